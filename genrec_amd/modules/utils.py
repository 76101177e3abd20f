"""Trainer utilities (parity: reference genrec/modules/utils.py).

parse_config lives in genrec_amd.config.ginlite (the CLI wrapper with
{split} substitution and --gin overrides, utils.py:85-117); this module
carries the decorators (utils.py:12-60) and the seq-length debug metrics
(utils.py:120-137).
"""

from __future__ import annotations

import functools
import torch

from genrec_amd.config.ginlite import parse_config  # re-export  # noqa: F401


def eval_mode(fn):
    """Run fn with the module in eval(), restoring the previous mode."""

    @functools.wraps(fn)
    def inner(self, *args, **kwargs):
        was_training = self.training
        self.eval()
        out = fn(self, *args, **kwargs)
        self.train(was_training)
        return out

    return inner


def reset_kv_cache(fn):
    """Clear decoder KV caches around fn (generation entry points)."""

    @functools.wraps(fn)
    def inner(self, *args, **kwargs):
        if hasattr(self, "reset_kv_cache"):
            self.reset_kv_cache()
        out = fn(self, *args, **kwargs)
        if hasattr(self, "reset_kv_cache"):
            self.reset_kv_cache()
        return out

    return inner


def reset_encoder_cache(fn):
    """Clear cached encoder output around fn."""

    @functools.wraps(fn)
    def inner(self, *args, **kwargs):
        if getattr(self, "cached_enc_output", None) is not None:
            self.cached_enc_output = None
        out = fn(self, *args, **kwargs)
        if getattr(self, "cached_enc_output", None) is not None:
            self.cached_enc_output = None
        return out

    return inner


@torch.no_grad()
def compute_debug_metrics(seq_mask: torch.Tensor, model_output=None,
                          prefix: str = "") -> dict:
    """Sequence-length quantiles (+ per-position losses when available)."""
    seq_lengths = seq_mask.sum(dim=1).to(torch.float32)
    prefix = prefix + "_" if prefix else ""
    out = {
        f"{prefix}seq_length_p{q}": torch.quantile(seq_lengths, q=q)
        .detach().cpu().item()
        for q in [0.25, 0.5, 0.75, 0.9, 1]
    }
    if model_output is not None and hasattr(model_output, "loss_d"):
        for d, v in enumerate(model_output.loss_d):
            out[f"{prefix}loss_{d}"] = float(v)
    return out
