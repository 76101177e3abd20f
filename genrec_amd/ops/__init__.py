"""genrec_amd.ops — single dispatch point for all hot-path compute.

Every op here has two implementations:

  * a hand-written CDNA4 HIP kernel (csrc/, compiled in-tree for gfx950)
    used whenever the input lives on a GPU;
  * a plain PyTorch eager implementation (ops/eager.py) used on CPU. The
    eager path is also the numerics reference that tests/ compare the HIP
    kernels against.

On a GPU box the HIP extension is REQUIRED: if a tensor is on CUDA/HIP and
the extension is missing, we raise instead of silently falling back — a
silent eager fallback would defeat the point of the framework and hide
build breakage.

Set GENREC_AMD_FORCE_EAGER=1 to force the eager path (debug only).
"""

from __future__ import annotations

import os

import torch

_EXT = None
_EXT_ERR: Exception | None = None


def _load_ext():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from genrec_amd import _C  # built in-tree by setup.py build_ext --inplace

        _EXT = _C
    except ImportError as e:  # pragma: no cover
        _EXT_ERR = e
        _EXT = None
    return _EXT


def ext():
    """Return the compiled extension module, raising if unavailable."""
    m = _load_ext()
    if m is None:
        raise RuntimeError(
            "genrec_amd HIP extension (genrec_amd._C) is not built. "
            "Run `python setup.py build_ext --inplace` (gfx950). "
            f"Import error: {_EXT_ERR}"
        )
    return m


def has_ext() -> bool:
    return _load_ext() is not None


def _force_eager() -> bool:
    return os.environ.get("GENREC_AMD_FORCE_EAGER", "0") == "1"


def use_hip(*tensors: torch.Tensor) -> bool:
    """HIP path iff all tensors are on GPU and eager isn't forced.

    On GPU, the extension is mandatory: ext() raises if it's missing, so a
    GPU run can never silently take the eager path.
    """
    if _force_eager():
        return False
    if not tensors or not tensors[0].is_cuda:
        return False
    ext()  # raises loudly if missing on a GPU box
    return True


from genrec_amd.ops import eager  # noqa: E402
from genrec_amd.ops.norms import (  # noqa: E402
    layer_norm, rms_norm, l2norm_op, swish_layer_norm,
)
from genrec_amd.ops.attention import (  # noqa: E402
    fused_attention,
    sasrec_attention,
    t5_attention,
    hstu_pointwise_attention,
    hstu_fused_attention,
)
from genrec_amd.ops.quantize import residual_quantize_step  # noqa: E402
from genrec_amd.ops.losses import softmax_ce, tied_softmax_ce, summed_ce  # noqa: E402
from genrec_amd.ops.metrics import topk_hit_ranks  # noqa: E402
from genrec_amd.ops.embedding import embedding  # noqa: E402
from genrec_amd.ops.fused import dropout_add, plain_dropout, relu_dropout  # noqa: E402

__all__ = [
    "ext",
    "has_ext",
    "use_hip",
    "eager",
    "rms_norm",
    "l2norm_op",
    "swish_layer_norm",
    "layer_norm",
    "fused_attention",
    "sasrec_attention",
    "t5_attention",
    "hstu_pointwise_attention",
    "hstu_fused_attention",
    "residual_quantize_step",
    "softmax_ce",
    "tied_softmax_ce",
    "summed_ce",
    "topk_hit_ranks",
    "embedding",
    "dropout_add",
    "plain_dropout",
    "relu_dropout",
]
