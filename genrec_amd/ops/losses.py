"""Loss ops (K3 tied-embedding CE, K16 summed CE — SURVEY.md §2.4).

The tied-logits CE is the SASRec/HSTU output path: logits = h @ E^T over the
full item vocab followed by mean-CE with ignore_index=0 (sasrec.py:121-128,
hstu.py:137-146). On GPU the logits GEMM runs on hipBLASLt (torch.matmul)
and the log-softmax+CE+backward runs in one fused HIP kernel so the [N,V]
probability tensor is never materialized twice.
"""

from __future__ import annotations

import torch
from torch import Tensor

from genrec_amd.ops import eager


class _SoftmaxCEFn(torch.autograd.Function):
    """Fused log-softmax + NLL over precomputed logits, ignore_index aware.

    Saves only (logits, targets, per-row lse); backward recomputes
    softmax from lse in one kernel pass: dlogits = (softmax - onehot) / n_valid.
    """

    @staticmethod
    def forward(ctx, logits: Tensor, targets: Tensor, ignore_index: int):
        from genrec_amd import ops

        loss, lse, n_valid = ops.ext().softmax_ce_fwd(
            logits.contiguous(), targets.contiguous(), ignore_index
        )
        ctx.save_for_backward(logits, targets, lse, n_valid)
        ctx.ignore_index = ignore_index
        return loss

    @staticmethod
    def backward(ctx, dloss: Tensor):
        from genrec_amd import ops

        logits, targets, lse, n_valid = ctx.saved_tensors
        dlogits = ops.ext().softmax_ce_bwd(
            dloss.contiguous(), logits, targets, lse, n_valid, ctx.ignore_index
        )
        return dlogits, None, None


def softmax_ce(logits: Tensor, targets: Tensor, ignore_index: int = -100) -> Tensor:
    import os

    from genrec_amd import ops

    if os.environ.get("GENREC_DISABLE_CE", "0") != "1" \
            and ops.use_hip(logits) and hasattr(ops.ext(), "softmax_ce_fwd"):
        return _SoftmaxCEFn.apply(logits, targets, ignore_index)
    return torch.nn.functional.cross_entropy(
        logits, targets, ignore_index=ignore_index
    )


def tied_softmax_ce(hidden: Tensor, emb_weight: Tensor, targets: Tensor,
                    ignore_index: int = 0) -> Tensor:
    logits = hidden @ emb_weight.t()
    return softmax_ce(logits, targets, ignore_index)


def summed_ce(logits: Tensor, targets: Tensor) -> Tensor:
    """TIGER loss: CE(reduction none) summed over sequence, mean over batch
    (tiger.py:232-240)."""
    from genrec_amd import ops

    import os

    B, T, V = logits.shape
    if os.environ.get("GENREC_DISABLE_CE", "0") != "1" \
            and ops.use_hip(logits) and hasattr(ops.ext(), "softmax_ce_fwd"):
        # fused per-token CE (no ignore) * T gives sum-then-mean semantics:
        # mean over B*T tokens * T == sum over T, mean over B.
        flat_loss = _SoftmaxCEFn.apply(
            logits.reshape(-1, V), targets.reshape(-1), -100
        )
        return flat_loss * T
    return eager.summed_ce(logits, targets)
