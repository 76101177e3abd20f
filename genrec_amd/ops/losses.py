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


def sampled_tied_softmax_ce(hidden: Tensor, emb_weight: Tensor,
                            targets: Tensor, num_negatives: int = 1024,
                            ignore_index: int = 0) -> Tensor:
    """Sampled-softmax tied-embedding CE (BASELINE north star).

    Instead of the full [N, V] logits (V up to ~60k for Amazon splits),
    draw `num_negatives` shared negative items uniformly, build the
    [N, 1 + M] candidate logits (target first) with a gather + GEMM, and
    apply the fused CE with the target at class 0. Rows whose target is
    `ignore_index` are masked out. Everything stays on device; the
    full-vocab logits tensor is never formed.

    This is a training-time approximation (uniform proposal, no
    correction term beyond excluding accidental hits by masking them to
    -1e9), matching common practice for retrieval towers.
    """
    n = hidden.size(0)
    device = hidden.device
    V = emb_weight.size(0)
    valid = targets != ignore_index
    if not bool(valid.any()):
        # all-padding batch: grad-connected zero instead of CE's 0/0 NaN
        return hidden.sum() * 0.0
    neg = torch.randint(1 if ignore_index == 0 else 0, V,
                        (num_negatives,), device=device)
    # logits: [N, 1+M] = [h . e_target, h @ E_neg^T]
    tgt_emb = emb_weight[targets.clamp(0, V - 1)]            # [N, D]
    pos_logit = (hidden * tgt_emb).sum(-1, keepdim=True)     # [N, 1]
    neg_logits = hidden @ emb_weight[neg].t()                # [N, M]
    # mask accidental hits (negative == target)
    hit = neg.unsqueeze(0) == targets.unsqueeze(1)           # [N, M]
    neg_logits = neg_logits.masked_fill(hit, -1e9)
    logits = torch.cat([pos_logit, neg_logits], dim=1)       # [N, 1+M]
    labels = torch.zeros(n, dtype=torch.long, device=device)
    labels = torch.where(valid, labels, torch.full_like(labels, -100))
    return softmax_ce(logits, labels, ignore_index=-100)
