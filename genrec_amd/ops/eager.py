"""Eager (plain PyTorch) reference implementations of every fused op.

These are the CPU execution path AND the numerics reference that the CDNA4
HIP kernels are tested against (tests/test_kernels_*.py). Semantics mirror
the reference implementations exactly; citations per function.
"""

from __future__ import annotations

from typing import Optional

import torch
from torch import Tensor
import torch.nn.functional as F

NEG_BIG = -1e9  # reference's mask value (sasrec.py:220, transformer.py:144)


def rms_norm(x: Tensor, weight: Tensor, eps: float, t5_style: bool) -> Tensor:
    if t5_style:
        # ref normalize.py:73-95 (RootMeanSquareLayerNorm)
        var = x.float().pow(2).mean(-1, keepdim=True)
        y = x * torch.rsqrt(var + eps)
        if weight.dtype in (torch.float16, torch.bfloat16):
            y = y.to(weight.dtype)
        return weight * y
    # ref normalize.py:38-55 (RMSNorm)
    xf = x.float()
    y = (xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)).to(x.dtype)
    return y * weight


def l2norm(x: Tensor, eps: float = 1e-12) -> Tensor:
    # ref normalize.py:11-16 — F.normalize semantics: x / max(||x||, eps)
    return F.normalize(x, p=2, dim=-1, eps=eps)


def swish_layer_norm(x: Tensor, weight: Tensor, bias: Tensor, eps: float) -> Tensor:
    # ref normalize.py:58-70
    return F.silu(F.layer_norm(x, x.shape[-1:], weight, bias, eps))


def fused_attention(
    q: Tensor,  # [B, H, Lq, D]
    k: Tensor,  # [B, H, Lk, D]
    v: Tensor,  # [B, H, Lk, D]
    *,
    scale: float = 1.0,
    bias: Optional[Tensor] = None,          # [H,Lq,Lk] or [B,H,Lq,Lk], added pre-mask
    key_pad_mask: Optional[Tensor] = None,  # [B, Lk] bool, True = PAD
    additive_mask: Optional[Tensor] = None,  # [Lq,Lk] float additive (-inf style)
    causal: bool = False,
    query_mask: Optional[Tensor] = None,    # [B, Lq] float, applied AFTER softmax
    score_act: str = "softmax",             # "softmax" (std) or "silu" (HSTU)
    dropout_p: float = 0.0,
    training: bool = False,
) -> Tensor:
    Lq, Lk = q.size(-2), k.size(-2)
    scores = torch.matmul(q, k.transpose(-2, -1))
    if scale != 1.0:
        scores = scores * scale
    if bias is not None:
        scores = scores + (bias if bias.dim() == 4 else bias.unsqueeze(0))
    if causal:
        cm = torch.triu(
            torch.ones(Lq, Lk, device=q.device, dtype=torch.bool), diagonal=1
        )
        scores = scores.masked_fill(cm, NEG_BIG)
    if key_pad_mask is not None:
        scores = scores.masked_fill(key_pad_mask[:, None, None, :], NEG_BIG)
    if additive_mask is not None:
        am = additive_mask
        while am.dim() < 4:
            am = am.unsqueeze(0)
        scores = scores + am.to(scores.dtype)
    if score_act == "softmax":
        attn = torch.softmax(scores, dim=-1)
    elif score_act == "silu":
        attn = F.silu(scores)
    else:
        raise ValueError(score_act)
    if query_mask is not None:
        attn = attn * query_mask[:, None, :, None]
    if dropout_p > 0.0 and training:
        if attn.is_cuda:  # replay-safe on GPU (ATen dropout corrupts
            from genrec_amd.ops.fused import plain_dropout  # in replay)

            attn = plain_dropout(attn, dropout_p, True)
        else:
            attn = F.dropout(attn, p=dropout_p, training=True)
    return torch.matmul(attn, v)


def pairwise_sqdist(x: Tensor, codebook: Tensor) -> Tensor:
    """L2 distance matrix ||x||^2 + ||c||^2 - 2 x c^T (ref rqvae.py:186-192)."""
    return (
        (x ** 2).sum(dim=1, keepdim=True)
        + (codebook.T ** 2).sum(dim=0, keepdim=True)
        - 2 * x @ codebook.T
    )


def tied_softmax_ce(
    hidden: Tensor,      # [N, D] flattened
    emb_weight: Tensor,  # [V, D]
    targets: Tensor,     # [N]
    ignore_index: int = 0,
) -> Tensor:
    """logits = h @ E^T then mean-CE with ignore_index (ref sasrec.py:121-128)."""
    logits = hidden @ emb_weight.t()
    return F.cross_entropy(logits, targets, ignore_index=ignore_index)


def summed_ce(logits: Tensor, targets: Tensor) -> Tensor:
    """Per-sequence summed CE then batch mean (ref tiger.py:232-240)."""
    B = logits.size(0)
    loss = F.cross_entropy(
        logits.reshape(-1, logits.size(-1)), targets.reshape(-1), reduction="none"
    ).reshape(B, -1)
    return loss.sum(dim=1).mean()


def topk_hit_ranks(actual: Tensor, top_k: Tensor) -> Tensor:
    """Rank (0-based) of the first exact match of `actual` in `top_k`.

    Args:
        actual: [B, D] ground-truth id tuples
        top_k:  [B, K, D] ranked predictions
    Returns:
        [B] int64 ranks; K when no match. (ref metrics.py:26-66)
    """
    K = top_k.size(1)
    matches = (actual.unsqueeze(1) == top_k).all(dim=-1)  # [B, K]
    found = matches.any(dim=1)
    first = matches.float().argmax(dim=1)
    return torch.where(found, first, torch.full_like(first, K))
