"""Linear layer with a split-K grad-weight backward.

Training GEMMs in this framework are tall-and-skinny: activations are
[B*L, d] with B*L ~ 15k rows while d is a few hundred (TIGER encoder:
15616 x 384). The forward/input-grad GEMMs shard fine, but the
grad-weight GEMM dW = dY^T X has a tiny MxN output (384x384) and a huge
K — hipBLASLt's heuristic picks a non-split-K kernel whose grid (~150
workgroups) cannot fill 256 CUs, and each of the ~25 such GEMMs per
TIGER step costs 75-105us (measured, tools/bench_splitk.py). Splitting K
into chunks and running one batched GEMM + fp32 partial-sum reduce fills
the chip and is ~2.7x faster (87 -> 31us at 384x384, K=15616).

`SplitKLinear` is a drop-in `nn.Linear` (bias supported) whose backward
uses the chunked-bmm path whenever K is large and divisible; otherwise it
falls back to the plain GEMM. Forward is a stock hipBLASLt GEMM either
way. hipGraph-capture safe (static shapes, no host branches on data).

Reference parity: replaces the implicit nn.Linear grad GEMMs behind
transformer.py:72-124 and tiger.py:161-207 of the reference.
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.nn.functional as F
from torch import nn

# below ~8k rows the single hipBLASLt GEMM wins (the chunked path adds a
# bmm + reduce launch per backward; SASRec/HSTU at K=6400 measured slower)
_SPLITK_MIN_K = 8192


def _pick_chunks(k: int) -> int:
    for nc in (16, 8, 4):
        if k % nc == 0 and k // nc >= 256:
            return nc
    return 1


def _use_skinny(x: torch.Tensor, weight: torch.Tensor) -> bool:
    """Hand MFMA GEMM (T14-pipelined staging) for the tall-skinny fwd
    family. Measured SLOWER than hipBLASLt in-graph even on its best
    shapes (46.7 vs 47.4k bench with the narrow-shape-only gate; the
    eager microbench win was launch-floor artifact), so OPT-IN:
    GENREC_SKINNY_GEMM=1 enables narrow shapes, =all every eligible
    shape."""
    import os

    mode = os.environ.get("GENREC_SKINNY_GEMM", "0")
    if mode == "0":
        return False
    if not (x.is_cuda and x.dtype == torch.bfloat16
            and weight.dtype == torch.bfloat16):
        return False
    n, k = weight.shape
    rows = x.numel() // x.shape[-1]
    if not (k % 64 == 0 and rows >= 4096):
        return False
    if mode != "all" and not (n <= 768 and k <= 512):
        return False
    from genrec_amd import ops

    return ops.has_ext()


def _use_skinny_dx(dy: torch.Tensor, weight: torch.Tensor) -> bool:
    """Wide-to-narrow dX backward (dy [M, C] @ W [C, N<=384]) through the
    hand TN kernel. MEASURED SLOWER than hipBLASLt (111 vs 21 us at
    C=1152: the in-register 8x8 transpose in the staging write pass
    serializes the load pipeline — each shuffle forces a full vmcnt
    wait), so opt-IN only (GENREC_SKINNY_DX=1); kept for kernel-layout
    experiments."""
    import os

    if os.environ.get("GENREC_SKINNY_DX", "0") != "1":
        return False
    if not (dy.is_cuda and dy.dtype == torch.bfloat16
            and weight.dtype == torch.bfloat16):
        return False
    c, n = weight.shape
    rows = dy.numel() // dy.shape[-1]
    if not (rows >= 4096 and c >= 768 and n <= 384 and n % 8 == 0):
        return False
    from genrec_amd import ops

    return ops.has_ext()


class _SplitKLinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, weight: torch.Tensor,
                bias: Optional[torch.Tensor]):
        ctx.save_for_backward(x, weight)
        ctx.has_bias = bias is not None
        ctx.bias_dtype = bias.dtype if bias is not None else None
        if _use_skinny(x, weight):
            from genrec_amd import ops

            x2 = x.reshape(-1, x.shape[-1]).contiguous()
            out = ops.ext().skinny_gemm(
                x2, weight.contiguous(),
                bias if bias is not None else None)
            return out.view(*x.shape[:-1], weight.shape[0])
        if bias is not None:
            # fused addmm forward (same kernel F.linear uses); only the
            # BACKWARD bias reduce is replaced (replay-safe colsum)
            x2 = x.reshape(-1, x.shape[-1])
            return torch.addmm(bias, x2, weight.t()) \
                .view(*x.shape[:-1], weight.shape[0])
        return x.matmul(weight.t())

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        x, weight = ctx.saved_tensors
        # under autocast the saved weight/x may be fp32 while dy is bf16;
        # match dtypes explicitly (no-op casts on the pure-bf16 path)
        dx = None
        if ctx.needs_input_grad[0]:
            if _use_skinny_dx(dy, weight):
                from genrec_amd import ops

                dy2 = dy.reshape(-1, dy.shape[-1]).contiguous()
                dx = ops.ext().skinny_gemm_tn(
                    dy2, weight.contiguous(), None) \
                    .view(*dy.shape[:-1], weight.shape[1])
            else:
                dx = dy.matmul(weight.to(dy.dtype))
        dw = None
        if ctx.needs_input_grad[1]:
            x2 = x.reshape(-1, x.shape[-1]).to(dy.dtype)
            dy2 = dy.reshape(-1, dy.shape[-1])
            k = x2.shape[0]
            nc = _pick_chunks(k) if k >= _SPLITK_MIN_K else 1
            if nc > 1 and x2.is_contiguous() and dy2.is_contiguous():
                part = torch.bmm(
                    dy2.view(nc, k // nc, -1).transpose(1, 2),
                    x2.view(nc, k // nc, -1))
                # fp32-accumulated chunk reduce. GENREC_CHUNK_SUM=1 swaps
                # ATen's sum(0) for the genrec chunk_sum kernel — BITWISE
                # identical (both accumulate bf16 in fp32, same row
                # order) but measured perf-NEUTRAL on the TIGER bench
                # (49.0k vs 48.9k same-box): the ~6 us per-dispatch
                # average rocprof shows for ATen's reduce is launch-ramp
                # inflation, not stream time. Opt-in, same policy as the
                # skinny GEMM.
                from genrec_amd import ops

                n_el = part.shape[1] * part.shape[2]
                if part.is_cuda and ops.has_ext() and n_el % 4 == 0 \
                        and os.environ.get("GENREC_CHUNK_SUM") == "1":
                    dw = ops.ext().chunk_sum(part.view(nc, n_el)) \
                        .view(part.shape[1], part.shape[2]) \
                        .to(weight.dtype)
                else:
                    dw = part.sum(0).to(weight.dtype)
            else:
                dw = dy2.t().matmul(x2).to(weight.dtype)
        db = None
        if ctx.has_bias and ctx.needs_input_grad[2]:
            dy2b = dy.reshape(-1, dy.shape[-1])
            if dy2b.is_cuda:
                # deterministic replay-safe column sum: ATen's outer-dim
                # reduce corrupts bias grads under hipGraph replay at
                # these shapes (ROCm 7 — BACKLOG hazard ledger)
                from genrec_amd import ops

                if ops.has_ext():
                    db = ops.ext().colsum(dy2b.contiguous())
                else:
                    db = dy2b.sum(0)
            else:
                db = dy2b.sum(0)
            if db.dtype != ctx.bias_dtype:  # autocast: fp32 bias leaf
                db = db.to(ctx.bias_dtype)
        return dx, dw, db


class SplitKLinear(nn.Linear):
    """nn.Linear with split-K grad-weight GEMMs (see module docstring)."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # CUDA + bias: always the custom Function, so the bias gradient
        # takes the replay-safe colsum kernel (ATen's fused addmm
        # backward reduce corrupts under hipGraph replay). Otherwise the
        # custom Function only pays for itself when the split-K backward
        # will be used.
        if x.is_cuda and self.bias is not None \
                and os.environ.get("GENREC_UNSAFE_BIAS", "0") != "1":
            return _SplitKLinearFn.apply(x, self.weight, self.bias)
        if x.numel() // x.shape[-1] < _SPLITK_MIN_K:
            return F.linear(x, self.weight, self.bias)
        return _SplitKLinearFn.apply(x, self.weight, self.bias)
