"""Norm op dispatch: CDNA4 HIP kernels on GPU, eager on CPU.

HIP kernels live in csrc/kernels/norms.hip. Forward saves the per-row
inverse RMS / inverse L2 norm for the backward kernel.
"""

from __future__ import annotations

import torch
from torch import Tensor

from genrec_amd.ops import eager


def _hip_ok(x: Tensor) -> bool:
    import os

    from genrec_amd import ops

    if os.environ.get("GENREC_DISABLE_NORM", "0") == "1":
        return False
    return ops.use_hip(x)


class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: Tensor, weight: Tensor, eps: float, t5_style: bool):
        from genrec_amd import ops

        x2 = x.contiguous()
        y, inv_rms = ops.ext().rms_norm_fwd(x2, weight, eps, t5_style)
        ctx.save_for_backward(x2, weight, inv_rms)
        ctx.t5_style = t5_style
        return y

    @staticmethod
    def backward(ctx, dy: Tensor):
        from genrec_amd import ops

        x, weight, inv_rms = ctx.saved_tensors
        dx, dw = ops.ext().rms_norm_bwd(
            dy.contiguous(), x, weight, inv_rms, ctx.t5_style
        )
        return dx, dw, None, None


def rms_norm(x: Tensor, weight: Tensor, eps: float, t5_style: bool) -> Tensor:
    if _hip_ok(x):
        return _RMSNormFn.apply(x, weight, eps, t5_style)
    return eager.rms_norm(x, weight, eps, t5_style)


class _L2NormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: Tensor, eps: float):
        from genrec_amd import ops

        x2 = x.contiguous()
        y, inv_norm = ops.ext().l2norm_fwd(x2, eps)
        ctx.save_for_backward(x2, inv_norm)
        ctx.eps = eps
        return y

    @staticmethod
    def backward(ctx, dy: Tensor):
        from genrec_amd import ops

        x, inv_norm = ctx.saved_tensors
        dx = ops.ext().l2norm_bwd(dy.contiguous(), x, inv_norm, ctx.eps)
        return dx, None


def l2norm_op(x: Tensor, eps: float = 1e-12) -> Tensor:
    if _hip_ok(x):
        return _L2NormFn.apply(x, eps)
    return eager.l2norm(x, eps)


def swish_layer_norm(x: Tensor, weight: Tensor, bias: Tensor, eps: float) -> Tensor:
    # Fused silu(LayerNorm(x)). HIP kernel pending; composed form is used on
    # GPU meanwhile (LayerNorm + silu are both ATen/MIOpen kernels).
    return eager.swish_layer_norm(x, weight, bias, eps)
