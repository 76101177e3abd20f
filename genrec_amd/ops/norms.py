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


class _LayerNormFn(torch.autograd.Function):
    """nn.LayerNorm (elementwise affine) on the genrec kernel: ATen's
    backward pair (cuComputeGradInput + cuComputePartGradGammaBeta) was
    ~7% of the COBRA step at [2560, 384]-class shapes."""

    @staticmethod
    def forward(ctx, x: Tensor, weight: Tensor, bias: Tensor, eps: float):
        from genrec_amd import ops

        x2 = x.contiguous()
        y, mean, rstd = ops.ext().layer_norm_fwd(x2, weight, bias, eps)
        ctx.save_for_backward(x2, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy: Tensor):
        from genrec_amd import ops

        x, weight, mean, rstd = ctx.saved_tensors
        dx, dw, db = ops.ext().layer_norm_bwd(
            dy.contiguous(), x, weight, mean, rstd)
        return dx, dw, db, None


def layer_norm(x: Tensor, weight: Tensor, bias: Tensor,
               eps: float = 1e-5) -> Tensor:
    """Drop-in F.layer_norm over the last dim (affine required)."""
    if (_hip_ok(x) and weight is not None and bias is not None
            and x.dtype == weight.dtype and x.dtype == bias.dtype
            and x.dtype in (torch.bfloat16, torch.float32)
            and x.shape[-1] <= 1024):
        return _LayerNormFn.apply(x, weight, bias, eps)
    return torch.nn.functional.layer_norm(
        x, (x.shape[-1],), weight, bias, eps)
