"""Normalization layers (parity: /root/reference/genrec/modules/normalize.py).

All forward semantics match the reference bit-for-bit in eager mode:
  * l2norm / L2Norm  — F.normalize p=2, eps=1e-12 (normalize.py:11-35)
  * RMSNorm          — fp32-upcast x, rsqrt(mean(x^2)+eps), cast back, *weight
                       (normalize.py:38-55)
  * SwishLayerNorm   — silu(layer_norm(x)) (normalize.py:58-70)
  * T5RMSNorm        — T5-style: variance in fp32, weight * x_normed, cast to
                       weight dtype when half (normalize.py:73-95; reference
                       name RootMeanSquareLayerNorm)

On GPU these dispatch to the fused CDNA4 HIP kernels in csrc/kernels/norms.hip
through genrec_amd.ops.
"""

from __future__ import annotations

import torch
from torch import Tensor, nn
import torch.nn.functional as F

from genrec_amd import ops


def l2norm(x: Tensor, dim: int = -1, eps: float = 1e-12) -> Tensor:
    if dim in (-1, x.dim() - 1):
        return ops.l2norm_op(x, eps)
    return F.normalize(x, p=2, dim=dim, eps=eps)


class L2Norm(nn.Module):
    def __init__(self, dim: int = -1, eps: float = 1e-12) -> None:
        super().__init__()
        self.dim = dim
        self.eps = eps

    def forward(self, x: Tensor) -> Tensor:
        return l2norm(x, dim=self.dim, eps=self.eps)


class RMSNorm(nn.Module):
    """RMS norm with fp32 upcast of the full normalization (ref normalize.py:38-55)."""

    def __init__(self, dim: int, eps: float = 1e-6) -> None:
        super().__init__()
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(dim))

    def forward(self, x: Tensor) -> Tensor:
        return ops.rms_norm(x, self.weight, self.eps, t5_style=False)


class T5RMSNorm(nn.Module):
    """T5-style RMS layer norm (ref RootMeanSquareLayerNorm, normalize.py:73-95).

    Variance computed in fp32, x scaled in original dtype, cast to the weight
    dtype when weight is half precision, then multiplied by weight.
    """

    def __init__(self, hidden_size: int, eps: float = 1e-6) -> None:
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden_size))
        self.variance_epsilon = eps

    def forward(self, x: Tensor) -> Tensor:
        return ops.rms_norm(x, self.weight, self.variance_epsilon, t5_style=True)


# Alias keeping the reference's class name importable for checkpoint parity.
RootMeanSquareLayerNorm = T5RMSNorm


class SwishLayerNorm(nn.Module):
    """silu(LayerNorm(x)) (ref normalize.py:58-70)."""

    def __init__(self, hidden_dim: int, eps: float = 1e-5) -> None:
        super().__init__()
        self.ln = nn.LayerNorm(hidden_dim, eps=eps)

    def forward(self, x: Tensor) -> Tensor:
        return ops.swish_layer_norm(x, self.ln.weight, self.ln.bias, self.ln.eps)


class FusedLayerNorm(nn.LayerNorm):
    """nn.LayerNorm whose forward dispatches to the genrec LayerNorm
    kernels on GPU (eager F.layer_norm elsewhere). State-dict compatible
    with nn.LayerNorm (same parameter names), normalizes the last dim."""

    def forward(self, x: Tensor) -> Tensor:
        return ops.layer_norm(x, self.weight, self.bias, self.eps)
