"""Fused elementwise ops: residual + dropout(x), dropout(relu(x)).

GPU: single HIP kernel each way (masks saved for exact backward). CPU /
no-extension: eager composition with identical semantics.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F
from torch import Tensor


def _seed_args(device, active: bool):
    from genrec_amd.ops.attention import _call_seed, _seed_counter

    if not active:
        return 0, None
    return _call_seed(), _seed_counter(device)


class _DropoutAddFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: Tensor, residual: Tensor, p: float):
        from genrec_amd import ops

        seed, seed_dev = _seed_args(x.device, True)
        out, mask = ops.ext().dropout_add_fwd(
            x.contiguous(), residual.contiguous(), p, seed, seed_dev)
        ctx.save_for_backward(mask)
        ctx.p = p
        return out

    @staticmethod
    def backward(ctx, dy: Tensor):
        from genrec_amd import ops

        (mask,) = ctx.saved_tensors
        dx = ops.ext().dropout_fuse_bwd(dy, mask, ctx.p, False)
        return dx, dy, None


class _ReluDropoutFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: Tensor, p: float):
        from genrec_amd import ops

        seed, seed_dev = _seed_args(x.device, True)
        out, mask = ops.ext().relu_dropout_fwd(x.contiguous(), p, seed,
                                               seed_dev)
        ctx.save_for_backward(mask)
        ctx.p = p
        return out

    @staticmethod
    def backward(ctx, dy: Tensor):
        from genrec_amd import ops

        (mask,) = ctx.saved_tensors
        dx = ops.ext().dropout_fuse_bwd(dy, mask, ctx.p, True)
        return dx, None


class _PlainDropoutFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: Tensor, p: float):
        from genrec_amd import ops

        seed, seed_dev = _seed_args(x.device, True)
        out, mask = ops.ext().plain_dropout_fwd(x.contiguous(), p, seed,
                                                seed_dev)
        ctx.save_for_backward(mask)
        ctx.p = p
        return out

    @staticmethod
    def backward(ctx, dy: Tensor):
        from genrec_amd import ops

        (mask,) = ctx.saved_tensors
        dx = ops.ext().dropout_fuse_bwd(dy, mask, ctx.p, False)
        return dx, None


def plain_dropout(x: Tensor, p: float, training: bool) -> Tensor:
    """dropout(x, p), hipGraph-replay-safe on GPU (ATen native_dropout
    corrupts on replay, ROCm 7 — see BACKLOG hazard ledger)."""
    from genrec_amd import ops

    if training and p > 0.0 and ops.use_hip(x) \
            and hasattr(ops.ext(), "plain_dropout_fwd"):
        return _PlainDropoutFn.apply(x, p)
    return F.dropout(x, p=p, training=training)


def dropout_add(x: Tensor, residual: Tensor, p: float,
                training: bool) -> Tensor:
    """residual + dropout(x, p) — fused on GPU."""
    from genrec_amd import ops

    if training and p > 0.0 and ops.use_hip(x) \
            and hasattr(ops.ext(), "dropout_add_fwd"):
        return _DropoutAddFn.apply(x, residual, p)
    return residual + F.dropout(x, p=p, training=training)


def relu_dropout(x: Tensor, p: float, training: bool) -> Tensor:
    """dropout(relu(x), p) — fused on GPU."""
    from genrec_amd import ops

    if training and p > 0.0 and ops.use_hip(x) \
            and hasattr(ops.ext(), "relu_dropout_fwd"):
        return _ReluDropoutFn.apply(x, p)
    return F.dropout(F.relu(x), p=p, training=training)
