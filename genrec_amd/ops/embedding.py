"""Embedding gather dispatch (K2/K15 — SURVEY.md §2.4).

On GPU this replaces F.embedding / nn.Embedding.forward with the
genrec_amd gather kernel whose backward is an fp32 atomic scatter-add —
graph-safe (ATen's ROCm sort-based embedding backward faults under
hipGraph replay) and faster at this zoo's index counts.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F
from torch import Tensor


class _EmbeddingFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, weight: Tensor, indices: Tensor, padding_idx: int):
        from genrec_amd import ops

        out = ops.ext().embedding_fwd(weight, indices)
        ctx.save_for_backward(indices)
        ctx.num_weights = weight.size(0)
        ctx.padding_idx = padding_idx
        ctx.w_dtype = weight.dtype
        return out

    @staticmethod
    def backward(ctx, dy: Tensor):
        from genrec_amd import ops

        (indices,) = ctx.saved_tensors
        dw = ops.ext().embedding_bwd(dy, indices, ctx.num_weights,
                                     ctx.padding_idx)
        if ctx.w_dtype != torch.float32:
            dw = dw.to(ctx.w_dtype)
        return dw, None, None


def embedding(weight: Tensor, indices: Tensor,
              padding_idx: Optional[int] = None) -> Tensor:
    from genrec_amd import ops

    if ops.use_hip(weight) and hasattr(ops.ext(), "embedding_fwd"):
        return _EmbeddingFn.apply(weight, indices,
                                  -1 if padding_idx is None else padding_idx)
    return F.embedding(indices, weight, padding_idx=padding_idx)
