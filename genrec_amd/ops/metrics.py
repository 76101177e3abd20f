"""On-device metric ops (K22 — SURVEY.md §2.4).

topk_hit_ranks: rank of first exact match of a target id-tuple within the
[B,K,D] beam output; used by TopKAccumulator which keeps Recall/NDCG
counters resident on device (the reference round-trips to Python per batch,
metrics.py:68-74 — we reduce once per epoch instead).
"""

from __future__ import annotations

from torch import Tensor

from genrec_amd.ops import eager


def topk_hit_ranks(actual: Tensor, top_k: Tensor) -> Tensor:
    from genrec_amd import ops

    if ops.use_hip(actual, top_k) and hasattr(ops.ext(), "topk_hit_ranks"):
        return ops.ext().topk_hit_ranks(actual.contiguous(), top_k.contiguous())
    return eager.topk_hit_ranks(actual, top_k)
