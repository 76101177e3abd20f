"""Recall@K / NDCG@K accumulation (parity: reference metrics.py:10-74).

MI355X-first redesign: counters stay resident on device as a single tensor
(the reference calls .item() per batch per K — a GPU sync each time,
metrics.py:55-66), and reduce() optionally all-reduces across ranks, fixing
the reference quirk where TIGER eval metrics are computed on rank 0's shard
only.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch
from torch import Tensor

from genrec_amd import ops


class TopKAccumulator:
    """Accumulates Recall@K and NDCG@K from exact-match ranks."""

    def __init__(self, ks: List[int] = [1, 5, 10]):
        self.ks = sorted(ks)
        self.reset()

    def reset(self) -> None:
        self.total = 0
        self._counters: Optional[Tensor] = None  # [2, len(ks)] recall/ndcg sums

    def _ensure(self, device: torch.device) -> Tensor:
        if self._counters is None:
            self._counters = torch.zeros(2, len(self.ks), dtype=torch.float64,
                                         device=device)
        return self._counters

    def accumulate(self, actual: Tensor, top_k: Tensor) -> None:
        """actual: [B, D] targets; top_k: [B, K, D] ranked predictions."""
        ranks = ops.topk_hit_ranks(actual, top_k)  # [B], K if miss
        c = self._ensure(ranks.device)
        dcg = 1.0 / torch.log2(ranks.float() + 2.0)
        for i, k in enumerate(self.ks):
            hit = ranks < k
            c[0, i] += hit.sum()
            c[1, i] += torch.where(hit, dcg, torch.zeros_like(dcg)).sum()
        self.total += actual.size(0)

    def reduce(self, all_reduce: bool = False) -> Dict[str, float]:
        dist_up = (all_reduce and torch.distributed.is_available()
                   and torch.distributed.is_initialized())
        if self._counters is None:
            if not dist_up:
                return {f"{m}@{k}": 0.0
                        for k in self.ks for m in ("Recall", "NDCG")}
            # this rank's eval shard was empty: still enter the collective
            # with zero counters, or the other ranks hang
            dev = (torch.device("cuda", torch.cuda.current_device())
                   if torch.cuda.is_available() else torch.device("cpu"))
            self._ensure(dev)
        c = self._counters.clone()
        total = torch.tensor([float(self.total)], dtype=torch.float64,
                             device=c.device)
        if all_reduce and torch.distributed.is_available() \
                and torch.distributed.is_initialized():
            torch.distributed.all_reduce(c)
            torch.distributed.all_reduce(total)
        c = c.cpu()
        t = max(total.item(), 1.0)
        out: Dict[str, float] = {}
        for i, k in enumerate(self.ks):
            out[f"Recall@{k}"] = c[0, i].item() / t
            out[f"NDCG@{k}"] = c[1, i].item() / t
        return out
