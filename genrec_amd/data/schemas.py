"""Dataset schemas (parity: reference genrec/data/schemas.py)."""

from __future__ import annotations

from typing import List, NamedTuple

from torch import Tensor

FUT_SUFFIX = "_fut"


class SeqData(NamedTuple):
    user_id: int
    item_ids: List[int]
    target_ids: List[int]


class SeqBatch(NamedTuple):
    user_ids: Tensor
    ids: Tensor
    ids_fut: Tensor
    x: Tensor
    x_fut: Tensor
    seq_mask: Tensor


class TokenizedSeqBatch(NamedTuple):
    user_ids: Tensor
    sem_ids: Tensor
    sem_ids_fut: Tensor
    seq_mask: Tensor
    token_type_ids: Tensor
    token_type_ids_fut: Tensor
