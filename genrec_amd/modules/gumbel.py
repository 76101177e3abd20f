"""Gumbel-softmax sampling (parity: reference gumbel.py)."""

from __future__ import annotations

import torch
import torch.nn.functional as F
from torch import Tensor


def sample_gumbel(shape, device, eps: float = 1e-20) -> Tensor:
    u = torch.rand(shape, device=device)
    return -torch.log(-torch.log(u + eps) + eps)


def gumbel_softmax_sample(logits: Tensor, temperature: float,
                          device: torch.device) -> Tensor:
    y = logits + sample_gumbel(logits.shape, device)
    return F.softmax(y / temperature, dim=-1)
