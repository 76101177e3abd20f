"""Sem-ID and user-ID embeddings (parity: reference embedding.py:7-74).

State-dict key names match the reference modules (`emb.weight`) so dict
checkpoints interoperate.
"""

from __future__ import annotations

from torch import Tensor, nn

from genrec_amd import ops


class SemIdEmbedding(nn.Module):
    """Flat table of C*V+1 rows indexed by token_type*V + id; last row is
    padding (ref embedding.py:7-43)."""

    def __init__(self, num_embeddings: int, sem_ids_dim: int,
                 embeddings_dim: int) -> None:
        super().__init__()
        self.num_embeddings = num_embeddings
        self.sem_ids_dim = sem_ids_dim
        self.padding_idx = sem_ids_dim * num_embeddings
        self.emb = nn.Embedding(
            num_embeddings * sem_ids_dim + 1, embeddings_dim,
            padding_idx=self.padding_idx,
        )

    def forward(self, input_ids: Tensor, token_type_ids: Tensor) -> Tensor:
        flat = token_type_ids * self.num_embeddings + input_ids
        return ops.embedding(self.emb.weight, flat, self.padding_idx)


class UserIdEmbedding(nn.Module):
    """User embedding with modulo hashing (ref embedding.py:46-74)."""

    def __init__(self, num_embeddings: int, embeddings_dim: int) -> None:
        super().__init__()
        self.num_embeddings = num_embeddings
        self.emb = nn.Embedding(num_embeddings, embeddings_dim)

    def forward(self, input_ids: Tensor) -> Tensor:
        return ops.embedding(self.emb.weight, input_ids % self.num_embeddings)
