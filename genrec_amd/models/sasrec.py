"""SASRec: self-attentive sequential recommendation.

Parity target: /root/reference/genrec/models/sasrec.py (266 LoC). The
load-bearing quirks of the official TF implementation are preserved exactly
(SURVEY.md §2.1):

  * sqrt(d)-scaled item embeddings, unscaled positional add (sasrec.py:103)
  * x * valid-mask after embedding and after EVERY block (sasrec.py:113-116)
  * key mask before softmax at -1e9, query mask AFTER softmax
    (sasrec.py:217-233)
  * residual added inside attention, on the normalized query (sasrec.py:244)
  * full-vocab tied logits x @ E^T and mean-CE ignore_index=0
    (sasrec.py:121-128)

MI355X design: the whole score->mask->softmax->query-mask->PV chain runs in
the fused CDNA4 attention kernel (ops.sasrec_attention); the tied-logits CE
runs as hipBLASLt GEMM + fused log-softmax-CE kernel (ops.tied_softmax_ce).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn.functional as F
from torch import Tensor, nn

from genrec_amd.modules.norms import FusedLayerNorm
from genrec_amd import ops
from genrec_amd.ops.linear import SplitKLinear
from genrec_amd.config import ginlite


class PointWiseFeedForward(nn.Module):
    """FFN with residual inside (ref sasrec.py:249-266)."""

    def __init__(self, embed_dim: int, ffn_dim: int, dropout: float) -> None:
        super().__init__()
        self.fc1 = SplitKLinear(embed_dim, ffn_dim)
        self.fc2 = SplitKLinear(ffn_dim, embed_dim)
        self.dropout = nn.Dropout(dropout)

    def forward(self, x: Tensor, residual: Tensor) -> Tensor:
        # genrec fused dropout kernels: replay-safe under hipGraph
        out = self.fc2(ops.relu_dropout(self.fc1(x), self.dropout.p,
                                        self.training))
        return ops.dropout_add(out, residual, self.dropout.p,
                               self.training)


class MultiHeadAttention(nn.Module):
    """SASRec attention with the official implementation's masking order."""

    def __init__(self, embed_dim: int, num_heads: int, dropout: float) -> None:
        super().__init__()
        assert embed_dim % num_heads == 0
        self.embed_dim = embed_dim
        self.num_heads = num_heads
        self.head_dim = embed_dim // num_heads
        self.scale = self.head_dim ** -0.5
        self.q_proj = SplitKLinear(embed_dim, embed_dim)
        self.k_proj = SplitKLinear(embed_dim, embed_dim)
        self.v_proj = SplitKLinear(embed_dim, embed_dim)
        self.dropout_p = dropout

    def forward(self, query: Tensor, key_value: Tensor, mask: Tensor) -> Tensor:
        """query: normalized x; key_value: raw x; mask: [B, L, 1] float."""
        b, l, _ = query.shape
        h, d = self.num_heads, self.head_dim

        def split(x: Tensor) -> Tensor:
            return x.view(b, l, h, d).transpose(1, 2)

        q = split(self.q_proj(query))
        k = split(self.k_proj(key_value))
        v = split(self.v_proj(key_value))
        valid = mask.squeeze(-1)  # [B, L] float 1/0
        out = ops.sasrec_attention(q, k, v, valid, self.scale,
                                   self.dropout_p, self.training)
        out = out.transpose(1, 2).reshape(b, l, self.embed_dim)
        return out + query  # residual on normalized query (ref sasrec.py:244)


class SASRecBlock(nn.Module):
    def __init__(self, embed_dim: int, num_heads: int, ffn_dim: int,
                 dropout: float) -> None:
        super().__init__()
        self.attention = MultiHeadAttention(embed_dim, num_heads, dropout)
        self.ffn = PointWiseFeedForward(embed_dim, ffn_dim, dropout)
        self.norm1 = FusedLayerNorm(embed_dim, eps=1e-8)
        self.norm2 = FusedLayerNorm(embed_dim, eps=1e-8)

    def forward(self, x: Tensor, mask: Tensor) -> Tensor:
        x = self.attention(self.norm1(x), x, mask)
        return self.ffn(self.norm2(x), x)


@ginlite.configurable(name="SASRec")
class SASRec(nn.Module):
    def __init__(self, num_items: int, max_seq_len: int = 50,
                 embed_dim: int = 64, num_heads: int = 2, num_blocks: int = 2,
                 ffn_dim: Optional[int] = None, dropout: float = 0.2,
                 loss_type: str = "full",
                 num_negatives: int = 1024) -> None:
        super().__init__()
        self.num_items = num_items
        self.loss_type = loss_type
        self.num_negatives = num_negatives
        self.max_seq_len = max_seq_len
        self.embed_dim = embed_dim
        ffn_dim = ffn_dim or embed_dim
        self.item_embedding = nn.Embedding(num_items + 1, embed_dim,
                                           padding_idx=0)
        self.position_embedding = nn.Embedding(max_seq_len, embed_dim)
        self.emb_dropout = nn.Dropout(dropout)
        self.blocks = nn.ModuleList([
            SASRecBlock(embed_dim, num_heads, ffn_dim, dropout)
            for _ in range(num_blocks)
        ])
        self.final_norm = FusedLayerNorm(embed_dim, eps=1e-8)
        self._init_weights()

    def _init_weights(self) -> None:
        for m in self.modules():
            if isinstance(m, (nn.Linear, nn.Embedding)):
                nn.init.xavier_uniform_(m.weight)
                if isinstance(m, nn.Linear) and m.bias is not None:
                    nn.init.zeros_(m.bias)
        with torch.no_grad():
            self.item_embedding.weight[0].zero_()

    def forward(self, input_ids: Tensor,
                targets: Optional[Tensor] = None
                ) -> Tuple[Tensor, Optional[Tensor]]:
        b, l = input_ids.shape
        mask = (input_ids != 0).unsqueeze(-1) \
            .to(self.item_embedding.weight.dtype)
        x = ops.embedding(self.item_embedding.weight, input_ids,
                          padding_idx=0) * (self.embed_dim ** 0.5)
        pos = torch.arange(l, device=input_ids.device).unsqueeze(0).expand(b, l)
        x = x + ops.embedding(self.position_embedding.weight, pos)
        x = ops.plain_dropout(x, self.emb_dropout.p, self.training) * mask
        for block in self.blocks:
            x = block(x, mask) * mask
        x = self.final_norm(x)

        loss = None
        if targets is not None:
            if self.loss_type == "sampled" and self.training:
                from genrec_amd.ops.losses import sampled_tied_softmax_ce

                loss = sampled_tied_softmax_ce(
                    x.reshape(-1, self.embed_dim),
                    self.item_embedding.weight, targets.reshape(-1),
                    num_negatives=self.num_negatives, ignore_index=0)
            else:
                loss = ops.tied_softmax_ce(
                    x.reshape(-1, self.embed_dim), self.item_embedding.weight,
                    targets.reshape(-1), ignore_index=0,
                )
            logits = None  # not materialized on the training path
            if not self.training:
                logits = x @ self.item_embedding.weight.t()
            return logits, loss
        logits = x @ self.item_embedding.weight.t()
        return logits, loss

    @torch.no_grad()
    def predict(self, input_ids: Tensor, top_k: int = 10) -> Tensor:
        logits, _ = self.forward(input_ids)
        last = logits[:, -1, :]
        last[:, 0] = float("-inf")
        return torch.topk(last, top_k, dim=-1).indices
