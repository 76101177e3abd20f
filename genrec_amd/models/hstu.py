"""HSTU: Hierarchical Sequential Transduction Unit.

Parity target: /root/reference/genrec/models/hstu.py (409 LoC). Signature
semantics preserved:
  * fused Linear(D,4D)+SiLU projection split into U,V,Q,K (hstu.py:189,234-235)
  * scores = QK^T (unscaled) + T5-log-bucket position bias + log2-bucket
    temporal bias from timestamps (hstu.py:283-409)
  * SiLU on scores instead of softmax (hstu.py:261-263)
  * LayerNorm(attn) ⊙ U gating, residual, 4D-SiLU FFN (hstu.py:268-276)
  * tied item-embedding output head + mean-CE ignore_index=0 (hstu.py:137-146)

MI355X design: bias adds + causal/pad masks + SiLU + PV run in the fused
CDNA4 attention kernel with score_act=silu (no softmax running-max needed);
bias tables are gathered by stock embedding kernels.
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn.functional as F
from torch import Tensor, nn

from genrec_amd.modules.norms import FusedLayerNorm
from genrec_amd import ops
from genrec_amd.ops.linear import SplitKLinear
from genrec_amd.config import ginlite


class RelativePositionBias(nn.Module):
    """T5-style half-exact / half-log bucketed position bias
    (ref hstu.py:283-349)."""

    def __init__(self, num_buckets: int = 32, max_distance: int = 128,
                 num_heads: int = 2) -> None:
        super().__init__()
        self.num_buckets = num_buckets
        self.max_distance = max_distance
        self.num_heads = num_heads
        self.relative_attention_bias = nn.Embedding(num_buckets, num_heads)
        self._bucket_cache: dict = {}

    def _bucket(self, rel: Tensor) -> Tensor:
        rel = torch.clamp(rel, min=0)
        max_exact = self.num_buckets // 2
        is_small = rel < max_exact
        large = max_exact + (
            torch.log(rel.float() / max_exact)
            / math.log(self.max_distance / max_exact)
            * (self.num_buckets - max_exact)
        ).long()
        large = torch.clamp(large, max=self.num_buckets - 1)
        return torch.where(is_small, rel, large)

    def bucket_table(self, seq_len: int, device) -> Tensor:
        key = (seq_len, str(device))
        buckets = self._bucket_cache.get(key)
        if buckets is None:
            pos = torch.arange(seq_len, device=device)
            buckets = self._bucket(pos.unsqueeze(0) - pos.unsqueeze(1))
            self._bucket_cache[key] = buckets
        return buckets

    def forward(self, seq_len: int, device) -> Tensor:
        buckets = self.bucket_table(seq_len, device)
        bias = ops.embedding(self.relative_attention_bias.weight, buckets)
        return bias.permute(2, 0, 1)  # [H, L, L]


class TemporalBias(nn.Module):
    """ln2-bucketed |timestamp difference| bias (ref hstu.py:352-409)."""

    def __init__(self, num_buckets: int = 64, num_heads: int = 2) -> None:
        super().__init__()
        self.num_buckets = num_buckets
        self.temporal_attention_bias = nn.Embedding(num_buckets, num_heads)

    def forward(self, timestamps: Tensor) -> Tensor:
        diff = timestamps.unsqueeze(2) - timestamps.unsqueeze(1)  # [B, L, L]
        abs_diff = torch.clamp(diff.abs(), min=1).float()
        buckets = (torch.log(abs_diff) / 0.693).long().clamp(
            min=0, max=self.num_buckets - 1)
        bias = ops.embedding(self.temporal_attention_bias.weight, buckets)
        return bias.permute(0, 3, 1, 2)  # [B, H, L, L]


class HSTULayer(nn.Module):
    def __init__(self, embed_dim: int, num_heads: int, dropout: float,
                 num_position_buckets: int, num_time_buckets: int,
                 max_position_distance: int, use_temporal_bias: bool) -> None:
        super().__init__()
        self.embed_dim = embed_dim
        self.num_heads = num_heads
        self.head_dim = embed_dim // num_heads
        self.use_temporal_bias = use_temporal_bias
        self.projection = SplitKLinear(embed_dim, 4 * embed_dim)
        self.position_bias = RelativePositionBias(
            num_buckets=num_position_buckets,
            max_distance=max_position_distance, num_heads=num_heads)
        if use_temporal_bias:
            self.temporal_bias = TemporalBias(num_buckets=num_time_buckets,
                                              num_heads=num_heads)
        self.attn_norm = FusedLayerNorm(embed_dim)
        self.ffn = nn.Sequential(
            SplitKLinear(embed_dim, 4 * embed_dim), nn.SiLU(),
            nn.Dropout(dropout), SplitKLinear(4 * embed_dim, embed_dim),
            nn.Dropout(dropout),
        )
        self.ffn_norm = FusedLayerNorm(embed_dim)
        self.dropout = nn.Dropout(dropout)

    def forward(self, x: Tensor, padding_mask: Tensor,
                timestamps: Optional[Tensor]) -> Tensor:
        b, l, d = x.shape
        residual = x
        u, v, q, k = F.silu(self.projection(x)).chunk(4, dim=-1)

        def split(t: Tensor) -> Tensor:
            return t.view(b, l, self.num_heads, self.head_dim).transpose(1, 2)

        q, k, v = split(q), split(k), split(v)
        use_time = self.use_temporal_bias and timestamps is not None
        # fully-fused path: bias bucketing + gathers + SiLU attention in
        # one MFMA kernel (falls through to the bias-tensor composition)
        attn = ops.hstu_fused_attention(
            q, k, v, self.position_bias.bucket_table(l, x.device),
            self.position_bias.relative_attention_bias.weight,
            self.temporal_bias.temporal_attention_bias.weight
            if use_time else None,
            timestamps if use_time else None, padding_mask)
        if attn is None:
            pos_bias = self.position_bias(l, x.device)  # [H, L, L]
            time_bias = self.temporal_bias(timestamps) if use_time else None
            attn = ops.hstu_pointwise_attention(q, k, v, pos_bias, time_bias,
                                                padding_mask)
        attn = attn.transpose(1, 2).reshape(b, l, d)
        attn = self.attn_norm(attn) * u
        # replay-safe genrec dropout (hipGraph hazard ledger, BACKLOG)
        x = ops.dropout_add(attn, residual, self.dropout.p, self.training)
        h = self.ffn[0](self.ffn_norm(x))           # lin1
        h = ops.plain_dropout(F.silu(h), self.ffn[2].p, self.training)
        h = self.ffn[3](h)                          # lin2
        return ops.dropout_add(h, x, self.ffn[4].p, self.training)


@ginlite.configurable(name="HSTU")
class HSTU(nn.Module):
    def __init__(self, num_items: int, max_seq_len: int = 50,
                 embed_dim: int = 64, num_heads: int = 2, num_blocks: int = 2,
                 dropout: float = 0.2, num_position_buckets: int = 32,
                 num_time_buckets: int = 64, max_position_distance: int = 128,
                 use_temporal_bias: bool = True) -> None:
        super().__init__()
        self.num_items = num_items
        self.max_seq_len = max_seq_len
        self.embed_dim = embed_dim
        self.use_temporal_bias = use_temporal_bias
        self.item_embedding = nn.Embedding(num_items + 1, embed_dim,
                                           padding_idx=0)
        self.emb_dropout = nn.Dropout(dropout)
        self.layers = nn.ModuleList([
            HSTULayer(embed_dim=embed_dim, num_heads=num_heads,
                      dropout=dropout,
                      num_position_buckets=num_position_buckets,
                      num_time_buckets=num_time_buckets,
                      max_position_distance=max_position_distance,
                      use_temporal_bias=use_temporal_bias)
            for _ in range(num_blocks)
        ])
        self.final_norm = FusedLayerNorm(embed_dim)
        self._init_weights()

    def _init_weights(self) -> None:
        for m in self.modules():
            if isinstance(m, nn.Linear):
                nn.init.trunc_normal_(m.weight, std=0.02)
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Embedding):
                nn.init.trunc_normal_(m.weight, std=0.02)
                if m.padding_idx is not None:
                    m.weight.data[m.padding_idx].zero_()
            elif isinstance(m, nn.LayerNorm):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)

    def forward(self, input_ids: Tensor,
                timestamps: Optional[Tensor] = None,
                targets: Optional[Tensor] = None
                ) -> Tuple[Optional[Tensor], Optional[Tensor]]:
        padding_mask = input_ids == 0
        x = ops.plain_dropout(
            ops.embedding(self.item_embedding.weight, input_ids,
                          padding_idx=0),
            self.emb_dropout.p, self.training)
        for layer in self.layers:
            x = layer(x, padding_mask, timestamps)
        x = self.final_norm(x)

        loss = None
        if targets is not None:
            loss = ops.tied_softmax_ce(
                x.reshape(-1, self.embed_dim), self.item_embedding.weight,
                targets.reshape(-1), ignore_index=0,
            )
            logits = None
            if not self.training:
                logits = x @ self.item_embedding.weight.t()
            return logits, loss
        return x @ self.item_embedding.weight.t(), None

    @torch.no_grad()
    def predict(self, input_ids: Tensor, timestamps: Optional[Tensor] = None,
                top_k: int = 10) -> Tensor:
        logits, _ = self.forward(input_ids, timestamps)
        last = logits[:, -1, :]
        last[:, 0] = float("-inf")
        return torch.topk(last, top_k, dim=-1).indices
