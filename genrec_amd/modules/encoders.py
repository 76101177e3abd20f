"""Text encoders (parity: reference genrec/modules/encoder.py, 425 LoC).

LightT5Encoder: random-init transformer encoder + masked mean pooling +
projection + L2 norm (encoder.py:15-106) — fully offline-capable, used by
COBRA's dense branch and the RQ-VAE synthetic feature path.

SentenceT5Encoder / ErnieEncoder / BgeEncoder wrap pretrained checkpoints
via sentence-transformers / transformers; the offline environment has no
model hub, so these require a local directory and raise a clear error
otherwise (encoder.py:108-377).
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.nn.functional as F
from torch import Tensor, nn

from genrec_amd.ops.linear import SplitKLinear
from genrec_amd.modules.norms import FusedLayerNorm


class CaptureSafeEncoderLayer(nn.Module):
    """Post-LN transformer encoder layer, numerics-equivalent to torch's
    nn.TransformerEncoderLayer (relu activation), built on the genrec
    fused attention + dropout kernels: ATen native_dropout produces
    corrupt values on hipGraph REPLAY on ROCm 7 (single-element NaN
    gradients, localized with tools/repro_cobra_nan.py), and the
    nested-tensor fast path does host-side mask syncs."""

    def __init__(self, d_model: int, nhead: int, ff_dim: int,
                 dropout: float) -> None:
        super().__init__()
        assert d_model % nhead == 0
        self.h = nhead
        self.hd = d_model // nhead
        self.scale = 1.0 / (self.hd ** 0.5)
        # SplitKLinear: bias grads go through the replay-safe colsum
        # kernel (drop-in nn.Linear, same state-dict keys)
        self.qkv = SplitKLinear(d_model, 3 * d_model)
        self.out = SplitKLinear(d_model, d_model)
        self.linear1 = SplitKLinear(d_model, ff_dim)
        self.linear2 = SplitKLinear(ff_dim, d_model)
        self.norm1 = FusedLayerNorm(d_model)
        self.norm2 = FusedLayerNorm(d_model)
        self.dropout_p = dropout

    def forward(self, x: Tensor,
                src_key_padding_mask: Optional[Tensor] = None) -> Tensor:
        from genrec_amd import ops

        b, l, d = x.shape
        qkv = self.qkv(x).view(b, l, 3, self.h, self.hd) \
            .permute(2, 0, 3, 1, 4)
        # zero-copy: the attention kernels are stride-aware
        att = ops.fused_attention(
            qkv[0], qkv[1], qkv[2],
            scale=self.scale, key_pad_mask=src_key_padding_mask,
            dropout_p=self.dropout_p, training=self.training)
        att = self.out(att.transpose(1, 2).reshape(b, l, d))
        x = self.norm1(ops.dropout_add(att, x, self.dropout_p,
                                       self.training))
        ff = self.linear2(ops.relu_dropout(self.linear1(x), self.dropout_p,
                                           self.training))
        return self.norm2(ops.dropout_add(ff, x, self.dropout_p,
                                          self.training))

    def _load_from_state_dict(self, state_dict, prefix, *args, **kwargs):
        # accept nn.TransformerEncoderLayer checkpoints
        ren = {"self_attn.in_proj_weight": "qkv.weight",
               "self_attn.in_proj_bias": "qkv.bias",
               "self_attn.out_proj.weight": "out.weight",
               "self_attn.out_proj.bias": "out.bias"}
        for old, new in ren.items():
            if prefix + old in state_dict:
                state_dict[prefix + new] = state_dict.pop(prefix + old)
        super()._load_from_state_dict(state_dict, prefix, *args, **kwargs)


class _EncoderStack(nn.Module):
    def __init__(self, layers) -> None:
        super().__init__()
        self.layers = nn.ModuleList(layers)

    def forward(self, x: Tensor,
                src_key_padding_mask: Optional[Tensor] = None) -> Tensor:
        for layer in self.layers:
            x = layer(x, src_key_padding_mask)
        return x


class LightT5Encoder(nn.Module):
    def __init__(self, n_layers: int = 1, hidden_dim: int = 768,
                 output_dim: int = 768, num_heads: int = 8,
                 ff_dim: int = 2048, vocab_size: int = 32128,
                 max_seq_len: int = 512, dropout: float = 0.1) -> None:
        super().__init__()
        self.embedding = nn.Embedding(vocab_size, hidden_dim)
        self.pos_embedding = nn.Embedding(max_seq_len, hidden_dim)
        self.encoder = _EncoderStack([
            CaptureSafeEncoderLayer(hidden_dim, num_heads, ff_dim, dropout)
            for _ in range(n_layers)])
        self.proj = SplitKLinear(hidden_dim, output_dim)
        self.layer_norm = FusedLayerNorm(hidden_dim)

    def forward(self, batch_tokens: Tensor) -> Tensor:
        """batch_tokens: (B, T, L) or (B, L); 0 = pad. Returns L2-normalized
        (B, T, D) or (B, D)."""
        if batch_tokens.dim() == 3:
            b, t, l = batch_tokens.shape
            flat = batch_tokens.reshape(b * t, l)
        else:
            b, l = batch_tokens.shape
            t = 1
            flat = batch_tokens
        pos = torch.arange(l, device=flat.device).unsqueeze(0)
        # graph-safe embedding gathers (ATen rocprim bwd faults in replay)
        from genrec_amd import ops

        x = ops.embedding(self.embedding.weight, flat) \
            + ops.embedding(self.pos_embedding.weight, pos)
        pad = flat == 0
        hidden = self.layer_norm(self.encoder(x, src_key_padding_mask=pad))
        m = (~pad).unsqueeze(-1).to(hidden.dtype)  # keep bf16 under the
        # pure-bf16 graph runner (a .float() mask silently promoted the
        # whole pooled path to fp32)
        pooled = (hidden * m).sum(dim=1) / m.sum(dim=1).clamp(min=1e-9)
        out = F.normalize(self.proj(pooled), p=2, dim=-1)
        return out.view(b, t, -1) if t > 1 else out


def _require_local(path: str, what: str) -> None:
    if not (path and os.path.exists(path)):
        raise FileNotFoundError(
            f"{what} requires a local pretrained checkpoint directory "
            f"(offline environment, no model hub): got {path!r}")


class SentenceT5Encoder(nn.Module):
    """Dissected sentence-transformers T5 (tokenizer/encoder/pooling/dense),
    grad-checkpointed (ref encoder.py:108-199). Local checkpoint only."""

    def __init__(self, model_name: str, output_dim: int = 768) -> None:
        super().__init__()
        _require_local(model_name, "SentenceT5Encoder")
        from sentence_transformers import SentenceTransformer

        full = SentenceTransformer(model_name)
        self.tokenizer = full.tokenizer
        self.encoder_model = full._modules["0"].auto_model.encoder
        self.pooling = full._modules["1"]
        self.dense = full._modules.get("2")
        enc_dim = self.encoder_model.config.d_model
        self.proj = (None if self.dense is not None or enc_dim == output_dim
                     else nn.Linear(enc_dim, output_dim))
        self.output_dim = output_dim
        self.encoder_model.gradient_checkpointing_enable()

    def forward(self, batch_tokens: Tensor) -> Tensor:
        if batch_tokens.dim() == 3:
            b, t, l = batch_tokens.shape
            flat = batch_tokens.reshape(b * t, l)
        else:
            b, l = batch_tokens.shape
            t = 1
            flat = batch_tokens
        attn = (flat != 0).long()
        hidden = self.encoder_model(input_ids=flat,
                                    attention_mask=attn).last_hidden_state
        pooled = self.pooling({"token_embeddings": hidden,
                               "attention_mask": attn})["sentence_embedding"]
        if self.dense is not None:
            pooled = self.dense({"sentence_embedding": pooled})[
                "sentence_embedding"]
        elif self.proj is not None:
            pooled = self.proj(pooled)
        out = F.normalize(pooled, p=2, dim=-1)
        return out.view(b, t, -1) if t > 1 else out


class _CLSEncoder(nn.Module):
    """CLS-pooling encoder over a local HF checkpoint (Ernie/BGE family,
    ref encoder.py:202-377)."""

    def __init__(self, model_name: str, output_dim: Optional[int] = None,
                 normalize: bool = True) -> None:
        super().__init__()
        _require_local(model_name, type(self).__name__)
        from transformers import AutoModel, AutoTokenizer

        self.tokenizer = AutoTokenizer.from_pretrained(model_name)
        self.model = AutoModel.from_pretrained(model_name)
        hid = self.model.config.hidden_size
        self.proj = (nn.Linear(hid, output_dim)
                     if output_dim and output_dim != hid else None)
        self.normalize = normalize

    def forward(self, batch_tokens: Tensor) -> Tensor:
        if batch_tokens.dim() == 3:
            b, t, l = batch_tokens.shape
            flat = batch_tokens.reshape(b * t, l)
        else:
            b, l = batch_tokens.shape
            t = 1
            flat = batch_tokens
        attn = (flat != 0).long()
        out = self.model(input_ids=flat, attention_mask=attn)
        pooled = out.last_hidden_state[:, 0]  # CLS
        if self.proj is not None:
            pooled = self.proj(pooled)
        if self.normalize:
            pooled = F.normalize(pooled, p=2, dim=-1)
        return pooled.view(b, t, -1) if t > 1 else pooled


class ErnieEncoder(_CLSEncoder):
    pass


class BgeEncoder(_CLSEncoder):
    pass
