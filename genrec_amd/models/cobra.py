"""COBRA: cascaded sparse-dense generative recommendation (arXiv:2503.02453).

Parity target: /root/reference/genrec/models/cobra.py (804 LoC). Reproduced
behavior:
  * CobraEmbedding: per-item interleaved [e^1..e^C, v] layout with
    codebook-offset IDs, token-type 0/1 (sparse/dense), absolute positions,
    partial-item support for generation (cobra.py:47-147)
  * causal TransformerDecoder over the interleaved sequence (cobra.py:150-224)
  * per-codebook sparse heads with the position-shifted CE: codebook 0 is
    predicted from the previous item's DENSE slot, codebook c>0 from the
    same item's previous sparse slot (cobra.py:417-457)
  * in-batch InfoNCE on L2-normalized dense predictions with same-sequence
    masking at -1e4 (cobra.py:466-495); the cross-batch MoCo queue exists
    but is disabled, as in the reference (cobra.py:283-321, 497-508)
  * codebook-entropy / accuracy metrics (cobra.py:459-517)
  * beam generate over codebooks re-running the decoder per level with
    partial-item embedding, dense vec taken from the last codebook slot
    (cobra.py:531-665)
  * beam_fusion: alpha-blend of softmaxed beam scores and max item
    similarity (cobra.py:679-760)

MI355X notes: interleaving is ONE indexed gather (K18); the decoder is
native CobraDecoderLayers on ops.fused_attention (flash-tiled MFMA
kernels at the interleaved length ~T*(C+1) <= 80, head_dim 64); the
sparse-head CE and dense InfoNCE run on the fused CE kernel (K3); the
InfoNCE sim GEMM and heads run on hipBLASLt.
"""

from __future__ import annotations

from typing import NamedTuple, Optional

import torch
import torch.nn.functional as F
from torch import Tensor, nn

from genrec_amd import ops
from genrec_amd.ops.linear import SplitKLinear

from genrec_amd.config import ginlite
from genrec_amd.modules.encoders import LightT5Encoder
from genrec_amd.modules.norms import FusedLayerNorm


class CobraOutput(NamedTuple):
    loss: Tensor
    loss_sparse: Tensor
    loss_dense: Tensor
    acc_correct: Tensor
    acc_total: Tensor
    recall_correct: Tensor
    recall_total: Tensor
    vec_cos_sim: Tensor
    codebook_entropy: Tensor


class CobraGenerationOutput(NamedTuple):
    sem_ids: Tensor      # (B, K, C)
    dense_vecs: Tensor   # (B, K, D)
    scores: Tensor       # (B, K)


class BeamFusionOutput(NamedTuple):
    item_ids: Tensor     # (B, K)
    sem_ids: Tensor      # (B, K, C)
    scores: Tensor       # (B, K)


class CobraEmbedding(nn.Module):
    """Interleaved sparse/dense item embedding (ref cobra.py:47-147)."""

    def __init__(self, id_vocab_size: int, n_codebooks: int = 3,
                 d_model: int = 768, max_len: int = 1024,
                 pad_id: Optional[int] = None) -> None:
        super().__init__()
        self.C = n_codebooks
        self.id_vocab_size = id_vocab_size
        self.pad_id = pad_id if pad_id is not None \
            else id_vocab_size * n_codebooks
        self.id_embed = nn.Embedding(
            id_vocab_size * n_codebooks + 1, d_model,
            padding_idx=id_vocab_size * n_codebooks)
        self.type_embed = nn.Embedding(2, d_model)
        self.pos_embed = nn.Embedding(max_len, d_model)

    def forward(self, input_ids: Tensor, input_vecs: Tensor, mask: Tensor,
                n_complete_items: Optional[int] = None) -> Tensor:
        b, l = input_ids.shape
        device = input_ids.device
        t_vecs = input_vecs.size(1)
        if n_complete_items is None:
            n_complete_items = l // self.C

        # codebook-offset gather (graph-safe genrec embedding op: ATen's
        # rocprim embedding backward faults under hipGraph replay)
        ttype = (torch.arange(l, device=device) % self.C).unsqueeze(0)
        valid = input_ids != self.pad_id
        flat_ids = torch.where(
            valid, input_ids + ttype * self.id_vocab_size, input_ids)
        sparse_emb = ops.embedding(self.id_embed.weight, flat_ids,
                                   self.id_embed.padding_idx)

        # K18 interleave as ONE indexed gather from [sparse_emb | vecs]
        # (the reference's chunk/cat loop is ~2T kernel launches,
        # cobra.py:323-377; a gather is bandwidth-optimal on HBM3E).
        # Sparse token j (chunk i=j//C) shifts right by the number of
        # dense slots inserted before it; dense slot i lands at
        # i*(C+1)+C.
        n_ct = n_complete_items * self.C
        n_ins = min(n_complete_items, t_vecs)
        out_len = l + n_ins
        orig = torch.arange(l, device=device)
        # clamp/full_like, NOT torch.tensor(scalar, device=...): an H2D
        # scalar upload is illegal inside hipGraph capture
        shift = torch.where(orig < n_ct,
                            (orig // self.C).clamp(max=n_ins),
                            torch.full_like(orig, n_ins))
        new_pos = orig + shift
        src_idx = torch.empty(out_len, dtype=torch.long, device=device)
        src_idx.scatter_(0, new_pos, orig)
        is_dense = torch.zeros(out_len, dtype=torch.bool, device=device)
        if n_ins > 0:
            ins_pos = torch.arange(n_ins, device=device) * (self.C + 1) \
                + self.C
            src_idx.scatter_(0, ins_pos,
                             torch.arange(n_ins, device=device) + l)
            is_dense.scatter_(0, ins_pos, True)
        src = torch.cat([sparse_emb, input_vecs[:, :n_ins]], dim=1)
        h = src.gather(1, src_idx.view(1, -1, 1).expand(b, -1, src.size(-1)))

        pos = torch.arange(out_len, device=device).unsqueeze(0).expand(b, -1)
        type_idx = is_dense.long().unsqueeze(0).expand(b, -1)
        m = mask.unsqueeze(-1).to(h.dtype)  # dtype-preserving (bf16 path)
        h = h * m
        h = h + ops.embedding(self.pos_embed.weight, pos) * m
        h = h + ops.embedding(self.type_embed.weight, type_idx) * m
        return h


class CobraDecoderLayer(nn.Module):
    """Native causal self-attention layer, numerics-equivalent to torch's
    post-LN ``nn.TransformerDecoderLayer`` driven with a zero-length
    memory (as the reference does, cobra.py:150-224): the cross-attn
    sublayer there contributes exactly zero, but its LayerNorm still
    applies, so the effective layer is
        x = norm1(x + SA(x)); x = norm2(x); x = norm3(x + FF(x)).
    Attention runs on ops.fused_attention — the flash-tiled MFMA kernels
    at the interleaved COBRA length L=(C+1)*T (~80), head_dim 64."""

    def __init__(self, d_model: int, n_heads: int, ff_dim: int,
                 dropout: float) -> None:
        super().__init__()
        assert d_model % n_heads == 0
        self.h = n_heads
        self.hd = d_model // n_heads
        self.scale = 1.0 / (self.hd ** 0.5)
        self.qkv = SplitKLinear(d_model, 3 * d_model, bias=True)
        self.out = SplitKLinear(d_model, d_model, bias=True)
        self.linear1 = SplitKLinear(d_model, ff_dim, bias=True)
        self.linear2 = SplitKLinear(ff_dim, d_model, bias=True)
        self.norm1 = FusedLayerNorm(d_model)
        self.norm2 = FusedLayerNorm(d_model)
        self.norm3 = FusedLayerNorm(d_model)
        self.dropout_p = dropout

    def forward(self, x: Tensor, key_pad_mask: Optional[Tensor],
                query_mask: Optional[Tensor]) -> Tensor:
        from genrec_amd import ops

        b, l, d = x.shape
        qkv = self.qkv(x).view(b, l, 3, self.h, self.hd) \
            .permute(2, 0, 3, 1, 4)
        # zero-copy: the flash/MFMA kernels are stride-aware, so the
        # permuted qkv views pass straight through (round-2 late sweep)
        att = ops.fused_attention(
            qkv[0], qkv[1], qkv[2],
            scale=self.scale, causal=True, key_pad_mask=key_pad_mask,
            query_mask=query_mask, dropout_p=self.dropout_p,
            training=self.training)
        att = self.out(att.transpose(1, 2).reshape(b, l, d))
        # genrec fused dropout kernels (device seed counter): ATen
        # native_dropout produces corrupt values on hipGraph REPLAY on
        # ROCm 7 (single-element NaN grads — tools/repro_cobra_nan.py)
        x = self.norm1(ops.dropout_add(att, x, self.dropout_p,
                                       self.training))
        x = self.norm2(x)  # vestigial cross-attn sublayer's norm
        ff = self.linear2(ops.relu_dropout(self.linear1(x), self.dropout_p,
                                           self.training))
        return self.norm3(ops.dropout_add(ff, x, self.dropout_p,
                                          self.training))


class CobraDecoder(nn.Module):
    """Causal transformer decoder (no cross-attn memory, ref cobra.py:150-224).

    Round 2: native layers on the fused/flash attention kernels replace
    the round-1 stock nn.TransformerDecoder (numerics-equivalence covered
    by tests/test_cobra_notellm.py::test_cobra_decoder_matches_torch)."""

    def __init__(self, hidden_dim: int = 768, n_layers: int = 6,
                 n_heads: int = 12, ff_dim: int = 2048,
                 dropout: float = 0.1) -> None:
        super().__init__()
        self.layers = nn.ModuleList([
            CobraDecoderLayer(hidden_dim, n_heads, ff_dim, dropout)
            for _ in range(n_layers)])

    def forward(self, tgt: Tensor, memory: Optional[Tensor] = None,
                tgt_key_padding_mask: Optional[Tensor] = None,
                memory_key_padding_mask: Optional[Tensor] = None) -> Tensor:
        # query_mask zeroes fully-padded query rows (torch emits NaN
        # there; those positions are never consumed — losses use
        # ignore_index and generation gathers valid last positions)
        qm = None
        if tgt_key_padding_mask is not None:
            qm = (~tgt_key_padding_mask).to(tgt.dtype)
        x = tgt
        for layer in self.layers:
            x = layer(x, tgt_key_padding_mask, qm)
        return x


@ginlite.configurable(name="Cobra")
class Cobra(nn.Module):
    def __init__(self, encoder_n_layers: int = 1,
                 encoder_hidden_dim: int = 768, encoder_num_heads: int = 8,
                 encoder_vocab_size: int = 32128, id_vocab_size: int = 512,
                 n_codebooks: int = 3, d_model: int = 768,
                 max_len: int = 1024, temperature: float = 0.2,
                 queue_size: int = 1024, decoder_n_layers: int = 8,
                 decoder_num_heads: int = 6, decoder_dropout: float = 0.1,
                 encoder_type: str = "light",
                 encoder_model_name: Optional[str] = None) -> None:
        super().__init__()
        self.C = n_codebooks
        self.d_model = d_model
        self.pad_id = id_vocab_size * n_codebooks
        if encoder_type == "pretrained":
            from genrec_amd.modules.encoders import SentenceT5Encoder

            self.encoder = SentenceT5Encoder(encoder_model_name,
                                             output_dim=d_model)
        else:
            self.encoder = LightT5Encoder(
                n_layers=encoder_n_layers, hidden_dim=encoder_hidden_dim,
                output_dim=d_model, num_heads=encoder_num_heads,
                vocab_size=encoder_vocab_size)
        self.cobra_emb = CobraEmbedding(
            id_vocab_size=id_vocab_size, n_codebooks=n_codebooks,
            d_model=d_model, max_len=max_len, pad_id=self.pad_id)
        self.decoder = CobraDecoder(d_model, n_layers=decoder_n_layers,
                                    n_heads=decoder_num_heads,
                                    dropout=decoder_dropout)
        self.sparse_head = nn.ModuleList([
            SplitKLinear(d_model, id_vocab_size) for _ in range(n_codebooks)])
        self.temperature = temperature
        # MoCo queue: present for parity but unused (ref cobra.py:497-508)
        self.register_buffer("feat_queue",
                             F.normalize(torch.randn(queue_size, d_model),
                                         dim=-1))
        self.register_buffer("queue_ptr", torch.zeros(1, dtype=torch.long))
        self.queue_size = queue_size

    @torch.no_grad()
    def _dequeue_and_enqueue(self, new_feats: Tensor) -> None:
        n, k = new_feats.size(0), self.queue_size
        ptr = int(self.queue_ptr)
        if n >= k:
            self.feat_queue.copy_(new_feats[-k:])
            self.queue_ptr[0] = 0
            return
        end = ptr + n
        if end <= k:
            self.feat_queue[ptr:end] = new_feats
        else:
            first = k - ptr
            self.feat_queue[ptr:] = new_feats[:first]
            self.feat_queue[:end - k] = new_feats[first:]
        self.queue_ptr[0] = end % k

    def interleave_seq_mask(self, seq_mask: Tensor, n: int,
                            n_complete_items: Optional[int] = None) -> Tensor:
        """Insert a dense slot after every n sparse positions
        (ref cobra.py:323-377)."""
        b, l = seq_mask.shape
        device = seq_mask.device
        if n_complete_items is None:
            n_complete_items = l // n
        orig = torch.arange(l, device=device)
        complete = orig < n_complete_items * n
        new_pos = torch.where(complete, orig + orig // n,
                              orig + n_complete_items)
        out = seq_mask.new_zeros(b, l + n_complete_items)
        out.scatter_(1, new_pos.expand(b, -1), seq_mask)
        if n_complete_items > 0:
            g = torch.arange(n_complete_items, device=device)
            ins_pos = g * (n + 1) + n
            prev = (g * n + (n - 1)).clamp(max=l - 1)
            out.scatter_(1, ins_pos.expand(b, -1), seq_mask[:, prev])
        return out

    # ------------------------------------------------------------ training

    def forward(self, input_ids: Tensor, encoder_input_ids: Tensor,
                mask=None) -> CobraOutput:
        vecs = self.encoder(encoder_input_ids)          # (B, T, D)
        b, tc = input_ids.shape
        seq_mask = self.interleave_seq_mask(input_ids != self.pad_id, self.C)
        emb = self.cobra_emb(input_ids, vecs, seq_mask)
        h = self.decoder(emb, tgt_key_padding_mask=~seq_mask)
        t = tc // self.C
        n_pos = t - 1

        loss_sparse = 0.0
        total_correct = torch.zeros((), device=h.device)
        total_top5 = torch.zeros((), device=h.device)
        total_tokens = torch.zeros((), device=h.device)
        all_item_correct = torch.ones(b, n_pos, dtype=torch.bool,
                                      device=h.device)
        all_valid = None
        for c in range(self.C):
            if c == 0:
                pos_c = torch.arange(0, t - 1, device=h.device) * (self.C + 1) \
                    + self.C
                target_pos = torch.arange(1, t, device=h.device) * self.C
            else:
                pos_c = torch.arange(1, t, device=h.device) * (self.C + 1) \
                    + (c - 1)
                target_pos = torch.arange(1, t, device=h.device) * self.C + c
            logits = self.sparse_head[c](h[:, pos_c, :])
            target = input_ids[:, target_pos]
            # fused CE (K3 kernel): mean over valid == the reference's
            # sum/n_valid (cobra.py:417-457)
            loss_c = ops.softmax_ce(
                logits.reshape(-1, logits.size(-1)).contiguous(),
                target.reshape(-1), ignore_index=self.pad_id)
            loss_sparse = loss_sparse + loss_c
            with torch.no_grad():
                valid = target != self.pad_id
                if all_valid is None:
                    all_valid = valid
                top1 = logits.argmax(-1)
                total_correct += ((top1 == target) & valid).sum()
                total_top5 += ((logits.topk(5, -1).indices
                                == target.unsqueeze(-1)).any(-1) & valid).sum()
                total_tokens += valid.sum()
                all_item_correct &= (top1 == target) | ~valid
        loss_sparse = loss_sparse / self.C

        item_correct = all_item_correct & all_valid
        recall_correct = item_correct.sum()
        recall_total = all_valid.sum()

        # dense InfoNCE (ref cobra.py:466-495)
        vec_pos = torch.arange(1, t, device=h.device) * (self.C + 1) \
            + (self.C - 1)
        vec_pred = h[:, vec_pos, :self.d_model]
        vec_gt = vecs[:, 1:, :].detach()
        q = b * n_pos
        valid_d = seq_mask[:, (self.C + 1)::(self.C + 1)].reshape(-1)
        if getattr(self, "static_infonce", False):
            # hipGraph-capturable variant: no boolean-mask gather (its
            # output shape is data-dependent and would be baked into the
            # capture). All q rows stay; invalid keys are masked to -1e4
            # (exp(-1e4/T) ~ 0 == excluded) and invalid rows contribute
            # zero to the masked mean. Numerically equivalent to the
            # filtered reference formulation.
            vp = F.normalize(vec_pred.reshape(q, -1), p=2, dim=-1,
                             eps=1e-12)
            vg = F.normalize(vec_gt.reshape(q, -1), p=2, dim=-1, eps=1e-12)
            seq_ids = torch.arange(b, device=h.device).unsqueeze(1) \
                .expand(-1, n_pos).reshape(-1)
            same = seq_ids.unsqueeze(0) == seq_ids.unsqueeze(1)
            same.fill_diagonal_(False)
            sim = (vp @ vg.T) / self.temperature
            sim = sim.masked_fill(same, -1e4)
            sim = sim.masked_fill(~valid_d.unsqueeze(0), -1e4)
            # keep each row's own positive so no row is all -1e4
            diag = torch.eye(q, dtype=torch.bool, device=sim.device)
            sim = torch.where(diag, (vp * vg).sum(-1, keepdim=True)
                              .expand(-1, q) / self.temperature, sim)
            labels = torch.arange(q, device=sim.device)
            per_row = F.cross_entropy(sim, labels, reduction="none")
            vmask = valid_d.float()
            nv = vmask.sum().clamp(min=1.0)
            loss_dense = (per_row * vmask).sum() / nv
            vec_cos_sim = (F.cosine_similarity(vp, vg) * vmask).sum() / nv
        else:
            vec_pred = F.normalize(
                vec_pred.reshape(q, -1)[valid_d], p=2, dim=-1, eps=1e-12)
            vec_gt = F.normalize(
                vec_gt.reshape(q, -1)[valid_d], p=2, dim=-1, eps=1e-12)
            seq_ids = torch.arange(b, device=h.device).unsqueeze(1) \
                .expand(-1, n_pos).reshape(-1)[valid_d]
            same = seq_ids.unsqueeze(0) == seq_ids.unsqueeze(1)
            same.fill_diagonal_(False)
            sim = (vec_pred @ vec_gt.T) / self.temperature
            sim = sim.masked_fill(same, -1e4)
            labels = torch.arange(sim.size(0), device=sim.device)
            loss_dense = ops.softmax_ce(sim.contiguous(), labels)

            vec_cos_sim = F.cosine_similarity(vec_pred, vec_gt).mean()
        with torch.no_grad():
            usage = torch.stack([
                F.one_hot(input_ids[:, c::self.C], self.pad_id + 1)
                .sum((0, 1)).float() for c in range(self.C)])
            prob = usage / usage.sum(1, keepdim=True)
            entropy = -(prob * prob.add(1e-12).log()).sum(1).mean()

        return CobraOutput(
            loss=loss_sparse + loss_dense, loss_sparse=loss_sparse,
            loss_dense=loss_dense, acc_correct=total_correct,
            acc_total=total_tokens, recall_correct=recall_correct,
            recall_total=recall_total, vec_cos_sim=vec_cos_sim,
            codebook_entropy=entropy)

    # ---------------------------------------------------------- generation

    @torch.no_grad()
    def generate(self, input_ids: Tensor, encoder_input_ids: Tensor,
                 n_candidates: int = 10,
                 temperature: float = 1.0) -> CobraGenerationOutput:
        b = input_ids.size(0)
        k = n_candidates
        device = input_ids.device
        vocab = self.sparse_head[0].out_features
        vecs = self.encoder(encoder_input_ids)
        t_items = vecs.size(1)

        beam_seqs = []
        beam_scores = None
        h_last = None
        for c in range(self.C):
            if c == 0:
                seq_mask = self.interleave_seq_mask(
                    input_ids != self.pad_id, self.C,
                    n_complete_items=t_items)
                emb = self.cobra_emb(input_ids, vecs, seq_mask,
                                     n_complete_items=t_items)
                h = self.decoder(emb, tgt_key_padding_mask=~seq_mask)
                last = seq_mask.sum(dim=1) - 1
                h_c = h[torch.arange(b, device=device), last]
                logits = self.sparse_head[0](h_c) / temperature
                if self.C == 1:
                    h_last = h_c.unsqueeze(1).expand(-1, k, -1)
                lp = F.log_softmax(logits, dim=-1)
                beam_scores, top_ids = lp.topk(k, dim=-1)
                beam_seqs = [top_ids]
            else:
                cur_k = beam_scores.size(1)
                exp_ids = input_ids.unsqueeze(1).expand(-1, cur_k, -1)
                gen = torch.stack(beam_seqs, dim=-1)
                flat_ids = torch.cat([exp_ids, gen], dim=-1) \
                    .reshape(b * cur_k, -1)
                flat_vecs = vecs.unsqueeze(1).expand(-1, cur_k, -1, -1) \
                    .reshape(b * cur_k, t_items, -1)
                seq_mask = self.interleave_seq_mask(
                    flat_ids != self.pad_id, self.C,
                    n_complete_items=t_items)
                emb = self.cobra_emb(flat_ids, flat_vecs, seq_mask,
                                     n_complete_items=t_items)
                h = self.decoder(emb, tgt_key_padding_mask=~seq_mask)
                last = seq_mask.sum(dim=1) - 1
                h_c = h[torch.arange(b * cur_k, device=device), last]
                lp = F.log_softmax(
                    self.sparse_head[c](h_c) / temperature, dim=-1) \
                    .view(b, cur_k, vocab)
                combined = (beam_scores.unsqueeze(-1) + lp).view(b, -1)
                beam_scores, pos = combined.topk(k, dim=-1)
                parent = pos // vocab
                tok = pos % vocab
                beam_seqs = [s.gather(1, parent) for s in beam_seqs]
                beam_seqs.append(tok)
                if c == self.C - 1:
                    h_r = h_c.view(b, cur_k, -1)
                    h_last = h_r.gather(
                        1, parent.unsqueeze(-1).expand(-1, -1, h_r.size(-1)))

        sem_ids = torch.stack(beam_seqs, dim=-1)
        dense_vecs = F.normalize(h_last, p=2, dim=-1)
        return CobraGenerationOutput(sem_ids=sem_ids, dense_vecs=dense_vecs,
                                     scores=beam_scores)

    @torch.no_grad()
    def generate_itemvec(self, encoder_input_ids: Tensor) -> Tensor:
        return F.normalize(self.encoder(encoder_input_ids), p=2, dim=-1,
                           eps=1e-12)

    @torch.no_grad()
    def beam_fusion(self, input_ids: Tensor, encoder_input_ids: Tensor,
                    item_dense_vecs: Tensor, item_sem_ids: Tensor,
                    n_candidates: int = 10, n_beam: int = 50,
                    temperature: float = 1.0,
                    alpha: float = 0.5) -> BeamFusionOutput:
        b = input_ids.size(0)
        n_beam = max(n_beam, n_candidates)
        gen = self.generate(input_ids, encoder_input_ids,
                            n_candidates=n_beam, temperature=temperature)
        item_dense_vecs = F.normalize(item_dense_vecs, p=2, dim=-1)
        sim = torch.einsum("bkd,nd->bkn", gen.dense_vecs, item_dense_vecs)
        max_sim, best_items = sim.max(dim=-1)
        beam_norm = torch.softmax(gen.scores, dim=-1)
        fused = alpha * beam_norm + (1 - alpha) * (max_sim + 1) / 2
        top_scores, top_idx = fused.topk(n_candidates, dim=-1)
        top_items = best_items.gather(1, top_idx)
        return BeamFusionOutput(item_ids=top_items,
                                sem_ids=item_sem_ids[top_items],
                                scores=top_scores)
