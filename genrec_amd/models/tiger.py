"""TIGER: generative retrieval over semantic IDs (arXiv:2305.05065).

Parity target: /root/reference/genrec/models/tiger.py (572 LoC). Training
forward matches the reference exactly: user-ID + sem-ID embeddings, RMSNorm
+ in-proj, T5 encoder-decoder (4+4 layers in the shipped config), flat
output head over num_item_embeddings*sem_id_dim+1 vocab, per-sequence summed
CE (tiger.py:146-240). Positional embeddings are allocated but unused, as in
the reference (tiger.py:172-179).

Generation is the MI355X-first redesign of tiger.py:312-452. The reference
walks a CPU dict trie and dedups beams in per-(batch,beam) Python loops with
a GPU sync every step; here the trie is a device-resident child-index table,
candidate sampling uses the Gumbel-top-K equivalent of the reference's
``multinomial`` (exactly the same distribution, but defined for rows with
fewer legal tokens than samples), and beam dedup/re-rank runs as a batched
double-sort on device. No host round-trips inside the decode loop.
"""

from __future__ import annotations

from typing import NamedTuple, Optional

import torch
from torch import Tensor, nn

from genrec_amd import ops
from genrec_amd.ops.linear import SplitKLinear
from genrec_amd.config import ginlite
from genrec_amd.modules.embeddings import SemIdEmbedding, UserIdEmbedding
from genrec_amd.modules.norms import RMSNorm, T5RMSNorm
from genrec_amd.modules.transformer import TransformerEncoderDecoder

NEG_INF_SCORE = -1e32  # reference's illegal-token fill (tiger.py:376)


class TigerOutput(NamedTuple):
    logits: Tensor
    loss: Optional[Tensor]


class TigerGenerationOutput(NamedTuple):
    sem_ids: Tensor
    log_probas: Tensor


class DeviceTrie:
    """Flattened trie over [N, C] id tuples: child-index table per node.

    Node 0 is the DEAD node (no children); node 1 is the root. Legal-token
    masks are `children[node] >= 0`. Replaces the reference's CPU dict trie
    (tiger.py:41-69) with one gather per decode step.
    """

    def __init__(self, valid_item_ids: Tensor, num_tokens: int):
        ids = valid_item_ids.reshape(-1, valid_item_ids.size(-1)).long().cpu()
        n, c = ids.shape
        edges = []  # (parent_node, token, child_node)
        prefix_nodes = torch.ones(n, dtype=torch.long)  # all at root (=1)
        next_free = 2
        for level in range(c):
            key = prefix_nodes * num_tokens + ids[:, level]
            uniq, inv = key.unique(return_inverse=True)
            child = torch.arange(next_free, next_free + uniq.numel())
            edges.append((uniq // num_tokens, uniq % num_tokens, child))
            prefix_nodes = child[inv]
            next_free += uniq.numel()
        children = torch.full((next_free, num_tokens), -1, dtype=torch.long)
        for parent, tok, child in edges:
            children[parent, tok] = child
        self.children = children
        self.num_tokens = num_tokens

    def to(self, device) -> "DeviceTrie":
        self.children = self.children.to(device)
        return self

    def legal_mask(self, nodes: Tensor) -> Tensor:
        """nodes: [M] -> bool [M, num_tokens]."""
        return self.children[nodes] >= 0

    def advance(self, nodes: Tensor, tokens: Tensor) -> Tensor:
        nxt = self.children[nodes, tokens]
        return nxt.clamp_min(0)  # -1 (no child) -> DEAD node 0


@ginlite.configurable(name="Tiger")
class Tiger(nn.Module):
    def __init__(self, embedding_dim: int, attn_dim: int, dropout: float,
                 num_heads: int, n_layers: int, num_item_embeddings: int,
                 num_user_embeddings: int, sem_id_dim: int,
                 max_pos: int = 2048) -> None:
        super().__init__()
        self.trie: Optional[DeviceTrie] = None
        self.embedding_dim = embedding_dim
        self.attn_dim = attn_dim
        self.num_heads = num_heads
        self.n_layers = n_layers
        self.num_item_embeddings = num_item_embeddings
        self.num_user_embeddings = num_user_embeddings
        self.sem_id_dim = sem_id_dim
        self.max_pos = max_pos

        self.bos_embedding = nn.Parameter(torch.randn(embedding_dim))
        self.norm = RMSNorm(embedding_dim)
        self.norm_context = RMSNorm(embedding_dim)
        self.drop = nn.Dropout(p=dropout)
        self.sem_id_embedding = SemIdEmbedding(
            num_embeddings=num_item_embeddings, sem_ids_dim=sem_id_dim,
            embeddings_dim=embedding_dim)
        self.user_id_embedding = UserIdEmbedding(
            num_embeddings=num_user_embeddings, embeddings_dim=embedding_dim)
        # allocated but unused, mirroring the reference (tiger.py:128-131)
        self.pos_embedding = nn.Embedding(max_pos, embedding_dim)
        self.decoder_pos_embedding = nn.Embedding(sem_id_dim, embedding_dim)

        self.in_proj = SplitKLinear(embedding_dim, attn_dim, bias=False)
        self.in_proj_context = SplitKLinear(embedding_dim, attn_dim, bias=False)
        self.transformer = TransformerEncoderDecoder(
            d_model=attn_dim, nhead=num_heads,
            num_encoder_layers=n_layers // 2,
            num_decoder_layers=n_layers // 2,
            dim_feedforward=1024, dropout=dropout, norm_cls=T5RMSNorm)
        self.out_proj = SplitKLinear(attn_dim, embedding_dim, bias=False)
        self.vocab_size = num_item_embeddings * sem_id_dim + 1
        self.output_head = SplitKLinear(attn_dim, self.vocab_size, bias=False)

    # ---------------- training forward ----------------

    def forward(self, user_input_ids: Tensor, item_input_ids: Tensor,
                token_type_ids: Tensor, target_input_ids: Tensor,
                target_token_type_ids: Tensor,
                seq_mask: Optional[Tensor]) -> TigerOutput:
        if seq_mask is None:
            seq_mask = torch.ones_like(item_input_ids)
        b = item_input_ids.size(0)

        user_emb = self.user_id_embedding(user_input_ids)
        item_emb = self.sem_id_embedding(item_input_ids, token_type_ids)
        encoder_input = torch.cat([user_emb, item_emb], dim=1)

        if target_input_ids is not None:
            target_emb = self.sem_id_embedding(target_input_ids,
                                               target_token_type_ids)
            decoder_input = torch.cat(
                [self.bos_embedding.expand(b, 1, -1), target_emb], dim=1)
        else:
            decoder_input = self.bos_embedding.expand(b, 1, -1)

        enc_valid = torch.cat([
            torch.ones(b, 1, dtype=seq_mask.dtype, device=seq_mask.device),
            seq_mask,
        ], dim=1)
        pad_mask = ~enc_valid.bool()  # True = PAD

        encoder_input = self.in_proj_context(
            self.drop(self.norm_context(encoder_input)))
        decoder_input = self.in_proj(self.drop(self.norm(decoder_input)))

        t = decoder_input.size(1)
        causal = torch.triu(
            torch.full((t, t), float("-inf"), device=decoder_input.device),
            diagonal=1)
        decoder_out = self.transformer(
            src=encoder_input, tgt=decoder_input, tgt_mask=causal,
            src_key_padding_mask=pad_mask, memory_key_padding_mask=pad_mask)
        logits = self.output_head(decoder_out)

        loss = None
        if target_input_ids is not None and \
                target_input_ids.size(1) == self.sem_id_dim:
            targets = (target_token_type_ids * self.num_item_embeddings
                       + target_input_ids)
            loss = ops.summed_ce(logits[:, :-1, :], targets)
        return TigerOutput(logits=logits, loss=loss)

    # ---------------- generation ----------------

    def _encode_context(self, user_input_ids, item_input_ids, token_type_ids,
                        seq_mask):
        user_emb = self.user_id_embedding(user_input_ids)
        item_emb = self.sem_id_embedding(item_input_ids, token_type_ids)
        encoder_input = torch.cat([user_emb, item_emb], dim=1)
        enc_valid = torch.cat([
            torch.ones(seq_mask.size(0), 1, dtype=seq_mask.dtype,
                       device=seq_mask.device),
            seq_mask,
        ], dim=1)
        pad_mask = enc_valid == 0
        encoder_input = self.in_proj_context(
            self.drop(self.norm_context(encoder_input)))
        memory = self.transformer.encoder(encoder_input,
                                          key_padding_mask=pad_mask)
        return memory, pad_mask

    def _decode_step(self, memory, memory_mask, tgt_ids, tgt_type):
        b = memory.size(0)
        if tgt_ids is None:
            decoder_input = self.bos_embedding.expand(b, 1, -1)
        else:
            target_emb = self.sem_id_embedding(tgt_ids, tgt_type)
            decoder_input = torch.cat(
                [self.bos_embedding.expand(b, 1, -1), target_emb], dim=1)
        decoder_input = self.in_proj(self.drop(self.norm(decoder_input)))
        t = decoder_input.size(1)
        causal = torch.triu(
            torch.full((t, t), float("-inf"), device=decoder_input.device),
            diagonal=1)
        out = self.transformer.decoder(
            decoder_input, memory=memory, attn_mask=causal,
            memory_key_padding_mask=memory_mask)
        return self.output_head(out)[:, -1, :]

    def _decode_step_cached(self, memory, memory_mask, last_tok, step,
                            kv_caches):
        """Incremental decode: feed only the newest target row, reusing
        per-layer self/cross K/V caches (redesign of the reference's full
        re-forward per step, tiger.py:283-310)."""
        b = memory.size(0)
        if step == 0:
            decoder_input = self.bos_embedding.expand(b, 1, -1)
        else:
            tgt_type = torch.full((b, 1), step - 1, dtype=torch.long,
                                  device=memory.device)
            decoder_input = self.sem_id_embedding(last_tok.view(b, 1),
                                                  tgt_type)
        decoder_input = self.in_proj(self.drop(self.norm(decoder_input)))
        out = self.transformer.decoder(
            decoder_input, memory=memory,
            memory_key_padding_mask=memory_mask, kv_caches=kv_caches)
        return self.output_head(out)[:, -1, :]

    @staticmethod
    def _permute_kv_caches(kv_caches, flat_parent):
        """Reorder cached SELF K/V rows after beam re-ranking. Cross-attn
        K/V rows are identical for all k beams of a user (memory was
        expanded), so gathering within a user is a no-op — skipped."""
        for layer_cache in kv_caches:
            sc = layer_cache.get("self", {})
            for key in ("k", "v"):
                if key in sc:
                    sc[key] = sc[key].index_select(0, flat_parent)

    @torch.no_grad()
    def generate(self, user_input_ids: Tensor, item_input_ids: Tensor,
                 token_type_ids: Tensor, seq_mask: Optional[Tensor] = None,
                 temperature: float = 0.2, n_top_k_candidates: int = 10,
                 valid_item_ids: Optional[Tensor] = None,
                 use_trie: bool = True,
                 use_kv_cache: bool = True) -> TigerGenerationOutput:
        b, k = user_input_ids.size(0), n_top_k_candidates
        device = user_input_ids.device
        vlevel = self.num_item_embeddings

        memory, memory_mask = self._encode_context(
            user_input_ids, item_input_ids, token_type_ids, seq_mask)
        lmem = memory.size(1)
        memory = memory.unsqueeze(1).expand(-1, k, -1, -1).reshape(b * k, lmem, -1)
        memory_mask = memory_mask.unsqueeze(1).expand(-1, k, -1).reshape(b * k, -1)

        beam_seqs = torch.zeros(b, k, self.sem_id_dim, dtype=torch.long,
                                device=device)
        beam_logps = torch.zeros(b, k, device=device)

        if use_trie:
            if self.trie is None:
                self.trie = DeviceTrie(valid_item_ids, vlevel).to(device)
            nodes = torch.ones(b, k, dtype=torch.long, device=device)  # root

        kk = min(k * 6, vlevel)  # reference R=6 (tiger.py:350-351)
        kv_caches = ([{} for _ in self.transformer.decoder.layers]
                     if use_kv_cache else None)

        for step in range(self.sem_id_dim):
            if use_kv_cache:
                last = (beam_seqs[:, :, step - 1].reshape(b * k)
                        if step > 0 else None)
                logits = self._decode_step_cached(
                    memory, memory_mask, last, step, kv_caches)
            else:
                if step == 0:
                    tgt_ids, tgt_type = None, None
                else:
                    tgt_ids = beam_seqs[:, :, :step].reshape(b * k, step)
                    tgt_type = torch.arange(step, device=device) \
                        .unsqueeze(0).expand(b * k, -1)
                logits = self._decode_step(memory, memory_mask, tgt_ids,
                                           tgt_type)

            offset = step * vlevel
            full_mask = torch.full_like(logits, False, dtype=torch.bool)
            if use_trie:
                full_mask[:, offset:offset + vlevel] = \
                    self.trie.legal_mask(nodes.reshape(-1))
            else:
                full_mask[:, offset:offset + vlevel] = True
            logits = logits.masked_fill(~full_mask, NEG_INF_SCORE)

            log_probs = torch.log_softmax(logits / temperature, dim=-1)
            # Gumbel-top-K == multinomial without replacement (tiger.py:386)
            gumbel = -torch.log(-torch.log(
                torch.rand_like(log_probs) + 1e-20) + 1e-20)
            # sort-based top-KK instead of torch.topk: ATen's multi-block
            # topk (chosen at b*k >= ~512 rows) keeps a workspace that is
            # not hipGraph-replay-safe (2nd replay faults, ROCm 7 —
            # tools/repro_gen512.py); radix sort replays cleanly and ties
            # are measure-zero under Gumbel noise
            cand_vocab = (log_probs + gumbel).sort(
                dim=-1, descending=True).indices[:, :kk].contiguous()
            cand_logp = torch.gather(log_probs, 1, cand_vocab)
            cand_tok = cand_vocab - offset

            total_logp = (beam_logps.unsqueeze(-1)
                          + cand_logp.view(b, k, kk)).reshape(b, k * kk)
            total_tok = cand_tok.view(b, k * kk)
            parent = torch.arange(k, device=device).view(1, k, 1) \
                .expand(b, k, kk).reshape(b, k * kk)

            # batched dedup: pack (parent-prefix, token) into a key; keep the
            # best-scoring occurrence of each key.
            if step == 0:
                key = total_tok.clamp(min=0)
            else:
                prefix_key = beam_keys.gather(1, parent)  # [b, k*kk]
                key = prefix_key * (vlevel + 1) + (total_tok.clamp(min=0) + 1)
            # within equal keys, keep the best score: sort scores desc first,
            # then stable-sort keys (preserves desc score order per key group)
            score_desc, score_order = total_logp.sort(dim=1, descending=True)
            key_by_score = key.gather(1, score_order)
            ks2, ko2 = key_by_score.sort(dim=1, stable=True)
            first = torch.ones_like(ks2, dtype=torch.bool)
            first[:, 1:] = ks2[:, 1:] != ks2[:, :-1]
            dedup_score = torch.where(
                first, score_desc.gather(1, ko2),
                torch.full_like(score_desc, NEG_INF_SCORE))
            # index back into the (k*kk) flat candidate list
            flat_idx = score_order.gather(1, ko2)
            srt = dedup_score.sort(dim=1, descending=True)
            top_scores = srt.values[:, :k]
            top_pos = srt.indices[:, :k]
            chosen = flat_idx.gather(1, top_pos)  # [b, k]

            new_tok = total_tok.gather(1, chosen)
            new_parent = parent.gather(1, chosen)
            new_seqs = beam_seqs.gather(
                1, new_parent.unsqueeze(-1).expand(b, k, self.sem_id_dim)).clone()
            new_seqs[:, :, step] = new_tok.clamp(min=0)
            padding = top_scores <= NEG_INF_SCORE / 2
            new_seqs[padding] = 0
            beam_seqs = new_seqs
            beam_logps = top_scores
            beam_keys = key.gather(1, chosen)
            if use_trie:
                parent_nodes = nodes.gather(1, new_parent.reshape(b, k))
                nodes = self.trie.advance(parent_nodes.reshape(-1),
                                          new_tok.clamp(min=0).reshape(-1)
                                          ).reshape(b, k)
                nodes[padding] = 1  # reference resets padded beams to root
            if use_kv_cache and step < self.sem_id_dim - 1:
                flat_parent = (torch.arange(b, device=device).unsqueeze(1) * k
                               + new_parent).reshape(-1)
                self._permute_kv_caches(kv_caches, flat_parent)

        return TigerGenerationOutput(sem_ids=beam_seqs, log_probas=beam_logps)

    def load_pretrained(self, path: str) -> None:
        """Load weights from a safetensors dir/file (ref tiger.py:248-253)
        or a reference-layout dict checkpoint (.pt)."""
        import os

        if os.path.isdir(path) or path.endswith(".safetensors"):
            from safetensors.torch import load_file

            f = path if path.endswith(".safetensors") else \
                os.path.join(path, "model.safetensors")
            state = load_file(f)
        else:
            obj = torch.load(path, map_location="cpu", weights_only=False)
            state = obj.get("model", obj) if isinstance(obj, dict) else obj
        self.load_state_dict(state, strict=True)
