"""LCRec: LLM + collaborative semantics (arXiv:2311.09049).

Parity target: /root/reference/genrec/models/lcrec.py (243 LoC). A causal
LLM backbone (Qwen2-family) extended with `<Ci_j>` codebook special tokens
(lcrec.py:48-60), SFT tokenization helpers (lcrec.py:88-112), HF-format
save/load (lcrec.py:135-162), and constrained top-k beam generation.

MI355X redesign:
  * offline-first: with no pretrained directory (this environment has no
    network), the backbone is a random-init Qwen2 built from a config and
    the tokenizer is a from-scratch byte-level BPE — the full pipeline
    (vocab resize, SFT, constrained decode) runs identically
  * generate_topk is a BATCHED KV-cached beam search: the reference
    re-runs the full forward per beam per step with no cache
    (lcrec.py:164-243 — B*W forwards of the whole prefix each step); here
    all beams advance in one forward on the incremental token with a
    DynamicCache reordered per beam hop, and the per-position legal-token
    mask is a device tensor rather than a Python callable per token.
"""

from __future__ import annotations

import os
from typing import Callable, Dict, List, Optional, Tuple

import logging

import torch
import torch.nn.functional as F
from torch import nn

from genrec_amd.config import ginlite

logger = logging.getLogger("genrec_amd")


def default_qwen_config(vocab_size: int = 512, hidden_size: int = 1536,
                        num_layers: int = 28, num_heads: int = 12,
                        num_kv_heads: int = 2, intermediate_size: int = 8960,
                        max_position_embeddings: int = 4096):
    """Qwen2.5-1.5B-shaped config (the reference's shipped backbone,
    config/base.gin:19) buildable offline."""
    from transformers import Qwen2Config

    return Qwen2Config(
        vocab_size=vocab_size, hidden_size=hidden_size,
        intermediate_size=intermediate_size, num_hidden_layers=num_layers,
        num_attention_heads=num_heads, num_key_value_heads=num_kv_heads,
        max_position_embeddings=max_position_embeddings, tie_word_embeddings=True)


@ginlite.configurable(name="LCRec")
class LCRec(nn.Module):
    def __init__(self, pretrained_path: Optional[str] = None,
                 config=None) -> None:
        super().__init__()
        from transformers import AutoModelForCausalLM, AutoTokenizer

        if pretrained_path and os.path.isdir(pretrained_path):
            self.tokenizer = AutoTokenizer.from_pretrained(pretrained_path)
            self.model = AutoModelForCausalLM.from_pretrained(pretrained_path)
        else:
            from genrec_amd.utils.tokenizer import build_offline_tokenizer

            self.tokenizer = build_offline_tokenizer()
            cfg = config or default_qwen_config(
                vocab_size=max(len(self.tokenizer), 512))
            if cfg.vocab_size < len(self.tokenizer):
                cfg.vocab_size = len(self.tokenizer)
            self.model = AutoModelForCausalLM.from_config(cfg)

    def gradient_checkpointing_enable(self):
        self.model.gradient_checkpointing_enable()

    def add_codebook_tokens(self, num_codebooks: int, codebook_size: int):
        """Append <Ci_j> special tokens and resize embeddings
        (ref lcrec.py:48-60)."""
        new_tokens = [f"<C{i}_{j}>" for i in range(num_codebooks)
                      for j in range(codebook_size)]
        num_added = self.tokenizer.add_special_tokens(
            {"additional_special_tokens": new_tokens})
        if num_added > 0:
            self.model.resize_token_embeddings(len(self.tokenizer))
            self.model.config.vocab_size = len(self.tokenizer)

    def codebook_token_ids(self, num_codebooks: int,
                           codebook_size: int) -> torch.Tensor:
        """[C, V] token ids of the codebook special tokens."""
        ids = [[self.tokenizer.convert_tokens_to_ids(f"<C{i}_{j}>")
                for j in range(codebook_size)] for i in range(num_codebooks)]
        return torch.tensor(ids, dtype=torch.long)

    def tokenize(self, prompt: str, *args, **kwargs):
        return self.tokenizer(prompt, *args, **kwargs)

    def decode(self, ids: torch.Tensor, *args, **kwargs) -> str:
        return self.tokenizer.decode(ids, *args, **kwargs)

    def tokenize_sft_format(self, prompt: str, response: str = "",
                            device=torch.device("cpu")) -> Dict:
        prompt_ids = self.tokenizer(prompt).input_ids
        response_ids = self.tokenizer(response).input_ids
        input_ids = prompt_ids + response_ids + [self.tokenizer.eos_token_id]
        t = torch.LongTensor([input_ids]).to(device)
        return {"input_ids": t, "prompt_seq_length": len(prompt_ids),
                "attention_mask": torch.ones_like(t)}

    def forward(self, input_ids, attention_mask=None, labels=None, **kw):
        return self.model(input_ids=input_ids, attention_mask=attention_mask,
                          labels=labels)

    def save_pretrained(self, save_dir: str, **kwargs):
        self.model.save_pretrained(save_dir, **kwargs)
        self.tokenizer.save_pretrained(save_dir)

    def load_pretrained(self, load_dir: str):
        from transformers import AutoModelForCausalLM, AutoTokenizer

        self.tokenizer = AutoTokenizer.from_pretrained(load_dir)
        self.model = AutoModelForCausalLM.from_pretrained(
            load_dir, torch_dtype=torch.bfloat16)
        logger.info("Loaded checkpoint from %s", load_dir)

    @torch.no_grad()
    def generate_topk(
        self,
        input_ids: torch.Tensor,      # [B, L] (left-padded)
        attention_mask: Optional[torch.Tensor] = None,
        max_new_tokens: int = 3,
        beam_width: int = 10,
        topk: Optional[int] = None,
        allowed_token_ids: Optional[List[torch.Tensor]] = None,
        allowed_token_fn: Optional[Callable[[int], bool]] = None,
        eos_token_id: Optional[int] = None,
        temperature: float = 1.0,
    ) -> List[List[Tuple[torch.Tensor, float]]]:
        """Batched KV-cached constrained beam search.

        allowed_token_ids: per-step 1-D tensors of legal token ids (the
        device-mask replacement for the reference's per-token Python
        callable); allowed_token_fn is still honored for parity when
        given (applied as a precomputed vocab mask).
        """
        from transformers import DynamicCache

        device = input_ids.device
        B, L = input_ids.shape
        W = beam_width
        topk = topk or beam_width
        eos = eos_token_id if eos_token_id is not None \
            else self.tokenizer.eos_token_id
        V = self.model.config.vocab_size
        if attention_mask is None:
            attention_mask = torch.ones_like(input_ids)

        fn_mask = None
        if allowed_token_fn is not None:
            keep = [allowed_token_fn(t) for t in range(V)]
            fn_mask = torch.tensor(keep, dtype=torch.bool, device=device)

        # prefill once per batch row
        cache = DynamicCache()
        out = self.model(input_ids=input_ids, attention_mask=attention_mask,
                         past_key_values=cache, use_cache=True)
        logits = out.logits[:, -1, :] / temperature          # [B, V]
        if hasattr(cache, "batch_repeat_interleave"):
            cache.batch_repeat_interleave(W)
        else:  # older transformers: expand via legacy tuples
            idx = torch.arange(B, device=device).repeat_interleave(W)
            legacy = tuple(
                (k.index_select(0, idx), v.index_select(0, idx))
                for k, v in cache.to_legacy_cache())
            cache = DynamicCache.from_legacy_cache(legacy)
        attn = attention_mask.repeat_interleave(W, dim=0)     # [B*W, L]

        def legal(step_logits: torch.Tensor, step: int) -> torch.Tensor:
            lp = F.log_softmax(step_logits, dim=-1)
            mask = torch.zeros(V, dtype=torch.bool, device=device)
            if allowed_token_ids is not None and step < len(allowed_token_ids):
                mask[allowed_token_ids[step].to(device)] = True
            else:
                mask[:] = True
            if fn_mask is not None:
                mask &= fn_mask
            return lp.masked_fill(~mask, float("-inf"))

        lp0 = legal(logits, 0)                                # [B, V]
        first_scores, first_tok = torch.topk(lp0, W, dim=-1)  # [B, W]
        beam_scores = first_scores.clone()                    # [B, W]
        beam_tokens = first_tok.unsqueeze(-1)                 # [B, W, 1]
        finished = first_tok == eos                           # [B, W]

        for step in range(1, max_new_tokens):
            if bool(finished.all()):
                break
            last = beam_tokens[:, :, -1].reshape(B * W, 1)
            attn = torch.cat([attn, torch.ones(B * W, 1, dtype=attn.dtype,
                                               device=device)], dim=1)
            out = self.model(input_ids=last, attention_mask=attn,
                             past_key_values=cache, use_cache=True)
            logits = out.logits[:, -1, :] / temperature       # [B*W, V]
            lp = legal(logits, step).view(B, W, V)
            # finished beams keep their score and emit only eos
            lp = torch.where(
                finished.unsqueeze(-1),
                torch.full_like(lp, float("-inf")).scatter(
                    -1, torch.full((B, W, 1), eos, dtype=torch.long,
                                   device=device), 0.0),
                lp)
            total = beam_scores.unsqueeze(-1) + lp            # [B, W, V]
            flat = total.view(B, W * V)
            beam_scores, pos = torch.topk(flat, W, dim=-1)
            parent = pos // V                                 # [B, W]
            tok = pos % V
            beam_tokens = torch.cat([
                beam_tokens.gather(1, parent.unsqueeze(-1).expand(
                    B, W, beam_tokens.size(-1))),
                tok.unsqueeze(-1),
            ], dim=-1)
            finished = finished.gather(1, parent) | (tok == eos)
            flat_parent = (torch.arange(B, device=device).unsqueeze(1) * W
                           + parent).reshape(-1)
            cache.reorder_cache(flat_parent)
            attn = attn.index_select(0, flat_parent)

        results: List[List[Tuple[torch.Tensor, float]]] = []
        for b in range(B):
            order = beam_scores[b].argsort(descending=True)
            row = []
            for w in order[:topk].tolist():
                seq = torch.cat([input_ids[b], beam_tokens[b, w]])
                row.append((seq, float(beam_scores[b, w])))
            results.append(row)
        return results
