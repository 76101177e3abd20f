"""Offline tokenizer construction.

The environment has no network, so HuggingFace tokenizers can't be
downloaded. When no local pretrained directory is supplied, LCRec/NoteLLM
build a byte-level BPE tokenizer from scratch (full byte alphabet, no
merges needed beyond bytes) — every string round-trips, and the `<Ci_j>`
codebook tokens are added as special tokens exactly as with a real
tokenizer (ref lcrec.py:48-60).
"""

from __future__ import annotations

from transformers import PreTrainedTokenizerFast


def build_offline_tokenizer(eos_token: str = "<|endoftext|>"
                            ) -> PreTrainedTokenizerFast:
    from tokenizers import Tokenizer, decoders, models, pre_tokenizers, trainers

    tok = Tokenizer(models.BPE())
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    tok.decoder = decoders.ByteLevel()
    trainer = trainers.BpeTrainer(
        vocab_size=512, special_tokens=[eos_token],
        initial_alphabet=pre_tokenizers.ByteLevel.alphabet())
    # tiny seed corpus just to materialize the byte alphabet
    tok.train_from_iterator(
        ["The user has interacted with items 0123456789.",
         "Please recommend the next item: title brand category"],
        trainer)
    return PreTrainedTokenizerFast(
        tokenizer_object=tok, eos_token=eos_token, pad_token=eos_token)
