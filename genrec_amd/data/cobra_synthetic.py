"""COBRA datasets + collates.

Parity target: /root/reference/genrec/data/amazon_cobra.py (274 LoC):
per-item sem-ID tuples + per-item tokenized text for the dense encoder
(max_text_len=128), ONE sample per user (no sliding window,
amazon_cobra.py:168-209), train-time target-appended collate vs eval-time
separated history/target (cobra_trainer.py:25-88).
"""

from __future__ import annotations

from typing import Dict, List

import numpy as np
import torch
from torch.utils.data import Dataset

from genrec_amd.config import ginlite
from genrec_amd.data.synthetic import _zipf_sequences


@ginlite.configurable(name="SyntheticCobraDataset")
class SyntheticCobraDataset(Dataset):
    def __init__(self, num_users: int = 500, num_items: int = 2000,
                 mean_len: float = 8.9, max_items_per_seq: int = 20,
                 n_codebooks: int = 3, id_vocab_size: int = 256,
                 text_vocab_size: int = 32128, max_text_len: int = 16,
                 split: str = "train", seed: int = 0) -> None:
        self.C = n_codebooks
        self.id_vocab_size = id_vocab_size
        self.num_items = num_items
        self.max_items_per_seq = max_items_per_seq
        rng = np.random.default_rng(seed + 71)
        self.item_sem_ids = rng.integers(
            0, id_vocab_size, size=(num_items + 1, n_codebooks))
        # token id 0 is pad; items get 4-16 text tokens
        lens = rng.integers(4, max_text_len + 1, size=num_items + 1)
        self.item_text = np.zeros((num_items + 1, max_text_len), dtype=np.int64)
        for i in range(1, num_items + 1):
            self.item_text[i, :lens[i]] = rng.integers(
                1, text_vocab_size, size=lens[i])
        seqs = _zipf_sequences(num_users, num_items, mean_len, seed)
        self.samples = []
        for u, full in enumerate(seqs):
            if split == "train":
                seq = full[:-2]
            elif split == "valid":
                seq = full[:-1]
            else:
                seq = full
            if len(seq) < 2:
                continue
            seq = seq[-(max_items_per_seq + 1):]
            self.samples.append({"user": u, "history": seq[:-1],
                                 "target": seq[-1]})

    def all_item_text(self) -> torch.Tensor:
        """[N, L_text] text tokens for item-vec precompute (item 1..N)."""
        return torch.tensor(self.item_text[1:], dtype=torch.long)

    def all_item_sem_ids(self) -> torch.Tensor:
        return torch.tensor(self.item_sem_ids[1:], dtype=torch.long)

    def __len__(self) -> int:
        return len(self.samples)

    def __getitem__(self, idx: int) -> Dict:
        s = self.samples[idx]
        return {
            "history_sem_ids": [list(map(int, self.item_sem_ids[i]))
                                for i in s["history"]],
            "history_text": [self.item_text[i] for i in s["history"]],
            "target_sem_ids": list(map(int, self.item_sem_ids[s["target"]])),
            "target_text": self.item_text[s["target"]],
            "target_item": s["target"] - 1,  # 0-based for eval tables
        }


def cobra_collate_fn(batch: List[Dict], pad_id: int, n_codebooks: int,
                     train: bool = True,
                     fixed_items: int = 0) -> Dict[str, torch.Tensor]:
    """Train: target appended to history (model's shifted loss covers it).
    Eval: history only, target kept aside (ref cobra_trainer.py:25-88).
    fixed_items > 0 pads every batch to that item count (static shapes
    for hipGraph capture)."""
    B = len(batch)
    if train:
        items = [b["history_sem_ids"] + [b["target_sem_ids"]] for b in batch]
        texts = [list(b["history_text"]) + [b["target_text"]] for b in batch]
    else:
        items = [b["history_sem_ids"] for b in batch]
        texts = [list(b["history_text"]) for b in batch]
    max_t = max(len(x) for x in items)
    if fixed_items:
        max_t = max(max_t, fixed_items)
    text_len = len(batch[0]["target_text"])
    input_ids = torch.full((B, max_t * n_codebooks), pad_id, dtype=torch.long)
    enc = torch.zeros(B, max_t, text_len, dtype=torch.long)
    for i, (its, txt) in enumerate(zip(items, texts)):
        flat = [c for item in its for c in item]
        input_ids[i, :len(flat)] = torch.tensor(flat)
        for t, tt in enumerate(txt):
            enc[i, t] = torch.as_tensor(tt)
    out = {"input_ids": input_ids, "encoder_input_ids": enc}
    if not train:
        out["target_sem_ids"] = torch.tensor(
            [b["target_sem_ids"] for b in batch])
        out["target_item"] = torch.tensor(
            [b["target_item"] for b in batch])
    return out
