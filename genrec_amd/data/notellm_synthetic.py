"""Synthetic query/note pairs for NoteLLM contrastive training.

The reference ships the NoteLLM MODEL only ("no
trainer/config ships for it"); this dataset + trainers/notellm_trainer.py
make the family trainable end to end in the offline environment. Pairs
share a topic word so in-batch contrastive retrieval is learnable; every
text carries the [EMB] pooling token the model gathers.
"""

from __future__ import annotations

from typing import Dict

import numpy as np
from torch.utils.data import Dataset

from genrec_amd.config import ginlite

_WORDS = ["travel", "food", "music", "fitness", "beauty", "coding",
          "books", "games", "garden", "finance"]


@ginlite.configurable(name="SyntheticNotePairDataset")
class SyntheticNotePairDataset(Dataset):
    def __init__(self, num_pairs: int = 512, n_topics: int = 10,
                 split: str = "train", seed: int = 0) -> None:
        rng = np.random.default_rng(seed + (0 if split == "train" else 1))
        self.samples = []
        for i in range(num_pairs):
            t = int(rng.integers(0, min(n_topics, len(_WORDS))))
            w = _WORDS[t]
            a = int(rng.integers(0, 1000))
            b = int(rng.integers(0, 1000))
            self.samples.append({
                "query": f"note about {w} number {a} [EMB]",
                "positive": f"another {w} note id {b} [EMB]",
                "category": w,
            })

    def __len__(self) -> int:
        return len(self.samples)

    def __getitem__(self, idx: int) -> Dict[str, str]:
        return self.samples[idx]
