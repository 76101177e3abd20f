"""Small data helpers (parity: reference genrec/data/utils.py)."""

from __future__ import annotations


def cycle(dataloader):
    """Infinite iterator over a DataLoader (re-iterates each epoch; used
    by iteration-keyed training loops — ref data/utils.py:7-12)."""
    while True:
        for batch in dataloader:
            yield batch
