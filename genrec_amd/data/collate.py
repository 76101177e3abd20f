"""Collate functions (parity: amazon_sasrec.py:125-181, amazon_hstu.py:137-200,
tiger_trainer.py:27-80)."""

from __future__ import annotations

from typing import Dict, List

import torch

from genrec_amd.data.schemas import SeqData


def sasrec_collate_fn(batch: List[Dict], max_seq_len: int = 50,
                      fixed_length: bool = False) -> Dict:
    """Left-pad; input = seq[:-1], target = seq[1:] (shifted next-item).
    fixed_length pads every batch to max_seq_len (hipGraph capture)."""
    histories = [b["history"] for b in batch]
    targets = [b["target"] for b in batch]
    max_len = max_seq_len if fixed_length else \
        min(max(len(h) for h in histories), max_seq_len)
    input_ids, target_ids = [], []
    for history, target in zip(histories, targets):
        if len(history) > max_len:
            history = history[-max_len:]
        seq = history + [target]
        pad = [0] * (max_len + 1 - len(seq))
        padded = pad + seq
        input_ids.append(padded[:-1])
        target_ids.append(padded[1:])
    return {
        "input_ids": torch.tensor(input_ids, dtype=torch.long),
        "targets": torch.tensor(target_ids, dtype=torch.long),
    }


def sasrec_eval_collate_fn(batch: List[Dict], max_seq_len: int = 50) -> Dict:
    histories = [b["history"] for b in batch]
    targets = [b["target"] for b in batch]
    max_len = min(max(len(h) for h in histories), max_seq_len)
    input_ids = []
    for history in histories:
        if len(history) > max_len:
            history = history[-max_len:]
        input_ids.append([0] * (max_len - len(history)) + history)
    return {
        "input_ids": torch.tensor(input_ids, dtype=torch.long),
        "targets": torch.tensor(targets, dtype=torch.long),
    }


def hstu_collate_fn(batch: List[Dict], max_seq_len: int = 50,
                    fixed_length: bool = False) -> Dict:
    """SASRec collate + per-position unix timestamps carried through."""
    out = sasrec_collate_fn(batch, max_seq_len, fixed_length=fixed_length)
    L = out["input_ids"].size(1)
    ts = []
    for b in batch:
        t = list(b.get("timestamps", []))[-L:]
        # align with input positions: timestamps of history items (the last
        # input position's timestamp repeats for the appended target slot)
        t = t[: L]
        ts.append([0] * (L - len(t)) + t)
    out["timestamps"] = torch.tensor(ts, dtype=torch.long)
    return out


def hstu_eval_collate_fn(batch: List[Dict], max_seq_len: int = 50) -> Dict:
    out = sasrec_eval_collate_fn(batch, max_seq_len)
    L = out["input_ids"].size(1)
    ts = []
    for b in batch:
        t = list(b.get("timestamps", []))[-L:]
        ts.append([0] * (L - len(t)) + t)
    out["timestamps"] = torch.tensor(ts, dtype=torch.long)
    return out


def tiger_pad_collate(batch: List[SeqData], pad_id: int = 0,
                      padding_side: str = "left",
                      sem_id_dim: int = 3,
                      fixed_length: int = 0) -> Dict[str, torch.Tensor]:
    """Flattened sem-ID history with token_type = pos % sem_id_dim
    (ref tiger_trainer.py:27-80; note the reference 'left' branch actually
    places ids at the sequence start — reproduced). fixed_length > 0 pads
    every batch to that length (required for hipGraph-captured steps)."""
    B = len(batch)
    natural = max(len(x.item_ids) for x in batch)
    if fixed_length:
        assert natural <= fixed_length, (
            f"tiger_pad_collate: sample length {natural} exceeds "
            f"fixed_length {fixed_length} (set max_items_per_seq to match "
            f"max_seq_len for hipGraph training)")
    max_len = fixed_length or natural
    tgt_len = len(batch[0].target_ids)
    user_ids = torch.zeros(B, 1, dtype=torch.long)
    ids = torch.full((B, max_len), pad_id, dtype=torch.long)
    mask = torch.zeros(B, max_len, dtype=torch.long)
    token_type = torch.zeros(B, max_len, dtype=torch.long)
    tgt_ids = torch.full((B, tgt_len), pad_id, dtype=torch.long)
    tgt_type = torch.zeros(B, tgt_len, dtype=torch.long)
    for i, s in enumerate(batch):
        n = len(s.item_ids)
        user_ids[i, 0] = s.user_id
        if padding_side == "left":
            ids[i, :n] = torch.tensor(s.item_ids)
            token_type[i, :n] = torch.arange(n) % sem_id_dim
            mask[i, :n] = 1
        else:
            ids[i, max_len - n:] = torch.tensor(s.item_ids)
            token_type[i, max_len - n:] = torch.arange(n) % sem_id_dim
            mask[i, max_len - n:] = 1
        tgt_ids[i] = torch.tensor(s.target_ids)
        tgt_type[i] = torch.arange(tgt_len)
    return {
        "user_input_ids": user_ids,
        "item_input_ids": ids,
        "token_type_ids": token_type,
        "target_input_ids": tgt_ids,
        "target_token_type_ids": tgt_type,
        "seq_mask": mask,
    }
