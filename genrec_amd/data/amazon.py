"""Amazon-Reviews-2014 data pipelines.

Parity targets: /root/reference/genrec/data/amazon.py (492 LoC),
amazon_sasrec.py (192), amazon_hstu.py (212), amazon_cobra.py (274),
amazon_lcrec.py (690). Same raw files (5-core reviews + metadata gz),
same processing:

  * tolerant gzip-JSON parsing (json then python-literal fallback,
    amazon.py:69-80)
  * AmazonItemDataset: item text -> encoder embeddings cached to parquet,
    seeded 95/5 split (amazon.py:84-239)
  * AmazonSeqDataset: loads a pretrained RQ-VAE to tokenize all items into
    semantic IDs, optional 4th disambiguation code, sliding-window train /
    leave-one-out valid/test, user hash % 10000 (amazon.py:259-459)
  * AmazonSASRecDataset / AmazonHSTUDataset: raw 1-based item-ID sequences
    (+ timestamps for HSTU)
  * AmazonCobraDataset / AmazonLCRecDataset: built on the shared review
    loader + RQ-VAE sem-IDs

This environment has NO network: downloads are not attempted; a missing
raw file raises with the expected path. The embedding encoder is pluggable
("light" = offline random-init LightT5Encoder; a sentence-transformers
directory when locally available).
"""

from __future__ import annotations

import gzip
import hashlib
import json
import logging
import os
from typing import Dict, List, Optional, Tuple

import numpy as np
import torch
from torch.utils.data import Dataset

from genrec_amd.config import ginlite
from genrec_amd.data.schemas import SeqData

logger = logging.getLogger("genrec_amd")

DATASET_CONFIGS = {
    "beauty": {"reviews": "reviews_Beauty_5.json.gz",
               "meta": "meta_Beauty.json.gz"},
    "sports": {"reviews": "reviews_Sports_and_Outdoors_5.json.gz",
               "meta": "meta_Sports_and_Outdoors.json.gz"},
    "toys": {"reviews": "reviews_Toys_and_Games_5.json.gz",
             "meta": "meta_Toys_and_Games.json.gz"},
    "clothing": {"reviews": "reviews_Clothing_Shoes_and_Jewelry_5.json.gz",
                 "meta": "meta_Clothing_Shoes_and_Jewelry.json.gz"},
}


def parse_gzip_json(path: str):
    """Line-by-line gz JSON with python-literal fallback (ref amazon.py:69-80)."""
    import ast

    with gzip.open(path, "rt", encoding="utf-8") as g:
        for line in g:
            line = line.strip()
            if not line:
                continue
            try:
                yield json.loads(line)
            except json.JSONDecodeError:
                try:
                    yield ast.literal_eval(line)
                except (ValueError, SyntaxError):
                    continue


def _require_raw(root: str, split: str, kind: str) -> str:
    cfg = DATASET_CONFIGS[split]
    path = os.path.join(root, "raw", split, cfg[kind])
    if not os.path.exists(path):
        # also accept flat layout root/<file>
        alt = os.path.join(root, cfg[kind])
        if os.path.exists(alt):
            return alt
        raise FileNotFoundError(
            f"Amazon raw file missing: {path}. This environment has no "
            f"network; place the 2014 5-core file there manually.")
    return path


def load_user_sequences(root: str, split: str, zero_based: bool = True,
                        min_len: int = 5
                        ) -> Tuple[List[List[int]], List[List[int]],
                                   List[str], Dict[str, int]]:
    """Per-user (items, timestamps) sorted by time + asin->id mapping."""
    reviews_path = _require_raw(root, split, "reviews")
    item_map: Dict[str, int] = {}
    users: Dict[str, List[Tuple[int, int]]] = {}
    base = 0 if zero_based else 1
    for r in parse_gzip_json(reviews_path):
        asin, uid = r.get("asin"), r.get("reviewerID")
        ts = r.get("unixReviewTime", 0)
        if not (asin and uid):
            continue
        if asin not in item_map:
            item_map[asin] = len(item_map) + base
        users.setdefault(uid, []).append((ts, item_map[asin]))
    seqs, ts_seqs, user_ids = [], [], []
    for uid, entries in users.items():
        entries.sort(key=lambda x: x[0])
        if len(entries) >= min_len:
            seqs.append([e[1] for e in entries])
            ts_seqs.append([e[0] for e in entries])
            user_ids.append(uid)
    return seqs, ts_seqs, user_ids, item_map


def load_item_metadata(root: str, split: str,
                       item_map: Dict[str, int]) -> Dict[int, dict]:
    meta_path = _require_raw(root, split, "meta")
    info: Dict[int, dict] = {}
    for m in parse_gzip_json(meta_path):
        asin = m.get("asin")
        if asin in item_map:
            info[item_map[asin]] = {
                "title": m.get("title"), "price": m.get("price"),
                "salesRank": m.get("salesRank"), "brand": m.get("brand"),
                "categories": m.get("categories"),
            }
    return info


def item_semantics_text(info: dict) -> str:
    """Text rendered for embedding (ref amazon.py:198-205 format)."""
    return (f"'title':{info.get('title', '')}\n"
            f" 'price':{info.get('price', '')}\n"
            f" 'salesRank':{info.get('salesRank', '')}\n"
            f" 'brand':{info.get('brand', '')}\n"
            f" 'categories':{info.get('categories', '')}")


@ginlite.configurable(name="AmazonItemDataset")
class AmazonItemDataset(Dataset):
    """Item embedding dataset for RQ-VAE training (ref amazon.py:84-239)."""

    def __init__(self, root: str = "dataset/amazon", split: str = "beauty",
                 train_test_split: str = "all",
                 encoder_model_name: str = "light",
                 force_regenerate: bool = False, embed_dim: int = 768,
                 embed_batch_size: int = 64) -> None:
        self.root, self.split = root, split.lower()
        self.train_test_split = train_test_split
        self.processed_dir = os.path.join(root, "processed", self.split)
        self.parquet_path = os.path.join(self.processed_dir,
                                         "item_embeddings.parquet")
        if os.path.exists(self.parquet_path) and not force_regenerate:
            self._load_cached()
        else:
            self._generate(encoder_model_name, embed_dim, embed_batch_size)
        self._apply_split()

    def _load_cached(self) -> None:
        import pandas as pd

        df = pd.read_parquet(self.parquet_path)
        self.embeddings = np.stack(df["embedding"].values, axis=0)
        self.dim = self.embeddings.shape[-1]

    def _generate(self, encoder_model_name: str, embed_dim: int,
                  batch_size: int) -> None:
        import pandas as pd

        os.makedirs(self.processed_dir, exist_ok=True)
        # item ids from reviews order, 1-based (ref amazon.py:165-171)
        reviews_path = _require_raw(self.root, self.split, "reviews")
        item_map: Dict[str, int] = {}
        for r in parse_gzip_json(reviews_path):
            asin = r.get("asin")
            if asin and asin not in item_map:
                item_map[asin] = len(item_map) + 1
        info = load_item_metadata(self.root, self.split, item_map)
        texts = [item_semantics_text(info.get(i, {}))
                 for i in sorted(set(item_map.values()))]

        if encoder_model_name and encoder_model_name != "light" \
                and os.path.exists(encoder_model_name):
            from sentence_transformers import SentenceTransformer

            model = SentenceTransformer(encoder_model_name)
            embs = model.encode(texts, batch_size=batch_size,
                                show_progress_bar=True)
        else:
            # offline: deterministic hashed bag-of-character-ngrams features
            # through a LightT5Encoder-style projection substitute
            embs = _hashed_text_embeddings(texts, embed_dim)
        df = pd.DataFrame({
            "ItemID": sorted(set(item_map.values())),
            "embedding": [np.asarray(e, dtype=np.float32).tolist()
                          for e in embs],
        })
        df.to_parquet(self.parquet_path, index=False)
        self.embeddings = np.asarray(
            [np.asarray(e, dtype=np.float32) for e in embs])
        self.dim = self.embeddings.shape[-1]

    def _apply_split(self) -> None:
        if self.train_test_split == "all":
            return
        gen = torch.Generator()
        gen.manual_seed(42)
        is_train = (torch.rand(len(self.embeddings), generator=gen)
                    > 0.05).numpy()
        self.embeddings = self.embeddings[is_train] \
            if self.train_test_split == "train" else self.embeddings[~is_train]

    def __len__(self) -> int:
        return len(self.embeddings)

    def __getitem__(self, idx: int):
        return torch.tensor(self.embeddings[idx], dtype=torch.float32)


def _hashed_text_embeddings(texts: List[str], dim: int) -> np.ndarray:
    """Deterministic offline text features: hashed char 3-gram counts,
    L2-normalized. A stand-in for sentence-T5 when no checkpoint exists —
    preserves the 'similar text -> similar vector' property the RQ-VAE
    needs."""
    out = np.zeros((len(texts), dim), dtype=np.float32)
    for i, t in enumerate(texts):
        t = t.lower()
        for j in range(len(t) - 2):
            g = t[j:j + 3]
            h = int(hashlib.md5(g.encode()).hexdigest()[:8], 16)
            out[i, h % dim] += 1.0
        n = np.linalg.norm(out[i])
        if n > 0:
            out[i] /= n
    return out


def tokenize_items_with_rqvae(item_embeddings: torch.Tensor,
                              pretrained_rqvae_path: str,
                              rqvae_input_dim: int = 768,
                              rqvae_embed_dim: int = 32,
                              rqvae_hidden_dims: List[int] = [512, 256, 128, 64],
                              rqvae_codebook_size: int = 256,
                              rqvae_n_layers: int = 3) -> List[List[int]]:
    """Tokenize all items to sem-IDs via a pretrained RQ-VAE
    (ref amazon.py:296-322)."""
    from genrec_amd.models.rqvae import RqVae

    rqvae = RqVae(input_dim=rqvae_input_dim, embed_dim=rqvae_embed_dim,
                  hidden_dims=list(rqvae_hidden_dims),
                  codebook_size=rqvae_codebook_size,
                  codebook_kmeans_init=False, n_layers=rqvae_n_layers,
                  n_cat_features=0)
    rqvae.load_pretrained(pretrained_rqvae_path)
    rqvae.eval()
    with torch.no_grad():
        return rqvae.get_semantic_ids(item_embeddings).sem_ids.tolist()


def add_disambiguation_suffix(sem_ids_list: List[List[int]]) -> List[List[int]]:
    """Append an incremental 4th code per collision group
    (ref amazon.py:323-353)."""
    from collections import defaultdict

    groups = defaultdict(list)
    for item_id, codes in enumerate(sem_ids_list):
        groups[tuple(codes)].append(item_id)
    out = []
    for item_id, codes in enumerate(sem_ids_list):
        out.append(list(codes) + [groups[tuple(codes)].index(item_id)])
    return out


@ginlite.configurable(name="AmazonSeqDataset")
class AmazonSeqDataset(Dataset):
    """Semantic-ID sequences for TIGER (ref amazon.py:243-459)."""

    def __init__(self, root: str = "dataset/amazon", split: str = "beauty",
                 train_test_split: str = "train", max_seq_len: int = 20,
                 max_items_per_seq: Optional[int] = None,
                 subsample: bool = True,
                 add_disambiguation: bool = True,
                 pretrained_rqvae_path: str = "./out/rqvae/{split}/checkpoint_final.pt",
                 encoder_model_name: str = "light",
                 rqvae_input_dim: int = 768, rqvae_embed_dim: int = 32,
                 rqvae_hidden_dims: List[int] = [512, 256, 128, 64],
                 rqvae_codebook_size: int = 256,
                 rqvae_n_layers: int = 3) -> None:
        self.split = split.lower()
        self.train_test_split = train_test_split
        self._max_seq_len = max_items_per_seq or max_seq_len
        pretrained_rqvae_path = pretrained_rqvae_path.format(split=self.split)

        item_ds = AmazonItemDataset(root=root, split=split,
                                    train_test_split="all",
                                    encoder_model_name=encoder_model_name)
        embs = torch.tensor(item_ds.embeddings, dtype=torch.float32)
        self.sem_ids_list = tokenize_items_with_rqvae(
            embs, pretrained_rqvae_path, rqvae_input_dim=rqvae_input_dim,
            rqvae_embed_dim=rqvae_embed_dim,
            rqvae_hidden_dims=rqvae_hidden_dims,
            rqvae_codebook_size=rqvae_codebook_size,
            rqvae_n_layers=rqvae_n_layers)
        if add_disambiguation:
            self.sem_ids_list = add_disambiguation_suffix(self.sem_ids_list)
        self.sem_id_dim = len(self.sem_ids_list[0])
        self.codebook_size = rqvae_codebook_size

        # zero-based item ids aligned with sem_ids_list rows
        self.sequences, _, self.user_ids, _ = load_user_sequences(
            root, self.split, zero_based=True)
        self._generate_samples()

    def _generate_samples(self) -> None:
        import zlib

        self.samples = []
        for uidx, full in enumerate(self.sequences):
            # deliberate fix of a reference quirk: the reference uses
            # python hash() % 10000 (amazon.py:412), which is randomized
            # per process (PYTHONHASHSEED) — under multi-process DDP the
            # SAME user maps to different embedding rows on different
            # ranks. crc32 is stable across processes and runs.
            user_id = zlib.crc32(str(self.user_ids[uidx]).encode()) % 10000
            if self.train_test_split == "train":
                seq = full[:-2]
                for i in range(1, len(seq)):
                    self.samples.append((user_id, seq[:i], seq[i]))
            elif self.train_test_split == "valid":
                seq = full[:-1]
                self.samples.append((user_id, seq[:-1], seq[-1]))
            else:
                self.samples.append((user_id, full[:-1], full[-1]))

    def all_valid_sem_ids(self) -> torch.Tensor:
        return torch.tensor(self.sem_ids_list, dtype=torch.long)

    @property
    def max_seq_len(self) -> int:
        return self._max_seq_len

    def __len__(self) -> int:
        return len(self.samples)

    def __getitem__(self, idx: int) -> SeqData:
        user_id, history, target = self.samples[idx]
        history = history[-self._max_seq_len:]
        hist = [c for it in history for c in self.sem_ids_list[it]]
        tgt = list(self.sem_ids_list[target])
        return SeqData(user_id=user_id, item_ids=hist, target_ids=tgt)


class _AmazonRawSeqBase(Dataset):
    """Raw 1-based item-ID sequences (SASRec/HSTU, ref amazon_sasrec.py)."""

    def __init__(self, root: str, split: str, train_test_split: str,
                 max_seq_len: int, with_timestamps: bool) -> None:
        self.max_seq_len = max_seq_len
        self.with_timestamps = with_timestamps
        seqs, ts, _, item_map = load_user_sequences(root, split.lower(),
                                                    zero_based=False)
        self.num_items = len(item_map)
        self.samples: List[Dict] = []
        for full, fts in zip(seqs, ts):
            if train_test_split == "train":
                seq, t = full[:-2], fts[:-2]
                if len(seq) < 2:
                    continue
                for i in range(1, len(seq)):
                    lo = max(0, i - max_seq_len)
                    self.samples.append({"history": seq[lo:i],
                                         "ts": t[lo:i], "target": seq[i]})
            else:
                seq = full[:-1] if train_test_split == "valid" else full
                t = fts[:-1] if train_test_split == "valid" else fts
                if len(seq) < 2:
                    continue
                lo = max(0, len(seq) - 1 - max_seq_len)
                self.samples.append({"history": seq[lo:-1], "ts": t[lo:-1],
                                     "target": seq[-1]})

    def __len__(self) -> int:
        return len(self.samples)

    def __getitem__(self, idx: int) -> Dict:
        s = self.samples[idx]
        out = {"history": s["history"], "target": s["target"]}
        if self.with_timestamps:
            out["timestamps"] = s["ts"]
        return out


@ginlite.configurable(name="AmazonSASRecDataset")
class AmazonSASRecDataset(_AmazonRawSeqBase):
    def __init__(self, root: str = "dataset/amazon", split: str = "beauty",
                 train_test_split: str = "train",
                 max_seq_len: int = 50) -> None:
        super().__init__(root, split, train_test_split, max_seq_len,
                         with_timestamps=False)


@ginlite.configurable(name="AmazonHSTUDataset")
class AmazonHSTUDataset(_AmazonRawSeqBase):
    def __init__(self, root: str = "dataset/amazon", split: str = "beauty",
                 train_test_split: str = "train",
                 max_seq_len: int = 50) -> None:
        super().__init__(root, split, train_test_split, max_seq_len,
                         with_timestamps=True)


@ginlite.configurable(name="AmazonCobraDataset")
class AmazonCobraDataset(Dataset):
    """Sem-IDs + tokenized item text for COBRA (ref amazon_cobra.py)."""

    def __init__(self, root: str = "dataset/amazon", split: str = "beauty",
                 train_test_split: str = "train",
                 max_items_per_seq: int = 20, max_text_len: int = 128,
                 n_codebooks: int = 3, id_vocab_size: int = 256,
                 pretrained_rqvae_path: str = "./out/rqvae/{split}/checkpoint_final.pt",
                 encoder_model_name: str = "light",
                 tokenizer=None, **rqvae_kw) -> None:
        self.C = n_codebooks
        self.id_vocab_size = id_vocab_size
        split = split.lower()
        item_ds = AmazonItemDataset(root=root, split=split,
                                    train_test_split="all",
                                    encoder_model_name=encoder_model_name)
        embs = torch.tensor(item_ds.embeddings, dtype=torch.float32)
        self.sem_ids = tokenize_items_with_rqvae(
            embs, pretrained_rqvae_path.format(split=split),
            rqvae_codebook_size=id_vocab_size,
            rqvae_n_layers=n_codebooks, **rqvae_kw)
        seqs, _, _, item_map = load_user_sequences(root, split,
                                                   zero_based=True)
        info = load_item_metadata(
            root, split, {a: i + 1 for a, i in item_map.items()})
        if tokenizer is None:
            from genrec_amd.utils.tokenizer import build_offline_tokenizer

            tokenizer = build_offline_tokenizer()
        n_items = len(item_map)
        self.item_text = np.zeros((n_items, max_text_len), dtype=np.int64)
        for i in range(n_items):
            meta = info.get(i + 1, {})
            text = f"{meta.get('title') or ''} {meta.get('brand') or ''}".strip()
            ids = tokenizer(text).input_ids[:max_text_len]
            self.item_text[i, :len(ids)] = ids
        self.num_items = n_items
        self.samples = []
        for full in seqs:
            if train_test_split == "train":
                seq = full[:-2]
            elif train_test_split == "valid":
                seq = full[:-1]
            else:
                seq = full
            if len(seq) < 2:
                continue
            seq = seq[-(max_items_per_seq + 1):]
            self.samples.append({"history": seq[:-1], "target": seq[-1]})

    def all_item_text(self) -> torch.Tensor:
        return torch.tensor(self.item_text, dtype=torch.long)

    def all_item_sem_ids(self) -> torch.Tensor:
        return torch.tensor(self.sem_ids, dtype=torch.long)

    def __len__(self) -> int:
        return len(self.samples)

    def __getitem__(self, idx: int) -> Dict:
        s = self.samples[idx]
        return {
            "history_sem_ids": [list(self.sem_ids[i]) for i in s["history"]],
            "history_text": [self.item_text[i] for i in s["history"]],
            "target_sem_ids": list(self.sem_ids[s["target"]]),
            "target_text": self.item_text[s["target"]],
            "target_item": s["target"],
        }


@ginlite.configurable(name="AmazonLCRecDataset")
class AmazonLCRecDataset(Dataset):
    """Multi-task SFT samples from Amazon reviews (ref amazon_lcrec.py)."""

    def __init__(self, root: str = "dataset/amazon", split: str = "beauty",
                 train_test_split: str = "train", max_history: int = 20,
                 sem_id_dim: int = 5, codebook_size: int = 256,
                 pretrained_rqvae_path: str = "./out/rqvae/{split}/checkpoint_final.pt",
                 encoder_model_name: str = "light",
                 max_samples: Optional[int] = None, **rqvae_kw) -> None:
        from genrec_amd.data.lcrec_sft import LCRecSFTDatasetBase

        split = split.lower()
        item_ds = AmazonItemDataset(root=root, split=split,
                                    train_test_split="all",
                                    encoder_model_name=encoder_model_name)
        embs = torch.tensor(item_ds.embeddings, dtype=torch.float32)
        sem_ids = tokenize_items_with_rqvae(
            embs, pretrained_rqvae_path.format(split=split),
            rqvae_codebook_size=codebook_size, rqvae_n_layers=sem_id_dim,
            **rqvae_kw)
        seqs, _, _, item_map = load_user_sequences(root, split,
                                                   zero_based=True)
        info = load_item_metadata(
            root, split, {a: i + 1 for a, i in item_map.items()})
        n = len(item_map)
        titles = [str(info.get(i + 1, {}).get("title") or f"item {i}")
                  for i in range(n)]
        descs = [f"{info.get(i + 1, {}).get('brand') or ''} "
                 f"{info.get(i + 1, {}).get('categories') or ''}".strip()
                 or f"item {i}" for i in range(n)]
        self._base = LCRecSFTDatasetBase(
            seqs, np.asarray(sem_ids), titles, descs,
            split=train_test_split, max_history=max_history,
            max_samples=max_samples)
        self.samples = self._base.samples
        self.sem_id_dim = sem_id_dim
        self.codebook_size = codebook_size

    def __len__(self) -> int:
        return len(self.samples)

    def __getitem__(self, idx: int) -> Dict:
        return self.samples[idx]
