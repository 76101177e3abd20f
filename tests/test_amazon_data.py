"""Amazon pipeline tests on tiny generated fixture files (offline)."""

import gzip
import json
import os

import numpy as np
import pytest
import torch


@pytest.fixture(scope="module")
def amazon_root(tmp_path_factory):
    root = tmp_path_factory.mktemp("amazon")
    raw = root / "raw" / "beauty"
    raw.mkdir(parents=True)
    rng = np.random.default_rng(0)
    n_items, n_users = 30, 12
    asins = [f"A{i:04d}" for i in range(n_items)]
    reviews = []
    for u in range(n_users):
        n = rng.integers(5, 9)
        items = rng.choice(n_items, size=n, replace=False)
        for j, it in enumerate(items):
            reviews.append({"asin": asins[it], "reviewerID": f"U{u}",
                            "unixReviewTime": 1_400_000_000 + u * 100 + j})
    with gzip.open(raw / "reviews_Beauty_5.json.gz", "wt") as f:
        for r in reviews:
            f.write(json.dumps(r) + "\n")
        f.write("this line is broken\n")  # malformed-line tolerance
    with gzip.open(raw / "meta_Beauty.json.gz", "wt") as f:
        for i, a in enumerate(asins):
            # python-literal style line (single quotes) like the real files
            f.write(str({"asin": a, "title": f"Item {i}", "brand": f"B{i % 3}",
                         "price": 9.99, "categories": [["Beauty"]]}) + "\n")
    return str(root)


@pytest.fixture(scope="module")
def rqvae_ckpt(tmp_path_factory, amazon_root):
    from genrec_amd.data.amazon import AmazonItemDataset
    from genrec_amd.models.rqvae import QuantizeForwardMode, RqVae

    ds = AmazonItemDataset(root=amazon_root, split="beauty",
                           train_test_split="all", embed_dim=48)
    m = RqVae(input_dim=48, embed_dim=8, hidden_dims=[16], codebook_size=8,
              codebook_mode=QuantizeForwardMode.STE,
              codebook_last_layer_mode=QuantizeForwardMode.STE,
              n_layers=3, n_cat_features=0)
    x = torch.stack([ds[i] for i in range(len(ds))])
    m(x, gumbel_t=0.2)  # kmeans init
    path = tmp_path_factory.mktemp("rq") / "ck.pt"
    torch.save({"epoch": 0, "model": m.state_dict()}, str(path))
    return str(path)


def test_item_dataset_and_split(amazon_root):
    from genrec_amd.data.amazon import AmazonItemDataset

    ds = AmazonItemDataset(root=amazon_root, split="beauty",
                           train_test_split="all", embed_dim=48)
    assert len(ds) == 30 and ds[0].shape == (48,)
    tr = AmazonItemDataset(root=amazon_root, split="beauty",
                           train_test_split="train", embed_dim=48)
    ev = AmazonItemDataset(root=amazon_root, split="beauty",
                           train_test_split="eval", embed_dim=48)
    assert len(tr) + len(ev) == 30
    # parquet cache reused
    assert os.path.exists(os.path.join(
        amazon_root, "processed", "beauty", "item_embeddings.parquet"))


def test_seq_dataset_with_rqvae(amazon_root, rqvae_ckpt):
    from genrec_amd.data.amazon import AmazonSeqDataset

    ds = AmazonSeqDataset(
        root=amazon_root, split="beauty", train_test_split="train",
        max_seq_len=10, add_disambiguation=True,
        pretrained_rqvae_path=rqvae_ckpt, rqvae_input_dim=48,
        rqvae_embed_dim=8, rqvae_hidden_dims=[16], rqvae_codebook_size=8,
        rqvae_n_layers=3)
    assert ds.sem_id_dim == 4  # 3 codes + disambiguation
    s = ds[0]
    assert len(s.item_ids) % 4 == 0 and len(s.target_ids) == 4
    valid = ds.all_valid_sem_ids()
    assert valid.shape == (30, 4)
    # leave-one-out valid/test have one sample per user
    dv = AmazonSeqDataset(
        root=amazon_root, split="beauty", train_test_split="valid",
        pretrained_rqvae_path=rqvae_ckpt, rqvae_input_dim=48,
        rqvae_embed_dim=8, rqvae_hidden_dims=[16], rqvae_codebook_size=8,
        rqvae_n_layers=3)
    assert len(dv) == 12


def test_disambiguation_suffix():
    from genrec_amd.data.amazon import add_disambiguation_suffix

    ids = [[1, 2], [1, 2], [3, 4], [1, 2]]
    out = add_disambiguation_suffix(ids)
    assert out == [[1, 2, 0], [1, 2, 1], [3, 4, 0], [1, 2, 2]]


def test_sasrec_hstu_datasets(amazon_root):
    from genrec_amd.data.amazon import AmazonHSTUDataset, AmazonSASRecDataset

    ds = AmazonSASRecDataset(root=amazon_root, split="beauty",
                             train_test_split="train", max_seq_len=10)
    assert ds.num_items == 30
    assert all(s["target"] >= 1 for s in ds.samples)  # 1-based ids
    h = AmazonHSTUDataset(root=amazon_root, split="beauty",
                          train_test_split="valid", max_seq_len=10)
    s = h[0]
    assert "timestamps" in s and len(s["timestamps"]) == len(s["history"])


def test_cobra_lcrec_datasets(amazon_root, rqvae_ckpt):
    from genrec_amd.data.amazon import AmazonCobraDataset, AmazonLCRecDataset

    kw = dict(rqvae_input_dim=48, rqvae_embed_dim=8,
              rqvae_hidden_dims=[16])
    ds = AmazonCobraDataset(
        root=amazon_root, split="beauty", train_test_split="train",
        n_codebooks=3, id_vocab_size=8, max_text_len=12,
        pretrained_rqvae_path=rqvae_ckpt, **kw)
    s = ds[0]
    assert len(s["target_sem_ids"]) == 3
    assert ds.all_item_text().shape == (30, 12)

    lc = AmazonLCRecDataset(
        root=amazon_root, split="beauty", train_test_split="train",
        sem_id_dim=3, codebook_size=8, pretrained_rqvae_path=rqvae_ckpt,
        max_samples=50, **kw)
    assert 0 < len(lc) <= 50
    assert "<C0_" in lc[0]["response"] or "<C0_" in lc[0]["prompt"]


def test_missing_raw_raises(tmp_path):
    from genrec_amd.data.amazon import AmazonSASRecDataset

    with pytest.raises(FileNotFoundError):
        AmazonSASRecDataset(root=str(tmp_path), split="beauty",
                            train_test_split="train")


def test_user_hash_is_process_stable(amazon_root, rqvae_ckpt):
    """User-id hashing must be deterministic across processes (the
    reference's python hash() is PYTHONHASHSEED-randomized, which maps
    the same user to different embedding rows on different DDP ranks)."""
    import subprocess
    import sys

    code = (
        "import zlib; print(zlib.crc32('A1B2C3'.encode()) % 10000)")
    outs = {subprocess.run([sys.executable, "-c", code],
                           capture_output=True, text=True,
                           env={"PYTHONHASHSEED": str(i)}).stdout
            for i in (0, 1)}
    assert len(outs) == 1  # same value under different hash seeds
