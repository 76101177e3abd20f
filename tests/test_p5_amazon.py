"""P5 pipeline tests on generated fixtures."""

import gzip
import json

import numpy as np
import pytest
import torch


@pytest.fixture(scope="module")
def p5_root(tmp_path_factory):
    root = tmp_path_factory.mktemp("p5")
    base = root / "beauty"
    base.mkdir()
    rng = np.random.default_rng(0)
    n_items, n_users = 25, 10
    with open(base / "sequential_data.txt", "w") as f:
        for u in range(1, n_users + 1):
            n = rng.integers(5, 10)
            items = rng.choice(n_items, size=n, replace=False) + 1
            f.write(f"{u} " + " ".join(map(str, items)) + "\n")
    with open(base / "datamaps.json", "w") as f:
        json.dump({"id2item": {str(i): f"A{i:04d}"
                               for i in range(1, n_items + 1)}}, f)
    with gzip.open(base / "meta.json.gz", "wt") as f:
        for i in range(1, n_items + 1):
            f.write(str({"asin": f"A{i:04d}", "title": f"P5 Item {i}",
                         "brand": "B", "price": 5.0}) + "\n")
    return str(root)


def test_amazon_reviews_loader(p5_root):
    from genrec_amd.data.p5_amazon import AmazonReviews

    data = AmazonReviews(p5_root, "beauty", embed_dim=32)
    assert data.num_items == 25
    assert data.item_embeddings.shape == (25, 32)
    wins = data.rolling_windows(window_size=5)
    assert len(wins) > 0
    trains = [w for w, t in wins if t]
    tests = [w for w, t in wins if not t]
    assert len(trains) > len(tests) > 0


def test_p5_item_and_seq_datasets(p5_root, tmp_path):
    from genrec_amd.data.p5_amazon import (
        P5AmazonReviewsItemDataset, P5AmazonReviewsSeqDataset,
    )
    from genrec_amd.models.rqvae import QuantizeForwardMode, RqVae

    ds = P5AmazonReviewsItemDataset(root=p5_root, split="beauty",
                                    embed_dim=32)
    assert len(ds) == 25 and ds[0].shape == (32,)

    m = RqVae(input_dim=32, embed_dim=8, hidden_dims=[16], codebook_size=8,
              codebook_mode=QuantizeForwardMode.STE,
              codebook_last_layer_mode=QuantizeForwardMode.STE,
              n_layers=3, n_cat_features=0)
    x = torch.stack([ds[i] for i in range(len(ds))])
    m(x, gumbel_t=0.2)
    ck = tmp_path / "rq.pt"
    torch.save({"epoch": 0, "model": m.state_dict()}, str(ck))

    seq = P5AmazonReviewsSeqDataset(
        root=p5_root, split="beauty", train_test_split="train",
        pretrained_rqvae_path=str(ck), rqvae_input_dim=32,
        rqvae_embed_dim=8, rqvae_hidden_dims=[16], rqvae_codebook_size=8,
        rqvae_n_layers=3)
    assert len(seq) > 0
    s = seq[0]
    assert len(s.item_ids) % 3 == 0 and len(s.target_ids) == 3
    ts = P5AmazonReviewsSeqDataset(
        root=p5_root, split="beauty", train_test_split="test",
        pretrained_rqvae_path=str(ck), rqvae_input_dim=32,
        rqvae_embed_dim=8, rqvae_hidden_dims=[16], rqvae_codebook_size=8,
        rqvae_n_layers=3)
    assert len(ts) == 10  # one per user


def test_processed_artifact_roundtrip(tmp_path):
    """build_processed mirrors the reference HeteroData artifact layout
    (p5_amazon.py:322-368): leave-two-out histories with -1-padded eval
    windows, item_x embeddings, 95/5 is_train mask (seed 42)."""
    import json

    from genrec_amd.data.p5_amazon import AmazonReviews

    base = tmp_path / "beauty"
    base.mkdir()
    seqs = [[1, 2, 3, 4, 5], [2, 3, 4], [5, 1, 2, 3]]
    (base / "sequential_data.txt").write_text("\n".join(
        f"{u} " + " ".join(map(str, s)) for u, s in enumerate(seqs)))
    (base / "datamaps.json").write_text(json.dumps(
        {"id2item": {str(i): f"A{i}" for i in range(1, 6)}}))
    ar = AmazonReviews(root=str(tmp_path), split="beauty", embed_dim=32)
    path = ar.save_processed(max_seq_len=4)
    art = AmazonReviews.load_processed(path)

    h = art["history"]
    # first user 1..5 -> 0-based 0..4: train=[0,1,2], fut=3;
    # test window = last max_seq_len+1 minus the target = [0,1,2,3], fut=4
    assert h["train"]["itemId"][0].tolist() == [0, 1, 2]
    assert h["train"]["itemId_fut"][0].item() == 3
    assert h["test"]["itemId"][0].tolist() == [0, 1, 2, 3]
    assert h["test"]["itemId_fut"][0].item() == 4
    assert h["val"]["itemId"][0].tolist() == [0, 1, 2, -1]
    # second user len 3: train=[1], fut=2
    assert h["train"]["itemId"][1].tolist() == [1]
    assert art["item_x"].shape == (5, 32)
    assert len(art["item_text"]) == 5
    assert art["item_is_train"].dtype == torch.bool
    assert art["item_is_train"].shape == (5,)
