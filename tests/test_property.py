"""Property-based tests (hypothesis) for foundational pieces: config value
parsing, metric math, bucketing."""

import math

import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from genrec_amd.config import ginlite
from genrec_amd.modules import TopKAccumulator, relative_position_bucket

literals = st.recursive(
    st.one_of(st.integers(-10**6, 10**6),
              st.floats(allow_nan=False, allow_infinity=False,
                        width=32),
              st.booleans(), st.none(),
              st.text(alphabet=st.characters(
                  whitelist_categories=("Ll", "Lu", "Nd"),
                  whitelist_characters=" _-"), max_size=12)),
    lambda c: st.lists(c, max_size=4),
    max_leaves=6)


@settings(max_examples=150, deadline=None)
@given(literals)
def test_ginlite_parses_any_python_literal(value):
    assert ginlite._parse_value(repr(value)) == value


@settings(max_examples=60, deadline=None)
@given(st.integers(0, 2**31), st.integers(1, 8), st.integers(1, 6),
       st.integers(2, 5))
def test_topk_accumulator_matches_bruteforce(seed, b, k, d):
    g = torch.Generator().manual_seed(seed)
    actual = torch.randint(0, 4, (b, d), generator=g)
    topk = torch.randint(0, 4, (b, k, d), generator=g)
    acc = TopKAccumulator(ks=[1, k])
    acc.accumulate(actual, topk)
    m = acc.reduce()
    # brute force: first exact-match rank per row
    hits1 = hitsk = ndcg = 0.0
    for i in range(b):
        rank = None
        for r in range(k):
            if torch.equal(topk[i, r], actual[i]):
                rank = r
                break
        if rank is not None:
            hitsk += 1
            ndcg += 1.0 / math.log2(rank + 2)
            if rank == 0:
                hits1 += 1
    assert abs(m["Recall@1"] - hits1 / b) < 1e-6
    assert abs(m[f"Recall@{k}"] - hitsk / b) < 1e-6 or k == 1
    assert abs(m[f"NDCG@{k}"] - ndcg / b) < 1e-6


@settings(max_examples=60, deadline=None)
@given(st.integers(-500, 500), st.integers(4, 64).filter(lambda x: x % 2 == 0),
       st.integers(16, 256))
def test_relative_position_bucket_bounds(rel, num_buckets, max_distance):
    b = relative_position_bucket(torch.tensor([[rel]]),
                                 num_buckets=num_buckets,
                                 max_distance=max_distance,
                                 bidirectional=True)
    assert 0 <= b.item() < num_buckets
    # monotone in |distance| within each sign
    if rel > 1:
        b2 = relative_position_bucket(torch.tensor([[rel - 1]]),
                                      num_buckets=num_buckets,
                                      max_distance=max_distance,
                                      bidirectional=True)
        assert b2.item() <= b.item()


@settings(max_examples=40, deadline=None)
@given(st.integers(2, 50), st.integers(1, 6), st.integers(0, 2**31))
def test_sem_id_flat_index_bijective(vocab, dim, seed):
    """SemIdEmbedding's type*V+id flat index is a bijection over
    (type, id) pairs — no two pairs share a row, padding row is last."""
    from genrec_amd.modules import SemIdEmbedding

    emb = SemIdEmbedding(num_embeddings=vocab, sem_ids_dim=dim,
                         embeddings_dim=4)
    g = torch.Generator().manual_seed(seed)
    ids = torch.randint(0, vocab, (1, dim), generator=g)
    types = torch.arange(dim).unsqueeze(0)
    flat = types * vocab + ids
    assert flat.max() < vocab * dim  # always below the padding row
    assert emb.padding_idx == vocab * dim
    # bijection: enumerate all pairs
    allp = torch.cartesian_prod(torch.arange(dim), torch.arange(vocab))
    rows = allp[:, 0] * vocab + allp[:, 1]
    assert rows.unique().numel() == dim * vocab


@settings(max_examples=30, deadline=None)
@given(st.integers(1, 40), st.integers(1, 8))
def test_user_id_embedding_modulo(num, uid):
    """UserIdEmbedding hashing: id % num_embeddings (ref embedding.py:62-74)
    for any id, including ids far beyond the table."""
    from genrec_amd.modules import UserIdEmbedding

    emb = UserIdEmbedding(num_embeddings=num, embeddings_dim=4)
    big = uid * 1_000_003 + 7
    out = emb(torch.tensor([[big]]))
    ref = emb.emb(torch.tensor([[big % num]]))
    assert torch.equal(out, ref)


@settings(max_examples=30, deadline=None)
@given(st.integers(1, 6), st.integers(1, 5), st.integers(0, 2**31),
       st.sampled_from(["left", "right"]))
def test_tiger_collate_shape_invariants(batch_size, sem_dim, seed, side):
    from genrec_amd.data.collate import tiger_pad_collate
    from genrec_amd.data.schemas import SeqData

    g = np.random.default_rng(seed)
    batch = []
    for b in range(batch_size):
        n_items = int(g.integers(1, 7))
        batch.append(SeqData(
            user_id=int(g.integers(0, 1000)),
            item_ids=[int(x) for x in g.integers(0, 16, n_items * sem_dim)],
            target_ids=[int(x) for x in g.integers(0, 16, sem_dim)]))
    out = tiger_pad_collate(batch, sem_id_dim=sem_dim, padding_side=side)
    L = out["item_input_ids"].shape[1]
    assert L == max(len(s.item_ids) for s in batch)
    assert out["seq_mask"].shape == (batch_size, L)
    assert out["token_type_ids"].max() < sem_dim
    # mask counts exactly the real tokens
    assert out["seq_mask"].sum().item() == sum(len(s.item_ids)
                                               for s in batch)
    # masked positions' token ids are pad (0)
    assert (out["item_input_ids"][out["seq_mask"] == 0] == 0).all()
