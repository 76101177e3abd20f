"""Model-level CPU tests: forward semantics, generation, checkpoints."""

import math

import pytest
import torch
import torch.nn.functional as F

from genrec_amd.models import (
    HSTU, RqVae, SASRec, Tiger, QuantizeForwardMode,
)
from genrec_amd.models.tiger import DeviceTrie

torch.manual_seed(0)


def _naive_sasrec_attention(q, k, v, mask, scale):
    """Independent re-implementation of the official masking order
    (sasrec.py:201-245) used to validate the fused op's semantics."""
    scores = q @ k.transpose(-2, -1) * scale
    key_mask = mask.squeeze(-1)[:, None, None, :]
    scores = scores.masked_fill(key_mask == 0, -1e9)
    L = q.size(-2)
    causal = torch.triu(torch.ones(L, L, dtype=torch.bool), diagonal=1)
    scores = scores.masked_fill(causal[None, None], -1e9)
    attn = torch.softmax(scores, dim=-1)
    attn = attn * mask.squeeze(-1)[:, None, :, None]
    return attn @ v


def test_sasrec_attention_masking_order():
    from genrec_amd import ops

    B, H, L, D = 2, 2, 8, 4
    q, k, v = torch.randn(B, H, L, D), torch.randn(B, H, L, D), \
        torch.randn(B, H, L, D)
    valid = torch.ones(B, L)
    valid[0, :3] = 0
    out = ops.sasrec_attention(q, k, v, valid, 0.5, 0.0, False)
    ref = _naive_sasrec_attention(q, k, v, valid.unsqueeze(-1), 0.5)
    assert torch.allclose(out, ref, atol=1e-6)


def test_sasrec_forward_and_loss():
    m = SASRec(num_items=50, max_seq_len=10, embed_dim=16, num_heads=2,
               num_blocks=2, ffn_dim=16, dropout=0.0)
    m.eval()
    ids = torch.randint(1, 51, (4, 10))
    ids[0, :4] = 0
    logits, loss = m(ids, ids)
    assert logits.shape == (4, 10, 51)
    assert loss.item() > 0
    # padding rows produce zero hidden prior to final norm: logits finite
    assert torch.isfinite(logits).all()
    # loss matches manual CE(ignore 0) on the same logits
    ref = F.cross_entropy(logits.reshape(-1, 51), ids.reshape(-1),
                          ignore_index=0)
    assert torch.allclose(loss, ref, atol=1e-5)
    top = m.predict(ids, top_k=5)
    assert top.shape == (4, 5) and (top != 0).all()


def test_hstu_forward_temporal():
    m = HSTU(num_items=60, embed_dim=16, num_heads=2, num_blocks=2,
             dropout=0.0)
    m.eval()
    ids = torch.randint(1, 61, (3, 12))
    ts = torch.arange(12).unsqueeze(0).expand(3, -1) * 86400 + 10 ** 9
    logits, loss = m(ids, ts, ids)
    assert logits.shape == (3, 12, 61) and loss.item() > 0
    # no-temporal variant differs
    m2 = HSTU(num_items=60, embed_dim=16, num_heads=2, num_blocks=2,
              dropout=0.0, use_temporal_bias=False)
    m2.eval()
    logits2, _ = m2(ids, None, ids)
    assert logits2.shape == (3, 12, 61)


@pytest.mark.parametrize("mode", [QuantizeForwardMode.GUMBEL_SOFTMAX,
                                  QuantizeForwardMode.STE,
                                  QuantizeForwardMode.ROTATION_TRICK,
                                  QuantizeForwardMode.SINKHORN])
def test_rqvae_modes_train_and_backward(mode):
    torch.manual_seed(0)
    m = RqVae(input_dim=24, embed_dim=8, hidden_dims=[16], codebook_size=12,
              codebook_mode=mode, codebook_last_layer_mode=mode,
              n_layers=2, n_cat_features=0)
    x = torch.randn(32, 24)
    out = m(x, gumbel_t=0.5)
    assert torch.isfinite(out.loss)
    out.loss.backward()
    grads = [p.grad for p in m.encoder.parameters()]
    assert all(g is not None and torch.isfinite(g).all() for g in grads)
    assert 0.0 <= out.p_unique_ids.item() <= 1.0


def test_rqvae_eval_ids_stable():
    m = RqVae(input_dim=24, embed_dim=8, hidden_dims=[16], codebook_size=12,
              codebook_mode=QuantizeForwardMode.STE,
              codebook_last_layer_mode=QuantizeForwardMode.STE,
              n_layers=3, n_cat_features=0)
    x = torch.randn(16, 24)
    m(x, gumbel_t=0.5)  # triggers kmeans init
    m.eval()
    a = m.get_semantic_ids(x).sem_ids
    b = m.get_semantic_ids(x).sem_ids
    assert torch.equal(a, b) and a.shape == (16, 3)


def test_device_trie_matches_python_trie():
    ids = torch.tensor([[0, 1, 2], [0, 1, 3], [1, 0, 0], [2, 2, 2]])
    trie = DeviceTrie(ids, num_tokens=4)
    # root legal tokens = {0, 1, 2}
    root_mask = trie.legal_mask(torch.tensor([1]))[0]
    assert root_mask.tolist() == [True, True, True, False]
    # walk 0 -> 1 -> legal {2,3}
    n = trie.advance(torch.tensor([1]), torch.tensor([0]))
    n = trie.advance(n, torch.tensor([1]))
    mask = trie.legal_mask(n)[0]
    assert mask.tolist() == [False, False, True, True]
    # dead path
    dead = trie.advance(torch.tensor([1]), torch.tensor([3]))
    assert dead.item() == 0
    assert not trie.legal_mask(dead).any()


def test_tiger_forward_loss_matches_manual():
    torch.manual_seed(0)
    m = Tiger(embedding_dim=16, attn_dim=24, dropout=0.0, num_heads=4,
              n_layers=2, num_item_embeddings=8, num_user_embeddings=10,
              sem_id_dim=3)
    m.eval()
    B, NI = 3, 4
    L = NI * 3
    item = torch.randint(0, 8, (B, L))
    ttype = (torch.arange(L) % 3).unsqueeze(0).expand(B, -1)
    tgt = torch.randint(0, 8, (B, 3))
    tgt_t = torch.arange(3).unsqueeze(0).expand(B, -1)
    mask = torch.ones(B, L, dtype=torch.long)
    out = m(torch.zeros(B, 1, dtype=torch.long), item, ttype, tgt, tgt_t, mask)
    assert out.logits.shape == (B, 4, 8 * 3 + 1)
    targets = tgt_t * 8 + tgt
    ref = F.cross_entropy(out.logits[:, :-1].reshape(-1, 25),
                          targets.reshape(-1), reduction="none"
                          ).reshape(B, 3).sum(1).mean()
    assert torch.allclose(out.loss, ref, atol=1e-5)


def test_tiger_generate_valid_unique_sorted():
    torch.manual_seed(1)
    m = Tiger(embedding_dim=16, attn_dim=24, dropout=0.0, num_heads=4,
              n_layers=2, num_item_embeddings=16, num_user_embeddings=10,
              sem_id_dim=3)
    m.eval()
    B, NI, K = 4, 4, 5
    L = NI * 3
    item = torch.randint(0, 16, (B, L))
    ttype = (torch.arange(L) % 3).unsqueeze(0).expand(B, -1)
    mask = torch.ones(B, L, dtype=torch.long)
    valid = torch.randint(0, 16, (60, 3))
    gen = m.generate(torch.zeros(B, 1, dtype=torch.long), item, ttype, mask,
                     n_top_k_candidates=K, valid_item_ids=valid)
    vs = set(map(tuple, valid.tolist()))
    for b in range(B):
        seqs = [tuple(gen.sem_ids[b, k].tolist()) for k in range(K)]
        scores = gen.log_probas[b].tolist()
        assert scores == sorted(scores, reverse=True)
        real = [s for s, sc in zip(seqs, scores) if sc > -1e30]
        assert all(s in vs for s in real)
        assert len(set(real)) == len(real)  # deduped


def test_checkpoint_roundtrip(tmp_path):
    from genrec_amd.trainers import common

    m = SASRec(num_items=20, embed_dim=8, num_heads=2, num_blocks=1,
               ffn_dim=8, dropout=0.0)
    opt = torch.optim.Adam(m.parameters())
    path = str(tmp_path / "ck.pt")
    common.save_checkpoint(path, m, opt, None, epoch=3,
                           model_config={"a": 1})
    m2 = SASRec(num_items=20, embed_dim=8, num_heads=2, num_blocks=1,
                ffn_dim=8, dropout=0.0)
    state = common.load_checkpoint(path, m2, torch.optim.Adam(m2.parameters()))
    assert state["epoch"] == 3 and state["model_config"] == {"a": 1}
    for p1, p2 in zip(m.parameters(), m2.parameters()):
        assert torch.equal(p1, p2)


def test_rqvae_reference_checkpoint_layout(tmp_path):
    """RqVae.load_pretrained reads the reference's dict layout
    (rqvae_trainer.py:315-324)."""
    m = RqVae(input_dim=24, embed_dim=8, hidden_dims=[16], codebook_size=12,
              codebook_mode=QuantizeForwardMode.STE,
              codebook_last_layer_mode=QuantizeForwardMode.STE,
              n_layers=2, n_cat_features=0)
    path = str(tmp_path / "rq.pt")
    torch.save({"epoch": 7, "model": m.state_dict(),
                "model_config": m.config}, path)
    m2 = RqVae(input_dim=24, embed_dim=8, hidden_dims=[16], codebook_size=12,
               codebook_mode=QuantizeForwardMode.STE,
               codebook_last_layer_mode=QuantizeForwardMode.STE,
               n_layers=2, n_cat_features=0)
    m2.load_pretrained(path)
    for p1, p2 in zip(m.parameters(), m2.parameters()):
        assert torch.equal(p1, p2)


def test_tiger_load_pretrained_both_formats(tmp_path):
    """safetensors (ref tiger.py:248-253) AND dict checkpoints load."""
    import os

    from safetensors.torch import save_file

    from genrec_amd.models.tiger import Tiger

    torch.manual_seed(0)
    kw = dict(embedding_dim=16, attn_dim=32, dropout=0.0, num_heads=2,
              n_layers=1, num_item_embeddings=8, num_user_embeddings=50,
              sem_id_dim=3)
    m1 = Tiger(**kw)
    st_dir = tmp_path / "hf"
    os.makedirs(st_dir)
    save_file({k: v.contiguous() for k, v in m1.state_dict().items()},
              str(st_dir / "model.safetensors"))
    torch.save({"model": m1.state_dict(), "epoch": 0},
               str(tmp_path / "ck.pt"))

    m2 = Tiger(**kw)
    m2.load_pretrained(str(st_dir))
    m3 = Tiger(**kw)
    m3.load_pretrained(str(tmp_path / "ck.pt"))
    for a, b, c in zip(m1.parameters(), m2.parameters(), m3.parameters()):
        assert torch.equal(a, b) and torch.equal(a, c)


def test_rotation_trick_identity_invariant():
    """When the input already equals the codeword, the rotation-trick
    transform is the identity (Householder reflection of u onto q with
    u == q) — and gradients flow through x (unlike STE)."""
    from genrec_amd.models.rqvae import efficient_rotation_trick_transform

    torch.manual_seed(0)
    x = torch.randn(6, 8, requires_grad=True)
    xn = x / x.norm(dim=-1, keepdim=True)
    out = efficient_rotation_trick_transform(xn.detach(), xn.detach(), x)
    assert torch.allclose(out, x, atol=1e-5)
    out.sum().backward()
    assert x.grad is not None and x.grad.abs().sum() > 0


def test_hstu_temporal_bucket_formula():
    """log2-bucketed |ts_i - ts_j| with clamp (ref hstu.py:368-409):
    bucket = clamp(floor(ln(max(|dt|,1)) / ln 2), 0, n-1)."""
    import math

    from genrec_amd.models.hstu import TemporalBias

    tb = TemporalBias(num_heads=2, num_buckets=8)
    ts = torch.tensor([[0, 1, 10, 1000, 10**9]], dtype=torch.long)
    bias = tb(ts)  # [B, H, L, L]
    assert bias.shape == (1, 2, 5, 5)
    diff = (ts.unsqueeze(2) - ts.unsqueeze(1)).abs().clamp(min=1).float()
    buckets = (diff.log() / 0.693).long().clamp(0, 7)
    expect = tb.temporal_attention_bias(buckets).permute(0, 3, 1, 2)
    assert torch.allclose(bias, expect)
    # same timestamp -> bucket 0; huge gap -> clamped top bucket
    assert buckets[0, 0, 0] == 0 and buckets[0, 0, 4] == 7


def test_tiger_generate_kv_cache_equivalent():
    """KV-cached incremental beam decode == the full re-forward path
    (same RNG stream -> identical Gumbel draws; dropout off). Uses a
    dense trie (all codes valid) so no beams are NEG_INF-padded."""
    torch.manual_seed(7)
    m = Tiger(embedding_dim=16, attn_dim=24, dropout=0.0, num_heads=4,
              n_layers=2, num_item_embeddings=8, num_user_embeddings=10,
              sem_id_dim=3)
    m.eval()
    B, NI, K = 3, 4, 4
    L = NI * 3
    item = torch.randint(0, 8, (B, L))
    ttype = (torch.arange(L) % 3).unsqueeze(0).expand(B, -1)
    mask = torch.ones(B, L, dtype=torch.long)
    valid = torch.cartesian_prod(*[torch.arange(8)] * 3)  # dense trie
    user = torch.zeros(B, 1, dtype=torch.long)

    torch.manual_seed(123)
    g1 = m.generate(user, item, ttype, mask, n_top_k_candidates=K,
                    valid_item_ids=valid, use_kv_cache=False)
    m.trie = None  # rebuild to keep paths independent
    torch.manual_seed(123)
    g2 = m.generate(user, item, ttype, mask, n_top_k_candidates=K,
                    valid_item_ids=valid, use_kv_cache=True)
    assert torch.equal(g1.sem_ids, g2.sem_ids)
    assert torch.allclose(g1.log_probas, g2.log_probas, atol=1e-5)
