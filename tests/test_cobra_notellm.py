"""COBRA + NoteLLM CPU tests."""

import pytest
import torch

torch.manual_seed(0)

TINY_COBRA = dict(encoder_n_layers=1, encoder_hidden_dim=32,
                  encoder_num_heads=4, encoder_vocab_size=100,
                  id_vocab_size=16, n_codebooks=3, d_model=32,
                  decoder_n_layers=2, decoder_num_heads=4,
                  decoder_dropout=0.0)


@pytest.fixture(scope="module")
def cobra():
    from genrec_amd.models.cobra import Cobra

    torch.manual_seed(0)
    return Cobra(**TINY_COBRA)


def _batch(model, B=3, T=4, L=6):
    C = model.C
    ids = torch.randint(0, 16, (B, T * C))
    enc = torch.randint(1, 100, (B, T, L))
    return ids, enc


def test_cobra_interleave_mask(cobra):
    m = torch.tensor([[1, 1, 1, 1, 1, 1], [1, 1, 1, 0, 0, 0]]).bool()
    out = cobra.interleave_seq_mask(m, 3)
    # [s s s d s s s d]
    assert out.shape == (2, 8)
    assert out[0].tolist() == [True] * 8
    assert out[1].tolist() == [True, True, True, True, False, False, False,
                               False]


def test_cobra_forward_losses(cobra):
    ids, enc = _batch(cobra)
    out = cobra(ids, enc)
    assert torch.isfinite(out.loss)
    assert out.acc_total > 0 and out.recall_total > 0
    assert out.codebook_entropy > 0
    out.loss.backward()


def test_cobra_padded_rows(cobra):
    ids, enc = _batch(cobra, B=2, T=4)
    ids[1, 6:] = cobra.pad_id  # second row has only 2 complete items
    out = cobra(ids, enc)
    assert torch.isfinite(out.loss)


def test_cobra_generate_and_fusion(cobra):
    cobra.eval()
    ids, enc = _batch(cobra, B=2, T=3)
    gen = cobra.generate(ids, enc, n_candidates=4)
    assert gen.sem_ids.shape == (2, 4, 3)
    assert gen.dense_vecs.shape == (2, 4, 32)
    assert torch.allclose(gen.dense_vecs.norm(dim=-1),
                          torch.ones(2, 4), atol=1e-4)
    # scores sorted desc
    assert (gen.scores[:, :-1] >= gen.scores[:, 1:]).all()

    n_items = 20
    item_vecs = torch.nn.functional.normalize(torch.randn(n_items, 32), dim=-1)
    item_sem = torch.randint(0, 16, (n_items, 3))
    fus = cobra.beam_fusion(ids, enc, item_vecs, item_sem,
                            n_candidates=5, n_beam=8)
    assert fus.item_ids.shape == (2, 5)
    assert fus.sem_ids.shape == (2, 5, 3)
    assert (fus.item_ids < n_items).all()


def test_cobra_trainer_smoke(tmp_path):
    from genrec_amd.data.cobra_synthetic import SyntheticCobraDataset
    from genrec_amd.trainers import cobra_trainer

    class Tiny(SyntheticCobraDataset):
        def __init__(self, **kw):
            kw.update(num_users=40, num_items=50, text_vocab_size=100,
                      max_text_len=6)
            super().__init__(**kw)

    cobra_trainer.train(
        dataset=Tiny, epochs=1, max_steps=2, batch_size=8,
        n_codebooks=3, id_vocab_size=16, d_model=32, decoder_n_layers=2,
        decoder_num_heads=4, encoder_n_layers=1, amp=False,
        save_dir_root=str(tmp_path), eval_every_epoch=1, eval_n_beam=6,
        eval_max_batches=1, num_warmup_steps=1)
    import os

    assert os.path.exists(os.path.join(str(tmp_path), "checkpoint_final.pt"))


def test_notellm_contrastive():
    from genrec_amd.models.lcrec import default_qwen_config
    from genrec_amd.models.notellm import Query2Embedding

    torch.manual_seed(0)
    m = Query2Embedding(config=default_qwen_config(
        vocab_size=512, hidden_size=32, num_layers=2, num_heads=4,
        num_kv_heads=2, intermediate_size=64), gradient_checkpointing=False)
    queries = []
    for i in range(4):
        queries.append(f"note {i} text [EMB]")
        queries.append(f"related note {i} [EMB]")
    tok = m.tokenize(queries)
    out = m(tok["input_ids"], tok["attention_mask"], tok["emb_token_idx"])
    assert torch.isfinite(out["loss"])
    out["loss"].backward()
    assert m.tau.grad is not None
    emb = out["sentence_embedding"].detach()
    assert torch.allclose(emb.norm(dim=1), torch.ones(8), atol=1e-4)
    acc = Query2Embedding.topk_retrieval_accuracy(emb, topk=2, batch_size=4)
    assert 0.0 <= acc <= 1.0


def test_notellm_hardneg_and_labels():
    from genrec_amd.models.lcrec import default_qwen_config
    from genrec_amd.models.notellm import Query2Embedding

    torch.manual_seed(1)
    m = Query2Embedding(config=default_qwen_config(
        vocab_size=512, hidden_size=32, num_layers=2, num_heads=4,
        num_kv_heads=2, intermediate_size=64), gradient_checkpointing=False)
    queries = [f"q{i} [EMB]" for i in range(6)]
    tok = m.tokenize(queries, score=[0.5, 0.05, 0.9])
    assert tok["hardneg"].tolist() == [False, True, False]
    out = m(tok["input_ids"], tok["attention_mask"], tok["emb_token_idx"],
            hardneg=tok["hardneg"])
    assert torch.isfinite(out["loss"])


def test_notellm_trainer_smoke(tmp_path):
    """End-to-end NoteLLM contrastive trainer on a tiny backbone
    (beyond-reference: the reference ships no NoteLLM trainer)."""
    from genrec_amd.data.notellm_synthetic import SyntheticNotePairDataset
    from genrec_amd.trainers import notellm_trainer

    class Tiny(SyntheticNotePairDataset):
        def __init__(self, **kw):
            kw.update(num_pairs=24)
            super().__init__(**kw)

    notellm_trainer.train(
        epochs=1, max_steps=2, batch_size=4, dataset=Tiny,
        backbone_config=dict(vocab_size=512, hidden_size=32, num_layers=1,
                             num_heads=4, num_kv_heads=2,
                             intermediate_size=64),
        gradient_checkpointing=False, do_eval=True, eval_max_batches=2,
        save_dir_root=str(tmp_path), save_every_epoch=1, num_workers=0)
    import os

    assert os.path.isdir(os.path.join(str(tmp_path), "epoch_0"))


def test_cobra_decoder_matches_torch():
    """Native CobraDecoder (fused/flash attention path) == torch's
    nn.TransformerDecoder with zero-length memory, weights copied over
    (reference decoder semantics, cobra.py:150-224). Padded query rows
    are excluded: torch emits NaN there, we zero them."""
    import torch.nn as tnn
    from genrec_amd.models.cobra import CobraDecoder

    torch.manual_seed(0)
    d, heads, ff, L, B = 48, 4, 96, 9, 3
    native = CobraDecoder(hidden_dim=d, n_layers=2, n_heads=heads,
                          ff_dim=ff, dropout=0.0)
    layer = tnn.TransformerDecoderLayer(d_model=d, nhead=heads,
                                        dim_feedforward=ff, dropout=0.0,
                                        batch_first=True)
    ref = tnn.TransformerDecoder(layer, num_layers=2)
    with torch.no_grad():
        for nl, rl in zip(native.layers, ref.layers):
            nl.qkv.weight.copy_(rl.self_attn.in_proj_weight)
            nl.qkv.bias.copy_(rl.self_attn.in_proj_bias)
            nl.out.weight.copy_(rl.self_attn.out_proj.weight)
            nl.out.bias.copy_(rl.self_attn.out_proj.bias)
            nl.linear1.weight.copy_(rl.linear1.weight)
            nl.linear1.bias.copy_(rl.linear1.bias)
            nl.linear2.weight.copy_(rl.linear2.weight)
            nl.linear2.bias.copy_(rl.linear2.bias)
            nl.norm1.weight.copy_(rl.norm1.weight)
            nl.norm1.bias.copy_(rl.norm1.bias)
            nl.norm2.weight.copy_(rl.norm2.weight)
            nl.norm2.bias.copy_(rl.norm2.bias)
            nl.norm3.weight.copy_(rl.norm3.weight)
            nl.norm3.bias.copy_(rl.norm3.bias)
    native.eval()
    ref.eval()
    x = torch.randn(B, L, d)
    pad = torch.zeros(B, L, dtype=torch.bool)
    pad[:, -2:] = True
    causal = torch.triu(torch.ones(L, L, dtype=torch.bool), 1)
    mem = torch.zeros(B, 0, d)
    out_ref = ref(x, mem, tgt_mask=causal, tgt_key_padding_mask=pad)
    out_nat = native(x, tgt_key_padding_mask=pad)
    valid = ~pad
    assert torch.allclose(out_nat[valid], out_ref[valid], atol=1e-5), \
        (out_nat[valid] - out_ref[valid]).abs().max()


def test_cobra_static_infonce_equivalent(cobra):
    """Graph-capturable fixed-shape InfoNCE (mask-to--1e4) == the
    filtered reference formulation, including padded rows."""
    torch.manual_seed(3)
    m = cobra
    ids, enc = _batch(m, B=3, T=5)
    ids[1, 9:] = m.pad_id  # pad the tail of one sequence
    m.eval()
    out_dyn = m(ids, enc)
    m.static_infonce = True
    try:
        out_sta = m(ids, enc)
    finally:
        m.static_infonce = False
    assert torch.allclose(out_sta.loss_dense, out_dyn.loss_dense,
                          atol=1e-5), (out_sta.loss_dense,
                                       out_dyn.loss_dense)
    assert torch.allclose(out_sta.vec_cos_sim, out_dyn.vec_cos_sim,
                          atol=1e-5)
    assert torch.allclose(out_sta.loss, out_dyn.loss, atol=1e-5)


def test_capture_safe_encoder_layer_matches_torch():
    """CaptureSafeEncoderLayer == nn.TransformerEncoderLayer (post-LN,
    relu) with weights copied via the back-compat state-dict mapping."""
    import torch.nn as tnn
    from genrec_amd.modules.encoders import CaptureSafeEncoderLayer

    torch.manual_seed(0)
    d, heads, ff, B, L = 48, 4, 96, 3, 9
    ref = tnn.TransformerEncoderLayer(d_model=d, nhead=heads,
                                      dim_feedforward=ff, dropout=0.0,
                                      batch_first=True)
    ours = CaptureSafeEncoderLayer(d, heads, ff, dropout=0.0)
    ours.load_state_dict(ref.state_dict())  # via _load_from_state_dict
    ref.eval()
    ours.eval()
    x = torch.randn(B, L, d)
    pad = torch.zeros(B, L, dtype=torch.bool)
    pad[:, -2:] = True
    o_ref = ref(x, src_key_padding_mask=pad)
    o_our = ours(x, src_key_padding_mask=pad)
    valid = ~pad
    assert torch.allclose(o_our[valid], o_ref[valid], atol=1e-5), \
        (o_our[valid] - o_ref[valid]).abs().max()
