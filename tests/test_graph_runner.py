"""GraphedTrainStep: CPU fallback path (capture itself is GPU-only and
covered by tests/test_kernels_gpu.py::test_graph_step_equals_eager_step)."""

import torch

from genrec_amd.models.tiger import Tiger
from genrec_amd.parallel import GraphedTrainStep


def _tiny_model():
    torch.manual_seed(0)
    return Tiger(embedding_dim=16, attn_dim=32, dropout=0.0, num_heads=2,
                 n_layers=1, num_item_embeddings=16,
                 num_user_embeddings=100, sem_id_dim=3)


def _batch(B=4, T=6, sem=3):
    torch.manual_seed(1)
    return {
        "user_input_ids": torch.randint(0, 100, (B, 1)),
        "item_input_ids": torch.randint(0, 16, (B, T * sem)),
        "token_type_ids": torch.arange(T * sem).remainder(sem).repeat(B, 1),
        "target_input_ids": torch.randint(0, 16, (B, sem)),
        "target_token_type_ids": torch.arange(sem).repeat(B, 1),
        "seq_mask": torch.ones(B, T * sem, dtype=torch.long),
    }


def test_graphed_train_step_cpu_fallback_trains():
    model = _tiny_model()
    runner = GraphedTrainStep(model, _batch(), lambda o: o.loss,
                              lr=1e-2, use_graph=False)
    assert not runner.captured
    losses = []
    for i in range(8):
        runner.set_lr(1e-2 * (i + 1) / 8)
        losses.append(runner.step(_batch()).item())
    assert all(torch.isfinite(torch.tensor(losses)))
    assert losses[-1] < losses[0]  # it actually optimizes
    assert runner.opt.param_groups[0]["lr"] == 1e-2


def test_graphed_train_step_state_roundtrip():
    model = _tiny_model()
    runner = GraphedTrainStep(model, _batch(), lambda o: o.loss,
                              lr=1e-3, use_graph=False)
    for _ in range(3):
        runner.step(_batch())
    state = runner.state_dict()

    model2 = _tiny_model()
    runner2 = GraphedTrainStep(model2, _batch(), lambda o: o.loss,
                               lr=1e-3, use_graph=False)
    runner2.load_state_dict(state)
    for m, n in zip(runner.masters, runner2.masters):
        assert torch.equal(m, n)
    l1 = runner.step(_batch()).item()
    l2 = runner2.step(_batch()).item()
    assert abs(l1 - l2) < 1e-3


def test_trainer_use_hip_graph_flag_is_cpu_safe(tmp_path):
    # on CPU the flag must be a no-op (graph mode requires CUDA)
    from genrec_amd.trainers import tiger_trainer

    tiger_trainer.train(
        epochs=1, max_steps=2, batch_size=8, embedding_dim=16, attn_dim=32,
        num_heads=2, n_layers=1, num_item_embeddings=16, sem_id_dim=3,
        max_seq_len=4, num_warmup_steps=2, do_eval=False, amp=False,
        use_hip_graph=True, num_workers=0,
        save_dir_root=str(tmp_path), wandb_logging=False,
        dataset=None, save_every_epoch=100)


def test_sasrec_hstu_use_hip_graph_flag_cpu_safe(tmp_path):
    # on CPU the flag is a no-op (graph mode requires CUDA); the fixed-
    # length collate path still produces correct shapes
    from genrec_amd.data.collate import hstu_collate_fn, sasrec_collate_fn
    from genrec_amd.trainers import hstu_trainer, sasrec_trainer

    b = [{"history": [1, 2], "target": 3, "timestamps": [10, 20]}]
    out = sasrec_collate_fn(b, max_seq_len=6, fixed_length=True)
    assert out["input_ids"].shape == (1, 6)
    out = hstu_collate_fn(b, max_seq_len=6, fixed_length=True)
    assert out["timestamps"].shape == (1, 6)

    common = dict(epochs=1, max_steps=2, batch_size=8, max_seq_len=6,
                  embed_dim=16, num_heads=2, num_blocks=1, do_eval=False,
                  amp=False, use_hip_graph=True, num_workers=0,
                  save_dir_root=str(tmp_path), wandb_logging=False,
                  save_every_epoch=100)
    sasrec_trainer.train(ffn_dim=16, **common)
    hstu_trainer.train(**common)


def test_tiger_fixed_length_collate_guard():
    import pytest

    from genrec_amd.data.collate import tiger_pad_collate
    from genrec_amd.data.schemas import SeqData

    b = [SeqData(0, [1] * 12, [1, 2, 3])]
    out = tiger_pad_collate(b, sem_id_dim=3, fixed_length=15)
    assert out["item_input_ids"].shape == (1, 15)
    with pytest.raises(AssertionError, match="exceeds"):
        tiger_pad_collate(b, sem_id_dim=3, fixed_length=9)


def test_fused_checkpoint_loads_into_cpu_runner():
    """A fused-path (GPU) runner checkpoint restores into the CPU
    fallback runner (cross-device resume)."""
    import torch

    model = _tiny_model()
    r = GraphedTrainStep(model, _batch(), lambda o: o.loss, use_graph=False)
    n = sum(p.numel() for p in r.params)
    fake = {"flat_master": torch.arange(n, dtype=torch.float32) * 1e-4,
            "m": torch.zeros(n), "v": torch.zeros(n),
            "step": torch.zeros(1, dtype=torch.int32), "lr": 5e-4}
    r.load_state_dict(fake)
    flat = torch.cat([m.reshape(-1) for m in r.masters])
    assert torch.allclose(flat, fake["flat_master"])
    assert r.opt.param_groups[0]["lr"] == 5e-4


def test_graphed_step_cobra_cpu():
    """GraphedTrainStep drives the COBRA model (static-InfoNCE mode) —
    the bf16-conversion path must keep every intermediate bf16 on CPU
    too (fp32 mask promotions broke the pure-bf16 runner in round 2)."""
    import torch

    from genrec_amd.data.cobra_synthetic import (SyntheticCobraDataset,
                                                 cobra_collate_fn)
    from genrec_amd.models.cobra import Cobra
    from genrec_amd.parallel.graph_runner import GraphedTrainStep

    torch.manual_seed(0)
    ds = SyntheticCobraDataset(num_users=30, num_items=50, split="train",
                               n_codebooks=3, id_vocab_size=16,
                               text_vocab_size=1000)
    m = Cobra(encoder_n_layers=1, encoder_hidden_dim=32,
              encoder_num_heads=4, encoder_vocab_size=1000,
              id_vocab_size=16, n_codebooks=3, d_model=32,
              decoder_n_layers=2, decoder_num_heads=4, decoder_dropout=0.0)
    m.static_infonce = True
    batch = cobra_collate_fn([ds[i] for i in range(4)], m.pad_id, 3,
                             train=True,
                             fixed_items=ds.max_items_per_seq + 1)
    ex = {"input_ids": batch["input_ids"],
          "encoder_input_ids": batch["encoder_input_ids"]}
    runner = GraphedTrainStep(
        m, ex, loss_getter=lambda out: out.loss_sparse + out.loss_dense,
        lr=1e-3, weight_decay=0.0, clip_norm=1.0, world=1, use_graph=False)
    losses = [float(runner.step(ex).detach()) for _ in range(3)]
    assert losses[-1] < losses[0]  # optimizes
