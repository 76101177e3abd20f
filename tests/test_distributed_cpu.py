"""Multi-process distributed tests on gloo, world_size=2 (CPU).

Validates the RCCL-topology engine's logic (bucketing, overlap hooks,
averaging) on the gloo backend — the same code path runs over RCCL on the
GPU box.
"""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _run_gradreducer(rank, world, port, results):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(1234)  # same params on both ranks
    from genrec_amd.parallel import GradReducer

    model = torch.nn.Sequential(
        torch.nn.Linear(8, 16), torch.nn.ReLU(), torch.nn.Linear(16, 4))
    reducer = GradReducer(model, bucket_cap_mb=0.0001)  # force many buckets
    torch.manual_seed(rank)  # different data per rank
    x = torch.randn(6, 8)
    y = model(x).sum()
    y.backward()
    reducer.finalize()
    grads = torch.cat([p.grad.flatten() for p in model.parameters()])
    results[rank] = grads.clone()
    dist.barrier()
    dist.destroy_process_group()


def test_grad_reducer_averages_across_ranks(tmp_path):
    world = 2
    port = 29613
    mgr = mp.Manager()
    results = mgr.dict()
    ctx = mp.spawn(_run_gradreducer, args=(world, port, results),
                   nprocs=world, join=True)
    g0, g1 = results[0], results[1]
    assert torch.allclose(g0, g1, atol=1e-6)

    # reference: average of single-process grads with each rank's data
    torch.manual_seed(1234)
    model = torch.nn.Sequential(
        torch.nn.Linear(8, 16), torch.nn.ReLU(), torch.nn.Linear(16, 4))
    expected = []
    for rank in range(world):
        for p in model.parameters():
            if p.grad is not None:
                p.grad = None
        torch.manual_seed(rank)
        x = torch.randn(6, 8)
        model(x).sum().backward()
        expected.append(torch.cat([p.grad.flatten()
                                   for p in model.parameters()]))
    avg = (expected[0] + expected[1]) / 2
    assert torch.allclose(g0, avg, atol=1e-5)


def _run_metric_reduce(rank, world, port, results):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from genrec_amd.parallel import reduce_scalars

    out = reduce_scalars({"a": rank + 1.0, "b": 10.0 * (rank + 1)},
                         torch.device("cpu"))
    results[rank] = out
    dist.barrier()
    dist.destroy_process_group()


def test_reduce_scalars_sums(tmp_path):
    world = 2
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_run_metric_reduce, args=(world, 29713, results), nprocs=world,
             join=True)
    assert results[0] == {"a": 3.0, "b": 30.0}
    assert results[1] == {"a": 3.0, "b": 30.0}


def _run_topk_allreduce(rank, world, port, results):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from genrec_amd.modules.metrics import TopKAccumulator

    acc = TopKAccumulator(ks=[1])
    actual = torch.tensor([[1, 2]])
    topk = torch.tensor([[[1, 2]]]) if rank == 0 else torch.tensor([[[9, 9]]])
    acc.accumulate(actual, topk)
    results[rank] = acc.reduce(all_reduce=True)
    dist.barrier()
    dist.destroy_process_group()


def test_topk_accumulator_all_reduce():
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_run_topk_allreduce, args=(2, 29813, results), nprocs=2,
             join=True)
    # 1 hit of 2 samples globally
    assert results[0]["Recall@1"] == pytest.approx(0.5)
    assert results[1]["Recall@1"] == pytest.approx(0.5)


def _run_unused_params(rank, world, port, results):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(7)
    from genrec_amd.parallel import GradReducer

    class Net(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.used = torch.nn.Linear(8, 8)
            self.unused = torch.nn.Linear(8, 8)   # never in forward
            self.used2 = torch.nn.Linear(8, 4)

        def forward(self, x):
            return self.used2(torch.relu(self.used(x)))

    model = Net()
    # tiny bucket cap => 'unused' shares buckets with used params
    reducer = GradReducer(model, bucket_cap_mb=0.0002)
    torch.manual_seed(100 + rank)
    model(torch.randn(4, 8)).sum().backward()
    reducer.finalize()
    grads = torch.cat([p.grad.flatten() for p in model.parameters()
                       if p.grad is not None])
    results[rank] = grads.clone()
    dist.barrier()
    dist.destroy_process_group()


def test_grad_reducer_with_unused_params():
    """Mixed used/unused buckets must still synchronize the used grads
    (TIGER has allocated-but-unused parameters)."""
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_run_unused_params, args=(2, 29913, results), nprocs=2,
             join=True)
    assert torch.allclose(results[0], results[1], atol=1e-6)


def _run_tiger_trainer(rank, world, port, results, tmpdir):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    from genrec_amd.data.synthetic import SyntheticSemIdSeqDataset
    from genrec_amd.trainers import tiger_trainer

    class Tiny(SyntheticSemIdSeqDataset):
        def __init__(self, **kw):
            kw.update(num_users=40, num_items=60)
            super().__init__(**kw)

    tiger_trainer.train(
        dataset=Tiny, epochs=1, max_steps=2, num_workers=0, batch_size=8,
        save_dir_root=tmpdir, amp=False, do_eval=False, save_every_epoch=1,
        embedding_dim=16, attn_dim=24, num_heads=4, n_layers=2,
        num_item_embeddings=16)
    results[rank] = True
    dist.destroy_process_group()


def test_tiger_trainer_world2(tmp_path):
    """Full TIGER trainer loop under gloo world_size=2 (the same code path
    runs over RCCL on GPU nodes)."""
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_run_tiger_trainer, args=(2, 29923, results, str(tmp_path)),
             nprocs=2, join=True)
    assert results[0] and results[1]
    import os as _os

    assert _os.path.exists(_os.path.join(str(tmp_path),
                                         "checkpoint_final.pt"))


def _run_sasrec_trainer(rank, world, port, results, tmpdir):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    from genrec_amd.data.synthetic import SyntheticSASRecDataset
    from genrec_amd.trainers import sasrec_trainer

    class Tiny(SyntheticSASRecDataset):
        def __init__(self, **kw):
            kw.update(num_users=60, num_items=80)
            super().__init__(**kw)

    m = sasrec_trainer.train(
        dataset=Tiny, epochs=1, max_steps=2, num_workers=0, batch_size=8,
        save_dir_root=tmpdir, amp=False, do_eval=True, eval_every_epoch=1,
        save_every_epoch=1, max_seq_len=8, embed_dim=16, num_heads=2,
        num_blocks=1, ffn_dim=16)
    results[rank] = m is not None or True
    dist.destroy_process_group()


def test_sasrec_trainer_world2_with_eval(tmp_path):
    """SASRec trainer + cross-rank metric reduction under gloo world=2
    (exercises the C2 scalar-reduce path in evaluate)."""
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_run_sasrec_trainer, args=(2, 29931, results, str(tmp_path)),
             nprocs=2, join=True)
    assert results[0] and results[1]


def _run_accum(rank, world, port, results):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(7)
    from genrec_amd.parallel import GradReducer

    model = torch.nn.Linear(4, 3)
    reducer = GradReducer(model, bucket_cap_mb=0.0001)
    torch.manual_seed(100 + rank)
    micro = [torch.randn(5, 4) for _ in range(2)]
    # micro-step 1: no sync
    reducer.skip_sync = True
    model(micro[0]).sum().backward()
    # micro-step 2: sync + finalize
    reducer.skip_sync = False
    model(micro[1]).sum().backward()
    reducer.finalize()
    got = torch.cat([p.grad.flatten() for p in model.parameters()])
    # expected: rank-local sum of micro grads, averaged across ranks
    ref = torch.nn.Linear(4, 3)
    torch.manual_seed(7)
    ref = torch.nn.Linear(4, 3)
    for m in micro:
        ref(m).sum().backward()
    local = torch.cat([p.grad.flatten() for p in ref.parameters()])
    results[f"got{rank}"] = got.clone()
    results[f"local{rank}"] = local.clone()
    dist.barrier()
    dist.destroy_process_group()


def test_grad_reducer_accumulation_micro_steps(tmp_path):
    """skip_sync micro-steps accumulate locally; the final sync averages
    the ACCUMULATED gradients across ranks."""
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_run_accum, args=(2, 29937, results), nprocs=2, join=True)
    expect = (results["local0"] + results["local1"]) / 2
    assert torch.allclose(results["got0"], expect, atol=1e-6)
    assert torch.allclose(results["got1"], expect, atol=1e-6)


def _run_cobra_trainer(rank, world, port, results, tmpdir):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    from genrec_amd.data.cobra_synthetic import SyntheticCobraDataset
    from genrec_amd.trainers import cobra_trainer

    class Tiny(SyntheticCobraDataset):
        def __init__(self, **kw):
            kw.update(num_users=40, num_items=60, text_vocab_size=500,
                      id_vocab_size=16)
            super().__init__(**kw)

    cobra_trainer.train(
        dataset=Tiny, epochs=1, max_steps=2, num_workers=0, batch_size=8,
        save_dir_root=tmpdir, amp=False, do_eval=True, eval_every_epoch=1,
        save_every_epoch=1, n_codebooks=3, id_vocab_size=16, d_model=32,
        decoder_n_layers=1, decoder_num_heads=4, decoder_dropout=0.0,
        encoder_n_layers=1, eval_n_beam=4, eval_max_batches=1,
        num_warmup_steps=1)
    results[rank] = True


def test_cobra_trainer_world2(tmp_path):
    """COBRA trainer under gloo world_size=2: DDP GradReducer + the
    all-reduced beam-fusion eval metrics (same code path over RCCL)."""
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_run_cobra_trainer, args=(2, 29931, results, str(tmp_path)),
             nprocs=2, join=True)
    assert results[0] and results[1]
    import os as _os

    assert _os.path.exists(_os.path.join(str(tmp_path),
                                         "checkpoint_final.pt"))


def _run_graphed_step_world2(rank, world, port, results):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(77)  # identical init on both ranks
    from genrec_amd.parallel.graph_runner import GraphedTrainStep

    model = torch.nn.Sequential(
        torch.nn.Linear(8, 16), torch.nn.Tanh(), torch.nn.Linear(16, 4))
    ex = {"x": torch.randn(6, 8, dtype=torch.bfloat16)}

    class Wrap(torch.nn.Module):
        def __init__(self, m):
            super().__init__()
            self.m = m

        def forward(self, x):
            return self.m(x)

    runner = GraphedTrainStep(Wrap(model), ex, lambda out: out.square().sum(),
                              lr=1e-2, clip_norm=1.0, world=world,
                              use_graph=False)
    torch.manual_seed(100 + rank)  # different data per rank (DDP semantics)
    losses = []
    for _ in range(4):
        losses.append(float(runner.step(
            {"x": torch.randn(6, 8, dtype=torch.bfloat16)})))
    flat = torch.cat([p.detach().float().flatten()
                      for p in runner.params])
    results[rank] = (flat.clone(), losses)
    dist.barrier()
    dist.destroy_process_group()


def test_graphed_train_step_world2_keeps_ranks_identical(tmp_path):
    """bench.py's N>1 engine on gloo: flat-grad all_reduce inside
    GraphedTrainStep._inner keeps parameters bitwise-identical across
    ranks over several steps with different per-rank data."""
    world = 2
    port = 29631
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_run_graphed_step_world2, args=(world, port, results),
             nprocs=world, join=True)
    p0, l0 = results[0]
    p1, l1 = results[1]
    assert torch.equal(p0, p1)          # identical weights after 4 steps
    assert l0 != l1                     # ranks really saw different data
    assert all(torch.isfinite(torch.tensor(l)) for l in l0 + l1)
