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
