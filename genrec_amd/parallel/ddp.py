"""Data-parallel engine: RCCL over xGMI.

The reference delegates DDP to HuggingFace accelerate (a thin wrapper over
torch DDP/NCCL). Here the engine is built directly on torch.distributed's
ProcessGroup (backend "nccl" IS RCCL on ROCm) with an explicit bucketed
gradient all-reducer designed for the xGMI topology:

  * one process per GPU; LOCAL_RANK pins the HIP device
  * gradients are packed into flat fp32/bf16 buckets in reverse parameter
    order (grads become ready back-to-front during backward) and
    all-reduced on a dedicated HIP stream, overlapping with the rest of
    backward (C1 in SURVEY.md §2.5)
  * MI355X xGMI is 7 point-to-point links per GPU (~153 GB/s each): ring
    collectives are per-link bound, so buckets are sized LARGE (default
    50 MB; LCRec-scale models use >=100 MB) to stay bandwidth-bound and
    let RCCL spread rings across links
  * tiny metric reductions are batched into one tensor (C2)

On CPU (tests) the same engine runs on the gloo backend; the comm stream
degenerates to synchronous all-reduce.
"""

from __future__ import annotations

import datetime
import os
from typing import Dict, List, Optional

import torch
import torch.distributed as dist
from torch import Tensor, nn


class DistributedContext:
    def __init__(self, rank: int, world_size: int, local_rank: int,
                 device: torch.device, backend: str):
        self.rank = rank
        self.world_size = world_size
        self.local_rank = local_rank
        self.device = device
        self.backend = backend

    @property
    def is_main(self) -> bool:
        return self.rank == 0

    def barrier(self) -> None:
        if self.world_size > 1:
            dist.barrier()


def init_distributed(backend: Optional[str] = None,
                     timeout_s: int = 1800) -> DistributedContext:
    """Initialize from torchrun env vars; single-process fallback."""
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_gpu = torch.cuda.is_available()
    if backend is None:
        backend = "nccl" if use_gpu else "gloo"
    if use_gpu:
        torch.cuda.set_device(local_rank)
        device = torch.device("cuda", local_rank)
    else:
        device = torch.device("cpu")
    if world > 1 and not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world,
            timeout=datetime.timedelta(seconds=timeout_s))
    return DistributedContext(rank, world, local_rank, device, backend)


def broadcast_parameters(model: nn.Module, src: int = 0) -> None:
    if not (dist.is_available() and dist.is_initialized()):
        return
    with torch.no_grad():
        for p in model.state_dict().values():
            if isinstance(p, Tensor):
                dist.broadcast(p, src=src)


class GradReducer:
    """Bucketed, overlapped gradient all-reduce.

    Usage per step:
        loss.backward()        # hooks fire as grads are accumulated
        reducer.finalize()     # wait for comms, unpack averaged grads
        optimizer.step()
    Call reducer.prepare() after zero_grad (or rely on auto re-arm).

    Set ``skip_sync`` True during gradient-accumulation micro-steps.
    """

    def __init__(self, model: nn.Module, bucket_cap_mb: float = 50.0,
                 process_group=None):
        self.model = model
        self.group = process_group
        self.world = dist.get_world_size(process_group) if (
            dist.is_available() and dist.is_initialized()) else 1
        self.skip_sync = False
        self._params: List[Tensor] = [
            p for p in model.parameters() if p.requires_grad]
        # reverse order: autograd produces grads roughly back-to-front
        self._params = self._params[::-1]
        cap = int(bucket_cap_mb * 1024 * 1024)
        self._buckets: List[List[Tensor]] = []
        cur, cur_bytes = [], 0
        for p in self._params:
            nbytes = p.numel() * p.element_size()
            if cur and cur_bytes + nbytes > cap:
                self._buckets.append(cur)
                cur, cur_bytes = [], 0
            cur.append(p)
            cur_bytes += nbytes
        if cur:
            self._buckets.append(cur)
        self._bucket_of: Dict[int, int] = {}
        for bi, ps in enumerate(self._buckets):
            for p in ps:
                self._bucket_of[id(p)] = bi
        self._pending = [0] * len(self._buckets)
        self._flat: List[Optional[Tensor]] = [None] * len(self._buckets)
        self._works: List = []
        self._comm_stream = (torch.cuda.Stream()
                             if torch.cuda.is_available() else None)
        self._hooks = []
        if self.world > 1:
            for p in self._params:
                self._hooks.append(p.register_post_accumulate_grad_hook(
                    self._on_grad_ready))
        self.prepare()

    def prepare(self) -> None:
        for bi, ps in enumerate(self._buckets):
            self._pending[bi] = len(ps)
        self._works = []
        self._next = 0

    def _on_grad_ready(self, p: Tensor) -> None:
        if self.world <= 1 or self.skip_sync:
            return
        bi = self._bucket_of[id(p)]
        self._pending[bi] -= 1
        self._maybe_launch()

    def _maybe_launch(self) -> None:
        # Launch strictly in bucket-index order: RCCL requires the same
        # collective sequence on every rank, and ready-order can diverge
        # when parameter usage is rank-dependent.
        while (self._next < len(self._buckets)
               and self._pending[self._next] == 0):
            self._launch(self._next)
            self._next += 1

    def _launch(self, bi: int) -> None:
        ps = self._buckets[bi]
        grads = [p.grad for p in ps]
        flat = torch._utils._flatten_dense_tensors(grads)
        if self._comm_stream is not None:
            # comm stream waits for grads produced on the compute stream
            self._comm_stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self._comm_stream):
                flat.record_stream(self._comm_stream)
                work = dist.all_reduce(flat, group=self.group, async_op=True)
        else:
            work = dist.all_reduce(flat, group=self.group, async_op=True)
        self._flat[bi] = flat
        self._works.append((bi, ps, work))

    def finalize(self) -> None:
        """Wait for all in-flight reductions; write averaged grads back.

        Buckets that never completed through the hooks (parameters the
        forward did not touch, e.g. TIGER's unused positional embeddings)
        are reduced here over the FULL bucket with zero grads substituted
        for untouched params — every rank contributes an identically
        shaped flat buffer even when parameter usage is rank-dependent
        (matching torch-DDP semantics: unused params end up with the
        cross-rank average, zero if unused everywhere).
        """
        if self.world <= 1 or self.skip_sync:
            return
        for bi in range(len(self._buckets)):
            if self._pending[bi] > 0:
                for p in self._buckets[bi]:
                    if p.grad is None:
                        p.grad = torch.zeros_like(p)
                self._pending[bi] = 0
        self._maybe_launch()
        for bi, ps, work in self._works:
            work.wait()
        if self._comm_stream is not None:
            torch.cuda.current_stream().wait_stream(self._comm_stream)
        inv = 1.0 / self.world
        for bi, ps, _ in self._works:
            flat = self._flat[bi]
            flat.mul_(inv)
            for p, g in zip(ps, torch._utils._unflatten_dense_tensors(
                    flat, [p.grad for p in ps])):
                p.grad.copy_(g)
            self._flat[bi] = None
        self.prepare()


def reduce_scalars(values: Dict[str, float], device,
                   op: str = "sum") -> Dict[str, float]:
    """Batch tiny metric reductions into ONE all-reduce (C2 in SURVEY §2.5)."""
    if not (dist.is_available() and dist.is_initialized()):
        return dict(values)
    keys = sorted(values.keys())
    t = torch.tensor([float(values[k]) for k in keys], dtype=torch.float64,
                     device=device)
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    if op == "mean":
        t /= dist.get_world_size()
    return {k: t[i].item() for i, k in enumerate(keys)}
