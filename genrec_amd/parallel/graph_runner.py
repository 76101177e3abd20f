"""hipGraph-captured training step (shared by bench.py and the trainers).

Captures one full training step — forward, backward, flat-gradient RCCL
all-reduce, global-norm clip, fused flat AdamW — and replays it per
batch. Requirements/properties:

  * the model runs natively in bf16 (fp32 master weights live in ONE flat
    buffer); no autocast, so no per-layer weight-cast kernels
  * every batch must have the SAME shapes (fixed-size collate + drop_last)
  * gradients are views into one flat bf16 buffer -> N>1 all-reduce is a
    single large RCCL message
  * on CUDA the optimizer is the genrec fused_adamw kernel
    (csrc/kernels/adamw.hip): masters/moments/params/grads are flat
    tensors and the whole AdamW step is ONE bandwidth-bound kernel.
    torch's capturable foreach AdamW with a tensor lr falls off the
    foreach fast route and launches ~4 tiny kernels per parameter per
    step (~1.5 ms/step measured on TIGER)
  * lr / clip-scale / step count are DEVICE scalars read by the kernel,
    so LR schedules (set_lr) and clipping keep working across replays
  * dropout stays live across replays (torch philox is graph-safe; the
    genrec_amd fused kernels read a captured device seed counter)
  * on ANY capture failure the same step runs eagerly — numerics are
    identical either way (tests/test_kernels_gpu.py::
    test_graph_step_equals_eager_step)
"""

from __future__ import annotations

import logging
from typing import Callable, Dict, Optional

import torch

logger = logging.getLogger("genrec_amd")


class GraphedTrainStep:
    def __init__(self, model: torch.nn.Module,
                 example_batch: Dict[str, torch.Tensor],
                 loss_getter: Callable,
                 lr: float = 1e-4, weight_decay: float = 0.0,
                 betas=(0.9, 0.999), eps: float = 1e-8,
                 clip_norm: Optional[float] = 1.0,
                 world: int = 1, use_graph: bool = True,
                 warmup_iters: int = 3):
        self.model = model.to(torch.bfloat16)
        self.model.train()
        self.world = world
        self.loss_getter = loss_getter
        self.clip_norm = clip_norm
        self.betas = betas
        self.eps = eps
        self.weight_decay = weight_decay
        device = next(model.parameters()).device
        self.device = device

        self.params = [p for p in model.parameters() if p.requires_grad]
        n_total = sum(p.numel() for p in self.params)
        self.lr_t = torch.tensor(float(lr), device=device,
                                 dtype=torch.float32)

        from genrec_amd import ops

        self.fused = device.type == "cuda" and ops.has_ext()
        if self.fused:
            # flat bf16 params (p.data re-pointed into views) + flat fp32
            # masters/moments: the fused AdamW kernel updates everything
            # in one pass and writes the bf16 params directly.
            self.flat_params = torch.empty(n_total, device=device,
                                           dtype=torch.bfloat16)
            off = 0
            with torch.no_grad():
                for p in self.params:
                    view = self.flat_params[off:off + p.numel()].view_as(p)
                    view.copy_(p.data)
                    p.data = view
                    off += p.numel()
            self.flat_master = self.flat_params.float()
            self.m = torch.zeros(n_total, device=device)
            self.v = torch.zeros(n_total, device=device)
            self.step_t = torch.zeros(1, device=device, dtype=torch.int32)
            self._no_scale = torch.empty(0, device=device)
            self.opt = None
        else:
            self.masters = [p.detach().float().clone() for p in self.params]
            self.opt = torch.optim.AdamW(self.masters, lr=lr, betas=betas,
                                         eps=eps,
                                         weight_decay=weight_decay)

        self.static = {k: v.clone() for k, v in example_batch.items()}
        # materialize grads once to learn which params receive them
        self.loss_getter(self.model(**self.static)).backward()
        self.flat_grads = torch.zeros(n_total, device=device,
                                      dtype=torch.bfloat16)
        self._views = []
        off = 0
        for p in self.params:
            self._views.append(self.flat_grads[off:off + p.numel()]
                               .view_as(p))
            off += p.numel()
        if self.fused:
            # Leave p.grad unset each step so autograd STEALS the computed
            # gradient (no AccumulateGrad add kernel per param — ~126 tiny
            # adds/step on TIGER); one foreach copy then moves the used
            # grads into the flat buffer. Unused params' flat regions stay
            # zero from init.
            self._used = [i for i, p in enumerate(self.params)
                          if p.grad is not None]
            self._used_views = [self._views[i] for i in self._used]
            for p in self.params:
                p.grad = None
        else:
            for p, view in zip(self.params, self._views):
                p.grad = view
        if not self.fused:
            self.flat_master_grad = torch.zeros(n_total, device=device)
            off = 0
            for mt in self.masters:
                mt.grad = self.flat_master_grad[off:off + mt.numel()] \
                    .view_as(mt)
                off += mt.numel()

        self._graph = None
        self._loss = None
        if use_graph and device.type == "cuda":
            # Warmup + capture + verification replay each run the full
            # optimizer update on the example batch; snapshot the
            # optimizer state first and restore it after (success OR
            # failure), so real training starts from step 0 exactly as
            # eager mode would.
            snap = self._opt_snapshot()
            graph = None
            try:
                s = torch.cuda.Stream()
                s.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(s):
                    for _ in range(warmup_iters):
                        self._inner()
                torch.cuda.current_stream().wait_stream(s)
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    self._loss = self._inner()
                graph = g
            except Exception as e:
                logger.warning(
                    "GraphedTrainStep: capture failed (%s); running the "
                    "same step eagerly", e)
            if self.world > 1:
                # all-or-none capture across ranks: a rank whose capture
                # failed would otherwise skip the verify replay's
                # all_reduce (and any per-step launch asymmetries) and
                # deadlock the job. Capture RECORDS without executing
                # collectives, so counts are still aligned here.
                import torch.distributed as dist

                ok = torch.tensor([1 if graph is not None else 0],
                                  device=device)
                dist.all_reduce(ok, op=dist.ReduceOp.MIN)
                if int(ok.item()) == 0 and graph is not None:
                    logger.warning(
                        "GraphedTrainStep: capture failed on another "
                        "rank; all ranks falling back to eager")
                    graph = None
            if graph is not None:
                graph.replay()  # verify (uniform across ranks)
                torch.cuda.synchronize()
                self._graph = graph
                logger.info("GraphedTrainStep: hipGraph capture active")
            self._opt_restore(snap)

    def _opt_snapshot(self):
        with torch.no_grad():
            if self.fused:
                return (self.flat_master.clone(), self.m.clone(),
                        self.v.clone(), self.step_t.clone())
            import copy

            return ([m.detach().clone() for m in self.masters],
                    copy.deepcopy(self.opt.state_dict()))

    def _opt_restore(self, snap) -> None:
        with torch.no_grad():
            if self.fused:
                fm, m, v, st = snap
                self.flat_master.copy_(fm)
                self.m.copy_(m)
                self.v.copy_(v)
                self.step_t.copy_(st)
                self.flat_params.copy_(self.flat_master.to(torch.bfloat16))
            else:
                masters, opt_state = snap
                for mt, s0 in zip(self.masters, masters):
                    mt.copy_(s0)
                self.opt.load_state_dict(opt_state)
                torch._foreach_copy_(self.params, self.masters)

    def _inner(self):
        if self.device.type == "cuda":
            from genrec_amd.ops.attention import advance_dropout_seeds

            advance_dropout_seeds(self.device)
        if self.fused:
            for p in self.params:
                p.grad = None
        else:
            self.flat_grads.zero_()
        out = self.model(**self.static)
        loss = self.loss_getter(out)
        loss.backward()
        if self.fused:
            torch._foreach_copy_(
                self._used_views,
                [self.params[i].grad for i in self._used])
        if self.world > 1:
            import torch.distributed as dist

            dist.all_reduce(self.flat_grads)
            self.flat_grads.mul_(1.0 / self.world)
        if self.fused:
            from genrec_amd import ops

            if self.clip_norm is not None:
                norm = torch.linalg.vector_norm(self.flat_grads, 2,
                                                dtype=torch.float32)
                scale = (self.clip_norm / (norm + 1e-6)).clamp(max=1.0)
            else:
                scale = self._no_scale
            ops.ext().fused_adamw(
                self.flat_master, self.flat_grads, self.m, self.v,
                self.flat_params, self.lr_t, scale, self.step_t,
                self.betas[0], self.betas[1], self.eps, self.weight_decay)
            return loss
        self.flat_master_grad.copy_(self.flat_grads)
        if self.clip_norm is not None:
            norm = self.flat_master_grad.norm()
            self.flat_master_grad.mul_(
                torch.clamp(self.clip_norm / (norm + 1e-6), max=1.0))
        self.opt.step()
        with torch.no_grad():
            torch._foreach_copy_(self.params, self.masters)
        return loss

    @property
    def captured(self) -> bool:
        return self._graph is not None

    def set_lr(self, lr: float) -> None:
        self.lr_t.fill_(lr)
        if self.opt is not None:
            for g in self.opt.param_groups:
                g["lr"] = lr

    def step(self, batch: Dict[str, torch.Tensor]) -> torch.Tensor:
        """Run one training step on `batch` (shapes must match the example
        batch). Returns the loss tensor (device; .item() syncs)."""
        for k in self.static:
            self.static[k].copy_(batch[k], non_blocking=True)
        if self._graph is not None:
            self._graph.replay()
            return self._loss
        return self._inner()

    def state_dict(self) -> dict:
        if self.fused:
            return {"flat_master": self.flat_master.cpu(),
                    "m": self.m.cpu(), "v": self.v.cpu(),
                    "step": self.step_t.cpu(), "lr": float(self.lr_t)}
        return {"masters": [m.detach().cpu() for m in self.masters],
                "optimizer": self.opt.state_dict(),
                "lr": float(self.lr_t)}

    def load_state_dict(self, state: dict) -> None:
        with torch.no_grad():
            if self.fused:
                if "flat_master" in state:
                    self.flat_master.copy_(state["flat_master"].to(
                        self.device))
                    self.m.copy_(state["m"].to(self.device))
                    self.v.copy_(state["v"].to(self.device))
                    self.step_t.copy_(state["step"].to(self.device))
                else:  # checkpoint written by the non-fused path
                    off = 0
                    for ms in state["masters"]:
                        n = ms.numel()
                        self.flat_master[off:off + n].copy_(
                            ms.reshape(-1).to(self.device))
                        off += n
                self.flat_params.copy_(
                    self.flat_master.to(torch.bfloat16))
            else:
                if "flat_master" in state:  # written by the fused path
                    flat = state["flat_master"]
                    off = 0
                    for mt in self.masters:
                        mt.copy_(flat[off:off + mt.numel()]
                                 .view_as(mt).to(mt.device))
                        off += mt.numel()
                else:
                    for mt, s in zip(self.masters, state["masters"]):
                        mt.copy_(s.to(mt.device))
                    self.opt.load_state_dict(state["optimizer"])
                torch._foreach_copy_(self.params, self.masters)
        self.set_lr(state.get("lr", float(self.lr_t)))
