"""hipGraph-captured training step (shared by bench.py and the trainers).

Captures one full training step — forward, backward, flat-gradient RCCL
all-reduce, global-norm clip, fp32-master AdamW, bf16 parameter refresh —
and replays it per batch. Requirements/properties:

  * the model runs natively in bf16 (fp32 master weights live in the
    optimizer); no autocast, so no per-layer weight-cast kernels
  * every batch must have the SAME shapes (fixed-size collate + drop_last)
  * gradients are views into one flat bf16 buffer -> N>1 all-reduce is a
    single large RCCL message
  * the learning rate is a device tensor read by the captured capturable
    AdamW, so LR schedules keep working across replays (set_lr / an
    lr_lambda evaluated on the host each step)
  * dropout stays live across replays (torch philox is graph-safe; the
    genrec_amd fused kernels read a captured device seed counter)
  * on ANY capture failure the same step runs eagerly — numerics are
    identical either way (tests/test_kernels_gpu.py::
    test_graph_step_equals_eager_step)
"""

from __future__ import annotations

from typing import Callable, Dict, Optional

import logging

import torch

logger = logging.getLogger("genrec_amd")


class GraphedTrainStep:
    def __init__(self, model: torch.nn.Module,
                 example_batch: Dict[str, torch.Tensor],
                 loss_getter: Callable,
                 lr: float = 1e-4, weight_decay: float = 0.0,
                 betas=(0.9, 0.999), clip_norm: Optional[float] = 1.0,
                 world: int = 1, use_graph: bool = True,
                 warmup_iters: int = 3):
        self.model = model.to(torch.bfloat16)
        self.model.train()
        self.world = world
        self.loss_getter = loss_getter
        self.clip_norm = clip_norm
        device = next(model.parameters()).device
        self.device = device

        self.params = [p for p in model.parameters() if p.requires_grad]
        self.masters = [p.detach().float().clone() for p in self.params]
        # capturable AdamW (CUDA only) reads a DEVICE lr tensor, so LR
        # schedules survive graph replay; the CPU fallback uses a plain
        # float lr updated through param_groups.
        self._capturable = device.type == "cuda"
        self.lr_t = torch.tensor(lr, device=device)
        self.opt = torch.optim.AdamW(
            self.masters, lr=self.lr_t if self._capturable else lr,
            betas=betas, weight_decay=weight_decay,
            capturable=self._capturable, foreach=True)

        self.static = {k: v.clone() for k, v in example_batch.items()}
        # materialize grads once, then re-point them into flat buffers
        self.loss_getter(self.model(**self.static)).backward()
        n_total = sum(p.numel() for p in self.params)
        self.flat_grads = torch.zeros(n_total, device=device,
                                      dtype=torch.bfloat16)
        off = 0
        for p in self.params:
            p.grad = self.flat_grads[off:off + p.numel()].view_as(p)
            off += p.numel()
        self.flat_master_grad = torch.zeros(n_total, device=device)
        off = 0
        for m in self.masters:
            m.grad = self.flat_master_grad[off:off + m.numel()].view_as(m)
            off += m.numel()

        self._graph = None
        self._loss = None
        if use_graph and device.type == "cuda":
            try:
                s = torch.cuda.Stream()
                s.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(s):
                    for _ in range(warmup_iters):
                        self._inner()
                torch.cuda.current_stream().wait_stream(s)
                graph = torch.cuda.CUDAGraph()
                with torch.cuda.graph(graph):
                    self._loss = self._inner()
                graph.replay()
                torch.cuda.synchronize()
                self._graph = graph
                logger.info("GraphedTrainStep: hipGraph capture active")
            except Exception as e:
                logger.warning(
                    "GraphedTrainStep: capture failed (%s); running the "
                    "same step eagerly", e)
                self._graph = None

    def _inner(self):
        self.flat_grads.zero_()
        out = self.model(**self.static)
        loss = self.loss_getter(out)
        loss.backward()
        if self.world > 1:
            import torch.distributed as dist

            dist.all_reduce(self.flat_grads)
            self.flat_grads.mul_(1.0 / self.world)
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
        if not self._capturable:
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
        return {"masters": [m.detach().cpu() for m in self.masters],
                "optimizer": self.opt.state_dict(),
                "lr": float(self.lr_t)}

    def load_state_dict(self, state: dict) -> None:
        with torch.no_grad():
            for m, s in zip(self.masters, state["masters"]):
                m.copy_(s.to(m.device))
            torch._foreach_copy_(self.params, self.masters)
        self.opt.load_state_dict(state["optimizer"])
        self.set_lr(state.get("lr", float(self.lr_t)))
