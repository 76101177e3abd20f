"""Shared trainer runtime.

The reference ships six copy-pasted ~200-490-line trainer scripts
(SURVEY.md §1 notes the absence of a shared base). This module centralizes
what they duplicate: seeding, logging, checkpoint I/O (dict format parity
with the reference writers, e.g. tiger_trainer.py:258-269), distributed
setup, AMP autocast and the DataLoader/sampler plumbing.
"""

from __future__ import annotations

import logging
import os
import random
import time
from typing import Any, Dict, Optional

import numpy as np
import torch
from torch.utils.data import DataLoader, Dataset, DistributedSampler

from genrec_amd.parallel import DistributedContext

logger = logging.getLogger("genrec_amd")


def enable_tuned_gemms() -> None:
    """Point TunableOp at the shipped hipBLASLt solution tables
    (benchmarks/tunableop<ordinal>.csv) when present — same mechanism
    bench.py uses. No-op if the env is already configured or the tables
    are absent; tuning itself stays off."""
    repo = os.path.dirname(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    csv0 = os.path.join(repo, "benchmarks", "tunableop0.csv")
    if os.path.exists(csv0) and \
            os.environ.get("PYTORCH_TUNABLEOP_TUNING", "0") != "1":
        os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
        os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
        os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME",
                              os.path.join(repo, "benchmarks",
                                           "tunableop.csv"))


def setup_logging(log_dir: Optional[str] = None, name: str = "train") -> None:
    handlers: list = [logging.StreamHandler()]
    if log_dir:
        os.makedirs(log_dir, exist_ok=True)
        stamp = time.strftime("%Y%m%d_%H%M%S")
        handlers.append(logging.FileHandler(
            os.path.join(log_dir, f"{name}_{stamp}.log")))
    logging.basicConfig(
        level=logging.INFO,
        format="%(asctime)s %(levelname)s %(name)s: %(message)s",
        handlers=handlers, force=True)


def set_seed(seed: int, rank: int = 0) -> None:
    random.seed(seed + rank)
    np.random.seed(seed + rank)
    torch.manual_seed(seed + rank)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed + rank)


def autocast_ctx(device: torch.device, mixed_precision: Optional[str]):
    if mixed_precision in ("bf16", "fp16") and device.type == "cuda":
        dtype = torch.bfloat16 if mixed_precision == "bf16" else torch.float16
        return torch.autocast(device_type="cuda", dtype=dtype)
    import contextlib

    return contextlib.nullcontext()


class UnpaddedShardSampler(torch.utils.data.Sampler):
    """Contiguous per-rank eval shard WITHOUT duplicate padding.

    DistributedSampler pads the last shard by repeating samples; the
    all-reduced Recall/NDCG counters would count those duplicates and bias
    metrics versus the reference's single-process eval. Here rank r takes
    indices [r*q + min(r, rem), ...) — shard sizes differ by at most 1 and
    every sample appears exactly once across ranks.
    """

    def __init__(self, dataset, num_replicas: int, rank: int):
        n = len(dataset)
        q, rem = divmod(n, num_replicas)
        start = rank * q + min(rank, rem)
        self._indices = list(range(start, start + q + (1 if rank < rem else 0)))

    def __iter__(self):
        return iter(self._indices)

    def __len__(self):
        return len(self._indices)


def make_loader(dataset: Dataset, batch_size: int, ctx: DistributedContext,
                shuffle: bool, collate_fn=None, num_workers: int = 4,
                drop_last: bool = False, seed: int = 0) -> DataLoader:
    sampler = None
    if ctx.world_size > 1:
        if shuffle:
            sampler = DistributedSampler(
                dataset, num_replicas=ctx.world_size, rank=ctx.rank,
                shuffle=True, seed=seed, drop_last=drop_last)
        else:  # eval: no duplicate padding (metrics are all-reduced)
            sampler = UnpaddedShardSampler(dataset, ctx.world_size, ctx.rank)
        shuffle = False
    return DataLoader(
        dataset, batch_size=batch_size, shuffle=shuffle, sampler=sampler,
        collate_fn=collate_fn, num_workers=num_workers,
        pin_memory=torch.cuda.is_available(), drop_last=drop_last,
        persistent_workers=num_workers > 0)


def save_checkpoint(path: str, model: torch.nn.Module, optimizer, scheduler,
                    *, epoch: Optional[int] = None, step: Optional[int] = None,
                    model_config: Optional[dict] = None,
                    is_main: bool = True, runner=None) -> None:
    """Dict checkpoint, reference-compatible layout
    ({epoch|iter, model, model_config, optimizer, scheduler}).

    `runner` (a GraphedTrainStep) saves the fused-path optimizer state —
    fp32 flat masters, AdamW moments, step counter, lr — under "runner";
    without it a hipGraph-mode resume would silently restart AdamW."""
    if not is_main:
        return
    os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
    state: Dict[str, Any] = {"model": model.state_dict()}
    if epoch is not None:
        state["epoch"] = epoch
    if step is not None:
        state["iter"] = step
    if model_config is not None:
        state["model_config"] = model_config
    if optimizer is not None:
        state["optimizer"] = optimizer.state_dict()
    if scheduler is not None:
        state["scheduler"] = scheduler.state_dict()
    if runner is not None:
        state["runner"] = runner.state_dict()
    tmp = path + ".tmp"
    torch.save(state, tmp)
    os.replace(tmp, path)
    logger.info("saved checkpoint to %s", path)


def load_checkpoint(path: str, model: torch.nn.Module, optimizer=None,
                    scheduler=None, map_location="cpu",
                    runner=None) -> Dict[str, Any]:
    state = torch.load(path, map_location=map_location, weights_only=False)
    model.load_state_dict(state["model"])
    if optimizer is not None and "optimizer" in state:
        optimizer.load_state_dict(state["optimizer"])
    if scheduler is not None and "scheduler" in state:
        scheduler.load_state_dict(state["scheduler"])
    if runner is not None and "runner" in state:
        runner.load_state_dict(state["runner"])
    logger.info("resumed from %s (epoch=%s iter=%s)", path,
                state.get("epoch"), state.get("iter"))
    return state


def to_device(batch, device: torch.device):
    if isinstance(batch, dict):
        return {k: to_device(v, device) for k, v in batch.items()}
    if isinstance(batch, torch.Tensor):
        return batch.to(device, non_blocking=True)
    if isinstance(batch, (list, tuple)):
        t = type(batch)
        vals = [to_device(v, device) for v in batch]
        return t(*vals) if hasattr(batch, "_fields") else t(vals)
    return batch


class WandbStub:
    """No-op logger used when wandb is unavailable (offline env)."""

    def log(self, *a, **k):
        pass

    def finish(self):
        pass


def init_wandb(project: str, config: dict, enabled: bool, is_main: bool):
    if not (enabled and is_main):
        return WandbStub()
    try:
        import wandb

        wandb.init(project=project, config=config)
        # step-keyed train metrics vs epoch-keyed eval metrics
        # (ref tiger_trainer.py:140-141)
        wandb.define_metric("train/*", step_metric="train/step")
        wandb.define_metric("eval/*", step_metric="eval/epoch")
        return wandb
    except Exception:
        logger.warning("wandb unavailable; logging disabled")
        return WandbStub()


def dataset_kwargs(ds_cls, wanted: Dict[str, Any]) -> Dict[str, Any]:
    """Filter dataset-constructor kwargs by signature; classes accepting
    **kwargs receive everything (subclasses wrapping a base via **kw)."""
    import inspect

    sig = inspect.signature(ds_cls.__init__)
    has_var_kw = any(p.kind == inspect.Parameter.VAR_KEYWORD
                     for p in sig.parameters.values())
    if has_var_kw:
        return dict(wanted)
    return {k: v for k, v in wanted.items() if k in sig.parameters}
