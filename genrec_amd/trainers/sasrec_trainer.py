"""SASRec trainer (parity: reference trainers/sasrec_trainer.py, 213 LoC).

Same CLI: ``python -m genrec_amd.trainers.sasrec_trainer config/sasrec/*.gin
--split beauty``. Differences from the reference are MI355X-idiomatic:
torch.distributed/RCCL directly instead of accelerate, device-resident
Recall/NDCG accumulation (the reference loops per-sample in Python,
sasrec_trainer.py:63-72), one batched metric all-reduce.
"""

from __future__ import annotations

import os
from typing import Optional

import torch
from torch.optim import Adam

from genrec_amd.config import ginlite
from genrec_amd.data.collate import sasrec_collate_fn, sasrec_eval_collate_fn
from genrec_amd.data.synthetic import SyntheticSASRecDataset
from genrec_amd.models.sasrec import SASRec
from genrec_amd.parallel import GradReducer, init_distributed, reduce_scalars
from genrec_amd.parallel.ddp import broadcast_parameters
from genrec_amd.trainers import common
from genrec_amd.trainers.common import logger


@torch.no_grad()
def evaluate(model, loader, device, ctx, ks=(1, 5, 10), amp_ctx=None):
    model.eval()
    sums = {f"recall@{k}": 0.0 for k in ks}
    sums.update({f"ndcg@{k}": 0.0 for k in ks})
    n = 0
    max_k = max(ks)
    for batch in loader:
        batch = common.to_device(batch, device)
        logits, _ = model(batch["input_ids"])
        last = logits[:, -1, :].float()
        last[:, 0] = float("-inf")
        topk = torch.topk(last, max_k, dim=-1).indices  # [B, K]
        tgt = batch["targets"].unsqueeze(1)
        match = topk == tgt  # [B, K]
        found = match.any(dim=1)
        rank = torch.where(found, match.float().argmax(dim=1),
                           torch.full_like(found.long(), max_k))
        dcg = 1.0 / torch.log2(rank.float() + 2.0)
        for k in ks:
            hit = rank < k
            sums[f"recall@{k}"] += hit.sum().item()
            sums[f"ndcg@{k}"] += torch.where(
                hit, dcg, torch.zeros_like(dcg)).sum().item()
        n += batch["input_ids"].size(0)
    sums["n"] = n
    sums = reduce_scalars(sums, device)
    n = max(sums.pop("n"), 1)
    return {k: v / n for k, v in sums.items()}


@ginlite.configurable(name="train")
def train(
    epochs: int = 10,
    batch_size: int = 128,
    learning_rate: float = 1e-3,
    weight_decay: float = 0.0,
    max_seq_len: int = 50,
    embed_dim: int = 64,
    num_heads: int = 2,
    num_blocks: int = 2,
    ffn_dim: int = 256,
    dropout: float = 0.2,
    dataset_folder: str = "dataset/amazon",
    split: str = "beauty",
    dataset=None,
    do_eval: bool = True,
    eval_every_epoch: int = 1,
    eval_batch_size: int = 256,
    save_dir_root: str = "out/sasrec/amazon",
    save_every_epoch: int = 50,
    wandb_logging: bool = False,
    wandb_project: str = "sasrec",
    wandb_log_interval: int = 100,
    amp: bool = True,
    mixed_precision_type: str = "bf16",
    seed: int = 42,
    max_steps: Optional[int] = None,
    num_workers: int = 4,
    use_hip_graph: bool = False,
):
    common.enable_tuned_gemms()
    ctx = init_distributed()
    common.setup_logging(save_dir_root if ctx.is_main else None, "sasrec")
    common.set_seed(seed, ctx.rank)
    device = ctx.device

    ds_cls = dataset or SyntheticSASRecDataset
    train_ds = _make(ds_cls, split=split, mode="train",
                     max_seq_len=max_seq_len, folder=dataset_folder)
    valid_ds = _make(ds_cls, split=split, mode="valid",
                     max_seq_len=max_seq_len, folder=dataset_folder)
    test_ds = _make(ds_cls, split=split, mode="test",
                    max_seq_len=max_seq_len, folder=dataset_folder)
    num_items = getattr(train_ds, "num_items")

    model = SASRec(num_items=num_items, max_seq_len=max_seq_len,
                   embed_dim=embed_dim, num_heads=num_heads,
                   num_blocks=num_blocks, ffn_dim=ffn_dim,
                   dropout=dropout).to(device)
    broadcast_parameters(model)
    opt = Adam(model.parameters(), lr=learning_rate, betas=(0.9, 0.98),
               weight_decay=weight_decay)
    graph_mode = use_hip_graph and device.type == "cuda"
    if graph_mode and weight_decay > 0 and ctx.is_main:
        logger.warning("use_hip_graph applies AdamW-style decoupled decay")
    reducer = None if graph_mode else GradReducer(model)

    coll = lambda b: sasrec_collate_fn(b, max_seq_len,
                                       fixed_length=graph_mode)
    ecoll = lambda b: sasrec_eval_collate_fn(b, max_seq_len)
    train_loader = common.make_loader(train_ds, batch_size, ctx, True, coll,
                                      num_workers=num_workers, seed=seed,
                                      drop_last=graph_mode)
    valid_loader = common.make_loader(valid_ds, eval_batch_size, ctx, False,
                                      ecoll, num_workers=num_workers)
    test_loader = common.make_loader(test_ds, eval_batch_size, ctx, False,
                                     ecoll, num_workers=num_workers)

    wb = common.init_wandb(wandb_project, {"model": "sasrec"},
                           wandb_logging, ctx.is_main)
    amp_ctx = common.autocast_ctx(device, mixed_precision_type if amp else None)

    runner = None
    if graph_mode:
        from genrec_amd.parallel.graph_runner import GraphedTrainStep

        example = common.to_device(next(iter(train_loader)), device)
        runner = GraphedTrainStep(
            model, example, loss_getter=lambda out: out[1],
            lr=learning_rate, weight_decay=weight_decay,
            betas=(0.9, 0.98), clip_norm=None, world=ctx.world_size)

    best_r10, step = -1.0, 0
    best_path = os.path.join(save_dir_root, "best_model.pt")
    for epoch in range(epochs):
        model.train()
        if hasattr(train_loader.sampler, "set_epoch"):
            train_loader.sampler.set_epoch(epoch)
        for batch in train_loader:
            batch = common.to_device(batch, device)
            if runner is not None:
                loss = runner.step(batch)
                step += 1
                if ctx.is_main and step % wandb_log_interval == 0:
                    logger.info("epoch %d step %d loss %.4f", epoch, step,
                                loss.item())
                    wb.log({"train/loss": loss.item(), "train/step": step})
                if max_steps is not None and step >= max_steps:
                    break
                continue
            opt.zero_grad(set_to_none=False)
            with amp_ctx:
                _, loss = model(batch["input_ids"], batch["targets"])
            loss.backward()
            reducer.finalize()
            opt.step()
            step += 1
            if ctx.is_main and step % wandb_log_interval == 0:
                logger.info("epoch %d step %d loss %.4f", epoch, step,
                            loss.item())
                wb.log({"train/loss": loss.item(), "train/step": step})
            if max_steps is not None and step >= max_steps:
                break
        if do_eval and (epoch + 1) % eval_every_epoch == 0:
            metrics = evaluate(model, valid_loader, device, ctx)
            if ctx.is_main:
                logger.info("epoch %d valid %s", epoch, metrics)
                wb.log({f"eval/{k}": v for k, v in metrics.items()})
            r10 = metrics.get("recall@10", 0.0)
            if r10 > best_r10:
                best_r10 = r10
                if ctx.is_main:
                    os.makedirs(save_dir_root, exist_ok=True)
                    torch.save(model.state_dict(), best_path)
        if ctx.is_main and (epoch + 1) % save_every_epoch == 0:
            common.save_checkpoint(
                os.path.join(save_dir_root, f"checkpoint_epoch_{epoch}.pt"),
                model, runner.opt if runner else opt, None,
                epoch=epoch, is_main=True, runner=runner)
        if max_steps is not None and step >= max_steps:
            break

    if do_eval:
        if os.path.exists(best_path):
            model.load_state_dict(torch.load(best_path, map_location=device))
        test_metrics = evaluate(model, test_loader, device, ctx)
        if ctx.is_main:
            logger.info("final test %s", test_metrics)
            wb.log({f"test/{k}": v for k, v in test_metrics.items()})
        wb.finish()
        return test_metrics
    wb.finish()
    return None


def _make(ds_cls, split: str, mode: str, max_seq_len: int, folder: str):
    import inspect

    sig = inspect.signature(ds_cls.__init__)
    if "root" in sig.parameters:  # real-data pipelines
        return ds_cls(root=folder, split=split, train_test_split=mode,
                      max_seq_len=max_seq_len)
    return ds_cls(**common.dataset_kwargs(
        ds_cls, {"split": mode, "max_seq_len": max_seq_len}))


if __name__ == "__main__":
    ginlite.parse_config()
    train()
