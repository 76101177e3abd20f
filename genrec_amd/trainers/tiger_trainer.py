"""TIGER trainer (parity: reference trainers/tiger_trainer.py, 381 LoC).

CLI: ``python -m genrec_amd.trainers.tiger_trainer config/tiger/amazon/tiger.gin
--split beauty``. Checkpoint dict layout matches the reference writer
(tiger_trainer.py:258-269). Eval = constrained beam generate + TopKAccumulator
(tiger_trainer.py:271-288), with the C5 fix: accumulator counters are
all-reduced across ranks.
"""

from __future__ import annotations

import os
from typing import Optional

import torch
from torch.optim import AdamW

from genrec_amd.config import ginlite
from genrec_amd.data.collate import tiger_pad_collate
from genrec_amd.data.synthetic import SyntheticSemIdSeqDataset
from genrec_amd.models.tiger import Tiger
from genrec_amd.modules.metrics import TopKAccumulator
from genrec_amd.modules.schedulers import get_cosine_schedule_with_warmup
from genrec_amd.parallel import GradReducer, init_distributed
from genrec_amd.parallel.ddp import broadcast_parameters
from genrec_amd.trainers import common
from genrec_amd.trainers.common import logger
from genrec_amd.utils.profiling import StepProfiler, roctx_range


@torch.no_grad()
def evaluate(model: Tiger, loader, device, valid_item_ids, amp_ctx,
             ks=(1, 5, 10), max_batches: Optional[int] = None):
    model.eval()
    acc = TopKAccumulator(ks=list(ks))
    sem_dim = model.sem_id_dim
    for i, batch in enumerate(loader):
        if max_batches is not None and i >= max_batches:
            break
        batch = common.to_device(batch, device)
        with amp_ctx:
            gen = model.generate(
                user_input_ids=batch["user_input_ids"],
                item_input_ids=batch["item_input_ids"],
                token_type_ids=batch["token_type_ids"],
                seq_mask=batch["seq_mask"],
                n_top_k_candidates=max(ks),
                valid_item_ids=valid_item_ids,
            )
        target = batch["target_input_ids"][:, :sem_dim]
        acc.accumulate(target, gen.sem_ids)
    return acc.reduce(all_reduce=True)


@ginlite.configurable(name="train")
def train(
    epochs: int = 100,
    learning_rate: float = 1e-4,
    num_warmup_steps: int = 100,
    weight_decay: float = 0.035,
    batch_size: int = 256,
    gradient_accumulate_every: int = 1,
    embedding_dim: int = 128,
    attn_dim: int = 384,
    dropout: float = 0.1,
    num_heads: int = 6,
    n_layers: int = 8,
    num_item_embeddings: int = 256,
    num_user_embeddings: int = 10000,
    sem_id_dim: int = 3,
    max_seq_len: int = 20,
    dataset=None,
    dataset_folder: str = "dataset/amazon",
    split: str = "beauty",
    pretrained_rqvae_path: Optional[str] = None,
    save_every_epoch: int = 50,
    eval_valid_every_epoch: int = 1,
    eval_test_every_epoch: int = 10,
    save_dir_root: str = "out/tiger/amazon",
    wandb_logging: bool = False,
    wandb_project: str = "tiger",
    wandb_log_interval: int = 10,
    do_eval: bool = True,
    amp: bool = True,
    mixed_precision_type: str = "bf16",
    seed: int = 42,
    max_steps: Optional[int] = None,
    resume_path: Optional[str] = None,
    num_workers: int = 4,
    eval_max_batches: Optional[int] = None,
    profile_steps: int = 0,
    profile_trace_path: Optional[str] = None,
    use_hip_graph: bool = False,
):
    common.enable_tuned_gemms()
    ctx = init_distributed()
    common.setup_logging(save_dir_root if ctx.is_main else None, "tiger")
    common.set_seed(seed, ctx.rank)
    device = ctx.device

    ds_cls = dataset or SyntheticSemIdSeqDataset
    mk = lambda mode: _make_dataset(
        ds_cls, split=split, mode=mode, folder=dataset_folder,
        max_items=max_seq_len, sem_id_dim=sem_id_dim,
        codebook_size=num_item_embeddings,
        rqvae_path=pretrained_rqvae_path)
    train_ds, valid_ds = mk("train"), mk("valid")
    valid_item_ids = train_ds.all_valid_sem_ids().to(device)

    model = Tiger(
        embedding_dim=embedding_dim, attn_dim=attn_dim, dropout=dropout,
        num_heads=num_heads, n_layers=n_layers,
        num_item_embeddings=num_item_embeddings,
        num_user_embeddings=num_user_embeddings,
        sem_id_dim=sem_id_dim).to(device)
    broadcast_parameters(model)
    opt = AdamW(model.parameters(), lr=learning_rate,
                weight_decay=weight_decay)

    graph_mode = (use_hip_graph and device.type == "cuda"
                  and gradient_accumulate_every == 1)
    fixed_len = max_seq_len * sem_id_dim if graph_mode else 0
    coll = lambda b: tiger_pad_collate(b, sem_id_dim=sem_id_dim,
                                       fixed_length=fixed_len)
    train_loader = common.make_loader(train_ds, batch_size, ctx, True, coll,
                                      num_workers=num_workers, seed=seed,
                                      drop_last=True)
    valid_loader = common.make_loader(valid_ds, batch_size, ctx, False, coll,
                                      num_workers=num_workers)
    steps_per_epoch = max(1, len(train_loader) // gradient_accumulate_every)
    sched = get_cosine_schedule_with_warmup(
        opt, num_warmup_steps, steps_per_epoch * epochs)
    # graph mode reduces via ONE flat all-reduce inside the captured step;
    # GradReducer's per-bucket hooks must not also fire.
    reducer = None if graph_mode else GradReducer(model)

    start_epoch, step = 0, 0
    resume_state = None
    if resume_path and os.path.exists(resume_path):
        resume_state = common.load_checkpoint(resume_path, model, opt, sched,
                                              map_location=device)
        start_epoch = resume_state.get("epoch", -1) + 1

    wb = common.init_wandb(wandb_project, {"model": "tiger"},
                           wandb_logging, ctx.is_main)
    amp_ctx = common.autocast_ctx(device, mixed_precision_type if amp else None)
    prof = StepProfiler(enabled=profile_steps > 0 and ctx.is_main,
                        active=profile_steps,
                        trace_path=profile_trace_path)

    runner = None
    if graph_mode:
        # hipGraph-captured full step (fwd+bwd+allreduce+clip+AdamW) at
        # fixed shapes; falls back to the same step eagerly on capture
        # failure. The cosine schedule drives a device LR tensor.
        from genrec_amd.parallel.graph_runner import GraphedTrainStep

        example = next(iter(train_loader))
        example = common.to_device(example, device)
        runner = GraphedTrainStep(
            model, example, loss_getter=lambda out: out.loss,
            lr=learning_rate, weight_decay=weight_decay,
            clip_norm=1.0, world=ctx.world_size)
        if resume_state is not None and "runner" in resume_state:
            # restore fp32 masters + AdamW moments/step for the fused path
            runner.load_state_dict(resume_state["runner"])

        import math as _math

        def _cosine_lr(st: int) -> float:
            total = steps_per_epoch * epochs
            if st < num_warmup_steps:
                return learning_rate * st / max(1, num_warmup_steps)
            prog = (st - num_warmup_steps) / max(1, total - num_warmup_steps)
            return learning_rate * max(
                0.0, 0.5 * (1.0 + _math.cos(_math.pi * prog)))

    for epoch in range(start_epoch, epochs):
        model.train()
        if hasattr(train_loader.sampler, "set_epoch"):
            train_loader.sampler.set_epoch(epoch)
        opt.zero_grad(set_to_none=False)
        for it, batch in enumerate(train_loader):
          with prof.step():
            with roctx_range("data_to_device"):
                batch = common.to_device(batch, device)
            if runner is not None:
                runner.set_lr(_cosine_lr(step))
                loss_t = runner.step(batch)
                step += 1
                if ctx.is_main and step % wandb_log_interval == 0:
                    logger.info("epoch %d step %d loss %.4f lr %.2e",
                                epoch, step, loss_t.item(),
                                float(runner.lr_t))
                    wb.log({"train/loss": loss_t.item()})
                if max_steps is not None and step >= max_steps:
                    break
                continue
            micro = (it + 1) % gradient_accumulate_every == 0
            reducer.skip_sync = not micro
            with roctx_range("forward"), amp_ctx:
                out = model(
                    user_input_ids=batch["user_input_ids"],
                    item_input_ids=batch["item_input_ids"],
                    token_type_ids=batch["token_type_ids"],
                    target_input_ids=batch["target_input_ids"],
                    target_token_type_ids=batch["target_token_type_ids"],
                    seq_mask=batch["seq_mask"],
                )
            with roctx_range("backward"):
                (out.loss / gradient_accumulate_every).backward()
            if micro:
                with roctx_range("optimizer"):
                    reducer.finalize()
                    torch.nn.utils.clip_grad_norm_(model.parameters(), 1.0)
                    opt.step()
                    sched.step()
                    opt.zero_grad(set_to_none=False)
                step += 1
                if ctx.is_main and step % wandb_log_interval == 0:
                    logger.info("epoch %d step %d loss %.4f lr %.2e",
                                epoch, step, out.loss.item(),
                                sched.get_last_lr()[0])
                    wb.log({"train/loss": out.loss.item(),
                            "train/lr": sched.get_last_lr()[0]})
            if max_steps is not None and step >= max_steps:
                break
        if do_eval and (epoch + 1) % eval_valid_every_epoch == 0:
            metrics = evaluate(model, valid_loader, device, valid_item_ids,
                               amp_ctx, max_batches=eval_max_batches)
            if ctx.is_main:
                logger.info("epoch %d valid %s", epoch, metrics)
                wb.log({f"eval/{k}": v for k, v in metrics.items()})
        if ctx.is_main and (epoch + 1) % save_every_epoch == 0:
            common.save_checkpoint(
                os.path.join(save_dir_root, f"checkpoint_epoch_{epoch}.pt"),
                model, runner.opt if runner else opt,
                None if runner else sched, epoch=epoch, is_main=True,
                runner=runner)
        if max_steps is not None and step >= max_steps:
            break
    prof.report()
    ctx.barrier()
    if ctx.is_main:
        common.save_checkpoint(
            os.path.join(save_dir_root, "checkpoint_final.pt"),
            model, runner.opt if runner else opt,
            None if runner else sched, epoch=epochs - 1, is_main=True,
            runner=runner)
    wb.finish()


def _make_dataset(ds_cls, split, mode, folder, max_items, sem_id_dim,
                  codebook_size, rqvae_path):
    import inspect

    sig = inspect.signature(ds_cls.__init__)
    if "root" in sig.parameters:  # real-data pipelines
        kw = dict(root=folder, split=split, train_test_split=mode,
                  max_items_per_seq=max_items)
        if rqvae_path:
            kw["pretrained_rqvae_path"] = rqvae_path
        return ds_cls(**kw)
    kw = common.dataset_kwargs(ds_cls, {
        "split": mode, "max_items_per_seq": max_items,
        "sem_id_dim": sem_id_dim, "codebook_size": codebook_size})
    return ds_cls(**kw)


if __name__ == "__main__":
    ginlite.parse_config()
    train()
