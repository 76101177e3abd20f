"""COBRA trainer (parity: reference trainers/cobra_trainer.py, 486 LoC).

Weighted sparse+dense loss (cobra_trainer.py:359-362), epoch-accumulated
codebook acc/recall (342-406), eval via beam_fusion against precomputed
item dense vectors (297-334, 414-452), reference dict checkpoints (284-294).
"""

from __future__ import annotations

import os
from typing import Optional

import torch
from torch.optim import AdamW

from genrec_amd.config import ginlite
from genrec_amd.data.cobra_synthetic import SyntheticCobraDataset, cobra_collate_fn
from genrec_amd.models.cobra import Cobra
from genrec_amd.modules.metrics import TopKAccumulator
from genrec_amd.modules.schedulers import get_cosine_schedule_with_warmup
from genrec_amd.parallel import GradReducer, init_distributed, reduce_scalars
from genrec_amd.parallel.ddp import broadcast_parameters
from genrec_amd.trainers import common
from genrec_amd.trainers.common import logger


@torch.no_grad()
def precompute_item_vecs(model: Cobra, item_text: torch.Tensor, device,
                         batch: int = 512) -> torch.Tensor:
    vecs = []
    for i in range(0, item_text.size(0), batch):
        chunk = item_text[i:i + batch].to(device)
        vecs.append(model.generate_itemvec(chunk.unsqueeze(1)).squeeze(1))
    return torch.cat(vecs)


@torch.no_grad()
def evaluate(model: Cobra, loader, device, item_vecs, item_sem_ids,
             ks=(1, 5, 10), n_beam: int = 50, alpha: float = 0.5,
             max_batches: Optional[int] = None):
    model.eval()
    acc = TopKAccumulator(ks=list(ks))
    for bi, batch in enumerate(loader):
        if max_batches is not None and bi >= max_batches:
            break
        out = model.beam_fusion(
            input_ids=batch["input_ids"].to(device),
            encoder_input_ids=batch["encoder_input_ids"].to(device),
            item_dense_vecs=item_vecs, item_sem_ids=item_sem_ids,
            n_candidates=max(ks), n_beam=n_beam, alpha=alpha)
        acc.accumulate(batch["target_sem_ids"].to(device), out.sem_ids)
    return acc.reduce(all_reduce=True)


@ginlite.configurable(name="train")
def train(
    epochs: int = 50,
    learning_rate: float = 3e-4,
    num_warmup_steps: int = 200,
    weight_decay: float = 0.01,
    batch_size: int = 32,
    gradient_accumulate_every: int = 1,
    sparse_loss_weight: float = 1.0,
    dense_loss_weight: float = 1.0,
    n_codebooks: int = 3,
    id_vocab_size: int = 256,
    d_model: int = 768,
    decoder_n_layers: int = 8,
    decoder_num_heads: int = 6,
    decoder_dropout: float = 0.1,
    encoder_n_layers: int = 1,
    encoder_hidden_dim: int = 768,
    encoder_num_heads: int = 8,
    encoder_type: str = "light",
    encoder_model_name: Optional[str] = None,
    dataset=None,
    dataset_folder: str = "dataset/amazon",
    split: str = "beauty",
    eval_n_beam: int = 50,
    eval_alpha: float = 0.5,
    save_dir_root: str = "out/cobra",
    save_every_epoch: int = 10,
    eval_every_epoch: int = 5,
    wandb_logging: bool = False,
    wandb_project: str = "cobra",
    wandb_log_interval: int = 50,
    do_eval: bool = True,
    amp: bool = True,
    mixed_precision_type: str = "bf16",
    seed: int = 42,
    max_steps: Optional[int] = None,
    resume_path: Optional[str] = None,
    num_workers: int = 0,
    eval_max_batches: Optional[int] = None,
    use_hip_graph: bool = False,
):
    common.enable_tuned_gemms()
    ctx = init_distributed()
    common.setup_logging(save_dir_root if ctx.is_main else None, "cobra")
    common.set_seed(seed, ctx.rank)
    device = ctx.device

    ds_cls = dataset or SyntheticCobraDataset
    mk = lambda mode: _make_dataset(ds_cls, mode, dataset_folder, split,
                                    n_codebooks, id_vocab_size)
    train_ds, valid_ds = mk("train"), mk("valid")

    model = Cobra(n_codebooks=n_codebooks, id_vocab_size=id_vocab_size,
                  d_model=d_model, decoder_n_layers=decoder_n_layers,
                  decoder_num_heads=decoder_num_heads,
                  decoder_dropout=decoder_dropout,
                  encoder_n_layers=encoder_n_layers,
                  encoder_hidden_dim=encoder_hidden_dim,
                  encoder_num_heads=encoder_num_heads,
                  encoder_type=encoder_type,
                  encoder_model_name=encoder_model_name).to(device)
    broadcast_parameters(model)
    opt = AdamW(model.parameters(), lr=learning_rate,
                weight_decay=weight_decay)
    pad_id = model.pad_id
    graph_mode = (use_hip_graph and device.type == "cuda"
                  and gradient_accumulate_every == 1)
    fixed_items = (getattr(train_ds, "max_items_per_seq", 20) + 1
                   if graph_mode else 0)
    model.static_infonce = graph_mode  # fixed-shape InfoNCE for capture
    tcoll = lambda b: cobra_collate_fn(b, pad_id, n_codebooks, train=True,
                                       fixed_items=fixed_items)
    ecoll = lambda b: cobra_collate_fn(b, pad_id, n_codebooks, train=False)
    train_loader = common.make_loader(train_ds, batch_size, ctx, True, tcoll,
                                      num_workers=num_workers, seed=seed,
                                      drop_last=True)
    valid_loader = common.make_loader(valid_ds, batch_size, ctx, False,
                                      ecoll, num_workers=num_workers)
    sched = get_cosine_schedule_with_warmup(
        opt, num_warmup_steps, max(1, len(train_loader)) * epochs)
    reducer = None if graph_mode else GradReducer(model)

    start_epoch, step = 0, 0
    resume_state = None
    if resume_path and os.path.exists(resume_path):
        resume_state = common.load_checkpoint(resume_path, model, opt, sched,
                                              map_location=device)
        start_epoch = resume_state.get("epoch", -1) + 1

    wb = common.init_wandb(wandb_project, {"model": "cobra"},
                           wandb_logging, ctx.is_main)
    amp_ctx = common.autocast_ctx(device, mixed_precision_type if amp else None)

    runner = None
    if graph_mode:
        # hipGraph-captured full step (fwd+bwd+allreduce+clip+AdamW) at
        # fixed shapes; eager fallback on capture failure. The fixed-shape
        # InfoNCE variant (models/cobra.py) keeps capture shape-safe.
        from genrec_amd.parallel.graph_runner import GraphedTrainStep

        example = common.to_device(next(iter(train_loader)), device)
        runner = GraphedTrainStep(
            model, {"input_ids": example["input_ids"],
                    "encoder_input_ids": example["encoder_input_ids"]},
            loss_getter=lambda out: (sparse_loss_weight * out.loss_sparse
                                     + dense_loss_weight * out.loss_dense),
            lr=learning_rate, weight_decay=weight_decay, clip_norm=1.0,
            world=ctx.world_size)
        if resume_state is not None and "runner" in resume_state:
            runner.load_state_dict(resume_state["runner"])

        import math as _math

        total_sched = max(1, len(train_loader)) * epochs

        def _cosine_lr(st: int) -> float:
            if st < num_warmup_steps:
                return learning_rate * st / max(1, num_warmup_steps)
            prog = (st - num_warmup_steps) / max(
                1, total_sched - num_warmup_steps)
            return learning_rate * max(
                0.0, 0.5 * (1.0 + _math.cos(_math.pi * prog)))

    for epoch in range(start_epoch, epochs):
        model.train()
        if hasattr(train_loader.sampler, "set_epoch"):
            train_loader.sampler.set_epoch(epoch)
        # device-resident epoch counters: no per-batch .item() syncs
        ep_t = torch.zeros(4, device=device)
        for it, batch in enumerate(train_loader):
            if runner is not None:
                runner.set_lr(_cosine_lr(step))
                loss_t = runner.step(
                    {"input_ids": batch["input_ids"].to(device),
                     "encoder_input_ids": batch["encoder_input_ids"]
                     .to(device)})
                step += 1
                if ctx.is_main and step % wandb_log_interval == 0:
                    logger.info("epoch %d step %d loss %.4f", epoch, step,
                                loss_t.item())
                    wb.log({"train/loss": loss_t.item()})
                if max_steps is not None and step >= max_steps:
                    break
                continue
            micro = (it + 1) % gradient_accumulate_every == 0
            reducer.skip_sync = not micro
            with amp_ctx:
                out = model(input_ids=batch["input_ids"].to(device),
                            encoder_input_ids=batch["encoder_input_ids"]
                            .to(device))
                loss = sparse_loss_weight * out.loss_sparse \
                    + dense_loss_weight * out.loss_dense
            (loss / gradient_accumulate_every).backward()
            with torch.no_grad():
                ep_t += torch.stack([out.acc_correct, out.acc_total,
                                     out.recall_correct,
                                     out.recall_total]).to(ep_t)
            if micro:
                reducer.finalize()
                torch.nn.utils.clip_grad_norm_(model.parameters(), 1.0)
                opt.step()
                sched.step()
                opt.zero_grad(set_to_none=False)
                step += 1
                if ctx.is_main and step % wandb_log_interval == 0:
                    logger.info(
                        "epoch %d step %d loss %.4f sparse %.4f dense %.4f "
                        "entropy %.3f", epoch, step, loss.item(),
                        out.loss_sparse.item(), out.loss_dense.item(),
                        out.codebook_entropy.item())
                    wb.log({"train/loss": loss.item(),
                            "train/codebook_entropy":
                                out.codebook_entropy.item()})
            if max_steps is not None and step >= max_steps:
                break
        ep = reduce_scalars(
            {k: v.item() for k, v in zip(
                ("acc_c", "acc_t", "rec_c", "rec_t"), ep_t)}, device)
        if ctx.is_main and ep["acc_t"] > 0:
            logger.info("epoch %d codebook_acc %.4f item_recall %.4f",
                        epoch, ep["acc_c"] / max(ep["acc_t"], 1),
                        ep["rec_c"] / max(ep["rec_t"], 1))
        if do_eval and (epoch + 1) % eval_every_epoch == 0:
            item_vecs = precompute_item_vecs(
                model, train_ds.all_item_text(), device)
            metrics = evaluate(model, valid_loader, device, item_vecs,
                               train_ds.all_item_sem_ids().to(device),
                               n_beam=eval_n_beam, alpha=eval_alpha,
                               max_batches=eval_max_batches)
            if ctx.is_main:
                logger.info("epoch %d valid %s", epoch, metrics)
                wb.log({f"eval/{k}": v for k, v in metrics.items()})
            model.train()
        if ctx.is_main and (epoch + 1) % save_every_epoch == 0:
            common.save_checkpoint(
                os.path.join(save_dir_root, f"checkpoint_epoch_{epoch}.pt"),
                model, opt, sched, epoch=epoch, is_main=True,
                runner=runner)
        if max_steps is not None and step >= max_steps:
            break
    if ctx.is_main:
        common.save_checkpoint(
            os.path.join(save_dir_root, "checkpoint_final.pt"),
            model, opt, sched, epoch=epochs - 1, is_main=True,
            runner=runner)
    wb.finish()
    return model


def _make_dataset(ds_cls, mode, folder, split, n_codebooks, id_vocab_size):
    import inspect

    sig = inspect.signature(ds_cls.__init__)
    if "root" in sig.parameters:  # real-data pipelines
        return ds_cls(root=folder, split=split, train_test_split=mode,
                      n_codebooks=n_codebooks, id_vocab_size=id_vocab_size)
    kw = common.dataset_kwargs(ds_cls, {
        "split": mode, "n_codebooks": n_codebooks,
        "id_vocab_size": id_vocab_size})
    return ds_cls(**kw)


if __name__ == "__main__":
    ginlite.parse_config()
    train()
