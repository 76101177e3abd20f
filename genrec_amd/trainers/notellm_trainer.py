"""NoteLLM contrastive trainer.

The reference ships Query2Embedding as a model-only capability (SURVEY.md
§2.1) — no trainer, no config. This trainer completes the family: pairs
are interleaved (anchor, positive) rows, the model's pairwise contrastive
loss (+ optional category-generation CE) trains the backbone, and eval is
the model's own top-k retrieval accuracy. Offline runs use a random-init
Qwen2-shaped backbone (genrec_amd/models/lcrec.py:default_qwen_config).

CLI: python -m genrec_amd.trainers.notellm_trainer config/notellm/synthetic.gin
"""

from __future__ import annotations

import os
from typing import Optional

import torch
from torch.utils.data import DataLoader

from genrec_amd.config import ginlite
from genrec_amd.data.notellm_synthetic import SyntheticNotePairDataset
from genrec_amd.models.notellm import Query2Embedding
from genrec_amd.parallel import GradReducer, init_distributed
from genrec_amd.parallel.ddp import broadcast_parameters
from genrec_amd.trainers import common
from genrec_amd.trainers.common import logger


def _pair_collate(batch):
    texts, cats = [], []
    for s in batch:  # interleave (anchor, positive); category on both rows
        texts.extend([s["query"], s["positive"]])
        cats.extend([s["category"], s["category"]])
    return texts, cats


@torch.no_grad()
def evaluate(model: Query2Embedding, loader, device, topk: int = 5,
             max_batches: Optional[int] = None) -> float:
    model.eval()
    embs = []
    for i, (texts, _) in enumerate(loader):
        if max_batches is not None and i >= max_batches:
            break
        tok = model.tokenize(texts)
        out = model(tok["input_ids"].to(device),
                    tok["attention_mask"].to(device),
                    tok["emb_token_idx"].to(device), return_loss=False)
        embs.append(out["sentence_embedding"])
    if not embs:
        return 0.0
    emb = torch.cat(embs)
    bs = min(64, emb.size(0) // 2)
    return Query2Embedding.topk_retrieval_accuracy(
        emb, topk=min(topk, bs), batch_size=bs)


@ginlite.configurable(name="train")
def train(
    epochs: int = 3,
    learning_rate: float = 3e-5,
    weight_decay: float = 0.01,
    batch_size: int = 16,
    use_category_loss: bool = True,
    pretrained_path: Optional[str] = None,
    backbone_config: Optional[dict] = None,
    gradient_checkpointing: bool = True,
    dataset=None,
    save_dir_root: str = "out/notellm",
    save_every_epoch: int = 1,
    eval_every_epoch: int = 1,
    eval_topk: int = 5,
    do_eval: bool = True,
    wandb_logging: bool = False,
    wandb_project: str = "notellm",
    wandb_log_interval: int = 10,
    seed: int = 42,
    max_steps: Optional[int] = None,
    num_workers: int = 0,
    eval_max_batches: Optional[int] = None,
):
    common.enable_tuned_gemms()
    ctx = init_distributed()
    common.setup_logging(save_dir_root if ctx.is_main else None, "notellm")
    common.set_seed(seed, ctx.rank)
    device = ctx.device

    cfg = None
    if backbone_config is not None:
        from genrec_amd.models.lcrec import default_qwen_config

        cfg = default_qwen_config(**backbone_config)
    model = Query2Embedding(pretrained_path=pretrained_path, config=cfg,
                            gradient_checkpointing=gradient_checkpointing
                            ).to(device)
    broadcast_parameters(model)
    opt = torch.optim.AdamW(model.parameters(), lr=learning_rate,
                            weight_decay=weight_decay)
    reducer = GradReducer(model, bucket_cap_mb=100.0)

    ds_cls = dataset or SyntheticNotePairDataset
    train_ds = ds_cls(**common.dataset_kwargs(ds_cls, {"split": "train"}))
    valid_ds = ds_cls(**common.dataset_kwargs(ds_cls, {"split": "valid"}))
    train_loader = common.make_loader(train_ds, batch_size, ctx, True,
                                      _pair_collate,
                                      num_workers=num_workers, seed=seed,
                                      drop_last=True)
    valid_loader = DataLoader(valid_ds, batch_size=batch_size,
                              collate_fn=_pair_collate)

    wb = common.init_wandb(wandb_project, {"model": "notellm"},
                           wandb_logging, ctx.is_main)
    step = 0
    for epoch in range(epochs):
        model.train()
        if hasattr(train_loader.sampler, "set_epoch"):
            train_loader.sampler.set_epoch(epoch)
        for texts, cats in train_loader:
            tok = model.tokenize(texts,
                                 cats if use_category_loss else None)
            out = model(
                tok["input_ids"].to(device),
                tok["attention_mask"].to(device),
                tok["emb_token_idx"].to(device),
                labels=tok.get("labels", torch.full_like(
                    tok["input_ids"], -100)).to(device))
            opt.zero_grad(set_to_none=False)
            out["loss"].backward()
            reducer.finalize()
            torch.nn.utils.clip_grad_norm_(model.parameters(), 1.0)
            opt.step()
            step += 1
            if ctx.is_main and step % wandb_log_interval == 0:
                logger.info("epoch %d step %d loss %.4f", epoch, step,
                            out["loss"].item())
                wb.log({"train/loss": out["loss"].item()})
            if max_steps is not None and step >= max_steps:
                break
        if do_eval and (epoch + 1) % eval_every_epoch == 0:
            acc = evaluate(model, valid_loader, device, topk=eval_topk,
                           max_batches=eval_max_batches)
            if ctx.is_main:
                logger.info("epoch %d retrieval acc@%d %.4f", epoch,
                            eval_topk, acc)
                wb.log({"eval/retrieval_acc": acc})
        if ctx.is_main and (epoch + 1) % save_every_epoch == 0:
            model.save_pretrained(
                os.path.join(save_dir_root, f"epoch_{epoch}"))
        if max_steps is not None and step >= max_steps:
            break
    ctx.barrier()
    wb.finish()


if __name__ == "__main__":
    ginlite.parse_config()
    train()
