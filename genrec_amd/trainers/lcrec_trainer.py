"""LCRec trainer (parity: reference trainers/lcrec_trainer.py, 442 LoC).

SFT collate with prompt-masked labels (-100) and left-padding for
generation (lcrec_trainer.py:43-84), ConstrainedDecodingHelper with
per-position codebook token sets (lcrec_trainer.py:87-128), 3-task eval
(seqrec constrained beam / item2index greedy / index2item substring match,
lcrec_trainer.py:131-239), eval-only mode, HF-format save_pretrained
checkpoints (lcrec_trainer.py:419-430). LoRA via peft when installed
(gated — peft is absent in this offline image).
"""

from __future__ import annotations

import os
import re
from typing import Dict, List, Optional

import torch

from genrec_amd.config import ginlite
from genrec_amd.data.lcrec_sft import SyntheticLCRecDataset
from genrec_amd.models.lcrec import LCRec
from genrec_amd.modules.metrics import TopKAccumulator
from genrec_amd.modules.schedulers import get_cosine_schedule_with_warmup
from genrec_amd.parallel import GradReducer, init_distributed, reduce_scalars
from genrec_amd.parallel.ddp import broadcast_parameters
from genrec_amd.trainers import common
from genrec_amd.trainers.common import logger


class ConstrainedDecodingHelper:
    """Per-position legal codebook token sets for generation
    (ref lcrec_trainer.py:87-128), as device id tensors."""

    def __init__(self, model: LCRec, n_codebooks: int, codebook_size: int):
        self.table = model.codebook_token_ids(n_codebooks, codebook_size)
        self.eos = model.tokenizer.eos_token_id
        self.n_codebooks = n_codebooks

    def allowed_token_ids(self) -> List[torch.Tensor]:
        out = [self.table[c] for c in range(self.n_codebooks)]
        out.append(torch.tensor([self.eos]))
        return out


def sft_collate(batch: List[Dict], model: LCRec, max_len: int = 512,
                for_generation: bool = False) -> Dict[str, torch.Tensor]:
    tok = model.tokenizer
    input_ids, labels, attn = [], [], []
    metas = []
    for s in batch:
        p_ids = tok(s["prompt"]).input_ids
        if for_generation:
            ids = p_ids[-max_len:]
            input_ids.append(ids)
            metas.append(s)
            continue
        r_ids = tok(s["response"]).input_ids + [tok.eos_token_id]
        ids = (p_ids + r_ids)[:max_len]
        lab = ([-100] * len(p_ids) + r_ids)[:max_len]
        input_ids.append(ids)
        labels.append(lab)
    max_l = max(len(x) for x in input_ids)
    pad = tok.pad_token_id
    out_ids, out_attn, out_lab = [], [], []
    for i, ids in enumerate(input_ids):
        n_pad = max_l - len(ids)
        if for_generation:  # left pad for decode
            out_ids.append([pad] * n_pad + ids)
            out_attn.append([0] * n_pad + [1] * len(ids))
        else:  # right pad for SFT
            out_ids.append(ids + [pad] * n_pad)
            out_attn.append([1] * len(ids) + [0] * n_pad)
            out_lab.append(labels[i] + [-100] * n_pad)
    res = {
        "input_ids": torch.tensor(out_ids, dtype=torch.long),
        "attention_mask": torch.tensor(out_attn, dtype=torch.long),
    }
    if not for_generation:
        res["labels"] = torch.tensor(out_lab, dtype=torch.long)
    else:
        res["meta"] = metas
    return res


SEM_ID_RE = re.compile(r"<C(\d+)_(\d+)>")


def extract_sem_ids(text: str, n_codebooks: int) -> Optional[List[int]]:
    found = SEM_ID_RE.findall(text)
    out = [-1] * n_codebooks
    for c, v in found[:n_codebooks]:
        c = int(c)
        if 0 <= c < n_codebooks:
            out[c] = int(v)
    return out


@torch.no_grad()
def evaluate_aux_tasks(model: LCRec, metas, device,
                       helper: ConstrainedDecodingHelper, n_codebooks: int,
                       max_samples: int = 64) -> Dict[str, float]:
    """item2index greedy accuracy + index2item substring match
    (ref lcrec_trainer.py:190-222)."""
    from genrec_amd.data.lcrec_sft import TEMPLATES, sem_ids_to_tokens

    allowed = helper.allowed_token_ids()
    metas = metas[:max_samples]
    out = {"item2index_correct": 0.0, "index2item_correct": 0.0,
           "aux_total": float(len(metas))}
    if not metas:
        return out
    # --- item2index: greedy constrained decode from the title prompt
    prompts = [TEMPLATES["item2index"][0].format(
        title=m.get("target_title", ""), description="") for m in metas]
    batch = sft_collate([{"prompt": p, "response": ""} for p in prompts],
                        model, for_generation=True)
    ids = batch["input_ids"].to(device)
    res = model.generate_topk(ids,
                              attention_mask=batch["attention_mask"]
                              .to(device),
                              max_new_tokens=n_codebooks, beam_width=1,
                              allowed_token_ids=allowed)
    L = ids.size(1)
    for m, row in zip(metas, res):
        text = model.decode(row[0][0][L:], skip_special_tokens=False)
        if extract_sem_ids(text, n_codebooks) == m["target_sem_ids"]:
            out["item2index_correct"] += 1.0
    # --- index2item: free generation, substring match of the title
    prompts = [TEMPLATES["index2item"][0].format(
        index=sem_ids_to_tokens(m["target_sem_ids"])) for m in metas]
    batch = sft_collate([{"prompt": p, "response": ""} for p in prompts],
                        model, for_generation=True)
    ids = batch["input_ids"].to(device)
    res = model.generate_topk(ids,
                              attention_mask=batch["attention_mask"]
                              .to(device),
                              max_new_tokens=24, beam_width=1)
    L = ids.size(1)
    for m, row in zip(metas, res):
        text = model.decode(row[0][0][L:], skip_special_tokens=True)
        title = str(m.get("target_title", "")).strip().lower()
        if title and (title in text.lower() or text.strip().lower() in title):
            out["index2item_correct"] += 1.0
    return out


@torch.no_grad()
def evaluate(model: LCRec, loader, device, helper: ConstrainedDecodingHelper,
             n_codebooks: int, ks=(1, 5, 10), beam_width: int = 10,
             max_batches: Optional[int] = None,
             aux_tasks: bool = True,
             debug_logging: bool = False) -> Dict[str, float]:
    model.eval()
    acc = TopKAccumulator(ks=list(ks))
    allowed = helper.allowed_token_ids()
    aux_metas = []
    for bi, batch in enumerate(loader):
        if max_batches is not None and bi >= max_batches:
            break
        ids = batch["input_ids"].to(device)
        attn = batch["attention_mask"].to(device)
        metas = batch["meta"]
        if aux_tasks and len(aux_metas) < 64:
            aux_metas.extend(metas)
        res = model.generate_topk(
            ids, attention_mask=attn, max_new_tokens=n_codebooks,
            beam_width=beam_width, allowed_token_ids=allowed)
        L = ids.size(1)
        preds, tgts = [], []
        for b, row in enumerate(res):
            tgts.append(metas[b]["target_sem_ids"])
            beams = []
            for seq, _ in row[:max(ks)]:
                text = model.decode(seq[L:], skip_special_tokens=False)
                sem = extract_sem_ids(text, n_codebooks)
                beams.append(sem if sem else [-1] * n_codebooks)
            while len(beams) < max(ks):
                beams.append([-1] * n_codebooks)
            preds.append(beams)
        if debug_logging and bi == 0:
            # target-vs-pred sample dump (ref lcrec_trainer.py:176-178)
            for b in range(min(3, len(tgts))):
                logger.info("debug sample %d: target=%s top1=%s", b,
                            tgts[b], preds[b][0])
        acc.accumulate(torch.tensor(tgts, device=device),
                       torch.tensor(preds, device=device))
    metrics = acc.reduce(all_reduce=True)
    if aux_tasks:
        aux = evaluate_aux_tasks(model, aux_metas, device, helper,
                                 n_codebooks)
        aux = reduce_scalars(aux, device)
        t = max(aux.pop("aux_total"), 1.0)
        metrics["item2index_acc"] = aux["item2index_correct"] / t
        metrics["index2item_acc"] = aux["index2item_correct"] / t
    return metrics


@ginlite.configurable(name="train")
def train(
    epochs: int = 4,
    learning_rate: float = 3e-5,
    num_warmup_steps: int = 100,
    weight_decay: float = 0.01,
    batch_size: int = 32,
    gradient_accumulate_every: int = 1,
    max_seq_len: int = 512,
    n_codebooks: int = 5,
    codebook_size: int = 256,
    pretrained_path: Optional[str] = None,
    backbone_config: Optional[dict] = None,
    use_lora: bool = False,
    gradient_checkpointing: bool = True,
    dataset=None,
    dataset_folder: str = "dataset/amazon",
    split: str = "beauty",
    pretrained_rqvae_path: Optional[str] = None,
    max_train_samples: Optional[int] = None,
    max_eval_samples: Optional[int] = None,
    save_dir_root: str = "out/lcrec",
    save_every_epoch: int = 1,
    eval_every_epoch: int = 1,
    eval_beam_width: int = 10,
    wandb_logging: bool = False,
    wandb_project: str = "lcrec",
    wandb_log_interval: int = 10,
    do_eval: bool = True,
    eval_only: bool = False,
    checkpoint_path: Optional[str] = None,
    mixed_precision_type: str = "bf16",
    seed: int = 42,
    max_steps: Optional[int] = None,
    num_workers: int = 0,
    eval_max_batches: Optional[int] = None,
    # ---- reference-config drop-in aliases (config/lcrec/amazon/lcrec.gin
    # of the reference binds these names)
    max_length: Optional[int] = None,          # == max_seq_len
    num_codebooks: Optional[int] = None,       # == n_codebooks
    warmup_ratio: Optional[float] = None,      # fraction -> num_warmup_steps
    eval_batch_size: Optional[int] = None,
    amp: bool = True,                          # False -> fp32 eval/train
    debug_logging: bool = False,               # print target-vs-pred samples
    max_text_len: Optional[int] = None,        # forwarded to the dataset
):
    if max_length is not None:
        max_seq_len = max_length
    if num_codebooks is not None:
        n_codebooks = num_codebooks
    if not amp:
        mixed_precision_type = None
    common.enable_tuned_gemms()
    ctx = init_distributed()
    common.setup_logging(save_dir_root if ctx.is_main else None, "lcrec")
    common.set_seed(seed, ctx.rank)
    device = ctx.device

    model = LCRec(pretrained_path=pretrained_path)
    if backbone_config is not None:
        from genrec_amd.models.lcrec import default_qwen_config

        model = LCRec(config=default_qwen_config(**backbone_config))
    model.add_codebook_tokens(n_codebooks, codebook_size)
    if checkpoint_path and os.path.isdir(checkpoint_path):
        model.load_pretrained(checkpoint_path)
    if use_lora:
        try:
            from peft import LoraConfig, get_peft_model

            model.model = get_peft_model(model.model, LoraConfig(
                r=16, lora_alpha=32, target_modules=["q_proj", "v_proj"]))
        except ImportError:
            logger.warning("peft not installed; training full model")
    if gradient_checkpointing:
        model.gradient_checkpointing_enable()
    model = model.to(device)
    broadcast_parameters(model)

    ds_cls = dataset or SyntheticLCRecDataset
    mk = lambda mode, cap: _make_dataset(
        ds_cls, mode, dataset_folder, split, n_codebooks, codebook_size,
        pretrained_rqvae_path, cap, max_text_len=max_text_len)
    train_ds = mk("train", max_train_samples)
    valid_ds = mk("valid", max_eval_samples)

    helper = ConstrainedDecodingHelper(model, n_codebooks, codebook_size)
    coll = lambda b: sft_collate(b, model, max_seq_len)
    gcoll = lambda b: sft_collate(b, model, max_seq_len, for_generation=True)
    train_loader = common.make_loader(train_ds, batch_size, ctx, True, coll,
                                      num_workers=num_workers, seed=seed,
                                      drop_last=True)
    valid_loader = common.make_loader(valid_ds,
                                      eval_batch_size or batch_size,
                                      ctx, False, gcoll,
                                      num_workers=num_workers)

    wb = common.init_wandb(wandb_project, {"model": "lcrec"},
                           wandb_logging, ctx.is_main)
    amp_ctx = common.autocast_ctx(device, mixed_precision_type)

    if eval_only:
        metrics = evaluate(model, valid_loader, device, helper, n_codebooks,
                           beam_width=eval_beam_width,
                           max_batches=eval_max_batches,
                           debug_logging=debug_logging)
        if ctx.is_main:
            logger.info("eval-only %s", metrics)
        wb.finish()
        return metrics

    opt = torch.optim.AdamW(model.parameters(), lr=learning_rate,
                            weight_decay=weight_decay)
    steps_per_epoch = max(1, len(train_loader) // gradient_accumulate_every)
    if warmup_ratio is not None:
        num_warmup_steps = max(1, int(warmup_ratio * steps_per_epoch
                                      * epochs))
    sched = get_cosine_schedule_with_warmup(opt, num_warmup_steps,
                                            steps_per_epoch * epochs)
    # LCRec-scale grads are bandwidth-bound: >=100 MB buckets
    reducer = GradReducer(model, bucket_cap_mb=100.0)

    step = 0
    for epoch in range(epochs):
        model.train()
        if hasattr(train_loader.sampler, "set_epoch"):
            train_loader.sampler.set_epoch(epoch)
        opt.zero_grad(set_to_none=False)
        for it, batch in enumerate(train_loader):
            batch = {k: v.to(device) for k, v in batch.items()}
            micro = (it + 1) % gradient_accumulate_every == 0
            reducer.skip_sync = not micro
            with amp_ctx:
                out = model(**batch)
            (out.loss / gradient_accumulate_every).backward()
            if micro:
                reducer.finalize()
                torch.nn.utils.clip_grad_norm_(model.parameters(), 1.0)
                opt.step()
                sched.step()
                opt.zero_grad(set_to_none=False)
                step += 1
                if ctx.is_main and step % wandb_log_interval == 0:
                    logger.info("epoch %d step %d loss %.4f", epoch, step,
                                out.loss.item())
                    wb.log({"train/loss": out.loss.item()})
            if max_steps is not None and step >= max_steps:
                break
        if do_eval and (epoch + 1) % eval_every_epoch == 0:
            metrics = evaluate(model, valid_loader, device, helper,
                               n_codebooks, beam_width=eval_beam_width,
                               max_batches=eval_max_batches,
                               debug_logging=debug_logging)
            if ctx.is_main:
                logger.info("epoch %d eval %s", epoch, metrics)
                wb.log({f"eval/{k}": v for k, v in metrics.items()})
        if ctx.is_main and (epoch + 1) % save_every_epoch == 0:
            model.save_pretrained(
                os.path.join(save_dir_root, f"epoch_{epoch}"))
        if max_steps is not None and step >= max_steps:
            break
    wb.finish()
    return model


def _make_dataset(ds_cls, mode, folder, split, n_codebooks, codebook_size,
                  rqvae_path, max_samples, max_text_len=None):
    import inspect

    sig = inspect.signature(ds_cls.__init__)
    if "root" in sig.parameters:  # real-data pipelines
        kw = dict(root=folder, split=split, train_test_split=mode,
                  max_samples=max_samples)
        if rqvae_path:
            kw["pretrained_rqvae_path"] = rqvae_path
        if max_text_len is not None and "max_text_len" in sig.parameters:
            kw["max_text_len"] = max_text_len
        return ds_cls(**kw)
    extra = {"split": mode, "sem_id_dim": n_codebooks,
             "codebook_size": codebook_size, "max_samples": max_samples}
    if max_text_len is not None:
        extra["max_text_len"] = max_text_len
    return ds_cls(**common.dataset_kwargs(ds_cls, extra))


if __name__ == "__main__":
    ginlite.parse_config()
    train()
