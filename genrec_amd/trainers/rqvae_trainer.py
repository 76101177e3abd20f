"""RQ-VAE trainer (parity: reference trainers/rqvae_trainer.py, 447 LoC).

Epoch- or iteration-based loop (mutually exclusive, rqvae_trainer.py:91-96),
20k-sample big-batch kmeans warmup on step 0 (rqvae_trainer.py:218-228),
collision-rate eval over the full train set (rqvae_trainer.py:26-47),
reference-layout dict checkpoints including model_config
(rqvae_trainer.py:315-324).
"""

from __future__ import annotations

import os
from typing import Optional

import torch
from torch.optim import AdamW
from torch.utils.data import DataLoader

from genrec_amd.config import ginlite
from genrec_amd.data.synthetic import SyntheticItemDataset
from genrec_amd.models.rqvae import QuantizeForwardMode, RqVae
from genrec_amd.modules.schedulers import get_linear_schedule_with_warmup
from genrec_amd.parallel import GradReducer, init_distributed
from genrec_amd.parallel.ddp import broadcast_parameters
from genrec_amd.trainers import common
from genrec_amd.trainers.common import logger


@torch.no_grad()
def compute_collision_rate(model: RqVae, loader, device) -> float:
    """Fraction of items whose full sem-ID tuple collides (ref
    rqvae_trainer.py:26-47) — computed on-device with tuple hashing."""
    model.eval()
    keys = []
    for batch in loader:
        x = common.to_device(batch, device)
        out = model.get_semantic_ids(x)
        ids = out.sem_ids  # [B, n_layers]
        key = torch.zeros(ids.size(0), dtype=torch.long, device=device)
        for l in range(ids.size(1)):
            key = key * model.codebook_size + ids[:, l]
        keys.append(key)
    keys = torch.cat(keys)
    n_unique = keys.unique().numel()
    return 1.0 - n_unique / keys.numel()


@ginlite.configurable(name="train")
def train(
    epochs: Optional[int] = 5000,
    iterations: Optional[int] = None,
    warmup_epochs: int = 50,
    learning_rate: float = 1e-3,
    weight_decay: float = 1e-4,
    batch_size: int = 1024,
    vae_input_dim: int = 768,
    vae_n_cat_feats: int = 0,
    vae_hidden_dims=[512, 256, 128, 64],
    vae_embed_dim: int = 32,
    vae_codebook_size: int = 256,
    vae_codebook_normalize: bool = False,
    vae_sim_vq: bool = False,
    vae_n_layers: int = 3,
    vae_codebook_mode: QuantizeForwardMode = QuantizeForwardMode.STE,
    vae_codebook_last_layer_mode: QuantizeForwardMode = QuantizeForwardMode.SINKHORN,
    commitment_weight: float = 0.25,
    gumbel_temperature: float = 0.2,
    dataset=None,
    dataset_folder: str = "dataset/amazon",
    encoder_model_name: Optional[str] = None,
    kmeans_warmup_samples: int = 20000,
    save_model_every: int = 50,
    eval_every: int = 50,
    save_dir_root: str = "out/rqvae",
    wandb_logging: bool = False,
    wandb_project: str = "rqvae",
    wandb_log_interval: int = 100,
    do_eval: bool = True,
    seed: int = 42,
    max_steps: Optional[int] = None,
    resume_path: Optional[str] = None,
    num_workers: int = 4,
):
    assert (epochs is None) != (iterations is None), \
        "epochs and iterations are mutually exclusive (ref rqvae_trainer.py:91-96)"
    common.enable_tuned_gemms()
    ctx = init_distributed()
    common.setup_logging(save_dir_root if ctx.is_main else None, "rqvae")
    common.set_seed(seed, ctx.rank)
    device = ctx.device

    ds_cls = dataset or SyntheticItemDataset
    train_ds = _make_dataset(ds_cls, dataset_folder, "train")
    train_loader = common.make_loader(train_ds, batch_size, ctx, True,
                                      num_workers=num_workers, seed=seed)
    eval_loader = DataLoader(train_ds, batch_size=batch_size, shuffle=False,
                             num_workers=0)

    model = RqVae(
        input_dim=vae_input_dim, embed_dim=vae_embed_dim,
        hidden_dims=list(vae_hidden_dims), codebook_size=vae_codebook_size,
        codebook_kmeans_init=True, codebook_normalize=vae_codebook_normalize,
        codebook_sim_vq=vae_sim_vq, codebook_mode=vae_codebook_mode,
        codebook_last_layer_mode=vae_codebook_last_layer_mode,
        n_layers=vae_n_layers, commitment_weight=commitment_weight,
        n_cat_features=vae_n_cat_feats).to(device)

    # kmeans warmup with a big batch BEFORE DDP broadcast so every rank gets
    # rank-0's initialized codebooks (ref rqvae_trainer.py:218-228)
    if ctx.is_main or ctx.world_size == 1:
        warm_n = min(kmeans_warmup_samples, len(train_ds))
        warm = torch.stack([train_ds[i] for i in range(warm_n)]).to(device)
        model.train()
        with torch.no_grad():
            model.get_semantic_ids(warm, gumbel_t=gumbel_temperature)
    for q in model.layers:
        q.kmeans_initted = True
    broadcast_parameters(model)

    opt = AdamW(model.parameters(), lr=learning_rate,
                weight_decay=weight_decay)
    total_steps = (iterations if iterations is not None
                   else epochs * max(1, len(train_loader)))
    sched = get_linear_schedule_with_warmup(
        opt, warmup_epochs * max(1, len(train_loader)), total_steps)
    reducer = GradReducer(model)

    start_epoch, step = 0, 0
    if resume_path and os.path.exists(resume_path):
        state = common.load_checkpoint(resume_path, model, opt, sched,
                                       map_location=device)
        start_epoch = state.get("epoch", -1) + 1
        step = state.get("iter", 0)

    wb = common.init_wandb(wandb_project, {"model": "rqvae"},
                           wandb_logging, ctx.is_main)

    n_epochs = epochs if epochs is not None else 10 ** 9
    done = False
    for epoch in range(start_epoch, n_epochs):
        model.train()
        if hasattr(train_loader.sampler, "set_epoch"):
            train_loader.sampler.set_epoch(epoch)
        for batch in train_loader:
            x = common.to_device(batch, device)
            opt.zero_grad(set_to_none=False)
            out = model(x, gumbel_t=gumbel_temperature)
            out.loss.backward()
            reducer.finalize()
            torch.nn.utils.clip_grad_norm_(model.parameters(), 1.0)
            opt.step()
            sched.step()
            step += 1
            if ctx.is_main and step % wandb_log_interval == 0:
                logger.info(
                    "epoch %d step %d loss %.4f recon %.4f vq %.4f uniq %.3f",
                    epoch, step, out.loss.item(),
                    out.reconstruction_loss.item(), out.rqvae_loss.item(),
                    out.p_unique_ids.item())
                wb.log({"train/loss": out.loss.item(),
                        "train/p_unique_ids": out.p_unique_ids.item()})
            if (iterations is not None and step >= iterations) or \
                    (max_steps is not None and step >= max_steps):
                done = True
                break
        if do_eval and (epoch + 1) % eval_every == 0 and ctx.is_main:
            cr = compute_collision_rate(model, eval_loader, device)
            logger.info("epoch %d collision_rate %.4f", epoch, cr)
            wb.log({"eval/collision_rate": cr})
            model.train()
        if ctx.is_main and (epoch + 1) % save_model_every == 0:
            common.save_checkpoint(
                os.path.join(save_dir_root, f"checkpoint_epoch_{epoch}.pt"),
                model, opt, sched, epoch=epoch,
                model_config=model.config, is_main=True)
        if done:
            break
    if ctx.is_main:
        common.save_checkpoint(
            os.path.join(save_dir_root, "checkpoint_final.pt"),
            model, opt, sched, epoch=epoch, step=step,
            model_config=model.config, is_main=True)
    wb.finish()
    return model


def _make_dataset(ds_cls, folder, mode):
    import inspect

    sig = inspect.signature(ds_cls.__init__)
    kwargs = {}
    if "split" in sig.parameters:
        kwargs["split"] = mode
    if "root" in sig.parameters:
        kwargs["root"] = folder
    return ds_cls(**kwargs)


if __name__ == "__main__":
    ginlite.parse_config()
    train()
