#!/usr/bin/env python3
"""Per-model training-step benchmarks (BASELINE.json configs 1-3, 5).

Measures train samples/sec for SASRec / HSTU / RQ-VAE (reference shipped
configs, synthetic data, random init) and a scaled LCRec SFT step. The
flagship TIGER benchmark lives in bench.py (the driver contract); this
harness provides the per-model evidence table.

Usage: python benchmarks/bench_models.py [--models sasrec,hstu,rqvae,lcrec]
       [--steps K] [--warmup W] [--out results.jsonl]
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def _timeit(step, steps, warmup, device):
    for i in range(warmup):
        step(i)
    if device.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(steps):
        step(warmup + i)
    if device.type == "cuda":
        torch.cuda.synchronize()
    return time.perf_counter() - t0


def _graph_step(model, batch, loss_getter, device, betas=(0.9, 0.999)):
    """hipGraph-captured full step on CUDA (the trainers' production
    path); eager AdamW fallback elsewhere."""
    if device.type == "cuda":
        from genrec_amd.parallel.graph_runner import GraphedTrainStep

        runner = GraphedTrainStep(model, batch, loss_getter, lr=1e-3,
                                  betas=betas, weight_decay=0.0,
                                  clip_norm=None, world=1, use_graph=True)
        return lambda i: runner.step(batch)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3, betas=betas)

    def step(i):
        opt.zero_grad(set_to_none=False)
        loss_getter(model(**batch)).backward()
        opt.step()

    return step


def bench_sasrec(device, steps, warmup):
    """SASRec: B=128, L=50, D=64, H=2, 2 blocks, V=12101 (sasrec/amazon.gin),
    graph-captured step on GPU."""
    from genrec_amd.models.sasrec import SASRec

    torch.manual_seed(0)
    B, L, V = 128, 50, 12101
    model = SASRec(num_items=V - 1, max_seq_len=L, embed_dim=64, num_heads=2,
                   num_blocks=2, ffn_dim=256, dropout=0.2).to(device)
    ids = torch.randint(1, V, (B, L), device=device)
    ids[::4, :20] = 0
    model.train()
    step = _graph_step(model, {"input_ids": ids, "targets": ids},
                       lambda out: out[1], device, betas=(0.9, 0.98))

    el = _timeit(step, steps, warmup, device)
    return dict(model="sasrec-amazon-beauty", batch=B,
                samples_per_s=B * steps / el,
                ms_per_step=el / steps * 1e3, hip_graph=device.type == "cuda")


def bench_hstu(device, steps, warmup):
    """HSTU bf16: B=128, L=50, D=64, H=2 + temporal bias (hstu/amazon.gin)."""
    from genrec_amd.models.hstu import HSTU

    torch.manual_seed(0)
    B, L, V = 128, 50, 12101
    model = HSTU(num_items=V - 1, max_seq_len=L, embed_dim=64, num_heads=2,
                 num_blocks=2, dropout=0.2, use_temporal_bias=True).to(device)
    ids = torch.randint(1, V, (B, L), device=device)
    ts = (torch.arange(L, device=device) * 86400 + 10 ** 9).unsqueeze(0) \
        .expand(B, -1).contiguous()
    model.train()
    step = _graph_step(
        model, {"input_ids": ids, "timestamps": ts, "targets": ids},
        lambda out: out[1], device, betas=(0.9, 0.98))

    el = _timeit(step, steps, warmup, device)
    return dict(model="hstu-amazon-beauty-bf16", batch=B,
                samples_per_s=B * steps / el,
                ms_per_step=el / steps * 1e3, hip_graph=device.type == "cuda")


def bench_rqvae(device, steps, warmup):
    """RQ-VAE: B=1024, 768->[512,256,128,64]->32, 3x256 codebooks,
    STE + Sinkhorn-last (tiger/amazon/rqvae.gin)."""
    from genrec_amd.models.rqvae import QuantizeForwardMode, RqVae

    torch.manual_seed(0)
    B = 1024
    model = RqVae(input_dim=768, embed_dim=32,
                  hidden_dims=[512, 256, 128, 64], codebook_size=256,
                  codebook_mode=QuantizeForwardMode.STE,
                  codebook_last_layer_mode=QuantizeForwardMode.SINKHORN,
                  n_layers=3, n_cat_features=0).to(device)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3, weight_decay=1e-4)
    x = torch.nn.functional.normalize(torch.randn(B, 768, device=device),
                                      dim=-1)
    model.train()
    model(x, gumbel_t=0.2)  # kmeans init outside the timed region

    def step(i):
        opt.zero_grad(set_to_none=False)
        out = model(x, gumbel_t=0.2)
        out.loss.backward()
        opt.step()

    el = _timeit(step, steps, warmup, device)
    return dict(model="rqvae-amazon-beauty", batch=B,
                samples_per_s=B * steps / el,
                ms_per_step=el / steps * 1e3)


def bench_lcrec(device, steps, warmup, full_size=False):
    """LCRec SFT step: Qwen2-1.5B-shaped backbone (lcrec/amazon/lcrec.gin:
    B=32, L=512, bf16, grad ckpt) — full size only on GPU."""
    from genrec_amd.models.lcrec import LCRec, default_qwen_config

    torch.manual_seed(0)
    B, L = (32, 512) if full_size else (2, 64)
    cfg = default_qwen_config() if full_size else default_qwen_config(
        vocab_size=2048, hidden_size=256, num_layers=4, num_heads=8,
        num_kv_heads=2, intermediate_size=512)
    model = LCRec(config=cfg)
    model.add_codebook_tokens(5, 256)
    if full_size:
        model.gradient_checkpointing_enable()
    model = model.to(device)
    if device.type == "cuda":
        model = model.to(torch.bfloat16)
    V = model.model.config.vocab_size
    ids = torch.randint(0, V, (B, L), device=device)
    attn = torch.ones_like(ids)
    model.train()
    # GraphedTrainStep: fused flat AdamW + steal-grads even when hipGraph
    # capture falls back to eager (HF checkpointing is not capture-safe)
    step = _graph_step(model,
                       {"input_ids": ids, "attention_mask": attn,
                        "labels": ids},
                       lambda out: out.loss, device)

    el = _timeit(step, steps, warmup, device)
    return dict(model="lcrec-qwen2-1.5b" if full_size else "lcrec-tiny",
                batch=B, seq_len=L, samples_per_s=B * steps / el,
                tokens_per_s=B * L * steps / el,
                ms_per_step=el / steps * 1e3)


def bench_cobra(device, steps, warmup):
    """COBRA train step (config/cobra/amazon.gin: B=32, C=3, d_model=384,
    6 heads, 4 layers, T=20 items -> interleaved L=80 on the flash
    attention path), bf16 autocast."""
    from genrec_amd.models.cobra import Cobra

    torch.manual_seed(0)
    B, T, Ltxt = 32, 20, 64
    model = Cobra(encoder_n_layers=2, encoder_hidden_dim=384,
                  encoder_num_heads=6, encoder_vocab_size=32128,
                  id_vocab_size=256, n_codebooks=3, d_model=384,
                  decoder_n_layers=4, decoder_num_heads=6,
                  decoder_dropout=0.1).to(device)
    ids = torch.randint(0, 256, (B, T * 3), device=device)
    enc = torch.randint(1, 32128, (B, T, Ltxt), device=device)
    model.train()
    model.static_infonce = device.type == "cuda"  # capture-safe variant
    step = _graph_step(
        model, {"input_ids": ids, "encoder_input_ids": enc},
        lambda out: out.loss_sparse + out.loss_dense, device)

    el = _timeit(step, steps, warmup, device)
    return dict(model="cobra-amazon-beauty", batch=B,
                samples_per_s=B * steps / el,
                ms_per_step=el / steps * 1e3, hip_graph=device.type == "cuda")


BENCHES = {"sasrec": bench_sasrec, "hstu": bench_hstu, "rqvae": bench_rqvae,
           "lcrec": bench_lcrec, "cobra": bench_cobra}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--models", default="sasrec,hstu,rqvae")
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--lcrec-full", action="store_true")
    p.add_argument("--out", default=None)
    args = p.parse_args()
    device = torch.device("cuda:0") if torch.cuda.is_available() \
        else torch.device("cpu")
    results = []
    for name in args.models.split(","):
        name = name.strip()
        kw = {}
        if name == "lcrec":
            kw["full_size"] = args.lcrec_full
        try:
            r = BENCHES[name](device, args.steps, args.warmup, **kw)
        except Exception as e:  # keep other models running
            print(json.dumps({"model": name, "error": str(e)}), flush=True)
            continue
        r.update(device=str(device), steps=args.steps, warmup=args.warmup,
                 dtype="bf16" if device.type == "cuda" else "fp32",
                 data="synthetic")
        print(json.dumps(r), flush=True)
        results.append(r)
    if args.out:
        with open(args.out, "w") as f:
            for r in results:
                f.write(json.dumps(r) + "\n")


if __name__ == "__main__":
    main()
