#!/usr/bin/env python3
"""TIGER inference benchmark: trie-constrained stochastic beam search
throughput/latency (the serving path — genrec_amd/serving/server.py).

Random-init TIGER at the Amazon-Beauty config, synthetic catalog of 12k
items with 3x256 semantic IDs, top-10 retrieval. Reports users/s at a
serving batch and per-request latency at small batches.

Usage: python benchmarks/bench_generate.py [--steps 20]
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--batch", type=int, default=256)
    p.add_argument("--topk", type=int, default=10)
    args = p.parse_args()

    from genrec_amd.models.tiger import Tiger

    device = torch.device("cuda:0") if torch.cuda.is_available() \
        else torch.device("cpu")
    torch.manual_seed(0)
    model = Tiger(embedding_dim=128, attn_dim=384, dropout=0.0, num_heads=6,
                  n_layers=8, num_item_embeddings=256,
                  num_user_embeddings=10000, sem_id_dim=3).to(device)
    if device.type == "cuda":
        model = model.to(torch.bfloat16)
    model.eval()

    n_items = 12101
    valid = torch.randint(0, 256, (n_items, 3), device=device)

    def make_batch(B, seed):
        g = torch.Generator().manual_seed(seed)
        L = 20 * 3
        return {
            "user_input_ids": torch.randint(0, 10000, (B, 1), generator=g
                                            ).to(device),
            "item_input_ids": torch.randint(0, 256, (B, L), generator=g
                                            ).to(device),
            "token_type_ids": (torch.arange(L) % 3).repeat(B, 1).to(device),
            "seq_mask": torch.ones(B, L, dtype=torch.long, device=device),
        }

    results = {}
    graphed = None
    if device.type == "cuda":
        from genrec_amd.serving.graphed_generate import GraphedGenerate

        graphed = GraphedGenerate(model, valid, n_top_k_candidates=args.topk)
    with torch.no_grad():
        for B in (1, 32, args.batch):
            batches = [make_batch(B, 7 + i) for i in range(4)]
            for kv in (True, False):
                for i in range(args.warmup):
                    model.generate(**batches[i % 4],
                                   n_top_k_candidates=args.topk,
                                   valid_item_ids=valid, use_kv_cache=kv)
                if device.type == "cuda":
                    torch.cuda.synchronize()
                t0 = time.perf_counter()
                for i in range(args.steps):
                    model.generate(**batches[i % 4],
                                   n_top_k_candidates=args.topk,
                                   valid_item_ids=valid, use_kv_cache=kv)
                if device.type == "cuda":
                    torch.cuda.synchronize()
                el = time.perf_counter() - t0
                key = f"batch_{B}" + ("" if kv else "_nocache")
                results[key] = {
                    "users_per_s": B * args.steps / el,
                    "ms_per_batch": el / args.steps * 1e3,
                }
            if graphed is not None:
                gin = [{k: v for k, v in b.items()} for b in batches]
                for i in range(args.warmup):
                    graphed(**gin[i % 4])
                torch.cuda.synchronize()
                t0 = time.perf_counter()
                for i in range(args.steps):
                    graphed(**gin[i % 4])
                torch.cuda.synchronize()
                el = time.perf_counter() - t0
                results[f"batch_{B}_graphed"] = {
                    "users_per_s": B * args.steps / el,
                    "ms_per_batch": el / args.steps * 1e3,
                }
    print(json.dumps({
        "metric": "tiger_generate", "topk": args.topk,
        "n_items": n_items, "device": str(device),
        "dtype": "bf16" if device.type == "cuda" else "fp32",
        **results}))


if __name__ == "__main__":
    main()
