"""Serving latency under concurrent load (BACKLOG: micro-batcher).

Drives RecommendationService through the async micro-batcher with many
concurrent requesters and reports throughput + latency percentiles per
concurrency level. Uses the hipGraph-captured decode on GPU.

Run on GPU: python tools/bench_serving.py [--seconds 3]
"""

from __future__ import annotations

import argparse
import asyncio
import json
import os
import random
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


async def run_level(batcher, concurrency: int, seconds: float, n_items: int):
    lat = []
    stop = time.perf_counter() + seconds

    async def worker(wid: int):
        rng = random.Random(wid)
        while time.perf_counter() < stop:
            hist = [rng.randrange(n_items) for _ in range(rng.randint(3, 20))]
            t0 = time.perf_counter()
            await batcher.submit(rng.randrange(10000), hist, 10)
            lat.append(time.perf_counter() - t0)

    await asyncio.gather(*(worker(i) for i in range(concurrency)))
    lat.sort()

    def pct(p):
        return lat[min(len(lat) - 1, int(p * len(lat)))] * 1e3

    return {
        "concurrency": concurrency,
        "requests": len(lat),
        "rps": len(lat) / seconds,
        "p50_ms": round(pct(0.50), 2),
        "p95_ms": round(pct(0.95), 2),
        "p99_ms": round(pct(0.99), 2),
    }


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--seconds", type=float, default=3.0)
    p.add_argument("--window-ms", type=float, default=2.0)
    p.add_argument("--max-batch", type=int, default=64)
    args = p.parse_args()

    from genrec_amd.models.tiger import Tiger
    from genrec_amd.serving.server import RecommendationService, _MicroBatcher

    torch.manual_seed(0)
    model = Tiger(embedding_dim=128, attn_dim=384, dropout=0.0, num_heads=6,
                  n_layers=8, num_item_embeddings=256,
                  num_user_embeddings=10000, sem_id_dim=3)
    if torch.cuda.is_available():
        model = model.to(torch.bfloat16)
    n_items = 12101
    sem = torch.randint(0, 256, (n_items, 3))
    svc = RecommendationService(model, sem, top_k=10)
    batcher = _MicroBatcher(svc, max_batch=args.max_batch,
                            window_ms=args.window_ms)

    results = []
    for conc in (1, 8, 32, 128):
        r = asyncio.run(run_level(batcher, conc, args.seconds, n_items))
        results.append(r)
        print(json.dumps(r), flush=True)
    print(json.dumps({"metric": "serving_load", "device":
                      "cuda" if torch.cuda.is_available() else "cpu",
                      "window_ms": args.window_ms,
                      "max_batch": args.max_batch, "levels": results}))


if __name__ == "__main__":
    main()
