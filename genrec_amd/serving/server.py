"""Online serving for TIGER generative retrieval.

The reference only sketches a FastAPI deployment in prose with
nonexistent APIs (docs/en/deployment.md:9-100 — SURVEY.md §1 notes it
calls `RqVae.load_from_checkpoint` which does not exist); this module is a
working implementation:

  * RecommendationService: loads a trained TIGER checkpoint (reference
    dict layout) + the item sem-ID table, builds the device trie once, and
    serves batched trie-constrained generation
  * create_app(): FastAPI app with /health, /recommend and /batch_recommend
  * micro-batching: concurrent requests within a small window are fused
    into one GPU generate call (MI355X throughput comes from batching)

Run: python -m genrec_amd.serving.server --checkpoint ck.pt --sem-ids ids.pt
"""

from __future__ import annotations

import argparse
import asyncio
import time
from typing import Dict, List, Optional

import torch

from genrec_amd.models.tiger import Tiger


class RecommendationService:
    def __init__(self, model: Tiger, item_sem_ids: torch.Tensor,
                 device: Optional[torch.device] = None,
                 max_items_per_seq: int = 20, top_k: int = 10,
                 use_graph: bool = True) -> None:
        self.device = device or (
            torch.device("cuda:0") if torch.cuda.is_available()
            else torch.device("cpu"))
        self.model = model.to(self.device).eval()
        self.item_sem_ids = item_sem_ids.to(self.device)
        self.sem_dim = model.sem_id_dim
        self.max_items = max_items_per_seq
        self.top_k = top_k
        # tuple -> item index for mapping generated sem-IDs back to items
        self._tuple_to_item: Dict[tuple, int] = {
            tuple(row): i for i, row in enumerate(item_sem_ids.tolist())}
        # warm the trie
        with torch.no_grad():
            self.model.generate(
                torch.zeros(1, 1, dtype=torch.long, device=self.device),
                self.item_sem_ids[:1].reshape(1, -1),
                torch.arange(self.sem_dim, device=self.device).unsqueeze(0),
                torch.ones(1, self.sem_dim, dtype=torch.long,
                           device=self.device),
                n_top_k_candidates=1, valid_item_ids=self.item_sem_ids)
        # hipGraph-captured decode (one graph per batch-shape) removes the
        # per-dispatch host cost of the launch-bound beam
        self._graphed = None
        if use_graph and self.device.type == "cuda":
            from genrec_amd.serving.graphed_generate import GraphedGenerate

            self._graphed = GraphedGenerate(
                self.model, self.item_sem_ids, n_top_k_candidates=top_k)

    @torch.no_grad()
    def recommend_batch(self, user_ids: List[int],
                        histories: List[List[int]],
                        top_k: Optional[int] = None) -> List[List[Dict]]:
        """histories: per request, a list of item indices (0-based)."""
        k = top_k or self.top_k
        b = len(histories)
        dim = self.sem_dim
        use_graph = self._graphed is not None and k == self.top_k
        if use_graph:
            # fixed shapes so ragged traffic reuses a handful of graphs:
            # histories padded to max_items, batch rounded up to the next
            # power of two (dummy rows sliced off the result)
            max_len = self.max_items
            b_pad = 1
            while b_pad < b:
                b_pad *= 2
        else:
            max_len = max(max((len(h) for h in histories), default=1), 1)
            max_len = min(max_len, self.max_items)
            b_pad = b
        L = max_len * dim
        item_ids = torch.zeros(b_pad, L, dtype=torch.long)
        mask = torch.zeros(b_pad, L, dtype=torch.long)
        ttype = (torch.arange(L) % dim).unsqueeze(0).expand(b_pad, -1) \
            .clone()
        n_items = self.item_sem_ids.size(0)
        sem_cpu = self.item_sem_ids.cpu()
        for i, h in enumerate(histories):
            # drop out-of-catalog indices instead of crashing the batch
            h = [x for x in h if 0 <= x < n_items][-max_len:]
            flat = sem_cpu[torch.tensor(h, dtype=torch.long)].reshape(-1) \
                if h else torch.zeros(0, dtype=torch.long)
            item_ids[i, :flat.numel()] = flat
            mask[i, :flat.numel()] = 1
        users = torch.tensor(
            list(user_ids) + [0] * (b_pad - b), dtype=torch.long) \
            .unsqueeze(1)
        if use_graph:
            gen = self._graphed(
                user_input_ids=users.to(self.device),
                item_input_ids=item_ids.to(self.device),
                token_type_ids=ttype.to(self.device),
                seq_mask=mask.to(self.device))
        else:
            gen = self.model.generate(
                users.to(self.device), item_ids.to(self.device),
                ttype.to(self.device), mask.to(self.device),
                n_top_k_candidates=k, valid_item_ids=self.item_sem_ids)
        out: List[List[Dict]] = []
        for i in range(b):
            row = []
            for j in range(k):
                tup = tuple(gen.sem_ids[i, j].tolist())
                score = float(gen.log_probas[i, j])
                if score <= -1e30:
                    continue
                row.append({
                    "item_id": self._tuple_to_item.get(tup, -1),
                    "sem_ids": list(tup),
                    "score": score,
                })
            out.append(row)
        return out

    @classmethod
    def from_checkpoint(cls, checkpoint_path: str, sem_ids_path: str,
                        model_kwargs: Optional[dict] = None,
                        **kwargs) -> "RecommendationService":
        state = torch.load(checkpoint_path, map_location="cpu",
                           weights_only=False)
        sd = state["model"] if "model" in state else state
        # model_config saved by the trainers wins unless overridden
        mk = dict(state.get("model_config") or {}) \
            if isinstance(state, dict) else {}
        mk.update(model_kwargs or {})
        model = Tiger(**mk)
        model.load_state_dict(sd)
        sem = torch.load(sem_ids_path, map_location="cpu",
                         weights_only=False)
        if isinstance(sem, dict):
            sem = sem["sem_ids"]
        return cls(model, sem.long(), **kwargs)


class _MicroBatcher:
    """Fuse concurrent requests into one generate call."""

    def __init__(self, service: RecommendationService,
                 max_batch: int = 64, window_ms: float = 2.0):
        self.service = service
        self.max_batch = max_batch
        self.window = window_ms / 1000.0
        self._queue: List = []
        self._lock = asyncio.Lock()
        self._event: Optional[asyncio.Event] = None

    async def submit(self, user_id: int, history: List[int], top_k: int):
        loop = asyncio.get_event_loop()
        fut = loop.create_future()
        async with self._lock:
            self._queue.append((user_id, history, top_k, fut))
            if len(self._queue) == 1:
                loop.create_task(self._flush_later())
            if len(self._queue) >= self.max_batch:
                await self._flush()
        return await fut

    async def _flush_later(self):
        await asyncio.sleep(self.window)
        async with self._lock:
            await self._flush()

    async def _flush(self):
        if not self._queue:
            return
        batch, self._queue = self._queue, []
        users = [b[0] for b in batch]
        hists = [b[1] for b in batch]
        k = max(b[2] for b in batch)
        loop = asyncio.get_event_loop()
        results = await loop.run_in_executor(
            None, lambda: self.service.recommend_batch(users, hists, k))
        for (u, h, tk, fut), res in zip(batch, results):
            if not fut.done():
                fut.set_result(res[:tk])


try:  # request schemas at module level (FastAPI resolves annotations here)
    from pydantic import BaseModel

    class RecommendRequest(BaseModel):
        user_id: int = 0
        history: List[int] = []
        top_k: int = 10

    class BatchRecommendRequest(BaseModel):
        user_ids: List[int]
        histories: List[List[int]]
        top_k: int = 10
except ImportError:  # pragma: no cover
    RecommendRequest = BatchRecommendRequest = None


def create_app(service: RecommendationService, max_batch: int = 64,
               window_ms: float = 2.0):
    from fastapi import FastAPI, Response

    app = FastAPI(title="genrec_amd TIGER serving")
    batcher = _MicroBatcher(service, max_batch=max_batch,
                            window_ms=window_ms)

    # Prometheus observability (the reference's deployment doc only
    # sketches monitoring; this is a working /metrics endpoint)
    try:
        from prometheus_client import (CollectorRegistry, Counter,
                                       Histogram, generate_latest)

        registry = CollectorRegistry()
        req_count = Counter("genrec_requests_total",
                            "recommendation requests", ["endpoint"],
                            registry=registry)
        req_lat = Histogram(
            "genrec_request_latency_seconds", "request latency",
            ["endpoint"], registry=registry,
            buckets=(.002, .005, .01, .025, .05, .1, .25, .5, 1., 2.5))
    except ImportError:  # pragma: no cover
        registry = None

    def _observe(endpoint: str, t0: float) -> float:
        dt = time.perf_counter() - t0
        if registry is not None:
            req_count.labels(endpoint).inc()
            req_lat.labels(endpoint).observe(dt)
        return dt * 1000.0

    @app.get("/health")
    def health():
        return {"status": "ok", "device": str(service.device),
                "num_items": service.item_sem_ids.size(0)}

    @app.get("/metrics")
    def metrics():
        if registry is None:
            return Response("prometheus_client not installed",
                            status_code=501)
        return Response(generate_latest(registry),
                        media_type="text/plain; version=0.0.4")

    @app.post("/recommend")
    async def recommend(req: "RecommendRequest"):
        t0 = time.perf_counter()
        recs = await batcher.submit(req.user_id, req.history, req.top_k)
        return {"recommendations": recs,
                "latency_ms": _observe("recommend", t0)}

    @app.post("/batch_recommend")
    def batch_recommend(req: "BatchRecommendRequest"):
        t0 = time.perf_counter()
        recs = service.recommend_batch(req.user_ids, req.histories,
                                       req.top_k)
        return {"recommendations": recs,
                "latency_ms": _observe("batch_recommend", t0)}

    return app


def main():
    import uvicorn

    p = argparse.ArgumentParser()
    p.add_argument("--checkpoint", required=True)
    p.add_argument("--sem-ids", required=True,
                   help="torch file with [N, sem_id_dim] item sem-IDs")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8000)
    p.add_argument("--embedding-dim", type=int, default=128)
    p.add_argument("--attn-dim", type=int, default=384)
    p.add_argument("--num-heads", type=int, default=6)
    p.add_argument("--n-layers", type=int, default=8)
    p.add_argument("--num-item-embeddings", type=int, default=256)
    p.add_argument("--num-user-embeddings", type=int, default=10000)
    p.add_argument("--sem-id-dim", type=int, default=3)
    args = p.parse_args()
    svc = RecommendationService.from_checkpoint(
        args.checkpoint, args.sem_ids,
        model_kwargs=dict(
            embedding_dim=args.embedding_dim, attn_dim=args.attn_dim,
            dropout=0.0, num_heads=args.num_heads, n_layers=args.n_layers,
            num_item_embeddings=args.num_item_embeddings,
            num_user_embeddings=args.num_user_embeddings,
            sem_id_dim=args.sem_id_dim))
    uvicorn.run(create_app(svc), host=args.host, port=args.port)


if __name__ == "__main__":
    main()
