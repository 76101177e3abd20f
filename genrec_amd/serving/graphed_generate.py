"""hipGraph-captured TIGER beam decode for serving.

`Tiger.generate` is launch-bound at serving batch sizes: the trie-
constrained beam runs ~1000 small dispatches, so batch-1 latency
(~7.7 ms measured) nearly equals batch-256 latency. Capturing the whole
generate — encoder, 3 KV-cached decode steps, trie masking, Gumbel
top-K, dedup double-sort, re-rank — into one hipGraph removes the
per-dispatch host cost; replays copy new inputs into static buffers.

RNG: torch philox is graph-safe (each replay draws fresh Gumbel noise).
Shapes must be fixed: one captured graph per (batch, K) pair, built
lazily. Falls back to eager generate on capture failure.

Reference pain point this redesigns: tiger.py:312-452 (per-step python
loop over beams; no capture, no KV cache).
"""

from __future__ import annotations

import logging
from typing import Dict, Tuple

import torch

from genrec_amd.models.tiger import Tiger, TigerGenerationOutput

logger = logging.getLogger("genrec_amd")


class GraphedGenerate:
    """Capture-once / replay-per-request wrapper around Tiger.generate."""

    # ATen's multi-block topk was not hipGraph-replay-safe (2nd replay
    # faulted at B=512, ROCm 7); generate now uses sort-based selection
    # (models/tiger.py). Cap kept as a generous guard for huge batches.
    MAX_GRAPH_BATCH = 1024

    def __init__(self, model: Tiger, valid_item_ids: torch.Tensor,
                 n_top_k_candidates: int = 10, temperature: float = 0.2,
                 warmup_iters: int = 2) -> None:
        self.model = model
        self.valid = valid_item_ids
        self.k = n_top_k_candidates
        self.temperature = temperature
        self.warmup_iters = warmup_iters
        self._graphs: Dict[Tuple[int, int], dict] = {}  # (B, L) -> state

    def _build(self, example: Dict[str, torch.Tensor]) -> dict:
        static = {k: v.clone() for k, v in example.items()}

        def run():
            return self.model.generate(
                user_input_ids=static["user_input_ids"],
                item_input_ids=static["item_input_ids"],
                token_type_ids=static["token_type_ids"],
                seq_mask=static["seq_mask"],
                n_top_k_candidates=self.k,
                valid_item_ids=self.valid,
                temperature=self.temperature)

        try:
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(self.warmup_iters):
                    run()
            torch.cuda.current_stream().wait_stream(s)
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                out = run()
            return {"static": static, "graph": graph, "out": out}
        except Exception as e:  # pragma: no cover - device dependent
            logger.warning("GraphedGenerate: capture failed (%s); "
                           "falling back to eager generate", e)
            return {"static": static, "graph": None, "out": None}

    @torch.no_grad()
    def __call__(self, **inputs) -> TigerGenerationOutput:
        key = (inputs["item_input_ids"].size(0),
               inputs["item_input_ids"].size(1))
        if key[0] > self.MAX_GRAPH_BATCH:
            return self.model.generate(
                **inputs, n_top_k_candidates=self.k,
                valid_item_ids=self.valid, temperature=self.temperature)
        state = self._graphs.get(key)
        if state is None:
            state = self._build(inputs)
            self._graphs[key] = state
        if state["graph"] is None:
            return self.model.generate(
                **inputs, n_top_k_candidates=self.k,
                valid_item_ids=self.valid, temperature=self.temperature)
        for k, v in inputs.items():
            state["static"][k].copy_(v, non_blocking=True)
        state["graph"].replay()
        out = state["out"]
        return TigerGenerationOutput(sem_ids=out.sem_ids.clone(),
                                     log_probas=out.log_probas.clone())
