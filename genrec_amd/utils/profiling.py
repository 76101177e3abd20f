"""Tracing / profiling utilities (SURVEY.md §5.1 — the reference has none).

* roctx_range: roctx markers via torch.cuda.nvtx (which maps onto
  roctracer/roctx on ROCm builds) so rocprofv3 --marker-trace groups
  kernels by phase (data / forward / backward / optimizer).
* StepProfiler: torch.profiler (kineto -> roctracer) wrapper for a
  `profile_steps` trainer flag; prints a per-kernel time table after the
  profiled window and optionally exports a chrome trace.
"""

from __future__ import annotations

import contextlib
import logging
from typing import Optional

import torch

logger = logging.getLogger("genrec_amd")


@contextlib.contextmanager
def roctx_range(name: str):
    """Marker range; no-op off-GPU."""
    if torch.cuda.is_available():
        torch.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield


class StepProfiler:
    """Profile a window of training steps.

    Usage:
        prof = StepProfiler(enabled=args.profile_steps > 0,
                            wait=2, active=args.profile_steps)
        for step, batch in enumerate(loader):
            with prof.step():
                train_step(batch)
        prof.report()
    """

    def __init__(self, enabled: bool = False, wait: int = 2,
                 active: int = 5, trace_path: Optional[str] = None):
        self.enabled = enabled and torch.cuda.is_available()
        self.trace_path = trace_path
        self._prof = None
        if self.enabled:
            self._prof = torch.profiler.profile(
                activities=[torch.profiler.ProfilerActivity.CPU,
                            torch.profiler.ProfilerActivity.CUDA],
                schedule=torch.profiler.schedule(wait=wait, warmup=1,
                                                 active=active, repeat=1),
                record_shapes=False, with_stack=False)
            self._prof.__enter__()

    @contextlib.contextmanager
    def step(self):
        yield
        if self._prof is not None:
            self._prof.step()

    def report(self, top: int = 25) -> Optional[str]:
        if self._prof is None:
            return None
        self._prof.__exit__(None, None, None)
        table = self._prof.key_averages().table(
            sort_by="self_cuda_time_total", row_limit=top)
        logger.info("profiler kernel table:\n%s", table)
        if self.trace_path:
            self._prof.export_chrome_trace(self.trace_path)
            logger.info("chrome trace written to %s", self.trace_path)
        return table
