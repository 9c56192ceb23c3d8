"""Metrics logging + tracing markers (SURVEY.md §5 'tracing/metrics').

The reference's only instrumentation is print statements
(`main_sac.py:71-72`, `enet_sac.py:608-625`). This module provides:

* :class:`MetricsLogger` — structured per-episode/per-step records to a
  JSONL file (+ optional stdout mirror);
* :func:`trace_range` — rocprof-visible range markers: on ROCm,
  ``torch.cuda.nvtx`` maps to roctx, so ranges show up in
  `rocprofv3 --marker-trace` timelines; a no-op on CPU.
"""

from __future__ import annotations

import contextlib
import json
import time

import torch

__all__ = ["MetricsLogger", "trace_range"]


class MetricsLogger:
    def __init__(self, path: str | None = None, stdout: bool = True):
        self.path = path
        self.stdout = stdout
        self._fh = open(path, "a") if path else None
        self._t0 = time.time()

    def log(self, kind: str, **fields):
        rec = {"t": round(time.time() - self._t0, 4), "kind": kind,
               **fields}
        if self._fh:
            self._fh.write(json.dumps(rec) + "\n")
            self._fh.flush()
        if self.stdout:
            print(" ".join(f"{k}={v}" for k, v in rec.items()))

    def episode(self, i: int, score: float, avg: float, **extra):
        self.log("episode", episode=i, score=round(float(score), 4),
                 avg100=round(float(avg), 4), **extra)

    def close(self):
        if self._fh:
            self._fh.close()


@contextlib.contextmanager
def trace_range(name: str):
    """roctx range marker (visible in rocprofv3 --marker-trace)."""
    if torch.cuda.is_available():
        torch.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield
