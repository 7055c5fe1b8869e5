"""Structured metrics + step timing.

The reference's only observability is ``time.time()`` deltas printed on rank 0
(reference: multi-gpu-distributed-cls.py:160-161,193-195). Here: a JSONL
metrics writer (step, loss, lr, samples/sec) for the benchmark harness, plus
a CUDA-event-aware step timer.
"""

from __future__ import annotations

import json
import time
from typing import Optional

import torch


class MetricsWriter:
    def __init__(self, path: Optional[str], rank: int = 0):
        self.path = path if rank == 0 else None
        self._f = open(path, "a") if self.path else None

    def write(self, **kv) -> None:
        if self._f is None:
            return
        kv.setdefault("ts", time.time())
        self._f.write(json.dumps(kv) + "\n")
        self._f.flush()

    def close(self) -> None:
        if self._f:
            self._f.close()
            self._f = None


class StepTimer:
    """Wall-clock step timer; synchronizes the device only when asked."""

    def __init__(self, device: Optional[torch.device] = None):
        self.device = device
        self.reset()

    def reset(self) -> None:
        self._t0 = time.perf_counter()
        self.steps = 0
        self.samples = 0

    def step(self, n_samples: int = 0) -> None:
        self.steps += 1
        self.samples += n_samples

    def elapsed(self, sync: bool = False) -> float:
        if sync and self.device is not None and self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        return time.perf_counter() - self._t0

    def samples_per_sec(self, sync: bool = False) -> float:
        e = self.elapsed(sync)
        return self.samples / e if e > 0 else 0.0


class TraceRange:
    """roctx-style range annotation; no-op off-GPU.

    Shows fwd/bwd/comm phases in rocprofv3 runtime traces (SURVEY.md §5.1:
    the rebuild adds first-class tracing the reference lacks)."""

    def __init__(self, name: str):
        self.name = name
        self._push = None
        try:
            if torch.cuda.is_available():
                from torch.cuda import nvtx  # maps to roctx on ROCm
                self._push = nvtx
        except Exception:
            self._push = None

    def __enter__(self):
        if self._push:
            self._push.range_push(self.name)
        return self

    def __exit__(self, *a):
        if self._push:
            self._push.range_pop()
        return False
