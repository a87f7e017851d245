"""Observability: roctx ranges + per-step throughput tracking.

The reference has no tracing/profiling at all. Here: optional
roctx range markers around the training-step phases (visible in rocprofv3
--sys-trace timelines) and a lightweight step-throughput tracker for the
structured logs the trainer emits.
"""

from __future__ import annotations

import contextlib
import ctypes
import time


class _Roctx:
    """Lazy binding to libroctx64 (present in /opt/rocm); silently inert when
    the library is unavailable (e.g. CPU-only dev sandbox)."""

    def __init__(self):
        self._lib = None
        self._tried = False

    def _load(self):
        if self._tried:
            return self._lib
        self._tried = True
        for name in ("libroctx64.so", "libroctx64.so.4", "libroctx64.so.1"):
            try:
                self._lib = ctypes.CDLL(name)
                break
            except OSError:
                continue
        return self._lib

    def push(self, msg: str):
        lib = self._load()
        if lib is not None:
            lib.roctxRangePushA(msg.encode())

    def pop(self):
        lib = self._load()
        if lib is not None:
            lib.roctxRangePop()


roctx = _Roctx()


@contextlib.contextmanager
def trace_range(name: str, enabled: bool = True):
    """roctx range context manager: `with trace_range("forward"): ...`"""
    if enabled:
        roctx.push(name)
    try:
        yield
    finally:
        if enabled:
            roctx.pop()


class ThroughputMeter:
    """Tracks samples/sec over a sliding window of steps."""

    def __init__(self, window: int = 50):
        self.window = window
        self._times: list[float] = []
        self._samples: list[int] = []

    def step(self, n_samples: int):
        now = time.perf_counter()
        self._times.append(now)
        self._samples.append(n_samples)
        if len(self._times) > self.window + 1:
            self._times.pop(0)
            self._samples.pop(0)

    @property
    def samples_per_sec(self) -> float:
        if len(self._times) < 2:
            return 0.0
        dt = self._times[-1] - self._times[0]
        return sum(self._samples[1:]) / dt if dt > 0 else 0.0
