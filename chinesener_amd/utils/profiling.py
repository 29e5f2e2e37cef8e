"""Profiling helpers (rocprof is the visibility path).

roctx ranges make training phases visible in `rocprofv3 --marker-trace`
timelines; they are no-ops when the marker library is absent (CPU boxes,
plain runs). Also a lightweight per-step wall/samples-per-sec meter used
by the trainer logs."""
from __future__ import annotations

import ctypes
import time
from contextlib import contextmanager
from typing import Optional

_roctx = None
_tried = False


def _lib() -> Optional[ctypes.CDLL]:
    global _roctx, _tried
    if not _tried:
        _tried = True
        for name in ("libroctx64.so", "libroctx64.so.4", "libroctx64.so.1"):
            try:
                _roctx = ctypes.CDLL(name)
                break
            except OSError:
                continue
    return _roctx


def range_push(name: str) -> None:
    lib = _lib()
    if lib is not None:
        lib.roctxRangePushA(name.encode())


def range_pop() -> None:
    lib = _lib()
    if lib is not None:
        lib.roctxRangePop()


@contextmanager
def roctx_range(name: str):
    range_push(name)
    try:
        yield
    finally:
        range_pop()


class StepMeter:
    """Rolling samples/sec + ms/step meter."""

    def __init__(self):
        self.t0 = time.perf_counter()
        self.samples = 0
        self.steps = 0

    def update(self, batch_size: int) -> None:
        self.samples += batch_size
        self.steps += 1

    def rate(self) -> dict:
        dt = max(time.perf_counter() - self.t0, 1e-9)
        return {"samples_per_sec": self.samples / dt,
                "ms_per_step": dt / max(self.steps, 1) * 1000}

    def reset(self) -> None:
        self.t0 = time.perf_counter()
        self.samples = 0
        self.steps = 0
