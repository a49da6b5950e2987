"""Tracing / profiling hooks.

The reference has no built-in tracing; the MI355X build makes
rocprof-friendly instrumentation first-class: rocTX ranges (visible in
rocprofv3 --marker-trace timelines), a torch.profiler context preconfigured
for ROCm, and a step-timer that brackets with hipDeviceSynchronize the way
bench.py does.
"""

from __future__ import annotations

import contextlib
import time
from typing import Dict, Iterator, Optional

_ROCTX = None


def _roctx():
    global _ROCTX
    if _ROCTX is None:
        try:
            from ctypes import CDLL

            _ROCTX = CDLL("libroctx64.so")
        except OSError:
            _ROCTX = False
    return _ROCTX


@contextlib.contextmanager
def roctx_range(name: str) -> Iterator[None]:
    """rocTX range marker; no-op when the runtime is unavailable (CPU)."""
    lib = _roctx()
    if lib:
        lib.roctxRangePushA(name.encode())
    try:
        yield
    finally:
        if lib:
            lib.roctxRangePop()


@contextlib.contextmanager
def torch_profile(out_dir: str = "profiles/torch_trace", with_stack: bool = False):
    """torch.profiler configured for the ROCm backend; exports a chrome
    trace under ``out_dir``."""
    import torch
    from torch.profiler import ProfilerActivity, profile

    activities = [ProfilerActivity.CPU]
    if torch.cuda.is_available():
        activities.append(ProfilerActivity.CUDA)
    with profile(activities=activities, with_stack=with_stack, record_shapes=True) as prof:
        yield prof
    import os

    os.makedirs(out_dir, exist_ok=True)
    prof.export_chrome_trace(os.path.join(out_dir, f"trace_{int(time.time())}.json"))


class StepTimer:
    """Per-step wall timing with device synchronization, aggregated."""

    def __init__(self) -> None:
        self.times: Dict[str, list] = {}

    @contextlib.contextmanager
    def time(self, name: str) -> Iterator[None]:
        import torch

        if torch.cuda.is_available():
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        with roctx_range(name):
            yield
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        self.times.setdefault(name, []).append(time.perf_counter() - t0)

    def summary(self) -> Dict[str, float]:
        return {name: sum(vals) / len(vals) for name, vals in self.times.items() if vals}
