"""Tracing / profiling subsystem (absent in the reference —
the MI355X framework provides per-step timing meters, torch.profiler
integration, and RCCL debug plumbing).

Usage:
    meter = StepTimer(device)
    with meter:
        ... one training step ...
    meter.ms  # wall ms of the step (device-synchronized)

    with torch_profile("trace_out"):   # chrome trace for rocprof-less boxes
        ...

    enable_rccl_debug()  # NCCL_DEBUG=INFO etc. before init_process_group

On-GPU kernel-level profiling uses rocprofv3 externally:
    rocprofv3 --kernel-trace --stats -d out/ -- python bench.py ...
"""

from __future__ import annotations

import contextlib
import os
import time
from collections import deque
from typing import Optional

import torch


class StepTimer:
    """Device-synchronized wall-clock step timer with a running window."""

    def __init__(self, device: Optional[torch.device] = None, window: int = 50):
        self.device = device
        self.history = deque(maxlen=window)
        self.ms = 0.0

    def __enter__(self):
        if self.device is not None and self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        self._t0 = time.perf_counter()
        return self

    def __exit__(self, *exc):
        if self.device is not None and self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        self.ms = (time.perf_counter() - self._t0) * 1000.0
        self.history.append(self.ms)
        return False

    @property
    def mean_ms(self) -> float:
        return sum(self.history) / len(self.history) if self.history else 0.0


@contextlib.contextmanager
def torch_profile(out_dir: str, record_shapes: bool = False, active: int = 3):
    """torch.profiler context writing a chrome trace to out_dir."""
    os.makedirs(out_dir, exist_ok=True)
    activities = [torch.profiler.ProfilerActivity.CPU]
    if torch.cuda.is_available():
        activities.append(torch.profiler.ProfilerActivity.CUDA)
    with torch.profiler.profile(
        activities=activities, record_shapes=record_shapes
    ) as prof:
        yield prof
    prof.export_chrome_trace(os.path.join(out_dir, f"trace_{os.getpid()}.json"))


def enable_rccl_debug(level: str = "INFO") -> None:
    """RCCL debug env plumbing — call before init_process_group."""
    os.environ.setdefault("NCCL_DEBUG", level)
    os.environ.setdefault("NCCL_DEBUG_SUBSYS", "INIT,COLL")


def device_memory_stats(device: Optional[torch.device] = None) -> dict:
    if not torch.cuda.is_available():
        return {}
    return {
        "allocated_gb": torch.cuda.memory_allocated(device) / 2**30,
        "reserved_gb": torch.cuda.memory_reserved(device) / 2**30,
        "peak_gb": torch.cuda.max_memory_allocated(device) / 2**30,
    }
