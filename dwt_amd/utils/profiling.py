"""Profiling helpers: wall-clock step timing with proper device sync, and an
optional torch.profiler (kineto/roctracer) capture for per-kernel traces.
rocprofv3 recipes live in profiles/README.md."""
from __future__ import annotations

import contextlib
import time

import torch


def cuda_sync():
    if torch.cuda.is_available():
        torch.cuda.synchronize()


class StepTimer:
    """Times exactly K steps, device-synchronized on both sides."""

    def __init__(self):
        self.t0 = None
        self.elapsed = None
        self.steps = 0

    def start(self):
        cuda_sync()
        self.t0 = time.perf_counter()
        self.steps = 0

    def step(self):
        self.steps += 1

    def stop(self):
        cuda_sync()
        self.elapsed = time.perf_counter() - self.t0
        return self.elapsed

    @property
    def ms_per_step(self):
        return 1000.0 * self.elapsed / max(self.steps, 1)


@contextlib.contextmanager
def torch_profile(path: str, enabled: bool = True):
    if not enabled:
        yield None
        return
    from torch.profiler import ProfilerActivity, profile
    acts = [ProfilerActivity.CPU]
    if torch.cuda.is_available():
        acts.append(ProfilerActivity.CUDA)
    with profile(activities=acts) as prof:
        yield prof
    prof.export_chrome_trace(path)
