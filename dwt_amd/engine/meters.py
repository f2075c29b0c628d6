"""Metrics / observability: meters + structured JSONL logging.

The reference logged with bare print(); we keep the identical
console strings for familiarity and add a structured JSONL sink
(loss/accuracy/imgs-sec per step) for tooling.
"""
from __future__ import annotations

import json
import os
import time
from typing import Optional


class AverageMeter:
    def __init__(self):
        self.reset()

    def reset(self):
        self.sum = 0.0
        self.count = 0

    def update(self, val, n=1):
        self.sum += float(val) * n
        self.count += n

    @property
    def avg(self):
        return self.sum / max(self.count, 1)


class ThroughputMeter:
    """images/sec over a sliding window of steps."""

    def __init__(self):
        self.t0 = None
        self.images = 0

    def tick(self, n_images: int):
        now = time.perf_counter()
        if self.t0 is None:
            self.t0 = now
            self.images = 0
            return None
        self.images += n_images
        return self.images / (now - self.t0)

    def reset(self):
        self.t0 = None
        self.images = 0


class JsonlLogger:
    def __init__(self, path: Optional[str] = None, rank: int = 0):
        self.path = path
        self.rank = rank
        self._fh = None
        if path and rank == 0:
            os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
            self._fh = open(path, "a", buffering=1)

    def log(self, **kv):
        if self._fh is not None:
            kv.setdefault("ts", time.time())
            self._fh.write(json.dumps(kv) + "\n")

    def close(self):
        if self._fh is not None:
            self._fh.close()
            self._fh = None
