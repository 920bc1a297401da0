"""Wall timers with optional device synchronization (ref mytimer utils.cpp:10
and the per-step MPI_Wtime spans throughout main.cpp/louvain.cpp)."""

from __future__ import annotations

import time
from collections import defaultdict

import torch


class Timer:
    def __init__(self, sync: bool = False):
        self.sync = sync
        self.t0 = 0.0

    def __enter__(self):
        if self.sync and torch.cuda.is_available():
            torch.cuda.synchronize()
        self.t0 = time.perf_counter()
        return self

    def __exit__(self, *exc):
        if self.sync and torch.cuda.is_available():
            torch.cuda.synchronize()
        self.elapsed = time.perf_counter() - self.t0
        return False


class Timers:
    """Named accumulating timers: `with timers('halo'): ...`."""

    def __init__(self, sync: bool = False):
        self.sync = sync
        self.acc = defaultdict(float)
        self.count = defaultdict(int)

    class _Span:
        def __init__(self, owner, name):
            self.owner, self.name = owner, name

        def __enter__(self):
            if self.owner.sync and torch.cuda.is_available():
                torch.cuda.synchronize()
            self.t0 = time.perf_counter()

        def __exit__(self, *exc):
            if self.owner.sync and torch.cuda.is_available():
                torch.cuda.synchronize()
            self.owner.acc[self.name] += time.perf_counter() - self.t0
            self.owner.count[self.name] += 1
            return False

    def __call__(self, name: str):
        return Timers._Span(self, name)

    def summary(self) -> str:
        return " ".join(f"{k}={v*1e3:.1f}ms/{self.count[k]}"
                        for k, v in sorted(self.acc.items()))
