"""Per-category time profiling (ref: include/timer.h TimeProfile +
interface_quda.cpp per-API profiles): context-manager categories, printed
at end, GPU-synchronizing when requested."""

from __future__ import annotations

import time
from collections import defaultdict
from contextlib import contextmanager
from typing import Dict


class TimeProfile:
    CATEGORIES = ("download", "upload", "init", "preamble", "compute",
                  "comms", "epilogue", "free", "io", "tune", "total")

    def __init__(self, name: str, sync_gpu: bool = False):
        self.name = name
        self.sync = sync_gpu
        self.seconds: Dict[str, float] = defaultdict(float)
        self.counts: Dict[str, int] = defaultdict(int)

    def _now(self):
        if self.sync:
            import torch
            if torch.cuda.is_available():
                torch.cuda.synchronize()
        return time.perf_counter()

    @contextmanager
    def __call__(self, category: str):
        t0 = self._now()
        try:
            yield
        finally:
            self.seconds[category] += self._now() - t0
            self.counts[category] += 1

    def summary(self) -> str:
        lines = [f"TimeProfile[{self.name}]"]
        tot = sum(v for k, v in self.seconds.items() if k != "total")
        for k, v in sorted(self.seconds.items(), key=lambda kv: -kv[1]):
            lines.append(f"  {k:10s} {v:10.4f} s  x{self.counts[k]:<6d} "
                         f"{100*v/max(tot,1e-30):5.1f}%")
        return "\n".join(lines)


global_profile = TimeProfile("quda_amd")
