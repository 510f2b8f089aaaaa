"""Launch-config autotuner with a persistent cache
(ref: lib/tune.cpp tuneLaunch + tunecache.tsv: measure candidates, keep
the winner per (kernel, volume, precision) key, persist under
$QUDA_AMD_RESOURCE_PATH, broadcast rank-0 winners so launch configs agree
grid-wide)."""

from __future__ import annotations

import os
import time
from typing import Callable, Dict, List, Optional, Tuple


class Tuner:
    def __init__(self, path: Optional[str] = None):
        rp = os.environ.get("QUDA_AMD_RESOURCE_PATH")
        self.path = path or (os.path.join(rp, "tunecache.tsv") if rp else None)
        self.cache: Dict[str, Tuple[str, float]] = {}
        self.counts: Dict[str, int] = {}  # per-key consults (profile role)
        if self.path and os.path.exists(self.path):
            self._load()

    def _load(self):
        with open(self.path) as f:
            for line in f:
                if line.startswith("#") or not line.strip():
                    continue
                key, cfg, t = line.rstrip("\n").split("\t")
                self.cache[key] = (cfg, float(t))

    def _save(self):
        if not self.path:
            return
        os.makedirs(os.path.dirname(self.path), exist_ok=True)
        with open(self.path, "w") as f:
            f.write("# quda_amd tunecache: key\tconfig\ttime_us\n")
            for k, (cfg, t) in sorted(self.cache.items()):
                f.write(f"{k}\t{cfg}\t{t}\n")

    def _broadcast(self):
        """Rank-0 winners everywhere (ref broadcastTuneCache tune.cpp:327)."""
        from ..parallel import comms
        if not comms.is_distributed():
            return
        import torch.distributed as dist
        obj = [self.cache]
        dist.broadcast_object_list(obj, src=0)
        self.cache = obj[0]

    def tune(self, key: str, candidates: List[str],
             setup: Callable[[str], None], run: Callable[[], None],
             warmup: int = 2, iters: int = 5) -> str:
        """Measure `run` under each candidate config (applied by `setup`),
        cache + persist the winner, and leave it applied."""
        self.counts[key] = self.counts.get(key, 0) + 1
        if key in self.cache:
            cfg = self.cache[key][0]
            setup(cfg)
            return cfg
        import torch
        best, best_t = None, float("inf")
        for cfg in candidates:
            setup(cfg)
            for _ in range(warmup):
                run()
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(iters):
                run()
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / iters * 1e6
            if dt < best_t:
                best, best_t = cfg, dt
        self.cache[key] = (best, best_t)
        # rank-0's winner becomes THE winner everywhere (launch configs
        # must agree grid-wide, ref broadcastTuneCache): re-read after
        # the broadcast before applying
        self._broadcast()
        best = self.cache[key][0]
        self._save()
        setup(best)
        return best

    def profile_dump(self, path: str) -> None:
        """Per-key consult-count x tuned-time table (role of the
        reference's profile_N.tsv, tune.cpp:566)."""
        with open(path, "w") as f:
            f.write("# key\tcount\ttuned_us\ttotal_us\n")
            for k in sorted(self.counts):
                cfg_t = self.cache.get(k, ("?", 0.0))
                n = self.counts[k]
                f.write(f"{k}\t{n}\t{cfg_t[1]:.3f}\t{n * cfg_t[1]:.3f}\n")


_TUNER: Optional[Tuner] = None


def get_tuner() -> Tuner:
    global _TUNER
    if _TUNER is None:
        _TUNER = Tuner()
    return _TUNER


def tune_dslash(example_call: Callable[[], None], key: str) -> str:
    """Tune the dslash workgroup size for a representative launch
    (winner applied via the set_dslash_block binding)."""
    from ..ops.dispatch import hip_ext
    ext = hip_ext()

    def setup(cfg):
        ext.set_dslash_block(int(cfg))

    return get_tuner().tune(key, ["64", "128", "256"], setup, example_call)


def tune_dslash_policy(example_call, key: str) -> str:
    """Autotune the comm-overlap policy (overlap vs fused) for a
    representative distributed dslash (ref: the policy-level autotuning
    of lib/dslash_policy.hpp, profile_async tsv)."""
    from ..ops.dispatch import set_dslash_policy

    return get_tuner().tune(key, ["overlap", "fused"],
                            lambda c: set_dslash_policy(c), example_call)
