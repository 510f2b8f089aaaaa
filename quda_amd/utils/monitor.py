"""Background power/temperature monitor (ref: lib/monitor.cpp — samples
into a tsv with trapezoidal energy integration; rocm-smi/amdsmi instead of
NVML)."""

from __future__ import annotations

import subprocess
import threading
import time
from typing import List, Optional


def _sample_rocm_smi():
    """Returns (power_W, temp_C) of GPU 0 or None."""
    try:
        out = subprocess.run(
            ["rocm-smi", "--showpower", "--showtemp", "--csv"],
            capture_output=True, text=True, timeout=5).stdout
        pw = tmp = None
        for line in out.splitlines():
            if line.startswith("card"):
                parts = line.split(",")
                for p in parts[1:]:
                    try:
                        v = float(p)
                    except ValueError:
                        continue
                    if pw is None:
                        pw = v
                    elif tmp is None:
                        tmp = v
                break
        return pw, tmp
    except Exception:
        return None


class PowerMonitor:
    """Sample power/temp every `period` s into rows; .energy_J integrates
    trapezoidally (env QUDA_ENABLE_MONITOR analogue)."""

    def __init__(self, period: float = 0.5, path: Optional[str] = None):
        self.period = period
        self.path = path
        self.rows: List[tuple] = []
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def start(self):
        self._stop.clear()
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def _loop(self):
        while not self._stop.is_set():
            s = _sample_rocm_smi()
            if s and s[0] is not None:
                self.rows.append((time.time(), s[0], s[1]))
            self._stop.wait(self.period)

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)
        if self.path and self.rows:
            with open(self.path, "w") as f:
                f.write("time\tpower_w\ttemp_c\n")
                for r in self.rows:
                    f.write(f"{r[0]:.3f}\t{r[1]}\t{r[2]}\n")

    @property
    def energy_J(self) -> float:
        e = 0.0
        for i in range(1, len(self.rows)):
            dt = self.rows[i][0] - self.rows[i - 1][0]
            e += 0.5 * (self.rows[i][1] + self.rows[i - 1][1]) * dt
        return e
