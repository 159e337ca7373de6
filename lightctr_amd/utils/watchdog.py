"""Heartbeat / GPU-health watchdog.

Parity with the reference Master's heartbeat monitor
(/root/reference/LightCTR/distribut/master.h:202-262: periodic pings,
soft-timeout warning, dead declaration) translated to the MI355X setting:
intra-node transport (xGMI) is reliable, so the watchdog monitors GPU and
rank HEALTH instead of message liveness — each rank posts a heartbeat
timestamp; a monitor thread flags ranks whose heartbeat goes stale (soft
at `soft_s`, dead at `dead_s`) and can query rocm-smi for device state.
"""

from __future__ import annotations

import subprocess
import threading
import time


class Watchdog:
    def __init__(self, n_ranks: int, soft_s: float = 10.0,
                 dead_s: float = 20.0, period_s: float = 5.0,
                 on_soft=None, on_dead=None):
        self.n_ranks = n_ranks
        self.soft_s, self.dead_s, self.period_s = soft_s, dead_s, period_s
        self.beats = [time.monotonic()] * n_ranks
        self.state = ["alive"] * n_ranks
        self.on_soft = on_soft or (lambda r: None)
        self.on_dead = on_dead or (lambda r: None)
        self._stop = threading.Event()
        self._thread = None
        self._lock = threading.Lock()

    def heartbeat(self, rank: int) -> None:
        with self._lock:
            self.beats[rank] = time.monotonic()
            if self.state[rank] != "dead":
                self.state[rank] = "alive"

    def _monitor(self):
        while not self._stop.wait(self.period_s):
            now = time.monotonic()
            with self._lock:
                for r in range(self.n_ranks):
                    age = now - self.beats[r]
                    if age > self.dead_s and self.state[r] != "dead":
                        self.state[r] = "dead"
                        self.on_dead(r)
                    elif age > self.soft_s and self.state[r] == "alive":
                        self.state[r] = "soft"
                        self.on_soft(r)

    def start(self):
        self._thread = threading.Thread(target=self._monitor, daemon=True)
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join()

    def snapshot(self) -> list[str]:
        with self._lock:
            return list(self.state)


def gpu_health() -> dict:
    """Host-side device health via rocm-smi (temperature/power/usage);
    returns {} when no GPU or rocm-smi unavailable."""
    try:
        out = subprocess.run(
            ["rocm-smi", "--showtemp", "--showpower", "--showuse",
             "--json"], capture_output=True, text=True, timeout=10)
        if out.returncode != 0:
            return {}
        import json

        return json.loads(out.stdout or "{}")
    except (OSError, subprocess.TimeoutExpired, ValueError):
        return {}
