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


class TrainGuard:
    """Failure-detection wiring for a training loop (reference
    master.h:202-262 heartbeat -> declare-dead semantics, modernized to
    checkpoint-restart per SURVEY §5.3).

    The training thread calls `step()` once per batch; a Watchdog monitor
    thread declares the loop dead when no step lands for `dead_s`
    seconds (GPU hang, dead peer rank blocking a collective, ...). On
    death the guard saves a checkpoint of the model's last COMPLETED
    step (train_step mutates state only after its collectives, so the
    snapshot is step-consistent per rank) and exits the process with
    status 3 so a supervisor can restart; restart resumes from the
    checkpoint via `maybe_resume()`.
    """

    EXIT_CODE = 3

    def __init__(self, model, ckpt_path: str | None, soft_s: float = 10.0,
                 dead_s: float = 20.0, period_s: float = 1.0,
                 ckpt_every: int = 0, log=print):
        self.model = model
        self.ckpt_path = ckpt_path
        self.ckpt_every = ckpt_every
        self.log = log
        self.steps = 0
        self._wd = Watchdog(1, soft_s=soft_s, dead_s=dead_s,
                            period_s=period_s,
                            on_soft=self._on_soft, on_dead=self._on_dead)

    def _on_soft(self, _rank):
        self.log(f"[watchdog] training loop stalled >"
                 f"{self._wd.soft_s:.0f}s at step {self.steps}")

    def _on_dead(self, _rank):
        self._die(f"training loop DEAD (> {self._wd.dead_s:.0f}s without "
                  f"a step)")

    def _die(self, why: str):
        import os

        self.log(f"[watchdog] {why} at step {self.steps}; checkpointing "
                 f"and aborting for restart-resume")
        if self.ckpt_path is not None:
            try:
                self.model.save(self.ckpt_path)
                self.log(f"[watchdog] checkpoint -> {self.ckpt_path}")
            except Exception as e:  # noqa: BLE001
                self.log(f"[watchdog] checkpoint failed: {e}")
        os._exit(self.EXIT_CODE)

    def run_step(self, fn, *a, **kw):
        """Run one training step under the guard: a raised communication
        / device error (gloo raises on a dead peer; RCCL may raise or
        hang — the hang case is the watchdog's) checkpoints the last
        completed step and exits for restart."""
        try:
            out = fn(*a, **kw)
        except Exception as e:  # noqa: BLE001 — any step fault is fatal
            self._die(f"step failed ({type(e).__name__}: {e})")
        self.step()
        return out

    def maybe_resume(self) -> bool:
        # sharded models save under "{prefix}.shardRofW.pt", single models
        # under the literal path — let load() resolve and treat a missing
        # file as "fresh start"
        if not self.ckpt_path:
            return False
        try:
            self.model.load(self.ckpt_path)
        except FileNotFoundError:
            return False
        self.log(f"[watchdog] resumed from {self.ckpt_path}")
        return True

    def start(self):
        self._wd.start()
        return self

    def step(self):
        self.steps += 1
        self._wd.heartbeat(0)
        if (self.ckpt_path and self.ckpt_every
                and self.steps % self.ckpt_every == 0):
            self.model.save(self.ckpt_path)

    def stop(self):
        self._wd.stop()


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
