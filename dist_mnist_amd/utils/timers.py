"""Lightweight step timing (SURVEY.md §5 tracing: hipEvent-based in-process
timers; rocprofv3 is the heavyweight proof artifact)."""

from __future__ import annotations

import json
import time

import torch


class StepTimer:
    """Wall-clock step timer with correct device synchronization at the
    measurement boundaries only (never inside the timed region)."""

    def __init__(self, device=None):
        self.device = device
        self.samples: list[float] = []
        self._t0 = None

    def _sync(self):
        if self.device is not None and torch.device(self.device).type == "cuda":
            torch.cuda.synchronize(self.device)

    def start(self):
        self._sync()
        self._t0 = time.perf_counter()

    def stop(self, n_steps: int = 1):
        self._sync()
        dt = time.perf_counter() - self._t0
        self.samples.append(dt / n_steps)
        return dt

    @property
    def ms_per_step(self):
        return 1000.0 * (sum(self.samples) / max(1, len(self.samples)))

    def dump(self, path: str, **extra):
        with open(path, "w") as f:
            json.dump({"ms_per_step": self.ms_per_step, "samples_s": self.samples, **extra}, f)
