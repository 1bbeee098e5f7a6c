"""Step-time breakdown tracing — data / H2D / forward / loss / backward /
optimizer / comm / postprocess per step.

The reference has no per-phase timing at all (wall-clock
per epoch only); this gives the train loop a cheap breakdown using CUDA
events on GPU (no global synchronize per phase) and perf_counter on CPU.
Enabled with --trace-step-time; rank 0 logs a summary every log interval.
"""

import time
from collections import defaultdict
from contextlib import contextmanager
from typing import Dict

import torch


class StepTimer:
    """Phase timer. GPU phases are bracketed with CUDA events whose
    elapsed times are read lazily at ``summary()`` (one sync per epoch,
    not per step)."""

    PHASES = ("data", "h2d", "forward", "loss", "backward", "optimizer",
              "comm", "postprocess", "metrics")

    def __init__(self, enabled: bool, device: torch.device):
        self.enabled = enabled
        self.use_events = enabled and device.type == "cuda"
        self._events = defaultdict(list)   # phase -> [(start_ev, end_ev)]
        self._cpu = defaultdict(float)
        self._steps = 0

    @contextmanager
    def phase(self, name: str):
        if not self.enabled:
            yield
            return
        if self.use_events:
            s = torch.cuda.Event(enable_timing=True)
            e = torch.cuda.Event(enable_timing=True)
            s.record()
            try:
                yield
            finally:
                e.record()
                self._events[name].append((s, e))
        else:
            t0 = time.perf_counter()
            try:
                yield
            finally:
                self._cpu[name] += time.perf_counter() - t0

    def step(self):
        if self.enabled:
            self._steps += 1

    def summary(self) -> Dict[str, float]:
        """Mean ms per phase per step (drains pending events)."""
        if not self.enabled or self._steps == 0:
            return {}
        out = {}
        if self.use_events:
            torch.cuda.synchronize()
            for name, pairs in self._events.items():
                out[name] = sum(s.elapsed_time(e) for s, e in pairs) / self._steps
            self._events.clear()
        else:
            for name, total in self._cpu.items():
                out[name] = total * 1e3 / self._steps
            self._cpu.clear()
        self._steps = 0
        return out

    def format(self) -> str:
        parts = [f"{k}={v:.1f}ms" for k, v in self.summary().items()]
        return "step breakdown: " + " ".join(parts) if parts else ""
