"""Hierarchical timer registry with verbosity levels.

Capability parity: the reference's global timer registry (src/timer.h:22-77
enum + inline start/stop, timer.c:59-91 report) — rebuilt as a registry of
named timers with GPU-aware stops (torch.cuda.synchronize around device
work when timing) and a sectioned report, plus optional HIP-event pairs for
per-kernel device timing.
"""
from __future__ import annotations

import time
from contextlib import contextmanager
from dataclasses import dataclass, field
from typing import Dict, Optional

import torch

VERB_QUIET, VERB_LOW, VERB_HIGH, VERB_MAX = 0, 1, 2, 3


@dataclass
class Timer:
    name: str
    level: int = VERB_LOW
    seconds: float = 0.0
    count: int = 0
    _start: Optional[float] = None

    def start(self):
        self._start = time.perf_counter()

    def stop(self):
        if self._start is not None:
            self.seconds += time.perf_counter() - self._start
            self.count += 1
            self._start = None


@dataclass
class TimerRegistry:
    verbosity: int = VERB_LOW
    sync_device: bool = False
    timers: Dict[str, Timer] = field(default_factory=dict)

    def get(self, name: str, level: int = VERB_LOW) -> Timer:
        if name not in self.timers:
            self.timers[name] = Timer(name, level)
        return self.timers[name]

    @contextmanager
    def time(self, name: str, level: int = VERB_LOW):
        t = self.get(name, level)
        if self.sync_device and torch.cuda.is_available():
            torch.cuda.synchronize()
        t.start()
        try:
            yield t
        finally:
            if self.sync_device and torch.cuda.is_available():
                torch.cuda.synchronize()
            t.stop()

    def report(self) -> str:
        lines = ["", "Timing information ---------------------------------"]
        width = max((len(n) for n in self.timers), default=4)
        for name, t in sorted(self.timers.items(), key=lambda kv: -kv[1].seconds):
            if t.level <= self.verbosity and t.count:
                lines.append(f"  {name:<{width}s}  {t.seconds:10.3f}s  (x{t.count})")
        return "\n".join(lines)


TIMERS = TimerRegistry()


class CudaEventTimer:
    """Device-side interval timing via HIP events (no host sync until read)."""

    def __init__(self):
        self._pairs = []

    @contextmanager
    def time(self):
        s = torch.cuda.Event(enable_timing=True)
        e = torch.cuda.Event(enable_timing=True)
        s.record()
        try:
            yield
        finally:
            e.record()
            self._pairs.append((s, e))

    def elapsed_ms(self) -> float:
        torch.cuda.synchronize()
        return sum(s.elapsed_time(e) for s, e in self._pairs)
