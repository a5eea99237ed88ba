"""Profiling and metrics helpers (tracing/observability plan).

- :class:`PhaseTimer` — named-phase CUDA-event timing with per-step rows and
  CSV export; cheap enough to leave in the bench loop.
- :func:`roctx_range` — optional roctx markers so ``rocprofv3 --kernel-trace``
  and ``torch.profiler`` traces carry phase names (no-op when roctx is
  unavailable).
"""

from __future__ import annotations

import contextlib
import csv
import os
from typing import Dict, List, Optional

import torch

try:  # roctx ships with torch on ROCm
    from torch.cuda import nvtx as _nvtx  # maps to roctx on ROCm builds
except Exception:  # pragma: no cover
    _nvtx = None


@contextlib.contextmanager
def roctx_range(name: str):
    if _nvtx is not None and torch.cuda.is_available():
        _nvtx.range_push(name)
        try:
            yield
        finally:
            _nvtx.range_pop()
    else:
        yield


class HopStats:
    """Global per-hop timing collector for the ring strategy.

    Enabled via SIGLIP_HOP_STATS=1 (bench --csv turns it on): the ring
    forward wraps each exchange wait and each chunk kernel in CUDA events;
    :func:`summary` syncs once and returns mean ms per label — the per-hop
    comm/compute visibility SURVEY §5 asks for, without an external
    profiler.
    """

    enabled = os.environ.get("SIGLIP_HOP_STATS", "0") == "1"
    _events: List = []

    @classmethod
    def set_enabled(cls, on: bool):
        cls.enabled = on

    @classmethod
    @contextlib.contextmanager
    def record(cls, label: str):
        if not cls.enabled or not torch.cuda.is_available():
            yield
            return
        s = torch.cuda.Event(enable_timing=True)
        e = torch.cuda.Event(enable_timing=True)
        s.record()
        try:
            yield
        finally:
            e.record()
            cls._events.append((label, s, e))

    @classmethod
    def summary(cls) -> Dict[str, float]:
        if not cls._events:
            return {}
        torch.cuda.synchronize()
        agg: Dict[str, List[float]] = {}
        for n, s, e in cls._events:
            agg.setdefault(n, []).append(s.elapsed_time(e))
        cls._events = []
        return {k: sum(v) / len(v) for k, v in sorted(agg.items())}


class PhaseTimer:
    """Per-phase wall timing via CUDA events (or perf_counter on CPU).

    Usage::
        timer = PhaseTimer(enabled=True)
        with timer.phase("forward"): ...
        with timer.phase("backward"): ...
        timer.step_end()
        timer.write_csv("steps.csv")
    """

    def __init__(self, enabled: bool = True, use_cuda: Optional[bool] = None):
        self.enabled = enabled
        self.use_cuda = (torch.cuda.is_available() if use_cuda is None
                         else use_cuda)
        self._events: List = []        # (name, start_ev, end_ev) per phase
        self.rows: List[Dict[str, float]] = []

    @contextlib.contextmanager
    def phase(self, name: str):
        if not self.enabled:
            yield
            return
        with roctx_range(name):
            if self.use_cuda:
                s = torch.cuda.Event(enable_timing=True)
                e = torch.cuda.Event(enable_timing=True)
                s.record()
                try:
                    yield
                finally:
                    e.record()
                    self._events.append((name, s, e))
            else:
                import time
                t0 = time.perf_counter()
                try:
                    yield
                finally:
                    self._events.append(
                        (name, t0, time.perf_counter()))

    def step_end(self):
        if not self.enabled or not self._events:
            return
        if self.use_cuda:
            torch.cuda.synchronize()
            row = {n: s.elapsed_time(e) for n, s, e in self._events}
        else:
            row = {n: (e - s) * 1000.0 for n, s, e in self._events}
        self.rows.append(row)
        self._events = []

    def summary(self) -> Dict[str, float]:
        """Mean ms per phase over recorded steps."""
        if not self.rows:
            return {}
        keys = self.rows[0].keys()
        return {k: sum(r.get(k, 0.0) for r in self.rows) / len(self.rows)
                for k in keys}

    def write_csv(self, path: str):
        if not self.rows:
            return
        keys = sorted({k for r in self.rows for k in r})
        with open(path, "w", newline="") as f:
            w = csv.DictWriter(f, fieldnames=["step"] + keys)
            w.writeheader()
            for i, r in enumerate(self.rows):
                w.writerow({"step": i, **{k: f"{r.get(k, 0.0):.4f}"
                                          for k in keys}})
