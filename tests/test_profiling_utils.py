"""Unit tests for the profiling helpers (PhaseTimer CPU path, CSV export)."""

import os
import tempfile
import time

from distributed_sigmoid_loss_amd.utils.profiling import (
    PhaseTimer,
    roctx_range,
)


def test_phase_timer_cpu():
    t = PhaseTimer(enabled=True, use_cuda=False)
    for _ in range(3):
        with t.phase("a"):
            time.sleep(0.002)
        with t.phase("b"):
            time.sleep(0.001)
        t.step_end()
    assert len(t.rows) == 3
    s = t.summary()
    assert set(s.keys()) == {"a", "b"}
    assert s["a"] > s["b"] > 0

    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "steps.csv")
        t.write_csv(path)
        lines = open(path).read().strip().splitlines()
        assert lines[0] == "step,a,b"
        assert len(lines) == 4


def test_phase_timer_disabled():
    t = PhaseTimer(enabled=False)
    with t.phase("x"):
        pass
    t.step_end()
    assert t.rows == []
    assert t.summary() == {}


def test_roctx_range_noop_on_cpu():
    with roctx_range("anything"):
        pass  # must not raise without a GPU
