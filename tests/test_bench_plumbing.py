"""Config-0 plumbing test: bench.py runs on CPU (batch=8, dim=64) and prints
one valid JSON line with the contract keys."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_cpu_plumbing():
    res = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--device", "cpu",
         "--global-batch", "8", "--dim", "64", "--steps", "2", "--warmup",
         "1"],
        capture_output=True, text=True, timeout=300, cwd=REPO)
    assert res.returncode == 0, res.stderr
    lines = [l for l in res.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, res.stdout
    out = json.loads(lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in out, key
    assert out["n_gpus"] == 1
    assert out["value"] > 0
    assert out["data"] == "synthetic"
