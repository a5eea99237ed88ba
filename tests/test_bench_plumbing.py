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


def test_bench_distributed_cpu_2proc():
    """The exact launch shape the driver uses (torch.distributed.run,
    nnodes=1, master 127.0.0.1), world_size 2 on CPU/gloo."""
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    res = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--device", "cpu", "--global-batch", "8", "--dim",
         "64", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=600, cwd=REPO)
    assert res.returncode == 0, res.stderr[-2000:]
    lines = [l for l in res.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, res.stdout
    out = json.loads(lines[0])
    assert out["n_gpus"] == 2


def test_bench_distributed_cpu_2proc_allgather_ddp():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    res = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), os.path.join(REPO, "bench.py"),
         "--device", "cpu", "--global-batch", "8", "--dim", "64",
         "--steps", "2", "--warmup", "1", "--strategy", "all_gather",
         "--ddp"],
        capture_output=True, text=True, timeout=600, cwd=REPO)
    assert res.returncode == 0, res.stderr[-2000:]


def test_example_cpu_smoke():
    """examples/train_siglip.py runs end to end on CPU (3 steps, tiny)."""
    res = subprocess.run(
        [sys.executable, os.path.join(REPO, "examples", "train_siglip.py"),
         "--device", "cpu", "--batch-per-gpu", "8", "--dim", "32",
         "--steps", "3", "--log-every", "2"],
        capture_output=True, text=True, timeout=300, cwd=REPO)
    assert res.returncode == 0, res.stderr[-2000:]
    assert "done:" in res.stdout


def test_bench_distributed_cpu_8proc():
    """The driver's full-node SCALE launch shape: torch.distributed.run with
    --nproc-per-node 8 on CPU/gloo — default strategy (bidir ring at W=8:
    three bidirectional rounds + one remainder hop) and default DDP towers."""
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    res = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
         "--master-port", str(port), os.path.join(REPO, "bench.py"),
         "--gpus", "8", "--device", "cpu", "--global-batch", "16", "--dim",
         "32", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=900, cwd=REPO)
    assert res.returncode == 0, res.stderr[-2000:]
    lines = [l for l in res.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, res.stdout
    out = json.loads(lines[0])
    assert out["n_gpus"] == 8
    assert out["config"]["parallelism"] == "dp8-ring"
    assert out["config"]["ddp_towers"] is True
