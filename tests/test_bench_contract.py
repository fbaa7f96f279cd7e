"""Benchmark-contract tests: the driver launches bench.py standalone and via
torch.distributed.run; verify both paths produce the contracted JSON line.
CPU/gloo here (the GPU path is exercised by the driver itself)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
ARGS = ["--steps", "1", "--warmup", "0", "--height", "80", "--width", "120",
        "--device", "cpu", "--dtype", "fp32"]


def _check_json(line: str, n: int):
    rec = json.loads(line)
    assert rec["metric"].startswith("imgs/sec train @320x960")
    assert rec["n_gpus"] == n
    assert rec["unit"] == "imgs/sec"
    assert rec["scaling"] == "weak"
    assert rec["higher_is_better"] is True
    assert rec["data"] == "synthetic"
    assert rec["value"] > 0
    assert rec["config"]["parallelism"] == f"dp{n}"
    return rec


def test_bench_single(tmp_path):
    r = subprocess.run([sys.executable, os.path.join(REPO, "bench.py")] + ARGS,
                       cwd=REPO, capture_output=True, text=True, timeout=900)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    _check_json(line, 1)


def _run_distributed(n: int):
    import socket
    with socket.socket() as sock:  # pick a free rendezvous port
        sock.bind(("127.0.0.1", 0))
        port = sock.getsockname()[1]
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", str(n), "--master-addr", "127.0.0.1",
         "--master-port", str(port), os.path.join(REPO, "bench.py")] + ARGS,
        cwd=REPO, env=env, capture_output=True, text=True, timeout=900)
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1  # rank 0 only
    _check_json(lines[-1], n)


def test_bench_distributed_two_ranks(tmp_path):
    _run_distributed(2)


def test_bench_distributed_four_ranks(tmp_path):
    """The driver's scaling launch shape (torch.distributed.run, one rank
    per GPU) at 4 ranks over gloo."""
    _run_distributed(4)
