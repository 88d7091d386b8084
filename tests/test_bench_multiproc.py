"""bench.py under torchrun (gloo, tiny model): DP replicas and TP serving —
the exact launch mode the driver uses for the scaling run."""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port() -> int:
    """OS-assigned port: fixed ports collide with a rendezvous left over
    from a previous suite run shutting down."""
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _run_bench(extra, port=None, nproc=2, model="tiny", timeout=420):
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", str(nproc), "--master-addr", "127.0.0.1",
           "--master-port", str(port or _free_port()),
           os.path.join(REPO, "bench.py"),
           "--gpus", str(nproc), "--steps", "1", "--warmup", "0",
           "--concurrency", "1", "--model", model, *extra]
    proc = subprocess.run(cmd, capture_output=True, text=True, timeout=timeout,
                          cwd=REPO)
    assert proc.returncode == 0, proc.stderr[-2000:]
    lines = [l for l in proc.stdout.splitlines() if l.startswith("{")]
    assert lines, proc.stdout[-2000:]
    return json.loads(lines[-1])


@pytest.mark.slow
@pytest.mark.timeout(500)
def test_bench_dp2():
    out = _run_bench([])
    assert out["config"]["parallelism"] == "dp2"
    assert out["value"] > 0
    assert out["config"]["offline_gate_pass_rate"] == 1.0


@pytest.mark.slow
@pytest.mark.timeout(500)
def test_bench_tp2():
    out = _run_bench(["--tp", "2"])
    assert out["config"]["parallelism"] == "tp2"
    assert out["value"] > 0


@pytest.mark.slow
@pytest.mark.timeout(600)
def test_bench_dp8():
    """8-rank DP smoke: the exact process-group construction the driver's
    8-GPU weak-scaling run uses — first hardware run must not be a debug
    round (round-1 verdict item 9)."""
    out = _run_bench([], nproc=8, timeout=540)
    assert out["config"]["parallelism"] == "dp8"
    assert out["value"] > 0


@pytest.mark.slow
@pytest.mark.timeout(600)
def test_bench_tp8():
    """8-way tensor-parallel shards (kv-head replication: 2 kv heads over
    8 ranks) through the serving broadcast loop on gloo."""
    out = _run_bench(["--tp", "8"], nproc=8, model="tiny8", timeout=540)
    assert out["config"]["parallelism"] == "tp8"
    assert out["value"] > 0
