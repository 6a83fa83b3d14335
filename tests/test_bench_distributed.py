"""bench.py distributed contract test: the exact launch shape the driver
uses (torch.distributed.run, one rank per GPU) on CPU/gloo with the tiny
model — validates rendezvous, per-rank engines, barrier placement, and the
cross-rank aggregation (MAX elapsed, SUM counts) producing one JSON line."""

import json
import os
import subprocess
import sys

import pytest


def _run_bench(nproc: int, port: int):
    env = dict(os.environ, MASTER_ADDR="127.0.0.1")
    proc = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", str(nproc),
            "--master-addr", "127.0.0.1", "--master-port", str(port),
            "bench.py", "--gpus", str(nproc), "--steps", "3", "--warmup", "1",
            "--concurrency", "4", "--model", "tiny", "--device", "cpu",
        ],
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        env=env,
        capture_output=True,
        text=True,
        timeout=220,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    line = [l for l in proc.stdout.splitlines() if l.startswith("{")][-1]
    return json.loads(line)


@pytest.mark.timeout(240)
def test_bench_two_rank_gloo(tmp_path):
    out = _run_bench(2, 29541)
    assert out["n_gpus"] == 2
    assert out["config"]["parallelism"] == "dp2"
    assert out["steps"] == 3
    assert out["ms_per_step"] > 0


@pytest.mark.timeout(240)
def test_bench_four_rank_gloo(tmp_path):
    """The driver's SCALE run launches N=4/8 the same way; exercise the
    rendezvous + MAX-elapsed/SUM-counts aggregation beyond 2 ranks."""
    out = _run_bench(4, 29543)
    assert out["n_gpus"] == 4
    assert out["config"]["parallelism"] == "dp4"
    assert out["config"]["global_batch"] == 16  # 4 ranks x conc 4
    assert out["ms_per_step"] > 0
