"""Bitrot protection for the benchmark harnesses: each runs end-to-end at a
tiny scale (the reference keeps its benchmark tool compiling in-tree the
same way)."""

import csv
import json
import subprocess
import sys
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent


def test_bench_router_small():
    proc = subprocess.run(
        [sys.executable, "benchmarks/bench_router.py",
         "--engines", "20", "--models", "10", "--requests", "200"],
        cwd=ROOT, capture_output=True, text=True, timeout=120,
    )
    assert proc.returncode == 0, proc.stderr[-1000:]
    out = json.loads(proc.stdout.strip().splitlines()[-1])
    assert out["requests"] == 200
    assert out["throughput_rps"] > 0


def test_sim_cli_csv(tmp_path):
    csv_path = tmp_path / "sweep.csv"
    proc = subprocess.run(
        [sys.executable, "-m", "llm_instance_gateway_amd.sim.main",
         "--routing-type", "least", "--rates", "5", "--duration", "5",
         "--num-servers", "2", "--csv", str(csv_path)],
        cwd=ROOT, capture_output=True, text=True, timeout=120,
    )
    assert proc.returncode == 0, proc.stderr[-1000:]
    rows = list(csv.DictReader(open(csv_path)))
    assert len(rows) == 1
    assert rows[0]["routing_type"] == "least"
    assert float(rows[0]["completed"]) > 0
