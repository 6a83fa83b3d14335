"""TunableOp policy glue (utils/tunableop.py) — CPU-side contract.

The GPU behavior (tuned algo pins, eager-only scoping) is exercised on
hardware; here we pin the policy decisions that must hold everywhere:
no-GPU means disabled, the in-tree results file exists and parses, and
eager_tuned() is a safe no-op without a GPU.
"""

import csv

import torch

from llm_instance_gateway_amd.utils import tunableop


def test_disabled_without_gpu():
    if torch.cuda.is_available():  # driver may run CPU suite on a GPU box
        return
    assert tunableop.maybe_enable_tunableop() is False


def test_eager_tuned_noop_without_gpu():
    with tunableop.eager_tuned():
        x = torch.ones(4, 4) @ torch.ones(4, 4)
    assert float(x[0, 0]) == 4.0


def test_results_file_ships_and_parses():
    """The gfx950 tuned-GEMM table must stay in-tree (serving engines load
    it read-only on a fresh box; losing it silently reverts the prefill
    GEMMs to heuristic picks — the r01 17%-of-peak defect)."""
    p = tunableop.RESULTS_FILE
    assert p.exists(), p
    with open(p, newline="") as f:
        rows = list(csv.reader(f))
    # PT_VERSION header + per-device validator lines + GemmTunableOp entries
    assert any("GemmTunableOp" in ",".join(r) for r in rows)
    tuned = [r for r in rows if r and r[0].startswith("GemmTunableOp")]
    assert len(tuned) >= 10  # the serving shapes, not a stub
