"""hipBLASLt GEMM algorithm selection via PyTorch TunableOp.

The serving GEMMs (skinny decode M=128 projections, big prefill GEMMs, the
128k-vocab logits GEMM) go through torch.matmul -> hipBLASLt heuristics,
which left the prefill GEMMs at ~430 TF (~17% of bf16 MFMA peak) in
profiles/r01_steady_state_breakdown.md.  TunableOp sweeps the hipBLASLt
algorithm space per GEMM shape and pins the winner.

Policy:
  * a tuned-results file for gfx950 ships in-tree (``data/tunableop_gfx950.csv``,
    produced by ``python benchmarks/tune_gemms.py`` on an MI355X); when it
    exists, engines enable TunableOp in read-only mode and load it — zero
    runtime tuning cost;
  * ``LIG_TUNE_GEMMS=1`` runs in tuning mode and writes/extends the file
    (only do this on a GPU box, it adds minutes to the first steps).
"""

from __future__ import annotations

import logging
import os
from pathlib import Path

import torch

log = logging.getLogger(__name__)

RESULTS_FILE = Path(__file__).resolve().parent.parent / "data" / "tunableop_gfx950.csv"


# set by maybe_enable_tunableop: tuned results loaded, eager-scoped enabling
# available through eager_tuned()
_EAGER_READY = False


def maybe_enable_tunableop() -> bool:
    """Call once per process before the first GEMM.  Returns True if enabled.

    Serving mode loads the tuned results but leaves TunableOp DISABLED
    globally; the model runner flips it on only around eager forwards
    (``eager_tuned()``).  MEASURED on MI355X (r01, gpurun_out/bench4.log):
    TunableOp active during hipGraph capture made the bench 3x slower
    (35.8 vs 12.4 ms/step) — the tunable GEMM wrapper defeats capture, so
    graphs keep the default algorithms and only eager (mixed prefill)
    GEMMs use the tuned picks.  LIG_TUNABLEOP=0 disables entirely."""
    global _EAGER_READY
    if not torch.cuda.is_available():
        return False
    tuning = os.environ.get("LIG_TUNE_GEMMS", "0") == "1"
    if os.environ.get("LIG_TUNABLEOP", "1") != "1" and not tuning:
        return False
    if not tuning and not RESULTS_FILE.exists():
        return False
    import torch.cuda.tunable as tunable

    if tuning:
        out = os.environ.get("LIG_TUNE_GEMMS_OUT", str(RESULTS_FILE))
        Path(out).parent.mkdir(parents=True, exist_ok=True)
        tunable.set_filename(out)
        tunable.enable(True)
        tunable.tuning_enable(True)
        log.info("TunableOp TUNING mode; results -> %s", out)
        return True
    try:
        tunable.enable(True)
        tunable.tuning_enable(False)
        tunable.read_file(str(RESULTS_FILE))
        tunable.enable(False)  # eager_tuned() scopes the enabling
        _EAGER_READY = True
        log.info("TunableOp: tuned GEMM algorithms loaded (eager-scoped)")
        return True
    except Exception as e:  # pragma: no cover
        log.warning("TunableOp results unreadable (%s); disabled", e)
        tunable.enable(False)
        return False


class eager_tuned:
    """Context manager: tuned GEMM algorithms for an eager region (no-op when
    no results are loaded or during tuning runs)."""

    def __enter__(self):
        if _EAGER_READY:
            import torch.cuda.tunable as tunable

            tunable.enable(True)
        return self

    def __exit__(self, *exc):
        if _EAGER_READY:
            import torch.cuda.tunable as tunable

            tunable.enable(False)
        return False


# Note: in tuning mode the C++ side persists results to the filename at
# normal process exit; scripts must not be SIGKILLed before that.
