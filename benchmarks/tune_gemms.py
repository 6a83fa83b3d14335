#!/usr/bin/env python3
"""One-off GEMM tuning run (GPU): sweeps hipBLASLt algorithms for the EAGER
serving GEMM shapes, writing llm_instance_gateway_amd/data/
tunableop_gfx950.csv (committed; loaded by every engine at startup and
enabled ONLY around eager forwards — graph capture/replay keeps the default
algorithms, which avoids the r01-measured capture regression).

Eager GEMMs are the mixed prefill+decode steps, whose flat batch is bucketed
to multiples of 256 tokens (engine/config.py pad_batch_multiple), so the
shape set is closed: M in {256..4096 step 256} x the four projection GEMMs.
Decode-only steps run inside hipGraphs where TunableOp stays disabled, so
their shapes need no tuning.

    python benchmarks/tune_gemms.py [--model llama3-8b]
"""

import argparse
import os
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--max-m", type=int, default=4096)
    ap.add_argument("--step-m", type=int, default=256)
    args = ap.parse_args()

    os.environ["LIG_TUNE_GEMMS"] = "1"
    import torch

    from llm_instance_gateway_amd.models.config import get_model_config
    from llm_instance_gateway_amd.utils.tunableop import (
        RESULTS_FILE,
        maybe_enable_tunableop,
    )

    assert torch.cuda.is_available()
    assert maybe_enable_tunableop()

    mc = get_model_config(args.model)
    h = mc.hidden_size
    q = mc.num_heads * mc.head_dim
    kv = mc.num_kv_heads * mc.head_dim
    shapes = [
        (h, q + 2 * kv),            # qkv_proj
        (q, h),                     # o_proj
    ]
    if not mc.is_moe:
        shapes += [
            (h, 2 * mc.intermediate_size),  # gate_up
            (mc.intermediate_size, h),      # down
        ]
    dev = "cuda:0"
    torch.manual_seed(0)
    weights = {
        (k_in, n_out): torch.randn(n_out, k_in, dtype=torch.bfloat16, device=dev)
        * 0.01
        for (k_in, n_out) in shapes
    }
    ms = list(range(args.step_m, args.max_m + 1, args.step_m))
    total = len(ms) * len(shapes)
    done = 0
    for m in ms:
        for (k_in, n_out) in shapes:
            x = torch.randn(m, k_in, dtype=torch.bfloat16, device=dev)
            for _ in range(3):
                torch.nn.functional.linear(x, weights[(k_in, n_out)])
            torch.cuda.synchronize()
            done += 1
            print(f"[{done}/{total}] tuned M={m} K={k_in} N={n_out}", flush=True)
    print(f"tuning pass done; results flush to {RESULTS_FILE} at exit")


if __name__ == "__main__":
    main()
