#!/usr/bin/env python3
"""One-off GEMM tuning run (GPU): sweeps hipBLASLt algorithms for every GEMM
shape the serving step hits, writing llm_instance_gateway_amd/data/
tunableop_gfx950.csv (committed; loaded read-only by every engine at startup).

    python benchmarks/tune_gemms.py [--model llama3-8b] [--steps 6]
"""

import argparse
import os
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--concurrency", type=int, default=128)
    args = ap.parse_args()

    os.environ["LIG_TUNE_GEMMS"] = "1"
    import torch

    from llm_instance_gateway_amd.utils.tunableop import RESULTS_FILE, maybe_enable_tunableop

    assert torch.cuda.is_available()
    assert maybe_enable_tunableop()

    from llm_instance_gateway_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from llm_instance_gateway_amd.models.config import get_model_config

    cfg = EngineConfig(
        model=get_model_config(args.model),
        num_kv_blocks=8192,
        max_num_seqs=256,
        max_num_batched_tokens=4096,
        max_model_len=2048,
        max_loras=0,
        enforce_eager=True,  # capture would freeze pre-tuning algorithms
    )
    torch.manual_seed(0)
    eng = LLMEngine(cfg, device="cuda:0")
    g = torch.Generator().manual_seed(1)
    # short prompts: a couple of prefill steps, then a long run of M=concurrency
    # decode steps — the decode GEMM shapes (M = graph bucket = concurrency)
    # are the stable, highest-leverage shapes to tune
    for i in range(args.concurrency):
        plen = int(torch.randint(8, 24, (1,), generator=g))
        eng.add_request(
            torch.randint(1, 128000, (plen,), generator=g).tolist(),
            SamplingParams(max_new_tokens=args.steps + 8),
        )
    for _ in range(args.steps):
        if not eng.has_work():
            break
        eng.step()
    torch.cuda.synchronize()
    print(f"tuning pass done; results flush to {RESULTS_FILE} at exit")


if __name__ == "__main__":
    main()
