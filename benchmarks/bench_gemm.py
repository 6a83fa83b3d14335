#!/usr/bin/env python3
"""Skinny-GEMM kernel vs hipBLASLt at the decode projection shapes (GPU).

    python benchmarks/bench_gemm.py [--m 128]
"""

import argparse
import sys
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from llm_instance_gateway_amd import ops

SHAPES = [  # (K, N) of llama3-8b decode projections + lm_head
    (4096, 6144),    # qkv
    (4096, 4096),    # o
    (4096, 28672),   # gate_up
    (14336, 4096),   # down
    (4096, 128256),  # lm_head
]


def bench(M, K, N, iters=200):
    dev = "cuda:0"
    torch.manual_seed(0)
    x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.1
    w = torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.02

    out = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
    ops._EXT.skinny_gemm(out, x, w, torch.cuda.current_stream().cuda_stream)
    ref = (x.float() @ w.float().t())
    err = (out.float() - ref).abs().max().item()
    rel = err / max(1.0, ref.abs().max().item())
    assert rel < 0.02, f"numerics: abs={err} rel={rel}"

    def timeit(fn):
        for _ in range(20):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters * 1e6

    stream = torch.cuda.current_stream().cuda_stream
    us_custom = timeit(lambda: ops._EXT.skinny_gemm(out, x, w, stream))
    us_blas = timeit(lambda: torch.nn.functional.linear(x, w))
    w_mb = N * K * 2 / 1e6
    floor_us = w_mb / 8.0  # MB / (TB/s) = us
    print(
        f"M={M:4d} K={K:5d} N={N:6d}: custom {us_custom:7.1f} us  "
        f"blaslt {us_blas:7.1f} us  ({us_blas/us_custom:4.2f}x, "
        f"W-stream floor {floor_us:5.1f} us, custom at {floor_us/us_custom*100:3.0f}%)"
    )


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--m", type=int, default=128)
    args = ap.parse_args()
    assert torch.cuda.is_available() and ops.native_available()
    for K, N in SHAPES:
        bench(args.m, K, N)
    for m in (1, 16, 64, 256):
        bench(m, 4096, 6144)
    print("OK")


if __name__ == "__main__":
    main()
