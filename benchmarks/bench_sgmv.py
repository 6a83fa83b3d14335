#!/usr/bin/env python3
"""SGMV kernel micro-benchmark + numerics gate (GPU).

Times sgmv shrink/expand at serving shapes (decode T=128, prefill T=512;
32 resident adapters) and checks against the fp32 torch reference.

    python benchmarks/bench_sgmv.py
"""

import sys
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from llm_instance_gateway_amd import ops
from llm_instance_gateway_amd.ops import torch_ref


def bench_case(T, H, O, R, S=32, iters=200, wide=False):
    dev = "cuda:0"
    torch.manual_seed(0)
    x = torch.randn(T, H, device=dev, dtype=torch.bfloat16) * 0.1
    a = torch.randn(S, R, H, device=dev, dtype=torch.bfloat16) * 0.02
    b = torch.randn(S, O, R, device=dev, dtype=torch.bfloat16) * 0.02
    if wide:
        # prefill regime: sequence-contiguous adapter segments (~200 tokens)
        seg = []
        g = torch.Generator().manual_seed(1)
        while len(seg) < T:
            seg += [int(torch.randint(0, S, (1,), generator=g))] * 200
        slots = torch.tensor(seg[:T], device=dev, dtype=torch.int32)
    else:
        slots = torch.randint(0, S, (T,), device=dev, dtype=torch.int32)
    y = torch.randn(T, O, device=dev, dtype=torch.bfloat16) * 0.1
    u_ws = torch.zeros(T, R, device=dev, dtype=torch.bfloat16)

    # numerics vs fp32 reference
    y_test = y.clone()
    ops._EXT.sgmv_apply(y_test, x, a, b, u_ws, slots, 1.0, wide, torch.cuda.current_stream().cuda_stream)
    torch.cuda.synchronize()
    u_ref = torch_ref.sgmv_shrink(x.float(), a.float(), slots)
    y_ref = y.clone().float()
    torch_ref.sgmv_expand(u_ref, b.float(), slots, y_ref, 1.0)
    err = (y_test.float() - y_ref).abs().max().item()
    scale = y_ref.abs().max().item()
    assert err <= 0.02 * max(1.0, scale), f"numerics off: {err} vs {scale}"

    stream = torch.cuda.current_stream().cuda_stream
    if wide:
        # MFMA tiled path (what the engine uses for prefill)
        sl = slots.tolist()
        t0s, lens, tslots = [], [], []
        i = 0
        while i < T:
            j = i
            while j < T and sl[j] == sl[i]:
                j += 1
            if sl[i] >= 0:
                for off in range(i, j, 64):
                    t0s.append(off)
                    lens.append(min(64, j - off))
                    tslots.append(sl[i])
            i = j
        tt0 = torch.tensor(t0s, dtype=torch.int32, device=dev)
        tl = torch.tensor(lens, dtype=torch.int32, device=dev)
        ts = torch.tensor(tslots, dtype=torch.int32, device=dev)

        def run():
            ops._EXT.sgmv_apply_tiled(y, x, a, b, u_ws, tt0, tl, ts, 1.0, stream)
    else:
        def run():
            ops._EXT.sgmv_apply(y, x, a, b, u_ws, slots, 1.0, wide, stream)
    for _ in range(20):
        run()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        run()
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / iters * 1e6
    gb = (T * H * 2 + T * R * 4 + S * R * H * 2 / 8 + T * O * 4 + T * R * O / O * 2) / 1e9
    print(
        f"T={T:5d} H={H} O={O:5d} R={R:2d}: {us:7.1f} us/apply "
        f"(shrink+expand pair)"
    )
    return us


def main():
    assert torch.cuda.is_available()
    assert ops.native_available()
    print("== decode shapes (qkv / o / gate_up / down of llama3-8b) ==")
    for (H, O) in ((4096, 6144), (4096, 4096), (4096, 28672), (14336, 4096)):
        bench_case(128, H, O, 16)
    print("== prefill shapes (segmented slots, wide kernels) ==")
    bench_case(512, 4096, 6144, 16, wide=True)
    bench_case(2048, 4096, 6144, 16, wide=True)
    bench_case(2048, 4096, 28672, 16, wide=True)
    print("== rank sweep (qkv) ==")
    for r in (8, 32, 64):
        bench_case(128, 4096, 6144, r)
    print("OK")


if __name__ == "__main__":
    main()
