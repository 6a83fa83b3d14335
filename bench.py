#!/usr/bin/env python3
"""Flagship serving benchmark — BASELINE.json's headline metric:

    goodput (req/s within TTFT SLO), Llama-3-8B (+32 LoRAs once the SGMV
    path is active) on N MI355X GPUs of one node.

One "step" = one continuous-batching engine iteration (mixed chunked-prefill
+ decode batch).  The engine is kept saturated by a closed-loop synthetic
workload shaped like the reference simulation's
(simulations/llm_ig_simulation/src/main.py: input 202±20 tokens, output
179±17); goodput counts requests that finished with TTFT <= SLO during the
timed window.

Multi-GPU (the driver launches one rank per GPU via torch.distributed.run):
data-parallel engines, weak scaling — each rank runs its own engine+workload;
the reported value is the whole-job aggregate (sum of per-rank goodput over
the max-across-ranks elapsed time).

Usage:  python bench.py [--gpus N] [--steps K] [--warmup W]
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from llm_instance_gateway_amd.engine import EngineConfig, LLMEngine, SamplingParams
from llm_instance_gateway_amd.models.config import get_model_config

TTFT_SLO_S = 2.0
# Tiered TPOT SLOs: the reference sim's two priority tiers (0.025 s/token
# lo-latency, 0.5 s/token hi-latency — simulations/llm_ig_simulation/src/
# main.py:17-37).  The headline `value` stays TTFT-goodput (BASELINE.json's
# metric); both tiered counts are reported alongside so "goodput" is
# distinguishable from raw throughput.
TPOT_SLO_LO_S = 0.025
TPOT_SLO_HI_S = 0.5
PROMPT_MEAN, PROMPT_STD = 202, 20
OUTPUT_MEAN, OUTPUT_STD = 179, 17
NUM_LORAS = 32  # workload tags requests with 32 adapters (BASELINE config #3)


def make_workload_gen(seed: int, vocab: int):
    g = torch.Generator().manual_seed(seed)
    count = [0]

    def next_request(prime: bool = False):
        plen = max(8, int(torch.normal(float(PROMPT_MEAN), float(PROMPT_STD), (1,), generator=g)))
        if prime:
            # Steady-state priming: the initial closed-loop fill draws its
            # REMAINING output length from the equilibrium residual
            # distribution (uniform over [4, 2*mean]), so completions flow
            # from the first steps instead of after ~mean decode steps.
            olen = int(torch.randint(4, 2 * OUTPUT_MEAN, (1,), generator=g))
        else:
            olen = max(4, int(torch.normal(float(OUTPUT_MEAN), float(OUTPUT_STD), (1,), generator=g)))
        prompt = torch.randint(1, vocab - 1, (plen,), generator=g).tolist()
        lora = f"lora-{int(torch.randint(0, NUM_LORAS, (1,), generator=g))}"
        count[0] += 1
        return prompt, olen, lora
    return next_request


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=128)
    ap.add_argument("--warmup", type=int, default=32)
    ap.add_argument("--model", default="llama3-8b")
    # conc sweep on MI355X (gpurun_out/sw_c*.log): 256→54.2, 384→67.0,
    # 512→67.0 req/s with worsening TPOT; 384 saturates the chip
    # (~25.4k tok/s) with TTFT p95 ~44 ms and is the default operating point.
    ap.add_argument("--concurrency", type=int, default=384)
    ap.add_argument("--device", default=None, help="override (e.g. cpu for dev)")
    ap.add_argument(
        "--quant",
        default="",
        choices=["", "fp8"],
        help="opt-in fp8 (e4m3) projection GEMMs; the headline metric stays "
        "bf16 (BASELINE's dtype) — fp8 runs are reported with dtype=fp8",
    )
    ap.add_argument("--max-batched-tokens", type=int, default=0,
                    help="prefill chunk budget override (0 = default)")
    ap.add_argument(
        "--lora",
        action=argparse.BooleanOptionalAction,
        default=True,
        help="LoRA adapters on the SGMV path (default ON: BASELINE's metric "
        "is Llama-3-8B + 32 LoRAs; --no-lora for the base-model A/B)",
    )
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    distributed = world > 1
    if distributed:
        torch.distributed.init_process_group(
            backend="nccl" if torch.cuda.is_available() else "gloo"
        )

    if args.device:
        device = args.device
    elif torch.cuda.is_available():
        device = f"cuda:{local_rank}"
        torch.cuda.set_device(device)
    else:
        device = "cpu"

    if device.startswith("cuda"):
        from llm_instance_gateway_amd.utils.tunableop import maybe_enable_tunableop

        maybe_enable_tunableop()

    mc = get_model_config(args.model)
    is_cuda = device.startswith("cuda")
    # KV pool sized for the closed-loop population: every in-flight request
    # can reach prompt+output ≈ 430 tokens (mean+3σ), so conc*32 blocks of 16
    # leaves ~20% headroom (33 GB at conc=512 — trivial against 288 GB HBM3E).
    cfg = EngineConfig(
        model=mc,
        num_kv_blocks=max(8192, args.concurrency * 32) if is_cuda else 256,
        max_num_seqs=args.concurrency,
        max_num_batched_tokens=(args.max_batched_tokens or (4096 if is_cuda else 512)),
        max_model_len=2048 if is_cuda else 512,
        gpu_memory_utilization=0.90,
        max_loras=NUM_LORAS,
        engine_id=f"engine-{rank}",
        gpu_id=local_rank,
        quant=args.quant,
    )
    torch.manual_seed(1234)  # identical weights across ranks
    eng = LLMEngine(cfg, device=device)

    if args.lora:
        from llm_instance_gateway_amd.lora.manager import AdapterManager

        mgr = AdapterManager(eng, max_slots=NUM_LORAS, rank=16)
        for i in range(NUM_LORAS):
            mgr.register_random(f"lora-{i}", seed=i)
        # pre-load all adapters (generation + H2D copies are setup, not
        # serving; measuring them as TTFT would be a benchmark artifact)
        mgr.prepare([f"lora-{i}" for i in range(NUM_LORAS)])
        if is_cuda:
            torch.cuda.synchronize()

    gen = make_workload_gen(seed=1000 + rank, vocab=mc.vocab_size)
    stats = {"done": 0, "good": 0, "good_lo": 0, "good_hi": 0, "tokens": 0}
    ttfts: list = []
    tpots: list = []
    queues: list = []  # arrival -> first scheduled (admission wait)
    trace: list = []   # LIG_BENCH_TRACE=1: per-request tail diagnosis

    primed = [False]
    flood: list = []  # the initial simultaneous fill (cold-start population)

    def top_up():
        in_flight = eng.scheduler.num_waiting + eng.scheduler.num_running
        prime = not primed[0]
        while in_flight < args.concurrency:
            prompt, olen, lora = gen(prime=prime)
            seq = eng.add_request(
                prompt,
                SamplingParams(max_new_tokens=olen),
                lora_name=lora if args.lora else "",
            )
            if prime:
                flood.append(seq)
            in_flight += 1
        primed[0] = True

    def run_steps(n: int, record: bool) -> int:
        completions = 0
        for _ in range(n):
            top_up()
            eng.step()
            for seq in eng.collect_finished():
                completions += 1
                if record:
                    stats["done"] += 1
                    stats["tokens"] += seq.total_len
                    ttft = seq.ttft()
                    tpot = seq.tpot()
                    if ttft is not None:
                        ttfts.append(ttft)
                    if tpot is not None:
                        tpots.append(tpot)
                    if seq.first_scheduled_time is not None:
                        q = seq.first_scheduled_time - seq.arrival_time
                        queues.append(q)
                        if os.environ.get("LIG_BENCH_TRACE") == "1":
                            trace.append(
                                (q, seq.arrival_time, seq.first_scheduled_time,
                                 seq.first_token_time, seq.prompt_len,
                                 seq.num_output_tokens, seq.num_preemptions)
                            )
                    if ttft is not None and ttft <= TTFT_SLO_S:
                        stats["good"] += 1
                        if tpot is not None and tpot <= TPOT_SLO_LO_S:
                            stats["good_lo"] += 1
                        if tpot is not None and tpot <= TPOT_SLO_HI_S:
                            stats["good_hi"] += 1
        return completions

    # -- warmup (untimed) ----------------------------------------------------
    # Contract floor: W untimed steps.  Then warm BY STATE: the closed loop
    # starts with `concurrency` simultaneous arrivals whose prefill flood
    # dominates any short timed window (r01's 5.95 vs 32.9 req/s defect).
    # Keep stepping until EVERY member of that initial flood has finished —
    # any survivor would carry its cold-start admission wait (measured: the
    # last flood prompts admit ~2 s after t=0 at conc=256) into the timed
    # window's TTFT tail.  After that, every in-flight request arrived via
    # steady-state replacement, so a timed window of ANY length samples the
    # steady state.
    run_steps(args.warmup, record=False)
    warm_extra = 0
    warm_cap_steps = 4096
    warm_deadline = time.monotonic() + 150.0
    while (
        any(not s.is_finished() for s in flood)
        and warm_extra < warm_cap_steps
        and time.monotonic() < warm_deadline
    ):
        run_steps(8, record=False)
        warm_extra += 8
    flood.clear()  # release the references

    if distributed:
        torch.distributed.barrier()
    if is_cuda:
        torch.cuda.synchronize()
    t0 = time.monotonic()
    run_steps(args.steps, record=True)
    if is_cuda:
        torch.cuda.synchronize()
    if distributed:
        torch.distributed.barrier()
    t1 = time.monotonic()
    elapsed = t1 - t0

    # aggregate: max elapsed across ranks, sum of goodput counts
    if distributed:
        dev_t = torch.tensor([elapsed], dtype=torch.float64)
        cnt = torch.tensor(
            [stats["good"], stats["done"], stats["tokens"],
             stats["good_lo"], stats["good_hi"]],
            dtype=torch.float64,
        )
        if torch.cuda.is_available():
            dev_t, cnt = dev_t.cuda(), cnt.cuda()
        torch.distributed.all_reduce(dev_t, op=torch.distributed.ReduceOp.MAX)
        torch.distributed.all_reduce(cnt, op=torch.distributed.ReduceOp.SUM)
        elapsed = float(dev_t.item())
        good, done, tokens, good_lo, good_hi = (int(x) for x in cnt.tolist())
        gathered_t: list = [None] * world
        gathered_p: list = [None] * world
        torch.distributed.all_gather_object(gathered_t, ttfts)
        torch.distributed.all_gather_object(gathered_p, tpots)
        ttfts = [x for sub in gathered_t for x in sub]
        tpots = [x for sub in gathered_p for x in sub]
    else:
        good, done, tokens, good_lo, good_hi = (
            stats["good"], stats["done"], stats["tokens"],
            stats["good_lo"], stats["good_hi"],
        )

    def pct(xs, q):
        if not xs:
            return None
        xs = sorted(xs)
        return round(xs[min(len(xs) - 1, int(q * len(xs)))], 4)

    goodput = good / elapsed if elapsed > 0 else 0.0
    if rank == 0:
        result = {
            "metric": "goodput (req/s within TTFT SLO)",
            "value": round(goodput, 3),
            "unit": "req/s",
            "n_gpus": world if distributed else (args.gpus if is_cuda else 1),
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference publishes no numbers
            "dtype": "fp8" if args.quant == "fp8" else "bf16",
            "data": "synthetic (random-init weights, Poisson-shaped 202±20 in / 179±17 out)",
            "config": {
                "model": args.model,
                "global_batch": args.concurrency * (world if distributed else 1),
                "seq_len": PROMPT_MEAN + OUTPUT_MEAN,
                "parallelism": f"dp{world if distributed else 1}",
                "ttft_slo_s": TTFT_SLO_S,
                "tpot_slo_s": {"lo": TPOT_SLO_LO_S, "hi": TPOT_SLO_HI_S},
                "num_loras": NUM_LORAS if args.lora else 0,
                "completed_req": done,
                "total_tokens_per_s": round(tokens / elapsed, 1),
                "goodput_tier_lo_req_per_s": round(good_lo / elapsed, 3) if elapsed > 0 else 0.0,
                "goodput_tier_hi_req_per_s": round(good_hi / elapsed, 3) if elapsed > 0 else 0.0,
                "warmup_extra_steps_to_steady_state": warm_extra,
                "ttft_p50_s": pct(ttfts, 0.50),
                "ttft_p95_s": pct(ttfts, 0.95),
                "ttft_p99_s": pct(ttfts, 0.99),
                "tpot_p50_s": pct(tpots, 0.50),
                "tpot_p95_s": pct(tpots, 0.95),
                "tpot_p99_s": pct(tpots, 0.99),
                "queue_p50_s": pct(queues, 0.50),
                "queue_p95_s": pct(queues, 0.95),
            },
        }
        print(json.dumps(result))
        if trace:
            t_ref = min(t[1] for t in trace)
            trace.sort(reverse=True)
            print("# worst queue waits (q_s, arrive_rel_s, sched_rel_s, "
                  "ttft_rel_s, plen, out, preempts):", file=sys.stderr)
            for q, a, sc, ft, pl, no, pre in trace[:12]:
                print(f"#  {q:8.3f} {a - t_ref:8.3f} {sc - t_ref:8.3f} "
                      f"{(ft - t_ref) if ft else -1:8.3f} {pl:4d} {no:4d} {pre}",
                      file=sys.stderr)
    if distributed:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
