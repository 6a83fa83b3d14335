"""Expert-parallel MoE correctness on CPU (gloo, world_size=2): each rank
owns half the experts at full intermediate width; the per-block all-reduce
sums the partial expert outputs — token-identical to the unsharded engine
(models/mixtral.py MoEBlock docstring; BASELINE config #5 stretch)."""

import dataclasses
import multiprocessing as mp
import os

import torch

from llm_instance_gateway_amd.models.config import get_model_config

PORT = 29531


def _copy_ep_slices(full, ep_model, rank: int, world: int) -> None:
    """EP weights are expert ranges of the full model; attention stays
    TP-sharded over the same group (the mixed TP-attention + EP-MoE engine),
    so qkv/o get their usual column/row shards; router, norms, embeddings
    replicated."""
    cfg = full.cfg
    hd, nh, nkv = cfg.head_dim, cfg.num_heads, cfg.num_kv_heads
    lq, lkv = nh // world * hd, nkv // world * hd
    E_local = cfg.num_experts // world
    lo, hi = rank * E_local, (rank + 1) * E_local
    sd = {}
    for k, v in full.state_dict().items():
        if ".block_sparse_moe.w13" in k or ".block_sparse_moe.w2" in k:
            sd[k] = v[lo:hi]
        elif k.endswith("self_attn.qkv_proj.weight"):
            qf, kf, vf = v.split([nh * hd, nkv * hd, nkv * hd], dim=0)
            sd[k] = torch.cat(
                [
                    qf[rank * lq:(rank + 1) * lq],
                    kf[rank * lkv:(rank + 1) * lkv],
                    vf[rank * lkv:(rank + 1) * lkv],
                ],
                dim=0,
            )
        elif k.endswith("self_attn.o_proj.weight"):
            sd[k] = v[:, rank * lq:(rank + 1) * lq]
        else:
            sd[k] = v
    ep_model.load_state_dict(sd)


def _ep_worker(rank: int, world: int, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(PORT)
        from llm_instance_gateway_amd.engine import EngineConfig, LLMEngine, SamplingParams
        from llm_instance_gateway_amd.parallel import TPGroup, init_distributed

        init_distributed(backend="gloo", rank=rank, world_size=world)
        tp = TPGroup()

        mc = get_model_config("tiny-moe")
        mc_ep = dataclasses.replace(mc, expert_parallel=True)
        ekw = dict(
            num_kv_blocks=64,
            max_num_seqs=4,
            max_num_batched_tokens=128,
            max_model_len=128,
            max_loras=0,
        )
        torch.manual_seed(7)
        full_eng = LLMEngine(EngineConfig(model=mc, **ekw), device="cpu")
        torch.manual_seed(123)
        ep_eng = LLMEngine(EngineConfig(model=mc_ep, **ekw), device="cpu", tp=tp)
        blk = ep_eng.runner.model.layers[0].block_sparse_moe
        assert blk.ep and blk.local_experts == mc.num_experts // world
        assert blk.w13.shape[2] == 2 * mc.intermediate_size  # full width
        _copy_ep_slices(full_eng.runner.model, ep_eng.runner.model, rank, world)

        prompt = list(range(11, 43))
        sp = SamplingParams(max_new_tokens=8)
        full_eng.add_request(prompt, sp)
        ep_eng.add_request(prompt, sp)
        ref = full_eng.generate_all()[0].output_token_ids
        out = ep_eng.generate_all()[0].output_token_ids
        assert out == ref, f"rank{rank}: EP tokens {out} != reference {ref}"
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, f"{e}\n{traceback.format_exc()}"))


def test_ep2_token_identical_generation():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_ep_worker, args=(r, 2, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=300) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"


def test_ep_block_single_rank_noop():
    """expert_parallel with world_size 1 degrades to the TP=1 path."""
    import dataclasses

    from llm_instance_gateway_amd.models.mixtral import MoEBlock

    mc = dataclasses.replace(get_model_config("tiny-moe"), expert_parallel=True)
    blk = MoEBlock(mc, device="cpu", dtype=torch.float32)  # tp=None
    assert not blk.ep and blk.local_experts == mc.num_experts
    x = torch.randn(5, mc.hidden_size)
    assert blk(x).shape == (5, mc.hidden_size)
