"""fp8 (e4m3) quantization semantics — CPU.

The CPU path of models/quant.py runs the SAME quantized values through an
fp32 matmul, so these tests pin the quantization scheme itself; the GPU
_scaled_mm path (tests/test_quant_gpu.py) must then agree with it up to
accumulation order.
"""

import pytest
import torch

from llm_instance_gateway_amd.models.quant import (
    FP8_MAX,
    Fp8Linear,
    convert_linears_to_fp8,
    quantize_fp8_rowwise,
)


def test_quantize_roundtrip_error_bounded():
    torch.manual_seed(0)
    w = torch.randn(64, 256, dtype=torch.bfloat16)
    q, s = quantize_fp8_rowwise(w)
    assert q.dtype == torch.float8_e4m3fn and s.shape == (64, 1)
    deq = q.float() * s
    # e4m3 has a 3-bit mantissa: elementwise relative error <= 2^-4 at the
    # top binade; allow 2x slack for values far below the row amax
    err = (deq - w.float()).abs()
    rel = err / w.float().abs().clamp(min=1e-3)
    assert rel.median() < 0.05
    assert (deq.abs() <= FP8_MAX * s).all()


def test_quantize_scale_is_per_row():
    w = torch.zeros(4, 8)
    w[0, 0] = 448.0  # row 0 amax exactly FP8_MAX -> scale 1
    w[1, 3] = 44.8
    q, s = quantize_fp8_rowwise(w)
    assert torch.isclose(s[0, 0], torch.tensor(1.0))
    assert torch.isclose(s[1, 0], torch.tensor(0.1), rtol=1e-3)
    assert float(q[0, 0].float()) == 448.0


def test_fp8_linear_close_to_bf16_linear():
    torch.manual_seed(1)
    lin = torch.nn.Linear(128, 96, bias=False, dtype=torch.bfloat16)
    with torch.no_grad():
        lin.weight.uniform_(-0.05, 0.05)
    qlin = Fp8Linear(lin)
    x = torch.randn(32, 128, dtype=torch.bfloat16)
    y_ref = lin(x).float()
    y_q = qlin(x).float()
    # weight + activation quant noise: relative Frobenius error ~1-3%
    rel = (y_q - y_ref).norm() / y_ref.norm()
    assert rel < 0.08, rel
    # shape/dtype contract
    assert y_q.shape == (32, 96)
    assert qlin(x).dtype == torch.bfloat16


def test_fp8_linear_3d_input():
    lin = torch.nn.Linear(16, 8, bias=False, dtype=torch.bfloat16)
    q = Fp8Linear(lin)
    x = torch.randn(2, 5, 16, dtype=torch.bfloat16)
    assert q(x).shape == (2, 5, 8)


def test_convert_llama_model_counts_and_runs():
    from llm_instance_gateway_amd.models.config import get_model_config
    from llm_instance_gateway_amd.models.llama import LlamaForCausalLM

    mc = get_model_config("tiny")
    torch.manual_seed(0)
    model = LlamaForCausalLM(mc, device="cpu", dtype=torch.bfloat16)
    n = convert_linears_to_fp8(model)
    assert n == 4 * mc.num_layers
    # projections converted, everything else untouched
    l0 = model.layers[0]
    assert isinstance(l0.self_attn.qkv_proj, Fp8Linear)
    assert isinstance(l0.mlp.down_proj, Fp8Linear)
    assert isinstance(model.embed_tokens, torch.nn.Embedding)


def test_fp8_model_logits_close_to_bf16():
    """Model-level numerics: fp8-projected forward vs the bf16 model on the
    same weights — logits cosine similarity stays high (quant noise is
    per-GEMM ~1-2% relative; depth compounds it but the residual stream
    keeps the signal dominant)."""
    from llm_instance_gateway_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from llm_instance_gateway_amd.models.config import get_model_config

    mc = get_model_config("tiny")

    def logits_of(quant: str):
        cfg = EngineConfig(
            model=mc, num_kv_blocks=64, max_num_seqs=2,
            max_num_batched_tokens=128, max_model_len=128,
            quant=quant, pad_batch_multiple=0,
        )
        torch.manual_seed(21)  # identical base weights both times
        eng = LLMEngine(cfg, device="cpu")
        captured = []
        import llm_instance_gateway_amd.engine.model_runner as mr

        orig = mr.sample_tokens

        def capture(logits, seqs):
            captured.append(logits.clone())
            return orig(logits, seqs)

        mr.sample_tokens = capture
        try:
            eng.add_request(list(range(5, 37)), SamplingParams(max_new_tokens=1))
            for _ in range(4):
                eng.step()
                if captured:
                    break
        finally:
            mr.sample_tokens = orig
        return captured[0][0].float()

    lb = logits_of("")
    lq = logits_of("fp8")
    cos = torch.nn.functional.cosine_similarity(lb, lq, dim=0)
    assert float(cos) > 0.98, float(cos)


def test_engine_fp8_generates_tokens():
    """End-to-end: a tiny engine with quant='fp8' decodes valid tokens on
    CPU (dequantized-reference path)."""
    from llm_instance_gateway_amd.engine import (
        EngineConfig,
        LLMEngine,
        SamplingParams,
    )
    from llm_instance_gateway_amd.models.config import get_model_config

    mc = get_model_config("tiny")
    cfg = EngineConfig(
        model=mc,
        num_kv_blocks=64,
        max_num_seqs=4,
        max_num_batched_tokens=128,
        max_model_len=128,
        quant="fp8",
        pad_batch_multiple=0,
    )
    torch.manual_seed(7)
    eng = LLMEngine(cfg, device="cpu")
    seq = eng.add_request([5, 6, 7, 8], SamplingParams(max_new_tokens=8))
    for _ in range(16):
        eng.step()
        if seq.is_finished():
            break
    assert seq.is_finished()
    assert len(seq.output_token_ids) == 8
    assert all(0 <= t < mc.vocab_size for t in seq.output_token_ids)


def test_engine_fp8_with_lora_adapters():
    """LoRA rides on top of fp8 base GEMMs unchanged: SGMV applies bf16
    deltas to the (bf16) GEMM output, so the adapter path must keep
    working when the projections are Fp8Linear."""
    from llm_instance_gateway_amd.engine import (
        EngineConfig,
        LLMEngine,
        SamplingParams,
    )
    from llm_instance_gateway_amd.lora.manager import AdapterManager
    from llm_instance_gateway_amd.models.config import get_model_config

    mc = get_model_config("tiny")
    cfg = EngineConfig(
        model=mc,
        num_kv_blocks=64,
        max_num_seqs=4,
        max_num_batched_tokens=128,
        max_model_len=128,
        quant="fp8",
        max_loras=2,
        pad_batch_multiple=0,
    )
    torch.manual_seed(11)
    eng = LLMEngine(cfg, device="cpu")
    mgr = AdapterManager(eng, max_slots=2, rank=8)
    mgr.register_random("a1", seed=1)
    mgr.prepare(["a1"])
    base = eng.add_request([5, 6, 7], SamplingParams(max_new_tokens=6))
    lora = eng.add_request([5, 6, 7], SamplingParams(max_new_tokens=6), lora_name="a1")
    for _ in range(24):
        eng.step()
        if base.is_finished() and lora.is_finished():
            break
    assert base.is_finished() and lora.is_finished()
    # the adapter delta must actually change the output
    assert base.output_token_ids != lora.output_token_ids


def _tp_fp8_worker(rank: int, world: int, q, port: int):
    try:
        import os

        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        from llm_instance_gateway_amd.engine import (
            EngineConfig,
            LLMEngine,
            SamplingParams,
        )
        from llm_instance_gateway_amd.models.config import get_model_config
        from llm_instance_gateway_amd.models.quant import Fp8Linear
        from llm_instance_gateway_amd.parallel import TPGroup, init_distributed

        init_distributed(backend="gloo", rank=rank, world_size=world)
        tp = TPGroup()
        mc = get_model_config("tiny")
        torch.manual_seed(5)
        eng = LLMEngine(
            EngineConfig(
                model=mc, num_kv_blocks=64, max_num_seqs=4,
                max_num_batched_tokens=128, max_model_len=128,
                quant="fp8", max_loras=0,
            ),
            device="cpu",
            tp=tp,
        )
        # sharded projections were converted (per-rank scales)
        assert isinstance(eng.runner.model.layers[0].self_attn.qkv_proj, Fp8Linear)
        seq = eng.add_request(list(range(8, 24)), SamplingParams(max_new_tokens=6))
        out = eng.generate_all()[0].output_token_ids
        assert len(out) == 6 and all(0 <= t < mc.vocab_size for t in out)
        # all ranks must agree token-for-token (deterministic TP execution)
        import torch.distributed as dist

        t = torch.tensor(out, dtype=torch.int64)
        ref = t.clone()
        dist.broadcast(ref, src=0)
        assert torch.equal(t, ref), (rank, out)
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        q.put((rank, f"FAIL: {type(e).__name__}: {e}"))


def test_tp2_fp8_engine_generates():
    """pool.quant=fp8 composes with tensor_parallel: sharded projections
    quantize per-rank and the group generates identical tokens on every
    rank (gloo; the GPU path runs the same code over RCCL)."""
    import multiprocessing as mp

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=_tp_fp8_worker, args=(r, 2, q, 29557))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in procs]
    for p in procs:
        p.join(timeout=30)
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"


def test_fp8_rejects_moe():
    from llm_instance_gateway_amd.engine import EngineConfig, LLMEngine
    from llm_instance_gateway_amd.models.config import get_model_config

    mc = get_model_config("tiny-moe")
    cfg = EngineConfig(
        model=mc, num_kv_blocks=16, max_num_seqs=2,
        max_num_batched_tokens=64, max_model_len=64, quant="fp8",
    )
    with pytest.raises(ValueError, match="dense"):
        LLMEngine(cfg, device="cpu")
