"""fp8 GPU path: _scaled_mm (hipBLASLt fp8 MFMA) vs the CPU
dequantized-fp32 reference of the SAME quantized values, plus an engine
e2e smoke with quant='fp8'."""

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():  # collected on CPU boxes; never run
    pytest.skip("GPU only", allow_module_level=True)

from llm_instance_gateway_amd.models.quant import Fp8Linear


@pytest.mark.parametrize("T,K", [(1, 4096), (384, 4096), (777, 14336)])
def test_fused_quant_kernel_matches_torch(T, K):
    from llm_instance_gateway_amd import ops
    from llm_instance_gateway_amd.models.quant import FP8_MAX

    torch.manual_seed(0)
    x = (torch.randn(T, K, dtype=torch.bfloat16) * 0.3).cuda()
    xq, xs = ops.quant_fp8_rowwise(x)
    assert xq.dtype == torch.float8_e4m3fn and xs.shape == (T, 1)
    # reference: torch elementwise path on the same device
    amax = x.abs().amax(dim=-1, keepdim=True).float().clamp_(min=1e-6)
    rs = amax / FP8_MAX
    rq = (x / rs).clamp_(-FP8_MAX, FP8_MAX).to(torch.float8_e4m3fn)
    assert torch.allclose(xs, rs, rtol=1e-6)
    # e4m3 values: identical up to an occasional 1-ulp rounding-boundary flip
    # from inv-scale multiply vs scale divide
    diff = (xq.float() - rq.float()).abs()
    denom = rq.float().abs().clamp(min=1e-3)
    assert (diff / denom > 0.15).float().mean() < 1e-3


@pytest.mark.parametrize("M,K,N", [(1, 4096, 6144), (256, 4096, 6144), (384, 14336, 4096)])
def test_scaled_mm_matches_dequant_reference(M, K, N):
    torch.manual_seed(0)
    lin = torch.nn.Linear(K, N, bias=False, dtype=torch.bfloat16)
    with torch.no_grad():
        lin.weight.uniform_(-0.02, 0.02)
    q_cpu = Fp8Linear(lin)
    q_gpu = Fp8Linear(lin).cuda()
    x = torch.randn(M, K, dtype=torch.bfloat16) * 0.5
    y_cpu = q_cpu(x).float()
    y_gpu = q_gpu(x.cuda()).float().cpu()
    # identical quantized operands; only the accumulation order differs
    rel = (y_gpu - y_cpu).norm() / y_cpu.norm().clamp(min=1e-9)
    assert rel < 2e-2, rel


def test_engine_fp8_gpu_generates():
    from llm_instance_gateway_amd.engine import (
        EngineConfig,
        LLMEngine,
        SamplingParams,
    )
    from llm_instance_gateway_amd.models.config import get_model_config

    mc = get_model_config("llama3-8b")
    cfg = EngineConfig(
        model=mc,
        num_kv_blocks=512,
        max_num_seqs=8,
        max_num_batched_tokens=512,
        max_model_len=256,
        quant="fp8",
    )
    torch.manual_seed(3)
    eng = LLMEngine(cfg, device="cuda:0")
    seqs = [
        eng.add_request(list(range(10 + i, 42 + i)), SamplingParams(max_new_tokens=16))
        for i in range(4)
    ]
    for _ in range(64):
        eng.step()
        if all(s.is_finished() for s in seqs):
            break
    assert all(s.is_finished() for s in seqs)
    for s in seqs:
        assert len(s.output_token_ids) == 16
        assert all(0 <= t < mc.vocab_size for t in s.output_token_ids)
