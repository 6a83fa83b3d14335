"""Property-based fp8 quantization invariants (hypothesis).

For ANY bf16 input, the row-wise e4m3 scheme must (a) round-trip within
the e4m3 precision bound relative to each row's amax, (b) never produce
non-finite quantized values, (c) be scale-invariant (quantizing c*x gives
c*scales and identical fp8 payloads for power-of-two c).
"""

import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from llm_instance_gateway_amd.models.quant import (
    FP8_MAX,
    quantize_fp8_rowwise,
)


@settings(max_examples=60, deadline=None)
@given(
    rows=st.integers(1, 8),
    cols=st.sampled_from([8, 16, 64, 256]),
    scale_exp=st.integers(-8, 8),
    seed=st.integers(0, 2**31 - 1),
)
def test_roundtrip_bounded_by_row_amax(rows, cols, scale_exp, seed):
    g = torch.Generator().manual_seed(seed)
    w = (torch.randn(rows, cols, generator=g) * (2.0 ** scale_exp)).to(
        torch.bfloat16
    )
    q, s = quantize_fp8_rowwise(w)
    deq = q.float() * s
    assert torch.isfinite(deq).all()
    # e4m3: 3 mantissa bits -> max quantization step is amax/448 * 32
    # (the largest binade's spacing); bound absolute error per row by
    # amax * 2^-3 / ... use the conservative bound step <= amax/448*64
    amax = w.float().abs().amax(dim=-1, keepdim=True)
    err = (deq - w.float()).abs()
    bound = amax / 448.0 * 32 + 1e-12
    assert (err <= bound + amax * 0.002).all(), (
        err.max(), bound.max(), amax.max())


@settings(max_examples=40, deadline=None)
@given(
    c_exp=st.integers(-4, 4),
    seed=st.integers(0, 2**31 - 1),
)
def test_power_of_two_scale_invariance(c_exp, seed):
    g = torch.Generator().manual_seed(seed)
    w = torch.randn(4, 64, generator=g).to(torch.bfloat16)
    c = 2.0 ** c_exp
    q1, s1 = quantize_fp8_rowwise(w)
    q2, s2 = quantize_fp8_rowwise((w.float() * c).to(torch.bfloat16))
    # bf16 holds powers of two exactly -> identical fp8 payloads,
    # scales exactly c apart
    assert torch.equal(q1.view(torch.uint8), q2.view(torch.uint8))
    assert torch.allclose(s2, s1 * c, rtol=1e-6)


@settings(max_examples=30, deadline=None)
@given(seed=st.integers(0, 2**31 - 1))
def test_zero_and_tiny_rows_safe(seed):
    g = torch.Generator().manual_seed(seed)
    w = torch.zeros(3, 32, dtype=torch.bfloat16)
    w[1] = (torch.randn(32, generator=g) * 1e-30).to(torch.bfloat16)
    q, s = quantize_fp8_rowwise(w)
    assert torch.isfinite(s).all() and (s > 0).all()
    assert torch.isfinite(q.float()).all()
    assert (q.float()[0] == 0).all()
