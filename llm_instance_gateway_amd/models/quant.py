"""Opt-in fp8 (OCP e4m3) weight+activation quantization for the dense
projection GEMMs.

MI355X rationale: the decode-path projections are WEIGHT-STREAMING bound
(hipBLASLt skinny-M GEMMs run at 2.8-3.7 TB/s of weight traffic —
profiles/r02_kernel_stats.md), and CDNA4's MFMA fp8 peak is 2x the bf16
peak.  Storing the projection weights as fp8 halves the bytes the decode
step must stream from HBM3E and lets prefill GEMMs use the fp8 pipes.

Scheme (the standard dynamic-quant serving recipe):
  * weights:     symmetric per-output-channel e4m3, quantized once at load;
  * activations: symmetric per-token (row-wise) e4m3, computed on the fly
    (elementwise amax + scale: graph-capturable, no host sync);
  * GEMM:        torch._scaled_mm(x_fp8, w_fp8.t(), scale_a=[M,1],
                 scale_b=[1,N], out_dtype=bf16) -> hipBLASLt fp8 MFMA path;
  * everything else (KV cache, attention, LoRA deltas, lm_head, norms)
    stays bf16 — LoRA SGMV applies bf16 deltas on the bf16 GEMM output
    exactly as in the bf16 path.

Opt-in via EngineConfig.quant = "fp8" (bench.py --quant fp8).  The
headline benchmark stays bf16 (BASELINE's dtype); fp8 numbers are
reported separately and marked as such.  Reference analogue: vLLM's
--quantization fp8 serving mode on the fleet the reference routes to
(SURVEY.md §2.3); implementation is original and MI355X-native.

MEASURED STATUS (r02, 1x MI355X, tests/test_quant_gpu.py 7/7 green):
  * with torch elementwise activation quant (~4 launches + 3 passes over
    x per projection): a NON-win — 69.4 vs 74.6 req/s at conc=384
    (gpurun_out/bench_fp8.log);
  * with the fused ONE-launch quant kernel (ops.quant_fp8_rowwise,
    ops/csrc/quant.hip): **84.97 vs 74.6 req/s at conc=384 (+14%)** and
    35.7 vs 31.6 at conc=64 (+13%) — bench_fp8_fused.log,
    bench_c64_fp8_fused.log.  The quant overhead was the entire penalty.
The headline benchmark still reports bf16 (BASELINE's dtype); fp8 runs
carry dtype="fp8" in the bench JSON.

CPU fallback (tests): the same quantized values are dequantized and the
matmul runs in fp32 — numerics match the GPU path up to accumulation
order, so CPU tests pin the quantization semantics.
"""

from __future__ import annotations

import torch
from torch import nn

FP8_DTYPE = torch.float8_e4m3fn
FP8_MAX = 448.0  # largest finite e4m3fn magnitude


def quantize_fp8_rowwise(w: torch.Tensor):
    """Symmetric per-row (dim=-1) e4m3 quantization.

    Returns (w_fp8 same shape, scale fp32 [..., 1]) with
    w ≈ w_fp8.float() * scale.
    """
    amax = w.abs().amax(dim=-1, keepdim=True).float().clamp_(min=1e-12)
    scale = amax / FP8_MAX
    q = (w.float() / scale).clamp_(-FP8_MAX, FP8_MAX).to(FP8_DTYPE)
    return q, scale


class Fp8Linear(nn.Module):
    """Drop-in replacement for a bias-free nn.Linear with fp8 weights.

    Built FROM an initialized nn.Linear; the bf16 weight is freed after
    quantization (that is the point: half the HBM bytes per decode step).
    """

    def __init__(self, linear: nn.Linear) -> None:
        super().__init__()
        assert linear.bias is None, "Fp8Linear supports bias-free projections"
        w = linear.weight.data  # [N, K]
        self.out_features, self.in_features = w.shape
        q, s = quantize_fp8_rowwise(w)
        # [N, K] row-major; .t() is the column-major [K, N] _scaled_mm wants
        self.register_buffer("weight_fp8", q, persistent=False)
        self.register_buffer("weight_scale", s, persistent=False)  # [N, 1] f32
        self._out_dtype = w.dtype

    def _quant_input(self, x2: torch.Tensor):
        if x2.device.type == "cuda":
            from .. import ops

            return ops.quant_fp8_rowwise(x2)  # ONE fused kernel
        amax = x2.abs().amax(dim=-1, keepdim=True).float().clamp_(min=1e-6)
        xs = amax / FP8_MAX
        xq = (x2 / xs).clamp_(-FP8_MAX, FP8_MAX).to(FP8_DTYPE)
        return xq, xs

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x2 = x.reshape(-1, self.in_features)
        xq, xs = self._quant_input(x2)
        if x2.device.type == "cuda":
            y = torch._scaled_mm(
                xq,
                self.weight_fp8.t(),
                scale_a=xs,
                scale_b=self.weight_scale.t(),
                out_dtype=self._out_dtype,
            )
        else:
            # fp32 reference of the same quantized values (tests)
            xdq = xq.float() * xs
            wdq = self.weight_fp8.float() * self.weight_scale
            y = (xdq @ wdq.t()).to(self._out_dtype)
        return y.reshape(*x.shape[:-1], self.out_features)

    def extra_repr(self) -> str:
        return f"in={self.in_features}, out={self.out_features}, fp8=e4m3"


def convert_linears_to_fp8(model: nn.Module) -> int:
    """Replace the projection linears of a Llama-family model with
    Fp8Linear.  Embedding, lm_head, norms, KV path stay bf16.  Returns the
    number of modules converted.  MoE expert weights (bmm layout) are not
    covered yet — dense models only."""
    n = 0
    layers = getattr(model, "layers", None)
    if layers is None:
        return 0
    for layer in layers:
        attn = getattr(layer, "self_attn", None)
        mlp = getattr(layer, "mlp", None)
        targets = []
        if attn is not None:
            targets += [(attn, "qkv_proj"), (attn, "o_proj")]
        if mlp is not None:
            targets += [(mlp, "gate_up_proj"), (mlp, "down_proj")]
        for owner, name in targets:
            old = getattr(owner, name, None)
            if isinstance(old, nn.Linear):
                setattr(owner, name, Fp8Linear(old))
                n += 1
    return n
