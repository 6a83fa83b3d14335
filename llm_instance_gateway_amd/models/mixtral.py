"""Mixtral-family sparse-MoE transformer (BASELINE config #5).

Same attention / norm / paged-KV path as the Llama family
(`models/llama.py` — the Attention module is reused verbatim); the dense
SwiGLU MLP is replaced by a top-k routed expert block.

MI355X design notes:

  * Experts are **TP-sharded** (every rank holds a 1/tp slice of every
    expert's intermediate dim) rather than expert-parallel: with 288 GB HBM
    per GPU the full 8x7B expert set fits on one device, so EP's all-to-all
    over xGMI is not needed for parity, and the per-block all-reduce pattern
    stays identical to the dense model.
  * Dispatch is sort-by-expert then one (gemm -> silu*mul -> gemm) per
    resident expert through hipBLASLt; tokens-per-expert at serving batch
    sizes keeps each GEMM MFMA-shaped.  A fused grouped-GEMM HIP kernel is
    the planned next step (same segmented pattern as the LoRA SGMV kernels,
    `ops/csrc/sgmv.hip`).
  * MoE engines run eager (no hipGraph decode capture): expert routing is
    token-dependent, so a captured graph would freeze the dispatch of the
    capture batch (`engine/model_runner.py` gates on `is_moe`).

LoRA: adapters attach to the attention projections only (qkv / o); routed
expert weights are not adapter targets — matching what the reference fleet
exercises (its 12 adapters ride a dense Llama base,
`vllm-lora-deployment.yaml:46-60`).
"""

from __future__ import annotations

from typing import List

import torch
import torch.nn as nn

from .. import ops
from ..engine.metadata import BatchMeta
from .config import ModelConfig
from .llama import Attention, _linear


class MoEBlock(nn.Module):
    """Top-k routed experts with fused gate+up per expert."""

    def __init__(self, cfg: ModelConfig, device, dtype, tp=None) -> None:
        super().__init__()
        self.tp = tp
        tp_size = tp.world_size if tp is not None else 1
        assert cfg.intermediate_size % tp_size == 0
        self.num_experts = cfg.num_experts
        self.top_k = cfg.num_experts_per_tok
        inter = cfg.intermediate_size // tp_size
        h = cfg.hidden_size
        # router is replicated (tiny GEMM, identical on every rank)
        self.gate = _linear(h, cfg.num_experts, device, dtype)
        # [E, 2*inter, h] and [E, h, inter], bounded random init
        self.w13 = nn.Parameter(
            torch.empty(cfg.num_experts, 2 * inter, h, device=device, dtype=dtype)
            .uniform_(-0.02, 0.02)
        )
        self.w2 = nn.Parameter(
            torch.empty(cfg.num_experts, h, inter, device=device, dtype=dtype)
            .uniform_(-0.02, 0.02)
        )

    def forward(self, x: torch.Tensor, meta=None) -> torch.Tensor:
        T, h = x.shape
        router_logits = self.gate(x)  # [T, E]
        weights, experts = torch.topk(router_logits, self.top_k, dim=-1)
        # Mixtral normalizes softmax over the selected k only
        weights = torch.softmax(weights.float(), dim=-1).to(x.dtype)  # [T, k]

        out = torch.zeros_like(x)
        flat_experts = experts.reshape(-1)  # [T*k]
        flat_tokens = (
            torch.arange(T, device=x.device).unsqueeze(1).expand(T, self.top_k).reshape(-1)
        )
        flat_w = weights.reshape(-1)
        # sort the (token, expert) pairs by expert -> contiguous per-expert runs
        order = torch.argsort(flat_experts, stable=True)
        sorted_experts = flat_experts[order]
        sorted_tokens = flat_tokens[order]
        sorted_w = flat_w[order]
        counts = torch.bincount(sorted_experts, minlength=self.num_experts)
        starts = torch.cumsum(counts, 0) - counts
        counts_l = counts.tolist()
        starts_l = starts.tolist()
        for e in range(self.num_experts):
            n = counts_l[e]
            if n == 0:
                continue
            sl = slice(starts_l[e], starts_l[e] + n)
            tok = sorted_tokens[sl]
            xe = x.index_select(0, tok)
            act = ops.silu_and_mul(xe @ self.w13[e].t())
            ye = act @ self.w2[e].t()
            out.index_add_(0, tok, ye * sorted_w[sl].unsqueeze(1))
        if self.tp is not None:
            out = self.tp.all_reduce(out)
        return out


class MixtralDecoderLayer(nn.Module):
    def __init__(self, cfg: ModelConfig, device, dtype, tp=None) -> None:
        super().__init__()
        self.input_layernorm = nn.Parameter(
            torch.ones(cfg.hidden_size, device=device, dtype=dtype)
        )
        self.post_attention_layernorm = nn.Parameter(
            torch.ones(cfg.hidden_size, device=device, dtype=dtype)
        )
        self.eps = cfg.rms_norm_eps
        self.self_attn = Attention(cfg, device, dtype, tp)
        self.block_sparse_moe = MoEBlock(cfg, device, dtype, tp)

    def forward(self, hidden, residual, positions, kv_cache, meta, cos_sin_cache):
        if residual is None:
            residual = hidden
            h = ops.rms_norm(hidden, self.input_layernorm, self.eps)
        else:
            h, residual = ops.fused_add_rms_norm(
                hidden, residual, self.input_layernorm, self.eps
            )
        h = self.self_attn(h, positions, kv_cache, meta, cos_sin_cache)
        h, residual = ops.fused_add_rms_norm(
            h, residual, self.post_attention_layernorm, self.eps
        )
        h = self.block_sparse_moe(h, meta)
        return h, residual


class MixtralForCausalLM(nn.Module):
    def __init__(self, cfg: ModelConfig, device="cpu", dtype=torch.bfloat16, tp=None) -> None:
        super().__init__()
        self.cfg = cfg
        self.tp = tp
        self.embed_tokens = nn.Embedding(
            cfg.vocab_size, cfg.hidden_size, device="meta", dtype=dtype
        ).to_empty(device=device)
        with torch.no_grad():
            self.embed_tokens.weight.uniform_(-0.02, 0.02)
        self.layers = nn.ModuleList(
            [MixtralDecoderLayer(cfg, device, dtype, tp) for _ in range(cfg.num_layers)]
        )
        self.norm = nn.Parameter(torch.ones(cfg.hidden_size, device=device, dtype=dtype))
        self.lm_head = _linear(cfg.hidden_size, cfg.vocab_size, device, dtype)
        self.cos_sin_cache = ops.build_cos_sin_cache(
            cfg.head_dim, cfg.max_position_embeddings, cfg.rope_theta, device
        ).to(dtype=torch.float32)

    def forward(
        self,
        input_ids: torch.Tensor,
        positions: torch.Tensor,
        kv_caches: List,
        meta: BatchMeta,
    ) -> torch.Tensor:
        hidden = self.embed_tokens(input_ids)
        residual = None
        for layer, kv in zip(self.layers, kv_caches):
            hidden, residual = layer(
                hidden, residual, positions, kv, meta, self.cos_sin_cache
            )
        hidden, _ = ops.fused_add_rms_norm(hidden, residual, self.norm, self.cfg.rms_norm_eps)
        return hidden

    def compute_logits(self, hidden: torch.Tensor, indices: torch.Tensor) -> torch.Tensor:
        h = hidden[indices]
        logits = h @ self.lm_head.weight.t()
        return logits.float()
