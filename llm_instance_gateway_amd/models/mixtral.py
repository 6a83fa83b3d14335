"""Mixtral-family sparse-MoE transformer (BASELINE config #5).

Same attention / norm / paged-KV path as the Llama family
(`models/llama.py` — the Attention module is reused verbatim); the dense
SwiGLU MLP is replaced by a top-k routed expert block.

MI355X design notes:

  * Two multi-GPU modes: **TP-sharded** experts (default: every rank holds a
    1/tp slice of every expert's intermediate dim) and **expert-parallel**
    (``ModelConfig.expert_parallel``: each rank owns E/world full experts —
    see MoEBlock docstring for why the combine stays one all-reduce over
    xGMI rather than an all-to-all in this replicated-activation design).
    With 288 GB HBM the full 8x7B expert set also fits one device (TP=1).
  * Dispatch is sort-by-expert into a capacity-padded [E, cap, h] batch and
    ONE bmm pair (w13 fused gate+up, then w2) through hipBLASLt — the
    per-expert Python loop was host-launch-bound (E x 3 tiny ops x layers).
    Padding rows read a zero row and scatter back with weight 0 (bit-exact,
    no token dropping; verified against the per-token reference in
    tests/test_mixtral.py).
  * Decode-only batches use a DENSE expert dispatch (every expert computes
    every token; unselected experts gated by weight 0): static shapes and no
    host sync make decode steps hipGraph-capturable, and decode MoE is bound
    by streaming all expert weights (2.8 GB/layer for 8x7B) regardless of
    routing, so the extra rows are nearly free.  Prefill/mixed batches —
    compute-bound — use the capacity-routed bmm dispatch.

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
    """Top-k routed experts with fused gate+up per expert.

    Parallel modes over the engine's process group:
      * TP (default): every rank holds a 1/tp slice of EVERY expert's
        intermediate dim; combine = the block all-reduce.
      * EP (``cfg.expert_parallel``): each rank OWNS num_experts/world full
        experts and computes partial token outputs for the experts it owns;
        the same all-reduce combines the partials.  This is the xGMI-shaped
        expert parallelism for this engine model: activations are replicated
        across the TP group by construction (every rank runs the identical
        batch), so the classic all-to-all token exchange would move data that
        is already resident — the collective stays ONE all-reduce per block
        while each expert's GEMMs run at full intermediate width.  A2A-based
        dispatch only pays when activations are sharded across ranks (a
        DP-coupled fleet), which this design avoids.
    """

    def __init__(self, cfg: ModelConfig, device, dtype, tp=None) -> None:
        super().__init__()
        self.tp = tp
        tp_size = tp.world_size if tp is not None else 1
        self.ep = bool(getattr(cfg, "expert_parallel", False)) and tp_size > 1
        if self.ep:
            assert cfg.num_experts % tp_size == 0, "EP needs E % world == 0"
            self.local_experts = cfg.num_experts // tp_size
            self.expert_base = tp.rank * self.local_experts
            inter = cfg.intermediate_size  # full width per owned expert
        else:
            assert cfg.intermediate_size % tp_size == 0
            self.local_experts = cfg.num_experts
            self.expert_base = 0
            inter = cfg.intermediate_size // tp_size
        self.num_experts = cfg.num_experts
        self.top_k = cfg.num_experts_per_tok
        h = cfg.hidden_size
        # router is replicated (tiny GEMM, identical on every rank)
        self.gate = _linear(h, cfg.num_experts, device, dtype)
        # Stored PRE-TRANSPOSED for bmm ([E, h, 2*inter] and [E, inter, h]):
        # feeding bmm a transposed non-contiguous weight view memory-faults
        # in hipBLASLt's strided-batched path at large batch (reproduced at
        # [8,1073,4096]x[8,4096,28672]^T on MI355X; contiguous operands are
        # also the faster layout).
        self.w13 = nn.Parameter(
            torch.empty(self.local_experts, h, 2 * inter, device=device, dtype=dtype)
            .uniform_(-0.02, 0.02)
        )
        self.w2 = nn.Parameter(
            torch.empty(self.local_experts, inter, h, device=device, dtype=dtype)
            .uniform_(-0.02, 0.02)
        )

    def forward(self, x: torch.Tensor, meta=None) -> torch.Tensor:
        T, h = x.shape
        router_logits = self.gate(x)  # [T, E]
        weights, experts = torch.topk(router_logits, self.top_k, dim=-1)
        # Mixtral normalizes softmax over the selected k only
        weights = torch.softmax(weights.float(), dim=-1).to(x.dtype)  # [T, k]

        # Decode-only batches take the DENSE dispatch: every expert computes
        # every token, unselected experts gated by weight 0.  Static shapes,
        # no host sync -> hipGraph-capturable; and decode MoE is bound by
        # streaming ALL expert weights regardless of routing (2.8 GB/layer
        # for 8x7B), so the extra rows are nearly free.  Prefill/mixed
        # batches (compute-bound) keep the capacity-routed path below.
        if meta is not None and meta.num_prefill_tokens == 0:
            wdense = torch.zeros(
                T, self.num_experts, dtype=x.dtype, device=x.device
            )
            wdense.scatter_(1, experts, weights)
            # EP: this rank computes only its owned expert slice; the block
            # all-reduce sums the partials
            wdense = wdense[:, self.expert_base:self.expert_base + self.local_experts]
            E = self.local_experts
            xe = x.unsqueeze(0).expand(E, T, h)
            act = ops.silu_and_mul(torch.bmm(xe, self.w13).view(E * T, -1))
            ye = torch.bmm(act.view(E, T, -1), self.w2)  # [E, T, h]
            out = (
                (ye.float() * wdense.t().unsqueeze(-1).float())
                .sum(dim=0)
                .to(x.dtype)
            )
            if self.tp is not None:
                out = self.tp.all_reduce(out)
            return out

        flat_experts = experts.reshape(-1)  # [T*k]
        flat_tokens = (
            torch.arange(T, device=x.device).unsqueeze(1).expand(T, self.top_k).reshape(-1)
        )
        flat_w = weights.reshape(-1)
        if self.ep:
            # keep only (token, expert) pairs owned by this rank; partial
            # outputs are summed by the block all-reduce
            mine = (flat_experts >= self.expert_base) & (
                flat_experts < self.expert_base + self.local_experts
            )
            flat_experts = flat_experts[mine] - self.expert_base
            flat_tokens = flat_tokens[mine]
            flat_w = flat_w[mine]
        # sort the (token, expert) pairs by expert -> contiguous per-expert runs
        order = torch.argsort(flat_experts, stable=True)
        sorted_experts = flat_experts[order]
        sorted_tokens = flat_tokens[order]
        sorted_w = flat_w[order]
        counts = torch.bincount(sorted_experts, minlength=self.local_experts)

        # Capacity-padded batched dispatch: ONE bmm pair for all experts
        # instead of E small GEMMs + Python loop (the loop was host-bound:
        # 8 experts x 3 ops x 32 layers of tiny launches).  cap = max tokens
        # on one expert this batch (one host sync per MoE block; MoE runs
        # eager, so there is no graph to break).  Padding rows index a zero
        # row appended to x and scatter back with weight 0 — bit-exact vs
        # the per-expert loop.
        cap = int(counts.max()) if counts.numel() else 0
        if cap == 0:
            out = x.new_zeros(T, h)
            # EP ranks with no owned tokens still join the combine
            return self.tp.all_reduce(out) if self.tp is not None else out
        starts = torch.cumsum(counts, 0) - counts
        E = self.local_experts
        dev = x.device
        # position of each sorted entry within its expert's run
        pos_in_e = (
            torch.arange(sorted_experts.numel(), device=dev)
            - starts.index_select(0, sorted_experts)
        )
        gather_idx = torch.full((E * cap,), T, dtype=torch.long, device=dev)
        dest = sorted_experts * cap + pos_in_e
        gather_idx.index_copy_(0, dest, sorted_tokens)
        x_pad = torch.cat([x, x.new_zeros(1, h)], dim=0)
        xe = x_pad.index_select(0, gather_idx).view(E, cap, h)
        act = ops.silu_and_mul(torch.bmm(xe, self.w13).view(E * cap, -1))
        ye = torch.bmm(act.view(E, cap, -1), self.w2).view(E * cap, h)
        w_pad = torch.zeros(E * cap, 1, dtype=x.dtype, device=dev)
        w_pad.index_copy_(0, dest, sorted_w.unsqueeze(1))
        out = torch.zeros_like(x)
        scatter_tok = gather_idx.clamp(max=T - 1)
        out.index_add_(0, scatter_tok, ye * w_pad)
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
