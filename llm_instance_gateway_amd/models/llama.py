"""Llama-family dense transformer with paged-KV attention.

The model family the reference's fleet configs pin
(`examples/poc/manifests/vllm/vllm-lora-deployment.yaml`: Llama-2-7b;
BASELINE.json: Llama-3-8B / 70B).  Built MI355X-first:

  * flat unpadded token batches (prefill chunks + decodes in one forward);
  * custom CDNA4 kernels via the ops layer for RMSNorm (fused residual add),
    RoPE, KV-cache scatter, paged prefill/decode attention and SwiGLU;
  * plain library GEMMs (rocBLAS/hipBLASLt via torch.matmul) for the
    projections; LoRA deltas ride the SGMV path (`..lora`);
  * optional tensor parallelism: column-sharded QKV/gate-up, row-sharded
    O/down with one RCCL all-reduce per block over xGMI (`..parallel`).

Weights are random-init (no network — synthetic serving benchmark).
"""

from __future__ import annotations

import math
from typing import List

import torch
import torch.nn as nn

from .. import ops
from ..engine.metadata import BatchMeta
from .config import ModelConfig


def _linear(in_f: int, out_f: int, device, dtype) -> nn.Linear:
    lin = nn.Linear(in_f, out_f, bias=False, device="meta", dtype=dtype)
    lin = lin.to_empty(device=device)
    # cheap bounded init (never materialize fp32 copies of 8B params)
    with torch.no_grad():
        lin.weight.uniform_(-0.02, 0.02)
    return lin


class Attention(nn.Module):
    def __init__(self, cfg: ModelConfig, device, dtype, tp=None) -> None:
        super().__init__()
        self.cfg = cfg
        self.tp = tp
        tp_size = tp.world_size if tp is not None else 1
        assert cfg.num_heads % tp_size == 0 and cfg.num_kv_heads % tp_size == 0
        self.num_heads = cfg.num_heads // tp_size
        self.num_kv_heads = cfg.num_kv_heads // tp_size
        self.head_dim = cfg.head_dim
        self.scale = 1.0 / math.sqrt(cfg.head_dim)
        h = cfg.hidden_size
        q_out = self.num_heads * self.head_dim
        kv_out = self.num_kv_heads * self.head_dim
        self.qkv_proj = _linear(h, q_out + 2 * kv_out, device, dtype)
        self.o_proj = _linear(q_out, h, device, dtype)
        self.q_size = q_out
        self.kv_size = kv_out
        # multi-LoRA hook (set by lora.manager.AdapterManager._attach)
        self._lora = None
        self._layer_idx = -1

    def forward(
        self,
        hidden: torch.Tensor,  # [T, H]
        positions: torch.Tensor,  # [T]
        kv_cache,  # (k_cache, v_cache) for this layer
        meta: BatchMeta,
        cos_sin_cache: torch.Tensor,
    ) -> torch.Tensor:
        qkv = self.qkv_proj(hidden)
        if self._lora is not None:
            self._lora.apply(self._layer_idx, "qkv", hidden, qkv, meta)
        q, k, v = qkv.split([self.q_size, self.kv_size, self.kv_size], dim=-1)
        q = q.contiguous()
        k = k.contiguous()
        q, k = ops.rotary_embedding(positions, q, k, cos_sin_cache, self.head_dim)

        k_cache, v_cache = kv_cache
        ops.reshape_and_cache(k, v.contiguous(), k_cache, v_cache, meta.slot_mapping)

        qh = q.view(-1, self.num_heads, self.head_dim)
        np_, nd = meta.num_prefill_tokens, meta.num_decode_tokens
        out = torch.empty_like(qh)
        if np_ > 0:
            out[:np_] = ops.paged_attention_prefill(
                qh[:np_],
                k_cache,
                v_cache,
                meta.prefill_block_tables,
                meta.prefill_query_start_locs,
                meta.prefill_context_lens,
                self.scale,
                tok_block_tables=meta.prefill_tok_block_tables,
                tok_context_lens=meta.prefill_tok_context_lens,
                tile_seq=meta.prefill_tile_seq,
                tile_qstart=meta.prefill_tile_qstart,
            )
        if nd > 0:
            # slice end is explicit: the flat batch may carry GEMM-bucketing
            # pad rows after the decode partition
            out[np_:np_ + nd] = ops.paged_attention_decode(
                qh[np_:np_ + nd],
                k_cache,
                v_cache,
                meta.decode_block_tables,
                meta.decode_context_lens,
                self.scale,
                max_ctx=meta.max_decode_ctx,
                nsplits=meta.decode_nsplits,
            )
        attn_flat = out.view(-1, self.num_heads * self.head_dim)
        o = self.o_proj(attn_flat)
        if self._lora is not None:
            self._lora.apply(self._layer_idx, "o", attn_flat, o, meta)
        if self.tp is not None:
            o = self.tp.all_reduce(o)
        return o


class MLP(nn.Module):
    def __init__(self, cfg: ModelConfig, device, dtype, tp=None) -> None:
        super().__init__()
        self.tp = tp
        tp_size = tp.world_size if tp is not None else 1
        assert cfg.intermediate_size % tp_size == 0
        inter = cfg.intermediate_size // tp_size
        h = cfg.hidden_size
        # gate and up fused into one GEMM (silu_and_mul splits them)
        self.gate_up_proj = _linear(h, 2 * inter, device, dtype)
        self.down_proj = _linear(inter, h, device, dtype)
        self._lora = None
        self._layer_idx = -1

    def forward(self, x: torch.Tensor, meta=None) -> torch.Tensor:
        gu = self.gate_up_proj(x)
        if self._lora is not None:
            self._lora.apply(self._layer_idx, "gate_up", x, gu, meta)
        act = ops.silu_and_mul(gu)
        y = self.down_proj(act)
        if self._lora is not None:
            self._lora.apply(self._layer_idx, "down", act, y, meta)
        if self.tp is not None:
            y = self.tp.all_reduce(y)
        return y


class DecoderLayer(nn.Module):
    def __init__(self, cfg: ModelConfig, device, dtype, tp=None) -> None:
        super().__init__()
        self.input_layernorm = nn.Parameter(torch.ones(cfg.hidden_size, device=device, dtype=dtype))
        self.post_attention_layernorm = nn.Parameter(
            torch.ones(cfg.hidden_size, device=device, dtype=dtype)
        )
        self.eps = cfg.rms_norm_eps
        self.self_attn = Attention(cfg, device, dtype, tp)
        self.mlp = MLP(cfg, device, dtype, tp)

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
        h = self.mlp(h, meta)
        return h, residual


class LlamaForCausalLM(nn.Module):
    def __init__(self, cfg: ModelConfig, device="cpu", dtype=torch.bfloat16, tp=None) -> None:
        super().__init__()
        self.cfg = cfg
        self.tp = tp
        self.device_ = torch.device(device)
        self.dtype_ = dtype
        self.embed_tokens = nn.Embedding(
            cfg.vocab_size, cfg.hidden_size, device="meta", dtype=dtype
        ).to_empty(device=device)
        with torch.no_grad():
            self.embed_tokens.weight.uniform_(-0.02, 0.02)
        self.layers = nn.ModuleList(
            [DecoderLayer(cfg, device, dtype, tp) for _ in range(cfg.num_layers)]
        )
        self.norm = nn.Parameter(torch.ones(cfg.hidden_size, device=device, dtype=dtype))
        if cfg.tie_word_embeddings:
            self.lm_head = None
        else:
            self.lm_head = _linear(cfg.hidden_size, cfg.vocab_size, device, dtype)
        self.cos_sin_cache = ops.build_cos_sin_cache(
            cfg.head_dim, cfg.max_position_embeddings, cfg.rope_theta, device
        ).to(dtype=torch.float32)

    def forward(
        self,
        input_ids: torch.Tensor,  # [T]
        positions: torch.Tensor,  # [T]
        kv_caches: List,  # per layer (k_cache, v_cache)
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
        """Logits only at the sampled positions (last token of each finished
        chunk / each decode)."""
        h = hidden[indices]
        w = self.embed_tokens.weight if self.lm_head is None else self.lm_head.weight
        # hidden is replicated across the TP group (each block ends in an
        # all-reduce), so every rank computes identical full-vocab logits.
        logits = h @ w.t()
        return logits.float()
