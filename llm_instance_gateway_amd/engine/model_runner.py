"""Model runner: owns the model, the physical paged KV cache, and one-step
execution (batch build → forward → sample).

MI355X notes: the KV cache is sized against the 288 GB HBM3E pool at engine
init (`determine_num_kv_blocks`); batches are flat/unpadded; all per-step
index tensors are built on host and copied once (a single H2D per tensor per
step).
"""

from __future__ import annotations

import logging
from typing import Dict, List, Optional, Tuple

import torch

from ..models.config import ModelConfig
from ..utils.tunableop import eager_tuned
from .config import EngineConfig
from .metadata import BatchMeta
from ..utils.tracing import gpu_range
from .sampling import sample_tokens
from .scheduler import ScheduledSeq, SchedulerOutput
from .sequence import Sequence

log = logging.getLogger(__name__)


def build_model(cfg: ModelConfig, device, dtype, tp=None):
    if cfg.is_moe:
        from ..models.mixtral import MixtralForCausalLM

        return MixtralForCausalLM(cfg, device=device, dtype=dtype, tp=tp)
    from ..models.llama import LlamaForCausalLM

    return LlamaForCausalLM(cfg, device=device, dtype=dtype, tp=tp)


def determine_num_kv_blocks(
    config: EngineConfig, device: torch.device, tp_size: int = 1
) -> int:
    """Size the paged KV pool from free HBM after weights, honoring
    gpu_memory_utilization (the reference fleet runs vLLM at 0.95 —
    tools/dynamic-lora-sidecar/deployment.yaml:40)."""
    mc = config.model
    bytes_per_block = (
        mc.num_layers
        * 2
        * (mc.num_kv_heads // tp_size)
        * config.block_size
        * mc.head_dim
        * 2  # bf16
    )
    if device.type == "cuda":
        free, total = torch.cuda.mem_get_info(device)
        budget = int(total * config.gpu_memory_utilization) - (total - free)
        budget = max(budget, 0)
        n = budget // bytes_per_block
        return int(n)
    return config.num_kv_blocks


class ModelRunner:
    def __init__(
        self,
        config: EngineConfig,
        device: str = "cpu",
        dtype: torch.dtype = torch.bfloat16,
        tp=None,
    ) -> None:
        self.config = config
        self.device = torch.device(device)
        self.dtype = dtype
        self.tp = tp
        mc = config.model
        log.info("building model %s on %s", mc.name, self.device)
        self.model = build_model(mc, self.device, dtype, tp)
        if config.quant == "fp8":
            if mc.is_moe:
                raise ValueError("quant='fp8' covers dense models only")
            from ..models.quant import convert_linears_to_fp8

            n_q = convert_linears_to_fp8(self.model)
            if self.device.type == "cuda":
                torch.cuda.empty_cache()  # bf16 projection weights are gone
            log.info("fp8 quantization: %d projection GEMMs converted", n_q)
        tp_size = tp.world_size if tp is not None else 1
        if self.device.type == "cuda":
            n = determine_num_kv_blocks(config, self.device, tp_size)
            if n <= 0:
                raise RuntimeError("no HBM left for KV cache after weights")
            config.num_kv_blocks = min(
                n, config.num_kv_blocks if config.num_kv_blocks > 0 else n
            )
        kvh = mc.num_kv_heads // tp_size
        # +1 scratch block: hipGraph decode padding rows write their K/V there
        shape = (config.num_kv_blocks + 1, kvh, config.block_size, mc.head_dim)
        self.kv_caches: List[Tuple[torch.Tensor, torch.Tensor]] = [
            (
                torch.zeros(shape, dtype=dtype, device=self.device),
                torch.zeros(shape, dtype=dtype, device=self.device),
            )
            for _ in range(mc.num_layers)
        ]
        log.info(
            "KV cache: %d blocks x %d tokens (%.1f GiB)",
            config.num_kv_blocks,
            config.block_size,
            2 * mc.num_layers * shape[0] * kvh * config.block_size * mc.head_dim * 2 / 2**30,
        )
        self.adapter_manager = None
        self.graph_runner = None
        # MoE decode is graph-safe too: decode-only batches use the dense
        # expert dispatch (static shapes, no host sync — models/mixtral.py).
        # TP engines decode EAGER by default: capturing the per-block RCCL
        # all-reduce inside a hipGraph is exactly the stream-semantics
        # surface that has never run on multi-GPU hardware (docs/scaling.md)
        # — conservative by construction for the first TP run; LIG_TP_GRAPHS=1
        # opts in once validated.
        import os as _os

        self._use_graphs = (
            self.device.type == "cuda"
            and not config.enforce_eager
            and (tp is None or _os.environ.get("LIG_TP_GRAPHS", "0") == "1")
        )
        # Mixed-step prefill/decode stream overlap — MEASURED DEAD END as a
        # default: 51.4 vs 56.5 req/s at the serving shape (gpurun
        # bench_mixed vs bench_mixed_noov) — the graph replay and the eager
        # prefill contend for the same CUs, so splitting the step into two
        # streams only added sync overhead; the single flat mixed batch
        # already keeps the chip full.  Kept opt-in (LIG_OVERLAP=1) for
        # shapes where decode is small vs prefill.
        self._overlap = _os.environ.get("LIG_OVERLAP", "0") == "1"
        self._decode_stream = None

    # -- batch construction ---------------------------------------------------
    def _build_batch(self, out):
        scheduled = out.scheduled if isinstance(out, SchedulerOutput) else out
        prefills: List[ScheduledSeq] = []
        decodes: List[ScheduledSeq] = []
        for ss in scheduled:
            (decodes if ss.num_tokens == 1 else prefills).append(ss)

        bs = self.config.block_size
        input_ids: List[int] = []
        positions: List[int] = []
        slots: List[int] = []
        sample_seqs: List[Sequence] = []
        sample_idx: List[int] = []

        qsl = [0]
        p_ctx: List[int] = []
        p_bts: List[List[int]] = []
        for ss in prefills:
            seq = ss.seq
            c, n = seq.num_computed_tokens, ss.num_tokens
            toks = seq.all_token_ids[c : c + n]
            input_ids.extend(toks)
            positions.extend(range(c, c + n))
            for pos in range(c, c + n):
                slots.append(seq.block_table[pos // bs] * bs + pos % bs)
            qsl.append(qsl[-1] + n)
            p_ctx.append(c + n)
            p_bts.append(seq.block_table)
            if c + n == seq.total_len:
                sample_seqs.append(seq)
                sample_idx.append(qsl[-1] - 1)

        num_prefill_tokens = len(input_ids)
        d_ctx: List[int] = []
        d_bts: List[List[int]] = []
        for ss in decodes:
            seq = ss.seq
            pos = seq.num_computed_tokens  # == total_len - 1
            input_ids.append(seq.all_token_ids[pos])
            positions.append(pos)
            slots.append(seq.block_table[pos // bs] * bs + pos % bs)
            d_ctx.append(pos + 1)
            d_bts.append(seq.block_table)
            sample_seqs.append(seq)
            sample_idx.append(len(input_ids) - 1)

        # GEMM M-bucketing: pad the flat batch tail so every projection GEMM
        # sees M % pad_batch_multiple == 0 (few stable shapes -> hipBLASLt
        # algo reuse + tile-friendly M).  Pad rows: token 0 at position 0,
        # K/V scattered to the scratch block (same slot — racy garbage is
        # fine there), outside both attention partitions and the sample set.
        num_pad_tokens = 0
        padm = self.config.pad_batch_multiple
        if (
            padm > 0
            and self.device.type == "cuda"
            and not self.config.model.is_moe
            and len(input_ids) % padm != 0
        ):
            num_pad_tokens = padm - len(input_ids) % padm
            scratch_slot = self.config.num_kv_blocks * bs
            input_ids.extend([0] * num_pad_tokens)
            positions.extend([0] * num_pad_tokens)
            slots.extend([scratch_slot] * num_pad_tokens)

        dev = self.device

        def pad_tables(bts: List[List[int]]) -> Optional[torch.Tensor]:
            if not bts:
                return None
            m = max(len(b) for b in bts)
            t = torch.zeros(len(bts), m, dtype=torch.int32)
            for i, b in enumerate(bts):
                t[i, : len(b)] = torch.tensor(b, dtype=torch.int32)
            return t.to(dev, non_blocking=True)

        meta = BatchMeta(
            num_prefill_tokens=num_prefill_tokens,
            num_decode_tokens=len(decodes),
            slot_mapping=torch.tensor(slots, dtype=torch.int64).to(dev, non_blocking=True),
            prefill_query_start_locs=(
                torch.tensor(qsl, dtype=torch.int32).to(dev, non_blocking=True)
                if prefills
                else None
            ),
            prefill_context_lens=(
                torch.tensor(p_ctx, dtype=torch.int32).to(dev, non_blocking=True)
                if prefills
                else None
            ),
            prefill_block_tables=pad_tables(p_bts),
            decode_block_tables=pad_tables(d_bts),
            decode_context_lens=(
                torch.tensor(d_ctx, dtype=torch.int32).to(dev, non_blocking=True)
                if decodes
                else None
            ),
            max_decode_ctx=max(d_ctx) if d_ctx else 0,
        )
        # Per-token prefill expansion, once per step (the attention layers of
        # every transformer block reuse it; see profiles/r01_bench_kernel_stats.md).
        if prefills:
            use_mfma = self.device.type == "cuda" and self.config.model.head_dim in (64, 128)
            if use_mfma:
                # 64-row query tiles for the MFMA flash-prefill kernel.
                ts: List[int] = []
                tq: List[int] = []
                for si, ss in enumerate(prefills):
                    for qb in range(0, ss.num_tokens, 64):
                        ts.append(si)
                        tq.append(qb)
                meta.prefill_tile_seq = torch.tensor(ts, dtype=torch.int32).to(
                    dev, non_blocking=True
                )
                meta.prefill_tile_qstart = torch.tensor(tq, dtype=torch.int32).to(
                    dev, non_blocking=True
                )
            else:
                ctx_tok: List[int] = []
                n_per: List[int] = []
                for ss in prefills:
                    c, n = ss.seq.num_computed_tokens, ss.num_tokens
                    ctx_tok.extend(range(c + 1, c + n + 1))
                    n_per.append(n)
                meta.prefill_tok_context_lens = torch.tensor(
                    ctx_tok, dtype=torch.int32
                ).to(dev, non_blocking=True)
                meta.prefill_tok_block_tables = torch.repeat_interleave(
                    meta.prefill_block_tables,
                    torch.tensor(n_per, dtype=torch.int64, device=dev),
                    dim=0,
                ).contiguous()

        # multi-LoRA: per-token adapter slot ids (-1 = base model only)
        mgr = getattr(self, "adapter_manager", None)
        if mgr is not None and any(ss.seq.lora_name for ss in prefills + decodes):
            slots: List[int] = []
            # prefill tokens are sequence-contiguous: emit 64-token segment
            # tiles per same-slot run for the MFMA SGMV path
            t0s: List[int] = []
            tlens: List[int] = []
            tslots: List[int] = []
            for ss in prefills:
                s_id = mgr.slot_of(ss.seq.lora_name)
                base = len(slots)
                slots.extend([s_id] * ss.num_tokens)
                if s_id >= 0:
                    for off in range(0, ss.num_tokens, 64):
                        t0s.append(base + off)
                        tlens.append(min(64, ss.num_tokens - off))
                        tslots.append(s_id)
            for ss in decodes:
                slots.append(mgr.slot_of(ss.seq.lora_name))
            if num_pad_tokens:
                slots.extend([-1] * num_pad_tokens)  # pad rows: base model only
            meta.lora_slot_per_token = torch.tensor(slots, dtype=torch.int32).to(
                dev, non_blocking=True
            )
            if self.device.type == "cuda":
                # slot-grouped SGMV units (lora/manager.py::build_lora_units)
                from ..lora.manager import build_lora_units

                order, ut0, ulen, uslot = build_lora_units(slots)
                if len(ut0):
                    meta.lora_order = torch.from_numpy(order).to(dev, non_blocking=True)
                    meta.lora_unit_t0 = torch.from_numpy(ut0).to(dev, non_blocking=True)
                    meta.lora_unit_len = torch.from_numpy(ulen).to(dev, non_blocking=True)
                    meta.lora_unit_slot = torch.from_numpy(uslot).to(dev, non_blocking=True)
            if t0s:
                meta.lora_tile_t0 = torch.tensor(t0s, dtype=torch.int32).to(
                    dev, non_blocking=True
                )
                meta.lora_tile_len = torch.tensor(tlens, dtype=torch.int32).to(
                    dev, non_blocking=True
                )
                meta.lora_tile_slot = torch.tensor(tslots, dtype=torch.int32).to(
                    dev, non_blocking=True
                )

        ids_t = torch.tensor(input_ids, dtype=torch.int64).to(dev, non_blocking=True)
        pos_t = torch.tensor(positions, dtype=torch.int64).to(dev, non_blocking=True)
        return ids_t, pos_t, meta, sample_seqs, sample_idx, prefills + decodes

    # -- one step --------------------------------------------------------------
    @torch.no_grad()
    def execute(self, out: SchedulerOutput) -> Dict[int, int]:
        """Run one mixed step; returns {seq_id: sampled_token} and advances
        num_computed_tokens for every scheduled sequence."""
        if out.is_empty:
            return {}
        if self._use_graphs and all(ss.num_tokens == 1 for ss in out.scheduled):
            with gpu_range("graph_replay"):
                res = self._execute_graph(out)
            if res is not None:
                return res
        if self._use_graphs and self._overlap:
            res = self._execute_overlapped(out)
            if res is not None:
                return res
        with gpu_range("build_batch"):
            ids, pos, meta, sample_seqs, sample_idx, ordered = self._build_batch(out)
        with gpu_range("forward"), eager_tuned():
            hidden = self.model(ids, pos, self.kv_caches, meta)
        sampled: Dict[int, int] = {}
        if sample_seqs:
            with gpu_range("sample"):
                # Bucket the logits GEMM's M (ROADMAP r02 #4): sample counts
                # vary 1..max_num_seqs per mixed step — pad the gather index
                # to a multiple of 64 (repeat row 0) so the 128k-vocab GEMM
                # sees a handful of stable shapes; pad rows sliced off.
                M = len(sample_idx)
                pad_m = 0
                if self.device.type == "cuda" and self.config.pad_batch_multiple:
                    pad_m = (-M) % 64
                idx = torch.tensor(
                    sample_idx + [sample_idx[0]] * pad_m,
                    dtype=torch.int64,
                    device=self.device,
                )
                logits = self.model.compute_logits(hidden, idx)
                if pad_m:
                    logits = logits[:M]
                tokens = sample_tokens(logits, sample_seqs)
            for seq, tok in zip(sample_seqs, tokens):
                sampled[seq.seq_id] = tok
        for ss in ordered:
            ss.seq.num_computed_tokens += ss.num_tokens
        return sampled

    def _execute_overlapped(self, out: SchedulerOutput) -> Optional[Dict[int, int]]:
        """Mixed-step stream overlap (ROADMAP r01 #9): the decode partition
        replays its hipGraph on a side stream while the prefill chunk runs
        eager on the main stream.  Safe by construction: KV slots are
        disjoint, adapter slot weights are read-only during a step, and the
        graph owns a private LoRA workspace (metadata.lora_ws)."""
        prefills = [ss for ss in out.scheduled if ss.num_tokens > 1]
        decodes = [ss for ss in out.scheduled if ss.num_tokens == 1]
        if not prefills or not decodes:
            return None
        gr = self._ensure_graph_runner()
        if gr is None or gr.pick_size(len(decodes)) is None:
            return None
        if self._decode_stream is None:
            self._decode_stream = torch.cuda.Stream(device=self.device)
        sd = self._decode_stream
        main = torch.cuda.current_stream(self.device)
        slot_of = (
            self.adapter_manager.slot_of if self.adapter_manager is not None else None
        )
        with gpu_range("overlap_step"):
            sd.wait_stream(main)  # adapter load fences etc. ride main
            with torch.cuda.stream(sd):
                logits_d = gr.run(decodes, slot_of)
            ids, pos, meta, sample_seqs_p, sample_idx_p, _ = self._build_batch(
                prefills
            )
            with eager_tuned():
                hidden = self.model(ids, pos, self.kv_caches, meta)
            logits_p = None
            if sample_seqs_p:
                idx = torch.tensor(
                    sample_idx_p, dtype=torch.int64, device=self.device
                )
                logits_p = self.model.compute_logits(hidden, idx)
            main.wait_stream(sd)
            sampled: Dict[int, int] = {}
            if logits_p is not None:
                for seq, tok in zip(
                    sample_seqs_p, sample_tokens(logits_p, sample_seqs_p)
                ):
                    sampled[seq.seq_id] = tok
            d_seqs = [ss.seq for ss in decodes]
            for seq, tok in zip(d_seqs, sample_tokens(logits_d.float(), d_seqs)):
                sampled[seq.seq_id] = tok
        for ss in out.scheduled:
            ss.seq.num_computed_tokens += ss.num_tokens
        return sampled

    def _ensure_graph_runner(self):
        from .graph_runner import DecodeGraphRunner

        want_lora = self.adapter_manager is not None
        if self.graph_runner is None or self.graph_runner.with_lora != want_lora:
            self.graph_runner = DecodeGraphRunner(self)
            self.graph_runner.capture(with_lora=want_lora)
        return self.graph_runner

    def _execute_graph(self, out: SchedulerOutput) -> Optional[Dict[int, int]]:
        """Decode-only fast path: one hipGraph replay for the whole step."""
        decodes = out.scheduled
        gr = self._ensure_graph_runner()
        if gr.pick_size(len(decodes)) is None:
            return None  # larger than any captured bucket: eager fallback
        slot_of = (
            self.adapter_manager.slot_of if self.adapter_manager is not None else None
        )
        logits = gr.run(decodes, slot_of)
        seqs = [ss.seq for ss in decodes]
        tokens = sample_tokens(logits.float(), seqs)
        for ss in decodes:
            ss.seq.num_computed_tokens += 1
        return {seq.seq_id: tok for seq, tok in zip(seqs, tokens)}
