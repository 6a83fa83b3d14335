"""Tensor parallelism over RCCL / xGMI.

The reference delegates TP to vLLM's NCCL path behind the
``--tensor-parallel-size`` flag (`examples/poc/manifests/vllm/
vllm-lora-deployment.yaml:39-40`); here it is first-class: column-sharded
QKV / gate-up, row-sharded O / down projections with ONE all-reduce per
sub-block (two per transformer layer), running over RCCL
(`torch.distributed` backend "nccl" IS RCCL on ROCm) on the xGMI mesh.

xGMI topology note: each MI355X has 7 point-to-point links
at ≈153 GB/s — there is no switch, so a ring all-reduce is bound by ONE
link.  For the small TP groups used here (TP=2/4/8) RCCL's one-shot /
direct-allreduce algorithms use all links concurrently; we keep messages
per-layer (hidden × tokens × 2 B, no bucketing delay) because decode-step
latency — not bandwidth — is the binding constraint at batch ≤ 256.

Process model: one process per GPU (the engine pool spawns `tp_size`
workers per TP engine; rank 0 fronts the request stream and broadcasts
per-step commands — `serving/pool.py`).  Groups are torch.distributed
subgroups so several TP engines (and DP singles) can coexist in one world,
which is exactly the reference's "mixed InferencePool" config
(BASELINE.json config #4: 8B TP=1 engines + 70B TP=4 engine in one pool).
"""

from __future__ import annotations

import datetime
import os
from typing import List, Optional, Sequence

import torch
import torch.distributed as dist


class TPGroup:
    """A tensor-parallel communicator wrapping one torch.distributed group.

    The model layer calls `all_reduce` after row-sharded projections
    (models/llama.py:111-112,138-139); everything else is setup.
    """

    def __init__(self, group=None, device: Optional[torch.device] = None):
        assert dist.is_initialized(), "init_distributed() first"
        self.group = group
        self.world_size = dist.get_world_size(group)
        self.rank = dist.get_rank(group)
        self.device = device

    def all_reduce(self, x: torch.Tensor) -> torch.Tensor:
        """Sum across the TP group, in place.  Hidden-sized message per call;
        on ROCm this is an RCCL all-reduce over xGMI."""
        if self.world_size == 1:
            return x
        dist.all_reduce(x, op=dist.ReduceOp.SUM, group=self.group)
        return x

    def all_gather(self, x: torch.Tensor, dim: int = -1) -> torch.Tensor:
        """Gather shards along ``dim`` (vocab-parallel logits path)."""
        if self.world_size == 1:
            return x
        parts = [torch.empty_like(x) for _ in range(self.world_size)]
        dist.all_gather(parts, x.contiguous(), group=self.group)
        return torch.cat(parts, dim=dim)

    def broadcast_obj(self, obj, src_rank_in_group: int = 0):
        """Broadcast a picklable object from the group's src rank (used by the
        TP engine front rank to fan out per-step commands)."""
        if self.world_size == 1:
            return obj
        ranks = dist.get_process_group_ranks(self.group) if self.group else list(
            range(dist.get_world_size())
        )
        box = [obj]
        dist.broadcast_object_list(box, src=ranks[src_rank_in_group], group=self.group)
        return box[0]

    def barrier(self) -> None:
        if self.world_size > 1:
            dist.barrier(group=self.group)


def init_distributed(
    backend: Optional[str] = None,
    rank: Optional[int] = None,
    world_size: Optional[int] = None,
    master_addr: str = "127.0.0.1",
    master_port: Optional[int] = None,
    timeout_s: float = 600.0,
) -> None:
    """Initialise the world process group (idempotent).

    Backend defaults to RCCL ("nccl") when a GPU is visible, else gloo —
    the CPU tests run the identical collective call pattern over gloo.
    """
    if dist.is_initialized():
        return
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    env = os.environ
    rank = rank if rank is not None else int(env.get("RANK", "0"))
    world_size = world_size if world_size is not None else int(env.get("WORLD_SIZE", "1"))
    env.setdefault("MASTER_ADDR", master_addr)
    if master_port is not None:
        env["MASTER_PORT"] = str(master_port)
    env.setdefault("MASTER_PORT", "29511")
    dist.init_process_group(
        backend=backend,
        rank=rank,
        world_size=world_size,
        timeout=datetime.timedelta(seconds=timeout_s),
    )


def make_tp_group(ranks: Sequence[int], device: Optional[torch.device] = None) -> Optional[TPGroup]:
    """Create a TP subgroup over ``ranks``.  Must be called by EVERY rank in
    the world with the same arguments (torch.distributed new_group contract);
    returns a TPGroup on member ranks, None elsewhere."""
    ranks = list(ranks)
    if len(ranks) == dist.get_world_size():
        grp = None  # WORLD
    else:
        grp = dist.new_group(ranks=ranks)
    if dist.get_rank() in ranks:
        return TPGroup(group=grp, device=device)
    return None


def shard_sizes(total: int, world_size: int) -> List[int]:
    assert total % world_size == 0, f"{total} not divisible by TP={world_size}"
    return [total // world_size] * world_size
