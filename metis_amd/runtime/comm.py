"""Process-group setup for (dp, tp, pp) plans.

Rank layout matches the planner's topology model
(metis_amd.planner.bandwidth.HomoTopology): rank = (p*dp + d)*tp + t —
TP innermost (consecutive ranks share a node's xGMI links, where the
all-to-all traffic of TP lives), DP next, PP outermost.

Backend: "nccl" IS RCCL on ROCm; CPU tests use "gloo".
"""

from __future__ import annotations

import datetime
import os
from dataclasses import dataclass
from typing import Optional

import torch
import torch.distributed as dist


@dataclass
class ParallelContext:
    rank: int
    world_size: int
    local_rank: int
    dp: int
    tp: int
    pp: int
    tp_group: Optional[object] = None
    dp_group: Optional[object] = None
    pp_group: Optional[object] = None
    device: Optional[torch.device] = None

    # coordinates
    @property
    def tp_rank(self) -> int:
        return self.rank % self.tp

    @property
    def dp_rank(self) -> int:
        return (self.rank // self.tp) % self.dp

    @property
    def pp_rank(self) -> int:
        return self.rank // (self.tp * self.dp)

    @property
    def is_first_stage(self) -> bool:
        return self.pp_rank == 0

    @property
    def is_last_stage(self) -> bool:
        return self.pp_rank == self.pp - 1

    def stage_neighbor(self, direction: int) -> int:
        """Global rank of the same (d, t) in stage pp_rank + direction."""
        p = self.pp_rank + direction
        return (p * self.dp + self.dp_rank) * self.tp + self.tp_rank


def init_parallel(
    dp: int, tp: int, pp: int, backend: Optional[str] = None
) -> ParallelContext:
    """Initialize torch.distributed (reading RANK/WORLD_SIZE/LOCAL_RANK from
    the env, torchrun-style) and carve tp/dp/pp subgroups."""
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    assert dp * tp * pp == world, (
        f"plan dp{dp} x tp{tp} x pp{pp} != world size {world}"
    )

    use_gpu = torch.cuda.is_available()
    if backend is None:
        backend = "nccl" if use_gpu else "gloo"
    device = None
    if use_gpu:
        device = torch.device("cuda", local_rank)
        torch.cuda.set_device(device)

    if world > 1 and not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29531")
        # failure detection: a hung collective (dead peer, deadlock)
        # surfaces as a timeout error instead of an infinite hang;
        # METIS_DIST_TIMEOUT_S tunes it (300 s default)
        timeout_s = int(os.environ.get("METIS_DIST_TIMEOUT_S", "300"))
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world,
            timeout=datetime.timedelta(seconds=timeout_s),
        )

    ctx = ParallelContext(rank=rank, world_size=world, local_rank=local_rank,
                          dp=dp, tp=tp, pp=pp, device=device)

    if world > 1:
        # Build every subgroup on every rank (new_group is collective).
        for p in range(pp):
            for d in range(dp):
                ranks = [(p * dp + d) * tp + t for t in range(tp)]
                g = dist.new_group(ranks) if tp > 1 else None
                if rank in ranks:
                    ctx.tp_group = g
        for p in range(pp):
            for t in range(tp):
                ranks = [(p * dp + d) * tp + t for d in range(dp)]
                g = dist.new_group(ranks) if dp > 1 else None
                if rank in ranks:
                    ctx.dp_group = g
        for d in range(dp):
            for t in range(tp):
                ranks = [(p * dp + d) * tp + t for p in range(pp)]
                g = dist.new_group(ranks) if pp > 1 else None
                if rank in ranks:
                    ctx.pp_group = g
    return ctx


def barrier_all() -> None:
    if dist.is_initialized():
        dist.barrier()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
