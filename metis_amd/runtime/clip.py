"""TP/PP-correct global gradient-norm computation.

Under tensor/expert parallelism a rank holds two kinds of parameters:
shards whose elements exist on exactly one rank (column/row-parallel
weights, expert weights) and replicated copies (norms, embeddings,
routers, row-parallel biases). The global L2 norm sums the sharded
contributions ACROSS the tp group, counts replicated ones once, and
sums stage contributions across the pp group. (The reference has no
runtime; this is the clipping rule its planner's plans need to train.)
"""

from __future__ import annotations

from typing import Iterable, Tuple

import torch
import torch.distributed as dist


def shard_flags(model: torch.nn.Module) -> dict:
    """id(param) -> True when the param is a distinct per-rank shard."""
    from metis_amd.models.gpt import ColumnParallelLinear, RowParallelLinear

    flags: dict = {}
    for m in model.modules():
        if isinstance(m, ColumnParallelLinear):
            flags[id(m.weight)] = True
            flags[id(m.bias)] = True
        elif isinstance(m, RowParallelLinear):
            flags[id(m.weight)] = True
            flags[id(m.bias)] = False
        elif m.__class__.__name__ == "_Experts":
            for p in m.parameters(recurse=False):
                flags[id(p)] = True
    return flags


def global_grad_norm(
    grads: Iterable[Tuple[torch.Tensor, bool]],
    tp_group=None,
    pp_group=None,
) -> float:
    """L2 norm over (grad, is_sharded) pairs with the reduction rule
    above; every tp/pp rank must call this (collective)."""
    a = b = None
    for g, sharded in grads:
        s = g.float().square().sum()
        if a is None:
            a, b = s.new_zeros(1), s.new_zeros(1)
        if sharded:
            a += s
        else:
            b += s
    if tp_group is not None and dist.get_world_size(tp_group) > 1:
        dist.all_reduce(a, group=tp_group)
    total = a + b
    if pp_group is not None and dist.get_world_size(pp_group) > 1:
        dist.all_reduce(total, group=pp_group)
    return float(total.sqrt())
