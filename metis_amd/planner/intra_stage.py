"""Intra-stage (dp, tp) strategy search under memory pressure.

Parity: reference search_space/plan.py:178-268, written as a generator
instead of a side-effecting ``has_next`` property.

Each stage starts at (dp = group_size, tp = 1); when the layer balancer
reports memory pressure, the stage with the least memory headroom
escalates (dp, tp) -> (dp/2, tp*2). Enumeration stops after a plan whose
layer partition fit on the first attempt (num_repartition == 1).
"""

from __future__ import annotations

import logging
from typing import Iterator, List, Optional, Sequence, Tuple

from metis_amd.planner.balancer import LayerLoadBalancer, StagePerformance
from metis_amd.planner.plans import InterStagePlan, IntraStagePlan

log = logging.getLogger(__name__)


def _next_strategy(
    strategies: List[Tuple[int, int]], memory_state: Optional[Sequence[float]]
) -> Optional[List[Tuple[int, int]]]:
    """Escalate the stage with least memory headroom; None when every stage
    is already at dp=1 (plan.py:251-268)."""
    if not memory_state:
        memory_state = [1.0 / dp for (dp, _tp) in strategies]
    order = sorted(range(len(memory_state)), key=lambda i: memory_state[i])
    out = list(strategies)
    for stage_id in order:
        dp, tp = out[stage_id]
        if dp != 1:
            out[stage_id] = (dp // 2, tp * 2)
            return out
    return None


def _strategies_valid(
    strategies: Sequence[Tuple[int, int]],
    gbs: int,
    batches: int,
    max_tp: int,
    max_bs: int,
) -> bool:
    for dp, tp in strategies:
        mbs = gbs // dp // batches
        if mbs == 0 or mbs > max_bs:
            return False
        if tp > max_tp:
            return False
    return True


def intra_stage_plans(
    plan: InterStagePlan,
    stage_performance: StagePerformance,
    layer_load_balancer: LayerLoadBalancer,
    max_tp: int,
    max_bs: int,
) -> Iterator[IntraStagePlan]:
    strategies: Optional[List[Tuple[int, int]]] = None
    memory_state: Optional[List[float]] = None

    while True:
        if strategies is None:
            strategies = [(group_size, 1) for group_size in plan.device_groups]
        else:
            strategies = _next_strategy(strategies, memory_state)
            if strategies is None:
                return

        if not _strategies_valid(strategies, plan.gbs, plan.batches, max_tp, max_bs):
            continue

        capacity = stage_performance.memory_capacity()
        compute = stage_performance.compute_performance(strategies, plan.gbs, plan.batches)
        layer_partition, num_repartition, mstate = layer_load_balancer.partition_layer(
            plan, strategies, compute, capacity
        )
        memory_state = mstate
        if layer_partition:
            yield IntraStagePlan(
                strategies=list(strategies),
                memory_state=list(mstate) if mstate else [],
                layer_partition=layer_partition,
                num_repartition=num_repartition,
            )
            if num_repartition == 1:
                return
