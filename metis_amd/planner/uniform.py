"""Uniform (Megatron-style) plan enumeration.

Same plan sequence as the reference's stateful odometer
(search_space/plan.py:40-97), written as a plain generator:

* strategies walk (pp, tp) with dp = N / tp / pp, keeping dp*pp*tp == N
  and tp <= max_tp;
* per strategy, gbs walks {dp} then divisors of max_gbs greater than dp;
* per gbs, mbs walks divisors of gbs with mbs * dp <= gbs.
"""

from __future__ import annotations

from typing import Iterator, List

from metis_amd.planner.plans import UniformPlan


def _divisors(n: int) -> List[int]:
    return [d for d in range(1, n + 1) if n % d == 0]


def uniform_plans(num_devices: int, max_tp: int, max_gbs: int) -> Iterator[UniformPlan]:
    for pp in range(1, num_devices + 1):
        for tp in range(1, max_tp + 1):
            if num_devices % (tp * pp) != 0:
                continue
            dp = num_devices // tp // pp
            # gbs = dp is always visited first (even when dp > max_gbs, in
            # which case only the mbs=1 plan is emitted) — reference parity.
            gbs_values = [dp] + [g for g in _divisors(max_gbs) if g > dp]
            for gbs in gbs_values:
                for mbs in _divisors(gbs):
                    if mbs * dp > gbs:
                        break
                    yield UniformPlan(dp=dp, pp=pp, tp=tp, mbs=mbs, gbs=gbs)
