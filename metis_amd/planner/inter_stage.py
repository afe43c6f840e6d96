"""Inter-stage plan enumeration.

Same plan sequence as the reference's stateful generator
(search_space/plan.py:100-175): node-type permutations x stage counts x
device-group assignments x microbatch counts (descending divisors of gbs).

Reference quirk Q12 is reproduced for parity (flagged below): when the
enumeration advances to the 2nd+ node sequence, the stage counter resets
to 1 while the device groups regenerate for the next stage count >= 2 —
so later node sequences skip their 1-stage groups and emit the first
multi-stage group list twice (once labelled num_stage=1). Single-type
clusters (one node sequence) are unaffected.
"""

from __future__ import annotations

from itertools import permutations
from typing import Iterator, List, Sequence

from metis_amd.cluster import DeviceSpec
from metis_amd.planner.groups import power_of_two_shapes, stage_device_groups
from metis_amd.planner.plans import InterStagePlan


def _descending_divisors(n: int) -> List[int]:
    return [d for d in range(n, 0, -1) if n % d == 0]


def inter_stage_plans(
    device_types: Sequence[DeviceSpec],
    num_devices: int,
    gbs: int,
    num_layers: int,
    variance: float,
    max_permute_len: int,
) -> Iterator[InterStagePlan]:
    node_sequences = list(permutations(device_types))
    shapes = power_of_two_shapes(num_devices)
    max_stage = min(num_devices, num_layers)
    batch_counts = _descending_divisors(gbs)

    def groups_for(num_stage: int) -> List[List[int]]:
        return stage_device_groups(num_stage, num_devices, shapes, variance, max_permute_len)

    def first_nonempty_from(n: int):
        """First stage count >= n with any device group (or n past max)."""
        dgs = groups_for(n)
        while not dgs and n <= max_stage:
            n += 1
            dgs = groups_for(n)
        return n, dgs

    for ns_idx, ns in enumerate(node_sequences):
        if ns_idx == 0:
            num_stage, dgs = 1, groups_for(1)
        else:
            # quirk Q12 (see module docstring): label stays 1, groups jump to >= 2
            _real_n, dgs = first_nonempty_from(2)
            num_stage = 1

        while num_stage <= max_stage:
            for dg_idx, dg in enumerate(dgs):
                for batches in batch_counts:
                    yield InterStagePlan(
                        ns_idx=ns_idx,
                        node_sequence=list(ns),
                        dg_idx=dg_idx,
                        device_groups=list(dg),
                        num_stage=num_stage,
                        batches=batches,
                        gbs=gbs,
                    )
            num_stage, dgs = first_nonempty_from(num_stage + 1)
