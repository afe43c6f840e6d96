"""Plan dataclasses (reference parity: search_space/plan.py:12-37)."""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Tuple

from metis_amd.cluster import DeviceSpec


@dataclass(frozen=True)
class UniformPlan:
    """One Megatron-style uniform 3D plan."""

    dp: int
    pp: int
    tp: int
    mbs: int
    gbs: int

    @property
    def num_microbatches(self) -> int:
        return self.gbs // self.mbs // self.dp


@dataclass
class InterStagePlan:
    """Pipeline-level plan: which devices form which stage, in what order.

    ``device_groups[i]`` is the GPU count of pipeline stage i;
    ``node_sequence`` orders the device types along the pipeline;
    ``batches`` is the number of microbatches.
    """

    ns_idx: int
    node_sequence: List[DeviceSpec]
    dg_idx: int
    device_groups: List[int]
    num_stage: int
    batches: int
    gbs: int


@dataclass
class IntraStagePlan:
    """Per-stage parallelism: strategies[i] = (dp_deg, tp_deg) of stage i."""

    strategies: List[Tuple[int, int]]
    memory_state: List[float]
    layer_partition: List[int]   # cumulative boundaries [0, ..., num_layers]
    num_repartition: int
