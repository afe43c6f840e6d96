"""Layer->stage and data->replica load balancing.

Behavior parity with reference model/load_balancer.py (cited per method).
The heuristics' magic constants are kept and named:

* ``MEM_COEF = 5.0``      — profiled-MB -> demand multiplier (:31)
* ``SLICES_PER_LAYER = 7``— "hallucination" sub-slice factor (:183)
* ``MAX_PARTITION_ATTEMPTS = 3`` (:122), ``MAX_REFINE_STEPS = 3`` (:353)
"""

from __future__ import annotations

import copy
from typing import Dict, List, Optional, Sequence, Tuple

from metis_amd.cluster import ClusterSpec, DeviceSpec
from metis_amd.config import ModelConfig
from metis_amd.planner.plans import InterStagePlan
from metis_amd.profiles import ProfileStore

MEM_COEF = 5.0
SLICES_PER_LAYER = 7
MAX_PARTITION_ATTEMPTS = 3
MAX_REFINE_STEPS = 3


def pow2_slices(bs: int) -> List[int]:
    """Decompose a batch size into its binary powers, descending
    (e.g. 5 -> [4, 1]); parity: cost_estimator.py:159-161."""
    if bs <= 0:
        return []
    return [1 << i for i in range(bs.bit_length() - 1, -1, -1) if bs & (1 << i)]


class DataLoadBalancer:
    """Split a stage's per-iteration batch across heterogeneous DP replicas
    proportional to 1/t(tp, bs=1), largest-remainder rounding.
    Parity: load_balancer.py:147-179."""

    def __init__(self, profiles: ProfileStore) -> None:
        self.profiles = profiles

    def partition_data(
        self, device_types: Sequence[str], strategy: Tuple[int, int], bs: int
    ) -> List[int]:
        dp_deg, tp_deg = strategy
        group_size = len(device_types) // dp_deg

        perf = []
        for i in range(dp_deg):
            dtype = device_types[i * group_size]
            total_time = sum(self.profiles.get(dtype, tp_deg, 1).layer_times_ms)
            perf.append(1.0 / total_time)

        total = sum(perf)
        shares = [p / total for p in perf]
        alloc = [int(bs * s) for s in shares]
        remainder = bs - sum(alloc)
        frac = [(bs * s) - int(bs * s) for s in shares]
        order = sorted(range(len(frac)), key=lambda i: frac[i], reverse=True)
        for i in range(remainder):
            alloc[order[i]] += 1
        return alloc


class StagePerformance:
    """Per-stage normalized compute throughput and pooled memory capacity.
    Parity: model/device_group.py:13-101."""

    def __init__(
        self,
        model_config: ModelConfig,
        profiles: ProfileStore,
        cluster: ClusterSpec,
        plan: InterStagePlan,
    ) -> None:
        self.config = model_config
        self.profiles = profiles
        self.cluster = cluster
        self.plan = plan
        self.rank_device_map = self._device_placement(plan.node_sequence)

    def _device_placement(self, node_sequence: Sequence[DeviceSpec]) -> Dict[int, str]:
        """rank -> device-type name, device types concatenated in
        node-sequence order (device_group.py:22-35)."""
        names: List[str] = []
        for spec in node_sequence:
            names.extend([spec.name] * self.cluster.num_devices_of_type(spec.name))
        return {rank: names[rank] for rank in range(self.cluster.total_devices)}

    def _total_layer_time(self, dtype: str, tp: int, bs: int) -> float:
        return sum(self.profiles.get(dtype, tp, bs).layer_times_ms)

    def _hetero_group_times(
        self, device_types: Sequence[str], strategy: Tuple[int, int], hetero_bs: List[int]
    ) -> List[float]:
        dp_deg, tp_deg = strategy
        times = []
        for dp_id, h_bs in enumerate(hetero_bs):
            dtype = device_types[(len(device_types) // dp_deg) * dp_id]
            t = 0.0
            for bs_slice in pow2_slices(h_bs):
                t += self._total_layer_time(dtype, tp_deg, bs_slice)
            times.append(t)
        return times

    def _stage_ranks(self, stage_id: int) -> range:
        groups = self.plan.device_groups
        return range(sum(groups[:stage_id]), sum(groups[: stage_id + 1]))

    def compute_performance(
        self, strategies: Sequence[Tuple[int, int]], gbs: int, batches: int
    ) -> List[float]:
        perf = []
        for stage_id, (dp_deg, tp_deg) in enumerate(strategies[: len(self.plan.device_groups)]):
            device_types = [self.rank_device_map[r] for r in self._stage_ranks(stage_id)]
            if len(set(device_types)) > 1:
                balancer = DataLoadBalancer(self.profiles)
                hetero_bs = balancer.partition_data(device_types, (dp_deg, tp_deg), gbs // batches)
                times = self._hetero_group_times(device_types, (dp_deg, tp_deg), hetero_bs)
                perf.append(1.0 / max(times) if max(times) != 0 else 0.0)
            else:
                bs = gbs // batches // dp_deg
                perf.append(1.0 / self._total_layer_time(device_types[0], tp_deg, bs))
        total = sum(perf)
        return [p / total for p in perf]

    def memory_capacity(self) -> List[float]:
        """Pooled per-stage memory capacity (MB): sum of each member
        device's capacity (device_group.py:87-101)."""
        capacities = []
        for stage_id in range(len(self.plan.device_groups)):
            device_types = [self.rank_device_map[r] for r in self._stage_ranks(stage_id)]
            cap = 0.0
            for dtype in set(device_types):
                cap += self.cluster.device_memory_mb_for_type(dtype) * device_types.count(dtype)
            capacities.append(cap)
        return capacities


class LayerComputeBalancer:
    """Heuristic bin-packer assigning layers to stages proportional to
    compute capacity; parity: load_balancer.py:182-372.

    Each layer is expanded into SLICES_PER_LAYER equal sub-slices; slices
    are placed by a greedy forward pass, a backward fill of the last stage,
    and an orphan pass into the 'gap' stage with most remaining capacity;
    a layer lands in a stage iff more than half its slices did; finally up
    to MAX_REFINE_STEPS boundary layers shift between neighbor stages.
    """

    def __init__(
        self,
        num_stage: int,
        num_layer: int,
        capacity: List[float],
        layer_demand: List[float],
    ) -> None:
        self.num_stage = num_stage
        self.h = SLICES_PER_LAYER
        self.num_slices = num_layer * self.h
        self.capacity_bak = capacity.copy()
        self.capacity = capacity
        self.layer_demand = layer_demand
        self.slice_demand: List[float] = []
        for d in layer_demand:
            self.slice_demand.extend([d / self.h] * self.h)
        # slice ids per stage (after rounding: real layer ids per stage)
        self.alloc: Dict[int, List[int]] = {s: [] for s in range(num_stage)}
        self.unassigned: List[int] = []

    def run(self) -> Tuple[List[int], List[float]]:
        self._forward_pass()
        self._backward_fill()
        self._place_orphans()
        self._round_to_layers()
        self._refine_boundaries()
        partition = self._partition()
        demand = [sum(self.layer_demand[partition[i]:partition[i + 1]])
                  for i in range(len(partition) - 1)]
        return partition, demand

    def _forward_pass(self) -> None:
        """Greedy slice placement over stages 0..S-2; the final h+1 slices
        are left for the later passes (load_balancer.py:216-231)."""
        k = 0
        for stage_id in range(self.num_stage - 1):
            for sid in range(k, self.num_slices - 1 - self.h):
                if self.capacity[stage_id] > self.slice_demand[sid]:
                    self.capacity[stage_id] -= self.slice_demand[sid]
                    self.alloc[stage_id].append(sid)
                    k = sid + 1
                else:
                    self.unassigned.append(sid)
                    k = sid + 1
                    break
        for sid in range(k, self.num_slices):
            self.unassigned.append(sid)
        self.unassigned = sorted(set(self.unassigned))

    def _backward_fill(self) -> None:
        """Fill the last stage from the tail (load_balancer.py:233-249):
        force the first h slices in regardless of capacity, then extend
        downward only while contiguous and affordable."""
        last = self.num_stage - 1
        for sid in sorted(self.unassigned, reverse=True):
            if len(self.alloc[last]) < self.h:
                self.capacity[last] -= self.slice_demand[sid]
                self.alloc[last].append(sid)
                self.unassigned.remove(sid)
                continue
            if (sid + 1) != min(self.alloc[last]):
                continue
            if self.capacity[last] > self.slice_demand[sid]:
                self.capacity[last] -= self.slice_demand[sid]
                self.alloc[last].append(sid)
                self.unassigned.remove(sid)

    def _gap_stage_for(self, sid: int) -> int:
        """The stage with the most remaining capacity among the stages
        bracketing slice sid (load_balancer.py:252-275)."""
        lo, hi = min(self.alloc), max(self.alloc)
        best_below, best_above = float("-inf"), float("inf")
        for stage_id, group in self.alloc.items():
            if not group:
                continue
            gmin, gmax = min(group), max(group)
            if sid > gmax and gmax > best_below:
                lo = stage_id
                best_below = gmax
            if sid < gmin and gmin < best_above:
                hi = stage_id
                best_above = gmin
        best_stage, best_cap = None, float("-inf")
        for s in range(lo, hi + 1):
            if self.capacity[s] > best_cap:
                best_cap = self.capacity[s]
                best_stage = s
        return best_stage

    def _place_orphans(self) -> None:
        for sid in sorted(self.unassigned):
            stage_id = self._gap_stage_for(sid)
            self.capacity[stage_id] -= self.slice_demand[sid]
            self.alloc[stage_id].append(sid)
            self.unassigned.remove(sid)
        for s in self.alloc:
            self.alloc[s] = sorted(self.alloc[s])

    def _round_to_layers(self) -> None:
        """A real layer belongs to a stage iff more than half its slices
        landed there (load_balancer.py:290-308); capacities recomputed from
        the un-mutated backup against the stage's [first..last] layer span."""
        rounded: Dict[int, List[int]] = {}
        for stage_id in range(self.num_stage):
            layer_ids = [sid // self.h for sid in self.alloc[stage_id]]
            kept = [lid for lid in layer_ids if layer_ids.count(lid) > (self.h / 2)]
            rounded[stage_id] = sorted(set(kept))
        self.alloc = rounded

        capacity = []
        for stage_id in range(self.num_stage):
            group = rounded[stage_id]
            if group:
                capacity.append(
                    self.capacity_bak[stage_id] - sum(self.layer_demand[group[0]:group[-1] + 1])
                )
            else:
                capacity.append(self.capacity_bak[stage_id])
        self.capacity = capacity

    def _refine_boundaries(self) -> None:
        """Shift one boundary layer per step from the fullest neighbor into
        the stage with most spare capacity; commit only while the global
        max spare capacity does not grow (load_balancer.py:310-356)."""
        trial_cap = self.capacity.copy()
        trial_alloc = copy.deepcopy(self.alloc)

        steps = 0
        while True:
            steps += 1
            stage_id = max(range(len(trial_cap)), key=lambda i: trial_cap[i])

            near = None
            near_val = float("inf")
            if stage_id - 1 >= 0 and trial_cap[stage_id - 1] < near_val:
                near, near_val = stage_id - 1, trial_cap[stage_id - 1]
            if stage_id + 1 < len(trial_cap) and trial_cap[stage_id + 1] < near_val:
                near, near_val = stage_id + 1, trial_cap[stage_id + 1]
            if near is not None and len(trial_alloc[near]) == 1:
                near = None

            if near is not None and len(trial_alloc[near]):
                if stage_id > near:
                    layer = trial_alloc[near].pop(-1)
                else:
                    layer = trial_alloc[near].pop(0)
                trial_alloc[stage_id] = sorted(trial_alloc[stage_id] + [layer])
                d = self.layer_demand[layer]
                trial_cap[stage_id] -= d
                trial_cap[near] += d

            if max(trial_cap) > max(self.capacity) or steps > MAX_REFINE_STEPS:
                break
            self.alloc = copy.deepcopy(trial_alloc)
            self.capacity = trial_cap.copy()

    def _partition(self) -> List[int]:
        partition = [0]
        for stage_id in range(self.num_stage):
            partition.append(partition[stage_id] + len(self.alloc[stage_id]))
        return partition


class LayerLoadBalancer:
    """Assign contiguous layer ranges to stages proportional to compute
    performance, under the pooled memory capacity; on OOM, shift compute
    capacity away from memory-starved stages and retry (<= 3 attempts).
    Parity: load_balancer.py:14-144."""

    def __init__(
        self,
        cluster: ClusterSpec,
        profiles: ProfileStore,
        model_config: ModelConfig,
        gbs: int,
        norm_device_type: Optional[str] = None,
    ) -> None:
        self.cluster = cluster
        self.profiles = profiles
        self.config = model_config
        self.gbs = gbs
        dtype = norm_device_type or profiles.device_type_names[0]
        times = profiles.get(dtype, 1, 1).layer_times_ms
        total = sum(times)
        self.norm_layer_duration = [t / total for t in times]

    def _device_types_by_node_sequence(self, node_sequence: Sequence[DeviceSpec]) -> List[str]:
        """Per-rank device types: per type, (#nodes of type) x devices/node
        (load_balancer.py:109-119 — assumes uniform devices per node)."""
        per_node = self.cluster.devices_per_node
        names: List[str] = []
        for spec in node_sequence:
            names.extend([spec.name] * (self.cluster.num_nodes_of_type(spec.name) * per_node))
        return names

    def _stage_memory_demand(
        self,
        layer_partition: List[int],
        strategies: Sequence[Tuple[int, int]],
        device_group: Sequence[int],
        device_types: Sequence[str],
        gbs: int,
        batches: int,
    ) -> List[float]:
        """Profiled MB x MEM_COEF per stage. NOTE the profile is always
        looked up under the FIRST rank's device type — reference quirk Q14
        (load_balancer.py:43,51), kept for plan-ranking parity."""
        demand = []
        for stage_id, (dp_deg, tp_deg) in enumerate(strategies):
            start_rank = sum(device_group[:stage_id])
            end_rank = sum(device_group[:stage_id + 1])
            stage_types = [device_types[r] for r in range(start_rank, end_rank)]
            start_l, end_l = layer_partition[stage_id], layer_partition[stage_id + 1]

            cur = 0.001
            if len(set(stage_types)) == 1:
                bs = gbs // batches // dp_deg
                prof = self.profiles.get(device_types[0], tp_deg, bs)
                cur += prof.memory_slice(start_l, end_l) * MEM_COEF
            else:
                balancer = DataLoadBalancer(self.profiles)
                hetero_bs = balancer.partition_data(stage_types, (dp_deg, tp_deg), gbs // batches)
                for h_bs in hetero_bs:
                    for bs_slice in pow2_slices(h_bs):
                        prof = self.profiles.get(device_types[0], tp_deg, bs_slice)
                        cur += prof.memory_slice(start_l, end_l) * MEM_COEF
            demand.append(cur)
        return demand

    @staticmethod
    def _memory_state(
        demand: Sequence[float], capacity: Sequence[float]
    ) -> Tuple[bool, List[float]]:
        usage = [c - d for c, d in zip(capacity, demand)]
        return (min(usage) < 0), usage

    def _shift_compute_capacity(
        self,
        compute: Sequence[float],
        mem_capacity: Sequence[float],
        mem_demand: Sequence[float],
    ) -> Optional[List[float]]:
        """On OOM: shrink starved stages' compute share (x cap/demand x0.9)
        and redistribute the freed share to stages with memory headroom
        proportional to their compute (load_balancer.py:71-107)."""
        adjusted: List[float] = []
        headroom: List[float] = []
        deficit = 0.0
        for c, m_cap, m_dem in zip(compute, mem_capacity, mem_demand):
            if m_cap > m_dem:
                adjusted.append(c)
                headroom.append(c * m_cap / m_dem - c)
            else:
                headroom.append(0.0)
                shrunk = c * (m_cap / m_dem) * 0.9
                adjusted.append(shrunk)
                deficit += c - shrunk

        if sum(headroom) < deficit:
            return None

        extra = [0.0] * len(adjusted)
        while deficit > 0.01:
            active_total = sum(c for h, c in zip(headroom, compute) if h > 0.001)
            ratios = [c / active_total if h > 0.001 else 0.0 for h, c in zip(headroom, compute)]
            for stage_id, (ratio, h) in enumerate(zip(ratios, list(headroom))):
                give = h if deficit * ratio > h else deficit * ratio
                extra[stage_id] += give
                headroom[stage_id] -= give
                deficit -= give
        return [e + a for e, a in zip(extra, adjusted)]

    def partition_layer(
        self,
        plan: InterStagePlan,
        strategies: Sequence[Tuple[int, int]],
        stage_compute_performance: List[float],
        stage_memory_capacity: Sequence[float],
    ) -> Tuple[Optional[List[int]], int, Optional[List[float]]]:
        device_types = self._device_types_by_node_sequence(plan.node_sequence)

        attempt = 1
        while attempt <= MAX_PARTITION_ATTEMPTS:
            balancer = LayerComputeBalancer(
                len(stage_compute_performance),
                self.config.num_layers,
                stage_compute_performance.copy(),
                self.norm_layer_duration,
            )
            layer_partition, _demand = balancer.run()
            mem_demand = self._stage_memory_demand(
                layer_partition, strategies, plan.device_groups, device_types,
                plan.gbs, plan.batches,
            )
            exceeded, memory_state = self._memory_state(mem_demand, stage_memory_capacity)
            if not exceeded:
                return layer_partition, attempt, memory_state

            stage_compute_performance = self._shift_compute_capacity(
                stage_compute_performance, stage_memory_capacity, mem_demand
            )
            if not stage_compute_performance:
                return None, -1, None
            attempt += 1
        return None, -1, None
