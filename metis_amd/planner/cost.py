"""Analytical cost estimators (time in ms).

Parity: reference model/cost_estimator.py:16-244. The cost formula
(SURVEY.md §3.4)::

    T(plan) = (B-1) * max_s t_s + sum_s t_s        # GPipe bubble
            + fb_sync(tp, mbs) * B                 # per-microbatch sync
            + T_opt                                # optimizer share
            + 2(d-1)/(d*BW_dp) * max_s P_s         # DP ring all-reduce
            + sum_{s<last} A_s / BW_pp(s, s+1)     # PP p2p activations
            + t_batch_gen * B

MI355X extension: ``comm_model="alpha_beta"`` adds a latency term
(alpha_us per collective / per p2p hop) on top of the bandwidth term —
the reference omits latency, which misprices the small LN/embedding-grad
reductions and short xGMI hops. Bandwidths are GB/s and are converted to
bytes/ms via *1024^2 exactly as the reference does (its "GB/s" is really
MB/ms — kept for parity; calibrate the clusterfile accordingly).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple

from metis_amd.cluster import ClusterSpec
from metis_amd.config import ModelConfig, PlannerArgs
from metis_amd.planner.balancer import DataLoadBalancer, pow2_slices
from metis_amd.planner.bandwidth import HeteroTopology, HomoTopology
from metis_amd.planner.plans import InterStagePlan, UniformPlan
from metis_amd.planner.volume import GPTVolume, uniform_layer_split
from metis_amd.profiles import ProfileStore


class CostEstimatorBase:
    def __init__(
        self,
        profiles: ProfileStore,
        model_config: ModelConfig,
        volume: GPTVolume,
        cluster: ClusterSpec,
        args: Optional[PlannerArgs] = None,
    ) -> None:
        self.profiles = profiles
        self.config = model_config
        self.volume = volume
        self.cluster = cluster
        self.args = args or PlannerArgs(gbs=1)

    # --- shared cost terms (cost_estimator.py:31-80) ----------------------
    def _batch_gen_cost(self, batches: int) -> float:
        return self.profiles.model.batch_generator_ms * batches

    def _alpha_ms(self) -> float:
        return self.args.alpha_us / 1000.0 if self.args.comm_model == "alpha_beta" else 0.0

    def _dp_cost(self, stage_parameters: Sequence[float], bandwidth: float, dp_deg: int) -> float:
        """Ring all-reduce of the largest stage's gradients."""
        bw = bandwidth * 1024 * 1024
        cost = 2 * (dp_deg - 1) / (dp_deg * bw) * max(stage_parameters)
        if dp_deg > 1:
            cost += self._alpha_ms()
        return cost

    def _pp_cost(self, activation_size: float, bandwidth: float) -> float:
        """Stage-boundary activation send/recv."""
        return activation_size / (bandwidth * 1024 * 1024) + self._alpha_ms()

    def _oom(self, stage_memory_mb: Sequence[float]) -> bool:
        return self.cluster.device_memory_mb(0) < max(stage_memory_mb)

    def _profile(self, dtype: str, tp: int, bs: int):
        """Profile lookup honoring the interpolate_bs extension."""
        if self.args.interpolate_bs:
            return self.profiles.get_interp(dtype, tp, bs)
        return self.profiles.get(dtype, tp, bs)

    def _use_marginal(self, prof) -> bool:
        return (self.args.microbatch_model == "marginal"
                and prof.marginal_mb_ms is not None)

    def _fb_sync_of(self, prof) -> float:
        """fb_sync of a (possibly interpolated) profile; a 0.0 value
        raises KeyError like a missing one — reference parity
        (cost_estimator.py:68-69, quirk Q15)."""
        if not prof.fb_sync_ms:
            raise KeyError("fb_sync missing (or 0.0) in profile data")
        return prof.fb_sync_ms


class HomoCostEstimator(CostEstimatorBase):
    """Uniform-plan estimator (cost_estimator.py:83-138)."""

    def __init__(self, *cargs, **kw) -> None:
        super().__init__(*cargs, **kw)
        self.topology = HomoTopology(self.cluster)

    def get_cost(self, plan: UniformPlan, device_type: str) -> Tuple[float, List[float], bool]:
        """Returns (time_ms, per-stage demand MB, oom). Raises KeyError for
        unprofiled (tp, bs) points — callers skip those plans."""
        tp, pp, dp = plan.tp, plan.pp, plan.dp
        bs = plan.mbs
        num_mbs = plan.gbs // plan.mbs // plan.dp

        param_sizes = self.volume.parameter_sizes(tp)
        stage_layers = uniform_layer_split(self.volume.num_layers, pp)

        lens: List[float] = []
        stage_params: List[float] = []
        stage_memory: List[float] = []
        pp_cost, fb_sync_cost = 0.0, 0.0
        for stage_id, _count in enumerate(stage_layers):
            start = sum(stage_layers[:stage_id])
            end = sum(stage_layers[: stage_id + 1])

            prof = self._profile(device_type, tp, bs)
            if self._use_marginal(prof):
                # per-microbatch stage time = measured accumulation
                # marginal; iteration residual charged ONCE (summed over
                # stage slices = whole-model residual)
                lens.append(prof.marginal_slice(start, end))
                fb_sync_cost += prof.residual_slice(start, end)
            else:
                lens.append(prof.time_slice(start, end))
                if stage_id == len(stage_layers) - 1:
                    fb_sync_cost = self._fb_sync_of(prof) * num_mbs
            stage_params.append(sum(param_sizes[start:end]))
            stage_memory.append(prof.memory_slice(start, end))

            if stage_id != len(stage_layers) - 1:
                act = self.volume.activation_size(end, bs, tp)
                pp_bw = self.topology.slowest_pp_bandwidth((pp, tp, dp), stage_id)
                pp_cost += self._pp_cost(act, pp_bw)

        # --- schedule pricing (MI355X extension; default keeps parity) ---
        # gpipe : (B-1)*max + sum (reference cost_estimator.py:129), one
        #         microbatch of activations charged (reference memory
        #         semantics).
        # 1f1b  : same bubble, but activation memory scales with the
        #         in-flight microbatch count min(B, pp - stage_id).
        # interleaved (vpp=v): bubble shrinks to (pp-1)/v chunk slots —
        #         execution = (B-1)*max + (sum + (v-1)*max)/v, which
        #         reduces to B*t + (pp-1)*t/v on uniform stages — at v x
        #         the p2p boundary crossings; in-flight count as 1f1b.
        sched, v = self.args.schedule, max(self.args.vpp, 1)
        if sched != "gpipe":
            state_mb = [p * 9.0 / (1024 * 1024) for p in
                        self.volume.parameter_sizes(tp)]
            for sid in range(len(stage_layers)):
                start = sum(stage_layers[:sid])
                end = sum(stage_layers[: sid + 1])
                state = sum(state_mb[start:end])
                act = max(stage_memory[sid] - state, 0.0)
                inflight = min(num_mbs, pp - sid) if pp > 1 else 1
                stage_memory[sid] = state + act * max(inflight, 1)
        oom = self._oom(stage_memory)
        if sched == "interleaved" and pp > 1:
            execution = ((num_mbs - 1) * max(lens)
                         + (sum(lens) + (v - 1) * max(lens)) / v)
            pp_cost *= v
        else:
            execution = (num_mbs - 1) * max(lens) + sum(lens)
        optimizer = self.profiles.model.optimizer_time_ms / pp / tp
        dp_bw = self.topology.slowest_dp_bandwidth((pp, tp, dp))
        dp_cost = self._dp_cost(stage_params, dp_bw, dp)
        total = (
            execution + fb_sync_cost + optimizer + dp_cost + pp_cost
            + self._batch_gen_cost(num_mbs)
        )
        return total, stage_memory, oom


class HeteroCostEstimator(CostEstimatorBase):
    """Non-uniform-plan estimator (cost_estimator.py:141-244)."""

    def _optimizer_cost(self, tp_deg: int, num_layers: int) -> float:
        ratio = num_layers / self.config.num_layers
        return self.profiles.model.optimizer_time_ms / tp_deg * ratio

    def _stage_execution_cost(
        self,
        device_types: Sequence[str],
        start_layer: int,
        end_layer: int,
        strategy: Tuple[int, int],
        gbs: int,
        batches: int,
    ) -> float:
        dp_deg, tp_deg = strategy
        if len(set(device_types)) == 1:
            prof = self._profile(device_types[0], tp_deg, gbs // dp_deg // batches)
            if self._use_marginal(prof):
                return prof.marginal_slice(start_layer, end_layer)
            return prof.time_slice(start_layer, end_layer)

        balancer = DataLoadBalancer(self.profiles)
        hetero_bs = balancer.partition_data(device_types, strategy, gbs // batches)
        costs = []
        for dp_id, h_bs in enumerate(hetero_bs):
            if h_bs == 0:
                continue
            dtype = device_types[(len(device_types) // dp_deg) * dp_id]
            t = 0.0
            for bs_slice in pow2_slices(h_bs):
                if bs_slice > self.args.max_profiled_batch_size:
                    raise KeyError(f"batch_size({bs_slice}) not profiled")
                prof = self._profile(dtype, tp_deg, bs_slice)
                if self._use_marginal(prof):
                    t += prof.marginal_slice(start_layer, end_layer)
                else:
                    t += prof.time_slice(start_layer, end_layer)
            costs.append(t)
        return max(costs)

    def get_cost(
        self,
        plan: InterStagePlan,
        strategies: Sequence[Tuple[int, int]],
        layer_partition: Sequence[int],
        rank_device_map: Dict[int, str],
    ) -> float:
        topology = HeteroTopology(self.cluster, plan)

        lens: List[float] = []
        dp_costs: List[float] = []
        opt_costs: List[float] = []
        pp_cost, fb_sync_cost = 0.0, 0.0
        for stage_id, strategy in zip(range(plan.num_stage), strategies):
            start_l, end_l = layer_partition[stage_id], layer_partition[stage_id + 1]
            start_rank = sum(plan.device_groups[:stage_id])
            end_rank = sum(plan.device_groups[: stage_id + 1])
            device_types = [rank_device_map[r] for r in range(start_rank, end_rank)]

            lens.append(
                self._stage_execution_cost(
                    device_types, start_l, end_l, strategy, plan.gbs, plan.batches
                )
            )

            dp_deg, tp_deg = strategy
            mbs = plan.gbs // dp_deg // plan.batches
            if self.args.microbatch_model == "marginal":
                profs = [self._profile(d, tp_deg, mbs)
                         for d in set(device_types)]
            else:
                profs = []
            if profs and all(self._use_marginal(p) for p in profs):
                # iteration residual charged once per stage slice (the
                # slowest device type's), not per microbatch
                fb_sync_cost += max(
                    p.residual_slice(start_l, end_l) for p in profs)
            elif stage_id == plan.num_stage - 1:
                fb_sync_cost = max(
                    self._fb_sync_of(self._profile(d, tp_deg, mbs))
                    for d in set(device_types)) * plan.batches
            if stage_id != plan.num_stage - 1:
                act = self.volume.activation_size(end_l, mbs, tp_deg)
                pp_cost += self._pp_cost(act, topology.slowest_pp_bandwidth(stage_id))

            stage_params = self.volume.stage_parameter_size(tp_deg, start_l, end_l)
            dp_bw = topology.slowest_dp_bandwidth(strategy, stage_id)
            dp_costs.append(self._dp_cost([stage_params], dp_bw, dp_deg))
            opt_costs.append(self._optimizer_cost(tp_deg, end_l - start_l))

        # schedule pricing mirrors HomoCostEstimator: interleaved (vpp=v)
        # shrinks the bubble to (pp-1)/v chunk slots at v x the p2p
        # volume; gpipe/1f1b keep the reference bubble
        sched, v = self.args.schedule, max(self.args.vpp, 1)
        if sched == "interleaved" and plan.num_stage > 1:
            execution = ((plan.batches - 1) * max(lens)
                         + (sum(lens) + (v - 1) * max(lens)) / v)
            pp_cost *= v
        else:
            execution = (plan.batches - 1) * max(lens) + sum(lens)
        total = (
            execution + fb_sync_cost + max(opt_costs) + max(dp_costs) + pp_cost
            + self._batch_gen_cost(plan.batches)
        )
        return total
