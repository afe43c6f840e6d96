"""Plan search space, load balancing and analytical cost model.

Layer map:

* ``plans``       — plan dataclasses (UniformPlan / InterStagePlan / IntraStagePlan)
* ``uniform``     — uniform (Megatron-style) plan enumeration
* ``groups``      — pipeline device-group compositions + bounded multiset permutation
* ``inter_stage`` — InterStagePlanGenerator (node seq x device groups x microbatches)
* ``intra_stage`` — per-stage (dp, tp) escalation under memory pressure
* ``stage_perf``  — per-stage normalized compute throughput & memory capacity
* ``balancer``    — layer->stage and data->replica load balancing
* ``volume``      — GPT activation/parameter volume model
* ``bandwidth``   — cluster communication topology (slowest-link classes)
* ``cost``        — Homo/Hetero analytical cost estimators
* ``validate``    — estimate-vs-measured validation (cost-model error %)
"""

from metis_amd.planner.plans import UniformPlan, InterStagePlan, IntraStagePlan
from metis_amd.planner.uniform import uniform_plans
from metis_amd.planner.cost import HomoCostEstimator, HeteroCostEstimator

__all__ = [
    "UniformPlan",
    "InterStagePlan",
    "IntraStagePlan",
    "uniform_plans",
    "HomoCostEstimator",
    "HeteroCostEstimator",
]
