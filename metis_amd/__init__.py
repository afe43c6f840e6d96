"""metis_amd — MI355X-native automatic distributed-training planner.

A from-scratch framework with the capabilities of SamsungLabs/Metis
(USENIX ATC'24): it enumerates 3D-parallel plans (DP x TP x PP with
non-uniform, heterogeneity-aware pipeline stages), balances layers and
data across stages, and ranks every plan with an analytical cost model.

Unlike the reference (planner only), this package also ships the
MI355X-native pieces the Metis README prescribes but never released:

* ``metis_amd.ops``       — hand-written CDNA4 (gfx950) HIP kernels for the
                            transformer hot ops (MFMA GEMM, flash attention,
                            LayerNorm, fused Adam), built in-tree.
* ``metis_amd.models``    — GPT/Llama model families built on those ops.
* ``metis_amd.runtime``   — plan runner: one process per GPU over RCCL/xGMI
                            executing a chosen (dp, tp, pp) plan.
* ``metis_amd.profiler``  — the per-layer profiler that emits
                            ``DeviceType.MI355X_tp{N}_bs{M}.json`` profiles
                            (schema: reference README.md:61-113), plus RCCL
                            bandwidth microbenchmarks for the clusterfile.

Layer map mirrors SURVEY.md §1; component parity tracked against
SURVEY.md §2.
"""

__version__ = "0.1.0"

from metis_amd.cluster import ClusterSpec, DeviceSpec, device_registry
from metis_amd.config import ModelConfig, PlannerArgs

__all__ = [
    "ClusterSpec",
    "DeviceSpec",
    "device_registry",
    "ModelConfig",
    "PlannerArgs",
]
