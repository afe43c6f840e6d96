"""Plan execution runtime: one process per GPU over RCCL/xGMI."""

from metis_amd.runtime.comm import ParallelContext
from metis_amd.runtime.runner import PlanRunner

__all__ = ["ParallelContext", "PlanRunner"]
