"""Offline profiler: produces the DeviceType.*.json profiles the planner
consumes, plus RCCL/xGMI bandwidth microbenchmarks for the clusterfile."""
