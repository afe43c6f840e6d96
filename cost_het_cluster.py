#!/usr/bin/env python3
"""Heterogeneous-cluster planner CLI (reference-compatible entry point)."""
from metis_amd.cli.het_cluster import main

if __name__ == "__main__":
    main()
