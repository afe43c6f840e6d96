#!/usr/bin/env python3
"""Homogeneous-cluster planner CLI (reference-compatible entry point)."""
from metis_amd.cli.homo_cluster import main

if __name__ == "__main__":
    main()
