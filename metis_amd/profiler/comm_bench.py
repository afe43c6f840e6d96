"""RCCL/xGMI microbenchmarks -> clusterfile bandwidth calibration.

Measures the all-reduce latency-bandwidth curve and p2p send/recv over
the world group (rccl-tests style): per payload size, bus bandwidth
``2(n-1)/n * bytes / time`` for all-reduce and ``bytes / time`` for p2p.

On one MI355X node the xGMI mesh is point-to-point (7 links x ~153 GB/s
per GPU); a ring all-reduce is single-link bound, so the *measured* bus
bandwidth here — not the marketing aggregate — is what belongs in the
clusterfile's ``intra_bandwidth``. The small-payload
latency feeds the alpha term of the planner's alpha_beta comm model.

Run: torchrun --nproc-per-node N -m metis_amd.profiler.comm_bench
"""

from __future__ import annotations

import argparse
import json
import os
import time
from typing import Dict, List

import torch
import torch.distributed as dist


def _time_collective(fn, iters: int, warmup: int) -> float:
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def bench_allreduce(sizes_bytes: List[int], iters: int = 20, warmup: int = 5) -> List[Dict]:
    n = dist.get_world_size()
    dev = torch.device("cuda", torch.cuda.current_device())
    rows = []
    for size in sizes_bytes:
        buf = torch.randn(size // 2, dtype=torch.bfloat16, device=dev)
        t = _time_collective(lambda: dist.all_reduce(buf), iters, warmup)
        bus_bw = 2 * (n - 1) / n * size / t / 1e9
        rows.append({"bytes": size, "time_us": t * 1e6, "busbw_GBps": bus_bw,
                     "algbw_GBps": size / t / 1e9})
    return rows


def bench_p2p(sizes_bytes: List[int], iters: int = 20, warmup: int = 5) -> List[Dict]:
    rank, n = dist.get_rank(), dist.get_world_size()
    if n < 2:
        return []
    dev = torch.device("cuda", torch.cuda.current_device())
    peer = rank ^ 1
    rows = []
    for size in sizes_bytes:
        buf = torch.randn(size // 2, dtype=torch.bfloat16, device=dev)

        def xfer():
            if rank >= n // 2 * 2:
                return  # odd world size: last rank sits out
            if rank % 2 == 0:
                dist.send(buf, dst=peer)
            else:
                dist.recv(buf, src=peer)

        t = _time_collective(xfer, iters, warmup)
        rows.append({"bytes": size, "time_us": t * 1e6,
                     "bw_GBps": size / t / 1e9})
    return rows


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--out", default="profiles/comm_bench.json")
    p.add_argument("--iters", type=int, default=20)
    args = p.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world < 2:
        print(json.dumps({"error": "needs WORLD_SIZE >= 2"}))
        return
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))
    dist.init_process_group("nccl", rank=rank, world_size=world)

    sizes = [1 << k for k in range(12, 31, 2)]  # 4 KB .. 1 GB
    ar = bench_allreduce(sizes, iters=args.iters)
    p2p = bench_p2p(sizes, iters=args.iters)

    if rank == 0:
        big_bw = max(r["busbw_GBps"] for r in ar)
        alpha_us = min(r["time_us"] for r in ar)
        doc = {
            "world_size": world,
            "allreduce": ar,
            "p2p": p2p,
            "clusterfile_suggestion": {
                "intra_bandwidth": round(big_bw, 1),
                "alpha_us": round(alpha_us, 1),
            },
        }
        os.makedirs(os.path.dirname(args.out) or ".", exist_ok=True)
        with open(args.out, "w") as fh:
            json.dump(doc, fh, indent=2)
        print(json.dumps(doc["clusterfile_suggestion"]))
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
