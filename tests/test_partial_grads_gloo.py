"""Direct unit tests for metis_amd.partial_grads (the once-per-step
replicated-partial-grad protocol; exercised end-to-end by the MoE/SP
suites, pinned down here on bare parameters)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from metis_amd.partial_grads import (defer_partial, mark_partial,
                                     partial_group, sync_partial_grads)


def _env(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)


def _worker(rank, world, port, out):
    _env(rank, world, port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    group = dist.group.WORLD

    # immediate mode: one backward -> hook sums across ranks
    p = torch.nn.Parameter(torch.ones(3))
    mark_partial(p, group)
    assert partial_group(p) is group
    (p * (rank + 1.0)).sum().backward()
    # rank r contributed grad (r+1); sum over 2 ranks = 3
    assert torch.allclose(p.grad, torch.full((3,), 3.0))

    # deferred mode: two accumulating backwards, ONE sum at the end
    q = torch.nn.Parameter(torch.ones(2))
    mark_partial(q, group)
    defer_partial([q])
    (q * (rank + 1.0)).sum().backward()
    (q * (rank + 1.0)).sum().backward()
    sync_partial_grads([q])
    # per-rank accumulated 2*(r+1); summed = 2*1 + 2*2 = 6
    assert torch.allclose(q.grad, torch.full((2,), 6.0))

    # marking twice (tied params) registers only one hook
    t = torch.nn.Parameter(torch.ones(1))
    mark_partial(t, group)
    mark_partial(t, group)
    t.sum().backward()
    assert torch.allclose(t.grad, torch.full((1,), float(world)))

    # world-size-1 group: mark is a no-op
    solo = dist.new_group([rank])
    u = torch.nn.Parameter(torch.ones(1))
    mark_partial(u, solo)
    assert partial_group(u) is None
    out.put(("ok", rank))
    dist.destroy_process_group()


def test_partial_grads_protocol():
    mp_ctx = mp.get_context("spawn")
    out = mp_ctx.Queue()
    procs = [mp_ctx.Process(target=_worker, args=(r, 2, 29645, out))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=120)
    for p in procs:
        assert p.exitcode == 0
    n = 0
    while not out.empty():
        out.get()
        n += 1
    assert n == 2
