"""MoE expert parallelism (gloo, world 2): loss/grad parity with a
single-process tp=1 model, and replicated-parameter consistency after an
optimizer step (the router grad-sum hook is what keeps EP ranks equal)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from metis_amd.models.moe import MoEModel, MOE_SPECS

SPEC = MOE_SPECS["moe-tiny"]


def _env(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)


def _shard_from_full(full: MoEModel, shard: MoEModel, r: int, tp: int) -> None:
    with torch.no_grad():
        shard.wte.weight.copy_(full.wte.weight)
        shard.wpe.weight.copy_(full.wpe.weight)
        for fb, sb in zip(full.blocks, shard.blocks):
            for name in ("ln_attn", "ln_mlp"):
                getattr(sb, name).weight.copy_(getattr(fb, name).weight)
                getattr(sb, name).bias.copy_(getattr(fb, name).bias)
            h = fb.qkv.weight.size(0) // 3
            hp = h // tp
            rows = torch.cat([
                torch.arange(blk * h + r * hp, blk * h + (r + 1) * hp)
                for blk in range(3)
            ])
            sb.qkv.weight.copy_(fb.qkv.weight[rows])
            sb.qkv.bias.copy_(fb.qkv.bias[rows])
            ipr = sb.proj.in_per_rank
            sb.proj.weight.copy_(fb.proj.weight[:, r * ipr:(r + 1) * ipr])
            sb.proj.bias.copy_(fb.proj.bias)
            # router replicated; experts sharded contiguously
            sb.router.weight.copy_(fb.router.weight)
            sb.router.bias.copy_(fb.router.bias)
            le = sb.experts.local_experts
            for t in ("w1", "b1", "w2", "b2"):
                getattr(sb.experts, t).copy_(
                    getattr(fb.experts, t)[r * le:(r + 1) * le])
        shard.ln_final.weight.copy_(full.ln_final.weight)
        shard.ln_final.bias.copy_(full.ln_final.bias)
        opr = shard.head.out_per_rank
        shard.head.weight.copy_(full.head.weight[r * opr:(r + 1) * opr])
        shard.head.bias.copy_(full.head.bias[r * opr:(r + 1) * opr])


def _ep_worker(rank, world, port, out):
    _env(rank, world, port)
    from metis_amd.runtime.comm import init_parallel

    ctx = init_parallel(dp=1, tp=2, pp=1)
    torch.manual_seed(7)
    full = MoEModel(SPEC, tp=1, dtype=torch.float32)
    shard = MoEModel(SPEC, tp=2, dtype=torch.float32, tp_group=ctx.tp_group)
    _shard_from_full(full, shard, rank, 2)

    g = torch.Generator().manual_seed(11)
    tokens = torch.randint(0, SPEC.vocab_size, (2, 32), generator=g)
    labels = torch.roll(tokens, -1, 1)

    ref_loss = full(tokens, labels=labels)
    ep_loss = shard(tokens, labels=labels)
    assert torch.allclose(ref_loss, ep_loss, atol=1e-4), (ref_loss, ep_loss)

    ref_loss.backward()
    ep_loss.backward()

    # expert grads: this rank's slice of the full model's expert grads
    le = shard.blocks[0].experts.local_experts
    fg = full.blocks[0].experts.w1.grad[rank * le:(rank + 1) * le]
    sg = shard.blocks[0].experts.w1.grad
    assert torch.allclose(fg, sg, atol=1e-4)

    # router grads: the post-accumulate hook summed the partial per-rank
    # contributions, so each EP rank must now hold the FULL router grad
    fr = full.blocks[0].router.weight.grad
    sr = shard.blocks[0].router.weight.grad
    assert torch.allclose(fr, sr, atol=1e-4), (fr - sr).abs().max()

    # replicated params must be bit-identical across the EP group
    flat = shard.blocks[0].router.weight.grad.reshape(-1)
    gathered = [torch.empty_like(flat) for _ in range(world)]
    dist.all_gather(gathered, flat)
    assert torch.equal(gathered[0], gathered[1])
    out.put(("ok", rank))
    dist.destroy_process_group()


def _moe_runner_worker(rank, world, port, out):
    _env(rank, world, port)
    from metis_amd.runtime.comm import init_parallel
    from metis_amd.runtime.runner import PlanRunner

    ctx = init_parallel(dp=1, tp=1, pp=2)
    torch.manual_seed(5)
    runner = PlanRunner(SPEC, ctx, mbs=1, gbs=4, dtype=torch.float32,
                        schedule="1f1b")
    loss = runner.train_step()
    loss2 = runner.train_step()   # second step: per-step state resets
    if ctx.is_last_stage:
        assert loss > 0 and loss2 > 0
    # MoE stages without the head still get router grads (aux backward path)
    for b in runner.model.blocks:
        assert b.router.weight.grad is not None
    out.put(("ok", rank))
    dist.destroy_process_group()


def _run_workers(fn, world=2, port=29621):
    mp_ctx = mp.get_context("spawn")
    out = mp_ctx.Queue()
    procs = [mp_ctx.Process(target=fn, args=(r, world, port, out))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
    for p in procs:
        assert p.exitcode == 0, f"worker failed with exit code {p.exitcode}"
    results = []
    while not out.empty():
        results.append(out.get())
    assert len(results) == world


def test_expert_parallel_matches_single_process():
    _run_workers(_ep_worker, port=29621)


def test_moe_pipeline_runner_1f1b():
    _run_workers(_moe_runner_worker, port=29622)


def _moe_interleaved_worker(rank, world, port, out):
    _env(rank, world, port)
    from metis_amd.runtime.comm import init_parallel
    from metis_amd.runtime.runner import PlanRunner

    ctx = init_parallel(dp=1, tp=1, pp=2)
    torch.manual_seed(5)
    runner = PlanRunner(SPEC, ctx, mbs=1, gbs=4, dtype=torch.float32,
                        schedule="interleaved", vpp=2)
    loss = runner.train_step()
    if rank == 1:  # head chunk (vs=3) lives on rank 1
        assert loss > 0
    # every chunk's routers got gradients (aux backward incl. non-head
    # virtual stages)
    for chunk in runner.model_chunks:
        for b in chunk.blocks:
            assert b.router.weight.grad is not None
    out.put(("ok", rank))
    dist.destroy_process_group()


def test_moe_interleaved_schedule():
    _run_workers(_moe_interleaved_worker, port=29626)


def _ep_accum_worker(rank, world, port, out):
    """Gradient accumulation over 2 microbatches with ep=2:
    the deferred once-per-step EP sum must reproduce the full model's
    accumulated router grad — the old per-backward hook re-summed earlier
    microbatches (3x instead of 2x on identical data)."""
    _env(rank, world, port)
    from metis_amd.runtime.comm import init_parallel
    from metis_amd.partial_grads import defer_partial, sync_partial_grads

    ctx = init_parallel(dp=1, tp=2, pp=1)
    torch.manual_seed(7)
    full = MoEModel(SPEC, tp=1, dtype=torch.float32)
    shard = MoEModel(SPEC, tp=2, dtype=torch.float32, tp_group=ctx.tp_group)
    _shard_from_full(full, shard, rank, 2)

    g = torch.Generator().manual_seed(13)
    mbs = [(torch.randint(0, SPEC.vocab_size, (2, 32), generator=g))
           for _ in range(2)]

    # accumulation driver contract: defer hooks, sum once after the last
    # backward (what PlanRunner/GradBucketSync do internally)
    defer_partial(shard.parameters())
    for tokens in mbs:
        labels = torch.roll(tokens, -1, 1)
        (full(tokens, labels=labels) / 2).backward()
        (shard(tokens, labels=labels) / 2).backward()
    sync_partial_grads(shard.parameters())

    fr = full.blocks[0].router.weight.grad
    sr = shard.blocks[0].router.weight.grad
    assert torch.allclose(fr, sr, atol=1e-4), (fr - sr).abs().max()
    fb = full.blocks[-1].router.bias.grad
    sb = shard.blocks[-1].router.bias.grad
    assert torch.allclose(fb, sb, atol=1e-4)
    out.put(("ok", rank))
    dist.destroy_process_group()


def test_ep_grad_accumulation_once_per_step():
    _run_workers(_ep_accum_worker, port=29631)


def _dp_ep_worker(rank, world, port, out):
    """dp=2 x ep=2 with accumulation: the EP sum must run
    BEFORE the DP bucket copy (it now happens inside the bucket hook), so
    router weights stay identical across TP ranks after optimizer steps."""
    _env(rank, world, port)
    from metis_amd.runtime.comm import init_parallel
    from metis_amd.runtime.runner import PlanRunner

    ctx = init_parallel(dp=2, tp=2, pp=1)
    torch.manual_seed(5)
    runner = PlanRunner(SPEC, ctx, mbs=1, gbs=4, dtype=torch.float32)
    assert runner.num_microbatches == 2     # accumulation active
    assert runner.grad_sync is not None     # DP bucket path active
    for _ in range(2):
        runner.train_step()

    # replicated router params must be bit-identical across the WORLD
    # (both across DP replicas and across TP/EP ranks)
    for blk in runner.model.blocks:
        for p in (blk.router.weight, blk.router.bias):
            flat = p.detach().reshape(-1)
            gathered = [torch.empty_like(flat) for _ in range(world)]
            dist.all_gather(gathered, flat)
            for gth in gathered[1:]:
                assert torch.equal(gathered[0], gth)
    out.put(("ok", rank))
    dist.destroy_process_group()


def test_dp_ep_router_stays_synced():
    _run_workers(_dp_ep_worker, world=4, port=29632)
