"""Resumable training driver: periodic per-rank checkpoints + restart.

The failure-recovery loop the reference has no runtime for: train under
torchrun, checkpoint every K steps (per-rank shard files + a COMPLETE
marker written only after a barrier), and on restart ``--resume`` finds
the latest COMPLETE step directory and continues from it — a crashed or
preempted job relaunches with the same command line and loses at most
K steps. Partial checkpoint directories (a rank died mid-write) are
ignored by resume.

  torchrun --nproc-per-node 8 -m metis_amd.cli.train \
      --model gpt3-2.7b --dp 8 --mbs 16 --gbs 128 --steps 1000 \
      --checkpoint-dir ckpts --checkpoint-every 100 --resume
"""

from __future__ import annotations

import argparse
import json
import os
import re
import time

import torch
import torch.distributed as dist

from metis_amd.models.gpt import MODEL_SPECS as _GPT_SPECS
from metis_amd.models.llama import LLAMA_SPECS
from metis_amd.models.moe import MOE_SPECS

MODEL_SPECS = {**_GPT_SPECS, **LLAMA_SPECS, **MOE_SPECS}
from metis_amd.runtime.comm import init_parallel
from metis_amd.runtime.runner import PlanRunner


def latest_complete_step(ckpt_dir: str) -> int:
    """Highest step with a COMPLETE marker, or -1."""
    best = -1
    if not os.path.isdir(ckpt_dir):
        return best
    for name in os.listdir(ckpt_dir):
        m = re.fullmatch(r"step_(\d+)", name)
        if m and os.path.exists(os.path.join(ckpt_dir, name, "COMPLETE")):
            best = max(best, int(m.group(1)))
    return best


def save_step(runner: PlanRunner, ckpt_dir: str, step: int,
              tag: str = None) -> None:
    """Periodic checkpoint: per-rank shard + a COMPLETE marker written
    only after a barrier. With ``tag`` (emergency save from the failure
    handler) the barrier and marker are skipped — peers may be dead,
    and resume must not trust a possibly-inconsistent snapshot."""
    name = f"step_{step}" if tag is None else f"step_{step}_{tag}"
    d = os.path.join(ckpt_dir, name)
    os.makedirs(d, exist_ok=True)
    runner.save_checkpoint(os.path.join(d, f"rank{runner.ctx.rank}.pt"))
    if tag is not None:
        return
    if dist.is_initialized():
        dist.barrier()  # marker only after EVERY rank's shard is on disk
    if runner.ctx.rank == 0:
        with open(os.path.join(d, "COMPLETE"), "w") as fh:
            fh.write(json.dumps({"step": step, "time": time.time()}))


def _train_one(args, ctx, runner, sched, step, start, t0,
               tokens_per_step, eval_loader, eval_loss) -> int:
    import time

    if sched is not None:
        sched.step(step)
    loss = runner.train_step()
    done = step + 1
    if ctx.rank == ctx.world_size - 1 and done % args.log_every == 0:
        tps = tokens_per_step * (done - start) / max(time.time() - t0, 1e-9)
        print(f"step {done}: loss {loss:.4f} lr {runner.optimizer.lr:.2e} "
              f"{tps:,.0f} tok/s", flush=True)
    if eval_loader is not None and done % args.eval_every == 0:
        el = eval_loss()
        if ctx.rank == ctx.world_size - 1:
            print(f"step {done}: eval loss {el:.4f}", flush=True)
    if (args.checkpoint_dir and args.checkpoint_every > 0
            and done % args.checkpoint_every == 0):
        save_step(runner, args.checkpoint_dir, done)
    return done


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--config", default=None,
                   help="YAML file of flag defaults (CLI flags override)")
    p.add_argument("--model", default="gpt2-small", choices=sorted(MODEL_SPECS))
    p.add_argument("--dp", type=int, default=1)
    p.add_argument("--tp", type=int, default=1)
    p.add_argument("--pp", type=int, default=1)
    p.add_argument("--mbs", type=int, default=1)
    p.add_argument("--gbs", type=int, default=1)
    p.add_argument("--steps", type=int, required=True)
    p.add_argument("--checkpoint-dir", default=None)
    p.add_argument("--checkpoint-every", type=int, default=100)
    p.add_argument("--resume", action="store_true",
                   help="continue from the latest COMPLETE checkpoint")
    p.add_argument("--schedule", default="gpipe", choices=("gpipe", "1f1b", "interleaved"))
    p.add_argument("--vpp", type=int, default=2,
                   help="virtual chunks per rank for --schedule interleaved")
    p.add_argument("--recompute", action="store_true")
    p.add_argument("--zero1", action="store_true")
    p.add_argument("--sp", action="store_true")
    p.add_argument("--clip-grad", type=float, default=None,
                   help="global grad-norm clip (TP/PP-correct)")
    p.add_argument("--lr", type=float, default=1e-4)
    p.add_argument("--lr-warmup", type=int, default=0,
                   help="linear warmup steps (then cosine decay to "
                        "--lr-min over --steps)")
    p.add_argument("--lr-min", type=float, default=0.0)
    p.add_argument("--data", default=None,
                   help="flat uint16 token file (default: synthetic)")
    p.add_argument("--eval-every", type=int, default=0,
                   help="eval-loss interval in steps (needs --data; the "
                        "last 2%% of the file is held out)")
    p.add_argument("--eval-batches", type=int, default=8)
    p.add_argument("--layer-partition", default=None)
    p.add_argument("--log-every", type=int, default=10)
    cfg_ns, _ = p.parse_known_args()
    if cfg_ns.config:
        import yaml

        with open(cfg_ns.config) as fh:
            doc = yaml.safe_load(fh) or {}
        unknown = [k for k in doc if not hasattr(cfg_ns, k.replace("-", "_"))]
        assert not unknown, f"unknown config keys: {unknown}"
        p.set_defaults(**{k.replace("-", "_"): v for k, v in doc.items()})
    args = p.parse_args()

    ctx = init_parallel(dp=args.dp, tp=args.tp, pp=args.pp)
    torch.manual_seed(1234)  # same init on every restart
    lp = ([int(x) for x in args.layer_partition.split(",")]
          if args.layer_partition else None)
    runner = PlanRunner(MODEL_SPECS[args.model], ctx, mbs=args.mbs,
                        gbs=args.gbs, layer_partition=lp, lr=args.lr,
                        schedule=args.schedule, recompute=args.recompute,
                        zero1=args.zero1, sp=args.sp, vpp=args.vpp,
                        clip_grad=args.clip_grad,
                        data_path=args.data)
    sched = None
    if args.lr_warmup:
        from metis_amd.runtime.lr import WarmupCosineLR

        sched = WarmupCosineLR(runner.optimizer, args.lr, args.lr_warmup,
                               args.steps, args.lr_min)

    start = 0
    if args.resume and args.checkpoint_dir:
        step = latest_complete_step(args.checkpoint_dir)
        if step >= 0:
            path = os.path.join(args.checkpoint_dir, f"step_{step}",
                                f"rank{ctx.rank}.pt")
            runner.load_checkpoint(path)
            start = step
            if ctx.rank == 0:
                print(f"resumed from {path} (step {step})")

    eval_loader = None
    if args.eval_every and ctx.pp > 1:
        # eval runs the pp==1 forward path only; don't skip silently
        if ctx.rank == 0:
            print("WARNING: --eval-every is not supported with pp > 1 "
                  "(eval loss will not be computed)")
    if args.eval_every and args.data and ctx.pp == 1:
        from metis_amd.data import TokenDataset, TokenLoader

        spec = MODEL_SPECS[args.model]
        eval_loader = TokenLoader(
            TokenDataset(args.data, spec.seq_length, split=(0.98, 1.0)),
            mbs=args.mbs, dp=ctx.dp, dp_rank=ctx.dp_rank,
            device=ctx.device or None)

    def eval_loss() -> float:
        eval_loader.load_state(0)
        total = 0.0
        n = min(args.eval_batches, eval_loader.microbatches_per_epoch)
        with torch.no_grad():
            for _ in range(n):
                tokens, labels = eval_loader.next_batch()
                total += float(runner.model(tokens, labels=labels))
        return total / n

    t0 = time.time()
    tokens_per_step = args.gbs * MODEL_SPECS[args.model].seq_length
    done = start
    try:
        for step in range(start, args.steps):
            done = _train_one(args, ctx, runner, sched, step, start, t0,
                              tokens_per_step, eval_loader, eval_loss)
    except Exception as e:
        # failure detection: a collective timeout / kernel fault lands
        # here — record a structured failure marker and (best effort,
        # the local state may still be intact) an emergency checkpoint
        # so --checkpoint-dir runs resume from the last good step
        import json as _json, traceback

        if args.checkpoint_dir:
            os.makedirs(args.checkpoint_dir, exist_ok=True)
            with open(os.path.join(args.checkpoint_dir,
                                   f"FAILED_rank{ctx.rank}.json"), "w") as fh:
                _json.dump({"rank": ctx.rank, "step": done,
                            "error": repr(e),
                            "traceback": traceback.format_exc()}, fh)
            try:
                save_step(runner, args.checkpoint_dir, done,
                          tag="emergency")
            except Exception:
                pass
        raise
    if args.checkpoint_dir and args.steps > start:
        save_step(runner, args.checkpoint_dir, args.steps)
    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
