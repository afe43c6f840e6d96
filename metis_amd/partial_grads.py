"""Cross-rank gradient sums for replicated parameters with partial grads.

Some parameters are REPLICATED across a process group but each rank
backprops only a slice of the loss terms that touch them, so the local
``p.grad`` is a partial sum that must be all-reduced across the group:

- the MoE router (every rank routes all tokens, but only its local
  experts' gate terms reach the router in backward) — EP group;
- SP-region params (LayerNorm/RMSNorm weights, row-parallel biases,
  embeddings) that see only this rank's sequence slice — TP group.

The reference (SamsungLabs/Metis) has no runtime, so no equivalent.

Protocol
--------
Models tag such parameters once at construction with ``mark_partial(p,
group)``. That registers an *immediate-mode* post-accumulate-grad hook:
the partial sum runs inside backward, which is correct when there is
exactly ONE backward per optimizer step (standalone use: tests,
profiler single-iteration timing).

Any driver that ACCUMULATES gradients over microbatches (PlanRunner)
must instead call ``defer_partial(params)`` once — the immediate hooks
then no-op — and perform the sum itself exactly once per step on the
fully-accumulated grad:

- ``GradBucketSync`` does it inside its own armed-final-microbatch hook,
  right BEFORE copying the grad into the flat fp32 buffer the DP
  all-reduce / clipping / FusedAdamW consume (ordering by construction,
  not by hook-registration order);
- the dp==1 runner path calls ``sync_partial_grads(params)`` after the
  last backward, before ``gather_grads``.

This fixes two round-1 advisor findings: accumulated grads being
re-summed on every microbatch backward, and the DP bucket copy racing
ahead of a lazily-registered EP sum.
"""

from __future__ import annotations

from typing import Iterable

import torch
import torch.distributed as dist

GROUP_ATTR = "_metis_partial_group"
DEFER_ATTR = "_metis_partial_defer"


def _immediate_hook(p: torch.nn.Parameter) -> None:
    if getattr(p, DEFER_ATTR, False):
        return
    if p.grad is not None:
        dist.all_reduce(p.grad, group=getattr(p, GROUP_ATTR))


def mark_partial(p: torch.nn.Parameter, group) -> None:
    """Tag ``p`` as replicated-with-partial-grads over ``group`` and
    install the immediate-mode sum hook. No-op for world size <= 1."""
    if group is None or dist.get_world_size(group) <= 1:
        return
    if hasattr(p, GROUP_ATTR):
        return  # already marked (e.g. shared/tied parameter)
    setattr(p, GROUP_ATTR, group)
    p.register_post_accumulate_grad_hook(_immediate_hook)


def defer_partial(params: Iterable[torch.nn.Parameter]) -> None:
    """Disable the immediate hooks; the caller owns the once-per-step sum."""
    for p in params:
        if hasattr(p, GROUP_ATTR):
            setattr(p, DEFER_ATTR, True)


def partial_group(p: torch.nn.Parameter):
    """The group ``p``'s grad must be summed over, or None."""
    return getattr(p, GROUP_ATTR, None)


def sync_partial_grads(params: Iterable[torch.nn.Parameter]) -> None:
    """One-shot sum of every tagged param's accumulated ``p.grad``."""
    for p in params:
        group = getattr(p, GROUP_ATTR, None)
        if group is not None and p.grad is not None:
            dist.all_reduce(p.grad, group=group)
