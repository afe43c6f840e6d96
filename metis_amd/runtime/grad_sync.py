"""Bucketed, overlapped DP gradient all-reduce.

Gradients land in the optimizer's flat fp32 buffer as autograd produces
them (post-accumulate-grad hooks, reverse parameter order = backward
order); each bucket launches its RCCL all-reduce as soon as it is
complete, overlapping communication with the rest of the backward pass.
On xGMI the ring all-reduce is single-link bound, so overlap — not raw
bus bandwidth — is what keeps DP scaling near-linear.

With gradient accumulation (num_microbatches > 1) the sync only arms for
the final microbatch's backward (DDP no_sync pattern).
"""

from __future__ import annotations

from typing import List

import torch
import torch.distributed as dist

from metis_amd.ops.adamw import FusedAdamW
from metis_amd.partial_grads import defer_partial, partial_group


class GradBucketSync:
    def __init__(
        self,
        optimizer: FusedAdamW,
        dp_group,
        dp_size: int,
        bucket_mb: float = 125.0,
    ) -> None:
        self.opt = optimizer
        self.group = dp_group
        self.dp = dp_size
        self._handles: List = []

        # buckets = contiguous ranges of the optimizer's flat grad buffer,
        # cut in reverse parameter order (the order backward completes)
        bucket_bytes = int(bucket_mb * 1024 * 1024)
        self._param_bucket = {}
        self._buckets = []  # (start, end, n_params)
        current: List[int] = []
        cur_start = None
        cur_bytes = 0
        params = list(enumerate(self.opt.params))
        for idx, p in reversed(params):
            off, n = self.opt._slices[idx]
            if cur_start is None:
                cur_start = off
            current.append(idx)
            cur_bytes += n * 4
            cur_end = off + n
            if cur_bytes >= bucket_bytes:
                self._close_bucket(current)
                current, cur_start, cur_bytes = [], None, 0
        if current:
            self._close_bucket(current)

        self._pending = [0] * len(self._buckets)
        self._armed = [False] * len(self.opt.params)
        self._step_active = False
        # replicated-with-partial-grads params (MoE router over EP, SP
        # norm/bias params over TP): their cross-rank sum moves INTO our
        # armed hook — before the bucket copy, once per step, on the
        # fully-accumulated grad (runtime.partial_grads protocol)
        defer_partial(self.opt.params)
        for idx, p in enumerate(self.opt.params):
            p.register_post_accumulate_grad_hook(self._make_hook(idx))

    def _close_bucket(self, param_idxs: List[int]) -> None:
        bucket_id = len(self._buckets)
        starts = [self.opt._slices[i][0] for i in param_idxs]
        ends = [self.opt._slices[i][0] + self.opt._slices[i][1] for i in param_idxs]
        self._buckets.append((min(starts), max(ends), len(param_idxs)))
        for i in param_idxs:
            self._param_bucket[i] = bucket_id

    def _make_hook(self, idx: int):
        def hook(param: torch.nn.Parameter) -> None:
            if not self._armed[idx]:
                return
            self._armed[idx] = False   # fire once per step per param
            pg = partial_group(param)
            if pg is not None and param.grad is not None:
                # partial-grad sum (EP/TP) before the flat-buffer copy the
                # DP all-reduce, clipping and FusedAdamW consume
                dist.all_reduce(param.grad, group=pg)
            off, n = self.opt._slices[idx]
            flat = self.opt.grad_flat
            g = param.grad
            if g is None:
                flat[off:off + n].zero_()
            else:
                flat[off:off + n].copy_(g.reshape(-1), non_blocking=True)
            b = self._param_bucket[idx]
            self._pending[b] -= 1
            if self._pending[b] == 0 and self.group is not None:
                start, end, _ = self._buckets[b]
                handle = dist.all_reduce(
                    flat[start:end], group=self.group, async_op=True
                )
                self._handles.append((handle, start, end))

        return hook

    def _begin_step(self) -> None:
        if self._step_active:
            return
        self._step_active = True
        self._handles = []
        for b, (_s, _e, n) in enumerate(self._buckets):
            self._pending[b] = n

    def arm(self) -> None:
        """Call before the FINAL backward that touches EVERY param (the
        last microbatch in GPipe/1F1B/no-pipeline)."""
        self._begin_step()
        self._armed = [True] * len(self.opt.params)

    def arm_params(self, idxs) -> None:
        """Arm a subset right before ITS final backward — the interleaved
        schedule finishes different chunks' gradients at different slots,
        so chunks arm separately (a bucket spanning two chunks launches
        once both chunks' members have fired)."""
        self._begin_step()
        for i in idxs:
            self._armed[i] = True

    def finish(self) -> None:
        """Wait for outstanding reduces and average; leaves grad_flat ready
        for FusedAdamW.step(pre_gathered=True)."""
        self._step_active = False
        if any(self._pending):
            # a parameter produced no gradient: ranks would disagree on the
            # bucket schedule — fail loudly rather than deadlock RCCL
            raise RuntimeError(
                f"grad buckets incomplete after backward: {self._pending}"
            )
        flat = self.opt.grad_flat
        for handle, start, end in self._handles:
            handle.wait()
            flat[start:end].div_(self.dp)
        self._handles = []
