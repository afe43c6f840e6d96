"""Fused AdamW with fp32 master weights over bf16 model parameters.

One flat fp32 master buffer + flat m/v moments per parameter group; the
fused gfx950 kernel updates master, moments and the bf16 working copy in
one pass. On CPU the same math runs in eager PyTorch (tests, gloo runs).

ZeRO-1 (``shard_group``): master/m/v are sharded 1/ws per rank of the
group (the DP group in practice — 12 B/param of optimizer state becomes
12/ws); each rank updates its shard and an all-gather of the updated
weights replaces the post-step scatter. Gradients still arrive via the
ordinary bucketed DP all-reduce (same wire bytes as reduce-scatter +
the weight all-gather on a ring).
"""

from __future__ import annotations

from typing import Iterable, List, Optional

import torch
import torch.distributed as dist

from metis_amd import ops as _ops


class FusedAdamW:
    def __init__(
        self,
        params: Iterable[torch.nn.Parameter],
        lr: float = 1e-4,
        betas=(0.9, 0.95),
        eps: float = 1e-8,
        weight_decay: float = 0.1,
        shard_group: Optional[object] = None,
    ) -> None:
        self.params: List[torch.nn.Parameter] = [p for p in params if p.requires_grad]
        if not self.params:
            raise ValueError("no trainable parameters")
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0
        self.shard_group = shard_group
        ws = dist.get_world_size(shard_group) if shard_group is not None else 1
        self._shard_ws = ws
        self._shard_rank = dist.get_rank(shard_group) if shard_group is not None else 0

        dev = self.params[0].device
        # pad the flat buffers to a multiple of 4 (16 B kernel path) x ws
        total = sum(p.numel() for p in self.params)
        self._n = (total + 4 * ws - 1) // (4 * ws) * (4 * ws)
        self._shard_n = self._n // ws
        self._shard_off = self._shard_rank * self._shard_n
        self._grad_flat = torch.zeros(self._n, dtype=torch.float32, device=dev)

        full = torch.zeros(self._n, dtype=torch.float32, device=dev)
        offset = 0
        self._slices = []
        for p in self.params:
            n = p.numel()
            full[offset:offset + n].copy_(p.detach().reshape(-1).float())
            self._slices.append((offset, n))
            offset += n
        if ws > 1:
            self.master = full[self._shard_off:self._shard_off + self._shard_n].clone()
        else:
            self.master = full
        self.m = torch.zeros_like(self.master)
        self.v = torch.zeros_like(self.master)

        # On GPU with all-bf16 params, re-bind every parameter to a view of
        # one flat bf16 buffer: the fused kernel then writes the bf16 copy
        # directly and no per-parameter scatter is needed after the step.
        self._model_flat = None
        if dev.type == "cuda" and all(p.dtype == torch.bfloat16 for p in self.params):
            flat = torch.zeros(self._n, dtype=torch.bfloat16, device=dev)
            with torch.no_grad():
                for p, (off, n) in zip(self.params, self._slices):
                    flat[off:off + n].copy_(p.detach().reshape(-1))
                    p.data = flat[off:off + n].view_as(p)
            self._model_flat = flat

    def zero_grad(self, set_to_none: bool = True) -> None:
        for p in self.params:
            if set_to_none:
                p.grad = None
            elif p.grad is not None:
                p.grad.zero_()

    @torch.no_grad()
    def _gather_grads(self) -> torch.Tensor:
        for p, (off, n) in zip(self.params, self._slices):
            g = p.grad
            if g is None:
                self._grad_flat[off:off + n].zero_()
            else:
                self._grad_flat[off:off + n].copy_(g.reshape(-1).float())
        return self._grad_flat

    @property
    def grad_flat(self) -> torch.Tensor:
        """Flat fp32 gradient buffer (for one-shot DP all-reduce)."""
        return self._grad_flat

    @torch.no_grad()
    def gather_grads(self) -> torch.Tensor:
        return self._gather_grads()

    def _update(self, grads: torch.Tensor, model: torch.Tensor,
                grad_scale: float) -> None:
        """AdamW update of (master, m, v) from ``grads`` (same length)."""
        if self.master.is_cuda:
            ext = _ops.require_extension()
            ext.adamw_step(
                self.master, model, grads, self.m, self.v,
                self.lr, self.beta1, self.beta2, self.eps,
                self.weight_decay, self.step_count, grad_scale,
            )
        else:
            g = grads * grad_scale
            self.m.mul_(self.beta1).add_(g, alpha=1 - self.beta1)
            self.v.mul_(self.beta2).addcmul_(g, g, value=1 - self.beta2)
            mhat = self.m / (1 - self.beta1 ** self.step_count)
            vhat = self.v / (1 - self.beta2 ** self.step_count)
            self.master.add_(
                -self.lr * (mhat / (vhat.sqrt() + self.eps)
                            + self.weight_decay * self.master)
            )

    def _gather_weight_shards(self) -> torch.Tensor:
        """All-gather the updated fp32 master shards into one full buffer."""
        full = torch.empty(self._n, dtype=torch.float32,
                           device=self.master.device)
        try:
            dist.all_gather_into_tensor(full, self.master, group=self.shard_group)
        except (RuntimeError, ValueError):
            parts = [torch.empty_like(self.master) for _ in range(self._shard_ws)]
            dist.all_gather(parts, self.master, group=self.shard_group)
            full = torch.cat(parts)
        return full

    @torch.no_grad()
    def step(self, grad_scale: float = 1.0, pre_gathered: bool = False) -> None:
        """Apply one AdamW step. With ``pre_gathered=True`` the caller has
        already filled (and possibly all-reduced) ``grad_flat``."""
        self.step_count += 1
        grads = self._grad_flat if pre_gathered else self._gather_grads()

        if self._shard_ws > 1:  # ZeRO-1: update the local shard, gather weights
            off, n = self._shard_off, self._shard_n
            model = (self._model_flat[off:off + n]
                     if self._model_flat is not None else torch.Tensor())
            self._update(grads[off:off + n], model, grad_scale)
            if self._model_flat is not None:
                # bf16 weight all-gather straight into the flat param buffer
                shard = self._model_flat[off:off + n].clone()
                dist.all_gather_into_tensor(self._model_flat, shard,
                                            group=self.shard_group)
                return
            full = self._gather_weight_shards()
            for p, (o, pn) in zip(self.params, self._slices):
                p.copy_(full[o:o + pn].view_as(p).to(p.dtype))
            return

        model = self._model_flat if self._model_flat is not None else torch.Tensor()
        self._update(grads, model, grad_scale)
        if self._model_flat is not None:
            return
        # scatter master back into the (possibly bf16) working params
        for p, (off, n) in zip(self.params, self._slices):
            p.copy_(self.master[off:off + n].view_as(p).to(p.dtype))

    def state_dict(self) -> dict:
        return {
            "step": self.step_count,
            "master": self.master,
            "m": self.m,
            "v": self.v,
        }

    def load_state_dict(self, state: dict) -> None:
        self.step_count = state["step"]
        self.master.copy_(state["master"])
        self.m.copy_(state["m"])
        self.v.copy_(state["v"])
