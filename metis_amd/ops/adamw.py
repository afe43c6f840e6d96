"""Fused AdamW with fp32 master weights over bf16 model parameters.

One flat fp32 master buffer + flat m/v moments per parameter group; the
fused gfx950 kernel updates master, moments and the bf16 working copy in
one pass. On CPU the same math runs in eager PyTorch (tests, gloo runs).
"""

from __future__ import annotations

from typing import Iterable, List

import torch

from metis_amd import ops as _ops


class FusedAdamW:
    def __init__(
        self,
        params: Iterable[torch.nn.Parameter],
        lr: float = 1e-4,
        betas=(0.9, 0.95),
        eps: float = 1e-8,
        weight_decay: float = 0.1,
    ) -> None:
        self.params: List[torch.nn.Parameter] = [p for p in params if p.requires_grad]
        if not self.params:
            raise ValueError("no trainable parameters")
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0

        dev = self.params[0].device
        # pad the flat buffers to a multiple of 4 for the 16 B kernel path
        total = sum(p.numel() for p in self.params)
        self._n = (total + 3) // 4 * 4
        self.master = torch.zeros(self._n, dtype=torch.float32, device=dev)
        self.m = torch.zeros_like(self.master)
        self.v = torch.zeros_like(self.master)
        self._grad_flat = torch.zeros_like(self.master)

        offset = 0
        self._slices = []
        for p in self.params:
            n = p.numel()
            self.master[offset:offset + n].copy_(p.detach().reshape(-1).float())
            self._slices.append((offset, n))
            offset += n

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

    @torch.no_grad()
    def step(self, grad_scale: float = 1.0, pre_gathered: bool = False) -> None:
        """Apply one AdamW step. With ``pre_gathered=True`` the caller has
        already filled (and possibly all-reduced) ``grad_flat``."""
        self.step_count += 1
        grads = self._grad_flat if pre_gathered else self._gather_grads()

        if self.master.is_cuda:
            ext = _ops.require_extension()
            model = self._model_flat if self._model_flat is not None else torch.Tensor()
            ext.adamw_step(
                self.master, model, grads, self.m, self.v,
                self.lr, self.beta1, self.beta2, self.eps,
                self.weight_decay, self.step_count, grad_scale,
            )
            if self._model_flat is not None:
                return
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
