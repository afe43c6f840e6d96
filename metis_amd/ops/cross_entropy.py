"""Fused cross entropy over bf16 logits (gfx950 kernel, CPU fallback).

Avoids materializing the fp32 [tokens, vocab] logits copy; the backward
writes bf16 dlogits directly.
"""

from __future__ import annotations

import torch

from metis_amd import ops as _ops


class _FusedCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels):
        ext = _ops.require_extension()
        loss, lse = ext.cross_entropy_fwd(logits, labels)
        ctx.save_for_backward(logits, labels, lse)
        return loss.mean()

    @staticmethod
    def backward(ctx, grad_out):
        ext = _ops.require_extension()
        logits, labels, lse = ctx.saved_tensors
        scale = (grad_out.float() / logits.size(0)).reshape(1)
        dlogits = ext.cross_entropy_bwd(logits, labels, lse, scale)
        return dlogits, None


def cross_entropy(logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """logits [N, V] (bf16 on GPU), labels int64 [N] -> mean loss."""
    if logits.is_cuda and logits.dtype == torch.bfloat16 and logits.size(1) % 8 == 0:
        return _FusedCE.apply(logits.contiguous(), labels)
    return torch.nn.functional.cross_entropy(logits.float(), labels)
