"""Flash attention wrapper (causal, GQA).

Forward runs the hand-written gfx950 MFMA kernel; backward currently
falls back to PyTorch SDPA recompute (the hand-written backward kernel is
the next optimization stage — until it lands, training graphs that need
grads route through SDPA, and the custom kernel serves no-grad forward
passes, decode and the profiler's forward timings).
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn.functional as F

from metis_amd import ops as _ops


def _supported(q: torch.Tensor) -> bool:
    d = q.size(-1)
    s = q.size(2)
    return (
        q.is_cuda
        and q.dtype == torch.bfloat16
        and d % 16 == 0
        and d <= 128
        and s % 64 == 0
    )


def flash_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    causal: bool = True,
    scale: Optional[float] = None,
) -> torch.Tensor:
    """q [B,H,S,D], k/v [B,Hkv,S,D] -> [B,H,S,D]."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.size(-1))
    needs_grad = torch.is_grad_enabled() and (
        q.requires_grad or k.requires_grad or v.requires_grad
    )
    if causal and not needs_grad and _supported(q):
        ext = _ops.require_extension()
        o, _lse = ext.attn_fwd(q, k, v, scale)
        return o
    return F.scaled_dot_product_attention(
        q, k, v, is_causal=causal, scale=scale, enable_gqa=q.size(1) != k.size(1)
    )
