"""RMSNorm, RoPE and SwiGLU wrappers (gfx950 kernels, CPU fallbacks)."""

from __future__ import annotations

from typing import Tuple

import torch
import torch.nn as nn

from metis_amd import ops as _ops


class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        ext = _ops.require_extension()
        x = x.contiguous()
        y, rstd = ext.rmsnorm_fwd(x, weight, eps)
        ctx.save_for_backward(x, weight, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _ops.require_extension()
        x, weight, rstd = ctx.saved_tensors
        dx, dgamma = ext.rmsnorm_bwd(dy, x, weight, rstd)
        return dx, dgamma.to(weight.dtype), None


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    if x.is_cuda and x.dtype == torch.bfloat16:
        return _RMSNormFn.apply(x, weight, eps)
    xf = x.float()
    y = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (y * weight.float()).to(x.dtype)


class RMSNorm(nn.Module):
    def __init__(self, hidden_size: int, eps: float = 1e-5):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden_size, dtype=torch.float32))
        self.eps = eps

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return rms_norm(x, self.weight, self.eps)


# --- RoPE ----------------------------------------------------------------
_rope_cache = {}


def rope_tables(
    seq_len: int, head_dim: int, base: float, device, dtype=torch.float32
) -> Tuple[torch.Tensor, torch.Tensor]:
    """[S, D/2] cos/sin tables, cached per (S, D, base, device)."""
    key = (seq_len, head_dim, base, str(device))
    if key not in _rope_cache:
        inv = 1.0 / (base ** (torch.arange(0, head_dim, 2, device=device,
                                           dtype=torch.float32) / head_dim))
        pos = torch.arange(seq_len, device=device, dtype=torch.float32)
        freqs = torch.outer(pos, inv)
        _rope_cache[key] = (freqs.cos().to(dtype), freqs.sin().to(dtype))
    return _rope_cache[key]


class _RoPEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos_t, sin_t):
        ext = _ops.require_extension()
        ctx.save_for_backward(cos_t, sin_t)
        return ext.rope_apply(x, cos_t, sin_t, False)

    @staticmethod
    def backward(ctx, dy):
        ext = _ops.require_extension()
        cos_t, sin_t = ctx.saved_tensors
        return ext.rope_apply(dy.contiguous(), cos_t, sin_t, True), None, None


def _rope_ref(x: torch.Tensor, cos_t: torch.Tensor, sin_t: torch.Tensor) -> torch.Tensor:
    d = x.size(-1)
    s = x.size(2)
    x1, x2 = x[..., : d // 2], x[..., d // 2:]
    c = cos_t[:s].view(1, 1, s, d // 2).to(x.dtype)
    sn = sin_t[:s].view(1, 1, s, d // 2).to(x.dtype)
    return torch.cat((x1 * c - x2 * sn, x2 * c + x1 * sn), dim=-1)


def apply_rope(x: torch.Tensor, base: float = 500000.0,
               pos_offset: int = 0) -> torch.Tensor:
    """x [B, H, S, D] -> rotated x (llama half-rotation convention);
    ``pos_offset`` shifts the absolute positions (KV-cache decode)."""
    s, d = x.size(2), x.size(3)
    cos_t, sin_t = rope_tables(pos_offset + s, d, base, x.device)
    cos_t, sin_t = cos_t[pos_offset:], sin_t[pos_offset:]
    if x.is_cuda and x.dtype == torch.bfloat16:
        return _RoPEFn.apply(x.contiguous(), cos_t.contiguous(),
                             sin_t.contiguous())
    return _rope_ref(x, cos_t, sin_t)


# --- SwiGLU ---------------------------------------------------------------
class _SwiGLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b):
        ext = _ops.require_extension()
        ctx.save_for_backward(a, b)
        return ext.swiglu_fwd(a, b)

    @staticmethod
    def backward(ctx, dy):
        ext = _ops.require_extension()
        a, b = ctx.saved_tensors
        da, db = ext.swiglu_bwd(dy, a, b)
        return da, db


def swiglu(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """silu(a) * b (fused on GPU bf16)."""
    if a.is_cuda and a.dtype == torch.bfloat16:
        return _SwiGLUFn.apply(a, b)
    return torch.nn.functional.silu(a) * b


def apply_rope_rows(x: torch.Tensor, base: float,
                    positions: torch.Tensor) -> torch.Tensor:
    """Decode-path RoPE with a PER-ROW position: x [b, h, 1, d],
    positions [b] (ragged batched decoding)."""
    d = x.size(-1)
    cos_t, sin_t = rope_tables(int(positions.max()) + 1, d, base, x.device)
    c = cos_t[positions].view(-1, 1, 1, d // 2).to(x.dtype)
    sn = sin_t[positions].view(-1, 1, 1, d // 2).to(x.dtype)
    x1, x2 = x[..., : d // 2], x[..., d // 2:]
    return torch.cat((x1 * c - x2 * sn, x2 * c + x1 * sn), dim=-1)
