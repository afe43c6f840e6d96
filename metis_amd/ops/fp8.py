"""fp8 (OCP e4m3) GEMM path with per-tensor scaling.

MI355X's fp8 MFMA rate is ~2x bf16, so running the projection/FFN GEMMs
in fp8 is the largest remaining perf lever after fusion. v1 scope:
forward-only fp8 (quantize x and w per-tensor to e4m3, hipBLASLt GEMM
with dequant scales, bf16 out); backward stays bf16. Opt-in until the
numerics and speed are validated on a box (round 2) — the gated GPU
test compares against the bf16 GEMM within fp8 quantization tolerance.
"""

from __future__ import annotations

import torch

from metis_amd import ops as _ops

_E4M3_MAX = 448.0


def quantize_e4m3(t: torch.Tensor):
    """Per-tensor symmetric quantization: returns (fp8 tensor, dequant
    scale as a 1-element fp32 tensor on t's device)."""
    amax = t.detach().abs().amax().clamp(min=1e-12).float()
    q = (t.float() * (_E4M3_MAX / amax)).clamp(-_E4M3_MAX, _E4M3_MAX)
    return q.to(torch.float8_e4m3fn), (amax / _E4M3_MAX).reshape(1)


def fp8_matmul(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """y = x @ w^T with both operands quantized to e4m3 per tensor;
    x [.., K] (bf16), w [N, K] (bf16) -> bf16 [.., N]."""
    lead = x.shape[:-1]
    x2 = x.reshape(-1, x.size(-1))
    if x.is_cuda and x.dtype == torch.bfloat16:
        ext = _ops.require_extension()
        x8, sx = quantize_e4m3(x2)
        w8, sw = quantize_e4m3(w)
        y = ext.lt_fp8_matmul(x8, w8, sx, sw)
    else:
        # CPU reference: matmul of the quantize-dequantized operands —
        # bit-faithful to what the fp8 GEMM computes
        x8, sx = quantize_e4m3(x2)
        w8, sw = quantize_e4m3(w)
        y = (x8.float() * sx) @ (w8.float() * sw).t()
        y = y.to(x.dtype)
    return y.reshape(*lead, w.size(0))
