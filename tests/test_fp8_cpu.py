"""fp8 quantization helpers (CPU reference path)."""

import torch

from metis_amd.ops.fp8 import _E4M3_MAX, fp8_matmul, quantize_e4m3


def test_quantize_roundtrip_error_bounded():
    torch.manual_seed(0)
    t = torch.randn(1000)
    q, scale = quantize_e4m3(t)
    back = q.float() * scale
    rel = (back - t).abs().max() / t.abs().max()
    assert rel < 0.07  # e4m3 has 3 mantissa bits near amax
    assert float(q.float().abs().max()) <= _E4M3_MAX


def test_fp8_matmul_cpu_reference():
    torch.manual_seed(0)
    x = torch.randn(8, 64)
    w = torch.randn(16, 64) * 0.1
    y = fp8_matmul(x, w)
    ref = x @ w.t()
    err = (y - ref).abs().mean() / ref.abs().mean()
    assert y.shape == (8, 16)
    assert err < 0.05
