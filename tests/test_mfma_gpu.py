"""GPU tests for the MFMA kernels (GEMM, flash attention) vs fp32 torch."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from metis_amd.ops import require_extension
else:
    pytest.skip("requires MI355X GPU", allow_module_level=True)


@pytest.fixture(scope="module")
def ext():
    return require_extension()


def test_mfma_tile_layout_probe(ext):
    """Asymmetric-input check of the assumed 16x16x32 fragment layout
    (guide G9: symmetric inputs miss operand/output transposes)."""
    torch.manual_seed(0)
    a = torch.randn(16, 32, device="cuda").to(torch.bfloat16)
    b = (torch.arange(32 * 16, device="cuda").reshape(32, 16).float() % 7 - 3)
    b = b.to(torch.bfloat16)
    d = ext.mfma_tile_probe(a, b)
    ref = a.float() @ b.float()
    assert torch.allclose(d, ref, atol=0.1, rtol=0.02), (d - ref).abs().max()


@pytest.mark.parametrize("m,n,k", [(128, 128, 64), (256, 384, 128), (512, 512, 2560)])
def test_gemm_bf16_matches_torch(ext, m, n, k):
    torch.manual_seed(1)
    a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
    c = ext.gemm_bf16(a, w, None, 0)
    ref = a.float() @ w.float().T
    err = (c.float() - ref).abs().max() / ref.abs().max()
    assert err < 0.02, err


def test_gemm_bias_gelu_epilogue(ext):
    torch.manual_seed(2)
    m, n, k = 256, 256, 128
    a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
    bias = torch.randn(n, device="cuda", dtype=torch.float32)

    c1 = ext.gemm_bf16(a, w, bias, 1)
    ref1 = a.float() @ w.float().T + bias
    assert (c1.float() - ref1).abs().max() / ref1.abs().max() < 0.02

    c2 = ext.gemm_bf16(a, w, bias, 2)
    ref2 = torch.nn.functional.gelu(ref1, approximate="tanh")
    assert (c2.float() - ref2).abs().max() / (ref2.abs().max() + 1) < 0.02


@pytest.mark.parametrize(
    "b,h,hkv,s,d",
    [(2, 4, 4, 256, 64), (1, 8, 2, 512, 128), (2, 4, 4, 256, 80)],
)
def test_attn_fwd_matches_fp32_sdpa(ext, b, h, hkv, s, d):
    torch.manual_seed(3)
    q = torch.randn(b, h, s, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(b, hkv, s, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(b, hkv, s, d, device="cuda", dtype=torch.bfloat16)
    scale = 1.0 / math.sqrt(d)

    o, lse = ext.attn_fwd(q, k, v, scale)

    ref = torch.nn.functional.scaled_dot_product_attention(
        q.float(), k.float(), v.float(), is_causal=True, scale=scale,
        enable_gqa=h != hkv,
    )
    err = (o.float() - ref).abs().max()
    assert err < 3e-2, err
    assert torch.isfinite(lse).all()


def test_attn_fwd_lse_values(ext):
    """LSE must equal logsumexp of the causal scaled scores."""
    torch.manual_seed(4)
    b, h, s, d = 1, 2, 128, 64
    q = torch.randn(b, h, s, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(b, h, s, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(b, h, s, d, device="cuda", dtype=torch.bfloat16)
    scale = 1.0 / math.sqrt(d)
    _o, lse = ext.attn_fwd(q, k, v, scale)

    scores = q.float() @ k.float().transpose(-1, -2) * scale
    mask = torch.tril(torch.ones(s, s, device="cuda", dtype=torch.bool))
    scores = scores.masked_fill(~mask, float("-inf"))
    ref_lse = torch.logsumexp(scores, dim=-1)
    assert torch.allclose(lse, ref_lse, atol=5e-2), (lse - ref_lse).abs().max()


def test_flash_attention_wrapper_dispatch():
    from metis_amd.ops.attention import flash_attention

    q = torch.randn(1, 4, 256, 64, device="cuda", dtype=torch.bfloat16)
    k, v = q.clone(), q.clone()
    with torch.no_grad():
        o = flash_attention(q, k, v)
    ref = torch.nn.functional.scaled_dot_product_attention(
        q.float(), k.float(), v.float(), is_causal=True,
        scale=1.0 / math.sqrt(64),
    )
    assert (o.float() - ref).abs().max() < 3e-2


@pytest.mark.gpu
def test_gemm8_matches_fp32():
    """256^2-tile 2-buffer glds GEMM (gemm8.hip) vs fp32 torch."""
    import metis_amd._hip_ops as ext

    torch.manual_seed(0)
    for (m, n, k) in ((256, 256, 64), (512, 256, 128), (512, 512, 192)):
        x = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
        y = ext.gemm8_bf16(x, w)
        ref = x.float() @ w.float().t()
        err = (y.float() - ref).abs().max() / ref.abs().max()
        assert err < 2e-2, (m, n, k, float(err))
