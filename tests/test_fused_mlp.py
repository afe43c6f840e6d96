"""Epilogue-fused MLP (ops/mlp.py): CPU-fallback parity with the eager
block path, and a GPU numerics test gated on METIS_EXPERIMENTAL (the
hipBLASLt epilogue path is validated on a GPU box before being enabled
by default — round-2 item)."""

import os

import pytest
import torch

from metis_amd.models.gpt import GPTModel, GPTModelSpec
from metis_amd.ops.mlp import fused_mlp

SPEC = GPTModelSpec("tiny", hidden_size=64, num_layers=2, num_heads=4,
                    vocab_size=512, seq_length=32)


def _models(monkeypatch, dtype=torch.float32):
    tok = torch.randint(0, 512, (2, 32))
    lab = torch.roll(tok, -1, 1)
    monkeypatch.delenv("METIS_FC1_EPILOGUE", raising=False)
    torch.manual_seed(0)
    plain = GPTModel(SPEC, dtype=dtype)
    monkeypatch.setenv("METIS_FC1_EPILOGUE", "1")
    torch.manual_seed(0)
    fused = GPTModel(SPEC, dtype=dtype)
    return plain, fused, tok, lab


def test_fused_mlp_cpu_matches_block(monkeypatch):
    plain, fused, tok, lab = _models(monkeypatch)
    l1 = plain(tok, labels=lab)
    l2 = fused(tok, labels=lab)
    assert torch.allclose(l1, l2, atol=1e-6)
    l1.backward()
    l2.backward()
    for (n, p1), (_, p2) in zip(plain.named_parameters(),
                                fused.named_parameters()):
        assert torch.allclose(p1.grad, p2.grad, atol=1e-5), n


def test_fused_mlp_autograd_shapes():
    x = torch.randn(8, 16, requires_grad=True)
    w1 = torch.randn(32, 16, requires_grad=True)
    b1 = torch.randn(32, requires_grad=True)
    w2 = torch.randn(16, 32, requires_grad=True)
    y = fused_mlp(x, w1, b1, w2)
    assert y.shape == (8, 16)
    y.sum().backward()
    assert x.grad.shape == x.shape and w1.grad.shape == w1.shape
    assert b1.grad.shape == b1.shape and w2.grad.shape == w2.shape


@pytest.mark.gpu
@pytest.mark.skipif(os.environ.get("METIS_EXPERIMENTAL") != "1",
                    reason="hipBLASLt epilogue path: ROCm 7.2 ships no "
                           "GELU_AUX_BIAS algorithm on gfx950 (retired, "
                           "see ops/mlp.py)")
def test_fused_mlp_gpu_matches_reference():
    import metis_amd._hip_ops as ext

    try:
        ext.lt_fc1_forward(
            torch.randn(64, 64, device="cuda", dtype=torch.bfloat16),
            torch.randn(64, 64, device="cuda", dtype=torch.bfloat16),
            torch.randn(64, device="cuda", dtype=torch.bfloat16))
    except RuntimeError as e:
        if "no algorithm" in str(e):
            pytest.skip("hipBLASLt build has no GELU_AUX_BIAS algorithm")
        raise

    torch.manual_seed(0)
    M, K, F, H = 256, 128, 512, 128
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    w1 = torch.randn(F, K, device="cuda", dtype=torch.bfloat16) * 0.05
    b1 = torch.randn(F, device="cuda", dtype=torch.bfloat16)
    w2 = torch.randn(H, F, device="cuda", dtype=torch.bfloat16) * 0.05

    y, aux = ext.lt_fc1_forward(x, w1, b1)
    ref_pre = (x.float() @ w1.float().t() + b1.float())
    ref_y = torch.nn.functional.gelu(ref_pre, approximate="tanh")
    assert torch.allclose(aux.float(), ref_pre, atol=0.15, rtol=0.05)
    assert torch.allclose(y.float(), ref_y, atol=0.15, rtol=0.05)

    dy = torch.randn(M, H, device="cuda", dtype=torch.bfloat16)
    dpre, db1 = ext.lt_matmul_dgelu_bgrad(dy, w2, aux)
    ref_h = ref_pre.clone().requires_grad_(True)
    ref_act = torch.nn.functional.gelu(ref_h, approximate="tanh")
    (ref_act * (dy.float() @ w2.float())).sum().backward()
    assert torch.allclose(dpre.float(), ref_h.grad, atol=0.2, rtol=0.05)
    assert torch.allclose(db1.float(), ref_h.grad.sum(0), atol=2.0, rtol=0.05)
