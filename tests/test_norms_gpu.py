"""GPU numerics: RMSNorm / RoPE / SwiGLU kernels vs fp32 references."""

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from metis_amd.ops import require_extension
    from metis_amd.ops.norms import apply_rope, rms_norm, swiglu
else:
    pytest.skip("requires MI355X GPU", allow_module_level=True)


@pytest.fixture(scope="module")
def ext():
    return require_extension()


@pytest.mark.parametrize("rows,hidden", [(128, 2048), (64, 4096)])
def test_rmsnorm_fwd_bwd(ext, rows, hidden):
    torch.manual_seed(0)
    x = torch.randn(rows, hidden, device="cuda", dtype=torch.bfloat16)
    w = (torch.rand(hidden, device="cuda") + 0.5).float()
    dy = torch.randn_like(x)

    y, rstd = ext.rmsnorm_fwd(x, w, 1e-5)
    xf = x.float().requires_grad_(True)
    wf = w.clone().requires_grad_(True)
    ref = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-5) * wf
    assert torch.allclose(y.float(), ref.detach(), atol=3e-2, rtol=3e-2)

    dx, dgamma = ext.rmsnorm_bwd(dy, x, w, rstd)
    ref.backward(dy.float())
    assert torch.allclose(dx.float(), xf.grad, atol=6e-2, rtol=6e-2)
    assert torch.allclose(dgamma, wf.grad, atol=0.5, rtol=2e-2)


def test_rope_gpu_matches_cpu_reference():
    torch.manual_seed(1)
    x = torch.randn(2, 4, 128, 64, device="cuda", dtype=torch.bfloat16)
    y = apply_rope(x, base=500000.0)
    from metis_amd.ops.norms import _rope_ref, rope_tables

    cos_t, sin_t = rope_tables(128, 64, 500000.0, x.device)
    ref = _rope_ref(x.float(), cos_t, sin_t)
    assert (y.float() - ref).abs().max() < 2e-2


def test_rope_gpu_grad_is_inverse_rotation():
    x = torch.randn(1, 2, 128, 64, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    y = apply_rope(x)
    dy = torch.randn_like(y)
    y.backward(dy)
    # d/dx of a rotation is the transpose rotation: |grad| == |dy| per pair
    gn = x.grad.float().pow(2).sum()
    dn = dy.float().pow(2).sum()
    assert torch.allclose(gn, dn, rtol=2e-2)


def test_swiglu_gpu_fwd_bwd():
    torch.manual_seed(2)
    a = torch.randn(4096, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    b = torch.randn(4096, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    y = swiglu(a, b)
    dy = torch.randn_like(y)
    y.backward(dy)

    af = a.detach().float().requires_grad_(True)
    bf = b.detach().float().requires_grad_(True)
    ref = torch.nn.functional.silu(af) * bf
    ref.backward(dy.float())
    assert (y.float() - ref.detach()).abs().max() < 2e-2
    assert (a.grad.float() - af.grad).abs().max() < 3e-2
    assert (b.grad.float() - bf.grad).abs().max() < 3e-2


def test_llama_model_trains_on_gpu():
    from metis_amd.models.llama import LLAMA_SPECS, LlamaModel
    from metis_amd.ops import FusedAdamW

    spec = LLAMA_SPECS["llama-tiny"]
    model = LlamaModel(spec, dtype=torch.bfloat16).to("cuda")
    opt = FusedAdamW(model.parameters(), lr=1e-3)
    tokens = torch.randint(0, spec.vocab_size, (2, spec.seq_length), device="cuda")
    losses = []
    for _ in range(6):
        opt.zero_grad()
        loss = model(tokens, labels=torch.roll(tokens, -1, 1))
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert losses[-1] < losses[0], losses
