"""GPU: MoE family runs the HIP kernel path end to end (single rank,
ep=1 — multi-rank EP is covered by the gloo suite; RCCL cannot host
two ranks on the one-GPU box)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("requires MI355X GPU", allow_module_level=True)

from metis_amd.models.moe import MoEModel, MOE_SPECS


@pytest.mark.parametrize("spec_name", ["moe-tiny", "moe-tiny-swiglu"])
def test_moe_forward_backward_matches_cpu(spec_name):
    spec = MOE_SPECS[spec_name]
    torch.manual_seed(3)
    model = MoEModel(spec, tp=1, dtype=torch.float32)
    tokens = torch.randint(0, spec.vocab_size, (2, spec.seq_length))
    labels = torch.roll(tokens, -1, 1)

    cpu_loss = model(tokens, labels=labels)
    cpu_loss.backward()
    cpu_router_grad = model.blocks[0].router.weight.grad.clone()

    gpu = MoEModel(spec, tp=1, dtype=torch.float32)
    gpu.load_state_dict(model.state_dict())
    gpu = gpu.to("cuda")
    gpu_loss = gpu(tokens.cuda(), labels=labels.cuda())
    gpu_loss.backward()

    assert abs(float(gpu_loss) - float(cpu_loss)) / abs(float(cpu_loss)) < 2e-2
    g = gpu.blocks[0].router.weight.grad.cpu()
    denom = cpu_router_grad.abs().max().clamp(min=1e-6)
    assert (g - cpu_router_grad).abs().max() / denom < 0.05


def test_moe_bf16_train_step():
    from metis_amd.ops import FusedAdamW

    spec = MOE_SPECS["moe-tiny"]
    torch.manual_seed(4)
    model = MoEModel(spec, tp=1, dtype=torch.bfloat16).to("cuda")
    opt = FusedAdamW(model.parameters(), lr=1e-3)
    losses = []
    tokens = torch.randint(0, spec.vocab_size, (4, spec.seq_length),
                           device="cuda")
    labels = torch.roll(tokens, -1, 1)
    for _ in range(8):
        opt.zero_grad()
        loss = model(tokens, labels=labels)
        loss.backward()
        opt.step(grad_scale=1.0)
        losses.append(float(loss))
    assert losses[-1] < losses[0], losses   # memorizes the fixed batch
