"""CPU tests for the GPT model family and fused AdamW (eager fallback)."""

import torch
import pytest

from metis_amd.models.gpt import GPTModel, GPTModelSpec, MODEL_SPECS
from metis_amd.ops import FusedAdamW


def tiny_spec(**kw):
    base = dict(name="tiny", hidden_size=64, num_layers=2, num_heads=4,
                vocab_size=512, seq_length=32)
    base.update(kw)
    return GPTModelSpec(**base)


def test_forward_backward_loss():
    torch.manual_seed(0)
    model = GPTModel(tiny_spec(), dtype=torch.float32)
    tokens = torch.randint(0, 512, (2, 32))
    labels = torch.roll(tokens, -1, 1)
    loss = model(tokens, labels=labels)
    assert torch.isfinite(loss)
    loss.backward()
    grads = [p.grad for p in model.parameters()]
    assert all(g is not None for g in grads)
    assert all(torch.isfinite(g).all() for g in grads)


def test_logits_shape_without_labels():
    model = GPTModel(tiny_spec(), dtype=torch.float32)
    tokens = torch.randint(0, 512, (2, 32))
    logits = model(tokens)
    assert logits.shape == (2, 32, 512)


def test_stage_slicing():
    spec = tiny_spec()
    total = spec.profile_num_layers  # 4: embed + 2 blocks + head
    s0 = GPTModel(spec, dtype=torch.float32, layer_range=(0, 2))
    s1 = GPTModel(spec, dtype=torch.float32, layer_range=(2, total))
    assert s0.has_embedding and not s0.has_head
    assert not s1.has_embedding and s1.has_head
    assert len(s0.blocks) + len(s1.blocks) == spec.num_layers

    tokens = torch.randint(0, 512, (2, 32))
    hidden = s0(tokens)
    assert hidden.shape == (2, 32, 64)
    loss = s1(hidden, labels=torch.roll(tokens, -1, 1))
    assert torch.isfinite(loss)


def test_param_count_2p7b():
    spec = MODEL_SPECS["gpt3-2.7b"]
    n = spec.num_parameters()
    assert 2.5e9 < n < 3.0e9, n


def test_fused_adamw_cpu_training_reduces_loss():
    torch.manual_seed(1)
    model = GPTModel(tiny_spec(num_layers=1), dtype=torch.float32)
    opt = FusedAdamW(model.parameters(), lr=3e-3, weight_decay=0.0)
    tokens = torch.randint(0, 512, (2, 32))
    labels = torch.roll(tokens, -1, 1)
    first = None
    for _ in range(8):
        opt.zero_grad()
        loss = model(tokens, labels=labels)
        loss.backward()
        opt.step()
        if first is None:
            first = float(loss)
    assert float(loss) < first, (first, float(loss))


def test_fused_adamw_matches_torch_adamw():
    torch.manual_seed(2)
    w = torch.nn.Parameter(torch.randn(64))
    w2 = torch.nn.Parameter(w.detach().clone())

    mine = FusedAdamW([w], lr=1e-2, betas=(0.9, 0.95), eps=1e-8, weight_decay=0.1)
    ref = torch.optim.AdamW([w2], lr=1e-2, betas=(0.9, 0.95), eps=1e-8,
                            weight_decay=0.1)
    for step in range(5):
        g = torch.randn(64)
        w.grad = g.clone()
        w2.grad = g.clone()
        mine.step()
        ref.step()
        assert torch.allclose(w.detach(), w2.detach(), atol=1e-6), step


def test_runner_checkpoint_roundtrip(tmp_path):
    from metis_amd.runtime.comm import ParallelContext
    from metis_amd.runtime.runner import PlanRunner

    torch.manual_seed(4)
    spec = tiny_spec(num_layers=1)
    ctx = ParallelContext(rank=0, world_size=1, local_rank=0, dp=1, tp=1, pp=1)
    runner = PlanRunner(spec, ctx, mbs=2, gbs=2, dtype=torch.float32)
    runner.train_step()
    runner.save_checkpoint(str(tmp_path / "ckpt.pt"))
    master_before = runner.optimizer.master.clone()
    step_before = runner.optimizer.step_count

    runner.train_step()  # diverge
    assert not torch.equal(master_before, runner.optimizer.master)

    runner.load_checkpoint(str(tmp_path / "ckpt.pt"))
    assert torch.equal(master_before, runner.optimizer.master)
    assert runner.optimizer.step_count == step_before
    runner.train_step()  # resumes cleanly


def test_warmup_cosine_lr():
    import math

    from metis_amd.ops import FusedAdamW
    from metis_amd.runtime.lr import WarmupCosineLR

    import torch

    p = torch.nn.Parameter(torch.ones(4))
    opt = FusedAdamW([p], lr=0.0)
    sched = WarmupCosineLR(opt, lr_max=1.0, warmup_steps=10, decay_steps=110,
                           lr_min=0.1)
    assert abs(sched.step(0) - 0.1) < 1e-9       # first warmup step
    assert abs(sched.step(9) - 1.0) < 1e-9       # warmup peak
    mid = sched.step(60)                          # cosine midpoint
    assert abs(mid - (0.1 + 0.45 * (1 + math.cos(math.pi / 2)))) < 1e-9
    assert abs(sched.step(200) - 0.1) < 1e-9     # floor after decay
    assert opt.lr == sched.lr_at(200)
