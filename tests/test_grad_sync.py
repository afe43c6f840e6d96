"""Unit tests for the bucketed DP gradient sync (single process)."""

import torch

from metis_amd.models.gpt import GPTModel, GPTModelSpec
from metis_amd.ops import FusedAdamW
from metis_amd.runtime.grad_sync import GradBucketSync


def _model():
    torch.manual_seed(0)
    spec = GPTModelSpec("t", hidden_size=64, num_layers=3, num_heads=4,
                        vocab_size=256, seq_length=32)
    return GPTModel(spec, dtype=torch.float32)


def test_buckets_partition_flat_buffer():
    model = _model()
    opt = FusedAdamW(model.parameters(), lr=1e-3)
    sync = GradBucketSync(opt, dp_group=None, dp_size=1, bucket_mb=0.25)

    # buckets tile the parameter span contiguously and cover every param
    assert len(sync._buckets) > 1
    spans = sorted((s, e) for s, e, _ in sync._buckets)
    for (s1, e1), (s2, e2) in zip(spans, spans[1:]):
        assert e1 == s2, "buckets must tile contiguously"
    assert spans[0][0] == 0
    assert spans[-1][1] == sum(n for _, n in opt._slices)
    covered = sum(n for _, _, n in sync._buckets)
    assert covered == len(opt.params)


def test_hooks_fill_flat_buffer_when_armed():
    model = _model()
    opt = FusedAdamW(model.parameters(), lr=1e-3)
    sync = GradBucketSync(opt, dp_group=None, dp_size=1, bucket_mb=0.25)

    tokens = torch.randint(0, 256, (2, 32))
    # un-armed backward: flat buffer untouched
    loss = model(tokens, labels=torch.roll(tokens, -1, 1))
    loss.backward()
    assert torch.count_nonzero(opt.grad_flat) == 0

    opt.zero_grad()
    sync.arm()
    loss = model(tokens, labels=torch.roll(tokens, -1, 1))
    loss.backward()
    assert all(p == 0 for p in sync._pending), sync._pending
    # flat buffer holds exactly the gathered grads
    expected = torch.cat([p.grad.reshape(-1) for p in opt.params])
    got = opt.grad_flat[: expected.numel()]
    assert torch.allclose(got, expected)
    sync.finish()


def test_step_tracer(tmp_path):
    from metis_amd.runtime.trace import StepTracer

    tracer = StepTracer(str(tmp_path / "trace.json"))
    for _ in range(2):
        with tracer.span("forward"):
            pass
        with tracer.span("optimizer"):
            pass
        tracer.next_step()
    summary = tracer.summary()
    assert set(summary) == {"forward", "optimizer"}
    tracer.export()
    import json as _json

    doc = _json.loads((tmp_path / "trace.json").read_text())
    assert len(doc["traceEvents"]) == 4
    assert "summary_ms" in doc["metadata"]
