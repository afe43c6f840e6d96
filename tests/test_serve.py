"""Completion server over the KV-cache decode path (in-process client):
checkpoint -> serve flow and deterministic greedy output."""

import torch

from metis_amd.cli.serve import build_app, load_model
from metis_amd.models.gpt import GPTModel, GPTModelSpec
from metis_amd.runtime.generate import generate

SPEC = GPTModelSpec("tiny", hidden_size=64, num_layers=2, num_heads=4,
                    vocab_size=512, seq_length=64)


def _client(model):
    from fastapi.testclient import TestClient

    return TestClient(build_app(model, SPEC))


def test_generate_endpoint_matches_direct_call():
    torch.manual_seed(0)
    model = GPTModel(SPEC, dtype=torch.float32)
    model.eval()
    client = _client(model)

    r = client.get("/health")
    assert r.status_code == 200 and r.json()["model"] == "tiny"

    g = torch.Generator().manual_seed(11)
    prompt = torch.randint(0, 512, (1, 8), generator=g)
    r = client.post("/generate", json={
        "tokens": prompt.tolist(), "max_new_tokens": 6, "temperature": 0.0})
    assert r.status_code == 200
    served = torch.tensor(r.json()["tokens"])
    ref = generate(model, prompt, 6, temperature=0.0)
    assert torch.equal(served, ref)


def test_checkpoint_to_serve_roundtrip(tmp_path):
    from metis_amd.runtime.comm import ParallelContext
    from metis_amd.runtime.runner import PlanRunner

    ctx = ParallelContext(rank=0, world_size=1, local_rank=0, dp=1, tp=1, pp=1)
    torch.manual_seed(3)
    runner = PlanRunner(SPEC, ctx, mbs=2, gbs=2, dtype=torch.float32)
    runner.train_step()
    ckpt = tmp_path / "rank0.pt"
    runner.save_checkpoint(str(ckpt))

    import metis_amd.cli.serve as serve

    serve.MODEL_SPECS = dict(serve.MODEL_SPECS)
    serve.MODEL_SPECS["tiny"] = SPEC
    model, spec = serve.load_model("tiny", str(ckpt), dtype=torch.float32)
    for p1, p2 in zip(runner.model.parameters(), model.parameters()):
        assert torch.equal(p1, p2)

    client = _client(model)
    r = client.post("/generate", json={
        "tokens": [[1, 2, 3, 4]], "max_new_tokens": 4})
    assert r.status_code == 200
    assert len(r.json()["tokens"][0]) == 8


def test_ragged_batch_endpoint():
    torch.manual_seed(0)
    model = GPTModel(SPEC, dtype=torch.float32)
    model.eval()
    client = _client(model)
    r = client.post("/generate", json={
        "tokens": [[1, 2, 3, 4, 5], [9, 8], [7, 6, 5]],
        "max_new_tokens": 4, "temperature": 0.0})
    assert r.status_code == 200
    rows = r.json()["tokens"]
    assert [len(x) for x in rows] == [9, 6, 7]
    # each row matches its single-prompt greedy continuation
    single = client.post("/generate", json={
        "tokens": [[9, 8]], "max_new_tokens": 4, "temperature": 0.0})
    assert rows[1] == single.json()["tokens"][0]


def test_bad_requests_rejected():
    torch.manual_seed(0)
    model = GPTModel(SPEC, dtype=torch.float32)
    model.eval()
    client = _client(model)
    assert client.post("/generate", json={"tokens": []}).status_code == 400
    assert client.post("/generate", json={"tokens": [[]]}).status_code == 400
    too_long = list(range(SPEC.seq_length + 1))
    assert client.post(
        "/generate", json={"tokens": [too_long]}).status_code == 400


def test_continuous_serving_interleaves_requests():
    """--continuous mode: concurrent HTTP requests share one decode
    loop and return the same greedy continuations as direct calls."""
    import threading

    from metis_amd.cli.serve import build_app
    from metis_amd.runtime.generate import generate
    from fastapi.testclient import TestClient

    torch.manual_seed(0)
    model = GPTModel(SPEC, dtype=torch.float32)
    model.eval()
    app = build_app(model, SPEC, continuous=True, max_batch=2)
    client = TestClient(app)

    g = torch.Generator().manual_seed(21)
    prompts = [torch.randint(0, SPEC.vocab_size, (1, n), generator=g)
               for n in (4, 6, 5)]
    expected = [generate(model, p, 5, temperature=0.0)[0].tolist()
                for p in prompts]

    results = [None] * 3

    def post(i):
        r = client.post("/generate", json={
            "tokens": [prompts[i][0].tolist()], "max_new_tokens": 5})
        assert r.status_code == 200
        results[i] = r.json()["tokens"][0]

    threads = [threading.Thread(target=post, args=(i,)) for i in range(3)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=120)
    assert results == expected
