"""Memory-mapped token dataset: sharding, determinism, resume."""

import numpy as np
import torch

from metis_amd.data import TokenDataset, TokenLoader


def _write_tokens(path, n=4097, vocab=512):
    rng = np.random.RandomState(0)
    rng.randint(0, vocab, size=n).astype(np.uint16).tofile(path)


def test_samples_and_labels_shift(tmp_path):
    p = tmp_path / "toks.bin"
    _write_tokens(str(p), n=257)
    ds = TokenDataset(str(p), seq_length=32)
    assert ds.num_samples == 8
    loader = TokenLoader(ds, mbs=2, dp=1, dp_rank=0)
    tokens, labels = loader.next_batch()
    assert tokens.shape == (2, 32) and labels.shape == (2, 32)
    assert torch.equal(tokens[:, 1:], labels[:, :-1])  # next-token shift


def test_dp_replicas_disjoint_and_cover(tmp_path):
    p = tmp_path / "toks.bin"
    _write_tokens(str(p))
    ds = TokenDataset(str(p), seq_length=32)
    l0 = TokenLoader(ds, mbs=4, dp=2, dp_rank=0, seed=7)
    l1 = TokenLoader(ds, mbs=4, dp=2, dp_rank=1, seed=7)
    seen0, seen1 = set(), set()
    for _ in range(l0.microbatches_per_epoch):
        t0, _ = l0.next_batch()
        t1, _ = l1.next_batch()
        seen0.add(t0.numpy().tobytes())
        seen1.add(t1.numpy().tobytes())
    assert not (seen0 & seen1)  # disjoint shards


def test_resume_replays_exact_stream(tmp_path):
    p = tmp_path / "toks.bin"
    _write_tokens(str(p))
    ds = TokenDataset(str(p), seq_length=32)
    a = TokenLoader(ds, mbs=2, dp=1, dp_rank=0, seed=3)
    stream = [a.next_batch()[0] for _ in range(20)]  # crosses an epoch edge

    b = TokenLoader(ds, mbs=2, dp=1, dp_rank=0, seed=3)
    for _ in range(7):
        b.next_batch()
    cursor = b.state()
    c = TokenLoader(ds, mbs=2, dp=1, dp_rank=0, seed=3)
    c.load_state(cursor)
    for i in range(7, 20):
        assert torch.equal(c.next_batch()[0], stream[i]), i


def test_runner_trains_on_file_data(tmp_path):
    from metis_amd.models.gpt import GPTModelSpec
    from metis_amd.runtime.comm import ParallelContext
    from metis_amd.runtime.runner import PlanRunner

    p = tmp_path / "toks.bin"
    _write_tokens(str(p), n=2049)
    spec = GPTModelSpec("tiny", hidden_size=64, num_layers=2, num_heads=4,
                        vocab_size=512, seq_length=32)
    ctx = ParallelContext(rank=0, world_size=1, local_rank=0, dp=1, tp=1, pp=1)
    torch.manual_seed(3)
    runner = PlanRunner(spec, ctx, mbs=2, gbs=4, dtype=torch.float32,
                        data_path=str(p))
    l1 = runner.train_step()
    l2 = runner.train_step()
    assert l1 > 0 and l2 > 0
    assert runner.data_loader.state() == 4  # 2 steps x 2 microbatches


def test_dataset_split_disjoint(tmp_path):
    p = tmp_path / "toks.bin"
    _write_tokens(str(p), n=32 * 100 + 1)
    train = TokenDataset(str(p), 32, split=(0.0, 0.9))
    ev = TokenDataset(str(p), 32, split=(0.9, 1.0))
    assert train.num_samples == 90 and ev.num_samples == 10
    assert np.array_equal(ev.sample(0), np.asarray(
        TokenDataset(str(p), 32).sample(90)))
