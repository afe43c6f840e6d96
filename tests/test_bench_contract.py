"""bench.py driver contract: one JSON line with the required fields."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


def test_bench_json_line_cpu():
    out = subprocess.run(
        [sys.executable, "bench.py", "--gpus", "1", "--steps", "1",
         "--warmup", "0", "--model", "llama-tiny", "--per-gpu-batch", "1",
         "--mbs", "1"],
        cwd=REPO, capture_output=True, text=True, timeout=300,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    doc = json.loads(lines[0])
    assert REQUIRED.issubset(doc.keys()), REQUIRED - set(doc.keys())
    assert doc["metric"] == "best_plan_iter_time_ms"
    assert doc["higher_is_better"] is False
    assert doc["scaling"] == "weak"
    assert doc["dtype"] == "bf16"
    assert doc["data"] == "synthetic"
    assert doc["value"] == doc["ms_per_step"] > 0
    cfg = doc["config"]
    for key in ("model", "global_batch", "seq_len", "parallelism"):
        assert key in cfg


def test_bench_json_line_two_ranks_torchrun():
    """The driver's N>1 launch shape: torch.distributed.run, one JSON line
    from rank 0, value = max over ranks (gloo on CPU, RCCL on the node)."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29733", "bench.py", "--gpus", "2",
         "--steps", "1", "--warmup", "0", "--model", "llama-tiny",
         "--per-gpu-batch", "1", "--mbs", "1"],
        cwd=REPO, capture_output=True, text=True, timeout=300,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    doc = json.loads(lines[0])
    assert doc["n_gpus"] == 2
    assert doc["config"]["parallelism"] == "dp2_tp1_pp1"
    assert doc["config"]["global_batch"] == 2
    assert doc["value"] > 0
