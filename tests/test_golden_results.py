"""Golden-output regression tests (the reference's results/ artifacts,
rebuilt for this framework): re-run the planner on the committed inputs
and compare against the captured ranked tables under results/."""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(script, args):
    out = subprocess.run(
        [sys.executable, script] + args,
        cwd=REPO, capture_output=True, text=True, check=True,
    ).stdout
    return [l for l in out.splitlines() if l and l[0].isdigit()]


def _golden(name):
    with open(os.path.join(REPO, "results", name)) as fh:
        return [l for l in fh.read().splitlines() if l and l[0].isdigit()]


def test_homo_mi355x_golden():
    rows = _run("cost_homo_cluster.py", [
        "--model_name", "gpt2-small", "--num_layers", "14", "--gbs", "32",
        "--hidden_size", "768", "--sequence_length", "1024",
        "--vocab_size", "51200",
        "--hostfile_path", "tests/data/mi355x_single_node/hostfile",
        "--clusterfile_path", "tests/data/mi355x_single_node/clusterfile.json",
        "--profile_data_path", "profiles/mi355x/gpt2-small",
        "--max_profiled_tp_degree", "1", "--max_profiled_batch_size", "8",
        "--comm_model", "alpha_beta",
    ])
    assert rows == _golden("mi355x_homo_gpt2s_gbs32.txt")


def test_hetero_synth_golden():
    rows = _run("cost_het_cluster.py", [
        "--model_name", "GPT", "--num_layers", "10", "--gbs", "16",
        "--hidden_size", "4096", "--sequence_length", "1024",
        "--vocab_size", "51200",
        "--hostfile_path", "tests/data/mi355x_hetero/hostfile",
        "--clusterfile_path", "tests/data/mi355x_hetero/clusterfile.json",
        "--profile_data_path", "tests/data/profiles_synth",
        "--max_profiled_tp_degree", "4", "--max_profiled_batch_size", "4",
        "--min_group_scale_variance", "1", "--max_permute_len", "4",
    ])
    golden = _golden("synth_hetero_2type_gbs16.txt")
    assert len(rows) == len(golden) == 159
    assert rows == golden
    # heterogeneity-aware: the best plan gives the slow type fewer layers
    best = rows[0]
    assert "MI355X_LC" in best and "[0, 4, 10]" in best
