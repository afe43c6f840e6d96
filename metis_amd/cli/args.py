"""Reference-compatible CLI surface (arguments.py:5-49) + MI355X extras."""

from __future__ import annotations

import argparse
from typing import List, Optional

from metis_amd.config import ModelConfig, PlannerArgs


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(description="metis_amd auto-parallelism planner")
    # model
    p.add_argument("--model_name", type=str, default="GPT")
    p.add_argument("--model_size", type=str, default=None)
    p.add_argument("--num_layers", type=int, required=True)
    p.add_argument("--gbs", type=int, required=True)
    # gpt-model
    p.add_argument("--hidden_size", type=int, required=True)
    p.add_argument("--sequence_length", type=int, required=True)
    p.add_argument("--vocab_size", type=int, required=True)
    p.add_argument("--attention_head_size", type=int, default=0)
    # cluster
    p.add_argument("--hostfile_path", required=True)
    p.add_argument("--clusterfile_path", required=True)
    # search-space / profiles ("hetspeed" group in the reference)
    p.add_argument("--profile_data_path", required=True)
    p.add_argument("--max_profiled_tp_degree", type=int, default=8)
    p.add_argument("--max_profiled_batch_size", type=int, default=16)
    p.add_argument("--min_group_scale_variance", type=float, default=1.0)
    p.add_argument("--max_permute_len", type=int, default=4)
    # env
    p.add_argument("--log_path", default=None)
    p.add_argument("--home_dir", default=None)
    # MI355X extensions
    p.add_argument("--comm_model", choices=["parity", "alpha_beta"], default="parity")
    p.add_argument("--alpha_us", type=float, default=20.0)
    p.add_argument("--microbatch_model", choices=["parity", "marginal"],
                   default="parity",
                   help="marginal: price accumulated microbatches at the "
                        "measured fwd_bwd_{2,4}mb probe marginal, iteration "
                        "residual charged once")
    p.add_argument("--interpolate_bs", action="store_true",
                   help="linearly interpolate profile quantities between "
                        "profiled batch sizes instead of skipping plans")
    p.add_argument("--schedule", choices=["gpipe", "1f1b", "interleaved"],
                   default="gpipe",
                   help="pipeline schedule the homo estimator prices")
    p.add_argument("--vpp", type=int, default=1,
                   help="virtual chunks per stage for --schedule interleaved")
    p.add_argument("--activation_dtype_bytes", type=int, default=1)
    p.add_argument("--drop_incomplete_partitions", action="store_true",
                   help="drop plans whose layer partition misses layers "
                        "(reference balancer quirk; under-costed, unrunnable)")
    p.add_argument("--evaluation_data_path", default=None,
                   help="measured-runtime JSON for cost-model validation")
    p.add_argument("--top_k", type=int, default=0, help="print only the top K plans (0 = all)")
    p.add_argument("--json_out", default=None,
                   help="also write the ranked plans as structured JSON")
    return p


def parse(argv: Optional[List[str]] = None):
    args = build_parser().parse_args(argv)
    model_config = ModelConfig(
        model_name=args.model_name,
        num_layers=args.num_layers,
        hidden_size=args.hidden_size,
        sequence_length=args.sequence_length,
        vocab_size=args.vocab_size,
        attention_head_size=args.attention_head_size,
    )
    planner_args = PlannerArgs(
        gbs=args.gbs,
        max_profiled_tp_degree=args.max_profiled_tp_degree,
        max_profiled_batch_size=args.max_profiled_batch_size,
        min_group_scale_variance=args.min_group_scale_variance,
        max_permute_len=args.max_permute_len,
        comm_model=args.comm_model,
        alpha_us=args.alpha_us,
        activation_dtype_bytes=args.activation_dtype_bytes,
        drop_incomplete_partitions=args.drop_incomplete_partitions,
        microbatch_model=args.microbatch_model,
        interpolate_bs=args.interpolate_bs,
        schedule=args.schedule,
        vpp=args.vpp,
    )
    return args, model_config, planner_args
