"""Model and planner configuration objects.

Replaces the reference's argparse namespace threaded through the whole
stack (and re-parsed deep inside the cost model, cost_estimator.py:154 —
quirk Q9): everything below is an explicit dataclass.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional


@dataclass
class ModelConfig:
    """GPT-family model description (reference: utils.py:71-79).

    ``num_layers`` counts the *profiled* layers: input embedding layer +
    (num_layers - 2) transformer blocks + output head, matching the
    profile JSON's per-layer arrays.
    """

    model_name: str
    num_layers: int
    hidden_size: int
    sequence_length: int
    vocab_size: int
    attention_head_size: int = 0
    ffn_hidden_size: Optional[int] = None   # defaults to 4*hidden for GPT
    num_attention_heads: Optional[int] = None
    num_kv_heads: Optional[int] = None      # GQA (Llama); None => MHA
    # MoE (EP axis, absent in the reference): >0 selects MoEVolume in
    # the planner — experts shard over the EP(=TP) group while router +
    # norms stay replicated, which changes the per-rank parameter bytes
    # the DP ring-all-reduce and memory terms price.
    num_experts: int = 0
    expert_weight_mul: int = 2              # 2 = gelu FFN, 3 = swiglu

    def __post_init__(self) -> None:
        if self.ffn_hidden_size is None:
            self.ffn_hidden_size = 4 * self.hidden_size


@dataclass
class PlannerArgs:
    """Search-space and profile-coverage limits.

    Mirrors the reference CLI surface (arguments.py:42-49) so the CLI
    entry points stay flag-compatible.
    """

    gbs: int
    max_profiled_tp_degree: int = 8
    max_profiled_batch_size: int = 16
    min_group_scale_variance: float = 1.0
    max_permute_len: int = 4

    # MI355X extensions (defaults keep reference-parity behavior):
    # "parity"     — bandwidth-class model: t = bytes / BW (reference semantics)
    # "alpha_beta" — t = alpha + bytes / BW with measured latency term
    comm_model: str = "parity"
    alpha_us: float = 20.0          # per-collective latency when comm_model="alpha_beta"
    # "parity"   — fb_sync residual charged once per MICROBATCH
    #              (reference cost_estimator.py:120); overprices gradient
    #              accumulation (+20% measured at gpt2-small mbs=2).
    # "marginal" — per-microbatch time = measured accumulation marginal
    #              (profile extension keys fwd_bwd_{1,2}mb_ms), iteration
    #              residual charged ONCE; falls back to parity for
    #              profiles without the keys.
    microbatch_model: str = "parity"
    # Linearly interpolate layer times/memory/fb_sync between profiled
    # batch sizes instead of skipping unprofiled-mbs plans (KeyError).
    interpolate_bs: bool = False
    # Pipeline schedule the homo estimator prices (the runtime executes
    # all three). "gpipe" keeps the reference bubble (B-1)*max + sum
    # (cost_estimator.py:129); "1f1b" has the same bubble but holds at
    # most pp microbatches of activations in flight instead of B;
    # "interleaved" with vpp virtual chunks per stage shrinks the bubble
    # to (pp-1)/vpp microbatch slots at vpp x the p2p volume.
    schedule: str = "gpipe"
    vpp: int = 1
    activation_dtype_bytes: int = 1  # 1 => element-count parity (quirk Q8); 2 for bf16 bytes
    # The reference's LayerComputeBalancer can emit partitions that do NOT
    # cover every layer on skewed many-stage inputs (slice rounding drops a
    # layer; reproduced live, e.g. 9 stages x 10 layers -> [..., 8, 9]).
    # Such plans are under-costed (the missing layer costs nothing) and
    # unrunnable. Default keeps them for plan-table parity; True drops them.
    drop_incomplete_partitions: bool = False
