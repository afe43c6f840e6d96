"""Mixture-of-Experts GPT family with expert parallelism over RCCL.

Beyond the reference (SamsungLabs/Metis models dense GPT only —
search_space/plan.py has no EP axis): a top-k routed MoE block with the
experts sharded across the TP process group (EP degree == tp).

Design, MI355X-first: our TP scheme replicates block inputs across the
group (ColumnParallelLinear applies f = identity-fwd/all-reduce-bwd), so
expert dispatch needs NO all-to-all — every rank sees every token,
computes only the experts it owns, and one all-reduce (the same g
operator row-parallel layers use) combines the weighted expert outputs.
On xGMI's point-to-point links (7x ~153 GB/s) that is one ring
collective per MoE layer instead of two all-to-alls plus a gather.

Replicated-parameter gradient rules (each rank computes the full router
but backprops only its local experts' gate terms):
- expert + router input passes through one f (_CopyToTP), so the
  partial per-rank activation grads are summed to the full grad;
- router weight/bias grads are summed across the EP group (tiny
  tensors: [E, h]) via the metis_amd.partial_grads protocol — an
  immediate in-backward hook standalone, or exactly once per step on
  the accumulated grad under the runner's GradBucketSync;
- the load-balance aux loss is computed identically on every rank and
  pre-scaled by 1/ep so the cross-rank sum restores the single logical
  contribution.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

import torch
import torch.utils.checkpoint
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from metis_amd.models.gpt import (
    ColumnParallelLinear,
    RowParallelLinear,
    _CopyToTP,
    _ReduceFromTP,
    vocab_parallel_ce,
    LayerNorm,
    _init_linear,
)
from metis_amd.ops.attention import flash_attention
from metis_amd.ops.cross_entropy import cross_entropy
from metis_amd.ops.relayout import heads_merge, qkv_split_transpose
from metis_amd.partial_grads import mark_partial


@dataclass(frozen=True)
class MoEModelSpec:
    name: str
    hidden_size: int
    num_layers: int
    num_heads: int
    vocab_size: int
    seq_length: int
    num_experts: int
    top_k: int = 2
    ffn_hidden_size: Optional[int] = None
    aux_loss_coef: float = 0.01
    expert_activation: str = "gelu"   # "gelu" (GPT FFN) | "swiglu" (Mixtral)

    @property
    def ffn(self) -> int:
        return self.ffn_hidden_size or 4 * self.hidden_size

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_heads

    @property
    def profile_num_layers(self) -> int:
        return self.num_layers + 2

    def num_parameters(self) -> int:
        h, v = self.hidden_size, self.vocab_size
        fmul = 3 if self.expert_activation == "swiglu" else 2
        per_block = (4 * h * h + 4 * h                          # attention + ln
                     + self.num_experts * fmul * h * self.ffn   # experts
                     + h * self.num_experts + 2 * h)            # router + ln
        return v * h + self.seq_length * h + self.num_layers * per_block + 2 * h


MOE_SPECS = {
    # GPT-3 1.3B-shaped attention with 8 experts (~5.4B params, top-2)
    "gpt3-moe-1.3b-8e": MoEModelSpec("gpt3-moe-1.3b-8e", 2048, 24, 16,
                                     51200, 2048, num_experts=8),
    # Mixtral-8x7B-class shape: swiglu experts, top-2 of 8
    "mixtral-8x7b-class": MoEModelSpec(
        "mixtral-8x7b-class", 4096, 32, 32, 32000, 4096, num_experts=8,
        ffn_hidden_size=14336, expert_activation="swiglu"),
    "moe-tiny": MoEModelSpec("moe-tiny", 64, 2, 4, 512, 32, num_experts=4),
    "moe-tiny-swiglu": MoEModelSpec("moe-tiny-swiglu", 64, 2, 4, 512, 32,
                                    num_experts=4,
                                    expert_activation="swiglu"),
}


class _Experts(nn.Module):
    """This rank's slice of the expert FFNs, stored as stacked weights
    ([local_e, ffn, h] / [local_e, h, ffn]) so a single bmm serves all
    local experts when token counts are balanced, with a per-expert
    gather path for the general case."""

    def __init__(self, spec: MoEModelSpec, ep: int, dtype):
        super().__init__()
        assert spec.num_experts % ep == 0, "num_experts must divide by ep"
        assert spec.expert_activation in ("gelu", "swiglu")
        self.local_experts = spec.num_experts // ep
        self.swiglu = spec.expert_activation == "swiglu"
        h, f = spec.hidden_size, spec.ffn
        # swiglu experts fuse gate & up into one [2f, h] weight (Mixtral)
        f1 = 2 * f if self.swiglu else f
        self.w1 = nn.Parameter(torch.empty(self.local_experts, f1, h, dtype=dtype))
        self.b1 = nn.Parameter(torch.zeros(self.local_experts, f1, dtype=dtype))
        self.w2 = nn.Parameter(torch.empty(self.local_experts, h, f, dtype=dtype))
        self.b2 = nn.Parameter(torch.zeros(self.local_experts, h, dtype=dtype))
        for e in range(self.local_experts):
            _init_linear(self.w1[e], h)
            _init_linear(self.w2[e], f)

    def expert_forward(self, e: int, x: torch.Tensor) -> torch.Tensor:
        y = F.linear(x, self.w1[e], self.b1[e])
        if self.swiglu:
            gate, up = y.chunk(2, dim=-1)
            y = F.silu(gate) * up
        else:
            y = F.gelu(y)
        return F.linear(y, self.w2[e], self.b2[e])


class MoEBlock(nn.Module):
    """Pre-LN attention (TP like GPTBlock) + top-k routed expert MLP (EP)."""

    def __init__(self, spec: MoEModelSpec, tp: int, dtype, tp_group=None):
        super().__init__()
        h = spec.hidden_size
        assert spec.num_heads % tp == 0
        self.heads_per_rank = spec.num_heads // tp
        self.head_dim = spec.head_dim
        self.spec = spec
        self.ep = tp

        self.ln_attn = LayerNorm(h)
        self.qkv = ColumnParallelLinear(h, 3 * h, tp, dtype)
        self.proj = RowParallelLinear(h, h, tp, dtype)
        self.ln_mlp = LayerNorm(h)
        # router in fp32 for routing stability
        self.router = nn.Linear(h, spec.num_experts, dtype=torch.float32)
        _init_linear(self.router.weight, h)
        self.experts = _Experts(spec, tp, dtype)
        # router grads are partial per EP rank; tag them EAGERLY (at
        # construction, before any GradBucketSync is built) so the EP sum
        # is ordered before the DP bucket copy — runtime.partial_grads
        mark_partial(self.router.weight, tp_group)
        mark_partial(self.router.bias, tp_group)
        self.last_aux_loss: Optional[torch.Tensor] = None

    def _moe_mlp(self, y: torch.Tensor, tp_group) -> torch.Tensor:
        spec = self.spec
        b, s, h = y.shape
        ep = dist.get_world_size(tp_group) if tp_group is not None else 1

        flat = _CopyToTP.apply(y, tp_group).reshape(-1, h)    # one f for both uses
        logits = self.router(flat.float())                    # [T, E] replicated
        probs = torch.softmax(logits, dim=-1)
        top_p, top_e = probs.topk(spec.top_k, dim=-1)         # [T, k]
        gates = (top_p / top_p.sum(dim=-1, keepdim=True)).to(y.dtype)

        # Switch-style load-balance loss: E * sum_e f_e * P_e, where f_e is
        # the token fraction routed to e (counts) and P_e the mean prob;
        # identical on every rank, pre-scaled by 1/ep (grad hook sums it back)
        with torch.no_grad():
            f_e = torch.zeros(spec.num_experts, device=y.device)
            f_e.scatter_add_(
                0, top_e.reshape(-1),
                torch.ones_like(top_e.reshape(-1), dtype=f_e.dtype))
            f_e /= top_e.numel()
        aux_full = spec.num_experts * (f_e * probs.mean(dim=0)).sum()
        # full VALUE, but gradient scaled by 1/ep (the router grad hook
        # sums the ep copies back to the single logical contribution)
        aux = aux_full / ep + aux_full.detach() * (1.0 - 1.0 / ep)
        self.last_aux_loss = aux

        local0 = (dist.get_rank(tp_group) if ep > 1 else 0) * self.experts.local_experts
        out = torch.zeros_like(flat)
        # every local expert must produce SOME grad every microbatch: a
        # token-starved expert would otherwise skip its post-accumulate
        # hook on the final microbatch and desync the DP bucket schedule
        # across ranks (runtime.grad_sync fails loudly on that). One
        # element per tensor is enough — autograd materializes the full
        # (zero) grad.
        ex = self.experts
        dummy = (ex.w1.flatten()[0] + ex.b1.flatten()[0]
                 + ex.w2.flatten()[0] + ex.b2.flatten()[0])
        out = out + (dummy * 0.0).to(out.dtype)
        for le in range(self.experts.local_experts):
            e = local0 + le
            slot = (top_e == e)                               # [T, k]
            tok = slot.any(dim=-1).nonzero(as_tuple=True)[0]
            if tok.numel() == 0:
                continue
            gate = (gates[tok] * slot[tok].to(gates.dtype)).sum(dim=-1)
            h_e = self.experts.expert_forward(le, flat[tok])
            out.index_add_(0, tok, gate[:, None] * h_e)
        out = _ReduceFromTP.apply(out, tp_group)              # one g combine
        return out.reshape(b, s, h)

    def forward(self, x: torch.Tensor, tp_group) -> torch.Tensor:
        residual = x
        y = self.ln_attn(x)
        qkv = self.qkv(y, tp_group)
        q, k, v = qkv_split_transpose(
            qkv, self.heads_per_rank, self.heads_per_rank, self.head_dim)
        attn = flash_attention(q, k, v, causal=True)
        x = residual + self.proj(heads_merge(attn), tp_group)
        x = x + self._moe_mlp(self.ln_mlp(x), tp_group)
        return x


class MoEModel(nn.Module):
    """Pipeline-stage slice of the MoE model (layer convention: 0 = embed,
    1..n-2 = MoE blocks, n-1 = final norm + head, as in GPTModel)."""

    def __init__(
        self,
        spec: MoEModelSpec,
        tp: int = 1,
        dtype: torch.dtype = torch.bfloat16,
        layer_range: Optional[tuple] = None,
        tp_group=None,
    ):
        super().__init__()
        self.spec = spec
        self.tp = tp
        self.tp_group = tp_group
        total = spec.profile_num_layers
        start, end = layer_range if layer_range is not None else (0, total)
        self.has_embedding = start == 0
        self.has_head = end == total

        h = spec.hidden_size
        if self.has_embedding:
            self.wte = nn.Embedding(spec.vocab_size, h, dtype=dtype)
            self.wpe = nn.Embedding(spec.seq_length, h, dtype=dtype)
            nn.init.normal_(self.wte.weight, std=0.02)
            nn.init.normal_(self.wpe.weight, std=0.02)

        block_start = max(start - 1, 0)
        block_end = min(end, total - 1) - 1
        self.blocks = nn.ModuleList(
            MoEBlock(spec, tp, dtype, tp_group=tp_group)
            for _ in range(max(block_end - block_start, 0)))

        if self.has_head:
            self.ln_final = LayerNorm(h)
            self.head = ColumnParallelLinear(h, spec.vocab_size, tp, dtype)
        self.recompute = False

    def consume_aux_loss(self) -> Optional[torch.Tensor]:
        """Scaled sum of this stage's routing aux losses for the microbatch
        just forwarded; the runner (or _loss) must include it in backward."""
        terms = [b.last_aux_loss for b in self.blocks if b.last_aux_loss is not None]
        for b in self.blocks:
            b.last_aux_loss = None
        if not terms:
            return None
        return self.spec.aux_loss_coef * torch.stack(terms).sum()

    def forward(self, x, labels=None):
        if self.has_embedding:
            b, s = x.shape
            pos = torch.arange(s, device=x.device)
            x = self.wte(x) + self.wpe(pos)[None, :, :]
        use_ckpt = self.recompute and torch.is_grad_enabled()
        for block in self.blocks:
            if use_ckpt:
                x = torch.utils.checkpoint.checkpoint(
                    block, x, self.tp_group, use_reentrant=False)
            else:
                x = block(x, self.tp_group)
        if self.has_head:
            x = self.ln_final(x)
            logits = self.head(x, self.tp_group)
            if labels is not None:
                labels = labels.reshape(-1)
                if self.tp_group is not None and dist.get_world_size(self.tp_group) > 1:
                    ce = vocab_parallel_ce(logits.view(-1, logits.size(-1)),
                                           labels, self.tp_group)
                else:
                    ce = cross_entropy(logits.view(-1, logits.size(-1)), labels)
                aux = self.consume_aux_loss()
                return ce + aux.to(ce.dtype) if aux is not None else ce
            return logits
        return x

    def layer_parameter_bytes(self) -> List[float]:
        spec = self.spec
        el = 2
        h, v = spec.hidden_size, spec.vocab_size
        embed = (v * h + spec.seq_length * h) * el
        fmul = 3 if spec.expert_activation == "swiglu" else 2
        per_block = (4 * h * h + 4 * h + 2 * h
                     + spec.num_experts * (fmul * h * spec.ffn + spec.ffn + h)) * el \
            + h * spec.num_experts * 4  # fp32 router
        head = (v * h + 2 * h) * el
        return [float(embed)] + [float(per_block)] * spec.num_layers + [float(head)]
