"""Llama model family: RMSNorm + RoPE + GQA flash attention + SwiGLU MLP,
with the same Megatron-style TP sharding and pipeline-stage slicing as
the GPT family (profile layer convention: 0 = embed, 1..n-2 = blocks,
n-1 = final norm + head)."""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

import torch
import torch.utils.checkpoint
import torch.distributed as dist
import torch.nn as nn

from metis_amd.models.gpt import (
    ColumnParallelLinear,
    RowParallelLinear,
    _GatherSeq,
    _ReduceScatterSeq,
    vocab_parallel_ce,
)
from metis_amd.ops.attention import decode_attention, flash_attention
from metis_amd.ops.cross_entropy import cross_entropy
from metis_amd.ops.norms import (RMSNorm, apply_rope,
                                 apply_rope_rows, swiglu)
from metis_amd.ops.relayout import (heads_merge, qkv_rope_split,
                                    qkv_split_transpose)
from metis_amd.partial_grads import mark_partial

import os as _os


@dataclass(frozen=True)
class LlamaModelSpec:
    name: str
    hidden_size: int
    num_layers: int
    num_heads: int
    num_kv_heads: int
    ffn_hidden_size: int
    vocab_size: int
    seq_length: int
    rope_base: float = 500000.0

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_heads

    @property
    def profile_num_layers(self) -> int:
        return self.num_layers + 2

    def num_parameters(self) -> int:
        h, v, f = self.hidden_size, self.vocab_size, self.ffn_hidden_size
        kvh = self.num_kv_heads * self.head_dim
        per_block = h * h + 2 * h * kvh + h * h + 3 * h * f + 2 * h
        return 2 * v * h + self.num_layers * per_block + h


LLAMA_SPECS = {
    # Llama-3-8B shape (vocab rounded to a 128-multiple for TP sharding)
    "llama3-8b": LlamaModelSpec("llama3-8b", 4096, 32, 32, 8, 14336,
                                128256, 8192),
    "llama3-1b": LlamaModelSpec("llama3-1b", 2048, 16, 32, 8, 8192,
                                128256, 2048),
    "llama-tiny": LlamaModelSpec("llama-tiny", 256, 2, 4, 2, 512, 2048, 256),
}


class LlamaBlock(nn.Module):
    def __init__(self, spec: LlamaModelSpec, tp: int, dtype):
        super().__init__()
        h = spec.hidden_size
        d = spec.head_dim
        assert spec.num_heads % tp == 0 and spec.num_kv_heads % tp == 0, (
            "heads and kv heads must divide by tp"
        )
        self.heads_per_rank = spec.num_heads // tp
        self.kv_heads_per_rank = spec.num_kv_heads // tp
        self.head_dim = d
        self.rope_base = spec.rope_base

        qkv_out = (spec.num_heads + 2 * spec.num_kv_heads) * d
        self.norm_attn = RMSNorm(h)
        self.qkv = ColumnParallelLinear(h, qkv_out, tp, dtype)
        self.proj = RowParallelLinear(spec.num_heads * d, h, tp, dtype)
        self.norm_mlp = RMSNorm(h)
        # gate & up fused in one column-parallel GEMM
        self.gate_up = ColumnParallelLinear(h, 2 * spec.ffn_hidden_size, tp, dtype)
        self.down = RowParallelLinear(spec.ffn_hidden_size, h, tp, dtype)
        self.ffn_per_rank = spec.ffn_hidden_size // tp
        # fused relayout+RoPE kernel (qkv_rope.hip); opt-in until
        # GPU-validated (CPU path is identical composition either way)
        self._fused_qkv_rope = _os.environ.get("METIS_QKV_ROPE") == "1"


    def forward(self, x: torch.Tensor, tp_group, sp: bool = False,
                cache=None, layer_idx: int = 0,
                pos_offset: int = 0) -> torch.Tensor:
        hq, hkv, d = self.heads_per_rank, self.kv_heads_per_rank, self.head_dim

        if cache is not None:
            import torch.nn.functional as F

            residual = x
            y = self.norm_attn(x)
            qkv = self.qkv(y, tp_group)
            q, k, v = qkv_split_transpose(qkv, hq, hkv, d)
            if torch.is_tensor(pos_offset):       # ragged per-row positions
                q = apply_rope_rows(q, self.rope_base, pos_offset)
                k = apply_rope_rows(k, self.rope_base, pos_offset)
            else:
                q = apply_rope(q, self.rope_base, pos_offset=pos_offset)
                k = apply_rope(k, self.rope_base, pos_offset=pos_offset)
            k, v = cache.append(layer_idx, k, v)
            am = getattr(cache, "attention_mask", None)
            rag_mask = am(k.size(2), q.device) if am is not None else None
            new, total = q.size(2), k.size(2)
            if rag_mask is not None:
                rep = hq // hkv
                attn = F.scaled_dot_product_attention(
                    q, k.repeat_interleave(rep, dim=1),
                    v.repeat_interleave(rep, dim=1), attn_mask=rag_mask)
            elif new == 1:
                attn = decode_attention(q, k, v)  # GQA mapped inside
            else:
                if hkv != hq:  # GQA: expand kv heads for SDPA
                    rep = hq // hkv
                    k = k.repeat_interleave(rep, dim=1)
                    v = v.repeat_interleave(rep, dim=1)
                mask = torch.tril(
                    torch.ones(new, total, dtype=torch.bool, device=q.device),
                    diagonal=total - new)
                attn = F.scaled_dot_product_attention(q, k, v, attn_mask=mask)
            x = residual + self.proj(heads_merge(attn), tp_group)
            residual = x
            gate_up = self.gate_up(self.norm_mlp(x), tp_group)
            gate, up = gate_up.split([self.ffn_per_rank, self.ffn_per_rank],
                                     dim=-1)
            return residual + self.down(
                swiglu(gate.contiguous(), up.contiguous()), tp_group)

        import torch.nn.functional as F

        residual = x
        y = self.norm_attn(x)
        if sp:  # seq all-gather replaces the f operator (see gpt.py SP)
            y = _GatherSeq.apply(y, tp_group)
            qkv = F.linear(y, self.qkv.weight, self.qkv.bias)
        else:
            qkv = self.qkv(y, tp_group)
        if self._fused_qkv_rope:
            q, k, v = qkv_rope_split(qkv, hq, hkv, d, self.rope_base)
        else:
            q, k, v = qkv_split_transpose(qkv, hq, hkv, d)
            q = apply_rope(q, self.rope_base)
            k = apply_rope(k, self.rope_base)
        attn = flash_attention(q, k, v, causal=True)
        if sp:
            part = F.linear(heads_merge(attn), self.proj.weight)
            part = _ReduceScatterSeq.apply(part, tp_group)
            x = residual + part + self.proj.bias
        else:
            x = residual + self.proj(heads_merge(attn), tp_group)

        residual = x
        y = self.norm_mlp(x)
        if sp:
            y = _GatherSeq.apply(y, tp_group)
            gate_up = F.linear(y, self.gate_up.weight, self.gate_up.bias)
        else:
            gate_up = self.gate_up(y, tp_group)
        gate, up = gate_up.split([self.ffn_per_rank, self.ffn_per_rank], dim=-1)
        act = swiglu(gate.contiguous(), up.contiguous())
        if sp:
            part = _ReduceScatterSeq.apply(F.linear(act, self.down.weight),
                                           tp_group)
            x = residual + part + self.down.bias
        else:
            x = residual + self.down(act, tp_group)
        return x


class LlamaModel(nn.Module):
    """A pipeline-stage slice of the Llama model (see GPTModel docs)."""

    def __init__(
        self,
        spec: LlamaModelSpec,
        tp: int = 1,
        dtype: torch.dtype = torch.bfloat16,
        layer_range: Optional[tuple] = None,
        tp_group=None,
        sp: bool = False,
    ):
        super().__init__()
        self.spec = spec
        self.tp = tp
        self.tp_group = tp_group
        self.sp = sp and tp > 1
        if self.sp:
            assert spec.seq_length % tp == 0, "sp needs seq % tp == 0"
        total = spec.profile_num_layers
        start, end = layer_range if layer_range is not None else (0, total)
        self.has_embedding = start == 0
        self.has_head = end == total

        h = spec.hidden_size
        if self.has_embedding:
            self.wte = nn.Embedding(spec.vocab_size, h, dtype=dtype)
            nn.init.normal_(self.wte.weight, std=0.02)

        block_start = max(start - 1, 0)
        block_end = min(end, total - 1) - 1
        self.blocks = nn.ModuleList(
            LlamaBlock(spec, tp, dtype)
            for _ in range(max(block_end - block_start, 0))
        )
        if self.has_head:
            self.norm_final = RMSNorm(h)
            self.head = ColumnParallelLinear(h, spec.vocab_size, tp, dtype)

        self.recompute = False  # see GPTModel.recompute

        # SP partial-grad tags for replicated params that see only a seq
        # slice (see GPTModel; llama adds rmsnorm weights + rp biases)
        if self.sp and tp_group is not None:
            hooked = []
            if self.has_embedding:
                hooked.append(self.wte.weight)
            for blk in self.blocks:
                hooked += [blk.norm_attn.weight, blk.norm_mlp.weight,
                           blk.proj.bias, blk.down.bias]
            if self.has_head:
                hooked.append(self.norm_final.weight)

            for p in hooked:
                mark_partial(p, tp_group)

    def forward(self, x, labels=None, cache=None, pos_offset: int = 0):
        if self.has_embedding:
            if self.sp and cache is None:
                r = dist.get_rank(self.tp_group)
                ss = x.size(1) // self.tp
                x = x[:, r * ss:(r + 1) * ss]
            x = self.wte(x)
        use_ckpt = self.recompute and torch.is_grad_enabled()
        for i, block in enumerate(self.blocks):
            if cache is not None:
                x = block(x, self.tp_group, cache=cache, layer_idx=i,
                          pos_offset=pos_offset)
            elif use_ckpt:
                x = torch.utils.checkpoint.checkpoint(
                    block, x, self.tp_group, self.sp, use_reentrant=False)
            else:
                x = block(x, self.tp_group, self.sp)
        if self.has_head:
            if self.sp and cache is None:
                import torch.nn.functional as F

                x = _GatherSeq.apply(x, self.tp_group)
                x = self.norm_final(x)
                logits = F.linear(x, self.head.weight, self.head.bias)
            else:
                x = self.norm_final(x)
                logits = self.head(x, self.tp_group)
            if labels is not None:
                labels = labels.reshape(-1)
                if self.tp_group is not None and dist.get_world_size(self.tp_group) > 1:
                    return vocab_parallel_ce(logits.view(-1, logits.size(-1)),
                                             labels, self.tp_group)
                return cross_entropy(logits.view(-1, logits.size(-1)), labels)
            return logits
        return x

    def layer_parameter_bytes(self) -> List[float]:
        spec = self.spec
        el = 2
        h, v, f = spec.hidden_size, spec.vocab_size, spec.ffn_hidden_size
        d = spec.head_dim
        kvh = spec.num_kv_heads * d
        embed = v * h * el
        per_block = (h * h + 2 * h * kvh + h * h + 3 * h * f + 2 * h) * el
        head = (v * h + h) * el
        return [float(embed)] + [float(per_block)] * spec.num_layers + [float(head)]
