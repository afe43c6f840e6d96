"""GPT model family with tensor-parallel sharding over RCCL.

Megatron-style TP: QKV / fc1 are column-parallel (f: identity fwd,
all-reduce bwd), proj / fc2 are row-parallel (g: all-reduce fwd, identity
bwd); one all-reduce per attention block and one per MLP block in each
direction. LayerNorm runs the fused gfx950 kernel (metis_amd.ops); GEMMs
go through torch.matmul (hipBLASLt on ROCm); attention uses
scaled_dot_product_attention until the hand-written flash kernel wires in.

The layer numbering matches the Metis profile convention: layer 0 =
embedding, 1..n-2 = transformer blocks, n-1 = LM head (+ final norm).
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import List, Optional

import torch
import torch.utils.checkpoint
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from metis_amd.ops import LayerNorm
from metis_amd.ops.attention import decode_attention, flash_attention
from metis_amd.ops.cross_entropy import cross_entropy
from metis_amd.ops.mlp import fused_mlp
from metis_amd.ops.relayout import heads_merge, qkv_split_transpose
from metis_amd.partial_grads import mark_partial

import os as _os


@dataclass(frozen=True)
class GPTModelSpec:
    name: str
    hidden_size: int
    num_layers: int          # transformer blocks
    num_heads: int
    vocab_size: int
    seq_length: int
    ffn_hidden_size: Optional[int] = None
    num_kv_heads: Optional[int] = None

    @property
    def ffn(self) -> int:
        return self.ffn_hidden_size or 4 * self.hidden_size

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_heads

    @property
    def profile_num_layers(self) -> int:
        """Layer count in the Metis profile convention (embed + blocks + head)."""
        return self.num_layers + 2

    def num_parameters(self) -> int:
        h, v = self.hidden_size, self.vocab_size
        per_block = 4 * h * h + 2 * h * self.ffn + 4 * h + self.ffn + h + 2 * h
        return v * h + self.seq_length * h + self.num_layers * per_block + 2 * h


MODEL_SPECS = {
    "gpt2-small": GPTModelSpec("gpt2-small", 768, 12, 12, 51200, 1024),
    "gpt3-1.3b": GPTModelSpec("gpt3-1.3b", 2048, 24, 16, 51200, 2048),
    "gpt3-2.7b": GPTModelSpec("gpt3-2.7b", 2560, 32, 32, 51200, 2048),
    "gpt3-6.7b": GPTModelSpec("gpt3-6.7b", 4096, 32, 32, 51200, 2048),
    # 10-layer GPT-3-shaped config matching the reference's bundled profiles
    "gpt3-10l": GPTModelSpec("gpt3-10l", 4096, 8, 32, 51200, 1024),
}


# --- tensor-parallel autograd collectives ---------------------------------
class _CopyToTP(torch.autograd.Function):
    """f: identity forward, grad all-reduce backward (column-parallel in)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, grad):
        if ctx.group is not None and dist.get_world_size(ctx.group) > 1:
            grad = grad.contiguous()
            dist.all_reduce(grad, group=ctx.group)
        return grad, None


class _ReduceFromTP(torch.autograd.Function):
    """g: all-reduce forward, identity backward (row-parallel out)."""

    @staticmethod
    def forward(ctx, x, group):
        if group is not None and dist.get_world_size(group) > 1:
            x = x.contiguous()
            dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, grad):
        return grad, None


# --- sequence-parallel collectives (Megatron-SP; opt-in via sp=True) ------
# With SP the norm/residual regions hold [b, s/tp, h] instead of the full
# sequence: entering a column-parallel layer the shard is all-gathered
# along seq (backward: reduce-scatter), and the row-parallel output is
# reduce-scattered (backward: all-gather). Same wire bytes as the TP
# ring all-reduce (ar = rs + ag) but the replicated-region activations
# shrink by 1/tp. gloo has no reduce-scatter, so CPU tests emulate it
# with all-reduce + slice (numerically identical).

def _rs_seq(x: torch.Tensor, group) -> torch.Tensor:
    """reduce-scatter along dim 1: [b, s, h] (partial sums) -> [b, s/ws, h]."""
    ws = dist.get_world_size(group)
    r = dist.get_rank(group)
    sl = x.size(1) // ws
    if dist.get_backend(group) == "gloo":
        x = x.contiguous()
        dist.all_reduce(x, group=group)
        return x[:, r * sl:(r + 1) * sl].contiguous()
    # [b, s, h] -> [ws, b, sl, h] so reduce_scatter_tensor splits on dim 0
    parts = x.reshape(x.size(0), ws, sl, x.size(2)).permute(1, 0, 2, 3).contiguous()
    out = torch.empty_like(parts[0])
    dist.reduce_scatter_tensor(out, parts, group=group)
    return out


def _ag_seq(x: torch.Tensor, group) -> torch.Tensor:
    """all-gather along dim 1: [b, s/ws, h] -> [b, s, h]."""
    ws = dist.get_world_size(group)
    x = x.contiguous()
    if dist.get_backend(group) == "gloo":
        parts = [torch.empty_like(x) for _ in range(ws)]
        dist.all_gather(parts, x, group=group)
        return torch.cat(parts, dim=1)
    out = torch.empty(ws, x.size(0), x.size(1), x.size(2),
                      dtype=x.dtype, device=x.device)
    dist.all_gather_into_tensor(out, x, group=group)
    return out.permute(1, 0, 2, 3).reshape(x.size(0), ws * x.size(1), x.size(2))


class _GatherSeq(torch.autograd.Function):
    """fwd: seq all-gather; bwd: seq reduce-scatter (SP's f operator)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        if group is None or dist.get_world_size(group) <= 1:
            return x
        return _ag_seq(x, group)

    @staticmethod
    def backward(ctx, grad):
        if ctx.group is None or dist.get_world_size(ctx.group) <= 1:
            return grad, None
        return _rs_seq(grad, ctx.group), None


class _ReduceScatterSeq(torch.autograd.Function):
    """fwd: seq reduce-scatter; bwd: seq all-gather (SP's g operator)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        if group is None or dist.get_world_size(group) <= 1:
            return x
        return _rs_seq(x, group)

    @staticmethod
    def backward(ctx, grad):
        if ctx.group is None or dist.get_world_size(ctx.group) <= 1:
            return grad, None
        return _ag_seq(grad.contiguous(), ctx.group), None


class _VocabParallelCrossEntropy(torch.autograd.Function):
    """Cross entropy over vocab-sharded logits without gathering them
    (Megatron-style): three small all-reduces (max, sum-exp, target logit)
    instead of moving the [tokens, vocab] tensor."""

    @staticmethod
    def forward(ctx, logits_shard, labels, group):
        rank = dist.get_rank(group)
        vp = logits_shard.size(-1)
        vocab_start = rank * vp

        m = logits_shard.max(dim=-1).values
        dist.all_reduce(m, op=dist.ReduceOp.MAX, group=group)
        exp = torch.exp(logits_shard - m[:, None])
        sumexp = exp.sum(dim=-1)
        dist.all_reduce(sumexp, group=group)

        local = (labels >= vocab_start) & (labels < vocab_start + vp)
        idx = (labels - vocab_start).clamp(0, vp - 1)
        target_logit = torch.where(
            local, logits_shard.gather(1, idx[:, None]).squeeze(1),
            torch.zeros_like(m),
        )
        dist.all_reduce(target_logit, group=group)

        loss = (torch.log(sumexp) + m - target_logit).mean()
        ctx.save_for_backward(exp, sumexp, local, idx)
        return loss

    @staticmethod
    def backward(ctx, grad_out):
        exp, sumexp, local, idx = ctx.saved_tensors
        n = exp.size(0)
        softmax = exp / sumexp[:, None]
        softmax.scatter_add_(
            1, idx[:, None],
            torch.where(local, -torch.ones_like(sumexp), torch.zeros_like(sumexp))[:, None],
        )
        return softmax * (grad_out / n), None, None


class _VocabParallelCrossEntropyBF16(torch.autograd.Function):
    """tp>1 cross entropy streaming the bf16 logits shard through the
    ce_row_max / ce_row_sumexp / cross_entropy_bwd kernels: the fp32
    [T, V/tp] copy is never materialized. Same Megatron flow (three small
    all-reduces) as _VocabParallelCrossEntropy; opt-in via METIS_VP_CE=1
    until GPU-validated."""

    @staticmethod
    def forward(ctx, logits_shard, labels, group):
        from metis_amd import ops as _mops

        ext = _mops.require_extension()
        rank = dist.get_rank(group)
        vp = logits_shard.size(-1)
        vocab_start = rank * vp

        m = ext.ce_row_max(logits_shard)
        dist.all_reduce(m, op=dist.ReduceOp.MAX, group=group)
        sumexp = ext.ce_row_sumexp(logits_shard, m)
        dist.all_reduce(sumexp, group=group)

        local = (labels >= vocab_start) & (labels < vocab_start + vp)
        idx = (labels - vocab_start).clamp(0, vp - 1)
        target = torch.where(
            local, logits_shard.gather(1, idx[:, None]).squeeze(1).float(),
            torch.zeros_like(m))
        dist.all_reduce(target, group=group)

        loss = (torch.log(sumexp) + m - target).mean()
        shifted = torch.where(local, idx, torch.full_like(idx, -1))
        ctx.save_for_backward(logits_shard, shifted, m, sumexp)
        return loss

    @staticmethod
    def backward(ctx, grad_out):
        from metis_amd import ops as _mops

        ext = _mops.require_extension()
        logits, shifted, m, sumexp = ctx.saved_tensors
        lse = m + torch.log(sumexp)
        scale = (grad_out.float() / logits.size(0)).reshape(1)
        dlogits = ext.cross_entropy_bwd(logits, shifted, lse, scale)
        return dlogits, None, None


def vocab_parallel_ce(logits2d: torch.Tensor, labels: torch.Tensor,
                      group) -> torch.Tensor:
    """Vocab-sharded CE; bf16-streaming kernel path under METIS_VP_CE=1."""
    if (logits2d.is_cuda and logits2d.dtype == torch.bfloat16
            and _os.environ.get("METIS_VP_CE") == "1"):
        return _VocabParallelCrossEntropyBF16.apply(
            logits2d.contiguous(), labels, group)
    return _VocabParallelCrossEntropy.apply(
        logits2d.float().contiguous(), labels, group)


def _init_linear(weight: torch.Tensor, fan_in: int) -> None:
    std = 1.0 / math.sqrt(fan_in)
    nn.init.normal_(weight, mean=0.0, std=std)


class ColumnParallelLinear(nn.Module):
    """y_shard = x @ W_shard^T + b_shard, W sharded over output dim."""

    def __init__(self, in_features: int, out_features: int, tp: int, dtype):
        super().__init__()
        assert out_features % tp == 0
        self.out_per_rank = out_features // tp
        self.weight = nn.Parameter(torch.empty(self.out_per_rank, in_features, dtype=dtype))
        self.bias = nn.Parameter(torch.zeros(self.out_per_rank, dtype=dtype))
        _init_linear(self.weight, in_features)

    def forward(self, x: torch.Tensor, tp_group) -> torch.Tensor:
        x = _CopyToTP.apply(x, tp_group)
        return F.linear(x, self.weight, self.bias)


class RowParallelLinear(nn.Module):
    """y = all_reduce(x_shard @ W_shard^T) + b, W sharded over input dim."""

    def __init__(self, in_features: int, out_features: int, tp: int, dtype):
        super().__init__()
        assert in_features % tp == 0
        self.in_per_rank = in_features // tp
        self.weight = nn.Parameter(torch.empty(out_features, self.in_per_rank, dtype=dtype))
        self.bias = nn.Parameter(torch.zeros(out_features, dtype=dtype))
        _init_linear(self.weight, in_features)

    def forward(self, x: torch.Tensor, tp_group) -> torch.Tensor:
        y = F.linear(x, self.weight)
        y = _ReduceFromTP.apply(y, tp_group)
        return y + self.bias


class GPTBlock(nn.Module):
    def __init__(self, spec: GPTModelSpec, tp: int, dtype):
        super().__init__()
        h = spec.hidden_size
        assert spec.num_heads % tp == 0, "num_heads must divide by tp"
        self.heads_per_rank = spec.num_heads // tp
        self.head_dim = spec.head_dim

        self.ln_attn = LayerNorm(h)
        self.qkv = ColumnParallelLinear(h, 3 * h, tp, dtype)
        self.proj = RowParallelLinear(h, h, tp, dtype)
        self.ln_mlp = LayerNorm(h)
        self.fc1 = ColumnParallelLinear(h, spec.ffn, tp, dtype)
        self.fc2 = RowParallelLinear(spec.ffn, h, tp, dtype)
        # hipBLASLt epilogue-fused MLP (ops/mlp.py); opt-in until the
        # GPU numerics of the GELU_AUX_BIAS/DGELU_BGRAD path are validated
        self._lt_mlp = _os.environ.get("METIS_FC1_EPILOGUE") == "1"


    def _cached_attention(self, q, k, v, mask=None):
        """Decode-path attention over cached K/V: q holds only the new
        tokens (rightmost positions). q_len == 1 attends everything (or
        the ragged cache's padding mask); a longer new chunk gets a
        prefix+causal mask."""
        new, total = q.size(2), k.size(2)
        if mask is not None:
            return F.scaled_dot_product_attention(q, k, v, attn_mask=mask)
        if new == 1:
            return decode_attention(q, k, v)
        mask = torch.ones(new, total, dtype=torch.bool, device=q.device)
        mask = torch.tril(mask, diagonal=total - new)
        return F.scaled_dot_product_attention(q, k, v, attn_mask=mask)

    def forward(self, x: torch.Tensor, tp_group, sp: bool = False,
                cache=None, layer_idx: int = 0) -> torch.Tensor:
        """With ``sp`` the block holds [b, s/tp, h] in the norm/residual
        regions: the f/g operators become seq all-gather / reduce-scatter
        (see the SP collectives above); attention always sees the full
        sequence. With ``cache`` (inference) the block runs incrementally:
        K/V of new tokens are appended to the cache and attention spans
        the cached prefix."""
        if cache is not None:
            residual = x
            y = self.ln_attn(x)
            qkv = self.qkv(y, tp_group)
            q, k, v = qkv_split_transpose(
                qkv, self.heads_per_rank, self.heads_per_rank, self.head_dim)
            k, v = cache.append(layer_idx, k, v)
            am = getattr(cache, "attention_mask", None)
            mask = am(k.size(2), q.device) if am is not None else None
            attn = self._cached_attention(q, k, v, mask)
            x = residual + self.proj(heads_merge(attn), tp_group)
            residual = x
            y = self.fc1(self.ln_mlp(x), tp_group)
            x = residual + self.fc2(F.gelu(y, approximate="tanh"), tp_group)
            return x

        residual = x
        y = self.ln_attn(x)
        # per-rank qkv layout: [q heads | k heads | v heads] blocks
        if sp:
            y_full = _GatherSeq.apply(y, tp_group)
            qkv = F.linear(y_full, self.qkv.weight, self.qkv.bias)
        else:
            qkv = self.qkv(y, tp_group)
        q, k, v = qkv_split_transpose(
            qkv, self.heads_per_rank, self.heads_per_rank, self.head_dim
        )
        attn = flash_attention(q, k, v, causal=True)
        if sp:
            part = F.linear(heads_merge(attn), self.proj.weight)
            part = _ReduceScatterSeq.apply(part, tp_group)
            x = residual + part + self.proj.bias
        else:
            x = residual + self.proj(heads_merge(attn), tp_group)

        residual = x
        y = self.ln_mlp(x)
        if self._lt_mlp:
            inp = (_GatherSeq.apply(y, tp_group) if sp
                   else _CopyToTP.apply(y, tp_group))
            b, s, hh = inp.shape
            flat = inp.reshape(-1, hh)
            part = fused_mlp(flat, self.fc1.weight, self.fc1.bias,
                             self.fc2.weight).reshape(b, s, hh)
            part = (_ReduceScatterSeq.apply(part, tp_group) if sp
                    else _ReduceFromTP.apply(part, tp_group))
            x = residual + part + self.fc2.bias
        elif sp:
            y_full = _GatherSeq.apply(y, tp_group)
            t = F.gelu(F.linear(y_full, self.fc1.weight, self.fc1.bias),
                       approximate="tanh")
            part = _ReduceScatterSeq.apply(F.linear(t, self.fc2.weight),
                                           tp_group)
            x = residual + part + self.fc2.bias
        else:
            y = self.fc1(y, tp_group)
            y = F.gelu(y, approximate="tanh")
            x = residual + self.fc2(y, tp_group)
        return x


class GPTModel(nn.Module):
    """A pipeline-stage slice of the GPT model.

    ``layer_range`` selects profile-convention layers [start, end) out of
    [0, num_layers+2): owning layer 0 adds the embeddings, owning the last
    layer adds the final norm + LM head.
    """

    def __init__(
        self,
        spec: GPTModelSpec,
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
        assert 0 <= start < end <= total
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
            GPTBlock(spec, tp, dtype) for _ in range(max(block_end - block_start, 0))
        )

        if self.has_head:
            self.ln_final = LayerNorm(h)
            self.head = ColumnParallelLinear(h, spec.vocab_size, tp, dtype)

        # activation recomputation: re-run each block's forward during
        # backward instead of keeping its activations (trades ~30% extra
        # compute for O(1) per-block activation memory)
        self.recompute = False

        # SP: params in the seq-sharded regions are replicated but see
        # only this rank's token slice, so their grads are partial — tag
        # them for the cross-TP sum (metis_amd.partial_grads: immediate
        # hook standalone, once-per-step under the runner)
        if self.sp and tp_group is not None:
            hooked = []
            if self.has_embedding:
                hooked += [self.wte.weight, self.wpe.weight]
            for blk in self.blocks:
                hooked += [blk.ln_attn.weight, blk.ln_attn.bias,
                           blk.ln_mlp.weight, blk.ln_mlp.bias,
                           blk.proj.bias, blk.fc2.bias]
            if self.has_head:
                # ln_final runs on the full sequence but its upstream grad
                # is the pre-reduce-scatter partial (the head bypasses f)
                hooked += [self.ln_final.weight, self.ln_final.bias]

            for p in hooked:
                mark_partial(p, tp_group)

    def forward(
        self, x: torch.Tensor, labels: Optional[torch.Tensor] = None,
        cache=None, pos_offset: int = 0,
    ) -> torch.Tensor:
        """x: token ids [b, s] on the first stage, hidden states elsewhere
        ([b, s/tp, h] between blocks when sp). Returns the loss when this
        stage has the head and labels are given, otherwise the stage's
        output hidden states. ``cache``/``pos_offset`` select the
        incremental KV-cache inference path (runtime.generate)."""
        if self.has_embedding:
            b, s = x.shape
            if cache is not None:
                if torch.is_tensor(pos_offset):   # ragged: per-row positions
                    pos = (pos_offset.to(x.device)[:, None]
                           + torch.arange(s, device=x.device)[None])
                else:
                    pos = torch.arange(pos_offset, pos_offset + s,
                                       device=x.device)
            elif self.sp:
                # embed only this rank's sequence slice
                r = dist.get_rank(self.tp_group)
                ss = s // self.tp
                x = x[:, r * ss:(r + 1) * ss]
                pos = torch.arange(r * ss, (r + 1) * ss, device=x.device)
            else:
                pos = torch.arange(s, device=x.device)
            wpe = self.wpe(pos)
            x = self.wte(x) + (wpe if wpe.dim() == 3 else wpe[None, :, :])

        use_ckpt = self.recompute and torch.is_grad_enabled()
        for i, block in enumerate(self.blocks):
            if cache is not None:
                x = block(x, self.tp_group, cache=cache, layer_idx=i)
            elif use_ckpt:
                x = torch.utils.checkpoint.checkpoint(
                    block, x, self.tp_group, self.sp, use_reentrant=False)
            else:
                x = block(x, self.tp_group, self.sp)

        if self.has_head:
            if self.sp:
                # the head + loss work on the full sequence; the gather
                # REPLACES the head's f operator (its backward
                # reduce-scatter does the cross-rank sum — going through
                # _CopyToTP too would double the gradient)
                x = _GatherSeq.apply(x, self.tp_group)
                x = self.ln_final(x)
                logits = F.linear(x, self.head.weight, self.head.bias)
            else:
                x = self.ln_final(x)
                logits = self.head(x, self.tp_group)  # [b, s, vocab/tp]
            if labels is not None:
                return self._loss(logits, labels)
            return logits
        return x

    def _loss(self, logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
        labels = labels.reshape(-1)
        if self.tp_group is not None and dist.get_world_size(self.tp_group) > 1:
            return vocab_parallel_ce(logits.view(-1, logits.size(-1)),
                                     labels, self.tp_group)
        return cross_entropy(logits.view(-1, logits.size(-1)), labels)

    def layer_parameter_bytes(self) -> List[float]:
        """parameters_per_layer_bytes for the profile JSON (full model)."""
        spec = self.spec
        el = 2  # bf16
        h, v_ = spec.hidden_size, spec.vocab_size
        embed = (v_ * h + spec.seq_length * h) * el
        per_block = (4 * h * h + 2 * h * spec.ffn + 4 * h + spec.ffn + h + 2 * h) * el
        head = (v_ * h + 2 * h) * el
        return [float(embed)] + [float(per_block)] * spec.num_layers + [float(head)]
