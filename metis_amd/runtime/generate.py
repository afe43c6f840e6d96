"""KV-cache incremental decoding (serving path, v1).

The reference is a training planner with no inference at all; this adds
the decode loop the models need to serve: per-block KV caches, one
prefill pass, then one-token incremental forwards. CPU tests prove
incremental logits match a full forward position-for-position; the
decode-optimized attention kernel (single-query flash) is a round-2 GPU
item — this path uses SDPA over the cache.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist


class KVCache:
    """Per-block K/V tensors [b, h, s, d], appended as decoding advances."""

    def __init__(self):
        self._kv = {}

    def append(self, layer_idx: int, k: torch.Tensor, v: torch.Tensor):
        if layer_idx in self._kv:
            pk, pv = self._kv[layer_idx]
            k = torch.cat([pk, k], dim=2)
            v = torch.cat([pv, v], dim=2)
        self._kv[layer_idx] = (k, v)
        return k, v

    @property
    def seq_len(self) -> int:
        if not self._kv:
            return 0
        return next(iter(self._kv.values()))[0].size(2)


@torch.no_grad()
def generate(
    model,
    tokens: torch.Tensor,
    max_new_tokens: int,
    temperature: float = 1.0,
    top_k: int = 0,
    tp_group=None,
    generator: Optional[torch.Generator] = None,
) -> torch.Tensor:
    """Greedy (temperature == 0) or top-k sampling from a full
    (non-pipelined) model; vocab-sharded logits are gathered across the
    TP group before sampling so every rank draws the same token."""
    model.eval()
    cache = KVCache()
    logits = model(tokens, cache=cache, pos_offset=0)   # prefill
    out = tokens
    for _ in range(max_new_tokens):
        last = logits[:, -1].float()                    # [b, vocab/tp]
        if tp_group is not None and dist.get_world_size(tp_group) > 1:
            parts = [torch.empty_like(last)
                     for _ in range(dist.get_world_size(tp_group))]
            dist.all_gather(parts, last.contiguous(), group=tp_group)
            last = torch.cat(parts, dim=-1)
        if temperature <= 0:
            nxt = last.argmax(dim=-1, keepdim=True)
        else:
            last = last / temperature
            if top_k:
                kth = last.topk(top_k, dim=-1).values[:, -1:]
                last = last.masked_fill(last < kth, float("-inf"))
            probs = torch.softmax(last, dim=-1)
            nxt = torch.multinomial(probs, 1, generator=generator)
        out = torch.cat([out, nxt], dim=1)
        logits = model(nxt, cache=cache, pos_offset=cache.seq_len)
    return out
