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


class RaggedKVCache:
    """Padded per-block K/V buffers for a decode batch whose sequences
    have DIFFERENT lengths: row i's valid prefix is lengths[i], new
    tokens land at per-row positions, and attention_mask() hides the
    padding. Built by generate_ragged() from per-sequence prefills."""

    def __init__(self, lengths: torch.Tensor):
        self.lengths = lengths.clone()          # [b] current per-row length
        self._kv = {}

    @classmethod
    def from_prefills(cls, caches, lengths, max_new: int):
        """Merge per-sequence KVCache objects (batch 1 each) into padded
        [b, h, max_len + max_new, d] buffers."""
        self = cls(torch.as_tensor(lengths))
        b = len(caches)
        total = int(self.lengths.max()) + max_new
        for idx in caches[0]._kv:
            k0, v0 = caches[0]._kv[idx]
            kbuf = k0.new_zeros(b, k0.size(1), total, k0.size(3))
            vbuf = torch.zeros_like(kbuf)
            for i, c in enumerate(caches):
                k, v = c._kv[idx]
                kbuf[i, :, :k.size(2)] = k[0]
                vbuf[i, :, :v.size(2)] = v[0]
            self._kv[idx] = (kbuf, vbuf)
        return self

    def append(self, layer_idx: int, k: torch.Tensor, v: torch.Tensor):
        assert k.size(2) == 1, "ragged cache decodes one token at a time"
        kbuf, vbuf = self._kv[layer_idx]
        rows = torch.arange(k.size(0), device=k.device)
        kbuf[rows, :, self.lengths] = k[:, :, 0]
        vbuf[rows, :, self.lengths] = v[:, :, 0]
        upto = int(self.lengths.max()) + 1
        return kbuf[:, :, :upto], vbuf[:, :, :upto]

    def attention_mask(self, total: int, device) -> torch.Tensor:
        """[b, 1, 1, total] bool: row i may attend cols <= lengths[i]
        (its prefix plus the token just appended)."""
        cols = torch.arange(total, device=device)
        return (cols[None] <= self.lengths.to(device)[:, None])[:, None, None]

    def advance(self) -> None:
        self.lengths += 1

    @property
    def seq_len(self):   # per-row positions for the NEXT forward
        return self.lengths


@torch.no_grad()
def generate_ragged(
    model,
    prompts,
    max_new_tokens: int,
    temperature: float = 1.0,
    top_k: int = 0,
    generator: Optional[torch.Generator] = None,
    device=None,
):
    """Batched decoding of prompts with different lengths: each prompt is
    prefilled separately (exact per-sequence caches), then all rows
    decode together against a padded ragged cache. Returns a list of
    token lists (prompt + continuation). Single process (tp=pp=1)."""
    model.eval()
    lengths = [len(p) for p in prompts]
    caches, last = [], []
    for p in prompts:
        c = KVCache()
        toks = torch.tensor([p], dtype=torch.long, device=device)
        logits = model(toks, cache=c, pos_offset=0)
        caches.append(c)
        last.append(logits[0, -1])
    cache = RaggedKVCache.from_prefills(caches, lengths, max_new_tokens)
    if device is not None:
        cache.lengths = cache.lengths.to(device)
    logits = torch.stack(last)                   # [b, vocab]
    outs = [list(p) for p in prompts]
    for _ in range(max_new_tokens):
        nxt = _sample(logits.float(), temperature, top_k, generator)
        for i, t in enumerate(nxt.tolist()):
            outs[i].append(t[0])
        pos = cache.lengths.clone()
        logits = model(nxt.to(cache.lengths.device), cache=cache,
                       pos_offset=pos)[:, -1]
        cache.advance()
    return outs


def _sample(last: torch.Tensor, temperature: float, top_k: int,
            generator) -> torch.Tensor:
    if temperature <= 0:
        return last.argmax(dim=-1, keepdim=True)
    last = last / temperature
    if top_k:
        kth = last.topk(top_k, dim=-1).values[:, -1:]
        last = last.masked_fill(last < kth, float("-inf"))
    return torch.multinomial(torch.softmax(last, dim=-1), 1,
                             generator=generator)


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
