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


class ContinuousBatcher:
    """Continuous batching over the padded ragged cache: requests join
    and leave the decode batch between steps instead of waiting for a
    whole batch to finish (the serving pattern vLLM popularized; the
    reference has no inference path at all).

    Fixed ``max_batch`` slots share per-layer K/V buffers of
    ``capacity`` positions. ``submit()`` queues a request; each
    ``step()`` admits pending requests into free slots (one exact
    batch-1 prefill each, spliced into the slot's buffer rows), decodes
    ONE token for every active slot, and retires rows that hit their
    token budget or ``eos_id``. Retired slots are reused immediately —
    a long request no longer blocks short ones behind it.

    Single process (tp=pp=1), device-agnostic (CPU tests; the GPU path
    uses the default-on single-query decode kernel via the models).
    """

    def __init__(self, model, capacity: int, max_batch: int = 8,
                 device=None, eos_id: Optional[int] = None):
        self.model = model
        self.capacity = capacity
        self.max_batch = max_batch
        self.device = device
        self.eos_id = eos_id
        self.cache: Optional[RaggedKVCache] = None
        self._pending = []      # submitted, not yet admitted
        self._slots = [None] * max_batch   # slot -> request state or None
        self._next_id = 0
        self._last_logits = {}  # slot -> [vocab] logits of last position
        self.finished = {}      # req_id -> token list

    def submit(self, prompt, max_new_tokens: int, temperature: float = 0.0,
               top_k: int = 0, generator=None) -> int:
        assert len(prompt) > 0
        assert len(prompt) + max_new_tokens <= self.capacity, (
            "prompt + budget exceeds cache capacity")
        rid = self._next_id
        self._next_id += 1
        self._pending.append(dict(
            rid=rid, prompt=list(prompt), max_new=max_new_tokens,
            temperature=temperature, top_k=top_k, generator=generator,
            out=list(prompt)))
        return rid

    @property
    def active(self) -> int:
        return sum(s is not None for s in self._slots)

    def _ensure_cache(self, proto_cache: "KVCache") -> None:
        if self.cache is not None:
            return
        self.cache = RaggedKVCache(torch.zeros(self.max_batch,
                                               dtype=torch.long))
        self.cache.active = torch.zeros(self.max_batch, dtype=torch.bool)
        for idx, (k, v) in proto_cache._kv.items():
            kbuf = k.new_zeros(self.max_batch, k.size(1), self.capacity,
                               k.size(3))
            self.cache._kv[idx] = (kbuf, torch.zeros_like(kbuf))
        if self.device is not None:
            self.cache.lengths = self.cache.lengths.to(self.device)
            self.cache.active = self.cache.active.to(self.device)

    def _admit(self) -> None:
        while self._pending and self.active < self.max_batch:
            req = self._pending.pop(0)
            slot = self._slots.index(None)
            toks = torch.tensor([req["prompt"]], dtype=torch.long,
                                device=self.device)
            pre = KVCache()
            logits = self.model(toks, cache=pre, pos_offset=0)
            self._ensure_cache(pre)
            plen = len(req["prompt"])
            for idx, (k, v) in pre._kv.items():
                kbuf, vbuf = self.cache._kv[idx]
                kbuf[slot, :, :plen] = k[0]
                vbuf[slot, :, :plen] = v[0]
            self.cache.lengths[slot] = plen
            self.cache.active[slot] = True
            self._slots[slot] = req
            self._last_logits[slot] = logits[0, -1]

    def step(self):
        """Admit + decode one token for every active row; returns the
        dict of requests that finished THIS step ({rid: tokens})."""
        self._admit()
        done_now = {}
        if self.active == 0:
            return done_now
        # sample per row (per-request params), inactive rows decode a
        # dummy token that the mask and retirement logic ignore
        nxt = torch.zeros(self.max_batch, 1, dtype=torch.long,
                          device=self.device)
        for slot, req in enumerate(self._slots):
            if req is None:
                continue
            last = self._last_logits[slot][None].float()
            tok = _sample(last, req["temperature"], req["top_k"],
                          req["generator"])
            req["out"].append(int(tok[0, 0]))
            nxt[slot, 0] = tok[0, 0]
        # batched single-token forward against the padded cache
        pos = self.cache.lengths.clone()
        # inactive rows: write their garbage token at a parked position
        # (their stale length) — overwritten at the next admit, hidden
        # by attention_mask() until then; clamp against the buffer end
        pos = torch.clamp(pos, max=self.capacity - 1)
        self.cache.lengths = pos
        logits = self.model(nxt, cache=self.cache, pos_offset=pos)[:, -1]
        # advance only active rows
        self.cache.lengths = pos + self.cache.active.long()
        for slot, req in enumerate(self._slots):
            if req is None:
                continue
            self._last_logits[slot] = logits[slot]
            tok = req["out"][-1]
            n_new = len(req["out"]) - len(req["prompt"])
            if n_new >= req["max_new"] or (self.eos_id is not None
                                           and tok == self.eos_id):
                done_now[req["rid"]] = req["out"]
                self.finished[req["rid"]] = req["out"]
                self._slots[slot] = None
                self.cache.active[slot] = False
                del self._last_logits[slot]
        return done_now

    def run_until_done(self, max_steps: int = 100000):
        for _ in range(max_steps):
            self.step()
            if self.active == 0 and not self._pending:
                return self.finished
        raise RuntimeError("continuous batcher did not drain")
