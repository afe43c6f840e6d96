"""Plan runner: execute a (dp, tp, pp) plan for one training step.

This is the component the reference's planner assumes exists but never
shipped — it runs the chosen plan on real GPUs (synthetic data, random
init) and reports measured iteration time, which feeds
metis_amd.planner.validate (the cost-model-error metric).

Schedule: GPipe (fill-drain), matching the cost model's
(B-1)*max_stage + sum_stages assumption. DP gradient sync is bucketed
and overlapped with the final backward (runtime.grad_sync); the fused
AdamW kernel then consumes the flat fp32 buffer directly.
"""

from __future__ import annotations

import os
import time
from typing import List, Optional, Tuple

import torch
import torch.distributed as dist

from metis_amd.models.gpt import GPTModel, GPTModelSpec
from metis_amd.models.llama import LlamaModel, LlamaModelSpec
from metis_amd.models.moe import MoEModel, MoEModelSpec
from metis_amd.ops import FusedAdamW
from metis_amd.planner.volume import uniform_layer_split
from metis_amd.runtime.clip import global_grad_norm, shard_flags
from metis_amd.runtime.comm import ParallelContext
from metis_amd.runtime.grad_sync import GradBucketSync
from metis_amd.partial_grads import defer_partial, sync_partial_grads
from metis_amd.runtime.trace import tracer_from_env


class PlanRunner:
    def __init__(
        self,
        spec: GPTModelSpec,
        ctx: ParallelContext,
        mbs: int,
        gbs: int,
        layer_partition: Optional[List[int]] = None,
        lr: float = 1e-4,
        dtype: torch.dtype = torch.bfloat16,
        schedule: str = "gpipe",
        recompute: bool = False,
        zero1: bool = False,
        sp: bool = False,
        vpp: int = 1,
        clip_grad: Optional[float] = None,
        data_path: Optional[str] = None,
    ) -> None:
        assert schedule in ("gpipe", "1f1b", "interleaved")
        self.schedule = schedule
        self.vpp = vpp if schedule == "interleaved" else 1
        if schedule == "interleaved":
            assert vpp >= 2, "interleaved needs >= 2 virtual chunks (--vpp)"
            assert ctx.pp > 1, "interleaved needs pp > 1"
        self.spec = spec
        self.ctx = ctx
        self.mbs = mbs
        self.gbs = gbs
        self.num_microbatches = gbs // mbs // ctx.dp
        assert self.num_microbatches >= ctx.pp or ctx.pp == 1, (
            "GPipe needs at least pp microbatches"
        )
        assert gbs % (mbs * ctx.dp) == 0, "gbs must divide by mbs*dp"
        if schedule == "interleaved":
            assert self.num_microbatches % ctx.pp == 0, (
                "interleaved schedule needs num_microbatches % pp == 0")

        total_layers = spec.profile_num_layers
        num_stages = ctx.pp * self.vpp
        if layer_partition is None:
            counts = uniform_layer_split(total_layers, num_stages)
            layer_partition = [0]
            for c in counts:
                layer_partition.append(layer_partition[-1] + c)
        assert len(layer_partition) == num_stages + 1
        self.layer_partition = layer_partition

        if isinstance(spec, MoEModelSpec):
            model_cls = MoEModel
        elif isinstance(spec, LlamaModelSpec):
            model_cls = LlamaModel
        else:
            model_cls = GPTModel
        sp_capable = model_cls in (GPTModel, LlamaModel)
        self.sp = bool(sp) and ctx.tp > 1 and sp_capable
        extra = {"sp": self.sp} if sp_capable else {}
        assert not (sp and not self.sp and ctx.tp > 1), (
            "sequence parallelism is implemented for the GPT and Llama "
            "families")

        # one model slice per virtual chunk: chunk c covers virtual stage
        # c*pp + pp_rank (vpp == 1 -> the ordinary single slice)
        chunks = []
        for c in range(self.vpp):
            vs = c * ctx.pp + ctx.pp_rank
            start, end = layer_partition[vs], layer_partition[vs + 1]
            chunks.append(model_cls(
                spec, tp=ctx.tp, dtype=dtype, layer_range=(start, end),
                tp_group=ctx.tp_group, **extra,
            ))
        self.model_chunks = chunks
        self.model = chunks[0] if self.vpp == 1 else torch.nn.ModuleList(chunks)
        for chunk in chunks:
            chunk.recompute = recompute
        if ctx.device is not None:
            self.model.to(ctx.device)
        shard_group = ctx.dp_group if (zero1 and ctx.dp > 1) else None
        self.optimizer = FusedAdamW(self.model.parameters(), lr=lr,
                                    shard_group=shard_group)
        self.clip_grad = clip_grad
        if clip_grad is not None:
            # classify each optimizer param: True if its elements are a
            # distinct shard per TP/EP rank (column/row-parallel weights,
            # expert weights), False if replicated (norms, embeddings,
            # routers, row-parallel biases) — the global grad norm sums
            # sharded contributions ACROSS the tp group but replicated
            # ones once
            flags = shard_flags(self.model)
            self._grad_sharded = [flags.get(id(p), False)
                                  for p in self.optimizer.params]
        self.dtype = dtype
        self.tracer = tracer_from_env(ctx.rank)
        self._check_sync = os.environ.get("METIS_CHECK_SYNC") == "1"
        _sw = os.environ.get("METIS_STRAGGLER_WARN")
        self._straggler_warn = float(_sw) if _sw else 0.0
        # METIS_HIPGRAPH=1: capture the pp==1 fwd+bwd as one hipGraph
        # (torch.cuda.make_graphed_callables) and replay it per
        # microbatch. Capture-validated on MI355X; measured SLOWER at
        # gpt2-small mbs1 (41.5 vs 39.0 ms/step — replay floor + input
        # copies beat the launch savings), so opt-in and off
        # (BENCHMARKS.md "hipGraph capture"). Requires static shapes
        # (always true here); tp=1 keeps collectives out of the graph.
        self._use_hipgraph = (os.environ.get("METIS_HIPGRAPH") == "1"
                              and ctx.pp == 1 and ctx.tp == 1
                              and torch.cuda.is_available())
        self._graphed = None
        self._data_gen = None
        # real-data path: memory-mapped token file with the same
        # determinism contract as the synthetic stream (data/dataset.py)
        self.data_loader = None
        if data_path is not None:
            from metis_amd.data import TokenDataset, TokenLoader

            self.data_loader = TokenLoader(
                TokenDataset(data_path, spec.seq_length), mbs=mbs,
                dp=ctx.dp, dp_rank=ctx.dp_rank,
                device=ctx.device or torch.device("cpu"))
        # bucketed, overlapped DP gradient all-reduce (dp > 1 only: the
        # per-parameter Python hooks cost more than the serial gather saves
        # when there is no collective to overlap — measured +40 ms/step)
        self.grad_sync = None
        if ctx.dp > 1 and ctx.dp_group is not None:
            self.grad_sync = GradBucketSync(self.optimizer, ctx.dp_group, ctx.dp)
        else:
            # the runner accumulates over microbatches: replicated
            # partial-grad params (MoE router / SP norms) must sum across
            # their group once per step, not per backward — defer the
            # model-level hooks and do it in _sync_and_step
            defer_partial(self.optimizer.params)

    # --- data -------------------------------------------------------------
    def synthetic_batch(self) -> Tuple[torch.Tensor, torch.Tensor]:
        """Random tokens + next-token labels of the plan's microbatch shape.

        Seeded per DP replica: all TP (and PP) ranks of one replica must
        draw the SAME tokens, or tensor-parallel math silently diverges
        across ranks; different DP replicas draw different data."""
        dev = self.ctx.device or torch.device("cpu")
        if self._data_gen is None:
            self._data_gen = torch.Generator(device=dev)
            self._data_gen.manual_seed(12345 + self.ctx.dp_rank)
        tokens = torch.randint(
            0, self.spec.vocab_size, (self.mbs, self.spec.seq_length),
            device=dev, generator=self._data_gen,
        )
        labels = torch.roll(tokens, -1, dims=1)
        return tokens, labels

    def next_batch(self) -> Tuple[torch.Tensor, torch.Tensor]:
        if self.data_loader is not None:
            return self.data_loader.next_batch()
        return self.synthetic_batch()

    # --- single-stage step (pp == 1) --------------------------------------
    def _forward(self, tokens, labels):
        if self._use_hipgraph:
            if self._graphed is None:
                # capture once at first use (shapes are static); the
                # graphed callable copies fresh inputs into its static
                # buffers and replays fwd (and bwd on .backward())
                self._graphed = torch.cuda.make_graphed_callables(
                    self.model, (tokens, labels))
            return self._graphed(tokens, labels)
        return self.model(tokens, labels=labels)

    def _step_no_pipeline(self) -> float:
        losses = []
        self.optimizer.zero_grad()
        for mb in range(self.num_microbatches):
            if self.grad_sync is not None and mb == self.num_microbatches - 1:
                self.grad_sync.arm()
            tokens, labels = self.next_batch()
            with self.tracer.span("forward"):
                loss = self._forward(tokens, labels)
            with self.tracer.span("backward"):
                (loss / self.num_microbatches).backward()
            losses.append(loss.detach())
        self._sync_and_step()
        self.tracer.next_step()
        return float(torch.stack(losses).mean())

    # --- GPipe step (pp > 1) ----------------------------------------------
    def _recv_activation(self, shape, src) -> torch.Tensor:
        buf = torch.empty(shape, dtype=self.dtype, device=self.ctx.device or "cpu")
        dist.recv(buf, src=src)
        return buf

    def _stage_aux(self) -> Optional[torch.Tensor]:
        """Routing aux loss of the microbatch just forwarded (MoE stages
        without the head; stages with the head fold it into the loss)."""
        consume = getattr(self.model, "consume_aux_loss", None)
        return consume() if consume is not None else None

    def _backward_stage(self, out, gout, aux) -> None:
        """Non-last-stage backward incl. this stage's aux loss in the same
        engine pass (the logical loss is mean_mb(ce + aux))."""
        if aux is not None:
            torch.autograd.backward(
                [out, aux],
                [gout, torch.full_like(aux, 1.0 / self.num_microbatches)])
        else:
            out.backward(gout)

    def _step_pipeline(self) -> float:
        ctx = self.ctx
        h = self.spec.hidden_size
        seq = self.spec.seq_length // (ctx.tp if self.sp else 1)
        act_shape = (self.mbs, seq, h)
        prev = ctx.stage_neighbor(-1) if not ctx.is_first_stage else None
        nxt = ctx.stage_neighbor(+1) if not ctx.is_last_stage else None
        dev = self.ctx.device or "cpu"

        self.optimizer.zero_grad()
        inputs: List[Optional[torch.Tensor]] = []
        outputs: List[torch.Tensor] = []
        losses: List[torch.Tensor] = []
        aux_terms: List[Optional[torch.Tensor]] = []

        # Pre-posted, order-safe p2p overlap (both gloo AND RCCL):
        # p2p ops between one pair of ranks complete in issue order, so
        # overlap is safe exactly when both ends issue their ops for a
        # pair in the same sequence. Fill recvs (from prev) are posted
        # up-front — prev sends them in the same 0..nm-1 order. Drain
        # recvs (from nxt) are posted only AFTER the last fill send to
        # nxt, in reverse order — matching nxt's grad-send order. (The
        # round-1 version pre-posted drain recvs at loop top, which is
        # only safe on gloo's host-driven transport; on RCCL they would
        # sit ahead of the fill sends in the pair's issue order.)
        # GPipe keeps all nm activations live anyway, so the buffers are
        # free; sends are isend with the tensor kept alive until waited.
        nm = self.num_microbatches
        pending_sends: List[Tuple] = []
        fwd_bufs, fwd_reqs = [], []
        if not ctx.is_first_stage:
            for _ in range(nm):
                buf = torch.empty(act_shape, dtype=self.dtype, device=dev)
                fwd_bufs.append(buf)
                fwd_reqs.append(dist.irecv(buf, src=prev))

        # forward fill
        for mb in range(nm):
            if ctx.is_first_stage:
                tokens, labels = self.next_batch()
                x = tokens
                inputs.append(None)
            else:
                fwd_reqs[mb].wait()
                x = fwd_bufs[mb].requires_grad_(True)
                inputs.append(x)
            if ctx.is_last_stage:
                if ctx.is_first_stage:
                    pass  # pp == 1 handled elsewhere
                else:
                    # labels generated on the last stage (same per-replica
                    # seeded stream as the first stage's tokens)
                    _, labels = self.next_batch()
                out = self.model(x, labels=labels)
                losses.append(out)
                outputs.append(out)
            else:
                out = self.model(x)
                outputs.append(out)
                aux_terms.append(self._stage_aux())
                t = out.detach().contiguous()
                pending_sends.append((dist.isend(t, dst=nxt), t))

        # drain-phase grad recvs: issue order to nxt is now
        # send(0..nm-1) then recv(nm-1..0) on both ends
        bwd_bufs, bwd_reqs = {}, {}
        if not ctx.is_last_stage:
            for i in reversed(range(nm)):
                buf = torch.empty(act_shape, dtype=self.dtype, device=dev)
                bwd_bufs[i] = buf
                bwd_reqs[i] = dist.irecv(buf, src=nxt)

        # backward drain (reverse order)
        for i in reversed(range(self.num_microbatches)):
            if self.grad_sync is not None and i == 0:
                self.grad_sync.arm()
            if ctx.is_last_stage:
                (outputs[i] / self.num_microbatches).backward()
            else:
                bwd_reqs[i].wait()
                self._backward_stage(outputs[i], bwd_bufs[i], aux_terms[i])
            if not ctx.is_first_stage:
                t = inputs[i].grad.contiguous()
                pending_sends.append((dist.isend(t, dst=prev), t))

        for work, _t in pending_sends:
            work.wait()
        self._sync_and_step()
        if losses:
            return float(torch.stack([l.detach() for l in losses]).mean())
        return 0.0

    def _step_pipeline_1f1b(self) -> float:
        """1F1B (one-forward-one-backward): same bubble as GPipe
        ((B-1)*max + sum) but at most pp - pp_rank activations are live per
        stage instead of all B — the memory-bound pipeline schedule the
        reference's cost model prices but cannot run. Sends are isend (a
        steady-state fwd-send/bwd-send pair of neighboring stages would
        deadlock with rendezvous sends); recvs stay blocking — and must:
        p2p ops of one rank pair share a RCCL stream in issue order, and
        1F1B interleaves the two directions, so a prefetched recv issued
        ahead of this rank's next grad send can form a cycle with the
        peer doing the same (recv-act(k+1) blocks send-grad(j) here,
        while the peer's recv-grad(j) blocks its send-act(k+1)). GPipe's
        phase structure has no such interleave, so only GPipe pre-posts
        (see _step_pipeline)."""
        ctx = self.ctx
        nm = self.num_microbatches
        seq = self.spec.seq_length // (ctx.tp if self.sp else 1)
        act_shape = (self.mbs, seq, self.spec.hidden_size)
        prev = ctx.stage_neighbor(-1) if not ctx.is_first_stage else None
        nxt = ctx.stage_neighbor(+1) if not ctx.is_last_stage else None
        self.optimizer.zero_grad()

        pending = []   # (work, tensor): keep the buffer alive until wait
        live = {}      # microbatch -> (stage input, stage output)
        losses: List[torch.Tensor] = []

        def fwd(i: int) -> None:
            inp = None
            if ctx.is_first_stage:
                x, _ = self.next_batch()
            else:
                x = self._recv_activation(act_shape, prev).requires_grad_(True)
                inp = x
            if ctx.is_last_stage:
                _, labels = self.next_batch()
                out = self.model(x, labels=labels)
                losses.append(out.detach())
            else:
                out = self.model(x)
                t = out.detach().contiguous()
                pending.append((dist.isend(t, dst=nxt), t))
            live[i] = (inp, out, self._stage_aux())

        def bwd(i: int) -> None:
            if self.grad_sync is not None and i == nm - 1:
                self.grad_sync.arm()
            inp, out, aux = live.pop(i)
            if ctx.is_last_stage:
                (out / nm).backward()
            else:
                gout = self._recv_activation(act_shape, nxt)
                self._backward_stage(out, gout, aux)
            if not ctx.is_first_stage:
                t = inp.grad.contiguous()
                pending.append((dist.isend(t, dst=prev), t))

        warm = min(nm, ctx.pp - 1 - ctx.pp_rank)
        for i in range(warm):
            fwd(i)
        for i in range(warm, nm):           # steady state
            fwd(i)
            bwd(i - warm)
        for i in range(nm - warm, nm):      # drain
            bwd(i)
        for work, _ in pending:
            work.wait()

        self._sync_and_step()
        if losses:
            return float(torch.stack(losses).mean())
        return 0.0

    def _rank_of_stage(self, p: int) -> int:
        ctx = self.ctx
        return (p * ctx.dp + ctx.dp_rank) * ctx.tp + ctx.tp_rank

    def _step_interleaved(self) -> float:
        """Interleaved 1F1B (virtual pipeline stages): rank r hosts vpp
        chunks, chunk c = virtual stage c*pp + r, so the pipeline bubble
        shrinks ~1/vpp vs plain 1F1B. Every rank enumerates the same
        global forward/backward slot orders (Megatron-style chunk cycling
        in pp-sized microbatch groups), which makes the send/recv sequence
        on each directed channel identical on both ends by construction;
        sends are isend, recvs block."""
        ctx = self.ctx
        pp, v, nm = ctx.pp, self.vpp, self.num_microbatches
        r = ctx.pp_rank
        total = nm * v
        last_vs = pp * v - 1
        seq = self.spec.seq_length // (ctx.tp if self.sp else 1)
        act_shape = (self.mbs, seq, self.spec.hidden_size)
        prev = self._rank_of_stage((r - 1) % pp)
        nxt = self._rank_of_stage((r + 1) % pp)

        def f_cm(k):
            return (k // pp) % v, (k // (pp * v)) * pp + k % pp

        def b_cm(k):
            return v - 1 - (k // pp) % v, (k // (pp * v)) * pp + k % pp

        # each chunk's grads finish at ITS last backward slot, not the
        # globally-last one: arm the dp grad-sync per chunk (grad_sync
        # buckets spanning chunks launch once both sides fired)
        last_bwd_slot = {}
        for k in range(total):
            last_bwd_slot[b_cm(k)[0]] = k
        if self.grad_sync is not None and not hasattr(self, "_chunk_param_idxs"):
            idxs = {}
            pos = 0
            for c, chunk in enumerate(self.model_chunks):
                n = sum(1 for _ in chunk.parameters())
                idxs[c] = list(range(pos, pos + n))
                pos += n
            self._chunk_param_idxs = idxs

        self.optimizer.zero_grad()
        pending = []
        live = {}
        losses: List[torch.Tensor] = []

        def fwd(k: int) -> None:
            c, mb = f_cm(k)
            chunk = self.model_chunks[c]
            vs = c * pp + r
            inp = None
            if vs == 0:
                x, _ = self.next_batch()
            else:
                buf = torch.empty(act_shape, dtype=self.dtype,
                                  device=ctx.device or "cpu")
                dist.recv(buf, src=prev)
                x = buf.requires_grad_(True)
                inp = x
            if vs == last_vs:
                _, labels = self.next_batch()
                out = chunk(x, labels=labels)
                losses.append(out.detach())
            else:
                out = chunk(x)
                t = out.detach().contiguous()
                pending.append((dist.isend(t, dst=nxt), t))
            aux = None
            consume = getattr(chunk, "consume_aux_loss", None)
            if consume is not None:
                aux = consume()
                if vs == last_vs:
                    aux = None  # folded into the loss by the chunk itself
            live[(c, mb)] = (inp, out, aux)

        def bwd(k: int) -> None:
            c, mb = b_cm(k)
            vs = c * pp + r
            if self.grad_sync is not None and k == last_bwd_slot[c]:
                self.grad_sync.arm_params(self._chunk_param_idxs[c])
            inp, out, aux = live.pop((c, mb))
            if vs == last_vs:
                (out / nm).backward()
            else:
                gout = torch.empty(act_shape, dtype=self.dtype,
                                   device=ctx.device or "cpu")
                dist.recv(gout, src=nxt)
                self._backward_stage(out, gout, aux)
            if vs > 0:
                t = inp.grad.contiguous()
                pending.append((dist.isend(t, dst=prev), t))

        warmup = min((pp - r - 1) * 2 + (v - 1) * pp, total)
        for k in range(warmup):
            fwd(k)
        fk, bk = warmup, 0
        while fk < total:
            fwd(fk)
            bwd(bk)
            fk += 1
            bk += 1
        while bk < total:
            bwd(bk)
            bk += 1
        for work, _ in pending:
            work.wait()

        self._sync_and_step()
        if losses:
            return float(torch.stack(losses).mean())
        return 0.0

    # --- gradient sync + optimizer ----------------------------------------
    def _global_grad_norm(self) -> float:
        """Global L2 norm of the (DP-synced) flat gradient (runtime.clip)."""
        flat = self.optimizer.grad_flat
        pairs = [(flat[off:off + n], sh) for sh, (off, n) in
                 zip(self._grad_sharded, self.optimizer._slices)]
        return global_grad_norm(pairs, self.ctx.tp_group, self.ctx.pp_group)

    def _sync_and_step(self) -> None:
        if getattr(self, "_step_t0", None) is not None:
            # straggler detector: compute-segment end (collectives that
            # equalize ranks come after this point)
            if (self.ctx.device or torch.device("cpu")).type == "cuda":
                torch.cuda.synchronize()
            self._compute_s = time.perf_counter() - self._step_t0
        with self.tracer.span("grad_sync"):
            if self.grad_sync is not None:
                # hooks copied + all-reduced the final-microbatch grads,
                # overlapped with backward; wait and average
                self.grad_sync.finish()
            else:
                sync_partial_grads(self.optimizer.params)
                self.optimizer.gather_grads()
        scale = 1.0
        if self.clip_grad is not None:
            norm = self._global_grad_norm()
            if norm > self.clip_grad:
                scale = self.clip_grad / (norm + 1e-6)
        with self.tracer.span("optimizer"):
            self.optimizer.step(pre_gathered=True, grad_scale=scale)
        if self._check_sync:
            self.verify_replicas_synced()

    def verify_replicas_synced(self) -> None:
        """Race detector for the DP path (METIS_CHECK_SYNC=1): bitwise
        checksum of the fp32 master weights compared across the DP group
        after the optimizer step; raises on divergence. The reference has
        no runtime at all, so no equivalent exists there."""
        if self.ctx.dp <= 1 or self.ctx.dp_group is None:
            return
        int_view = {torch.bfloat16: torch.int16, torch.float16: torch.int16,
                    torch.float32: torch.int32, torch.float64: torch.int64}
        digest = sum(
            p.detach().reshape(-1).view(int_view[p.dtype]).long().sum()
            for p in self.model.parameters()
        )[None]
        gathered = [torch.zeros_like(digest) for _ in range(self.ctx.dp)]
        dist.all_gather(gathered, digest, group=self.ctx.dp_group)
        if any(int(g[0]) != int(gathered[0][0]) for g in gathered):
            raise RuntimeError(
                "DP replicas diverged after optimizer step: parameter "
                f"checksums {[int(g[0]) for g in gathered]} "
                f"(dp_rank {self.ctx.dp_rank})"
            )

    def train_step(self) -> float:
        self._step_t0 = time.perf_counter() if self._straggler_warn else None
        if self.ctx.pp == 1:
            loss = self._step_no_pipeline()
        elif self.schedule == "interleaved":
            loss = self._step_interleaved()
        elif self.schedule == "1f1b":
            loss = self._step_pipeline_1f1b()
        else:
            loss = self._step_pipeline()
        if self._step_t0 is not None:
            self._check_stragglers()
        return loss

    def _check_stragglers(self) -> None:
        """Straggler detector (METIS_STRAGGLER_WARN=<ratio>): all-gather
        per-rank COMPUTE segment times (step start to _sync_and_step —
        whole-step wall times equalize at the blocking collectives, the
        pre-sync segment is where a throttled or failing GPU shows up);
        rank 0 warns when max/min exceeds the ratio. Synchronizing, so
        it is opt-in diagnostics, not an always-on cost."""
        if not dist.is_initialized() or self.ctx.world_size < 2:
            return
        dev = self.ctx.device or torch.device("cpu")
        t = torch.tensor([self._compute_s], dtype=torch.float64, device=dev)
        gathered = [torch.zeros_like(t) for _ in range(self.ctx.world_size)]
        dist.all_gather(gathered, t)
        times = [float(g) for g in gathered]
        lo = max(min(times), 1e-9)
        if self.ctx.rank == 0 and max(times) / lo > self._straggler_warn:
            worst = times.index(max(times))
            print(f"WARNING: rank {worst} compute time "
                  f"{max(times)*1e3:.1f} ms vs fastest {lo*1e3:.1f} ms "
                  f"({max(times)/lo:.2f}x > {self._straggler_warn:.2f}x "
                  "threshold)", flush=True)

    # --- checkpoint ---------------------------------------------------------
    def save_checkpoint(self, path: str) -> None:
        """Per-rank checkpoint (model shard + optimizer state + plan)."""
        torch.save({
            "model": self.model.state_dict(),
            "optimizer": self.optimizer.state_dict(),
            "plan": {"dp": self.ctx.dp, "tp": self.ctx.tp, "pp": self.ctx.pp,
                     "mbs": self.mbs, "gbs": self.gbs, "vpp": self.vpp,
                     "layer_partition": self.layer_partition},
            "rank": self.ctx.rank,
            # synthetic-data generator state: resume replays the exact
            # token stream a continuous run would have drawn
            "data_gen": (self._data_gen.get_state()
                         if self._data_gen is not None else None),
            "data_cursor": (self.data_loader.state()
                            if self.data_loader is not None else None),
        }, path)

    def load_checkpoint(self, path: str) -> None:
        state = torch.load(path, map_location="cpu", weights_only=True)
        plan = state["plan"]
        assert (plan["dp"], plan["tp"], plan["pp"]) == (
            self.ctx.dp, self.ctx.tp, self.ctx.pp
        ), "checkpoint plan does not match the running plan"
        self.model.load_state_dict(state["model"])
        if state.get("data_gen") is not None:
            dev = self.ctx.device or torch.device("cpu")
            self._data_gen = torch.Generator(device=dev)
            self._data_gen.set_state(state["data_gen"])
        if state.get("data_cursor") is not None and self.data_loader is not None:
            self.data_loader.load_state(int(state["data_cursor"]))
        opt_state = state["optimizer"]
        self.optimizer.load_state_dict({
            "step": opt_state["step"],
            "master": opt_state["master"].to(self.optimizer.master.device),
            "m": opt_state["m"].to(self.optimizer.m.device),
            "v": opt_state["v"].to(self.optimizer.v.device),
        })

    # --- timing -----------------------------------------------------------
    def timed_steps(self, steps: int, warmup: int) -> float:
        """Run warmup + timed steps; returns mean ms/step on this rank."""
        for _ in range(warmup):
            self.train_step()
        if dist.is_initialized():
            dist.barrier()
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(steps):
            self.train_step()
        if dist.is_initialized():
            dist.barrier()
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        self.tracer.export()
        return (time.perf_counter() - t0) * 1000.0 / steps
