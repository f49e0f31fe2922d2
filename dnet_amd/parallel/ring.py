"""Pipelined-ring executor: one rank per GPU, RCCL send/recv activation hops.

The MI355X-native redesign of the reference's decode driver + shard compute
loop (reference: src/dnet/api/inference.py generate_stream +
src/dnet/shard/policies/fit_in_memory.py process): instead of gRPC frames
between asyncio workers, each rank runs a synchronous schedule over
microbatches — recv hidden, run its layer window, send to the next rank —
which fills the pipeline exactly like the reference's in-flight nonce
pipelining. The last rank samples (one token id per hop back, like the
reference's SendToken) and rank 0 embeds.

Decode steps are captured into hipGraphs per microbatch (static buffers;
sequence position lives in a device tensor) so the per-token critical path
is graph replays + p2p hops, not ~500 kernel launches.
"""
from __future__ import annotations

import logging
from dataclasses import dataclass, field
from typing import Optional

import torch
import torch.distributed as dist

from ..core.sampler import DecodingConfig, Sampler
from ..models import KVCache, ModelConfig, get_ring_model
from .comm import Ring

log = logging.getLogger("dnet")


def split_layers(num_layers: int, world: int) -> list[list[int]]:
    """Contiguous equal split (the solver produces smarter ones)."""
    base = num_layers // world
    rem = num_layers % world
    out = []
    start = 0
    for r in range(world):
        n = base + (1 if r < rem else 0)
        out.append(list(range(start, start + n)))
        start += n
    return out


@dataclass
class RingPlan:
    """Per-rank layer windows; rounds[r] = list of windows (k>1 = multiple
    ring laps per token, reference: api/utils.py compute_layer_assignments)."""
    assignments: list  # [world][round][layer_ids]

    @classmethod
    def contiguous(cls, num_layers: int, world: int) -> "RingPlan":
        return cls([[w] for w in split_layers(num_layers, world)])

    @property
    def rounds(self) -> int:
        return max(len(a) for a in self.assignments)


class RingExecutor:
    def __init__(self, cfg: ModelConfig, rank: int, world: int,
                 device: torch.device, plan: Optional[RingPlan] = None,
                 mb_count: int = 1, mb_size: int = 1, smax: int = 4096,
                 seed: int = 0, decoding: Optional[DecodingConfig] = None,
                 use_graphs: Optional[bool] = None, init_weights: bool = True,
                 residency: int = 0, compress_ratio: float = 0.0,
                 tp: int = 1, cp: int = 1, kv_bits: int = 16):
        self.cfg = cfg
        self.rank = rank
        self.world = world
        self.device = torch.device(device)
        # tensor OR context parallelism inside each pipeline stage: ranks
        # are laid out stage-major (rank = stage*grp + grp_rank); ring hops
        # are pairwise rank -> rank+grp; collectives run in the per-stage
        # group (all-reduce for TP, partials all-gather for CP).
        assert tp == 1 or cp == 1, "tp and cp are mutually exclusive (v1)"
        grp = tp * cp
        assert world % grp == 0, "world must be a multiple of tp*cp"
        self.tp = grp               # stage group width (topology)
        self.cp = cp
        self.stages = world // grp
        self.stage = rank // grp
        self.tp_rank = rank % grp
        self.plan = plan or RingPlan.contiguous(cfg.num_layers, self.stages)
        self.rounds = self.plan.rounds
        # per-round layer windows for this stage (padded to k with empty
        # pass-through windows); one token step = k laps of the ring
        # (reference: api/utils.py compute_layer_assignments k-round
        # interleaving per prima.cpp)
        self.windows = list(self.plan.assignments[self.stage])
        while len(self.windows) < self.rounds:
            self.windows.append([])
        self.my_layers = [l for w in self.windows for l in w]
        self.mb_count = mb_count
        self.mb_size = mb_size
        self.smax = smax
        self.is_first = self.stage == 0
        self.is_last = self.stage == self.stages - 1
        self.ring = None
        stage_group = None
        if world > 1:
            self.ring = Ring(rank, world, self.device)
            self.ring.next = (rank + grp) % world
            self.ring.prev = (rank - grp) % world
        if grp > 1:
            for st in range(self.stages):  # collective: create every group
                g = dist.new_group(list(range(st * grp, (st + 1) * grp)))
                if st == self.stage:
                    stage_group = g
        # token return path: last-stage rank pairs with its first-stage peer
        self.token_src = (self.stages - 1) * grp + self.tp_rank
        self.token_dst = self.tp_rank
        cls = get_ring_model(cfg.model_type)
        if cp > 1:
            use_graphs = False  # the partials all-gather is not captured
            self.model = cls(cfg, self.my_layers, self.device, self.is_first,
                             self.is_last, smax=smax,
                             cp_rank=self.tp_rank, cp_size=cp,
                             cp_group=stage_group)
        else:
            self.model = cls(cfg, self.my_layers, self.device, self.is_first,
                             self.is_last, smax=smax, tp_rank=self.tp_rank,
                             tp_size=tp, tp_group=stage_group)
        self.model.kv_bits = kv_bits
        if init_weights:
            self.model.init_random(seed)
        self.weight_cache = None
        if init_weights and 0 < residency < len(self.my_layers):
            from ..shard.policies import enable_offload  # lazy: avoids cycle
            self.weight_cache = enable_offload(self.model, residency)
            use_graphs = False  # slot addresses change per step
        self.kvs = [self.model.make_kv_cache(mb_size, smax)
                    for _ in range(mb_count)]
        self.sampler = Sampler(decoding or DecodingConfig())
        H = cfg.hidden_size
        # static buffers (recv targets / graph inputs)
        self.hbuf = [torch.zeros(mb_size, H, dtype=torch.bfloat16,
                                 device=self.device) for _ in range(mb_count)]
        self.tokbuf = [torch.zeros(mb_size, dtype=torch.int64,
                                   device=self.device) for _ in range(mb_count)]
        self.logits_buf = None
        if self.is_last:
            self.logits_buf = [torch.zeros(mb_size, cfg.vocab_size,
                                           dtype=torch.bfloat16, device=self.device)
                               for _ in range(mb_count)]
        if use_graphs is None:
            use_graphs = self.device.type == "cuda"
        self.use_graphs = use_graphs
        self._graphs: list = []
        self.last_logprob = None   # last sample's logprobs (last rank only)
        self.last_tops = None
        # optional column-sparsified activation hops (fixed keep-count so
        # the wire stays RCCL-shaped); reference: DNET_TRANSPORT_COMPRESS
        self.compress_ratio = compress_ratio if 0.0 < compress_ratio < 1.0 else 0.0
        if self.compress_ratio:
            from ..compression import keep_count
            k = keep_count(cfg.hidden_size, self.compress_ratio)
            self._cidx = [torch.zeros(k, dtype=torch.int32, device=self.device)
                          for _ in range(mb_count)]
            self._cpacked = [torch.zeros(mb_size, k, dtype=torch.bfloat16,
                                         device=self.device)
                             for _ in range(mb_count)]

    class _RecvPipe:
        """Depth-2 pre-posted receive pipeline for a fixed-count sequence
        of identical ring hops (the decode loop used blocking
        recv per hop; pre-posting lets the upstream rank's send proceed
        while this rank still computes the previous microbatch). `total`
        must be EXACT — every posted receive is matched before the loop
        exits, so no dangling requests survive (gloo would hang on
        shutdown otherwise)."""

        def __init__(self, ring, like: torch.Tensor, total: int,
                     depth: int = 2):
            self.ring = ring
            self.remaining = total
            self.q: list = []
            self.bufs = [torch.empty_like(like) for _ in range(min(depth,
                                                                  total))]
            self._free = list(self.bufs)
            self._post()

        def _post(self):
            while self._free and len(self.q) < self.remaining:
                b = self._free.pop()
                self.q.append((self.ring.irecv(b), b))

        def next_into(self, dst: torch.Tensor):
            req, b = self.q.pop(0)
            req.wait()
            dst.copy_(b)
            self.remaining -= 1
            self._free.append(b)
            self._post()

    def _send_hidden(self, mb: int):
        if not self.compress_ratio:
            self.ring.send(self.hbuf[mb])
            return
        from ..compression import column_sparsify
        idx, packed = column_sparsify(self.hbuf[mb], self.compress_ratio)
        self.ring.send(idx)
        self.ring.send(packed)

    def _recv_hidden(self, mb: int):
        if not self.compress_ratio:
            self.ring.recv(self.hbuf[mb])
            return
        from ..compression import column_unsparsify
        self.ring.recv(self._cidx[mb])
        self.ring.recv(self._cpacked[mb])
        self.hbuf[mb].copy_(
            column_unsparsify(self._cpacked[mb], self._cidx[mb],
                              self.cfg.hidden_size))

    # ------------- one-rank step bodies (graph-capturable) -------------

    def _decode_body(self, mb: int, r: int = 0):
        if self.is_first and r == 0:
            self.hbuf[mb].copy_(
                torch.nn.functional.embedding(self.tokbuf[mb], self.model.embed))
        if self.windows[r]:
            self.model.decode_window(self.hbuf[mb], self.windows[r],
                                     self.kvs[mb])
        if self.is_last and r == self.rounds - 1:
            self.logits_buf[mb].copy_(
                self.model.normalize_project(self.hbuf[mb]))

    def _capture_graphs(self):
        pool = torch.cuda.graph_pool_handle()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for mb in range(self.mb_count):
                for r in range(self.rounds):
                    for _ in range(2):  # warmup
                        self._decode_body(mb, r)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        # warmup advanced nothing persistent except KV garbage at pos; KV pos
        # unchanged (we don't advance pos in the body), cache rows at pos get
        # rewritten by real steps.
        # TWO alternating execs per step: relaunching the SAME graphExec
        # while its previous launch is still running blocks the host in
        # hipGraphLaunch — the slot scheduler's pipelined tick (launch
        # n+1, then emit n) measured the whole step time inside the
        # "launch" otherwise. Same static buffers; stream order keeps
        # them correct.
        self._graphs = {}
        self._graph_flip = {}
        for mb in range(self.mb_count):
            for r in range(self.rounds):
                pair = []
                for _ in range(2):
                    g = torch.cuda.CUDAGraph()
                    with torch.cuda.graph(g, pool=pool):
                        self._decode_body(mb, r)
                    pair.append(g)
                self._graphs[(mb, r)] = pair
                self._graph_flip[(mb, r)] = 0
        log.info("captured %d decode graphs (x2 execs)", len(self._graphs))

    def _run_decode(self, mb: int, r: int = 0):
        if self.use_graphs and not self._graphs:
            try:
                self._capture_graphs()
            except Exception as e:  # pragma: no cover
                log.warning("hipGraph capture failed (%s); falling back to eager", e)
                self.use_graphs = False
                self._graphs = {}
        if self.use_graphs:
            f = self._graph_flip[(mb, r)]
            self._graphs[(mb, r)][f].replay()
            self._graph_flip[(mb, r)] = f ^ 1
        else:
            self._decode_body(mb, r)

    # ------------- collective schedules -------------

    def prefill(self, tokens: torch.Tensor,
                chunk: int = 0) -> torch.Tensor | None:
        """tokens: [mb_count, mb_size, T] int64 (significant on rank 0; other
        ranks use it for shape only). Fills KV, samples the first token per
        sequence. Returns first-token tensor [mb_count, mb_size] on rank 0
        (and on the last rank), else None.

        ``chunk`` > 0 prefills in position chunks of that size (row-wise
        identical math — same tokens — while bounding the activation
        buffer to [B, chunk, H] for long prompts)."""
        M, B, T = tokens.shape
        assert M == self.mb_count and B == self.mb_size
        H = self.cfg.hidden_size
        tok_reqs = []
        first_tokens = torch.zeros(M, B, dtype=torch.int64, device=self.device)
        last_r = self.rounds - 1
        step = chunk if 0 < chunk < T else T
        for mb in range(M):
            for p0 in range(0, T, step):
                p1 = min(p0 + step, T)
                h = None
                for r in range(self.rounds):
                    if self.is_first and r == 0:
                        h = self.model.embed_tokens(
                            tokens[mb, :, p0:p1].to(self.device)).clone()
                    else:
                        if h is None:
                            h = torch.empty(B, p1 - p0, H,
                                            dtype=torch.bfloat16,
                                            device=self.device)
                        self.ring.recv(h)
                    if self.windows[r]:
                        self.model.prefill_window(h, self.windows[r],
                                                  self.kvs[mb], p0)
                    if not (self.is_last and r == last_r):
                        if self.stages > 1:
                            self.ring.send(h)
            self.kvs[mb].pos.fill_(T)
            if not self.is_last:
                if self.is_first:
                    tok_reqs.append(self.ring.irecv(first_tokens[mb],
                                                    src=self.token_src))
            else:
                logits = self.model.normalize_project(h[:, -1].contiguous())
                tok, logprob, tops = self.sampler.sample(logits.float())
                self.last_logprob, self.last_tops = logprob, tops
                first_tokens[mb] = tok
                if not self.is_first:
                    self.ring.send(first_tokens[mb], dst=self.token_dst)
                    # (single-stage TP: is_first too, so no self-send)
        for r in tok_reqs:
            r.wait()
        if self.is_first:
            for mb in range(M):
                self.tokbuf[mb].copy_(first_tokens[mb])
            return first_tokens
        return first_tokens if self.is_last else None

    def decode_rounds(self, n: int, collect: bool = True) -> torch.Tensor | None:
        """Run n decode rounds (each sequence advances n tokens).

        Returns generated tokens [mb_count, mb_size, n] on rank 0, else None.
        rank 0's tokbuf must hold the current token (set by prefill)."""
        M = self.mb_count
        out = (torch.zeros(M, self.mb_size, n, dtype=torch.int64,
                           device=self.device) if (collect and self.is_first)
               else None)
        tok_req: dict[int, object] = {}
        last_r = self.rounds - 1
        # pre-posted hidden receives (exact total: the loop consumes
        # per_step hops per (s, mb)) + async sends drained just before
        # their buffer is rewritten — no host-blocking comm between the
        # compute launches (reference pipelined its
        # in-flight frames, src/dnet/shard/adapters/ring.py:226-299)
        # first stage excluded: its lap-wrap hops and the token frames
        # arrive on the SAME (last->first) channel, and p2p matching is
        # FIFO per pair — a pre-posted hidden recv would swallow a token
        # frame (size mismatch/hang). Middle/last stages only ever
        # receive hidden frames from prev, so pre-posting is safe there.
        pipe = None
        if self.stages > 1 and not self.is_first and not self.compress_ratio:
            if n > 0:
                pipe = self._RecvPipe(self.ring, self.hbuf[0],
                                      n * M * self.rounds)
        send_req: dict[int, list] = {}

        def recv_h(mb):
            if pipe is not None:
                pipe.next_into(self.hbuf[mb])
            else:
                self._recv_hidden(mb)

        def send_h(mb):
            if pipe is not None:   # same no-compress fast path
                send_req.setdefault(mb, []).append(
                    self.ring.isend(self.hbuf[mb]))
            else:
                self._send_hidden(mb)

        def drain_sends(mb):
            for rq in send_req.pop(mb, ()):
                rq.wait()

        for s in range(n):
            for mb in range(M):
                if self.is_first:
                    if mb in tok_req:           # token from previous round
                        tok_req.pop(mb).wait()
                    if out is not None and s > 0:
                        out[mb, :, s - 1] = self.tokbuf[mb]
                    for r in range(self.rounds):
                        if r > 0:
                            recv_h(mb)          # lap wrap from last stage
                        drain_sends(mb)         # hbuf[mb] rewritten below
                        self._run_decode(mb, r)
                        if self.stages > 1 and not (self.is_last and r == last_r):
                            send_h(mb)
                    if self.stages > 1:
                        tok_req[mb] = self.ring.irecv(self.tokbuf[mb],
                                                      src=self.token_src)
                if self.stages > 1 and not self.is_first:
                    for r in range(self.rounds):
                        recv_h(mb)
                        drain_sends(mb)
                        self._run_decode(mb, r)
                        if not (self.is_last and r == last_r):
                            send_h(mb)
                if self.is_last:
                    tok, _, _ = self.sampler.sample(self.logits_buf[mb].float())
                    if self.stages > 1:
                        self.ring.send(tok, dst=self.token_dst)
                    else:
                        self.tokbuf[mb].copy_(tok)
                        if out is not None and self.rank == 0:
                            out[mb, :, s] = tok
                self.kvs[mb].pos.add_(1)
        for mb in list(send_req):
            drain_sends(mb)
        # drain last round's tokens on rank 0
        if self.is_first and self.stages > 1:
            for mb in range(M):
                if mb in tok_req:
                    tok_req.pop(mb).wait()
                if out is not None:
                    out[mb, :, n - 1] = self.tokbuf[mb]
        return out

    def reset(self):
        for kv in self.kvs:
            kv.reset()

    # ------------- slot-batched serving (continuous batching) -------------

    def prefill_slot(self, si: int, tokens: torch.Tensor,
                     chunk: int = 2048):
        """Collective single-slot prefill through the ring: one sequence
        flows stage to stage (position-chunked to bound activations) while
        other slots keep decoding state. ``tokens`` [T] int64 (significant
        on the first stage). Returns the last-position logits [1, V] on
        the LAST stage, else None."""
        T = int(tokens.shape[-1])
        H = self.cfg.hidden_size
        kvslot = self.kvs[0].slot(si)
        kvslot.pos.fill_(0)
        step = chunk if 0 < chunk < T else T
        h = None
        for p0 in range(0, T, step):
            p1 = min(p0 + step, T)
            h = None
            for r in range(self.rounds):
                if self.is_first and r == 0:
                    h = self.model.embed_tokens(
                        tokens.view(1, T)[:, p0:p1].to(self.device)).clone()
                else:
                    if h is None:
                        h = torch.empty(1, p1 - p0, H, dtype=torch.bfloat16,
                                        device=self.device)
                    self.ring.recv(h)
                if self.windows[r]:
                    self.model.prefill_window(h, self.windows[r], kvslot, p0)
                if not (self.is_last and r == self.rounds - 1):
                    if self.stages > 1:
                        self.ring.send(h)
        kvslot.pos.fill_(T)
        if self.is_last:
            return self.model.normalize_project(h[:, -1].contiguous())
        return None

    t_replay = 0.0   # cumulative; slot-tick profiling

    def slot_step_compute(self, mb: int = 0) -> None:
        """One decode step over the whole slot batch, including the ring
        hops — no sampling (the caller samples on the last stage)."""
        import time as _t
        _t0 = _t.perf_counter()
        self._slot_step_compute_inner(mb)
        self.t_replay += _t.perf_counter() - _t0

    def _slot_step_compute_inner(self, mb: int = 0) -> None:
        last_r = self.rounds - 1
        if self.is_first:
            for r in range(self.rounds):
                if r > 0:
                    self._recv_hidden(mb)
                self._run_decode(mb, r)
                if self.stages > 1 and not (self.is_last and r == last_r):
                    self._send_hidden(mb)
        elif self.stages > 1:
            for r in range(self.rounds):
                self._recv_hidden(mb)
                self._run_decode(mb, r)
                if not (self.is_last and r == last_r):
                    self._send_hidden(mb)

    def set_decoding(self, cfg: DecodingConfig, seed=None):
        g = None
        if seed is not None:
            g = torch.Generator(device=self.device)
            g.manual_seed(int(seed))
        self.sampler = Sampler(cfg, generator=g)

    def decode_stream(self, max_tokens: int, stop_ids=(), on_token=None,
                      mb: int = 0, should_stop=None):
        """Serving decode: one microbatch, token broadcast from the last rank
        to ALL ranks each step (so every rank stops identically on EOS), and
        ``on_token(step, tokens_tensor)`` called per step on every rank.

        rank 0's tokbuf must hold the current token (set by prefill). The
        first generated token is the prefill's sample; this generates up to
        ``max_tokens - 1`` more.
        """
        import torch.distributed as dist
        stop = torch.tensor(sorted(stop_ids), dtype=torch.int64,
                            device=self.device) if stop_ids else None

        def is_stop(tok: torch.Tensor) -> bool:
            if stop is None or stop.numel() == 0:
                return False
            return bool(torch.isin(tok, stop).all())

        produced = 0
        for s in range(max_tokens - 1):
            # external early stop (API cancel). Single-process only: with
            # world > 1 the stop decision must stay collective (every rank
            # breaks on the same broadcast token), so callers pass None.
            if should_stop is not None and should_stop():
                break
            last_r = self.rounds - 1
            if self.is_first:
                for r in range(self.rounds):
                    if r > 0:
                        self._recv_hidden(mb)
                    self._run_decode(mb, r)
                    if self.stages > 1 and not (self.is_last and r == last_r):
                        self._send_hidden(mb)
            elif self.stages > 1:
                for r in range(self.rounds):
                    self._recv_hidden(mb)
                    self._run_decode(mb, r)
                    if not (self.is_last and r == last_r):
                        self._send_hidden(mb)
            if self.is_last:
                tok, logprob, tops = self.sampler.sample(
                    self.logits_buf[mb].float())
                self.last_logprob, self.last_tops = logprob, tops
                self.tokbuf[mb].copy_(tok)
            if self.world > 1:
                dist.broadcast(self.tokbuf[mb], src=(self.stages - 1) * self.tp)
            self.kvs[mb].pos.add_(1)
            produced += 1
            done = is_stop(self.tokbuf[mb])
            if on_token is not None:
                on_token(s, self.tokbuf[mb], done or s == max_tokens - 2)
            if done:
                break
        return produced
