"""Shard runtime: one process per GPU; driver thread owns all torch state.

MI355X redesign of the reference's ShardRuntime + RingAdapter + policies
(reference: src/dnet/shard/runtime.py, adapters/ring.py): instead of gRPC
activation frames between asyncio workers, the shard joins a
torch.distributed group (RCCL over xGMI on GPU, gloo on CPU rigs) at
load_model time; rank 0 pulls inference requests from the wire-protocol
data server and drives the ring with collective broadcasts, so every rank
runs the same schedule. The reference's three-lock concurrency model is
replaced by ONE driver thread that owns the model, the process group and
the GPU (SURVEY.md §7 hard-part (3)).
"""
from __future__ import annotations

import json
import os
import queue
import socket
import struct
import threading
import time
from pathlib import Path
from typing import Optional

import msgpack
import torch

from ..config import get_settings
from ..core.sampler import DecodingConfig
from ..core.types import ShardLoadModelRequest
from ..models import ModelConfig, PRESETS, QuantConfig
from ..parallel.profiler import DeviceProfile, profile_device
from ..parallel.ring import RingExecutor, RingPlan
from ..utils.logger import get_logger
from ..utils.model_meta import get_model_metadata, load_tensors

log = get_logger("shard")

CMD_NOOP, CMD_INFER, CMD_UNLOAD, CMD_SHUTDOWN = 0, 1, 2, 3
CMD_SLOT_ADMIT, CMD_SLOT_STEP, CMD_SLOT_CANCEL = 4, 5, 6


class SyncWireClient:
    """Blocking wire-protocol client for the driver thread (token path)."""

    def __init__(self, host: str, port: int):
        self.host, self.port = host, port
        self.sock: Optional[socket.socket] = None

    def send(self, frame: dict):
        if self.sock is None:
            self.sock = socket.create_connection((self.host, self.port),
                                                 timeout=30)
        body = msgpack.packb(frame, use_bin_type=True)
        self.sock.sendall(struct.pack(">I", len(body)) + body)

    def close(self):
        if self.sock is not None:
            try:
                self.sock.close()
            finally:
                self.sock = None


class ShardRuntime:
    def __init__(self, instance: str = "shard0"):
        self.instance = instance
        self.settings = get_settings()
        self.executor: Optional[RingExecutor] = None
        self.load_req: Optional[ShardLoadModelRequest] = None
        self.model_name: str = ""
        self.ctrl_q: "queue.Queue[tuple]" = queue.Queue()
        self.infer_q: "queue.Queue[dict]" = queue.Queue()
        self._callback: Optional[SyncWireClient] = None
        self.slots: Optional[list] = None   # continuous batching (world==1)
        self._cancelled: dict = {}          # nonce -> ts, cancelled by API
        self._stop = threading.Event()
        self.status = "idle"
        self.last_error = ""

    # ---------- control-plane entry points (called from HTTP threads) ----------

    def submit_load(self, req: ShardLoadModelRequest) -> None:
        done = threading.Event()
        box: dict = {}
        self.ctrl_q.put(("load", req, done, box))
        done.wait(timeout=1800)
        if box.get("error"):
            raise RuntimeError(box["error"])

    def submit_unload(self) -> None:
        done = threading.Event()
        box: dict = {}
        self.ctrl_q.put(("unload", None, done, box))
        done.wait(timeout=120)

    def submit_infer(self, frame: dict) -> None:
        if self.executor is None:
            raise RuntimeError("no model loaded")
        log.debug("infer queued nonce=%s prompt_len=%s", frame.get("nonce"),
                 frame.get("prompt_len"))
        self.infer_q.put(frame)

    def submit_cancel(self, nonce: str) -> None:
        """API-side early stop (user stop strings matched in the detok
        text): frees the request's slot in slots mode; stops the legacy
        single-rank decode loop at the next token. Entries expire after
        60 s so cancels for already-finished requests don't accumulate."""
        self._cancelled[nonce] = time.monotonic()

    def shutdown(self) -> None:
        self._stop.set()

    def profile(self, quick: bool = True) -> DeviceProfile:
        dev = "cuda:0" if torch.cuda.is_available() else "cpu"
        return profile_device(self.instance, dev, quick=quick)

    # ---------- model load / unload (driver thread) ----------

    def _resolve_config(self, req: ShardLoadModelRequest) -> ModelConfig:
        quant = None
        if req.quant.startswith("int"):
            bits = int(req.quant[3])
            group = int(req.quant.split("-g")[1]) if "-g" in req.quant else 128
            quant = QuantConfig(bits, group)
        p = Path(req.model_path).expanduser()
        if (p / "config.json").exists():
            return ModelConfig.from_hf(json.loads((p / "config.json").read_text()),
                                       quant=quant)
        name = req.model_name or req.model_path
        if name in PRESETS:
            return ModelConfig.from_hf(dict(PRESETS[name]), quant=quant)
        raise FileNotFoundError(
            f"model not found: {req.model_path!r} (no config.json, not a preset)")

    def _load(self, req: ShardLoadModelRequest) -> None:
        import torch.distributed as dist
        cfg = self._resolve_config(req)
        if torch.cuda.is_available():
            device = torch.device(f"cuda:{req.gpu_index}")
            torch.cuda.set_device(device)
        else:
            device = torch.device("cpu")
        if req.world_size > 1 and not dist.is_initialized():
            backend = "nccl" if device.type == "cuda" else "gloo"
            dist.init_process_group(
                backend=backend,
                init_method=f"tcp://{req.master_addr}:{req.master_port}",
                rank=req.rank, world_size=req.world_size)
        rounds = req.layer_rounds or [sorted(req.layers)]
        if req.world_size == 1:
            plan = RingPlan([rounds])
        else:
            # every rank gets the full plan via its own request's windows;
            # the executor only reads its own slot (pad others to the same
            # round count so plan.rounds is consistent)
            k = len(rounds)
            plan = RingPlan([[[] for _ in range(k)]
                             for _ in range(req.world_size)])
            plan.assignments[req.rank] = rounds
        synthetic = not (Path(req.model_path).expanduser() / "config.json").exists()
        residency = req.residency_size if \
            0 < req.residency_size < len(req.layers) else 0
        slots_mode = req.max_batch > 1 and req.world_size == 1
        ex = RingExecutor(cfg, req.rank, req.world_size, device, plan=plan,
                          mb_count=1, mb_size=req.max_batch,
                          smax=req.max_seq,
                          # hipGraph decode is ON by default, including
                          # under slot churn (admissions prefill eagerly
                          # into the same static KV storage the captured
                          # graph reads — verified token-exact vs eager
                          # on hardware, tests/test_ops_gpu.py slots
                          # churn test). DNET_SLOTS_GRAPHS=0 opts out.
                          use_graphs=(device.type == "cuda"
                                      and self.settings.compute.use_graphs
                                      and residency == 0
                                      and (not slots_mode or os.environ.get(
                                          "DNET_SLOTS_GRAPHS", "1") == "1")),
                          init_weights=synthetic, residency=residency,
                          kv_bits=req.kv_bits)
        if not synthetic:
            self._load_weights(ex, req)
            if residency:
                from .policies import enable_offload
                ex.weight_cache = enable_offload(ex.model, residency)
                ex.use_graphs = False
        self.executor = ex
        # slot-based continuous batching: several single-stream requests
        # share the decode batch (each owns one KV slot); enabled by
        # loading with max_batch > 1. Multi-rank rings coordinate slot
        # admits/steps via the command broadcasts (every rank keeps the
        # same slot state; the last stage samples and emits).
        self.slots = [None] * req.max_batch if req.max_batch > 1 else None
        if self.slots is not None:
            from ..core.sampler import RowSampler
            self._row_sampler = RowSampler(req.max_batch, device=ex.device)
            # device-resident park masks (updated only on slot-state
            # CHANGES: building a park index tensor per tick was a
            # synchronous H2D that serialized the host against the
            # in-flight decode graph, ~17 ms/tick measured).
            #   _park_dev: row's dummy append pinned in place (freed or
            #              mid-chunked-prefill slots)
            #   _free_dev: additionally FREE -> pinned at pos 0, so the
            #              dead row's attention scans 1 position, not
            #              smax (prefilling slots must stay at smax-1:
            #              their low rows hold real prefill KV)
            self._park_dev = torch.ones(req.max_batch, dtype=torch.bool,
                                        device=ex.device)
            self._free_dev = torch.ones(req.max_batch, dtype=torch.bool,
                                        device=ex.device)
            # per-slot generation counter: bumped on every admit so a
            # pending (pipelined) emit can detect that its slot was freed
            # and re-admitted and must not deliver the stale token under
            # the new request's nonce
            self._slot_gen = [0] * req.max_batch
        self.load_req = req
        self.model_name = req.model_name or req.model_path
        # one emitter per ring: the grp-rank-0 member of the last stage
        # (TP/CP peers compute identical logits; two callbacks would
        # double-emit every token)
        if ex.is_last and ex.tp_rank == 0 and req.api_callback_address:
            host, _, port = req.api_callback_address.rpartition(":")
            self._callback = SyncWireClient(host or "127.0.0.1", int(port))
        self.status = "loaded"
        self.link_profile = {}
        self.link_matrix = {}
        self.rank = req.rank
        if req.world_size > 1:
            try:
                from ..parallel.profiler import (measure_link_matrix,
                                                 measure_ring_links)
                self.link_profile = measure_ring_links(
                    req.rank, req.world_size, ex.device,
                    sizes=(65536,), reps=5)
                for size, r in self.link_profile.items():
                    log.info("[PROFILE][XGMI] payload=%d latency=%.3fms "
                             "bw=%.1fGB/s", size, r["latency_ms"], r["gbps"])
                # full per-pair fabric map (identical on every rank);
                # surfaces via /health so the NEXT prepare_topology orders
                # the ring by measured xGMI links instead of TCP RTT
                if req.world_size <= 16:
                    self.link_matrix = measure_link_matrix(
                        req.rank, req.world_size, ex.device)
                    if req.rank == 0:
                        for k, r in sorted(self.link_matrix.items()):
                            log.info("[PROFILE][XGMI-LINK] %s latency="
                                     "%.3fms bw=%.1fGB/s", k,
                                     r["latency_ms"], r["gbps"])
            except Exception:
                log.exception("xGMI link probe failed (non-fatal)")
        log.info("model loaded: %s rank %d/%d %d layers %s...", self.model_name,
                 req.rank, req.world_size, len(req.layers), req.layers[:4])

    def _load_weights(self, ex: RingExecutor, req: ShardLoadModelRequest) -> None:
        from ..utils import repack as rp
        model_id = req.model_name or req.model_path
        need_api = ex.is_first or ex.is_last
        if self.settings.storage.repack_on_load:
            sd = rp.load_repacked(model_id, ex.my_layers, include_api=need_api)
            if sd is not None:
                log.info("repack fastpath: loaded %d tensors from %s",
                         len(sd), rp.repack_dir_for(model_id, ex.my_layers))
                ex.model.load_state_dict(sd)
                return
        meta = get_model_metadata(req.model_path)
        names: list[str] = []
        for lid in ex.my_layers:
            names += meta.layers.get(lid, [])
        if ex.is_first:
            names += meta.embed
        if ex.is_last:
            names += meta.final_norm + meta.lm_head + meta.embed
        sd = load_tensors(meta, sorted(set(names)))
        if self.settings.storage.repack_on_load:
            try:
                # repack for free from the just-loaded tensors: the next
                # load of this assignment skips the full-safetensors parse
                rp.ensure_repacked_for_layers(req.model_path, model_id,
                                              ex.my_layers,
                                              include_api=need_api, sd=sd)
            except Exception:
                log.exception("repack write failed (non-fatal)")
        ex.model.load_state_dict(sd)

    def _unload(self) -> None:
        import torch.distributed as dist
        self.executor = None
        self.slots = None
        self._pending = None
        self.load_req = None
        self.model_name = ""
        if self._callback:
            self._callback.close()
            self._callback = None
        if dist.is_initialized():
            dist.destroy_process_group()
        if torch.cuda.is_available():
            torch.cuda.empty_cache()
        self.status = "idle"

    # ---------- the driver loop ----------

    def run(self) -> None:
        """Main driver loop — owns the model and the process group."""
        while not self._stop.is_set():
            try:
                # BUSY (active slots / queued work): poll the control
                # queue without blocking — the 20 ms timeout here ran
                # once per slot TICK and capped serving at ~26 tok/s
                # (each token paid the control poll)
                busy = (self.slots is not None
                        and (any(st is not None for st in self.slots)
                             or not self.infer_q.empty()
                             or self._pending is not None))
                if busy:
                    kind, arg, done, box = self.ctrl_q.get_nowait()
                else:
                    kind, arg, done, box = self.ctrl_q.get(timeout=0.02)
                try:
                    if kind == "load":
                        self._load(arg)
                    elif kind == "unload":
                        self._broadcast_cmd(CMD_UNLOAD)
                        self._unload()
                except Exception as e:  # surface to the HTTP caller
                    log.exception("control command failed")
                    box["error"] = str(e)
                    self.last_error = str(e)
                finally:
                    done.set()
                continue
            except queue.Empty:
                pass
            ex = self.executor
            if ex is None:
                continue
            if ex.rank == 0:
                if self.slots is not None:
                    self._slots_tick()
                    continue
                try:
                    frame = self.infer_q.get(timeout=0.02)
                except queue.Empty:
                    if ex.world > 1:
                        self._broadcast_cmd(CMD_NOOP)
                    continue
                try:
                    log.debug("running infer nonce=%s", frame.get("nonce"))
                    self._run_infer_rank0(frame)
                except Exception:
                    log.exception("inference failed")
                    self._send_error(frame.get("nonce", ""))
            else:
                cmd = self._recv_cmd()
                if cmd[0] == CMD_INFER:
                    self._run_infer_follower(cmd)
                elif cmd[0] == CMD_SLOT_ADMIT:
                    self._slot_admit_follower(cmd)
                elif cmd[0] == CMD_SLOT_STEP:
                    self._slot_step_exec()
                elif cmd[0] == CMD_SLOT_CANCEL:
                    self.slots[int(cmd[1])] = None
                    self._row_sampler.clear_row(int(cmd[1]))
                    self._park_dev[int(cmd[1])] = True
                    self._free_dev[int(cmd[1])] = True
                elif cmd[0] == CMD_UNLOAD:
                    self._unload()

    # ---------- collective command plumbing ----------

    _CMD_LEN = 16

    def _cmd_tensor(self, vals=()):
        t = torch.zeros(self._CMD_LEN, dtype=torch.float64,
                        device=self._comm_device())
        for i, v in enumerate(vals):
            t[i] = float(v)
        return t

    def _comm_device(self):
        ex = self.executor
        return ex.device if (ex and ex.device.type == "cuda") else torch.device("cpu")

    def _broadcast_cmd(self, *vals):
        import torch.distributed as dist
        if self.executor is None or self.executor.world == 1:
            return
        t = self._cmd_tensor(vals)
        dist.broadcast(t, src=0)

    def _recv_cmd(self):
        import torch.distributed as dist
        t = self._cmd_tensor()
        dist.broadcast(t, src=0)
        return [t[i].item() for i in range(self._CMD_LEN)]

    # ---------- inference ----------

    def _run_infer_rank0(self, frame: dict) -> None:
        import numpy as np
        ex = self.executor
        tokens = torch.from_numpy(
            np.frombuffer(frame["tokens"], dtype=np.int32).copy()).long()
        T = int(frame.get("prompt_len", tokens.numel()))
        tokens = tokens.view(1, 1, T)
        p = frame.get("params", {})
        max_tokens = int(frame.get("max_tokens", 128))
        stop_ids = list(frame.get("stop_ids", []))
        nonce = frame.get("nonce", "")
        nonce_ids = list(nonce.encode("utf-8"))[:64]
        if ex.world > 1:
            self._broadcast_cmd(CMD_INFER, T, max_tokens, len(stop_ids),
                                p.get("temperature", 0.0), p.get("top_p", 1.0),
                                p.get("top_k", 0), p.get("min_p", 0.0),
                                int(p.get("logprobs", False)),
                                int(p.get("top_logprobs", 0)), len(nonce_ids))
            import torch.distributed as dist
            payload = torch.cat([
                tokens.flatten().to(self._comm_device()),
                torch.tensor(stop_ids + nonce_ids, dtype=torch.int64,
                             device=self._comm_device())])
            dist.broadcast(payload, src=0)
        self._execute_infer(nonce, tokens, max_tokens, stop_ids, p)

    def _run_infer_follower(self, cmd) -> None:
        import torch.distributed as dist
        ex = self.executor
        T, max_tokens, n_stop = int(cmd[1]), int(cmd[2]), int(cmd[3])
        n_nonce = int(cmd[10])
        p = {"temperature": cmd[4], "top_p": cmd[5], "top_k": int(cmd[6]),
             "min_p": cmd[7], "logprobs": bool(cmd[8]),
             "top_logprobs": int(cmd[9])}
        payload = torch.zeros(T + n_stop + n_nonce, dtype=torch.int64,
                              device=self._comm_device())
        dist.broadcast(payload, src=0)
        tokens = payload[:T].view(1, 1, T).cpu()
        stop_ids = payload[T:T + n_stop].tolist()
        nonce = bytes(payload[T + n_stop:].tolist()).decode("utf-8", "replace")
        self._execute_infer(nonce, tokens, max_tokens, stop_ids, p)

    def _execute_infer(self, nonce: str, tokens: torch.Tensor,
                       max_tokens: int, stop_ids: list, p: dict) -> None:
        ex = self.executor
        ex.reset()
        # never decode past the KV capacity (rope_append at pos >= smax
        # would write out of range)
        T_prompt = tokens.shape[-1]
        max_tokens = max(1, min(max_tokens, ex.smax - T_prompt))
        ex.set_decoding(DecodingConfig(
            temperature=p.get("temperature", 0.0), top_p=p.get("top_p", 1.0),
            top_k=int(p.get("top_k", 0)), min_p=p.get("min_p", 0.0),
            logprobs=bool(p.get("logprobs", False)),
            top_logprobs=int(p.get("top_logprobs", 0))),
            seed=p.get("seed"))
        # pad the single sequence to the executor's batch width
        B = ex.mb_size
        toks = tokens.expand(1, B, tokens.shape[-1]).contiguous().to(ex.device)
        first = ex.prefill(toks, chunk=2048)   # bound activation memory
        # every rank must agree on EOS-after-first-token before entering the
        # collective decode loop; rank 0 holds the first token in tokbuf.
        tok0_t = ex.tokbuf[0][:1].clone()
        if ex.world > 1:
            import torch.distributed as dist
            dist.broadcast(tok0_t, src=0)
        tok0 = int(tok0_t[0])
        done0 = tok0 in stop_ids or max_tokens <= 1
        if ex.is_last:
            self._emit_token(nonce, tok0, finished=done0)
        if done0:
            return

        def on_token(step, tokt, last):
            if ex.is_last:
                self._emit_token(nonce, int(tokt[0]),
                                 finished=last or int(tokt[0]) in stop_ids)

        self._emit_s = 0.0
        t0 = time.perf_counter()
        stopper = ((lambda: nonce in self._cancelled)
                   if ex.world == 1 else None)
        n = ex.decode_stream(max_tokens, stop_ids=stop_ids, on_token=on_token,
                             should_stop=stopper)
        self._cancelled.pop(nonce, None)
        dt = time.perf_counter() - t0
        log.info("[PROFILE][DECODE] nonce=%s tokens=%d ms=%.1f tok_s=%.1f "
                 "emit_ms=%.1f", nonce[:18], n, dt * 1e3, n / max(dt, 1e-9),
                 self._emit_s * 1e3)

    # ---------- slot-based continuous batching ----------

    def _slots_tick(self) -> None:
        """One scheduler iteration: admit queued requests into free KV
        slots (prefill them while other slots hold their state), then run
        one decode step for every active slot. Reference has no equivalent
        (requests serialize there); this keeps N single-stream requests at
        ~single-stream latency each."""
        ex = self.executor
        if self._cancelled:
            for i, st in enumerate(self.slots):
                if st is not None and st.get("nonce") in self._cancelled:
                    self._cancelled.pop(st["nonce"], None)
                    if ex.world > 1:
                        self._broadcast_cmd(CMD_SLOT_CANCEL, i)
                    self.slots[i] = None
                    self._row_sampler.clear_row(i)
                    self._park_dev[i] = True
                    self._free_dev[i] = True
                    log.info("[PROFILE][SLOT] cancel slot=%d nonce=%s", i,
                             st.get("nonce", "")[:18])
            now = time.monotonic()
            for n, ts in list(self._cancelled.items()):
                if now - ts > 60.0:
                    del self._cancelled[n]
        active = any(s is not None for s in self.slots)
        block = not active
        progressed = False
        while any(s is None for s in self.slots):
            free = next(i for i, s in enumerate(self.slots) if s is None)
            try:
                frame = (self.infer_q.get(timeout=0.02) if block
                         else self.infer_q.get_nowait())
            except queue.Empty:
                break
            block = False
            progressed = True
            try:
                self._slot_admit(free, frame)
            except Exception:
                log.exception("slot admit failed")
                self._send_error(frame.get("nonce", ""))
        prefilling = [i for i, st in enumerate(self.slots)
                      if st is not None and st.get("state") == "prefill"]
        if prefilling:
            # interleaved admission: one prompt chunk per tick, so long
            # prompts trickle in BETWEEN decode steps instead of blocking
            # the in-flight streams (single-rank path)
            progressed = True
            self._slot_prefill_chunk(prefilling[0])
        if any(st is not None and st.get("state") != "prefill"
               for st in self.slots):
            progressed = True
            if ex.world > 1:
                # multi-rank: broadcast the step command; emit synchronously
                self._broadcast_cmd(CMD_SLOT_STEP)
                self._slot_step_exec()
            else:
                # single rank: pipeline — launch step n+1 (device-side
                # deps only) FIRST, then emit step n's tokens: the host
                # sync inside emit overlaps the just-launched GPU work.
                # Slot-reuse safety comes from the per-slot generation
                # guard in _slot_emit (admission may have re-populated a
                # slot the pending active list still names).
                t0 = time.perf_counter()
                launched = self._slot_step_launch()
                self._tick_launch_s += time.perf_counter() - t0
                if self._pending is not None:
                    t0e = time.perf_counter()
                    self._slot_emit(*self._pending)
                    self._tick_emit_s += time.perf_counter() - t0e
                    if self._pending_evs is not None:
                        # emit synced the pending step -> both events done
                        self._tick_gpu_ms += self._pending_evs[0].\
                            elapsed_time(self._pending_evs[1])
                self._pending = launched
                self._pending_evs = self._launch_evs
                self._tick_n += 1
                if self._tick_n >= 128:
                    log.info("[PROFILE][TICK] n=%d launch_ms=%.2f "
                             "emit_ms=%.2f replay_ms=%.2f gpu_ms=%.2f",
                             self._tick_n,
                             self._tick_launch_s / self._tick_n * 1e3,
                             self._tick_emit_s / self._tick_n * 1e3,
                             ex.t_replay / self._tick_n * 1e3,
                             self._tick_gpu_ms / self._tick_n)
                    log.info("[PROFILE][TICK2] sample_ms=%.2f aux_ms=%.2f",
                             self._t_sample / self._tick_n * 1e3,
                             self._t_aux / self._tick_n * 1e3)
                    self._tick_n = 0
                    self._tick_launch_s = self._tick_emit_s = 0.0
                    ex.t_replay = 0.0
                    self._t_sample = self._t_aux = 0.0
                    self._tick_gpu_ms = 0.0
        if (self._pending is not None
                and not any(st is not None and st.get("state") != "prefill"
                            for st in self.slots)):
            # nothing left to launch: flush the final pipelined step
            self._slot_emit(*self._pending)
            self._pending = None
        if not progressed and ex.world > 1:
            self._broadcast_cmd(CMD_NOOP)

    def _slot_admit(self, si: int, frame: dict) -> None:
        """Rank-0 admit: broadcast the slot command + payload, then run the
        collective admit on this rank too."""
        import numpy as np
        ex = self.executor
        tokens = torch.from_numpy(
            np.frombuffer(frame["tokens"], dtype=np.int32).copy()).long()
        T = int(frame.get("prompt_len", tokens.numel()))
        p = frame.get("params", {})
        cfg = DecodingConfig(
            temperature=p.get("temperature", 0.0), top_p=p.get("top_p", 1.0),
            top_k=int(p.get("top_k", 0)), min_p=p.get("min_p", 0.0),
            logprobs=bool(p.get("logprobs", False)),
            top_logprobs=int(p.get("top_logprobs", 0)))
        stop_ids = list(frame.get("stop_ids", []))
        max_tokens = int(frame.get("max_tokens", 128))
        nonce = frame.get("nonce", "")
        nonce_ids = list(nonce.encode("utf-8"))[:64]
        seed = p.get("seed")
        if ex.world > 1:
            import torch.distributed as dist
            self._broadcast_cmd(CMD_SLOT_ADMIT, si, T, max_tokens,
                                len(stop_ids), len(nonce_ids),
                                cfg.temperature, cfg.top_p, cfg.top_k,
                                cfg.min_p,
                                -1.0 if seed is None else int(seed),
                                1.0 if cfg.logprobs else 0.0,
                                cfg.top_logprobs)
            payload = torch.cat([
                tokens.flatten().to(self._comm_device()),
                torch.tensor(stop_ids + nonce_ids, dtype=torch.int64,
                             device=self._comm_device())])
            dist.broadcast(payload, src=0)
        self._slot_admit_exec(si, tokens.view(-1), max_tokens, stop_ids,
                              nonce, cfg, seed)

    def _slot_admit_follower(self, cmd) -> None:
        import torch.distributed as dist
        si, T, max_tokens = int(cmd[1]), int(cmd[2]), int(cmd[3])
        n_stop, n_nonce = int(cmd[4]), int(cmd[5])
        cfg = DecodingConfig(temperature=cmd[6], top_p=cmd[7],
                             top_k=int(cmd[8]), min_p=cmd[9],
                             logprobs=bool(cmd[11]),
                             top_logprobs=int(cmd[12]))
        seed = None if cmd[10] < 0 else int(cmd[10])
        payload = torch.zeros(T + n_stop + n_nonce, dtype=torch.int64,
                              device=self._comm_device())
        dist.broadcast(payload, src=0)
        tokens = payload[:T].cpu()
        stop_ids = payload[T:T + n_stop].tolist()
        nonce = bytes(payload[T + n_stop:].tolist()).decode("utf-8", "replace")
        self._slot_admit_exec(si, tokens, max_tokens, stop_ids, nonce, cfg,
                              seed)

    def _slot_admit_exec(self, si, tokens, max_tokens, stop_ids, nonce,
                         cfg, seed=None) -> None:
        """Collective slot admit (every rank): single-slot ring prefill,
        sample on the last stage, broadcast the first token, register the
        slot state identically everywhere."""
        ex = self.executor
        self._slot_gen[si] += 1
        max_tokens = max(1, min(max_tokens,
                                ex.smax - int(tokens.shape[-1])))
        gen = self._row_sampler.set_row(si, cfg, seed)
        chunk = int(os.environ.get("DNET_PREFILL_CHUNK", "2048"))
        if ex.world == 1 and int(tokens.shape[-1]) > chunk:
            # long prompt: admit in "prefill" state — the tick loop feeds
            # one chunk per iteration between decode steps (slot stays
            # PARKED until the prefill completes)
            self._park_dev[si] = True
            self._free_dev[si] = False
            # the row was free-parked at pos 0; move it to smax-1 NOW —
            # the next decode step appends at the CURRENT pos before the
            # masks are re-applied, and an append at 0 would overwrite
            # the first prefill chunk
            ex.kvs[0].pos[si] = ex.smax - 1
            self.slots[si] = {"state": "prefill", "tokens": tokens,
                              "p0": 0, "cfg": cfg, "nonce": nonce,
                              "produced": 0, "max_tokens": max_tokens,
                              "stop_ids": set(stop_ids)}
            log.info("[PROFILE][SLOT] admit slot=%d nonce=%s prompt=%d "
                     "(chunked)", si, nonce[:18], int(tokens.shape[-1]))
            return
        logits = ex.prefill_slot(si, tokens)
        t0_t = torch.zeros(1, dtype=torch.int64, device=self._comm_device())
        lp0 = tops0 = None
        if ex.is_last:
            from ..core.sampler import Sampler
            tok, lp, tops = Sampler(cfg, generator=gen).sample(logits.float())
            t0_t[0] = int(tok[0])
            if lp is not None:
                lp0 = float(lp[0])
                tops0 = tops[0] if tops else None
        if ex.world > 1:
            import torch.distributed as dist
            dist.broadcast(t0_t, src=(ex.stages - 1) * ex.tp)
        t0 = int(t0_t[0])
        ex.tokbuf[0][si] = t0
        st = {"nonce": nonce, "produced": 1, "max_tokens": max_tokens,
              "stop_ids": set(stop_ids)}
        done = t0 in st["stop_ids"] or max_tokens <= 1
        if ex.is_last:
            self._emit_token(nonce, t0, finished=done, logprob=lp0,
                             tops=tops0)
        if not done:
            self.slots[si] = st
            self._park_dev[si] = False
            self._free_dev[si] = False
        log.info("[PROFILE][SLOT] admit slot=%d nonce=%s prompt=%d", si,
                 nonce[:18], int(tokens.shape[-1]))

    def _slot_prefill_chunk(self, si: int) -> None:
        """Feed one position chunk of a prefilling slot's prompt; on the
        final chunk, sample the first token and flip the slot active."""
        ex = self.executor
        st = self.slots[si]
        chunk = int(os.environ.get("DNET_PREFILL_CHUNK", "2048"))
        toks = st["tokens"]
        T = int(toks.shape[-1])
        p0 = st["p0"]
        p1 = min(p0 + chunk, T)
        kvslot = ex.kvs[0].slot(si)
        h = ex.model.embed_tokens(
            toks.view(1, T)[:, p0:p1].to(ex.device)).clone()
        ex.model.prefill_window(h, ex.my_layers, kvslot, p0)
        st["p0"] = p1
        if p1 < T:
            return
        kvslot.pos.fill_(T)
        logits = ex.model.normalize_project(h[:, -1].contiguous())
        from ..core.sampler import Sampler
        tok, lp, tops = Sampler(st["cfg"],
                                generator=self._row_sampler.gens[si]
                                ).sample(logits.float())
        t0 = int(tok[0])
        lp0 = float(lp[0]) if lp is not None else None
        tops0 = tops[0] if tops else None
        ex.tokbuf[0][si] = t0
        st["produced"] = 1
        st["state"] = "active"
        self._park_dev[si] = False
        self._free_dev[si] = False
        st.pop("tokens")
        done = t0 in st["stop_ids"] or st["max_tokens"] <= 1
        if ex.is_last:
            self._emit_token(st["nonce"], t0, finished=done, logprob=lp0,
                             tops=tops0)
        if done:
            self.slots[si] = None
            self._row_sampler.clear_row(si)
            self._park_dev[si] = True
            self._free_dev[si] = True

    def _slot_step_exec(self) -> None:
        """Collective decode step for all slots (multi-rank path, emits
        synchronously): hops + compute, sample on the last stage, token
        broadcast, identical slot-state update on every rank."""
        ex = self.executor
        rs = self._row_sampler
        ex.slot_step_compute()
        if ex.is_last:
            toks_t = self._row_sampler.sample(ex.logits_buf[0].float())
            ex.tokbuf[0].copy_(toks_t)
        if ex.world > 1:
            import torch.distributed as dist
            dist.broadcast(ex.tokbuf[0], src=(ex.stages - 1) * ex.tp)
        ex.kvs[0].pos.add_(1)
        ex.kvs[0].pos.masked_fill_(self._park_dev, ex.smax - 1)
        ex.kvs[0].pos.masked_fill_(self._free_dev, 0)
        self._slot_emit(ex.tokbuf[0],
                        [(i, self._slot_gen[i])
                         for i, st in enumerate(self.slots)
                         if st is not None],
                        rs.last_logp, rs.last_topv, rs.last_topi)

    _pending = None   # (device tokens, [(slot, gen)]) of the in-flight step
    _pending_evs = None   # CUDA events bracketing the pending step
    _launch_evs = None
    _tick_n = 0
    _tick_launch_s = 0.0
    _tick_emit_s = 0.0
    _tick_gpu_ms = 0.0
    _slot_gen: list = []   # per-slot admit generation (see _slot_emit)

    _t_sample = 0.0
    _t_aux = 0.0

    def _slot_step_launch(self):
        """Enqueue one decode step for the whole batch — device ops only,
        no host sync. Returns (tokens, active slots, logprob tensors)."""
        ex = self.executor
        rs = self._row_sampler
        if ex.device.type == "cuda":
            self._launch_evs = (torch.cuda.Event(enable_timing=True),
                                torch.cuda.Event(enable_timing=True))
            self._launch_evs[0].record()
        ex.slot_step_compute()
        t0 = time.perf_counter()
        toks_t = self._row_sampler.sample(ex.logits_buf[0].float())
        self._t_sample += time.perf_counter() - t0
        t0 = time.perf_counter()
        ex.tokbuf[0].copy_(toks_t)
        ex.kvs[0].pos.add_(1)
        # park idle/prefilling slots at the last row: the dummy append
        # stays in range and that row is rewritten by a real token before
        # any active slot ever attends to it. Device-resident mask:
        # building a park index tensor here was a synchronous H2D that
        # serialized the host against the in-flight decode graph
        # (~17 ms/tick measured).
        ex.kvs[0].pos.masked_fill_(self._park_dev, ex.smax - 1)
        ex.kvs[0].pos.masked_fill_(self._free_dev, 0)
        if self._launch_evs is not None:
            self._launch_evs[1].record()
        self._t_aux += time.perf_counter() - t0
        return (toks_t, [(i, self._slot_gen[i])
                         for i, st in enumerate(self.slots)
                         if st is not None and st.get("state") != "prefill"],
                rs.last_logp, rs.last_topv, rs.last_topi)

    def _slot_emit(self, toks_t, active, logp=None, topv=None,
                   topi=None) -> None:
        toks = toks_t.tolist()   # syncs; overlaps the already-launched step
        rs = self._row_sampler
        lp_l = logp.tolist() if logp is not None else None
        tv_l = topv.tolist() if topv is not None else None
        ti_l = topi.tolist() if topi is not None else None
        for i, gen in active:
            st = self.slots[i]
            if st is None or self._slot_gen[i] != gen:
                # freed since launch (stop lag), or freed AND re-admitted —
                # either way this row's token belongs to the old request
                continue
            t = int(toks[i])
            st["produced"] += 1
            done = (t in st["stop_ids"]
                    or st["produced"] >= st["max_tokens"])
            lp = lp_l[i] if (lp_l is not None and rs.want_lp[i]) else None
            tops = None
            if tv_l is not None and rs.n_top[i] > 0:
                k = rs.n_top[i]
                tops = {int(ti_l[i][j]): float(tv_l[i][j]) for j in range(k)}
            self._emit_token(st["nonce"], t, finished=done, logprob=lp,
                             tops=tops)
            if done:
                self.slots[i] = None
                self._row_sampler.clear_row(i)
                self._park_dev[i] = True
                self._free_dev[i] = True

    _emit_s = 0.0

    def _emit_token(self, nonce: str, token_id: int, finished: bool = False,
                    logprob=None, tops=None):
        ex = self.executor
        if self._callback is None:
            return
        t0 = time.perf_counter()
        frame = {"t": "token", "nonce": nonce, "token_id": token_id,
                 "ts_ms": int(time.time() * 1e3), "finished": finished}
        if logprob is None and ex.last_logprob is not None:
            logprob = float(ex.last_logprob[0])     # legacy serial path
        if tops is None and ex.last_tops is not None:
            tops = ex.last_tops[0]
        if logprob is not None:
            frame["logprob"] = logprob
        if tops is not None:
            frame["top_logprobs"] = tops
        try:
            self._callback.send(frame)
        except OSError:
            log.warning("token callback failed (api down?)")
            self._callback.close()
        finally:
            self._emit_s += time.perf_counter() - t0

    def _send_error(self, nonce: str):
        if self._callback is not None:
            try:
                self._callback.send({"t": "error", "nonce": nonce,
                                     "failed_node": self.instance,
                                     "code": 500, "error": self.last_error})
            except OSError:
                pass
