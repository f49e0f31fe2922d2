"""InferenceManager: the decode driver on the API node.

Reference counterpart: src/dnet/api/inference.py generate_stream — chat
templating, nonce allocation, sending the request to the head shard over
the wire protocol, awaiting per-token callbacks (Future map keyed by
nonce), incremental detokenization and SSE chunk emission with usage and
optional `profile: true` metrics (ttfb_ms / tps_decoding — the same metrics
shape as the reference so harnesses stay comparable).
"""
from __future__ import annotations

import asyncio
import json
import time
from typing import AsyncIterator, Optional

import numpy as np

from ..protos.wire import WireClient
from ..utils.logger import get_logger
from .models import (ChatChunkModel, ChatRequestModel, ChatResponseModel,
                     Choice, ChoiceDelta, ChoiceMessage, StreamChoice,
                     UsageModel, new_nonce)
from .tokenizer import Detokenizer

log = get_logger("api")


class InferenceManager:
    def __init__(self, model_manager, token_timeout_s: float = 300.0):
        self.mm = model_manager
        self.token_timeout_s = token_timeout_s
        self.pending: dict[str, asyncio.Queue] = {}
        self.head_client: Optional[WireClient] = None
        self.callback_addr: str = ""
        # set by the API server when auto-recover is enabled: invoked on
        # error frames / token timeouts (fire-and-forget)
        self.on_failure = None

    def _notify_failure(self):
        if self.on_failure is not None:
            try:
                self.on_failure()
            except Exception:
                log.exception("on_failure hook failed")

    def connect_head(self, host: str, port: int, callback_addr: str):
        self.head_client = WireClient(host, port)
        self.callback_addr = callback_addr

    def resolve_token(self, frame: dict):
        """Called by the API wire server on an incoming token frame."""
        q = self.pending.get(frame.get("nonce", ""))
        if q is not None:
            q.put_nowait(frame)

    def _encode_prompt(self, request: ChatRequestModel) -> list[int]:
        tok = self.mm.tokenizer
        msgs = [m.model_dump() for m in request.messages]
        try:
            text = tok.apply_chat_template(msgs, add_generation_prompt=True,
                                           tokenize=False)
        except Exception:
            text = "\n".join(m.get("content") or "" for m in msgs)
        return tok.encode(text)

    async def generate_stream(self, request: ChatRequestModel
                              ) -> AsyncIterator[ChatChunkModel]:
        if self.head_client is None:
            raise RuntimeError("no ring connected — load a model first")
        nonce = new_nonce()
        prompt_ids = self._encode_prompt(request)
        q: asyncio.Queue = asyncio.Queue()
        self.pending[nonce] = q
        detok = Detokenizer(self.mm.tokenizer)
        t_start = time.perf_counter()
        ttfb_ms = None
        n_tokens = 0
        finish_reason = "stop"
        # OpenAI `stop` strings: matched API-side in the detokenized text
        # (reference ignores them silently — advisor r1). Text that could
        # still grow into a stop string is held back until disambiguated.
        stops = request.stop
        stops = ([stops] if isinstance(stops, str) else list(stops or []))
        stops = [s for s in stops if s]
        pend = ""

        def _split_pending(buf: str):
            """(emit_now, keep, hit): earliest stop occurrence wins; else
            hold back the longest tail that is a proper prefix of a stop."""
            cut = min((i for i in (buf.find(s) for s in stops) if i >= 0),
                      default=-1)
            if cut >= 0:
                return buf[:cut], "", True
            hold = 0
            for s in stops:
                for k in range(min(len(s) - 1, len(buf)), hold, -1):
                    if buf.endswith(s[:k]):
                        hold = k
                        break
            return (buf[:-hold] if hold else buf), (buf[-hold:] if hold
                                                    else ""), False
        try:
            await self.head_client.request({
                "t": "infer", "nonce": nonce,
                "tokens": np.asarray(prompt_ids, dtype=np.int32).tobytes(),
                "prompt_len": len(prompt_ids),
                "max_tokens": request.effective_max_tokens,
                "params": {
                    "seed": request.seed,
                    "temperature": request.temperature, "top_p": request.top_p,
                    "top_k": request.top_k, "min_p": request.min_p,
                    "logprobs": request.logprobs,
                    "top_logprobs": request.top_logprobs},
                "stop_ids": list(self.mm.stop_ids),
                "callback": self.callback_addr})
            yield ChatChunkModel(id=nonce, model=request.model, choices=[
                StreamChoice(delta=ChoiceDelta(role="assistant", content=""))])
            while True:
                try:
                    frame = await asyncio.wait_for(
                        q.get(), timeout=self.token_timeout_s)
                except asyncio.TimeoutError:
                    self._notify_failure()
                    raise
                if frame.get("t") == "error":
                    self._notify_failure()
                    raise RuntimeError(frame.get("error", "ring error"))
                tid = int(frame["token_id"])
                n_tokens += 1
                if ttfb_ms is None:
                    ttfb_ms = (time.perf_counter() - t_start) * 1e3
                is_stop = tid in self.mm.stop_ids
                hit_stop_str = False
                if not is_stop:
                    delta = detok.add_token(tid)
                    if delta and stops:
                        pend += delta
                        delta, pend, hit_stop_str = _split_pending(pend)
                    if delta:
                        lp = None
                        if request.logprobs and "logprob" in frame:
                            lp = {"content": [{
                                "token": delta,
                                "logprob": frame.get("logprob"),
                                "top_logprobs": [
                                    {"token": str(k), "logprob": v}
                                    for k, v in (frame.get("top_logprobs")
                                                 or {}).items()]}]}
                        yield ChatChunkModel(
                            id=nonce, model=request.model,
                            choices=[StreamChoice(
                                delta=ChoiceDelta(content=delta),
                                logprobs=lp)])
                if hit_stop_str:
                    # stop the ring early (frees the slot / legacy loop);
                    # any in-flight tokens after this are dropped with the
                    # pending queue
                    try:
                        await self.head_client.request(
                            {"t": "cancel", "nonce": nonce})
                    except Exception:
                        log.warning("cancel send failed", exc_info=True)
                    break
                if frame.get("finished") or is_stop:
                    if not is_stop and n_tokens >= request.effective_max_tokens:
                        finish_reason = "length"
                    break
        finally:
            self.pending.pop(nonce, None)
        if pend:     # held-back text that never completed a stop string
            yield ChatChunkModel(id=nonce, model=request.model, choices=[
                StreamChoice(delta=ChoiceDelta(content=pend))])
        total_ms = (time.perf_counter() - t_start) * 1e3
        usage = UsageModel(prompt_tokens=len(prompt_ids),
                           completion_tokens=n_tokens,
                           total_tokens=len(prompt_ids) + n_tokens)
        metrics = None
        if request.profile:
            gen_ms = total_ms - (ttfb_ms or 0.0)
            metrics = {
                "total_ms": total_ms, "ttfb_ms": ttfb_ms,
                "token_gen_ms": gen_ms, "tokens_generated": n_tokens,
                "tps_overall": n_tokens / (total_ms / 1e3) if total_ms else 0,
                "tps_decoding": ((n_tokens - 1) / (gen_ms / 1e3)
                                 if gen_ms > 0 and n_tokens > 1 else 0)}
        yield ChatChunkModel(
            id=nonce, model=request.model,
            choices=[StreamChoice(delta=ChoiceDelta(),
                                  finish_reason=finish_reason)],
            usage=usage, metrics=metrics)

    async def chat_completions(self, request: ChatRequestModel
                               ) -> ChatResponseModel:
        """Non-streaming = collect the stream (reference: inference.py:255)."""
        content = []
        usage = UsageModel()
        metrics = None
        finish = "stop"
        nonce = ""
        async for chunk in self.generate_stream(request):
            nonce = chunk.id
            for c in chunk.choices:
                if c.delta.content:
                    content.append(c.delta.content)
                if c.finish_reason:
                    finish = c.finish_reason
            if chunk.usage:
                usage = chunk.usage
            if chunk.metrics:
                metrics = chunk.metrics
        return ChatResponseModel(
            id=nonce, model=request.model,
            choices=[Choice(message=ChoiceMessage(content="".join(content)),
                            finish_reason=finish)],
            usage=usage, metrics=metrics)
