"""Shard control plane (HTTP) + data plane (wire protocol) servers.

Reference counterpart: src/dnet/shard/http_api.py (control: /health,
/load_model, /unload_model, /profile, /measure_latency, /cleanup_repacked)
+ grpc_servicer (data plane — here the wire-protocol TCP server that
receives "infer"/"reset" frames on the head shard).
"""
from __future__ import annotations

import asyncio
import threading
import time
from typing import Optional

from fastapi import FastAPI, HTTPException
from pydantic import BaseModel

from ..core.types import ShardLoadModelRequest
from ..protos.wire import WireServer
from ..utils.logger import get_logger
from .runtime import ShardRuntime

log = get_logger("shard")


class HealthResponse(BaseModel):
    status: str
    instance: str
    model: str = ""
    queue_depth: int = 0
    error: str = ""
    rank: int = -1                 # rank in the active ring (-1 = none)
    xgmi: dict = {}                # all-pairs link matrix ("i-j" -> ms/GBps)


class MeasureLatencyRequest(BaseModel):
    peers: list[dict]                # [{instance, host, port}]
    payload_sizes: list[int] = [4096, 1048576]
    reps: int = 5


def build_shard_app(rt: ShardRuntime) -> FastAPI:
    app = FastAPI(title="dnet_amd shard")

    @app.get("/health")
    def health() -> HealthResponse:
        return HealthResponse(status=rt.status, instance=rt.instance,
                              model=rt.model_name,
                              queue_depth=rt.infer_q.qsize(),
                              error=rt.last_error,
                              rank=getattr(rt, "rank", -1),
                              xgmi=getattr(rt, "link_matrix", {}) or {})

    @app.post("/load_model")
    def load_model(req: ShardLoadModelRequest) -> dict:
        try:
            rt.submit_load(req)
        except Exception as e:
            raise HTTPException(500, str(e))
        return {"status": "ok", "instance": rt.instance}

    @app.post("/unload_model")
    def unload_model() -> dict:
        rt.submit_unload()
        return {"status": "ok"}

    @app.post("/profile")
    def profile(quick: bool = True) -> dict:
        return rt.profile(quick=quick).to_dict()

    @app.post("/measure_latency")
    def measure_latency(req: MeasureLatencyRequest) -> dict:
        """TCP round-trip probe to each peer's wire port per payload size
        (reference: shard /measure_latency -> gRPC MeasureLatency sweep;
        the xGMI in-group sweep runs separately at load time)."""
        import socket
        import struct

        import msgpack
        out = {}
        for peer in req.peers:
            per_size = {}
            for size in req.payload_sizes:
                samples = []
                try:
                    with socket.create_connection(
                            (peer["host"], int(peer["port"])), timeout=5) as s:
                        for _ in range(req.reps):
                            body = msgpack.packb(
                                {"t": "latency_probe", "payload": b"x" * size},
                                use_bin_type=True)
                            t0 = time.perf_counter()
                            s.sendall(struct.pack(">I", len(body)) + body)
                            hdr = _recv_exact(s, 4)
                            (n,) = struct.unpack(">I", hdr)
                            _recv_exact(s, n)
                            samples.append((time.perf_counter() - t0) * 1e3)
                except OSError as e:
                    per_size[str(size)] = {"error": str(e)}
                    continue
                samples.sort()
                per_size[str(size)] = {
                    "median_ms": samples[len(samples) // 2],
                    "min_ms": samples[0]}
            out[peer.get("instance", peer["host"])] = per_size
        return {"latencies": out}

    @app.post("/cleanup_repacked")
    def cleanup_repacked(model: str = "") -> dict:
        from ..utils.repack import delete_repacked_layers
        n = delete_repacked_layers(model or None)
        return {"status": "ok", "deleted": n}

    return app


def _recv_exact(sock, n: int) -> bytes:
    buf = b""
    while len(buf) < n:
        chunk = sock.recv(n - len(buf))
        if not chunk:
            raise ConnectionError("peer closed")
        buf += chunk
    return buf


async def wire_handler_factory(rt: ShardRuntime):
    async def handler(frame, writer):
        t = frame.get("t")
        if t == "infer":
            rt.submit_infer(frame)
            return {"t": "ack"}
        if t == "reset":
            return {"t": "ack"}
        if t == "cancel":
            rt.submit_cancel(frame.get("nonce", ""))
            return {"t": "ack"}
        if t == "ping":
            return {"t": "pong"}
        if t == "latency_probe":
            return {"t": "latency_probe_ack", "n": len(frame.get("payload", b""))}
        return {"t": "error", "error": f"unknown frame type {t}"}
    return handler


def start_servers(rt: ShardRuntime, host: str, http_port: int,
                  wire_port: int) -> threading.Thread:
    """Run HTTP + wire servers on a background asyncio/uvicorn thread; the
    caller then runs rt.run() (the driver loop) on the main thread."""
    import uvicorn

    app = build_shard_app(rt)

    def _serve():
        loop = asyncio.new_event_loop()
        asyncio.set_event_loop(loop)

        async def main():
            handler = await wire_handler_factory(rt)
            wire = WireServer(host, wire_port, handler)
            await wire.start()
            config = uvicorn.Config(app, host=host, port=http_port,
                                    log_level="warning", loop="asyncio")
            server = uvicorn.Server(config)
            await server.serve()

        loop.run_until_complete(main())

    th = threading.Thread(target=_serve, daemon=True, name="shard-io")
    th.start()
    return th
