"""dnet_amd wire protocol: length-prefixed msgpack frames over TCP.

Replaces the reference's gRPC data plane (reference: src/dnet/protos/
dnet_ring.proto + shard_api_comm.proto). On MI355X the activation ring runs
over RCCL/xGMI, so the remaining wire traffic is small control/token
messages — a persistent TCP connection with 4-byte big-endian length +
msgpack body is lower-latency than HTTP/2 and needs no codegen.

Frame types (the protocol contract):
  api -> shard0 : {"t": "infer", "nonce", "tokens": bytes(int32 LE),
                   "prompt_len", "max_tokens", "params": {...},
                   "stop_ids": [...], "callback": "host:port"}
  api -> shard0 : {"t": "reset", "nonce"}
  api -> shard0 : {"t": "ping"} -> {"t": "pong"}
  last -> api   : {"t": "token", "nonce", "token_id", "ts_ms",
                   "logprob"?, "top_logprobs"?, "finished": bool}
  any  -> any   : {"t": "latency_probe", "payload": bytes} (echoed)
  shard-> api   : {"t": "error", "nonce", "failed_node", "code", "error"}
"""
from __future__ import annotations

import asyncio
import struct
from typing import Awaitable, Callable, Optional

import msgpack

Frame = dict
MAX_FRAME = 256 * 1024 * 1024


async def read_frame(reader: asyncio.StreamReader) -> Optional[Frame]:
    try:
        hdr = await reader.readexactly(4)
    except (asyncio.IncompleteReadError, ConnectionResetError):
        return None
    (n,) = struct.unpack(">I", hdr)
    if n > MAX_FRAME:
        raise ValueError(f"frame too large: {n}")
    body = await reader.readexactly(n)
    return msgpack.unpackb(body, raw=False)


async def write_frame(writer: asyncio.StreamWriter, frame: Frame) -> None:
    body = msgpack.packb(frame, use_bin_type=True)
    writer.write(struct.pack(">I", len(body)) + body)
    await writer.drain()


class WireServer:
    """Asyncio TCP server dispatching frames to a handler.

    handler(frame, writer) -> optional response frame.
    """

    def __init__(self, host: str, port: int,
                 handler: Callable[[Frame, asyncio.StreamWriter],
                                   Awaitable[Optional[Frame]]]):
        self.host = host
        self.port = port
        self.handler = handler
        self._server: Optional[asyncio.AbstractServer] = None

    async def start(self):
        self._server = await asyncio.start_server(self._on_conn, self.host,
                                                  self.port)

    async def _on_conn(self, reader: asyncio.StreamReader,
                       writer: asyncio.StreamWriter):
        try:
            while True:
                frame = await read_frame(reader)
                if frame is None:
                    break
                resp = await self.handler(frame, writer)
                if resp is not None:
                    await write_frame(writer, resp)
        except (ConnectionResetError, asyncio.CancelledError):
            pass
        finally:
            writer.close()

    async def stop(self):
        if self._server:
            self._server.close()
            await self._server.wait_closed()


class WireClient:
    """Persistent client connection with lazy reconnect."""

    def __init__(self, host: str, port: int):
        self.host = host
        self.port = port
        self._reader: Optional[asyncio.StreamReader] = None
        self._writer: Optional[asyncio.StreamWriter] = None
        self._lock = asyncio.Lock()

    async def _ensure(self):
        if self._writer is None or self._writer.is_closing():
            self._reader, self._writer = await asyncio.open_connection(
                self.host, self.port)

    async def send(self, frame: Frame) -> None:
        async with self._lock:
            await self._ensure()
            await write_frame(self._writer, frame)

    async def request(self, frame: Frame) -> Optional[Frame]:
        async with self._lock:
            await self._ensure()
            await write_frame(self._writer, frame)
            return await read_frame(self._reader)

    async def close(self):
        if self._writer is not None:
            self._writer.close()
            self._writer = None
