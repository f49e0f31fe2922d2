"""API node HTTP server: OpenAI-compatible endpoints + cluster control.

Reference counterpart: src/dnet/api/http_api.py (routes /health,
/v1/chat/completions, /v1/load_model, /v1/unload_model, /v1/models,
/v1/topology, /v1/prepare_topology, /v1/prepare_topology_manual,
/v1/devices).
"""
from __future__ import annotations

import asyncio

import json
from typing import Optional

from fastapi import FastAPI, HTTPException
from fastapi.responses import StreamingResponse

from ..core.types import LayerAssignment, TopologyInfo
from ..utils.logger import get_logger
from .catalog import get_entry
from .cluster import ClusterManager
from .inference import InferenceManager
from .model_manager import ModelManager, resolve_model_config
from .models import (APILoadModelRequest, ChatRequestModel, ModelInfo,
                     ModelListResponse, PrepareTopologyManualRequest,
                     PrepareTopologyRequest)

log = get_logger("api")


class ApiState:
    def __init__(self, cluster: ClusterManager, settings):
        self.cluster = cluster
        self.last_load = None          # APILoadModelRequest for recovery
        self.models = ModelManager(cluster)
        self.inference = InferenceManager(
            self.models, token_timeout_s=settings.api.request_timeout_s)
        self.settings = settings


def build_api_app(state: ApiState) -> FastAPI:
    app = FastAPI(title="dnet_amd api")
    s = state

    @app.get("/health")
    async def health():
        return {"status": "ok", "model": s.models.loaded_model,
                "devices": len(s.cluster.devices)}

    @app.get("/v1/models")
    async def models():
        return ModelListResponse(data=[
            ModelInfo(id=e.id) for e in s.models.catalog_entries()])

    @app.get("/v1/devices")
    async def devices():
        await s.cluster.scan_devices()
        return {k: vars(v) for k, v in s.cluster.devices.items()}

    @app.get("/v1/topology")
    async def topology():
        if s.cluster.topology is None:
            raise HTTPException(404, "no topology prepared")
        return s.cluster.topology.model_dump()

    @app.post("/v1/prepare_topology")
    async def prepare_topology(req: PrepareTopologyRequest):
        entry = get_entry(req.model)
        if entry is None:
            raise HTTPException(404, f"unknown model {req.model}")
        cfg = resolve_model_config(entry, req.quant)
        s.cluster.excluded = set()   # explicit re-prepare forgives failures
        await s.cluster.profile_cluster()
        topo = s.cluster.solve_topology(
            req.model, cfg, master_port=s.settings.transport.master_port,
            kv_bits=req.kv_bits, batch=req.batch_size, seq_len=req.seq_len)
        return topo.model_dump()

    @app.post("/v1/prepare_topology_manual")
    async def prepare_topology_manual(req: PrepareTopologyManualRequest):
        entry = get_entry(req.model)
        if entry is None:
            raise HTTPException(404, f"unknown model {req.model}")
        cfg = resolve_model_config(entry, req.quant)
        await s.cluster.scan_devices()
        assignments = []
        insts = [a.instance for a in req.assignments]
        for i, a in enumerate(req.assignments):
            dev = s.cluster.devices.get(a.instance)
            if dev is None:
                raise HTTPException(400, f"unknown device {a.instance}")
            assignments.append(LayerAssignment(
                instance=a.instance, layers=[sorted(a.layers)],
                next_instance=insts[(i + 1) % len(insts)],  # auto ring closure
                window_size=len(a.layers), residency_size=len(a.layers),
                gpu_index=max(dev.gpu_index, 0)))
        covered = sorted(l for a in assignments for r in a.layers for l in r)
        if covered != list(range(cfg.num_layers)):
            raise HTTPException(400, "assignments must cover every layer once")
        head = s.cluster.devices[assignments[0].instance]
        topo = TopologyInfo(
            model=req.model, kv_bits=req.kv_bits, num_layers=cfg.num_layers,
            devices=insts, assignments=assignments,
            master_addr=head.local_ip,
            master_port=s.settings.transport.master_port)
        s.cluster.topology = topo
        return topo.model_dump()

    @app.post("/v1/load_model")
    async def load_model(req: APILoadModelRequest):
        entry = get_entry(req.model)
        if entry is None:
            raise HTTPException(404, f"unknown model {req.model}")
        if s.cluster.topology is None or s.cluster.topology.model != req.model:
            # bootstrap a topology inline (reference: http_api.py:144-181)
            cfg = resolve_model_config(entry, req.quant)
            await s.cluster.profile_cluster()
            s.cluster.solve_topology(
                req.model, cfg, master_port=s.settings.transport.master_port)
        cb = s.settings.api.callback_addr or \
            f"127.0.0.1:{s.settings.api.grpc_port}"
        try:
            await s.models.load_model(
                s.cluster.topology, entry, quant=req.quant,
                max_batch=req.max_batch, max_seq=req.max_seq,
                api_callback_address=cb)
        except Exception as e:
            log.exception("load_model failed")
            raise HTTPException(500, str(e))
        head = s.cluster.get_head_node()
        s.inference.connect_head(head.local_ip, head.shard_port, cb)
        s.last_load = req
        return {"status": "ok", "model": req.model,
                "topology": s.cluster.topology.model_dump()}

    @app.post("/v1/unload_model")
    async def unload_model():
        await s.models.unload_model()
        return {"status": "ok"}

    async def _do_recover():
        """Elastic recovery after a shard failure: health-sweep, exclude
        dead shards, re-solve the ring over the survivors and reload the
        last-loaded model (drop-and-reload; in-flight requests error out
        via the token timeout). Reference defines RingError but never
        recovers — a dead mid-ring shard stays a timeout there."""
        if s.last_load is None:
            raise HTTPException(400, "nothing was loaded")
        healthy = await s.cluster.healthy_shards()
        if not healthy:
            raise HTTPException(503, "no healthy shards")
        all_shards = {d.instance for d in s.cluster.devices.values()
                      if not d.is_manager}
        s.cluster.excluded = all_shards - {d.instance for d in healthy}
        await s.models.unload_model()          # best-effort on survivors
        req = s.last_load
        entry = get_entry(req.model)
        cfg = resolve_model_config(entry, req.quant)
        s.cluster.solve_topology(
            req.model, cfg, master_port=s.settings.transport.master_port)
        cb = s.settings.api.callback_addr or \
            f"127.0.0.1:{s.settings.api.grpc_port}"
        await s.models.load_model(
            s.cluster.topology, entry, quant=req.quant,
            max_batch=req.max_batch, max_seq=req.max_seq,
            api_callback_address=cb)
        head = s.cluster.get_head_node()
        s.inference.connect_head(head.local_ip, head.shard_port, cb)
        return {"status": "ok", "excluded": sorted(s.cluster.excluded),
                "topology": s.cluster.topology.model_dump()}

    @app.post("/v1/recover")
    async def recover():
        return await _do_recover()

    # failure -> automatic re-solve (recovery was
    # operator-only): the inference manager reports error frames / token
    # timeouts; one debounced background recover runs at a time
    if s.settings.api.auto_recover:
        async def _auto_recover():
            if getattr(s, "_recovering", False):
                return
            s._recovering = True
            try:
                log.warning("auto-recover: failure reported, re-solving")
                out = await _do_recover()
                log.warning("auto-recover done: excluded=%s",
                            out.get("excluded"))
            except Exception:
                log.exception("auto-recover failed")
            finally:
                s._recovering = False

        s.inference.on_failure = lambda: asyncio.get_event_loop(
            ).create_task(_auto_recover())

    @app.post("/v1/chat/completions")
    async def chat_completions(req: ChatRequestModel):
        if s.models.loaded_model == "":
            raise HTTPException(400, "no model loaded")
        if req.stream:
            async def sse():
                async for chunk in s.inference.generate_stream(req):
                    yield f"data: {chunk.model_dump_json(exclude_none=True)}\n\n"
                yield "data: [DONE]\n\n"
            return StreamingResponse(sse(), media_type="text/event-stream")
        return await s.inference.chat_completions(req)

    return app


async def api_wire_handler(state: ApiState):
    """Token-callback data plane on the API node (reference:
    ShardApiService.SendToken, src/dnet/api/grpc_servicer/servicer.py)."""
    async def handler(frame, writer):
        t = frame.get("t")
        if t in ("token", "error"):
            state.inference.resolve_token(frame)
            return None
        if t == "ping":
            return {"t": "pong"}
        return None
    return handler
