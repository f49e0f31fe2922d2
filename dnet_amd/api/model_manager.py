"""ModelManager: per-shard load/unload fan-out + tokenizer ownership
(reference: src/dnet/api/model_manager.py).
"""
from __future__ import annotations

from pathlib import Path
from typing import Optional

import httpx

from ..core.types import ShardLoadModelRequest, TopologyInfo
from ..models import ModelConfig, PRESETS, QuantConfig
from ..utils.hostfile import DeviceProperties
from ..utils.logger import get_logger
from .catalog import CatalogEntry, get_entry, model_catalog
from .tokenizer import load_tokenizer, stop_token_ids

log = get_logger("api")


def resolve_model_config(entry: CatalogEntry, quant: str = "") -> ModelConfig:
    import json
    q = quant or entry.quant
    qc = None
    if q.startswith("int"):
        bits = int(q[3])
        group = int(q.split("-g")[1]) if "-g" in q else 128
        qc = QuantConfig(bits, group)
    if entry.preset:
        return ModelConfig.from_hf(dict(PRESETS[entry.preset]), quant=qc)
    p = Path(entry.repo).expanduser()
    if (p / "config.json").exists():
        return ModelConfig.from_hf(json.loads((p / "config.json").read_text()),
                                   quant=qc)
    raise FileNotFoundError(
        f"{entry.id}: no local checkpoint at {entry.repo} and no preset")


class ModelManager:
    def __init__(self, cluster):
        self.cluster = cluster
        self.loaded_model: str = ""
        self.tokenizer = None
        self.stop_ids: list[int] = []
        self.config: Optional[ModelConfig] = None

    def is_model_available(self, model_id: str) -> bool:
        return get_entry(model_id) is not None

    async def load_model(self, topology: TopologyInfo, entry: CatalogEntry,
                         quant: str = "", max_batch: int = 1,
                         max_seq: int = 4096,
                         api_callback_address: str = "") -> None:
        cfg = resolve_model_config(entry, quant)
        self.config = cfg
        world = len(topology.assignments)
        # Concurrent fan-out: ranks block in the collective process-group
        # join, so sequential posting would deadlock (unlike the reference's
        # independent gRPC shards).
        import asyncio
        async with httpx.AsyncClient(timeout=None) as client:
            posts = []
            for rank, a in enumerate(topology.assignments):
                dev = self.cluster.devices.get(a.instance)
                if dev is None:
                    raise RuntimeError(f"unknown device {a.instance}")
                req = ShardLoadModelRequest(
                    model_path=entry.repo or (entry.preset or entry.id),
                    model_name=entry.preset or entry.id,
                    total_layers=topology.num_layers,
                    layers=[l for round_ in a.layers for l in round_],
                    layer_rounds=a.layers,
                    next_node=a.next_instance,
                    window_size=a.window_size,
                    residency_size=a.residency_size,
                    kv_bits=topology.kv_bits,
                    api_callback_address=api_callback_address,
                    rank=rank, world_size=world,
                    master_addr=topology.master_addr,
                    master_port=topology.master_port,
                    gpu_index=max(a.gpu_index, 0),
                    max_batch=max_batch, max_seq=max_seq,
                    quant=quant or entry.quant)
                posts.append(client.post(
                    f"http://{dev.local_ip}:{dev.server_port}/load_model",
                    json=req.model_dump()))
            results = await asyncio.gather(*posts, return_exceptions=True)
            for a, r in zip(topology.assignments, results):
                if isinstance(r, Exception):
                    raise RuntimeError(f"load_model failed on {a.instance}: {r}")
                if r.status_code != 200:
                    raise RuntimeError(
                        f"load_model failed on {a.instance}: {r.text}")
        self.tokenizer = load_tokenizer(entry.tokenizer, entry.repo or "",
                                        vocab_size=cfg.vocab_size)
        self.stop_ids = stop_token_ids(self.tokenizer)
        self.loaded_model = entry.id
        log.info("model %s loaded on %d shard(s)", entry.id, world)

    async def unload_model(self) -> None:
        async with httpx.AsyncClient(timeout=60.0) as client:
            for d in self.cluster.shard_devices():
                try:
                    await client.post(
                        f"http://{d.local_ip}:{d.server_port}/unload_model")
                except httpx.HTTPError:
                    log.warning("unload failed on %s", d.instance)
        self.loaded_model = ""
        self.tokenizer = None
        self.config = None

    def catalog_entries(self):
        return model_catalog
