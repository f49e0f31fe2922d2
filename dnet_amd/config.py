"""dnet_amd settings tree: nested groups, DNET_<GROUP>_* env overrides.

Mirrors the reference's pydantic-settings tree (reference: src/dnet/config.py
DnetSettings with per-group env prefixes) on plain pydantic (pydantic-settings
is not in this image). ``get_settings()`` is cached; ``reset_settings()``
re-reads the environment (tests).
"""
from __future__ import annotations

import os
from functools import lru_cache

from pydantic import BaseModel, Field


def _env_override(model: BaseModel, prefix: str) -> None:
    for name, field in type(model).model_fields.items():
        env = f"{prefix}{name.upper()}"
        if env in os.environ:
            raw = os.environ[env]
            ann = field.annotation
            try:
                if ann is bool:
                    val = raw.lower() in ("1", "true", "yes", "on")
                elif ann is int:
                    val = int(raw)
                elif ann is float:
                    val = float(raw)
                else:
                    val = raw
                setattr(model, name, val)
            except ValueError:
                pass


class LoggingSettings(BaseModel):
    level: str = "INFO"
    dir: str = "~/.dnet_amd/logs"


class ApiSettings(BaseModel):
    host: str = "0.0.0.0"
    port: int = 8080
    grpc_port: int = 50051          # token-callback data plane
    callback_addr: str = ""          # override for SendToken target
    request_timeout_s: float = 300.0  # per-token timeout (reference: 300 s)
    # on an error frame or token timeout, automatically health-sweep,
    # exclude dead shards, re-solve and reload (the /v1/recover flow) —
    # the reference never recovers (RingError defined but unsent)
    auto_recover: bool = True


class ShardSettings(BaseModel):
    host: str = "0.0.0.0"
    http_port: int = 8081
    grpc_port: int = 50052
    queue_size: int = 256
    kv_ttl_s: float = 30.0


class ComputeSettings(BaseModel):
    device: str = "auto"             # cuda:N or cpu
    use_graphs: bool = True
    max_batch: int = 16
    max_seq: int = 4096
    wire_dtype: str = "bfloat16"     # activation dtype on the wire


class KVCacheSettings(BaseModel):
    bits: int = 16                   # 16 = bf16 (8/4-bit quantized KV later)
    group: int = 64


class TransportSettings(BaseModel):
    compress: bool = False           # column-sparsification wire compression
    compress_ratio: float = 90.0     # percent of columns kept
    master_port: int = 29500         # torch.distributed rendezvous


class StorageSettings(BaseModel):
    model_dir: str = "~/.dnet_amd/models"
    repack_dir: str = "~/.dnet_amd/repacked_layers"
    # write per-layer repacked safetensors on first cold load and load
    # from them on later loads (reference wires repack into OffloadPolicy,
    # src/dnet/shard/policies/offload.py:46-77; here it serves every load)
    repack_on_load: bool = True


class TopologySettings(BaseModel):
    window_size: int = 0             # 0 = solver decides
    residency_size: int = 0
    mip_gap: float = 1e-4


class ObsSettings(BaseModel):
    enabled: bool = False
    profile: bool = False            # [PROFILE] log lines
    sync_per_layer: bool = False
    sync_every_n: int = 0


class DnetSettings(BaseModel):
    logging: LoggingSettings = Field(default_factory=LoggingSettings)
    api: ApiSettings = Field(default_factory=ApiSettings)
    shard: ShardSettings = Field(default_factory=ShardSettings)
    compute: ComputeSettings = Field(default_factory=ComputeSettings)
    kv_cache: KVCacheSettings = Field(default_factory=KVCacheSettings)
    transport: TransportSettings = Field(default_factory=TransportSettings)
    storage: StorageSettings = Field(default_factory=StorageSettings)
    topology: TopologySettings = Field(default_factory=TopologySettings)
    observability: ObsSettings = Field(default_factory=ObsSettings)


_PREFIXES = {
    "logging": "DNET_LOG_",
    "api": "DNET_API_",
    "shard": "DNET_SHARD_",
    "compute": "DNET_COMPUTE_",
    "kv_cache": "DNET_KV_",
    "transport": "DNET_TRANSPORT_",
    "storage": "DNET_STORAGE_",
    "topology": "DNET_TOPOLOGY_",
    "observability": "DNET_OBS_",
}


@lru_cache(maxsize=1)
def get_settings() -> DnetSettings:
    s = DnetSettings()
    for group, prefix in _PREFIXES.items():
        _env_override(getattr(s, group), prefix)
    return s


def reset_settings() -> None:
    get_settings.cache_clear()
