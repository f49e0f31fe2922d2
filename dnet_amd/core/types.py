"""Core DTOs: the topology checkpoint format + inference messages.

``TopologyInfo`` / ``LayerAssignment`` keep the reference's JSON shape
(reference: src/dnet/core/types/topology.py:14-51 — the format returned by
/v1/prepare_topology and consumed by /v1/load_model), extended with the
torch.distributed rendezvous info the RCCL ring needs.
"""
from __future__ import annotations

from typing import Optional

from pydantic import BaseModel, Field


class LayerAssignment(BaseModel):
    instance: str
    layers: list[list[int]]          # per-round layer lists (k rounds)
    next_instance: str = ""
    window_size: int = 0
    residency_size: int = 0
    gpu_index: int = -1


class TopologyInfo(BaseModel):
    model: str
    kv_bits: int = 16
    num_layers: int = 0
    devices: list[str] = Field(default_factory=list)
    assignments: list[LayerAssignment] = Field(default_factory=list)
    solution: dict = Field(default_factory=dict)     # solver diagnostics
    # RCCL/xGMI ring bootstrap (MI355X extension)
    master_addr: str = "127.0.0.1"
    master_port: int = 29500

    def assignment_for(self, instance: str) -> Optional[LayerAssignment]:
        for a in self.assignments:
            if a.instance == instance:
                return a
        return None


class DecodingParams(BaseModel):
    temperature: float = 0.0
    top_p: float = 1.0
    top_k: int = 0
    min_p: float = 0.0
    repetition_penalty: float = 1.0
    logprobs: bool = False
    top_logprobs: int = 0


class TokenResult(BaseModel):
    nonce: str
    token_id: int
    timestamp_ms: int = 0
    logprob: Optional[float] = None
    top_logprobs: Optional[dict] = None
    finished: bool = False


class ShardLoadModelRequest(BaseModel):
    """Per-shard load request — the shard-side checkpoint format
    (reference: src/dnet/shard/models.py ShardLoadModelRequest)."""
    model_path: str
    model_name: str = ""
    total_layers: int = 0
    layers: list[int] = Field(default_factory=list)   # flattened local layers
    layer_rounds: Optional[list] = None               # per-round windows (k>1)
    next_node: str = ""
    window_size: int = 0
    residency_size: int = 0
    kv_bits: int = 16
    api_callback_address: str = ""
    # RCCL group bootstrap
    rank: int = 0
    world_size: int = 1
    master_addr: str = "127.0.0.1"
    master_port: int = 29500
    gpu_index: int = 0
    max_batch: int = 8
    max_seq: int = 4096
    quant: str = ""                   # "" | "int8-g128" | ...
    warmup: bool = False
