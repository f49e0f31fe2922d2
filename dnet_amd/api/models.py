"""API DTOs: OpenAI-compatible chat types + topology/load requests.

Reference counterpart: src/dnet/api/models.py.
"""
from __future__ import annotations

import time
import uuid
from typing import Any, Optional

from pydantic import BaseModel, Field


class ChatMessage(BaseModel):
    role: str
    content: Optional[str] = None
    name: Optional[str] = None


class ChatRequestModel(BaseModel):
    model: str
    messages: list[ChatMessage]
    max_tokens: Optional[int] = None
    max_completion_tokens: Optional[int] = None
    temperature: float = 0.0
    seed: int | None = None
    top_p: float = 1.0
    top_k: int = 0
    min_p: float = 0.0
    repetition_penalty: float = 1.0
    stream: bool = False
    stop: Optional[list[str] | str] = None
    logprobs: bool = False
    top_logprobs: int = 0
    profile: bool = False            # emit metrics in the final chunk
    n: int = 1

    @property
    def effective_max_tokens(self) -> int:
        return self.max_completion_tokens or self.max_tokens or 128


class UsageModel(BaseModel):
    prompt_tokens: int = 0
    completion_tokens: int = 0
    total_tokens: int = 0


class ChoiceDelta(BaseModel):
    role: Optional[str] = None
    content: Optional[str] = None


class StreamChoice(BaseModel):
    index: int = 0
    delta: ChoiceDelta = Field(default_factory=ChoiceDelta)
    finish_reason: Optional[str] = None
    logprobs: Optional[dict] = None


class ChatChunkModel(BaseModel):
    id: str
    object: str = "chat.completion.chunk"
    created: int = Field(default_factory=lambda: int(time.time()))
    model: str = ""
    choices: list[StreamChoice] = Field(default_factory=list)
    usage: Optional[UsageModel] = None
    metrics: Optional[dict] = None


class ChoiceMessage(BaseModel):
    role: str = "assistant"
    content: str = ""


class Choice(BaseModel):
    index: int = 0
    message: ChoiceMessage = Field(default_factory=ChoiceMessage)
    finish_reason: str = "stop"
    logprobs: Optional[dict] = None


class ChatResponseModel(BaseModel):
    id: str
    object: str = "chat.completion"
    created: int = Field(default_factory=lambda: int(time.time()))
    model: str = ""
    choices: list[Choice] = Field(default_factory=list)
    usage: UsageModel = Field(default_factory=UsageModel)
    metrics: Optional[dict] = None


def new_nonce() -> str:
    return f"chatcmpl-{uuid.uuid4()}"


class ModelInfo(BaseModel):
    id: str
    object: str = "model"
    created: int = 0
    owned_by: str = "dnet_amd"


class ModelListResponse(BaseModel):
    object: str = "list"
    data: list[ModelInfo] = Field(default_factory=list)


class PrepareTopologyRequest(BaseModel):
    model: str
    seq_len: int = 512
    batch_size: int = 1
    kv_bits: int = 16
    quant: str = ""


class ManualAssignment(BaseModel):
    instance: str
    layers: list[int]


class PrepareTopologyManualRequest(BaseModel):
    model: str
    assignments: list[ManualAssignment]
    kv_bits: int = 16
    quant: str = ""


class APILoadModelRequest(BaseModel):
    model: str
    quant: str = ""
    # >1 = slot-scheduled continuous batching (the production serving
    # mode: concurrent requests each own a KV slot; per-slot sampling,
    # seeds, logprobs and hipGraph decode all supported). 1 = serial.
    max_batch: int = 8
    max_seq: int = 4096
    warmup: bool = False
