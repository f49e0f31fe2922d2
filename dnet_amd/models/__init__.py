"""Model registry: model_type -> RingModel class.

Reference counterpart: src/dnet/core/models/__init__.py get_ring_model.
"""
from __future__ import annotations

from .base import KVCache, Linear, RingModel
from .config import PRESETS, ModelConfig, QuantConfig

_REGISTRY: dict[str, type] = {}


def register(cls):
    for mt in getattr(cls, "model_types", [cls.model_type]):
        _REGISTRY[mt] = cls
    return cls


def get_ring_model(model_type: str) -> type:
    if model_type in _REGISTRY:
        return _REGISTRY[model_type]
    raise ValueError(f"unsupported model_type: {model_type} "
                     f"(have {sorted(_REGISTRY)})")


# llama family covers llama / mistral / qwen2 (bias) / qwen3 (qk-norm)
RingModel.model_types = ["llama", "mistral", "qwen2", "qwen3"]
register(RingModel)

from .moe import MoERingModel  # noqa: E402

register(MoERingModel)  # mixtral / qwen2_moe / qwen3_moe

from .gpt_oss import GptOssRingModel  # noqa: E402

register(GptOssRingModel)

from .deepseek_v2 import DeepseekV2RingModel  # noqa: E402

register(DeepseekV2RingModel)

__all__ = ["ModelConfig", "QuantConfig", "RingModel", "MoERingModel", "KVCache",
           "Linear", "PRESETS", "get_ring_model", "register"]
