"""Supported-model catalog (reference: src/dnet/api/catalog.py).

Each entry: HF repo id (or local dir / synthetic preset), quantization and
test flags. ``ci_test`` models run in the localhost integration suite with
random-init weights + the byte tokenizer (no network in CI).
"""
from __future__ import annotations

from dataclasses import dataclass, field


@dataclass
class CatalogEntry:
    id: str
    preset: str = ""                 # synthetic preset key (models.config.PRESETS)
    repo: str = ""                   # HF repo / local dir with safetensors
    quant: str = ""                  # "" (bf16) | "int8-g128" | "int8-g64"
    tokenizer: str = "auto"          # "auto" (transformers) | "byte"
    ci_test: bool = False


model_catalog: list[CatalogEntry] = [
    # synthetic presets (random-init; always available — no network)
    CatalogEntry("tiny-random", preset="tiny", tokenizer="byte", ci_test=True),
    CatalogEntry("opt-125m-like", preset="opt-125m-like", tokenizer="byte",
                 ci_test=True),
    CatalogEntry("llama-3-8b-synthetic", preset="llama-3-8b", tokenizer="byte"),
    CatalogEntry("qwen-2.5-32b-int8-synthetic", preset="qwen-2.5-32b",
                 quant="int8-g128", tokenizer="byte"),
    CatalogEntry("qwen-2.5-32b-int4-synthetic", preset="qwen-2.5-32b",
                 quant="int4-g128", tokenizer="byte"),
    CatalogEntry("llama-3-70b-int4-synthetic", preset="llama-3-70b",
                 quant="int4-g128", tokenizer="byte"),
    CatalogEntry("llama-3-70b-synthetic", preset="llama-3-70b", tokenizer="byte"),
    CatalogEntry("mixtral-8x7b-synthetic", preset="mixtral-8x7b", tokenizer="byte"),
    CatalogEntry("gpt-oss-20b-synthetic", preset="gpt-oss-20b", tokenizer="byte"),
    CatalogEntry("deepseek-v2-lite-synthetic", preset="deepseek-v2-lite",
                 tokenizer="byte"),
    CatalogEntry("deepseek-v3-lite-synthetic", preset="deepseek-v3-lite",
                 tokenizer="byte"),
    CatalogEntry("openai/gpt-oss-20b", repo="openai/gpt-oss-20b"),
    CatalogEntry("deepseek-ai/DeepSeek-V2-Lite", repo="deepseek-ai/DeepSeek-V2-Lite"),
    # real checkpoints (local safetensors dir or pre-downloaded HF cache)
    CatalogEntry("meta-llama/Llama-3.1-8B-Instruct",
                 repo="meta-llama/Llama-3.1-8B-Instruct"),
    CatalogEntry("meta-llama/Llama-3.3-70B-Instruct",
                 repo="meta-llama/Llama-3.3-70B-Instruct"),
    CatalogEntry("Qwen/Qwen2.5-32B-Instruct",
                 repo="Qwen/Qwen2.5-32B-Instruct", quant="int8-g128"),
    CatalogEntry("Qwen/Qwen3-8B", repo="Qwen/Qwen3-8B"),
    CatalogEntry("mistralai/Mixtral-8x7B-Instruct-v0.1",
                 repo="mistralai/Mixtral-8x7B-Instruct-v0.1"),
]


def get_entry(model_id: str) -> CatalogEntry | None:
    for e in model_catalog:
        if e.id == model_id:
            return e
    return None


def get_ci_test_models() -> list[CatalogEntry]:
    return [e for e in model_catalog if e.ci_test]
