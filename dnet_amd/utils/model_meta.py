"""Model metadata: safetensors header parsing + per-layer weight buckets.

Reference counterpart: src/dnet/utils/model.py (TensorInfo / ModelMetadata /
get_model_metadata). Headers are parsed directly (8-byte length + JSON) so
per-layer byte sizes are known without loading tensors; actual tensor reads
go through safetensors.safe_open (mmap-backed lazy loads feeding the
pinned-host staging of the weight cache).
"""
from __future__ import annotations

import json
import re
import struct
from dataclasses import dataclass, field
from pathlib import Path
from typing import Iterator, Optional

_LAYER_RE = re.compile(r"^(?:model\.)?layers\.(\d+)\.(.+)$")
_DTYPE_BYTES = {"F64": 8, "F32": 4, "F16": 2, "BF16": 2, "I64": 8, "I32": 4,
                "I16": 2, "I8": 1, "U8": 1, "BOOL": 1, "F8_E4M3": 1,
                "F8_E5M2": 1}


@dataclass
class TensorInfo:
    name: str
    dtype: str
    shape: tuple
    offset: int        # data offset within the file's data section
    nbytes: int
    filename: str


@dataclass
class ModelMetadata:
    """Per-layer / embed / lm_head / norm tensor buckets for a model dir."""
    model_dir: str
    config: dict = field(default_factory=dict)
    tensors: dict = field(default_factory=dict)          # name -> TensorInfo
    layers: dict = field(default_factory=dict)           # layer id -> [names]
    embed: list = field(default_factory=list)
    lm_head: list = field(default_factory=list)
    final_norm: list = field(default_factory=list)
    other: list = field(default_factory=list)

    @property
    def num_layers(self) -> int:
        return (max(self.layers) + 1) if self.layers else 0

    def layer_bytes(self, lid: int) -> int:
        return sum(self.tensors[n].nbytes for n in self.layers.get(lid, []))

    def bucket_bytes(self, names: list) -> int:
        return sum(self.tensors[n].nbytes for n in names)


def parse_safetensors_header(path: Path) -> Iterator[TensorInfo]:
    with open(path, "rb") as f:
        n = struct.unpack("<Q", f.read(8))[0]
        header = json.loads(f.read(n))
    for name, info in header.items():
        if name == "__metadata__":
            continue
        o0, o1 = info["data_offsets"]
        yield TensorInfo(name=name, dtype=info["dtype"],
                         shape=tuple(info["shape"]), offset=o0,
                         nbytes=o1 - o0, filename=str(path))


def get_model_metadata(model_dir: str) -> ModelMetadata:
    d = Path(model_dir).expanduser()
    meta = ModelMetadata(model_dir=str(d))
    cfg_path = d / "config.json"
    if cfg_path.exists():
        meta.config = json.loads(cfg_path.read_text())
    for st in sorted(d.glob("*.safetensors")):
        for ti in parse_safetensors_header(st):
            meta.tensors[ti.name] = ti
            m = _LAYER_RE.match(ti.name)
            if m:
                meta.layers.setdefault(int(m.group(1)), []).append(ti.name)
            elif "embed_tokens" in ti.name:
                meta.embed.append(ti.name)
            elif "lm_head" in ti.name:
                meta.lm_head.append(ti.name)
            elif ti.name in ("model.norm.weight", "norm.weight"):
                meta.final_norm.append(ti.name)
            else:
                meta.other.append(ti.name)
    return meta


def load_tensors(meta: ModelMetadata, names: list, device="cpu") -> dict:
    """Load the named tensors (grouped per file) via safetensors mmap."""
    from safetensors import safe_open
    by_file: dict[str, list] = {}
    for n in names:
        by_file.setdefault(meta.tensors[n].filename, []).append(n)
    out = {}
    for fn, ns in by_file.items():
        with safe_open(fn, framework="pt", device=str(device)) as f:
            for n in ns:
                out[n] = f.get_tensor(n)
    return out
