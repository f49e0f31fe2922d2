"""Per-layer weight repacking for fast cold loads / host-DRAM streaming.

Reference counterpart: src/dnet/utils/repack.py (repack assigned layers to
layer_%04d.safetensors + api_layers.safetensors + repack-manifest.json under
<repack_dir>/<model>/<sha1(layers)[:10]>, skipped when the manifest already
matches). On MI355X the repacked files feed the pinned-host weight cache of
the offload policy instead of mmap/madvise streaming.
"""
from __future__ import annotations

import hashlib
import json
import re
import shutil
from pathlib import Path

from .model_meta import ModelMetadata, get_model_metadata, load_tensors

MANIFEST = "repack-manifest.json"
VERSION = 1


def _root() -> Path:
    from ..config import get_settings
    return Path(get_settings().storage.repack_dir).expanduser()


def _sanitize(model_id: str) -> str:
    return re.sub(r"[^A-Za-z0-9_.-]", "_", model_id)


def layers_hash(layers: list[int]) -> str:
    return hashlib.sha1(",".join(map(str, sorted(layers))).encode()).hexdigest()[:10]


def repack_dir_for(model_id: str, layers: list[int]) -> Path:
    return _root() / _sanitize(model_id) / layers_hash(layers)


def _manifest_ok(out: Path, layers: list[int]) -> bool:
    man_path = out / MANIFEST
    if not man_path.exists():
        return False
    try:
        man = json.loads(man_path.read_text())
        return (man.get("version") == VERSION
                and man.get("layers_hash") == layers_hash(layers))
    except (json.JSONDecodeError, OSError):
        return False


def ensure_repacked_for_layers(model_dir: str, model_id: str,
                               layers: list[int], include_api: bool = True,
                               sd: dict | None = None) -> Path:
    """Rewrite the assigned layers into one safetensors file per layer.
    Returns the repack directory; no-op if the manifest already matches.
    When ``sd`` (an already-loaded name->tensor dict) is given, tensors
    come from it instead of re-reading the source safetensors — the cold
    load path repacks for free from what it just loaded."""
    from safetensors.torch import save_file
    out = repack_dir_for(model_id, layers)
    if _manifest_ok(out, layers):
        return out
    meta = get_model_metadata(model_dir)

    def grab(names):
        if sd is not None:
            got = {k: sd[k] for k in names if k in sd}
            return got if len(got) == len(names) else None
        return load_tensors(meta, names)

    out.mkdir(parents=True, exist_ok=True)
    files = []
    for lid in sorted(layers):
        names = meta.layers.get(lid, [])
        if not names:
            continue
        d = grab(names)
        if d is None:
            d = load_tensors(meta, names)
        fn = out / f"layer_{lid:04d}.safetensors"
        save_file({k: v.contiguous() for k, v in d.items()}, str(fn))
        files.append(fn.name)
    if include_api:
        api_names = meta.embed + meta.final_norm + meta.lm_head
        if api_names:
            d = grab(api_names)
            if d is None:
                d = load_tensors(meta, api_names)
            fn = out / "api_layers.safetensors"
            save_file({k: v.contiguous() for k, v in d.items()}, str(fn))
            files.append(fn.name)
    man_path = out / MANIFEST
    man_path.write_text(json.dumps({
        "version": VERSION, "model_id": model_id,
        "assigned_layers": sorted(layers),
        "layers_hash": layers_hash(layers), "files": files}))
    return out


def load_repacked(model_id: str, layers: list[int],
                  include_api: bool = True) -> dict | None:
    """Load the assigned layers from the repacked per-layer files if a
    matching manifest exists (the cold-load fastpath: no full-safetensors
    header parse / sharded reads). Returns name->tensor or None."""
    from safetensors.torch import load_file
    out = repack_dir_for(model_id, layers)
    if not _manifest_ok(out, layers):
        return None
    sd: dict = {}
    try:
        for lid in sorted(layers):
            fn = out / f"layer_{lid:04d}.safetensors"
            if fn.exists():
                sd.update(load_file(str(fn)))
        if include_api:
            fn = out / "api_layers.safetensors"
            if fn.exists():
                sd.update(load_file(str(fn)))
    except (OSError, RuntimeError):
        return None
    return sd


def delete_repacked_layers(model_id: str | None = None) -> int:
    """Delete repacked trees for one model (or all). Returns dirs removed."""
    root = _root()
    if not root.exists():
        return 0
    n = 0
    targets = [root / _sanitize(model_id)] if model_id else list(root.iterdir())
    for t in targets:
        if t.exists() and t.is_dir():
            shutil.rmtree(t, ignore_errors=True)
            n += 1
    return n
