"""Tensor <-> wire bytes helpers + canonical dtype alias maps.

Reference counterpart: src/dnet/utils/serialization.py (np/mlx/safetensors
dtype maps, tensor_to_bytes/bytes_to_tensor with BF16 fallback).
"""
from __future__ import annotations

import numpy as np
import torch

TORCH_DTYPES = {
    "float32": torch.float32, "float16": torch.float16,
    "bfloat16": torch.bfloat16, "int64": torch.int64, "int32": torch.int32,
    "int8": torch.int8, "uint8": torch.uint8, "bool": torch.bool,
}
SAFETENSORS_DTYPES = {"F32": "float32", "F16": "float16", "BF16": "bfloat16",
                      "I64": "int64", "I32": "int32", "I8": "int8",
                      "U8": "uint8", "BOOL": "bool"}


def dtype_name(t: torch.Tensor) -> str:
    return str(t.dtype).replace("torch.", "")


def tensor_to_bytes(t: torch.Tensor) -> tuple[bytes, str, tuple]:
    """-> (raw little-endian bytes, dtype name, shape). bf16 rides as its
    raw 16-bit pattern (numpy has no bf16)."""
    t = t.detach().contiguous().cpu()
    shape = tuple(t.shape)
    name = dtype_name(t)
    if t.dtype == torch.bfloat16:
        return t.view(torch.int16).numpy().tobytes(), name, shape
    return t.numpy().tobytes(), name, shape


def bytes_to_tensor(data: bytes, dtype: str, shape, device="cpu") -> torch.Tensor:
    td = TORCH_DTYPES[dtype]
    if td == torch.bfloat16:
        arr = np.frombuffer(data, dtype=np.int16).copy()
        return torch.from_numpy(arr).view(torch.bfloat16).view(*shape).to(device)
    np_dtype = np.dtype(dtype if dtype != "bool" else np.bool_)
    arr = np.frombuffer(data, dtype=np_dtype).copy()
    return torch.from_numpy(arr).view(*shape).to(device)
