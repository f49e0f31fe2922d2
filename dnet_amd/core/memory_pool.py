"""Size-bucketed tensor pools: pinned-host staging + device scratch.

Reference counterpart: src/dnet/core/memory/memory_pool.py
(DynamicMemoryPool / LayerAwareMemoryPool — exact-size buffer reuse with a
byte budget and LRU eviction of free buffers). On MI355X the torch caching
allocator already pools DEVICE memory, so the load-bearing use here is
PINNED HOST staging buffers (torch does not pool pin_memory allocations)
and long-lived per-layer recv buffers whose stats the layer-aware wrapper
tracks.
"""
from __future__ import annotations

import threading
from collections import OrderedDict
from typing import Optional

import torch


class DynamicMemoryPool:
    """Exact-(shape,dtype) buffer reuse with a byte budget; LRU eviction of
    FREE buffers only. Thread-safe; single-owner use is lock-cheap."""

    def __init__(self, max_bytes: int = 2 << 30, device: str = "cpu",
                 pin: bool = False):
        self.max_bytes = max_bytes
        self.device = torch.device(device)
        self.pin = pin and torch.cuda.is_available() and self.device.type == "cpu"
        self._free: OrderedDict[tuple, list] = OrderedDict()
        self._bytes = 0
        self._lock = threading.Lock()
        self.hits = 0
        self.misses = 0

    @staticmethod
    def _key(shape, dtype) -> tuple:
        return (tuple(shape), dtype)

    def acquire(self, shape, dtype=torch.bfloat16) -> torch.Tensor:
        key = self._key(shape, dtype)
        with self._lock:
            bucket = self._free.get(key)
            if bucket:
                t = bucket.pop()
                if not bucket:
                    del self._free[key]
                self._bytes -= t.numel() * t.element_size()
                self.hits += 1
                return t
            self.misses += 1
        t = torch.empty(shape, dtype=dtype, device=self.device)
        if self.pin:
            t = t.pin_memory()
        return t

    def release(self, t: torch.Tensor) -> None:
        nb = t.numel() * t.element_size()
        key = self._key(t.shape, t.dtype)
        with self._lock:
            self._free.setdefault(key, []).append(t)
            self._free.move_to_end(key)
            self._bytes += nb
            while self._bytes > self.max_bytes and self._free:
                k, bucket = next(iter(self._free.items()))
                victim = bucket.pop(0)
                self._bytes -= victim.numel() * victim.element_size()
                if not bucket:
                    del self._free[k]

    @property
    def free_bytes(self) -> int:
        return self._bytes


class LayerAwareMemoryPool(DynamicMemoryPool):
    """DynamicMemoryPool + per-layer acquire stats (reference:
    memory_pool.py LayerAwareMemoryPool)."""

    def __init__(self, *a, **kw):
        super().__init__(*a, **kw)
        self.layer_stats: dict[int, dict] = {}

    def acquire_for_layer(self, layer_id: int, shape,
                          dtype=torch.bfloat16) -> torch.Tensor:
        t = self.acquire(shape, dtype)
        st = self.layer_stats.setdefault(layer_id,
                                         {"count": 0, "bytes": 0})
        st["count"] += 1
        st["bytes"] += t.numel() * t.element_size()
        return t
