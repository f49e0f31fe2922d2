"""Compute policies: fit / offload / sliding_fit residency planning.

Reference counterpart: src/dnet/shard/policies/ (plan_policy decides mode
from m = local layers, w = window, n = residency; FitInMemoryPolicy keeps
everything resident; OffloadPolicy streams windows with async prefetch of
the next window overlapped with compute). Here the streaming engine is the
pinned-host WeightCache (core/weight_cache.py); ``OffloadBinder`` is the
weight-provider hook installed on the ring model — on every bind it
prefetches the next layers of the ring order on the copy stream, which is
the compute/IO overlap that lets models exceed HBM.
"""
from __future__ import annotations

from typing import Sequence

import torch

from ..core.weight_cache import PinnedLayerStore, WeightCache
from ..models.base import LayerWeights, RingModel
from ..utils.logger import get_logger

log = get_logger("shard")


def plan_policy(m: int, w: int, n: int) -> str:
    """m=local layer count, w=window size, n=GPU-resident layers
    (reference: src/dnet/shard/policies/__init__.py plan_policy)."""
    if n >= m:
        return "fit"
    if n < w:
        return "sliding_fit"
    return "offload"


class OffloadBinder:
    """Weight provider over a WeightCache with ring-order prefetch."""

    def __init__(self, cache: WeightCache, order: Sequence[int], group: int,
                 packed: bool, prefetch_depth: int = 3):
        self.cache = cache
        self.order = list(order)
        self.group = group
        self.packed = packed
        self.depth = min(prefetch_depth, max(cache.residency - 1, 1))
        self._idx = {lid: i for i, lid in enumerate(self.order)}

    def __call__(self, lid: int) -> LayerWeights:
        tensors = self.cache.bind(lid)
        # overlap: schedule the next layers of the ring order on the copy
        # stream while this layer computes (wraps to the first window —
        # reference: offload.py:395-421)
        i = self._idx.get(lid, 0)
        for d in range(1, self.depth + 1):
            self.cache.prefetch(self.order[(i + d) % len(self.order)])
        return LayerWeights.from_tensor_dict(tensors, self.group, self.packed)


def enable_offload(model: RingModel, residency: int,
                   prefetch_depth: int = 3) -> WeightCache:
    """Move the model's layer weights to pinned host memory and install the
    windowed weight cache as the model's weight provider. Returns the cache
    (for stats). MoE expert banks stream in their stacked [E, ...] form;
    layers must share one tensor-shape template (true for uniform stacks —
    deepseek's mixed dense/MoE stack is not offloadable yet)."""
    assert model.layers, "load or init weights first"
    group = model.cfg.quant.group if model.cfg.quant else 0
    packed = False
    # group layers by tensor-shape template: the LARGEST uniform group
    # streams through the slot cache; the rest (e.g. deepseek's leading
    # dense layers) stay resident (partial offload)
    dicts = {lid: lw.to_tensor_dict()
             for lid, lw in sorted(model.layers.items())}
    groups: dict = {}
    for lid, td in dicts.items():
        key = tuple(sorted((k, tuple(t.shape)) for k, t in td.items()))
        groups.setdefault(key, []).append(lid)
    order = max(groups.values(), key=len)
    resident = [lid for lid in dicts if lid not in set(order)]
    if resident:
        log.info("partial offload: %d non-uniform layer(s) stay resident "
                 "(%s)", len(resident), resident)
    store = PinnedLayerStore(pin=True)
    for lid in order:
        lw = model.layers[lid]
        if lw.qkv is not None and lw.qkv.is_quant:
            packed = lw.qkv.packed
        store.put_layer(lid, dicts[lid])
    model.layers = {lid: model.layers[lid] for lid in resident}
    if model.device.type == "cuda":
        torch.cuda.empty_cache()
    cache = WeightCache(store, residency, model.device, order=order)
    model.weight_provider = OffloadBinder(cache, order, group, packed,
                                          prefetch_depth)
    # warm the first window
    for lid in order[:cache.residency]:
        cache.prefetch(lid)
    log.info("offload enabled: %d layers, residency %d (%s)", len(order),
             cache.residency, plan_policy(len(order), prefetch_depth + 1,
                                          residency))
    return cache
