"""Mixtral-style MoE ring model (top-k routed experts, SwiGLU experts).

Reference counterpart: src/dnet/core/models/gpt_oss.py (MoE layer pattern);
this is the dense-correct implementation — per-expert token gather + the
same fused SwiGLU/GEMV kernels. Expert-parallel execution across ranks is a
later milestone.
"""
from __future__ import annotations

import torch

from .. import ops
from .base import LayerWeights, Linear, RingModel
from .config import ModelConfig


class MoERingModel(RingModel):
    model_type = "mixtral"
    model_types = ["mixtral", "qwen2_moe", "qwen3_moe"]

    def _init_layer(self, rand, lid: int) -> LayerWeights:
        lw = super()._init_layer(rand, lid)
        c = self.cfg
        inter = c.moe_intermediate_size or c.intermediate_size
        dev = self.device
        lw.gateup = None
        lw.down = None
        lw.router = Linear(rand(c.num_experts, c.hidden_size).to(dev))
        lw.experts_gateup = [
            Linear.make(rand(2 * inter, c.hidden_size).to(dev), None, c.quant)
            for _ in range(c.num_experts)]
        lw.experts_down = [
            Linear.make(rand(c.hidden_size, inter).to(dev), None, c.quant)
            for _ in range(c.num_experts)]
        return lw

    def load_state_dict(self, sd: dict):
        raise NotImplementedError(
            "MoE safetensors loading lands with the weight-cache milestone; "
            "use init_random for synthetic runs")

    def _mlp(self, y: torch.Tensor, lw: LayerWeights) -> torch.Tensor:
        c = self.cfg
        T = y.shape[0]
        logits = lw.router(y).float()
        weights, idx = torch.topk(logits, c.num_experts_per_tok, dim=-1)
        weights = torch.softmax(weights, dim=-1)
        out = torch.zeros_like(y, dtype=torch.float32)
        for e in range(c.num_experts):
            if self.tp_size > 1 and e % self.tp_size != self.tp_rank:
                continue  # expert parallelism: partial sum reduced by caller
            mask = (idx == e).any(dim=-1)
            if not bool(mask.any()):
                continue
            rows = mask.nonzero(as_tuple=True)[0]
            xe = y[rows].contiguous()
            gu = lw.experts_gateup[e](xe)
            a = ops.swiglu(gu)
            d = lw.experts_down[e](a)
            we = (weights * (idx == e)).sum(dim=-1)[rows]
            out[rows] += d.float() * we.unsqueeze(-1)
        return out.to(y.dtype)
