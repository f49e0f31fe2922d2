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

    # below this many rows, run every expert on the whole batch and
    # weight-sum (no data-dependent host syncs -> hipGraph-capturable; the
    # expert weight read — the decode bound — is once per expert either way)
    DENSE_MOE_MAX_T = 64

    def _mlp(self, y: torch.Tensor, lw: LayerWeights) -> torch.Tensor:
        c = self.cfg
        T = y.shape[0]
        logits = lw.router(y).float()
        weights, idx = torch.topk(logits, c.num_experts_per_tok, dim=-1)
        weights = torch.softmax(weights, dim=-1)
        out = torch.zeros_like(y, dtype=torch.float32)
        dense = T <= self.DENSE_MOE_MAX_T
        for e in range(c.num_experts):
            if self.tp_size > 1 and e % self.tp_size != self.tp_rank:
                continue  # expert parallelism: partial sum reduced by caller
            we_full = (weights * (idx == e)).sum(dim=-1)
            if dense:
                d = lw.experts_down[e](ops.swiglu(lw.experts_gateup[e](y)))
                out += d.float() * we_full.unsqueeze(-1)
                continue
            mask = we_full > 0
            if not bool(mask.any()):
                continue
            rows = mask.nonzero(as_tuple=True)[0]
            xe = y[rows].contiguous()
            d = lw.experts_down[e](ops.swiglu(lw.experts_gateup[e](xe)))
            out[rows] += d.float() * we_full[rows].unsqueeze(-1)
        return out.to(y.dtype)
