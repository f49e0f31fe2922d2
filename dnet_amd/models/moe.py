"""Mixtral-style MoE ring model (top-k routed experts, SwiGLU experts).

Reference counterpart: src/dnet/core/models/gpt_oss.py (MoE layer pattern);
this is the dense-correct implementation — per-expert token gather + the
same fused SwiGLU/GEMV kernels. Expert-parallel execution across ranks is a
later milestone.
"""
from __future__ import annotations

import os

import torch

from .. import ops
from .base import LayerWeights, Linear, RingModel
from .config import ModelConfig


def stack_experts(lw: LayerWeights, local: list) -> dict:
    """Stack the given experts' weights [E_local, ...] for the grouped MoE
    kernels; the per-expert Linears become views of the stack (the sparse
    large-T path keeps working on the same storage, nothing is duplicated
    after the originals are freed).

    The returned dict's "local"/"local_list" name the experts actually in
    the stack IN STACK ORDER — possibly a superset of the request (the
    offload path restores the FULL bank per slot); callers align router
    weights with stack_route_weights, which zeroes non-owned experts under
    EP. A cached stack that does not COVER the request (advisor r1: a
    TP-subset cache vs a full-bank serialization, or vice versa) is
    rebuilt from the per-expert Linears instead of being returned as-is —
    returning the wrong subset silently double-counted experts under
    TP+offload."""
    local = list(local)
    st = getattr(lw, "experts_stacked", None)
    if st is not None:
        cached = st.get("local_list")
        if cached is None:
            cached = st["local"].tolist()
            st["local_list"] = cached
        if set(local).issubset(cached):
            return st
        # fall through: rebuild a stack covering the request
    gls = [lw.experts_gateup[e] for e in local]
    dls = [lw.experts_down[e] for e in local]

    def stack(attr, ls):
        ts = [getattr(l, attr) for l in ls]
        return torch.stack(ts).contiguous() if ts[0] is not None else None

    gw, gs, gb = stack("w", gls), stack("scales", gls), stack("bias", gls)
    dw, ds, db = stack("w", dls), stack("scales", dls), stack("bias", dls)
    for j, e in enumerate(local):
        lw.experts_gateup[e].w = gw[j]
        lw.experts_down[e].w = dw[j]
        for lin, s, b in ((lw.experts_gateup[e], gs, gb),
                          (lw.experts_down[e], ds, db)):
            if s is not None:
                lin.scales = s[j]
            if b is not None:
                lin.bias = b[j]
    st = {"local": torch.tensor(local, dtype=torch.long, device=gw.device),
          "local_list": local,
          "gw": gw, "gs": gs, "gb": gb, "dw": dw, "ds": ds, "db": db,
          "group": gls[0].group, "packed": gls[0].packed}
    lw.experts_stacked = st
    return st


def stack_route_weights(st: dict, we: torch.Tensor, tp_size: int,
                        tp_rank: int) -> torch.Tensor:
    """Align dense router weights [T, E_total] with the stack's expert
    rows and zero experts this rank does not own (EP rule: e % tp == rank).
    Device-only ops — graph-safe. The zeroed columns make the grouped
    kernels' per-block early exit skip non-owned experts, so a full-bank
    stack under EP reads only the owned experts' weights."""
    if len(st["local_list"]) != we.shape[1]:
        we = we.index_select(1, st["local"]).contiguous()
    if tp_size > 1:
        key = ("own", tp_size, tp_rank)
        own = st.get(key)
        if own is None:
            own = ((st["local"] % tp_size) == tp_rank).to(we.dtype)
            st[key] = own
        we = we * own
    return we


class MoERingModel(RingModel):
    model_type = "mixtral"
    model_types = ["mixtral", "qwen2_moe", "qwen3_moe"]

    def _init_layer(self, rand, lid: int) -> LayerWeights:
        # mxfp4 applies to the EXPERT banks only (gpt-oss checkpoint
        # shape); attention/router stay bf16 — so the base init runs
        # unquantized in that mode
        c = self.cfg
        mx = c.quant is not None and c.quant.fmt == "mxfp4"
        if mx:
            import copy
            base_self = copy.copy(self)
            base_self.cfg = copy.copy(c)
            base_self.cfg.quant = None
            lw = RingModel._init_layer(base_self, rand, lid)
        else:
            lw = super()._init_layer(rand, lid)
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
        if c.shared_expert_intermediate_size:       # qwen2-moe
            si = c.shared_expert_intermediate_size
            lw.shared_gateup = Linear.make(self._slice_gateup(
                rand(2 * si, c.hidden_size)).to(dev), None, c.quant)
            lw.shared_down = Linear.make(self._slice_cols(
                rand(c.hidden_size, si)).to(dev), None, c.quant)
            lw.shared_gate = Linear(rand(1, c.hidden_size).to(dev))
        return lw

    def load_state_dict(self, sd: dict):
        """HF mixtral layout (`block_sparse_moe.gate` +
        `experts.E.w1/w3/w2` = gate/up/down) and qwen-moe layout
        (`mlp.gate` + `mlp.experts.E.{gate,up,down}_proj`); attention and
        norms follow the llama-family naming (reference:
        src/dnet/core/models/gpt_oss.py MoE weight mapping)."""
        c = self.cfg

        def get(name):
            for pref in ("model.", ""):
                if pref + name in sd:
                    return sd[pref + name].to(torch.bfloat16)
            return None

        dev = self.device
        for lid in self.layer_ids:
            p = f"layers.{lid}."
            qw, kw, vw = (get(p + f"self_attn.{x}_proj.weight") for x in "qkv")
            qb = get(p + "self_attn.q_proj.bias")
            bias = None
            if qb is not None:
                bias = torch.cat([qb, get(p + "self_attn.k_proj.bias"),
                                  get(p + "self_attn.v_proj.bias")]).to(dev)
            lw = LayerWeights(
                attn_norm=get(p + "input_layernorm.weight").to(dev),
                qkv=Linear.make(torch.cat([qw, kw, vw]).to(dev), bias,
                                c.quant),
                o=Linear.make(get(p + "self_attn.o_proj.weight").to(dev),
                              None, c.quant),
                mlp_norm=get(p + "post_attention_layernorm.weight").to(dev),
            )
            qn = get(p + "self_attn.q_norm.weight")
            if qn is not None:
                lw.q_norm = qn.to(dev)
                lw.k_norm = get(p + "self_attn.k_norm.weight").to(dev)
            router = get(p + "block_sparse_moe.gate.weight")
            if router is None:
                router = get(p + "mlp.gate.weight")
            lw.router = Linear(router.to(dev))
            lw.gateup = None
            lw.down = None
            lw.experts_gateup, lw.experts_down = [], []
            gub = get(p + "mlp.experts.gate_up_proj")   # batched [E, 2I, H]
            dnb = get(p + "mlp.experts.down_proj")      # batched [E, H, I]
            for e in range(c.num_experts):
                if gub is not None:     # batched layout (gate/up halves)
                    gu_e, dn_e = gub[e].contiguous(), dnb[e].contiguous()
                else:                   # per-expert tensors
                    mx = f"{p}block_sparse_moe.experts.{e}."
                    qw2 = f"{p}mlp.experts.{e}."
                    gate = get(mx + "w1.weight")
                    if gate is not None:               # legacy mixtral
                        up = get(mx + "w3.weight")
                        dn_e = get(mx + "w2.weight")
                    else:                              # qwen-moe naming
                        gate = get(qw2 + "gate_proj.weight")
                        up = get(qw2 + "up_proj.weight")
                        dn_e = get(qw2 + "down_proj.weight")
                    gu_e = torch.cat([gate, up])
                lw.experts_gateup.append(Linear.make(gu_e.to(dev), None,
                                                     c.quant))
                lw.experts_down.append(Linear.make(dn_e.to(dev), None,
                                                   c.quant))
            sg = get(p + "mlp.shared_expert.gate_proj.weight")
            if sg is not None:                      # qwen2-moe shared expert
                lw.shared_gateup = Linear.make(self._slice_gateup(
                    torch.cat([sg, get(p + "mlp.shared_expert.up_proj.weight")]
                              )).to(dev), None, c.quant)
                lw.shared_down = Linear.make(self._slice_cols(
                    get(p + "mlp.shared_expert.down_proj.weight")).to(dev),
                    None, c.quant)
                lw.shared_gate = Linear(
                    get(p + "mlp.shared_expert_gate.weight").to(dev))
            self.layers[lid] = lw
        if self.is_first:
            self.embed = get("embed_tokens.weight").to(dev)
        if self.is_last:
            self.final_norm = get("norm.weight").to(dev)
            head = sd.get("lm_head.weight")
            if head is None or c.tie_word_embeddings:
                self.lm_head = Linear(get("embed_tokens.weight").to(dev))
            else:
                self.lm_head = Linear(head.to(torch.bfloat16).to(dev))

    # below this many rows, route through the grouped-expert kernels (no
    # data-dependent host syncs -> hipGraph-capturable; unrouted experts are
    # skipped per-block on device, so a single stream reads only the top-k
    # experts' weights)
    DENSE_MOE_MAX_T = int(os.environ.get("DNET_DENSE_MOE_T", "64"))
    GLU = 0                  # 0 = SwiGLU, 1 = gpt-oss clamped GLU
    GLU_ALPHA = 1.702
    GLU_LIMIT = 7.0

    def _act(self, gu: torch.Tensor) -> torch.Tensor:
        return ops.swiglu(gu)

    def _expert_stack(self, lw: LayerWeights) -> dict:
        local = [e for e in range(self.cfg.num_experts)
                 if self.tp_size <= 1 or e % self.tp_size == self.tp_rank]
        return stack_experts(lw, local)

    def _route(self, logits: torch.Tensor):
        """(weights, idx). mixtral: softmax over the top-k logits;
        qwen-moe: softmax over ALL experts -> top-k (-> renorm when
        norm_topk_prob)."""
        c = self.cfg
        if c.model_type in ("qwen2_moe", "qwen3_moe"):
            probs = torch.softmax(logits, dim=-1)
            weights, idx = torch.topk(probs, c.num_experts_per_tok, dim=-1)
            if c.norm_topk_prob:
                weights = weights / weights.sum(-1, keepdim=True)
            return weights, idx
        weights, idx = torch.topk(logits, c.num_experts_per_tok, dim=-1)
        return torch.softmax(weights, dim=-1), idx

    def _shared_expert(self, y: torch.Tensor, lw: LayerWeights):
        """qwen2-moe always-on shared expert, sigmoid-gated; returns a
        PARTIAL under TP (rows/cols sliced; the caller's all-reduce sums).
        """
        sh = lw.shared_down(ops.swiglu(lw.shared_gateup(y)))
        gate = torch.sigmoid(lw.shared_gate(y).float())
        return sh.float() * gate

    def _mlp(self, y: torch.Tensor, lw: LayerWeights) -> torch.Tensor:
        c = self.cfg
        T = y.shape[0]
        logits = lw.router(y).float()
        weights, idx = self._route(logits)
        eg0 = lw.experts_gateup[0]
        if (T <= self.DENSE_MOE_MAX_T
                and (eg0.bits in (8, 16) or eg0.mxfp4)):
            st = self._expert_stack(lw)
            we = torch.zeros(T, c.num_experts, dtype=torch.float32,
                             device=y.device)
            we.scatter_(1, idx, weights)
            we = stack_route_weights(st, we, self.tp_size, self.tp_rank)
            act = ops.moe_gateup(y, st["gw"], st["gs"], st["gb"], we,
                                 st["group"], st["packed"], self.GLU,
                                 self.GLU_ALPHA, self.GLU_LIMIT)
            out = ops.moe_down(act, st["dw"], st["ds"], st["db"], we,
                               st["group"], st["packed"])
            if getattr(lw, "shared_gateup", None) is not None:
                out = out + self._shared_expert(y, lw)
            return out.to(y.dtype)
        out = torch.zeros_like(y, dtype=torch.float32)
        for e in range(c.num_experts):
            if self.tp_size > 1 and e % self.tp_size != self.tp_rank:
                continue  # expert parallelism: partial sum reduced by caller
            we_full = (weights * (idx == e)).sum(dim=-1)
            mask = we_full > 0
            if not bool(mask.any()):
                continue
            rows = mask.nonzero(as_tuple=True)[0]
            xe = y[rows].contiguous()
            d = lw.experts_down[e](self._act(lw.experts_gateup[e](xe)))
            out[rows] += d.float() * we_full[rows].unsqueeze(-1)
        if getattr(lw, "shared_gateup", None) is not None:
            out = out + self._shared_expert(y, lw)
        return out.to(y.dtype)
