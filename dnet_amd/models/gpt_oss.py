"""gpt-oss ring model: MoE + alternating sliding/full attention + sinks.

Reference counterpart: src/dnet/core/models/gpt_oss.py (mlx_lm gpt_oss ring
wrapper with per-type masks and RotatingKVCache). MI355X version: the
sliding window and the learned attention-sink logit are handled inside the
decode-attention HIP kernel (window/sinks params); experts use the gpt-oss
clamped-GLU activation; router is softmax-over-top-k with bias. The
checkpoint's interleaved gate/up expert layout is de-interleaved at load to
the [gate; up] concat the kernels use. MXFP4 checkpoints are dequantized to
bf16/int8 at load (no MXFP4 runtime yet).
"""
from __future__ import annotations

import torch

from .base import LayerWeights, Linear, RingModel
from .config import ModelConfig
from .moe import MoERingModel


def gpt_oss_glu(gu: torch.Tensor, alpha: float = 1.702,
                limit: float = 7.0) -> torch.Tensor:
    i = gu.shape[-1] // 2
    gate = gu[..., :i].float().clamp(max=limit)
    up = gu[..., i:].float().clamp(min=-limit, max=limit)
    glu = gate * torch.sigmoid(gate * alpha)
    return ((up + 1.0) * glu).to(gu.dtype)


class GptOssRingModel(MoERingModel):
    model_type = "gpt_oss"
    model_types = ["gpt_oss"]

    def _attn_params(self, lid: int, lw=None):
        types = self.cfg.sliding_window_pattern
        if types is not None and lid < len(types):
            sliding = types[lid] == "sliding_attention"
        else:
            sliding = lid % 2 == 0  # gpt-oss default: even layers slide
        if lw is None:
            lw = self.layers.get(lid)
        sinks = getattr(lw, "sinks", None) if lw is not None else None
        return (self.cfg.sliding_window if sliding else 0), sinks

    def _init_layer(self, rand, lid: int) -> LayerWeights:
        lw = super()._init_layer(rand, lid)
        c = self.cfg
        dev = self.device
        # gpt-oss: biases everywhere + learned sink logit per q head.
        # TP: qkv bias and sinks slice with the heads; the o bias lives on
        # rank 0 only (the stage all-reduce sums partials — a copy per
        # rank would count it tp times)
        lw.qkv.bias = self._slice_qkv(
            rand(c.qkv_out).view(-1, 1)).view(-1).to(dev)
        ob = rand(c.hidden_size).to(dev)
        lw.o.bias = ob if self.tp_rank == 0 else None
        lw.router.bias = rand(c.num_experts).to(dev)
        nh = c.num_q_heads // self.tp_size
        lw.sinks = rand(c.num_q_heads)[self.tp_rank * nh:
                                       (self.tp_rank + 1) * nh].to(dev)
        lw.expert_biases_gu = [rand(2 * (c.moe_intermediate_size
                                         or c.intermediate_size)).to(dev)
                               for _ in range(c.num_experts)]
        lw.expert_biases_down = [rand(c.hidden_size).to(dev)
                                 for _ in range(c.num_experts)]
        for e in range(c.num_experts):
            lw.experts_gateup[e].bias = lw.expert_biases_gu[e]
            lw.experts_down[e].bias = lw.expert_biases_down[e]
        return lw

    def load_state_dict(self, sd: dict):
        """HF gpt-oss layout: batched expert tensors
        mlp.experts.gate_up_proj [E, H, 2I] (interleaved gate/up, input-major)
        + per-expert biases; self_attn.sinks [Hq]."""
        c = self.cfg

        def get(name):
            for pref in ("model.", ""):
                if pref + name in sd:
                    return sd[pref + name].to(torch.bfloat16)
            return None

        # MXFP4 checkpoints ship <name>_blocks (uint8 nibble pairs) +
        # <name>_scales (uint8 E8M0), output-major [E, rows, K/32, 16] /
        # [E, rows, K/32]. Default: keep them PACKED and execute natively
        # in the grouped MoE kernels (~4.25 bit/weight resident;
        # reference executes MXFP4 via MLX nn.quantize,
        # src/dnet/core/models/gpt_oss.py:216-287). An explicit int8/int4
        # quant request, or DNET_MXFP4_DEQUANT=1, dequantizes to bf16 at
        # load instead (round-1 behavior).
        import os
        mx_native = c.quant is None and not os.environ.get(
            "DNET_MXFP4_DEQUANT")

        def get_mx(name):
            for pref in ("model.", ""):
                b = sd.get(pref + name + "_blocks")
                s = sd.get(pref + name + "_scales")
                if b is not None and s is not None:
                    from .. import ops
                    if mx_native:
                        e, r, g, _ = b.shape
                        return (b.reshape(e, r, g * 16).contiguous(),
                                s.reshape(e, r, g).contiguous())
                    return ops.mxfp4_dequant(b, s)
            return None

        inter = c.moe_intermediate_size or c.intermediate_size
        for lid in self.layer_ids:
            p = f"layers.{lid}."
            qw, kw, vw = (get(p + f"self_attn.{x}_proj.weight") for x in "qkv")
            qb, kb, vb = (get(p + f"self_attn.{x}_proj.bias") for x in "qkv")
            bias = (self._slice_qkv(torch.cat([qb, kb, vb]).view(-1, 1))
                    .view(-1).to(self.device) if qb is not None else None)
            ob = get(p + "self_attn.o_proj.bias")
            if ob is not None and self.tp_rank != 0:
                ob = None          # bias counted once across the stage sum
            lw = LayerWeights(
                attn_norm=get(p + "input_layernorm.weight").to(self.device),
                qkv=Linear.make(self._slice_qkv(
                    torch.cat([qw, kw, vw])).to(self.device), bias, c.quant),
                o=Linear.make(self._slice_cols(
                    get(p + "self_attn.o_proj.weight")).to(self.device),
                    ob, c.quant),
                mlp_norm=get(p + "post_attention_layernorm.weight").to(self.device),
            )
            lw.router = Linear(get(p + "mlp.router.weight").to(self.device),
                               get(p + "mlp.router.bias"))
            gu = get(p + "mlp.experts.gate_up_proj")        # [E, H, 2I]
            gub = get(p + "mlp.experts.gate_up_proj_bias")  # [E, 2I]
            dn = get(p + "mlp.experts.down_proj")           # [E, I, H]
            dnb = get(p + "mlp.experts.down_proj_bias")     # [E, H]
            mx = gu is None
            gu_s = dn_s = None
            if mx:
                gu = get_mx(p + "mlp.experts.gate_up_proj")  # [E, 2I, H(/2)]
                dn = get_mx(p + "mlp.experts.down_proj")     # [E, H, I(/2)]
                if mx_native:
                    gu, gu_s = gu
                    dn, dn_s = dn
            lw.experts_gateup, lw.experts_down = [], []
            for e in range(c.num_experts):
                # bf16 layout is input-major (transpose); MXFP4 blocks are
                # already output-major. De-interleave rows: gate = even
                # rows, up = odd rows (nibble packing is along K, so row
                # gathers work unchanged on the packed layout).
                w_e = (gu[e] if mx else gu[e].t()).contiguous()  # interleaved
                w_e = torch.cat([w_e[0::2], w_e[1::2]])     # [2I, ...] concat
                b_e = torch.cat([gub[e][0::2], gub[e][1::2]]) if gub is not None else None
                b_e = b_e.to(self.device) if b_e is not None else None
                if gu_s is not None:
                    s_e = torch.cat([gu_s[e][0::2], gu_s[e][1::2]])
                    lw.experts_gateup.append(Linear.make_mxfp4(
                        w_e.to(self.device), s_e.contiguous().to(self.device),
                        b_e))
                    lw.experts_down.append(Linear.make_mxfp4(
                        dn[e].contiguous().to(self.device),
                        dn_s[e].contiguous().to(self.device),
                        dnb[e].to(self.device) if dnb is not None else None))
                    continue
                lw.experts_gateup.append(Linear.make(
                    w_e.to(self.device), b_e, c.quant))
                lw.experts_down.append(Linear.make(
                    (dn[e] if mx else dn[e].t()).contiguous().to(self.device),
                    dnb[e].to(self.device) if dnb is not None else None,
                    c.quant))
            sk = get(p + "self_attn.sinks")
            nh = c.num_q_heads // self.tp_size
            lw.sinks = sk[self.tp_rank * nh:
                          (self.tp_rank + 1) * nh].to(self.device)
            self.layers[lid] = lw
        if self.is_first:
            self.embed = get("embed_tokens.weight").to(self.device)
        if self.is_last:
            self.final_norm = get("norm.weight").to(self.device)
            head = sd.get("lm_head.weight")
            emb = get("embed_tokens.weight")
            self.lm_head = Linear((head.to(torch.bfloat16)
                                   if head is not None else emb).to(self.device))

    # dense/sparse expert routing is inherited from MoERingModel._mlp;
    # only the activation differs (clamped GLU, fused in the grouped kernel
    # via GLU=1)
    GLU = 1

    def _act(self, gu: torch.Tensor) -> torch.Tensor:
        return gpt_oss_glu(gu, self.GLU_ALPHA, self.GLU_LIMIT)
