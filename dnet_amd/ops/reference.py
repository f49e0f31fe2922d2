"""Plain-PyTorch fp32 reference implementations of every dnet_amd HIP kernel.

These serve two purposes:
  * golden references for the GPU numerics tests (kernel vs fp32 eager), and
  * the CPU execution path, so the whole engine (ring executor, policies,
    API server) runs and is testable on machines with no GPU.

They intentionally mirror the kernel contracts exactly (shapes, in-place
semantics, bf16 rounding points).
"""
from __future__ import annotations

import torch


def rmsnorm(x: torch.Tensor, residual: torch.Tensor | None, w: torch.Tensor,
            eps: float) -> torch.Tensor:
    """y = rms_norm(x or x+residual) * w; residual (if given) updated in place."""
    if residual is not None:
        h = (x.float() + residual.float()).to(x.dtype)
        residual.copy_(h)
        src = h.float()
    else:
        src = x.float()
    scale = torch.rsqrt(src.pow(2).mean(-1, keepdim=True) + eps)
    return (src * scale * w.float()).to(x.dtype)


def gemv_bf16(x: torch.Tensor, w: torch.Tensor,
              bias: torch.Tensor | None = None) -> torch.Tensor:
    y = x.float() @ w.float().t()
    if bias is not None:
        y = y + bias.float()
    return y.to(x.dtype)


def dequant_int8(w: torch.Tensor, scales: torch.Tensor, group: int) -> torch.Tensor:
    n, k = w.shape
    wf = w.float().view(n, k // group, group)
    return (wf * scales.float().unsqueeze(-1)).view(n, k).to(torch.bfloat16)


def gemv_int8(x: torch.Tensor, w: torch.Tensor, scales: torch.Tensor,
              group: int, bias: torch.Tensor | None = None) -> torch.Tensor:
    wd = dequant_int8(w, scales, group)
    y = x.float() @ wd.float().t()
    if bias is not None:
        y = y + bias.float()
    return y.to(x.dtype)


def pack_int8_mfma(q: torch.Tensor) -> torch.Tensor:
    """Permute each row's K dim into MFMA chunk-pair order so a lane's
    16 B load covers its 8-elem B-fragment slices of two adjacent K=32
    chunks (full 64 B HBM bursts per 16-row group instead of 32 B halves).

    orig [N, cp, half, slice, j] -> packed [N, cp, slice, half, j]
    (cp = 64-k pair, half = which chunk, slice = lane>>4, j = 0..7).
    """
    n, k = q.shape
    assert k % 64 == 0
    v = q.view(n, k // 64, 2, 4, 8)
    return v.permute(0, 1, 3, 2, 4).reshape(n, k).contiguous()


def unpack_int8_mfma(qp: torch.Tensor) -> torch.Tensor:
    n, k = qp.shape
    v = qp.view(n, k // 64, 4, 2, 8)
    return v.permute(0, 1, 3, 2, 4).reshape(n, k).contiguous()


def quantize_int8(w: torch.Tensor, group: int) -> tuple[torch.Tensor, torch.Tensor]:
    """Symmetric per-group int8 quantization along the K (last) dim."""
    n, k = w.shape
    assert k % group == 0
    wf = w.float().view(n, k // group, group)
    amax = wf.abs().amax(dim=-1).clamp_min(1e-8)
    scales = (amax / 127.0).to(torch.bfloat16)
    q = torch.round(wf / scales.float().unsqueeze(-1)).clamp(-127, 127).to(torch.int8)
    return q.view(n, k), scales


def quantize_kv_rows(x: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
    """int8 group-64 KV quantization over the last dim (matches the
    rope_append kernel: bf16-rounded scale = amax/127)."""
    *lead, d = x.shape
    ng = d // 64
    xf = x.float().view(*lead, ng, 64)
    amax = xf.abs().amax(dim=-1).clamp_min(1e-8)
    scales = (amax / 127.0).to(torch.bfloat16)
    codes = torch.round(xf / scales.float().unsqueeze(-1)).clamp(-127, 127)
    return codes.to(torch.int8).view(*lead, d), scales.view(*lead, ng)


def dequant_kv(codes: torch.Tensor, scales: torch.Tensor) -> torch.Tensor:
    *lead, d = codes.shape
    ng = d // 64
    xf = codes.float().view(*lead, ng, 64) * scales.float().unsqueeze(-1)
    return xf.view(*lead, d).to(torch.bfloat16)


def quantize_int4(w: torch.Tensor, group: int = 128
                  ) -> tuple[torch.Tensor, torch.Tensor]:
    """Symmetric per-group int4 quantization along K: values in [-7, 7]
    stored offset-by-8 as nibbles, two per byte (even k = low nibble)."""
    n, k = w.shape
    assert k % group == 0 and k % 2 == 0
    wf = w.float().view(n, k // group, group)
    amax = wf.abs().amax(dim=-1).clamp_min(1e-8)
    scales = (amax / 7.0).to(torch.bfloat16)
    q = torch.round(wf / scales.float().unsqueeze(-1)).clamp(-7, 7)
    q = (q + 8).to(torch.uint8).view(n, k)
    packed = (q[:, 0::2] | (q[:, 1::2] << 4)).contiguous()
    return packed, scales


def unpack_int4(q4: torch.Tensor) -> torch.Tensor:
    """[N, K/2] packed nibbles -> [N, K] int8 in [-7, 7] (device-preserving)."""
    n, kb = q4.shape
    lo = (q4 & 0xF).to(torch.int8) - 8
    hi = (q4 >> 4).to(torch.int8) - 8
    out = torch.empty(n, kb * 2, dtype=torch.int8, device=q4.device)
    out[:, 0::2] = lo
    out[:, 1::2] = hi
    return out


def dequant_int4(q4: torch.Tensor, scales: torch.Tensor, group: int
                 ) -> torch.Tensor:
    return dequant_int8(unpack_int4(q4), scales.to(q4.device), group)


def gemv_int4(x: torch.Tensor, q4: torch.Tensor, scales: torch.Tensor,
              group: int, bias: torch.Tensor | None = None) -> torch.Tensor:
    wd = dequant_int4(q4, scales, group)
    y = x.float() @ wd.float().t()
    if bias is not None:
        y = y + bias.float()
    return y.to(x.dtype)


# ---- MXFP4 (OCP microscaling fp4): e2m1 nibbles + one e8m0 (power-of-2
# uint8, bias 127) scale per 32-element block. The gpt-oss checkpoint
# format (reference executes it via MLX nn.quantize,
# src/dnet/core/models/gpt_oss.py:216-287); here it executes natively in
# the grouped MoE kernels at ~4.25 bits/weight resident.

_E2M1 = torch.tensor([0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0])


def quantize_mxfp4(w: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
    """[N, K] -> (packed nibbles [N, K/2] uint8 (even k = low nibble),
    e8m0 scales [N, K/32] uint8). Round-to-nearest onto the e2m1 grid."""
    n, k = w.shape
    assert k % 32 == 0
    wf = w.float().view(n, k // 32, 32)
    amax = wf.abs().amax(dim=-1).clamp_min(1e-30)
    # scale = 2^e with amax/2^e <= 6 (largest e2m1 magnitude)
    e = torch.ceil(torch.log2(amax / 6.0)).clamp(-127, 127)
    scales = (e + 127).to(torch.uint8)
    v = wf / torch.pow(2.0, e).unsqueeze(-1)
    mag = v.abs().unsqueeze(-1)                     # [n, nb, 32, 1]
    idx = (mag - _E2M1.to(w.device).view(1, 1, 1, 8)).abs().argmin(-1)
    code = (idx + torch.where(v < 0, 8, 0)).to(torch.uint8).view(n, k)
    packed = (code[:, 0::2] | (code[:, 1::2] << 4)).contiguous()
    return packed, scales.view(n, k // 32).contiguous()


def dequant_mxfp4(packed: torch.Tensor, scales: torch.Tensor) -> torch.Tensor:
    """(nibbles [N, K/2], e8m0 [N, K/32]) -> bf16 [N, K]."""
    n, kb = packed.shape
    k = kb * 2
    code = torch.empty(n, k, dtype=torch.uint8, device=packed.device)
    code[:, 0::2] = packed & 0xF
    code[:, 1::2] = packed >> 4
    lut = _E2M1.to(packed.device)
    val = lut[(code & 7).long()] * torch.where(code >= 8, -1.0, 1.0)
    sc = torch.pow(2.0, scales.float() - 127.0)
    out = val.view(n, k // 32, 32) * sc.unsqueeze(-1)
    return out.view(n, k).to(torch.bfloat16)


def gemv_mxfp4(x: torch.Tensor, packed: torch.Tensor, scales: torch.Tensor,
               bias: torch.Tensor | None = None) -> torch.Tensor:
    wd = dequant_mxfp4(packed, scales)
    y = x.float() @ wd.float().t()
    if bias is not None:
        y = y + bias.float()
    return y.to(x.dtype)


def pack_int4_mfma(q4: torch.Tensor) -> torch.Tensor:
    """Permute packed-nibble rows into MFMA chunk-quad order: a lane's
    16 B load covers its 8-elem B slices of FOUR adjacent K=32 chunks.

    unpacked orig [N, quad, chunk(4), slice(4), j(8)] ->
    packed nibble bytes [N, quad, slice, chunk, j/2]."""
    n, kb = q4.shape
    k = kb * 2
    assert k % 128 == 0
    vals = unpack_int4(q4) + 8               # [N, K] in [1,15]
    v = vals.view(n, k // 128, 4, 4, 8)      # [N, quad, chunk, slice, j]
    v = v.permute(0, 1, 3, 2, 4).reshape(n, k)  # slice-major
    packed = (v[:, 0::2] | (v[:, 1::2] << 4)).to(torch.uint8)
    return packed.contiguous()


def attn_decode(q: torch.Tensor, kcache: torch.Tensor, vcache: torch.Tensor,
                pos: torch.Tensor, scale: float, window: int = 0,
                sinks: torch.Tensor | None = None,
                kscale: torch.Tensor | None = None,
                vscale: torch.Tensor | None = None) -> torch.Tensor:
    """out[b,h] = softmax(q . K^T * scale) @ V over positions
    [max(0, len-window), len) (window=0 -> full)."""
    B, Hq, D = q.shape
    Hkv = kcache.shape[1]
    Dv = vcache.shape[-1]
    G = Hq // Hkv
    out = torch.zeros(B, Hq, Dv, dtype=q.dtype, device=q.device)
    for b in range(B):
        ln = int(pos[b])
        if ln == 0:
            continue
        s0 = max(0, ln - window) if window > 0 else 0
        if kcache.dtype == torch.int8:
            k = dequant_kv(kcache[b, :, s0:ln], kscale[b, :, s0:ln]).float()
            v = dequant_kv(vcache[b, :, s0:ln], vscale[b, :, s0:ln]).float()
        else:
            k = kcache[b, :, s0:ln].float()          # [Hkv, ln-s0, D]
            v = vcache[b, :, s0:ln].float()
        qq = q[b].float().view(Hkv, G, D)      # [Hkv, G, D]
        s = torch.einsum("hgd,hld->hgl", qq, k) * scale
        if sinks is not None:
            sk = sinks.float().view(Hkv, G, 1)
            p = torch.softmax(torch.cat([s, sk], dim=-1), dim=-1)[..., :-1]
        else:
            p = torch.softmax(s, dim=-1)
        o = torch.einsum("hgl,hld->hgd", p, v)
        out[b] = o.reshape(Hq, Dv).to(q.dtype)
    return out


def rope_tables(smax: int, d: int, theta: float,
                device=None, scaling: dict | None = None) -> tuple[torch.Tensor, torch.Tensor]:
    """Precompute [Smax, D/2] f32 cos/sin tables (neox convention).

    ``scaling`` supports the llama3 rope-scaling dict
    (factor / low_freq_factor / high_freq_factor / original_max_position_embeddings).
    """
    half = d // 2
    inv = 1.0 / (theta ** (torch.arange(0, half, dtype=torch.float32) / half))
    attn_factor = 1.0
    if scaling and scaling.get("rope_type", scaling.get("type")) == "yarn":
        # YaRN (gpt-oss / deepseek style): NTK-by-parts interpolation +
        # attention temperature on cos/sin.
        import math
        factor = scaling.get("factor", 1.0)
        orig = scaling.get("original_max_position_embeddings", 4096)
        beta_fast = scaling.get("beta_fast", 32.0)
        beta_slow = scaling.get("beta_slow", 1.0)

        def corr_dim(n_rot):
            return (d * math.log(orig / (n_rot * 2 * math.pi))
                    / (2 * math.log(theta)))

        low = max(math.floor(corr_dim(beta_fast)), 0)
        high = min(math.ceil(corr_dim(beta_slow)), half - 1)
        rng = torch.arange(half, dtype=torch.float32)
        ramp = ((rng - low) / max(high - low, 1e-3)).clamp(0, 1)
        mask = 1 - ramp  # 1 = no interpolation (high freq), 0 = full
        inv = inv / factor * (1 - mask) + inv * mask
        attn_factor = scaling.get("attention_factor") or \
            (0.1 * math.log(factor) + 1.0 if factor > 1 else 1.0)
        mscale = scaling.get("mscale")
        if mscale is not None:
            # deepseek variant: factor from mscale/mscale_all_dim
            def ds_mscale(scale, m):
                return 1.0 if scale <= 1 else 0.1 * m * math.log(scale) + 1.0
            attn_factor = (ds_mscale(factor, mscale)
                           / ds_mscale(factor, scaling.get("mscale_all_dim", 0)))
    if scaling and scaling.get("rope_type", scaling.get("type")) == "llama3":
        factor = scaling["factor"]
        lo = scaling.get("low_freq_factor", 1.0)
        hi = scaling.get("high_freq_factor", 4.0)
        orig = scaling.get("original_max_position_embeddings", 8192)
        wavelen = 2 * torch.pi / inv
        low_wl = orig / lo
        high_wl = orig / hi
        smooth = (orig / wavelen - lo) / (hi - lo)
        scaled = torch.where(wavelen > low_wl, inv / factor, inv)
        mid = (1 - smooth) * inv / factor + smooth * inv
        use_mid = (wavelen <= low_wl) & (wavelen >= high_wl)
        inv = torch.where(use_mid, mid, scaled)
    t = torch.arange(smax, dtype=torch.float32)
    freqs = torch.outer(t, inv)
    return (freqs.cos().mul_(attn_factor).to(device),
            freqs.sin().mul_(attn_factor).to(device))


def rope_apply(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
               positions: torch.Tensor) -> torch.Tensor:
    """Neox rotate-half rope. x: [..., T, H, D] or [B, H, D] with positions [T]/[B]."""
    half = x.shape[-1] // 2
    c = cos[positions].float()  # [P, half]
    s = sin[positions].float()
    # positions index the dim at x.dim()-3 (T for [B,T,H,D], B for [B,H,D])
    shape = [1] * (x.dim() - 3) + [c.shape[0], 1, half]
    c = c.view(shape)
    s = s.view(shape)
    x1 = x[..., :half].float()
    x2 = x[..., half:].float()
    return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1).to(x.dtype)


def rope_append(q, k, v, kcache, vcache, pos, cos, sin,
                kscale=None, vscale=None, wpos=None):
    """Decode-step fused rope+append reference (in-place on q, k and caches).
    ``wpos`` (context parallelism): local cache write positions; rows with
    wpos outside [0, Smax) are rotated but not stored."""
    B = q.shape[0]
    positions = pos.long()
    q.copy_(rope_apply(q, cos, sin, positions))
    k.copy_(rope_apply(k, cos, sin, positions))
    q8 = kcache.dtype == torch.int8
    smax = kcache.shape[2]
    for b in range(B):
        p = int(pos[b]) if wpos is None else int(wpos[b])
        if p < 0 or p >= smax:
            continue
        if q8:
            kc, ks = quantize_kv_rows(k[b])
            vc, vs = quantize_kv_rows(v[b])
            kcache[b, :, p] = kc
            kscale[b, :, p] = ks
            vcache[b, :, p] = vc
            vscale[b, :, p] = vs
        else:
            kcache[b, :, p] = k[b]
            vcache[b, :, p] = v[b]


def swiglu(gu: torch.Tensor) -> torch.Tensor:
    i = gu.shape[-1] // 2
    g = gu[..., :i].float()
    u = gu[..., i:].float()
    return (torch.nn.functional.silu(g) * u).to(gu.dtype)


def _moe_dense_w(w: torch.Tensor, scales, group: int, packed: bool
                 ) -> torch.Tensor:
    """Stacked expert weights [E, N, K(/2)] (bf16, grouped-int8, or
    mxfp4 nibble rows) -> f32."""
    if w.dtype == torch.uint8:     # mxfp4: [E, N, K/2] + e8m0 [E, N, K/32]
        e, n, kb = w.shape
        return dequant_mxfp4(w.reshape(e * n, kb),
                             scales.reshape(e * n, -1)).float().view(
                                 e, n, kb * 2)
    if w.dtype != torch.int8:
        return w.float()
    e, n, k = w.shape
    q = w.reshape(e * n, k)
    if packed:
        q = unpack_int8_mfma(q)
    return dequant_int8(q, scales.reshape(e * n, -1), group).float().view(
        e, n, k)


def moe_gateup(x: torch.Tensor, w: torch.Tensor, scales, bias,
               we: torch.Tensor, group: int = 0, packed: bool = False,
               glu: int = 0, alpha: float = 1.702, limit: float = 7.0
               ) -> torch.Tensor:
    """x [M,K], w [E,2I,K], we [M,E] -> act [E,M,I] (glu 0=SwiGLU,
    1=gpt-oss clamped GLU). Experts with all-zero we still computed here
    (the kernel skips them; their act rows are never read by moe_down)."""
    wf = _moe_dense_w(w, scales, group, packed)
    gu = torch.einsum("mk,enk->emn", x.float(), wf)
    if bias is not None:
        gu = gu + bias.float().unsqueeze(1)
    i = gu.shape[-1] // 2
    g, u = gu[..., :i], gu[..., i:]
    if glu == 1:
        g = g.clamp(max=limit)
        u = u.clamp(min=-limit, max=limit)
        act = (u + 1.0) * g * torch.sigmoid(g * alpha)
    else:
        act = torch.nn.functional.silu(g) * u
    return act.to(torch.bfloat16)


def moe_down(act: torch.Tensor, w: torch.Tensor, scales, bias,
             we: torch.Tensor, group: int = 0, packed: bool = False
             ) -> torch.Tensor:
    """act [E,M,I], w [E,H,I], we [M,E] -> out [M,H] f32 (weighted sum of
    per-expert down projections; bias applied inside the weighting)."""
    wf = _moe_dense_w(w, scales, group, packed)
    d = torch.einsum("emi,ehi->emh", act.float(), wf)
    if bias is not None:
        d = d + bias.float().unsqueeze(1)
    return (we.t().unsqueeze(-1) * d).sum(dim=0)


_MXFP4_LUT = [0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0,
              -0.0, -0.5, -1.0, -1.5, -2.0, -3.0, -4.0, -6.0]


def mxfp4_dequant(blocks: torch.Tensor, scales: torch.Tensor) -> torch.Tensor:
    """OCP MXFP4 -> bf16 (gpt-oss checkpoint format; reference counterpart:
    src/dnet/core/models/gpt_oss.py MXFP4 weight sanitization).

    blocks: uint8 [..., B, 16] — two E2M1 nibbles per byte, low nibble
    first, 32 values per block. scales: uint8 [..., B] — E8M0 power-of-two
    exponents, bias 127. Returns bf16 [..., B*32].
    """
    lut = torch.tensor(_MXFP4_LUT, dtype=torch.float32, device=blocks.device)
    lo = lut[(blocks & 0xF).long()]
    hi = lut[(blocks >> 4).long()]
    vals = torch.stack([lo, hi], dim=-1).reshape(*blocks.shape[:-1],
                                                 blocks.shape[-1] * 2)
    ex = torch.pow(2.0, scales.float() - 127.0)
    out = vals * ex.unsqueeze(-1)
    return out.reshape(*blocks.shape[:-2],
                       blocks.shape[-2] * blocks.shape[-1] * 2).to(
                           torch.bfloat16)


def attn_decode_partials(q, kcache, vcache, pos, scale,
                         kscale=None, vscale=None) -> torch.Tensor:
    """Unnormalized flash-decode partials over this cache shard:
    [B, Hq, 1, Dv+2] f32 with acc (exp(s-m) @ V), m, l — the cross-rank /
    cross-split combinable form (matches the kernel's split-S partials)."""
    B, Hq, D = q.shape
    Hkv = kcache.shape[1]
    Dv = vcache.shape[-1]
    G = Hq // Hkv
    out = torch.zeros(B, Hq, 1, Dv + 2, dtype=torch.float32)
    out[..., Dv] = -1e30
    for b in range(B):
        ln = int(pos[b])
        if ln == 0:
            continue
        if kcache.dtype == torch.int8:
            k = dequant_kv(kcache[b, :, :ln], kscale[b, :, :ln]).float()
            v = dequant_kv(vcache[b, :, :ln], vscale[b, :, :ln]).float()
        else:
            k = kcache[b, :, :ln].float()
            v = vcache[b, :, :ln].float()
        qq = q[b].float().view(Hkv, G, D)
        s = torch.einsum("hgd,hld->hgl", qq, k) * scale
        m = s.amax(dim=-1, keepdim=True)
        p = torch.exp(s - m)
        acc = torch.einsum("hgl,hld->hgd", p, v)
        out[b, :, 0, :Dv] = acc.reshape(Hq, Dv)
        out[b, :, 0, Dv] = m.reshape(Hq)
        out[b, :, 0, Dv + 1] = p.sum(-1).reshape(Hq)
    return out


def attn_combine(partials: torch.Tensor,
                 sinks: torch.Tensor | None = None) -> torch.Tensor:
    """partials [B, Hq, S, Dv+2] -> out [B, Hq, Dv] bf16 (merge shards)."""
    B, Hq, S, W = partials.shape
    Dv = W - 2
    acc, m, l = partials[..., :Dv], partials[..., Dv], partials[..., Dv + 1]
    M = m.amax(dim=-1)
    if sinks is not None:
        M = torch.maximum(M, sinks.float().view(1, Hq))
    w = torch.exp(m - M.unsqueeze(-1))
    denom = (l * w).sum(-1)
    if sinks is not None:
        denom = denom + torch.exp(sinks.float().view(1, Hq) - M)
    num = (acc * w.unsqueeze(-1)).sum(2)
    denom = denom.clamp_min(1e-30)
    return (num / denom.unsqueeze(-1)).to(torch.bfloat16)
