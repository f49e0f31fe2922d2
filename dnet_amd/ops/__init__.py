"""dnet_amd.ops — gfx950 HIP kernels with CPU reference fallback.

GPU tensors dispatch to the in-tree ``_C.so`` extension (hand-written
CDNA4/HIP kernels); if the extension is missing on a GPU machine the ops
FAIL LOUDLY — there is no silent eager fallback on the GPU path. CPU
tensors use the fp32 reference implementations (tests / no-GPU machines).
"""
from __future__ import annotations

import importlib.util
from pathlib import Path

import torch

from . import reference as ref

_SO = Path(__file__).resolve().parent / "_C.so"
_C = None
_load_error: Exception | None = None


def _try_load():
    global _C, _load_error
    if _C is not None or _load_error is not None:
        return _C
    try:
        spec = importlib.util.spec_from_file_location("dnet_amd.ops._C", _SO)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        _C = mod
    except Exception as e:  # pragma: no cover
        _load_error = e
    return _C


def has_native() -> bool:
    return _try_load() is not None


def _native():
    c = _try_load()
    if c is None:
        raise RuntimeError(
            f"dnet_amd native extension not available ({_SO}): {_load_error}. "
            "Run `python -m dnet_amd.ops.build` — GPU tensors never fall back "
            "to eager.")
    return c


def rmsnorm(x: torch.Tensor, residual: torch.Tensor | None, w: torch.Tensor,
            eps: float = 1e-6) -> torch.Tensor:
    if x.is_cuda:
        y = torch.empty_like(x)
        _native().rmsnorm(x, residual, w, y, eps)
        return y
    return ref.rmsnorm(x, residual, w, eps)


# split-K f32 scratch, cached per device (max 64 rows x N<8192 cols)
_scratch: dict = {}


def _get_scratch(device) -> torch.Tensor:
    key = str(device)
    if key not in _scratch:
        # zero-initialized: the split-k path relies on it (the combine
        # kernel re-zeroes after each use instead of a per-launch memset)
        # sized for the largest M=64 pass (lm_head-class N): too small a
        # scratch silently disables split-k (launch_m16 falls back to
        # sk=1) — measured on gateup N=55296: 864 blocks quantize badly
        # onto 256 CUs at 3 blocks/CU
        _scratch[key] = torch.zeros(64 * 152064, dtype=torch.float32,
                                    device=device)
    return _scratch[key]


def gemv_bf16(x: torch.Tensor, w: torch.Tensor,
              bias: torch.Tensor | None = None) -> torch.Tensor:
    if x.is_cuda:
        out = torch.empty(x.shape[0], w.shape[0], dtype=x.dtype, device=x.device)
        if x.shape[1] % 64 == 0:
            _native().gemm_m16(x, w, None, bias, out, _get_scratch(x.device),
                               0, False, 16)
        else:
            _native().gemv_bf16(x, w, out, bias)
        return out
    return ref.gemv_bf16(x, w, bias)


def gemv_int8(x: torch.Tensor, w: torch.Tensor, scales: torch.Tensor,
              group: int, bias: torch.Tensor | None = None,
              packed: bool = False) -> torch.Tensor:
    if x.is_cuda:
        out = torch.empty(x.shape[0], w.shape[0], dtype=x.dtype, device=x.device)
        if packed and x.shape[0] > 2:
            _native().gemm_m16(x, w, scales, bias, out, _get_scratch(x.device),
                               group, True, 8)
        else:
            # M<=2: the scalar GEMV reads 16B/lane and hits 4-5.6 TB/s
            _native().gemv_int8(x, w, scales, out, group, bias, packed)
        return out
    if packed:
        w = ref.unpack_int8_mfma(w)
    return ref.gemv_int8(x, w, scales, group, bias)


def gemv_int4(x: torch.Tensor, w: torch.Tensor, scales: torch.Tensor,
              group: int, bias: torch.Tensor | None = None,
              packed: bool = False) -> torch.Tensor:
    if x.is_cuda:
        assert packed, "GPU int4 path needs the packed layout"
        out = torch.empty(x.shape[0], w.shape[0], dtype=x.dtype, device=x.device)
        _native().gemm_m16(x, w, scales, bias, out, _get_scratch(x.device),
                           group, True, 4)
        return out
    return ref.gemv_int4(x, w, scales, group, bias)


def dequant_int4(w: torch.Tensor, scales: torch.Tensor, group: int,
                 packed: bool = True) -> torch.Tensor:
    if w.is_cuda:
        assert packed
        out = torch.empty(w.shape[0], w.shape[1] * 2, dtype=torch.bfloat16,
                          device=w.device)
        _native().dequant_int4(w, scales, out, group)
        return out
    return ref.dequant_int4(w, scales, group)


def dequant_int8(w: torch.Tensor, scales: torch.Tensor, group: int,
                 packed: bool = False) -> torch.Tensor:
    if w.is_cuda:
        out = torch.empty(w.shape, dtype=torch.bfloat16, device=w.device)
        _native().dequant_int8(w, scales, out, group, packed)
        return out
    if packed:
        w = ref.unpack_int8_mfma(w)
    return ref.dequant_int8(w, scales, group)


def _attn_splits(b: int, hkv: int, smax: int) -> int:
    # fill ~1024 blocks (4/CU); cap so each split owns >= ~512 positions of
    # capacity (short actual sequences leave idle splits + combine overhead)
    import os
    force = os.environ.get("DNET_ATTN_SPLITS")
    if force:
        return max(1, int(force))
    splits = 1
    cap = max(1, smax // 512)
    # target ~2048 blocks (8/CU): measured at 2k ctx batch 32 (qwen-32b,
    # 512 base blocks) splits=4 runs 21.5 ms/step vs 27.2 at splits=2 —
    # serial chunks per block hurt more than combine overhead until the
    # per-split span drops under ~512 positions (the cap)
    while (b * hkv * splits * 2 <= 2048 and splits * 2 <= cap):
        splits *= 2
    return splits


def attn_decode(q: torch.Tensor, kcache: torch.Tensor, vcache: torch.Tensor,
                pos: torch.Tensor, scale: float, window: int = 0,
                sinks: torch.Tensor | None = None,
                kscale: torch.Tensor | None = None,
                vscale: torch.Tensor | None = None) -> torch.Tensor:
    if q.is_cuda:
        # q may be a strided slice of the fused QKV buffer; out is contiguous.
        B, Hq, D = q.shape
        dv = vcache.shape[-1]           # MLA: v head dim < qk head dim
        out = torch.empty(B, Hq, dv, dtype=q.dtype, device=q.device)
        splits = _attn_splits(B, kcache.shape[1], kcache.shape[2])
        partials = None
        if splits > 1:
            partials = torch.empty(B * Hq * splits * (dv + 2),
                                   dtype=torch.float32, device=q.device)
        _native().attn_decode(q, kcache, vcache, pos, out, scale, window,
                              sinks, kscale, vscale, partials, splits)
        return out
    return ref.attn_decode(q, kcache, vcache, pos, scale, window, sinks,
                           kscale, vscale)


_PREFILL_DIMS = {(128, 128), (64, 64), (192, 128), (96, 96)}


def attn_prefill_supported(d: int, dv: int) -> bool:
    return (d, dv) in _PREFILL_DIMS


def attn_prefill(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                 scale: float, q_off: int, window: int = 0,
                 sinks: torch.Tensor | None = None) -> torch.Tensor:
    """Fused MFMA flash prefill attention (causal + window + sinks).

    q [B,Hq,T,D], k/v [B,Hkv,S,D(v)] -> [B,Hq,T,Dv]. GPU-only; callers
    gate on attn_prefill_supported and fall back to the chunked-einsum
    path otherwise (models/base.py)."""
    B, Hq, T, D = q.shape
    Dv = v.shape[-1]
    out = torch.empty(B, Hq, T, Dv, dtype=q.dtype, device=q.device)
    _native().attn_prefill(q.contiguous(), k.contiguous(), v.contiguous(),
                           sinks, out, q_off, window, scale)
    return out


def rope_append(q, k, v, kcache, vcache, pos, cos, sin,
                kscale=None, vscale=None, wpos=None) -> None:
    if q.is_cuda:
        _native().rope_append(q, k, v, kcache, vcache, pos, cos, sin,
                              kscale, vscale, wpos)
        return
    ref.rope_append(q, k, v, kcache, vcache, pos, cos, sin, kscale, vscale,
                    wpos)


_defer_cache: dict = {}


def _will_defer(M: int, N: int, K: int, group: int, bits: int,
                scratch_elems: int) -> bool:
    key = (M, N, K, group, bits)
    v = _defer_cache.get(key)
    if v is None:
        v = _native().gemm_m16_will_defer(M, N, K, group, bits, scratch_elems)
        _defer_cache[key] = v
    return v


def gemv_qkv_rope(y: torch.Tensor, w: torch.Tensor, scales: torch.Tensor,
                  group: int, bits: int, bias, nq: int, nkv: int, d: int,
                  kcache, vcache, pos, cos, sin, kscale=None, vscale=None,
                  wpos=None):
    """Fused qkv GEMM + RoPE + KV append for the decode path: when the
    GEMM runs split-k, RoPE reads (and re-zeroes) the f32 scratch
    directly, applies the qkv bias itself, writes the rotated q to a
    dedicated buffer and k/v straight to the cache — the combine kernel
    and the bf16 k/v round trip disappear from the dependency chain.
    Returns the rotated q [B, nq, d]."""
    B = y.shape[0]
    scratch = _get_scratch(y.device)
    if _will_defer(B, w.shape[0], y.shape[1], group, bits, scratch.numel()):
        out = torch.empty(B, w.shape[0], dtype=y.dtype, device=y.device)
        deferred = _native().gemm_m16(y, w, scales, None, out, scratch,
                                      group, True, bits, True)
        assert deferred, "will_defer disagreed with gemm_m16"
        q = torch.empty(B, nq, d, dtype=y.dtype, device=y.device)
        _native().rope_append_f32(scratch, bias, q, nkv, kcache, vcache,
                                  pos, cos, sin, kscale, vscale, wpos)
        return q
    gemv = gemv_int8 if bits == 8 else gemv_int4
    qkv = gemv(y, w, scales, group, bias, True)
    q = qkv[:, :nq * d].view(B, nq, d)
    k = qkv[:, nq * d:(nq + nkv) * d].view(B, nkv, d)
    v = qkv[:, (nq + nkv) * d:].view(B, nkv, d)
    rope_append(q, k, v, kcache, vcache, pos, cos, sin, kscale, vscale,
                wpos=wpos)
    return q


class _Deferred:
    """Sentinel: the projection's result lives (un-combined, f32) in the
    split-k scratch; the consumer must read+re-zero it."""


DEFERRED = _Deferred()


def gemv_defer(x: torch.Tensor, w: torch.Tensor, scales: torch.Tensor,
               group: int, bits: int) -> bool:
    """Launch the GEMM leaving the f32 split-k result in the scratch
    (caller checked _will_defer). Returns True."""
    out = torch.empty(x.shape[0], w.shape[0], dtype=x.dtype, device=x.device)
    d = _native().gemm_m16(x, w, scales, None, out, _get_scratch(x.device),
                           group, True, bits, True)
    assert d, "will_defer disagreed with gemm_m16"
    return True


def rmsnorm_f32_scratch(residual: torch.Tensor, wn: torch.Tensor,
                        eps: float) -> torch.Tensor:
    """RMSNorm(+residual) whose x input is the deferred f32 scratch."""
    y = torch.empty_like(residual)
    _native().rmsnorm_f32(_get_scratch(residual.device), residual, wn, y,
                          eps)
    return y


def resid_add_scratch(h: torch.Tensor) -> None:
    _native().resid_add_f32(h, _get_scratch(h.device))


def linear_will_defer(x: torch.Tensor, w: torch.Tensor, group: int,
                      bits: int) -> bool:
    return _will_defer(x.shape[0], w.shape[0], x.shape[1], group, bits,
                       _get_scratch(x.device).numel())


def gemv_rmsnorm(x: torch.Tensor, w: torch.Tensor, scales: torch.Tensor,
                 group: int, bits: int, residual: torch.Tensor,
                 wn: torch.Tensor, eps: float) -> torch.Tensor:
    """Fused projection GEMM + RMSNorm(+residual) for the decode path:
    when the GEMM runs split-k, the norm reads (and re-zeroes) the f32
    scratch directly — the combine kernel disappears from the chain.
    Updates `residual` in place (residual += proj) like ops.rmsnorm."""
    scratch = _get_scratch(x.device)
    if _will_defer(x.shape[0], w.shape[0], x.shape[1], group, bits,
                   scratch.numel()):
        out = torch.empty(x.shape[0], w.shape[0], dtype=x.dtype,
                          device=x.device)
        _native().gemm_m16(x, w, scales, None, out, scratch, group, True,
                           bits, True)
        y = torch.empty_like(residual)
        _native().rmsnorm_f32(scratch, residual, wn, y, eps)
        return y
    gemv = gemv_int8 if bits == 8 else gemv_int4
    o = gemv(x, w, scales, group, None, True)
    return rmsnorm(o, residual, wn, eps)


def gemv_swiglu(x: torch.Tensor, w: torch.Tensor, scales: torch.Tensor,
                group: int, bits: int = 8,
                packed: bool = True) -> torch.Tensor:
    """Fused gate/up GEMM + SwiGLU for the decode path: when the GEMM
    runs split-k, SwiGLU reads (and re-zeroes) the f32 scratch directly
    instead of going through the f32->bf16 combine kernel — one fewer
    dependent ~5 us kernel per layer per step."""
    if not (x.is_cuda and packed and 2 < x.shape[0] <= 64):
        gemv = gemv_int8 if bits == 8 else gemv_int4
        return swiglu(gemv(x, w, scales, group, None, packed))
    M, N = x.shape[0], w.shape[0]
    out = torch.empty(M, N, dtype=x.dtype, device=x.device)
    scratch = _get_scratch(x.device)
    deferred = _native().gemm_m16(x, w, scales, None, out, scratch, group,
                                  True, bits, True)
    y = torch.empty(M, N // 2, dtype=x.dtype, device=x.device)
    if deferred:
        _native().swiglu_f32(scratch, y, N)
    else:
        _native().swiglu(out, y)
    return y


def swiglu(gu: torch.Tensor) -> torch.Tensor:
    if gu.is_cuda:
        i = gu.shape[-1] // 2
        y = torch.empty(*gu.shape[:-1], i, dtype=gu.dtype, device=gu.device)
        _native().swiglu(gu, y)
        return y
    return ref.swiglu(gu)


def moe_gateup(x: torch.Tensor, w: torch.Tensor, scales, bias,
               we: torch.Tensor, group: int = 0, packed: bool = False,
               glu: int = 0, alpha: float = 1.702,
               limit: float = 7.0) -> torch.Tensor:
    """Grouped expert gate+up with fused GLU: x [M,K], stacked w [E,2I,K]
    (bf16 or grouped-int8), we [M,E] f32 routing weights -> act [E,M,I].
    Experts with all-zero we are skipped on GPU (their act rows hold
    garbage; moe_down never reads them)."""
    if x.is_cuda:
        e, i2, _ = w.shape
        act = torch.empty(e, x.shape[0], i2 // 2, dtype=torch.bfloat16,
                          device=x.device)
        _native().moe_gateup(x, w, scales, bias, we, act, group, packed,
                             glu, alpha, limit)
        return act
    return ref.moe_gateup(x, w, scales, bias, we, group, packed, glu,
                          alpha, limit)


def moe_down(act: torch.Tensor, w: torch.Tensor, scales, bias,
             we: torch.Tensor, group: int = 0,
             packed: bool = False) -> torch.Tensor:
    """Grouped expert down projection, weighted f32 accumulation:
    act [E,M,I], stacked w [E,H,I], we [M,E] -> out [M,H] f32."""
    if act.is_cuda:
        out = torch.zeros(act.shape[1], w.shape[1], dtype=torch.float32,
                          device=act.device)
        _native().moe_down(act, w, scales, bias, we, out, group, packed)
        return out
    return ref.moe_down(act, w, scales, bias, we, group, packed)


quantize_int8 = ref.quantize_int8
quantize_int4 = ref.quantize_int4
pack_int8_mfma = ref.pack_int8_mfma
pack_int4_mfma = ref.pack_int4_mfma
unpack_int8_mfma = ref.unpack_int8_mfma
rope_tables = ref.rope_tables
rope_apply = ref.rope_apply
mxfp4_dequant = ref.mxfp4_dequant
quantize_mxfp4 = ref.quantize_mxfp4


def dequant_mxfp4(w: torch.Tensor, scales: torch.Tensor) -> torch.Tensor:
    """MXFP4 row-major (nibbles [N, K/2] + e8m0 [N, K/32]) -> bf16."""
    if w.is_cuda and has_native():
        out = torch.empty(w.shape[0], w.shape[1] * 2, dtype=torch.bfloat16,
                          device=w.device)
        _native().dequant_mxfp4(w, scales, out)
        return out
    return ref.dequant_mxfp4(w, scales)


def attn_decode_partials(q, kcache, vcache, pos, scale,
                         kscale=None, vscale=None) -> torch.Tensor:
    """Local flash-decode partials [B, Hq, splits, Dv+2] f32 over THIS
    cache shard — the combinable form for context-parallel attention
    (combine with ``attn_combine`` after gathering shards)."""
    if q.is_cuda:
        B, Hq, _ = q.shape
        dv = vcache.shape[-1]
        splits = _attn_splits(B, kcache.shape[1], kcache.shape[2])
        partials = torch.zeros(B, Hq, splits, dv + 2, dtype=torch.float32,
                               device=q.device)
        partials[..., dv] = -1e30   # empty shards stay neutral in combine
        out = torch.empty(B, Hq, dv, dtype=q.dtype, device=q.device)
        _native().attn_decode(q, kcache, vcache, pos, out, scale, 0, None,
                              kscale, vscale, partials.view(-1), splits,
                              False)
        return partials
    return ref.attn_decode_partials(q, kcache, vcache, pos, scale,
                                    kscale, vscale)


def attn_combine(partials: torch.Tensor,
                 sinks: torch.Tensor | None = None) -> torch.Tensor:
    """partials [B, Hq, S, Dv+2] -> out [B, Hq, Dv]."""
    if partials.is_cuda:
        B, Hq, S, W = partials.shape
        out = torch.empty(B, Hq, W - 2, dtype=torch.bfloat16,
                          device=partials.device)
        _native().attn_combine(partials.contiguous().view(-1), sinks, out, S)
        return out
    return ref.attn_combine(partials, sinks)
