// Fused decode-step RoPE + KV-cache append.
//
// Applies neox-style rotary embedding (precomputed f32 cos/sin tables — the
// CDNA4 guide's rule for trig-heavy ops: tables on host, never sinf/cosf per
// element) to q in place and to k, then writes the rotated k and raw v into
// the KV cache at position pos[b]. Position comes from a device tensor so the
// whole decode step is hipGraph-replayable.
//
// Reference equivalent: mlx_lm RoPE + KVCache.update_and_fetch
// (reference: src/dnet/core/models/llama.py, utils/model.py make_cache).
#include "common.h"

namespace dnet {

// grid: (B, Hq + Hkv); block: D/2 lanes, lane i rotates pair (i, i + D/2).
// Q8: KV cache stored int8 with one bf16 scale per 64-dim group
// (reference: utils/model.py make_cache kv quantization, 8-bit group 64).
// F32: split-k-fused variant — q/k/v come from the UN-COMBINED f32
// split-k scratch of the qkv GEMM (layout [B, (Hq+2*Hkv)*D], row stride
// ldq), bias (bf16, optional) is applied here, the rotated q goes to a
// dedicated bf16 buffer `qout`, and every scratch element read is
// re-zeroed for the next split-k use. Removes the f32->bf16 combine
// kernel from the decode dependency chain.
template <bool Q8, bool F32 = false>
__global__ void rope_append_kernel(short* __restrict__ q,
                                   short* __restrict__ k,
                                   const short* __restrict__ v,
                                   void* __restrict__ kcache,
                                   void* __restrict__ vcache,
                                   short* __restrict__ kscale,
                                   short* __restrict__ vscale,
                                   const int* __restrict__ pos,
                                   const int* __restrict__ wpos,
                                   const float* __restrict__ cost,
                                   const float* __restrict__ sint,
                                   const int Hq, const int Hkv, const int Smax,
                                   const int D, const int ldq, const int ldk,
                                   const int ldv,
                                   float* __restrict__ src = nullptr,
                                   const short* __restrict__ biasv = nullptr,
                                   short* __restrict__ qout = nullptr) {
  const int b = blockIdx.x;
  const int h = blockIdx.y;
  const int i = threadIdx.x;  // 0 .. max(D/2,64)-1; lanes >= D/2 idle
  const int p = pos[b];
  const int half = D / 2;
  const bool act = i < half;
  const float c = act ? cost[(int64_t)p * half + i] : 0.f;
  const float s = act ? sint[(int64_t)p * half + i] : 0.f;
  if (h < Hq) {
    if (!act) return;
    if (F32) {
      float* row = src + (int64_t)b * ldq + h * D;
      float x1 = row[i], x2 = row[i + half];
      if (biasv != nullptr) {
        x1 += bits2f(biasv[h * D + i]);
        x2 += bits2f(biasv[h * D + i + half]);
      }
      row[i] = 0.f;
      row[i + half] = 0.f;
      short* qrow = qout + (int64_t)b * Hq * D + h * D;
      qrow[i] = f2bits(x1 * c - x2 * s);
      qrow[i + half] = f2bits(x2 * c + x1 * s);
      return;
    }
    short* row = q + (int64_t)b * ldq + h * D;
    const float x1 = bits2f(row[i]), x2 = bits2f(row[i + half]);
    row[i] = f2bits(x1 * c - x2 * s);
    row[i + half] = f2bits(x2 * c + x1 * s);
    return;
  }
  const int hk = h - Hq;
  short* krow = k + (int64_t)b * ldk + hk * D;
  float k1 = 0.f, k2 = 0.f, v1 = 0.f, v2 = 0.f;
  if (act && F32) {
    const int kcol = (Hq + hk) * D, vcol = (Hq + Hkv + hk) * D;
    float* krowf = src + (int64_t)b * ldq + kcol;
    float* vrowf = src + (int64_t)b * ldq + vcol;
    float x1 = krowf[i], x2 = krowf[i + half];
    v1 = vrowf[i];
    v2 = vrowf[i + half];
    if (biasv != nullptr) {
      x1 += bits2f(biasv[kcol + i]);
      x2 += bits2f(biasv[kcol + i + half]);
      v1 += bits2f(biasv[vcol + i]);
      v2 += bits2f(biasv[vcol + i + half]);
    }
    krowf[i] = 0.f;
    krowf[i + half] = 0.f;
    vrowf[i] = 0.f;
    vrowf[i + half] = 0.f;
    k1 = x1 * c - x2 * s;
    k2 = x2 * c + x1 * s;
  } else if (act) {
    const float x1 = bits2f(krow[i]), x2 = bits2f(krow[i + half]);
    k1 = x1 * c - x2 * s;
    k2 = x2 * c + x1 * s;
    krow[i] = f2bits(k1);
    krow[i + half] = f2bits(k2);
    const short* vrow = v + (int64_t)b * ldv + hk * D;
    v1 = bits2f(vrow[i]);
    v2 = bits2f(vrow[i + half]);
  }
  // context parallelism: the cache write targets the LOCAL position
  // wpos[b] (rope still uses the global p); rows outside this rank's
  // shard are skipped (uniform per block, so the Q8 wave reductions
  // below never diverge)
  const int wp = (wpos != nullptr) ? wpos[b] : p;
  if (wp < 0 || wp >= Smax) return;
  const int64_t rowbase = ((int64_t)b * Hkv + hk) * Smax + wp;
  if (!Q8) {
    if (!act) return;
    short* kdst = (short*)kcache + rowbase * D;
    kdst[i] = f2bits(k1);
    kdst[i + half] = f2bits(k2);
    short* vdst = (short*)vcache + rowbase * D;
    vdst[i] = f2bits(v1);
    vdst[i + half] = f2bits(v2);
    return;
  }
  // int8 group-64 quantization. For D=128 (half=64): elem i is in group 0,
  // elem i+half in group 1. For D=64 (half=32): both elems in group 0.
  // One full wave participates in the reductions (idle lanes contribute 0).
  const int ng = D / 64;
  auto wave_max = [](float m) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      m = fmaxf(m, __shfl_xor(m, off, 64));
    return m;
  };
  auto q8scale = [](float amax) {
    return bits2f(f2bits(fmaxf(amax, 1e-8f) / 127.f));  // bf16-rounded
  };
  {
    const float a0 = wave_max(ng == 2 ? fabsf(k1) : fmaxf(fabsf(k1), fabsf(k2)));
    const float a1 = ng == 2 ? wave_max(fabsf(k2)) : a0;
    if (act) {
      const float s1 = q8scale(a0), s2 = q8scale(a1);
      int8_t* kdst = (int8_t*)kcache + rowbase * D;
      kdst[i] = (int8_t)min(max(__float2int_rn(k1 / s1), -127), 127);
      kdst[i + half] = (int8_t)min(max(__float2int_rn(k2 / s2), -127), 127);
      short* ksrow = kscale + rowbase * ng;
      if (i == 0) ksrow[0] = f2bits(q8scale(a0));
      if (i == 0 && ng == 2) ksrow[1] = f2bits(q8scale(a1));
    }
  }
  {
    const float a0 = wave_max(ng == 2 ? fabsf(v1) : fmaxf(fabsf(v1), fabsf(v2)));
    const float a1 = ng == 2 ? wave_max(fabsf(v2)) : a0;
    if (act) {
      const float s1 = q8scale(a0), s2 = q8scale(a1);
      int8_t* vdst = (int8_t*)vcache + rowbase * D;
      vdst[i] = (int8_t)min(max(__float2int_rn(v1 / s1), -127), 127);
      vdst[i + half] = (int8_t)min(max(__float2int_rn(v2 / s2), -127), 127);
      short* vsrow = vscale + rowbase * ng;
      if (i == 0) vsrow[0] = f2bits(q8scale(a0));
      if (i == 0 && ng == 2) vsrow[1] = f2bits(q8scale(a1));
    }
  }
}

// Fused qkv-split-k-combine + RoPE + append: src = f32 scratch of the
// qkv GEMM (un-combined), qout receives the rotated+biased q rows.
void rope_append_f32(torch::Tensor src, c10::optional<torch::Tensor> bias,
                     torch::Tensor qout, int64_t Hkv, torch::Tensor kcache,
                     torch::Tensor vcache, torch::Tensor pos,
                     torch::Tensor cos_table, torch::Tensor sin_table,
                     c10::optional<torch::Tensor> kscale,
                     c10::optional<torch::Tensor> vscale,
                     c10::optional<torch::Tensor> wpos) {
  const int64_t B = qout.size(0), Hq = qout.size(1), D = qout.size(2);
  const int64_t Smax = kcache.size(2);
  const int64_t N = (Hq + 2 * Hkv) * D;
  DNET_CHECK(D % 2 == 0 && D / 2 <= 1024, "head_dim");
  DNET_CHECK(qout.is_contiguous() && src.is_contiguous(), "contig");
  DNET_CHECK(src.numel() >= B * N, "scratch too small");
  DNET_CHECK(cos_table.dtype() == torch::kFloat32, "cos table f32");
  auto stream = current_stream();
  const bool q8 = kcache.dtype() == torch::kInt8;
  const int* wpp = wpos.has_value() ? (const int*)wpos->data_ptr() : nullptr;
  const short* bp = bias.has_value() ? (const short*)bias->data_ptr()
                                     : nullptr;
  const unsigned thr = q8 ? 64u : (unsigned)(D / 2);
  if (q8)
    hipLaunchKernelGGL((rope_append_kernel<true, true>),
                       dim3((unsigned)B, (unsigned)(Hq + Hkv)), dim3(thr), 0,
                       stream, nullptr, nullptr, nullptr, kcache.data_ptr(),
                       vcache.data_ptr(), (short*)kscale->data_ptr(),
                       (short*)vscale->data_ptr(),
                       (const int*)pos.data_ptr(), wpp,
                       (const float*)cos_table.data_ptr(),
                       (const float*)sin_table.data_ptr(), (int)Hq, (int)Hkv,
                       (int)Smax, (int)D, (int)N, 0, 0,
                       (float*)src.data_ptr(), bp, (short*)qout.data_ptr());
  else
    hipLaunchKernelGGL((rope_append_kernel<false, true>),
                       dim3((unsigned)B, (unsigned)(Hq + Hkv)), dim3(thr), 0,
                       stream, nullptr, nullptr, nullptr, kcache.data_ptr(),
                       vcache.data_ptr(), nullptr, nullptr,
                       (const int*)pos.data_ptr(), wpp,
                       (const float*)cos_table.data_ptr(),
                       (const float*)sin_table.data_ptr(), (int)Hq, (int)Hkv,
                       (int)Smax, (int)D, (int)N, 0, 0,
                       (float*)src.data_ptr(), bp, (short*)qout.data_ptr());
}

void rope_append(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                 torch::Tensor kcache, torch::Tensor vcache, torch::Tensor pos,
                 torch::Tensor cos_table, torch::Tensor sin_table,
                 c10::optional<torch::Tensor> kscale,
                 c10::optional<torch::Tensor> vscale,
                 c10::optional<torch::Tensor> wpos) {
  const int64_t B = q.size(0), Hq = q.size(1), D = q.size(2);
  const int64_t Hkv = k.size(1), Smax = kcache.size(2);
  DNET_CHECK(D % 2 == 0 && D / 2 <= 1024, "head_dim");
  DNET_CHECK(cos_table.dtype() == torch::kFloat32, "cos table f32");
  // q/k/v may be column slices of one fused QKV buffer: strides (ld, D, 1).
  DNET_CHECK(q.stride(2) == 1 && q.stride(1) == D, "q inner contiguous");
  DNET_CHECK(k.stride(2) == 1 && k.stride(1) == D, "k inner contiguous");
  DNET_CHECK(v.stride(2) == 1 && v.stride(1) == D, "v inner contiguous");
  auto stream = current_stream();
  const bool q8 = kcache.dtype() == torch::kInt8;
  const int* wpp = wpos.has_value() ? (const int*)wpos->data_ptr() : nullptr;
  if (q8) {
    DNET_CHECK(kscale.has_value() && vscale.has_value(), "q8 needs scales");
    DNET_CHECK(D == 64 || D == 128, "q8 kv needs D 64/128");
    hipLaunchKernelGGL(rope_append_kernel<true>,
                       dim3((unsigned)B, (unsigned)(Hq + Hkv)),
                       dim3(64), 0, stream, (short*)q.data_ptr(),
                       (short*)k.data_ptr(), (const short*)v.data_ptr(),
                       kcache.data_ptr(), vcache.data_ptr(),
                       (short*)kscale->data_ptr(), (short*)vscale->data_ptr(),
                       (const int*)pos.data_ptr(), wpp,
                       (const float*)cos_table.data_ptr(),
                       (const float*)sin_table.data_ptr(), (int)Hq, (int)Hkv,
                       (int)Smax, (int)D, (int)q.stride(0), (int)k.stride(0),
                       (int)v.stride(0));
  } else {
    hipLaunchKernelGGL(rope_append_kernel<false>,
                       dim3((unsigned)B, (unsigned)(Hq + Hkv)),
                       dim3((unsigned)(D / 2)), 0, stream, (short*)q.data_ptr(),
                       (short*)k.data_ptr(), (const short*)v.data_ptr(),
                       kcache.data_ptr(), vcache.data_ptr(), nullptr, nullptr,
                       (const int*)pos.data_ptr(), wpp,
                       (const float*)cos_table.data_ptr(),
                       (const float*)sin_table.data_ptr(), (int)Hq, (int)Hkv,
                       (int)Smax, (int)D, (int)q.stride(0), (int)k.stride(0),
                       (int)v.stride(0));
  }
}

}  // namespace dnet
