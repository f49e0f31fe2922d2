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
__global__ void rope_append_kernel(short* __restrict__ q,
                                   short* __restrict__ k,
                                   const short* __restrict__ v,
                                   short* __restrict__ kcache,
                                   short* __restrict__ vcache,
                                   const int* __restrict__ pos,
                                   const float* __restrict__ cost,
                                   const float* __restrict__ sint,
                                   const int Hq, const int Hkv, const int Smax,
                                   const int D, const int ldq, const int ldk,
                                   const int ldv) {
  const int b = blockIdx.x;
  const int h = blockIdx.y;
  const int i = threadIdx.x;  // 0 .. D/2-1
  const int p = pos[b];
  const int half = D / 2;
  const float c = cost[(int64_t)p * half + i];
  const float s = sint[(int64_t)p * half + i];
  if (h < Hq) {
    short* row = q + (int64_t)b * ldq + h * D;
    const float x1 = bits2f(row[i]), x2 = bits2f(row[i + half]);
    row[i] = f2bits(x1 * c - x2 * s);
    row[i + half] = f2bits(x2 * c + x1 * s);
  } else {
    const int hk = h - Hq;
    short* krow = k + (int64_t)b * ldk + hk * D;
    const float x1 = bits2f(krow[i]), x2 = bits2f(krow[i + half]);
    const short r1 = f2bits(x1 * c - x2 * s);
    const short r2 = f2bits(x2 * c + x1 * s);
    krow[i] = r1;
    krow[i + half] = r2;
    short* kdst = kcache + (((int64_t)b * Hkv + hk) * Smax + p) * D;
    kdst[i] = r1;
    kdst[i + half] = r2;
    const short* vrow = v + (int64_t)b * ldv + hk * D;
    short* vdst = vcache + (((int64_t)b * Hkv + hk) * Smax + p) * D;
    vdst[i] = vrow[i];
    vdst[i + half] = vrow[i + half];
  }
}

void rope_append(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                 torch::Tensor kcache, torch::Tensor vcache, torch::Tensor pos,
                 torch::Tensor cos_table, torch::Tensor sin_table) {
  const int64_t B = q.size(0), Hq = q.size(1), D = q.size(2);
  const int64_t Hkv = k.size(1), Smax = kcache.size(2);
  DNET_CHECK(D % 2 == 0 && D / 2 <= 1024, "head_dim");
  DNET_CHECK(cos_table.dtype() == torch::kFloat32, "cos table f32");
  // q/k/v may be column slices of one fused QKV buffer: strides (ld, D, 1).
  DNET_CHECK(q.stride(2) == 1 && q.stride(1) == D, "q inner contiguous");
  DNET_CHECK(k.stride(2) == 1 && k.stride(1) == D, "k inner contiguous");
  DNET_CHECK(v.stride(2) == 1 && v.stride(1) == D, "v inner contiguous");
  auto stream = current_stream();
  hipLaunchKernelGGL(rope_append_kernel, dim3((unsigned)B, (unsigned)(Hq + Hkv)),
                     dim3((unsigned)(D / 2)), 0, stream, (short*)q.data_ptr(),
                     (short*)k.data_ptr(), (const short*)v.data_ptr(),
                     (short*)kcache.data_ptr(), (short*)vcache.data_ptr(),
                     (const int*)pos.data_ptr(), (const float*)cos_table.data_ptr(),
                     (const float*)sin_table.data_ptr(), (int)Hq, (int)Hkv,
                     (int)Smax, (int)D, (int)q.stride(0), (int)k.stride(0),
                     (int)v.stride(0));
}

}  // namespace dnet
