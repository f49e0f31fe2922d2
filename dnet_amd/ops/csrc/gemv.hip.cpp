// Decode-path skinny GEMM / GEMV kernels: out[M,N] = x[M,K] @ W[N,K]^T.
//
// M is the number of sequences decoding together (1..16 per pass); the
// kernels are weight-bandwidth-bound, so the W read is amortized across all
// M rows (batching is nearly free). Weights are read straight to VGPRs with
// 16 B/lane vector loads and a deep K loop — per the CDNA4 guide, GEMV
// operands streamed once per block want no LDS round trip
// (cdna_hip_programming.md §5 "GEMV / M<=16 decode weights" row).
//
//  - gemv_bf16:  W bf16 [N,K] row-major.
//  - gemv_int8:  W int8 [N,K] + per-group symmetric scales bf16 [N,K/G]
//                (grouped-affine W8A16; dequant fused into the dot).
//
// Replaces the reference's MLX quantized-matmul built-ins
// (reference: src/dnet/core/models/base.py nn.quantize usage) with a
// CDNA4-native fused path.
#include "common.h"

namespace dnet {

constexpr int kRowsPerBlock = 4;  // 4 waves, one output row each

template <int M>
__global__ void gemv_bf16_kernel(const short* __restrict__ x,
                                 const short* __restrict__ w,
                                 const short* __restrict__ bias,
                                 short* __restrict__ out, const int K,
                                 const int N, const int ldx) {
  const int wid = threadIdx.x / kWave;
  const int lane = threadIdx.x & (kWave - 1);
  const int n = blockIdx.x * kRowsPerBlock + wid;
  if (n >= N) return;
  const short8* wrow = reinterpret_cast<const short8*>(w + (int64_t)n * K);
  float acc[M];
#pragma unroll
  for (int m = 0; m < M; ++m) acc[m] = 0.f;
  const int vecs = K / 8;  // K % 8 == 0
  for (int i = lane; i < vecs; i += kWave) {
    const short8 wv = wrow[i];
#pragma unroll
    for (int m = 0; m < M; ++m) {
      const short8 xv = reinterpret_cast<const short8*>(x + (int64_t)m * ldx)[i];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        acc[m] = fmaf(bits2f(wv.x[j]), bits2f(xv.x[j]), acc[m]);
    }
  }
#pragma unroll
  for (int m = 0; m < M; ++m) {
    float r = wave_reduce_sum(acc[m]);
    if (lane == 0) {
      if (bias != nullptr) r += bits2f(bias[n]);
      out[(int64_t)m * N + n] = f2bits(r);
    }
  }
}

// PACKED: weights in pack_int8_mfma chunk-pair order — the dot is
// order-independent, so only the x vector indices change (vec i covers
// orig k = pr*64 + sl*8 (+32): pr = i/4, sl = i%4).
template <int M, bool PACKED>
__global__ void gemv_int8_kernel(const short* __restrict__ x,
                                 const int8_t* __restrict__ w,
                                 const short* __restrict__ scales,
                                 const short* __restrict__ bias,
                                 short* __restrict__ out, const int K,
                                 const int N, const int G, const int ldx) {
  const int wid = threadIdx.x / kWave;
  const int lane = threadIdx.x & (kWave - 1);
  const int n = blockIdx.x * kRowsPerBlock + wid;
  if (n >= N) return;
  const int4* wrow = reinterpret_cast<const int4*>(w + (int64_t)n * K);  // 16 int8
  const short* srow = scales + (int64_t)n * (K / G);
  float acc[M];
#pragma unroll
  for (int m = 0; m < M; ++m) acc[m] = 0.f;
  const int vecs = K / 16;  // K % 16 == 0, G % 16 == 0
  for (int i = lane; i < vecs; i += kWave) {
    const int4 wv = wrow[i];
    const int8_t* q = reinterpret_cast<const int8_t*>(&wv);
    const float s = bits2f(srow[(i * 16) / G]);
    float wq[16];
#pragma unroll
    for (int j = 0; j < 16; ++j) wq[j] = (float)q[j] * s;
#pragma unroll
    for (int m = 0; m < M; ++m) {
      const short8* xrow = reinterpret_cast<const short8*>(x + (int64_t)m * ldx);
      short8 x0, x1;
      if (PACKED) {
        const int pr = i / 4, sl = i % 4;
        x0 = xrow[pr * 8 + sl];
        x1 = xrow[pr * 8 + sl + 4];
      } else {
        x0 = xrow[2 * i];
        x1 = xrow[2 * i + 1];
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        acc[m] = fmaf(wq[j], bits2f(x0.x[j]), acc[m]);
        acc[m] = fmaf(wq[8 + j], bits2f(x1.x[j]), acc[m]);
      }
    }
  }
#pragma unroll
  for (int m = 0; m < M; ++m) {
    float r = wave_reduce_sum(acc[m]);
    if (lane == 0) {
      if (bias != nullptr) r += bits2f(bias[n]);
      out[(int64_t)m * N + n] = f2bits(r);
    }
  }
}

template <typename LaunchFn>
static void dispatch_m(int M, LaunchFn&& fn) {
  switch (M) {
    case 1: fn(std::integral_constant<int, 1>{}); break;
    case 2: fn(std::integral_constant<int, 2>{}); break;
    case 3: fn(std::integral_constant<int, 3>{}); break;
    case 4: fn(std::integral_constant<int, 4>{}); break;
    case 5: fn(std::integral_constant<int, 5>{}); break;
    case 6: fn(std::integral_constant<int, 6>{}); break;
    case 8: fn(std::integral_constant<int, 8>{}); break;
    case 12: fn(std::integral_constant<int, 12>{}); break;
    case 16: fn(std::integral_constant<int, 16>{}); break;
    default: TORCH_CHECK(false, "gemv: unsupported M tile ", M);
  }
}

// Host entry: loops over M in tiles of <=16 rows.
void gemv_bf16(torch::Tensor x, torch::Tensor w, torch::Tensor out,
               c10::optional<torch::Tensor> bias) {
  const int64_t M = x.size(0), K = x.size(1), N = w.size(0);
  DNET_CHECK(w.size(1) == K && out.size(0) == M && out.size(1) == N, "shape");
  DNET_CHECK(K % 8 == 0, "K % 8");
  DNET_CHECK(x.is_contiguous() && w.is_contiguous() && out.is_contiguous(), "contig");
  auto stream = current_stream();
  const short* bptr = bias.has_value() ? (const short*)bias->data_ptr() : nullptr;
  const int grid = cdiv(N, kRowsPerBlock);
  int64_t m0 = 0;
  while (m0 < M) {
    int mt = (int)std::min<int64_t>(M - m0, 16);
    if (mt > 8 && mt < 12) mt = 8;
    else if (mt > 12 && mt < 16) mt = 12;
    else if (mt == 7) mt = 6;
    dispatch_m(mt, [&](auto mc) {
      hipLaunchKernelGGL((gemv_bf16_kernel<decltype(mc)::value>), dim3(grid),
                         dim3(kRowsPerBlock * kWave), 0, stream,
                         (const short*)x.data_ptr() + m0 * K,
                         (const short*)w.data_ptr(), bptr,
                         (short*)out.data_ptr() + m0 * N, (int)K, (int)N, (int)K);
    });
    m0 += mt;
  }
}

void gemv_int8(torch::Tensor x, torch::Tensor w, torch::Tensor scales,
               torch::Tensor out, int64_t group,
               c10::optional<torch::Tensor> bias, bool packed) {
  const int64_t M = x.size(0), K = x.size(1), N = w.size(0);
  DNET_CHECK(w.size(1) == K && out.size(1) == N, "shape");
  DNET_CHECK(K % 16 == 0 && group % 16 == 0 && K % group == 0, "K/group align");
  if (packed) DNET_CHECK(K % 64 == 0 && group % 64 == 0, "packed align");
  DNET_CHECK(scales.size(0) == N && scales.size(1) == K / group, "scales shape");
  DNET_CHECK(x.is_contiguous() && w.is_contiguous() && out.is_contiguous() &&
                 scales.is_contiguous(), "contig");
  auto stream = current_stream();
  const short* bptr = bias.has_value() ? (const short*)bias->data_ptr() : nullptr;
  const int grid = cdiv(N, kRowsPerBlock);
  int64_t m0 = 0;
  while (m0 < M) {
    int mt = (int)std::min<int64_t>(M - m0, 16);
    if (mt > 8 && mt < 12) mt = 8;
    else if (mt > 12 && mt < 16) mt = 12;
    else if (mt == 7) mt = 6;
    dispatch_m(mt, [&](auto mc) {
      if (packed)
        hipLaunchKernelGGL((gemv_int8_kernel<decltype(mc)::value, true>),
                           dim3(grid), dim3(kRowsPerBlock * kWave), 0, stream,
                           (const short*)x.data_ptr() + m0 * K,
                           (const int8_t*)w.data_ptr(),
                           (const short*)scales.data_ptr(), bptr,
                           (short*)out.data_ptr() + m0 * N, (int)K, (int)N,
                           (int)group, (int)K);
      else
        hipLaunchKernelGGL((gemv_int8_kernel<decltype(mc)::value, false>),
                           dim3(grid), dim3(kRowsPerBlock * kWave), 0, stream,
                           (const short*)x.data_ptr() + m0 * K,
                           (const int8_t*)w.data_ptr(),
                           (const short*)scales.data_ptr(), bptr,
                           (short*)out.data_ptr() + m0 * N, (int)K, (int)N,
                           (int)group, (int)K);
    });
    m0 += mt;
  }
}

}  // namespace dnet
