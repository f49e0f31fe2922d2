// MFMA decode GEMM for M<=16: out[M,N] = x[M,K] @ W[N,K]^T (+bias).
//
// The scalar GEMV (gemv.hip.cpp) is VALU-bound at M>=4: every weight element
// costs ~2 VALU ops per output row. Here one v_mfma_f32_16x16x32_bf16
// covers a 16(M)x16(N)x32(K) tile, so the per-weight-element cost collapses
// to the in-register int8->bf16 dequant (QUANT path) or nothing (bf16 path),
// and the kernel runs at the weight-read HBM bound. Fragment layout for
// 16x16x32 (cdna_hip_programming.md §3): lane l holds A[row=l&15][k=(l>>4)*8+j],
// B[col=l&15][k=(l>>4)*8+j], C[col=l&15][row=(l>>4)*4+r].
//
// Wave tile: 16 N-columns; block = 4 waves = 64 columns; gridDim.y = SPLITK
// K-splits (chosen so blocks ~ fill 256 CUs; partials combined by f32
// atomics into a scratch accumulator, then a tiny convert+bias kernel).
#include "common.h"

namespace dnet {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

__device__ __forceinline__ bf16x8 deq8(const int8_t* q, const float s) {
  bf16x8 b;
#pragma unroll
  for (int j = 0; j < 8; ++j) b[j] = (__bf16)((float)q[j] * s);
  return b;
}

template <bool QUANT, bool PACKED>
__global__ void gemm_m16_kernel(const short* __restrict__ x,
                                const void* __restrict__ w,
                                const short* __restrict__ scales,
                                const short* __restrict__ bias,
                                short* __restrict__ out,        // SPLITK==1
                                float* __restrict__ out_f32,    // SPLITK>1
                                const int M, const int K, const int N,
                                const int G, const int splitk) {
  const int wave = threadIdx.x / kWave;
  const int lane = threadIdx.x & (kWave - 1);
  const int n0 = (blockIdx.x * 4 + wave) * 16;
  // LDS-staged paths have block-wide barriers: OOB waves must stay alive
  // (their stores are skipped); only the barrier-free path may exit early.
  if (QUANT && !PACKED && n0 >= N) return;
  const int row = lane & 15;          // A row (x row = output m), B col
  const int ks = (lane >> 4) * 8;     // k-offset of this lane's 8-elem slice
  const int n_w = min(n0 + row, N - 1);     // this lane's W row

  const int chunks = K / 32;
  const int per_split = chunks / splitk;
  const int c_begin = blockIdx.y * per_split;
  const int c_end = (blockIdx.y == splitk - 1) ? chunks : c_begin + per_split;

  // Rows >= M read a clamped (valid) row and produce garbage C rows that are
  // never stored — cheaper than per-load zero-masking.
  const short* xrow = x + (int64_t)min(row, M - 1) * K;
  f32x4 acc0 = {0.f, 0.f, 0.f, 0.f};
  f32x4 acc1 = {0.f, 0.f, 0.f, 0.f};

  // 4-chunk unrolled main loop: all 8-12 loads issue before the first
  // dequant+MFMA, so ~4 HBM loads stay in flight per wave (the single-chunk
  // loop was load-latency-bound at ~1 chunk / 970 cycles).
  if constexpr (QUANT && PACKED) {
    // W stored in MFMA chunk-pair order (pack_int8_mfma): one b128 load per
    // lane covers its B slices of two adjacent chunks -> full 64 B bursts.
    // x is staged tile-wise into LDS with coalesced full-line loads (the
    // fragment-shaped direct read touches 16 scattered rows per instruction
    // and costs up to +45% — cdna_hip_programming.md §5 M=256 GEMM row).
    constexpr int XT = 1024;              // k values per x tile
    constexpr int SE = XT + 8;            // row stride in bf16 elems (16B pad)
    __shared__ short x_lds[16 * SE];
    const int8_t* wrow = (const int8_t*)w + (int64_t)n_w * K;
    const short* srow = scales + (int64_t)n_w * (K / G);
    const int pairs = K / 64;
    const int pp = pairs / splitk;
    const int p_begin = blockIdx.y * pp;
    const int p_end = (blockIdx.y == splitk - 1) ? pairs : p_begin + pp;
    const int woff = (lane >> 4) * 16;
    for (int k0 = p_begin * 64; k0 < p_end * 64; k0 += XT) {
      const int tk = min(XT, p_end * 64 - k0);
      __syncthreads();
      for (int idx = threadIdx.x; idx < 16 * (tk / 8); idx += 256) {
        const int r = idx / (tk / 8);
        const int vec = idx % (tk / 8);
        *reinterpret_cast<short8*>(&x_lds[r * SE + vec * 8]) =
            *reinterpret_cast<const short8*>(
                &x[(int64_t)min(r, M - 1) * K + k0 + vec * 8]);
      }
      __syncthreads();
      const short* arow = &x_lds[row * SE];
      int pl = 0;
      const int pl_end = tk / 64;
      for (; pl + 4 <= pl_end; pl += 4) {
        bf16x8 a[8];
        int4 wq[4];
        float s2[4];
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          const int p = (k0 / 64) + pl + u;
          wq[u] = *reinterpret_cast<const int4*>(&wrow[p * 64 + woff]);
          s2[u] = bits2f(srow[(p * 64) / G]);
          a[2 * u] = *reinterpret_cast<const bf16x8*>(&arow[(pl + u) * 64 + ks]);
          a[2 * u + 1] = *reinterpret_cast<const bf16x8*>(&arow[(pl + u) * 64 + 32 + ks]);
        }
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          const int8_t* q = reinterpret_cast<const int8_t*>(&wq[u]);
          acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[2 * u], deq8(q, s2[u]),
                                                         acc0, 0, 0, 0);
          acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[2 * u + 1],
                                                         deq8(q + 8, s2[u]),
                                                         acc1, 0, 0, 0);
        }
      }
      for (; pl < pl_end; ++pl) {
        const int p = (k0 / 64) + pl;
        const int4 wq = *reinterpret_cast<const int4*>(&wrow[p * 64 + woff]);
        const int8_t* q = reinterpret_cast<const int8_t*>(&wq);
        const float sv = bits2f(srow[(p * 64) / G]);
        const bf16x8 al = *reinterpret_cast<const bf16x8*>(&arow[pl * 64 + ks]);
        const bf16x8 ah = *reinterpret_cast<const bf16x8*>(&arow[pl * 64 + 32 + ks]);
        acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(al, deq8(q, sv), acc0, 0, 0, 0);
        acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ah, deq8(q + 8, sv), acc1, 0, 0, 0);
      }
    }
  } else if constexpr (QUANT) {
    const int8_t* wrow = (const int8_t*)w + (int64_t)n_w * K;
    const short* srow = scales + (int64_t)n_w * (K / G);
    int c = c_begin;
    for (; c + 4 <= c_end; c += 4) {
      bf16x8 a[4];
      int2 wq[4];
      float s[4];
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const int k = (c + u) * 32 + ks;
        a[u] = *reinterpret_cast<const bf16x8*>(&xrow[k]);
        wq[u] = *reinterpret_cast<const int2*>(&wrow[k]);
        s[u] = bits2f(srow[k / G]);
      }
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const int8_t* q = reinterpret_cast<const int8_t*>(&wq[u]);
        bf16x8 b;
#pragma unroll
        for (int j = 0; j < 8; ++j) b[j] = (__bf16)((float)q[j] * s[u]);
        if (u & 1)
          acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[u], b, acc1, 0, 0, 0);
        else
          acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[u], b, acc0, 0, 0, 0);
      }
    }
    for (; c < c_end; ++c) {
      const int k = c * 32 + ks;
      const bf16x8 a = *reinterpret_cast<const bf16x8*>(&xrow[k]);
      const int2 wq8 = *reinterpret_cast<const int2*>(&wrow[k]);
      const int8_t* q = reinterpret_cast<const int8_t*>(&wq8);
      const float s = bits2f(srow[k / G]);
      bf16x8 b;
#pragma unroll
      for (int j = 0; j < 8; ++j) b[j] = (__bf16)((float)q[j] * s);
      acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc0, 0, 0, 0);
    }
  } else {
    // bf16 path: same LDS x-staging as the packed int8 path (direct
    // fragment-shaped x reads touch 16 scattered rows per instruction).
    constexpr int XT = 1024;
    constexpr int SE = XT + 8;
    __shared__ short x_lds_b[16 * SE];
    const short* wrow = (const short*)w + (int64_t)n_w * K;
    for (int k0 = c_begin * 32; k0 < c_end * 32; k0 += XT) {
      const int tk = min(XT, c_end * 32 - k0);
      __syncthreads();
      for (int idx = threadIdx.x; idx < 16 * (tk / 8); idx += 256) {
        const int r = idx / (tk / 8);
        const int vec = idx % (tk / 8);
        *reinterpret_cast<short8*>(&x_lds_b[r * SE + vec * 8]) =
            *reinterpret_cast<const short8*>(
                &x[(int64_t)min(r, M - 1) * K + k0 + vec * 8]);
      }
      __syncthreads();
      const short* arow = &x_lds_b[row * SE];
      int cl = 0;
      const int cl_end = tk / 32;
      for (; cl + 4 <= cl_end; cl += 4) {
        bf16x8 a[4], b[4];
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          a[u] = *reinterpret_cast<const bf16x8*>(&arow[(cl + u) * 32 + ks]);
          b[u] = *reinterpret_cast<const bf16x8*>(&wrow[k0 + (cl + u) * 32 + ks]);
        }
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          if (u & 1)
            acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[u], b[u], acc1, 0, 0, 0);
          else
            acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[u], b[u], acc0, 0, 0, 0);
        }
      }
      for (; cl < cl_end; ++cl) {
        const bf16x8 a = *reinterpret_cast<const bf16x8*>(&arow[cl * 32 + ks]);
        const bf16x8 b = *reinterpret_cast<const bf16x8*>(&wrow[k0 + cl * 32 + ks]);
        acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc0, 0, 0, 0);
      }
    }
  }
  const f32x4 acc = acc0 + acc1;

  // C write: lane covers col = n0 + (lane&15), rows (lane>>4)*4 + 0..3.
  const int n = n0 + (lane & 15);
  if (n >= N) return;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int m = (lane >> 4) * 4 + r;
    if (m >= M) continue;
    if (splitk > 1) {
      atomicAdd(out_f32 + (int64_t)m * N + n, acc[r]);
    } else {
      float v = acc[r];
      if (bias != nullptr) v += bits2f(bias[n]);
      out[(int64_t)m * N + n] = f2bits(v);
    }
  }
}

__global__ void f32_to_bf16_bias_kernel(const float* __restrict__ in,
                                        const short* __restrict__ bias,
                                        short* __restrict__ out,
                                        const int64_t total, const int N) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    float v = in[i];
    if (bias != nullptr) v += bits2f(bias[i % N]);
    out[i] = f2bits(v);
  }
}

static int pick_splitk(int64_t N, int64_t K, int64_t M, int64_t scratch_elems) {
  // Target ~1024 blocks (~4 blocks / 16 waves per CU) so HBM latency is
  // covered by wave overlap; the last split absorbs any chunk remainder.
  // Split-K needs the f32 scratch to hold the [M, N] accumulator, and each
  // split should keep >= 8 K-chunks of work.
  const int blocks = (int)((N + 63) / 64);
  int sk = 1;
  while (sk < 32 && blocks * sk < 1024 && (K / 32) / (sk * 2) >= 8) sk *= 2;
  if (sk > 1 && M * N > scratch_elems) sk = 1;
  return sk;
}

// out must be [M<=16, N]; scratch_f32 (nullable) must be [M, N] f32 if
// splitk>1 would be chosen (the wrapper supplies it).
static void launch_m16(torch::Tensor x, torch::Tensor w,
                       c10::optional<torch::Tensor> scales,
                       c10::optional<torch::Tensor> bias, torch::Tensor out,
                       c10::optional<torch::Tensor> scratch, int group,
                       int64_t m0, int M, bool packed) {
  const int64_t K = x.size(1), N = w.size(0);
  auto stream = current_stream();
  const bool quant = scales.has_value();
  const int64_t scratch_elems = scratch.has_value() ? scratch->numel() : 0;
  const int sk = pick_splitk(N, K, M, scratch_elems);
  const short* bptr = bias.has_value() ? (const short*)bias->data_ptr() : nullptr;
  const dim3 grid((unsigned)((N + 63) / 64), sk);
  const short* xp = (const short*)x.data_ptr() + m0 * K;
  short* op = (short*)out.data_ptr() + m0 * N;
  float* fp = nullptr;
  if (sk > 1) {
    TORCH_CHECK(scratch.has_value() && scratch->numel() >= (int64_t)M * N,
                "split-k scratch required");
    fp = (float*)scratch->data_ptr();
    DNET_CHECK_HIP(hipMemsetAsync(fp, 0, sizeof(float) * M * N, stream));
  }
  if (quant && packed) {
    hipLaunchKernelGGL((gemm_m16_kernel<true, true>), grid, dim3(256), 0,
                       stream, xp, w.data_ptr(), (const short*)scales->data_ptr(),
                       sk > 1 ? nullptr : bptr, op, fp, M, (int)K, (int)N,
                       group, sk);
  } else if (quant) {
    hipLaunchKernelGGL((gemm_m16_kernel<true, false>), grid, dim3(256), 0,
                       stream, xp, w.data_ptr(), (const short*)scales->data_ptr(),
                       sk > 1 ? nullptr : bptr, op, fp, M, (int)K, (int)N,
                       group, sk);
  } else {
    hipLaunchKernelGGL((gemm_m16_kernel<false, false>), grid, dim3(256), 0,
                       stream, xp, w.data_ptr(), nullptr,
                       sk > 1 ? nullptr : bptr, op, fp, M, (int)K, (int)N,
                       group, sk);
  }
  if (sk > 1) {
    const int64_t total = (int64_t)M * N;
    const int cgrid = (int)std::min<int64_t>((total + 255) / 256, 2048);
    hipLaunchKernelGGL(f32_to_bf16_bias_kernel, dim3(cgrid), dim3(256), 0,
                       stream, fp, bptr, op, total, (int)N);
  }
}

void gemm_m16(torch::Tensor x, torch::Tensor w,
              c10::optional<torch::Tensor> scales,
              c10::optional<torch::Tensor> bias, torch::Tensor out,
              c10::optional<torch::Tensor> scratch, int64_t group,
              bool packed) {
  const int64_t M = x.size(0), K = x.size(1), N = w.size(0);
  DNET_CHECK(K % 32 == 0, "K % 32 == 0 required for the MFMA path");
  DNET_CHECK(w.size(1) == K && out.size(0) == M && out.size(1) == N, "shape");
  DNET_CHECK(x.is_contiguous() && w.is_contiguous() && out.is_contiguous(), "contig");
  if (scales.has_value()) {
    DNET_CHECK(group % 8 == 0 && K % group == 0, "group align");
    DNET_CHECK(scales->is_contiguous(), "scales contig");
    if (packed) DNET_CHECK(K % 64 == 0 && group % 64 == 0, "packed layout align");
  }
  int64_t m0 = 0;
  while (m0 < M) {
    const int mt = (int)std::min<int64_t>(M - m0, 16);
    launch_m16(x, w, scales, bias, out, scratch, (int)group, m0, mt, packed);
    m0 += mt;
  }
}

}  // namespace dnet
