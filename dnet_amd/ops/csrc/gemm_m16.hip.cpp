// MFMA decode GEMM: out[M,N] = x[M,K] @ W[N,K]^T (+bias), M <= 64 per pass.
//
// The scalar GEMV (gemv.hip.cpp) is VALU-bound at M>=4: every weight element
// costs ~2 VALU ops per output row. Here one v_mfma_f32_16x16x32_bf16
// covers a 16(M)x16(N)x32(K) tile; MT in {1,2,4} stacks up to 4 M-tiles per
// wave so ONE weight read (and for int8 ONE in-register dequant) serves up
// to 64 batch rows — decode batching at constant weight traffic. Fragment
// layout for 16x16x32 (cdna_hip_programming.md §3): lane l holds
// A[row=l&15][k=(l>>4)*8+j], B[col=l&15][k=(l>>4)*8+j],
// C[col=l&15][row=(l>>4)*4+r].
//
// x is staged tile-wise into LDS with coalesced full-line loads (direct
// fragment-shaped x reads touch 16 scattered rows per instruction, up to
// +45% — guide §5 M=256 GEMM row). int8 weights use the chunk-pair packed
// layout (pack_int8_mfma) so each lane loads 16 B contiguous (full 64 B
// HBM bursts).
//
// Wave tile: 16 N-columns; block = 4 waves = 64 columns; gridDim.y = SPLITK
// K-splits (partials combined by f32 atomics + a convert/bias kernel).
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

// 4 packed nibble-bytes (offset-8) -> 8 bf16 values * scale
__device__ __forceinline__ bf16x8 deq4(const uint8_t* qb, const float s) {
  bf16x8 b;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int byte = qb[j];
    b[2 * j] = (__bf16)((float)((byte & 0xF) - 8) * s);
    b[2 * j + 1] = (__bf16)((float)((byte >> 4) - 8) * s);
  }
  return b;
}

// QBITS: 16 = bf16 weights, 8 = packed int8 (pack_int8_mfma chunk-pair
// layout), 4 = packed int4 (pack_int4_mfma chunk-quad layout).
// MT: number of stacked 16-row A tiles (M <= 16*MT).
template <int QBITS, int MT>
__global__ void gemm_m16_kernel(const short* __restrict__ x,
                                const void* __restrict__ w,
                                const short* __restrict__ scales,
                                const short* __restrict__ bias,
                                short* __restrict__ out,        // SPLITK==1
                                float* __restrict__ out_f32,    // SPLITK>1
                                const int M, const int K, const int N,
                                const int G, const int splitk) {
  constexpr int XT = 1024 / MT;        // k per x tile (LDS ~33 KB, 4 blocks/CU)
  constexpr int SE = XT + 8;           // row stride in bf16 elems (16B pad)
  __shared__ short x_lds[16 * MT * SE];
  const int wave = threadIdx.x / kWave;
  const int lane = threadIdx.x & (kWave - 1);
  const int n0 = (blockIdx.x * 4 + wave) * 16;
  // Block-wide barriers: OOB waves must stay alive; their stores are
  // skipped at the end.
  const int row = lane & 15;          // A row within a tile, B col
  const int ks = (lane >> 4) * 8;     // k-offset of this lane's 8-elem slice
  const int n_w = min(n0 + row, N - 1);     // this lane's W row

  const int pairs = K / 64;
  int p_begin, p_end;
  if (QBITS == 4) {
    // partition at quad (128-k) granularity so every split starts on an
    // even pair (the int4 layout packs four chunks per lane load)
    const int quads = K / 128;
    const int qq = quads / splitk;
    p_begin = blockIdx.y * qq * 2;
    p_end = (blockIdx.y == splitk - 1) ? quads * 2 : p_begin + qq * 2;
  } else {
    const int pp = pairs / splitk;
    p_begin = blockIdx.y * pp;
    p_end = (blockIdx.y == splitk - 1) ? pairs : p_begin + pp;
  }
  const int woff = (lane >> 4) * 16;

  f32x4 acc[MT][2];
#pragma unroll
  for (int t = 0; t < MT; ++t)
#pragma unroll
    for (int u = 0; u < 2; ++u) acc[t][u] = {0.f, 0.f, 0.f, 0.f};

  const int8_t* wrow_q = (const int8_t*)w + (int64_t)n_w * K;
  const uint8_t* wrow_q4 = (const uint8_t*)w + (int64_t)n_w * (K / 2);
  const short* wrow_b = (const short*)w + (int64_t)n_w * K;
  const short* srow = QBITS < 16 ? scales + (int64_t)n_w * (K / G) : nullptr;

  for (int k0 = p_begin * 64; k0 < p_end * 64; k0 += XT) {
    const int tk = min(XT, p_end * 64 - k0);
    // raw barrier (lgkmcnt only): __syncthreads() would wait vmcnt(0) and
    // drain the in-flight weight stream at EVERY x-tile boundary — at MT=4
    // (XT=256) that is a full memory-pipeline restart every 4 pairs
    // (guide §6: counted waits, not vmcnt(0), across barriers)
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    asm volatile("s_barrier" ::: "memory");
    for (int idx = threadIdx.x; idx < 16 * MT * (tk / 8); idx += 256) {
      const int r = idx / (tk / 8);
      const int vec = idx % (tk / 8);
      // rows >= M read a clamped row; their C rows are never stored
      *reinterpret_cast<short8*>(&x_lds[r * SE + vec * 8]) =
          *reinterpret_cast<const short8*>(
              &x[(int64_t)min(r, M - 1) * K + k0 + vec * 8]);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    asm volatile("s_barrier" ::: "memory");
    const short* arow = &x_lds[row * SE];
    int pl = 0;
    const int pl_end = tk / 64;
    for (; pl + 4 <= pl_end; pl += 4) {  // 4 pairs, all weight loads first
      int4 wq8[4];
      int4 wq4[2];
      bf16x8 wb[4][2];
      float sv[4];
      if (QBITS == 8) {
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          const int p = (k0 / 64) + pl + u;
          wq8[u] = *reinterpret_cast<const int4*>(&wrow_q[p * 64 + woff]);
          sv[u] = bits2f(srow[(p * 64) / G]);
        }
      } else if (QBITS == 4) {
#pragma unroll
        for (int v = 0; v < 2; ++v) {
          const int quad = ((k0 / 64) + pl) / 2 + v;
          wq4[v] = *reinterpret_cast<const int4*>(&wrow_q4[quad * 64 + woff]);
          sv[v] = bits2f(srow[(quad * 128) / G]);
        }
      } else {
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          const int p = (k0 / 64) + pl + u;
          wb[u][0] = *reinterpret_cast<const bf16x8*>(&wrow_b[p * 64 + ks]);
          wb[u][1] = *reinterpret_cast<const bf16x8*>(
              &wrow_b[p * 64 + 32 + ks]);
        }
      }
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        bf16x8 b0, b1;
        if (QBITS == 8) {
          const int8_t* q8 = reinterpret_cast<const int8_t*>(&wq8[u]);
          b0 = deq8(q8, sv[u]);
          b1 = deq8(q8 + 8, sv[u]);
        } else if (QBITS == 4) {
          const uint8_t* q4 =
              reinterpret_cast<const uint8_t*>(&wq4[u / 2]) + (u & 1) * 8;
          b0 = deq4(q4, sv[u / 2]);
          b1 = deq4(q4 + 4, sv[u / 2]);
        } else {
          b0 = wb[u][0];
          b1 = wb[u][1];
        }
#pragma unroll
        for (int t = 0; t < MT; ++t) {
          const short* at = arow + t * 16 * SE;
          const bf16x8 a0 =
              *reinterpret_cast<const bf16x8*>(&at[(pl + u) * 64 + ks]);
          const bf16x8 a1 =
              *reinterpret_cast<const bf16x8*>(&at[(pl + u) * 64 + 32 + ks]);
          acc[t][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a0, b0, acc[t][0], 0, 0, 0);
          acc[t][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a1, b1, acc[t][1], 0, 0, 0);
        }
      }
    }
    for (; pl < pl_end; ++pl) {
      const int p = (k0 / 64) + pl;
      bf16x8 b0, b1;
      if (QBITS == 8) {
        const int4 wq = *reinterpret_cast<const int4*>(&wrow_q[p * 64 + woff]);
        const int8_t* q8 = reinterpret_cast<const int8_t*>(&wq);
        const float sv = bits2f(srow[(p * 64) / G]);
        b0 = deq8(q8, sv);
        b1 = deq8(q8 + 8, sv);
      } else if (QBITS == 4) {
        const int quad = p / 2;
        const int half = p & 1;
        const int2 wq = *reinterpret_cast<const int2*>(
            &wrow_q4[quad * 64 + woff + half * 8]);
        const uint8_t* q4 = reinterpret_cast<const uint8_t*>(&wq);
        const float sv = bits2f(srow[(quad * 128) / G]);
        b0 = deq4(q4, sv);
        b1 = deq4(q4 + 4, sv);
      } else {
        b0 = *reinterpret_cast<const bf16x8*>(&wrow_b[p * 64 + ks]);
        b1 = *reinterpret_cast<const bf16x8*>(&wrow_b[p * 64 + 32 + ks]);
      }
#pragma unroll
      for (int t = 0; t < MT; ++t) {
        const short* at = arow + t * 16 * SE;
        const bf16x8 a0 = *reinterpret_cast<const bf16x8*>(&at[pl * 64 + ks]);
        const bf16x8 a1 = *reinterpret_cast<const bf16x8*>(&at[pl * 64 + 32 + ks]);
        acc[t][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[t][0],
                                                            0, 0, 0);
        acc[t][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc[t][1],
                                                            0, 0, 0);
      }
    }
  }

  // C write: lane covers col = n0 + (lane&15), rows (lane>>4)*4 + 0..3 of
  // each stacked tile.
  const int n = n0 + (lane & 15);
  if (n >= N) return;
#pragma unroll
  for (int t = 0; t < MT; ++t) {
    const f32x4 a2 = acc[t][0] + acc[t][1];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = t * 16 + (lane >> 4) * 4 + r;
      if (m >= M) continue;
      if (splitk > 1) {
        atomicAdd(out_f32 + (int64_t)m * N + n, a2[r]);
      } else {
        float v = a2[r];
        if (bias != nullptr) v += bits2f(bias[n]);
        out[(int64_t)m * N + n] = f2bits(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Streamed int8/int4 schedule (round-2).
//
// The generic kernel above is pair-rate bound on quantized weights: PMC
// showed SQ_WAIT 69% with int8 wall time == bf16 wall time at every shape
// (profiles/r01_gemm_ubench.md) — hipcc keeps only ~2 weight loads in
// flight and serializes the x staging as load;vmcnt(0);ds_write per chunk
// (one full round trip per 16-byte chunk). Per 256-k x-tile this schedule
// instead takes ONE combined round trip:
//
//   raw s_barrier (lgkmcnt only — previous tile's consumers done)
//   issue x staging as global_load_lds DMA (no VGPR round trip)
//   ONE asm block: issue the tile's weight+scale loads, s_waitcnt vmcnt(0)
//   raw s_barrier
//   dequant + MFMA from registers (no VMEM waits at all)
//
// The weight loads and the staging DMA are all in flight together
// (10-14 VMEM ops, 5-9 KB per lane-group), so their latencies overlap in
// one drain; wave overlap (4 blocks x 4 waves / CU) covers the per-wave
// consume phase. Every async VMEM value is produced AND waited for
// INSIDE a single asm statement: hipcc models asm outputs as ready when
// the statement ends, so an async result that escapes a statement before
// its s_waitcnt can be copied/rematerialized from an in-flight register
// (measured: garbage numerics and corrupted-address page faults from the
// earlier cross-tile ping-pong prefetch). Outputs are early-clobber so
// the in-flight destinations can never alias the address operands.
//
// glds writes LDS at (wave-uniform base + lane*16), so the LDS x image is
// UNPADDED (16*MT rows x 32 16B-chunks); bank conflicts on the 16-lane
// fragment reads are killed by an XOR swizzle applied on the SOURCE
// address instead of row padding (guide §5 rule 21): source chunk c lands
// at image chunk c ^ (row & 15) — the 16 reader lanes (rows 0..15, same
// c) then touch 16 different 16B slots = all 64 LDS banks.
// ---------------------------------------------------------------------------

using i32x4 = __attribute__((ext_vector_type(4))) int;

// NSC = scales per 256-k tile = 256/G (G compile-time via dispatch; the
// host streams only G in {64,128}, and requires the split start k to be
// G-aligned so the in-tile scale index u*NSC/4 is exact).
template <int QBITS, int NSC>
struct WSet {
  i32x4 w[QBITS == 8 ? 4 : 2];
  unsigned s[NSC];
};

// Load one tile's weights+scales and drain: single asm statement (see
// header comment). wb already includes the lane's row offset
// (n_w*K + woff) plus the tile's k byte offset; sb points at the tile's
// first scale.
template <int QBITS, int NSC>
__device__ __forceinline__ void wset_load(WSet<QBITS, NSC>& o,
                                          const void* wb, const short* sb) {
  if constexpr (QBITS == 8 && NSC == 2) {
    asm volatile(
        "global_load_dwordx4 %0, %6, off\n\t"
        "global_load_dwordx4 %1, %6, off offset:64\n\t"
        "global_load_dwordx4 %2, %6, off offset:128\n\t"
        "global_load_dwordx4 %3, %6, off offset:192\n\t"
        "global_load_ushort %4, %7, off\n\t"
        "global_load_ushort %5, %7, off offset:2\n\t"
        "s_waitcnt vmcnt(0)"
        : "=&v"(o.w[0]), "=&v"(o.w[1]), "=&v"(o.w[2]), "=&v"(o.w[3]),
          "=&v"(o.s[0]), "=&v"(o.s[1])
        : "v"(wb), "v"(sb)
        : "memory");
  } else if constexpr (QBITS == 8 && NSC == 4) {
    asm volatile(
        "global_load_dwordx4 %0, %8, off\n\t"
        "global_load_dwordx4 %1, %8, off offset:64\n\t"
        "global_load_dwordx4 %2, %8, off offset:128\n\t"
        "global_load_dwordx4 %3, %8, off offset:192\n\t"
        "global_load_ushort %4, %9, off\n\t"
        "global_load_ushort %5, %9, off offset:2\n\t"
        "global_load_ushort %6, %9, off offset:4\n\t"
        "global_load_ushort %7, %9, off offset:6\n\t"
        "s_waitcnt vmcnt(0)"
        : "=&v"(o.w[0]), "=&v"(o.w[1]), "=&v"(o.w[2]), "=&v"(o.w[3]),
          "=&v"(o.s[0]), "=&v"(o.s[1]), "=&v"(o.s[2]), "=&v"(o.s[3])
        : "v"(wb), "v"(sb)
        : "memory");
  } else {
    static_assert(QBITS == 4 && NSC == 2, "unsupported stream variant");
    asm volatile(
        "global_load_dwordx4 %0, %4, off\n\t"
        "global_load_dwordx4 %1, %4, off offset:64\n\t"
        "global_load_ushort %2, %5, off\n\t"
        "global_load_ushort %3, %5, off offset:2\n\t"
        "s_waitcnt vmcnt(0)"
        : "=&v"(o.w[0]), "=&v"(o.w[1]), "=&v"(o.s[0]), "=&v"(o.s[1])
        : "v"(wb), "v"(sb)
        : "memory");
  }
}

// 2-tile weight batch: one statement issues BOTH tiles' weights (+ the
// first tile's staging already in flight) and drains once — halves the
// exposed weight round trips vs per-tile loads. Values are final at the
// statement end, so holding tile B's registers across tile A's consume
// and tile B's staging is safe (only IN-FLIGHT values must not escape).
template <int QBITS, int NSC>
struct WSet2 {
  i32x4 w[QBITS == 8 ? 8 : 4];
  unsigned s[2 * NSC];
};

template <int QBITS, int NSC>
__device__ __forceinline__ void wset_load2(WSet2<QBITS, NSC>& o,
                                           const void* wb, const short* sb) {
  if constexpr (QBITS == 8 && NSC == 2) {
    asm volatile(
        "global_load_dwordx4 %0, %12, off\n\t"
        "global_load_dwordx4 %1, %12, off offset:64\n\t"
        "global_load_dwordx4 %2, %12, off offset:128\n\t"
        "global_load_dwordx4 %3, %12, off offset:192\n\t"
        "global_load_dwordx4 %4, %12, off offset:256\n\t"
        "global_load_dwordx4 %5, %12, off offset:320\n\t"
        "global_load_dwordx4 %6, %12, off offset:384\n\t"
        "global_load_dwordx4 %7, %12, off offset:448\n\t"
        "global_load_ushort %8, %13, off\n\t"
        "global_load_ushort %9, %13, off offset:2\n\t"
        "global_load_ushort %10, %13, off offset:4\n\t"
        "global_load_ushort %11, %13, off offset:6\n\t"
        "s_waitcnt vmcnt(0)"
        : "=&v"(o.w[0]), "=&v"(o.w[1]), "=&v"(o.w[2]), "=&v"(o.w[3]),
          "=&v"(o.w[4]), "=&v"(o.w[5]), "=&v"(o.w[6]), "=&v"(o.w[7]),
          "=&v"(o.s[0]), "=&v"(o.s[1]), "=&v"(o.s[2]), "=&v"(o.s[3])
        : "v"(wb), "v"(sb)
        : "memory");
  } else if constexpr (QBITS == 8 && NSC == 4) {
    asm volatile(
        "global_load_dwordx4 %0, %16, off\n\t"
        "global_load_dwordx4 %1, %16, off offset:64\n\t"
        "global_load_dwordx4 %2, %16, off offset:128\n\t"
        "global_load_dwordx4 %3, %16, off offset:192\n\t"
        "global_load_dwordx4 %4, %16, off offset:256\n\t"
        "global_load_dwordx4 %5, %16, off offset:320\n\t"
        "global_load_dwordx4 %6, %16, off offset:384\n\t"
        "global_load_dwordx4 %7, %16, off offset:448\n\t"
        "global_load_ushort %8, %17, off\n\t"
        "global_load_ushort %9, %17, off offset:2\n\t"
        "global_load_ushort %10, %17, off offset:4\n\t"
        "global_load_ushort %11, %17, off offset:6\n\t"
        "global_load_ushort %12, %17, off offset:8\n\t"
        "global_load_ushort %13, %17, off offset:10\n\t"
        "global_load_ushort %14, %17, off offset:12\n\t"
        "global_load_ushort %15, %17, off offset:14\n\t"
        "s_waitcnt vmcnt(0)"
        : "=&v"(o.w[0]), "=&v"(o.w[1]), "=&v"(o.w[2]), "=&v"(o.w[3]),
          "=&v"(o.w[4]), "=&v"(o.w[5]), "=&v"(o.w[6]), "=&v"(o.w[7]),
          "=&v"(o.s[0]), "=&v"(o.s[1]), "=&v"(o.s[2]), "=&v"(o.s[3]),
          "=&v"(o.s[4]), "=&v"(o.s[5]), "=&v"(o.s[6]), "=&v"(o.s[7])
        : "v"(wb), "v"(sb)
        : "memory");
  } else {
    static_assert(QBITS == 4 && NSC == 2, "unsupported stream variant");
    asm volatile(
        "global_load_dwordx4 %0, %8, off\n\t"
        "global_load_dwordx4 %1, %8, off offset:64\n\t"
        "global_load_dwordx4 %2, %8, off offset:128\n\t"
        "global_load_dwordx4 %3, %8, off offset:192\n\t"
        "global_load_ushort %4, %9, off\n\t"
        "global_load_ushort %5, %9, off offset:2\n\t"
        "global_load_ushort %6, %9, off offset:4\n\t"
        "global_load_ushort %7, %9, off offset:6\n\t"
        "s_waitcnt vmcnt(0)"
        : "=&v"(o.w[0]), "=&v"(o.w[1]), "=&v"(o.w[2]), "=&v"(o.w[3]),
          "=&v"(o.s[0]), "=&v"(o.s[1]), "=&v"(o.s[2]), "=&v"(o.s[3])
        : "v"(wb), "v"(sb)
        : "memory");
  }
}

template <int MT, bool GLDS>
__device__ __forceinline__ void stage_tile(const short* __restrict__ x,
                                           short* x_lds, const int k0,
                                           const int M, const int K) {
  constexpr int NSG = MT * 2;  // glds per wave (1 KB each)
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  asm volatile("s_barrier" ::: "memory");
  const int wavei = (int)threadIdx.x >> 6, lanei = (int)threadIdx.x & 63;
  if constexpr (GLDS) {
#pragma unroll
    for (int g = 0; g < NSG; ++g) {
      const int L = (wavei * NSG + g) * 64 + lanei;  // image chunk
      const int r = L >> 5;
      const int cs = (L & 31) ^ (r & 15);            // source chunk (swizzle)
      const short* ga = &x[(int64_t)min(r, M - 1) * K + k0 + cs * 8];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)ga,
          (__attribute__((address_space(3))) void*)&x_lds[(wavei * NSG + g) *
                                                          64 * 8],
          16, 0, 0);
    }
  } else {
    // debug/fallback: same swizzled image via plain loads + ds_write
#pragma unroll
    for (int g = 0; g < NSG; ++g) {
      const int L = (wavei * NSG + g) * 64 + lanei;
      const int r = L >> 5;
      const int cs = (L & 31) ^ (r & 15);
      *reinterpret_cast<short8*>(&x_lds[L * 8]) =
          *reinterpret_cast<const short8*>(
              &x[(int64_t)min(r, M - 1) * K + k0 + cs * 8]);
    }
  }
}

template <bool GLDS>
__device__ __forceinline__ void end_stage() {
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  asm volatile("s_barrier" ::: "memory");
  if constexpr (GLDS)  // DS ops bound-check against m0 on gfx9-family
    asm volatile("s_mov_b32 m0, -1" ::: "memory");
}

// dequant + MFMA for one tile's 4 pairs from already-final registers
template <int QBITS, int MT, int NSC>
__device__ __forceinline__ void consume_tile(const i32x4* __restrict__ wv,
                                             const unsigned* __restrict__ sc,
                                             short* x_lds,
                                             f32x4 (&acc)[MT][2],
                                             const int row, const int ks) {
  const int ln4 = ks >> 3;  // lane>>4
#pragma unroll
  for (int u = 0; u < 4; ++u) {
    bf16x8 b0, b1;
    if (QBITS == 8) {
      const int8_t* q8 = reinterpret_cast<const int8_t*>(&wv[u]);
      const float sv = bits2f((short)sc[u * NSC / 4]);
      b0 = deq8(q8, sv);
      b1 = deq8(q8 + 8, sv);
    } else {
      const uint8_t* q4 =
          reinterpret_cast<const uint8_t*>(&wv[u / 2]) + (u & 1) * 8;
      const float sv = bits2f((short)sc[(u / 2) * NSC / 2]);
      b0 = deq4(q4, sv);
      b1 = deq4(q4 + 4, sv);
    }
    const int c0 = (u * 8 + ln4) ^ row;      // swizzled image chunks
    const int c1 = (u * 8 + 4 + ln4) ^ row;
#pragma unroll
    for (int t = 0; t < MT; ++t) {
      const short* rb = &x_lds[(t * 16 + row) * 256];
      const bf16x8 a0 = *reinterpret_cast<const bf16x8*>(&rb[c0 * 8]);
      const bf16x8 a1 = *reinterpret_cast<const bf16x8*>(&rb[c1 * 8]);
      acc[t][0] =
          __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[t][0], 0, 0, 0);
      acc[t][1] =
          __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc[t][1], 0, 0, 0);
    }
  }
}

// MINW: min waves per EU hint (occupancy/VGPR trade, see profiles/).
template <int QBITS, int MT, int NSC, int MINW, bool GLDS = true>
__global__ __launch_bounds__(256, MINW) void gemm_m16_stream_kernel(
    const short* __restrict__ x, const void* __restrict__ w,
    const short* __restrict__ scales, const short* __restrict__ bias,
    short* __restrict__ out, float* __restrict__ out_f32, const int M,
    const int K, const int N, const int G, const int splitk) {
  constexpr int XT = 256;
  __shared__ short x_lds[16 * MT * XT];  // unpadded (glds image, swizzled)
  const int wave = threadIdx.x / kWave;
  const int lane = threadIdx.x & (kWave - 1);
  const int n0 = (blockIdx.x * 4 + wave) * 16;
  const int row = lane & 15;
  const int ks = (lane >> 4) * 8;
  const int n_w = min(n0 + row, N - 1);

  const int pairs = K / 64;
  int p_begin, p_end;
  if (QBITS == 4) {
    const int quads = K / 128;
    const int qq = quads / splitk;
    p_begin = blockIdx.y * qq * 2;
    p_end = (blockIdx.y == splitk - 1) ? quads * 2 : p_begin + qq * 2;
  } else {
    const int pp = pairs / splitk;
    p_begin = blockIdx.y * pp;
    p_end = (blockIdx.y == splitk - 1) ? pairs : p_begin + pp;
  }
  const int woff = (lane >> 4) * 16;

  f32x4 acc[MT][2];
#pragma unroll
  for (int t = 0; t < MT; ++t)
#pragma unroll
    for (int u = 0; u < 2; ++u) acc[t][u] = {0.f, 0.f, 0.f, 0.f};

  const void* wrow =
      QBITS == 8
          ? (const void*)((const int8_t*)w + (int64_t)n_w * K + woff)
          : (const void*)((const uint8_t*)w + (int64_t)n_w * (K / 2) + woff);
  const short* srow = scales + (int64_t)n_w * (K / G);

  const int kbeg = p_begin * 64, kend = p_end * 64;
  const int kfull = kbeg + ((kend - kbeg) / XT) * XT;
  constexpr int GS = 256 / NSC;
  int k0 = kbeg;
  // 2-tile groups: one weight statement serves both tiles (one exposed
  // round trip instead of two); tile B's staging waits alone
  for (; k0 + 2 * XT <= kfull; k0 += 2 * XT) {
    WSet2<QBITS, NSC> w2;
    stage_tile<MT, GLDS>(x, x_lds, k0, M, K);
    if (QBITS == 8)
      wset_load2<QBITS, NSC>(w2, (const int8_t*)wrow + k0, srow + k0 / GS);
    else
      wset_load2<QBITS, NSC>(w2, (const uint8_t*)wrow + k0 / 2,
                             srow + k0 / GS);
    end_stage<GLDS>();
    consume_tile<QBITS, MT, NSC>(&w2.w[0], &w2.s[0], x_lds, acc, row, ks);
    stage_tile<MT, GLDS>(x, x_lds, k0 + XT, M, K);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // staging only
    end_stage<GLDS>();
    consume_tile<QBITS, MT, NSC>(&w2.w[QBITS == 8 ? 4 : 2], &w2.s[NSC],
                                 x_lds, acc, row, ks);
  }
  for (; k0 < kfull; k0 += XT) {
    WSet<QBITS, NSC> w1;
    stage_tile<MT, GLDS>(x, x_lds, k0, M, K);
    if (QBITS == 8)
      wset_load<QBITS, NSC>(w1, (const int8_t*)wrow + k0, srow + k0 / GS);
    else
      wset_load<QBITS, NSC>(w1, (const uint8_t*)wrow + k0 / 2,
                            srow + k0 / GS);
    end_stage<GLDS>();
    consume_tile<QBITS, MT, NSC>(&w1.w[0], &w1.s[0], x_lds, acc, row, ks);
  }

  // tail (< XT k): generic serial staging + pair loop, executed by the
  // whole block (condition uniform)
  const int ln4 = ks >> 3;
  for (; k0 < kend; k0 += XT) {
    const int tk = kend - k0;  // < XT, multiple of 64
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    asm volatile("s_barrier" ::: "memory");
    for (int idx = threadIdx.x; idx < 16 * MT * (tk / 8); idx += 256) {
      const int r = idx / (tk / 8);
      const int vec = idx % (tk / 8);
      // same swizzled image as the glds path (rows are full 32-chunk
      // strides even when tk < XT)
      *reinterpret_cast<short8*>(&x_lds[(r * 32 + (vec ^ (r & 15))) * 8]) =
          *reinterpret_cast<const short8*>(
              &x[(int64_t)min(r, M - 1) * K + k0 + vec * 8]);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    asm volatile("s_barrier" ::: "memory");
    for (int pl = 0; pl < tk / 64; ++pl) {
      const int p = (k0 / 64) + pl;
      bf16x8 b0, b1;
      if (QBITS == 8) {
        const int8_t* wq = (const int8_t*)wrow;
        const int4 wv = *reinterpret_cast<const int4*>(&wq[p * 64]);
        const int8_t* q8 = reinterpret_cast<const int8_t*>(&wv);
        const float sv = bits2f(srow[(p * 64) / G]);
        b0 = deq8(q8, sv);
        b1 = deq8(q8 + 8, sv);
      } else {
        const int quad = p / 2;
        const int half = p & 1;
        const uint8_t* wq = (const uint8_t*)wrow;
        const int2 wv =
            *reinterpret_cast<const int2*>(&wq[quad * 64 + half * 8]);
        const uint8_t* q4 = reinterpret_cast<const uint8_t*>(&wv);
        const float sv = bits2f(srow[(quad * 128) / G]);
        b0 = deq4(q4, sv);
        b1 = deq4(q4 + 4, sv);
      }
      const int c0 = (pl * 8 + ln4) ^ row;
      const int c1 = (pl * 8 + 4 + ln4) ^ row;
#pragma unroll
      for (int t = 0; t < MT; ++t) {
        const short* rb = &x_lds[(t * 16 + row) * 256];
        const bf16x8 a0 = *reinterpret_cast<const bf16x8*>(&rb[c0 * 8]);
        const bf16x8 a1 = *reinterpret_cast<const bf16x8*>(&rb[c1 * 8]);
        acc[t][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[t][0],
                                                            0, 0, 0);
        acc[t][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc[t][1],
                                                            0, 0, 0);
      }
    }
  }

  const int n = n0 + (lane & 15);
  if (n >= N) return;
#pragma unroll
  for (int t = 0; t < MT; ++t) {
    const f32x4 a2 = acc[t][0] + acc[t][1];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = t * 16 + (lane >> 4) * 4 + r;
      if (m >= M) continue;
      if (splitk > 1) {
        atomicAdd(out_f32 + (int64_t)m * N + n, a2[r]);
      } else {
        float v = a2[r];
        if (bias != nullptr) v += bits2f(bias[n]);
        out[(int64_t)m * N + n] = f2bits(v);
      }
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

// One device per process (shard design): number of leading scratch
// elements holding un-zeroed partials from plain (non-deferred)
// combines. A bool is NOT enough: a small GEMM's memset after a big
// combine would clear the flag while stale partials remain beyond its
// own M*N region and corrupt the next bigger user.
static int64_t g_scratch_dirty_elems = 0;

static int pick_splitk(int64_t N, int64_t K, int64_t M, int64_t scratch_elems,
                       int min_pairs, int g_align) {
  // Target ~1024 blocks (~4 blocks / 16 waves per CU) so HBM latency is
  // covered by wave overlap; each split keeps >= 8 K-chunk-pairs of work.
  static const int forced = []() {
    const char* e = getenv("DNET_GEMM_SPLITK");  // debug/bench override
    return e ? atoi(e) : 0;
  }();
  const int blocks = (int)((N + 63) / 64);
  if (forced > 0)
    return (M * N > scratch_elems || (K / 64) < forced) ? 1 : forced;
  int sk = 1;
  while (sk < 32 && blocks * sk < 1024 &&
         (K / 64) / (sk * 2) >= min_pairs &&
         ((K / 64) / (sk * 2)) * 64 % g_align == 0)  // split starts G-aligned
    sk *= 2;
  // Measured (profiles/r02, sk sweep): a split span that is not a
  // multiple of 256 pushes its remainder through the serial tail path;
  // on SHORT spans that tail dominates (qkv K=5120 sk=8: span 640,
  // 20% tail, 27.7us vs 23.6us at sk=4). Halve sk when the tail is
  // >=15% of the span, the halved span is tail-free, and the halved
  // grid still covers the chip (o N=5120 shows 320 blocks is too few).
  if (sk > 1) {
    const int64_t span = ((K / 64) / sk) * 64;
    const int64_t span2 = ((K / 64) / (sk / 2)) * 64;
    if (span % 256 != 0 && (double)(span % 256) / span >= 0.15 &&
        span2 % 256 == 0 && span2 % g_align == 0 &&
        blocks * (sk / 2) >= 440)
      sk /= 2;
  }
  if (sk > 1 && M * N > scratch_elems) sk = 1;
  return sk;
}

static bool launch_m16(torch::Tensor x, torch::Tensor w,
                       c10::optional<torch::Tensor> scales,
                       c10::optional<torch::Tensor> bias, torch::Tensor out,
                       c10::optional<torch::Tensor> scratch, int group,
                       int64_t m0, int M, int bits, bool defer_combine) {
  const int64_t K = x.size(1), N = w.size(0);
  auto stream = current_stream();
  const int64_t scratch_elems = scratch.has_value() ? scratch->numel() : 0;
  // the streamed schedule tolerates short splits (one combined round
  // trip per 4-pair tile), so quantized paths split deeper to fill the
  // 256 CUs on small-N shapes (qkv/o run at <2 blocks/CU otherwise)
  const int sk = pick_splitk(N, K, M, scratch_elems, bits < 16 ? 4 : 8,
                             bits < 16 ? std::max<int>((int)group, 64) : 64);
  const short* bptr = bias.has_value() ? (const short*)bias->data_ptr() : nullptr;
  const dim3 grid((unsigned)((N + 63) / 64), sk);
  const short* xp = (const short*)x.data_ptr() + m0 * K;
  short* op = (short*)out.data_ptr() + m0 * N;
  float* fp = nullptr;
  if (sk > 1) {
    TORCH_CHECK(scratch.has_value() && scratch->numel() >= (int64_t)M * N,
                "split-k scratch required");
    fp = (float*)scratch->data_ptr();
    // The scratch must be zero before the atomics. Deferred-combine
    // consumers (rope/swiglu/rmsnorm f32 variants) re-zero what they
    // read, so a chain of deferred GEMMs never needs a memset; the
    // plain combine does NOT re-zero — a combine-side in[i]=0 write
    // measured a 3.5x MoE decode slowdown (the grouped expert kernels
    // dropped ~3.9 -> ~2.5 TB/s whenever the combine left dirtied
    // scratch lines behind; bisected to that single change) — so the
    // NEXT split-k launch after a combine memsets first.
    if (g_scratch_dirty_elems > 0) {
      const int64_t span = std::max<int64_t>(g_scratch_dirty_elems,
                                             (int64_t)M * N);
      DNET_CHECK_HIP(hipMemsetAsync(fp, 0, sizeof(float) * span, stream));
      g_scratch_dirty_elems = 0;
    }
  }
  const short* sp = bits < 16 ? (const short*)scales->data_ptr() : nullptr;
  const short* bp1 = sk > 1 ? nullptr : bptr;
#define LAUNCH(QQ, TT)                                                      \
  hipLaunchKernelGGL((gemm_m16_kernel<QQ, TT>), grid, dim3(256), 0, stream, \
                     xp, w.data_ptr(), sp, bp1, op, fp, M, (int)K, (int)N,  \
                     group, sk)
  static const bool mt4occ3 = []() {
    const char* e = getenv("DNET_GEMM_MT4OCC");
    return e == nullptr || e[0] == '3';  // default: spill-free 3-wave MT4
  }();
  static const bool noglds = getenv("DNET_GEMM_NOGLDS") != nullptr;
#define LAUNCH_STREAM(QQ, TT, NSC, MW)                                      \
  do {                                                                      \
    if (noglds)                                                             \
      hipLaunchKernelGGL((gemm_m16_stream_kernel<QQ, TT, NSC, MW, false>), \
                         grid, dim3(256), 0, stream, xp, w.data_ptr(), sp,  \
                         bp1, op, fp, M, (int)K, (int)N, group, sk);        \
    else                                                                    \
      hipLaunchKernelGGL((gemm_m16_stream_kernel<QQ, TT, NSC, MW, true>),   \
                         grid, dim3(256), 0, stream, xp, w.data_ptr(), sp,  \
                         bp1, op, fp, M, (int)K, (int)N, group, sk);        \
  } while (0)
#define LAUNCH_STREAM_G8(TT)                                            \
  do {                                                                    \
    if (TT == 4 && mt4occ3) {                                             \
      if (group == 64) LAUNCH_STREAM(8, TT, 4, 3);                        \
      else LAUNCH_STREAM(8, TT, 2, 3);                                    \
    } else {                                                              \
      if (group == 64) LAUNCH_STREAM(8, TT, 4, 4);                        \
      else LAUNCH_STREAM(8, TT, 2, 4);                                    \
    }                                                                     \
  } while (0)
#define LAUNCH_STREAM_G4(TT)                                              \
  do {                                                                    \
    if (TT == 4 && mt4occ3) LAUNCH_STREAM(4, TT, 2, 3);                   \
    else LAUNCH_STREAM(4, TT, 2, 4);                                      \
  } while (0)
  // the streamed counted-vmcnt schedule needs a compile-time scale count
  // per 256-k tile; other group sizes fall back to the generic kernel
  // G must divide every tile start (k0 is any multiple of 64 under
  // split-k partitioning, and the in-tile scale index u*NSC/4 assumes
  // k0 % G == 0), so only G <= 128 streams; G=256 falls back
  // additionally every split start kbeg = by*(pairs/sk)*64 must be
  // G-aligned (the in-tile scale index assumes k0 % G == 0)
  const bool split_aligned =
      group > 0 && (sk == 1 || ((K / 64 / sk) * 64) % group == 0);
  const bool can_stream =
      split_aligned && ((bits == 8 && (group == 64 || group == 128)) ||
                        (bits == 4 && group == 128));
  static const bool nostream = getenv("DNET_GEMM_NOSTREAM") != nullptr;
  const bool use_stream = can_stream && !nostream;
  if (bits == 8) {
    if (M > 32) { if (use_stream) LAUNCH_STREAM_G8(4); else LAUNCH(8, 4); }
    else if (M > 16) { if (use_stream) LAUNCH_STREAM_G8(2); else LAUNCH(8, 2); }
    else { if (use_stream) LAUNCH_STREAM_G8(1); else LAUNCH(8, 1); }
  } else if (bits == 4) {
    if (M > 32) { if (use_stream) LAUNCH_STREAM_G4(4); else LAUNCH(4, 4); }
    else if (M > 16) { if (use_stream) LAUNCH_STREAM_G4(2); else LAUNCH(4, 2); }
    else { if (use_stream) LAUNCH_STREAM_G4(1); else LAUNCH(4, 1); }
  } else {
    if (M > 32) LAUNCH(16, 4);
    else if (M > 16) LAUNCH(16, 2);
    else LAUNCH(16, 1);
  }
#undef LAUNCH
#undef LAUNCH_STREAM
#undef LAUNCH_STREAM_G8
#undef LAUNCH_STREAM_G4
  if (sk > 1) {
    // defer_combine: the caller's next kernel reads (and re-zeroes) the
    // f32 scratch itself — skip the convert/bias pass
    if (defer_combine && bptr == nullptr) return true;
    const int64_t total = (int64_t)M * N;
    const int cgrid = (int)std::min<int64_t>((total + 255) / 256, 2048);
    hipLaunchKernelGGL(f32_to_bf16_bias_kernel, dim3(cgrid), dim3(256), 0,
                       stream, fp, bptr, op, total, (int)N);
    g_scratch_dirty_elems =
        std::max<int64_t>(g_scratch_dirty_elems, (int64_t)M * N);
  }
  return false;
}

// Would gemm_m16(defer_combine=true) actually defer for this shape?
// (Pure: lets callers decide between the fused-consumer and the
// bias-in-combine paths BEFORE launching.)
bool gemm_m16_will_defer(int64_t M, int64_t N, int64_t K, int64_t group,
                         int64_t bits, int64_t scratch_elems) {
  if (M > 64 || M <= 2) return false;
  const int sk = pick_splitk(N, K, M, scratch_elems, bits < 16 ? 4 : 8,
                             bits < 16 ? std::max<int>((int)group, 64) : 64);
  return sk > 1;
}

bool gemm_m16(torch::Tensor x, torch::Tensor w,
              c10::optional<torch::Tensor> scales,
              c10::optional<torch::Tensor> bias, torch::Tensor out,
              c10::optional<torch::Tensor> scratch, int64_t group,
              bool packed, int64_t bits, bool defer_combine) {
  const int64_t M = x.size(0), K = x.size(1), N = w.size(0);
  DNET_CHECK(K % 64 == 0, "K % 64 == 0 required for the MFMA path");
  DNET_CHECK(out.size(0) == M && out.size(1) == N, "shape");
  DNET_CHECK(x.is_contiguous() && w.is_contiguous() && out.is_contiguous(), "contig");
  if (bits == 16) {
    DNET_CHECK(w.size(1) == K, "w shape");
  } else {
    DNET_CHECK(packed, "quantized MFMA path expects the packed weight layout");
    DNET_CHECK(scales.has_value() && scales->is_contiguous(), "scales");
    DNET_CHECK(group % 64 == 0 && K % group == 0, "group align");
    if (bits == 8) DNET_CHECK(w.size(1) == K, "w shape (int8)");
    if (bits == 4) {
      DNET_CHECK(w.size(1) == K / 2 && K % 128 == 0 && group % 128 == 0,
                 "w shape / align (int4)");
    }
  }
  // defer only meaningful for the single-chunk (M <= 64) decode path:
  // multiple chunks could pick different sk and leave a mixed state
  const bool defer = defer_combine && M <= 64 && !bias.has_value();
  bool deferred = false;
  int64_t m0 = 0;
  while (m0 < M) {
    const int mt = (int)std::min<int64_t>(M - m0, 64);
    deferred = launch_m16(x, w, scales, bias, out, scratch, (int)group, m0,
                          mt, (int)bits, defer);
    m0 += mt;
  }
  return deferred;
}

}  // namespace dnet
