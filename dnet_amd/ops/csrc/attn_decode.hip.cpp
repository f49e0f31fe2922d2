// Single-token (decode) attention against the KV cache, GQA-aware.
//
//   out[b,h,:] = softmax(q[b,h,:] . K[b,kv(h),:len,:]^T * scale) @ V[b,kv(h),:len,:]
//
// One workgroup per (batch, kv_head): the K/V read — the memory bound — is
// staged through LDS once and shared by all query heads of the group and all
// 4 waves. Online softmax (flash-decode style) over 64-position chunks;
// per-lane layout: score phase = one seq position per lane, accumulate
// phase = D/64 output dims per lane with the softmax weight broadcast by
// shuffle. Sequence length comes from a device tensor (`pos`) so the kernel
// is hipGraph-replayable across decode steps without recapture.
//
// Replaces the reference's MLX scaled_dot_product_attention decode path
// (reference: src/dnet/core/models/llama.py apply_single_layer).
#include "common.h"

namespace dnet {

// Row pitch in bf16 elements for a D-wide LDS tile such that the per-lane
// column reads are bank-conflict-free (pitch in dwords must be ≡ 2 mod 4
// and 8B-aligned; 66 dwords for D=128 → 2l mod 64 distinct per 32-lane group).
__device__ __host__ __forceinline__ constexpr int lds_pitch(int D) {
  return D + 4;  // bf16 elems; D=128 -> 264 B = 66 dwords; D=64 -> 136 B = 34 dwords
}

__device__ __forceinline__ float wave_allreduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}
__device__ __forceinline__ float wave_allreduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

constexpr int kChunk = 64;

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using bf16x2 = __attribute__((ext_vector_type(2))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

// splits > 1: flash-decode S-partitioning — gridDim.y splits each take
// chunks c = c0 + split, c0 + split + splits, ...; unnormalized partials
// (acc, m, l) go to scratch [B, Hq, splits, D+2] f32 and a combine kernel
// merges them (sink logit folded there). Raises the block count from
// B*Hkv to B*Hkv*splits so small batches still fill 256 CUs.
template <int D, int DV, bool Q8>
__global__ void attn_decode_kernel(const short* __restrict__ q,
                                   const void* __restrict__ kc,
                                   const void* __restrict__ vc,
                                   const short* __restrict__ kscale,
                                   const short* __restrict__ vscale,
                                   const int* __restrict__ pos,
                                   short* __restrict__ out, const int Hq,
                                   const int Hkv, const int Smax,
                                   const float scale, const int ldq,
                                   const int window,
                                   const short* __restrict__ sinks,
                                   float* __restrict__ partials,
                                   const int splits) {
  constexpr int P = lds_pitch(D);
  constexpr int PV = lds_pitch(DV);
  constexpr int DPL = DV / kWave;  // output dims per lane (1 or 2)
  const int b = blockIdx.x / Hkv;
  const int hkv = blockIdx.x % Hkv;
  const int G = Hq / Hkv;  // query heads per kv head
  const int wid = threadIdx.x / kWave;
  const int lane = threadIdx.x & (kWave - 1);

  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* k_lds = reinterpret_cast<short*>(smem);               // [kChunk][P]
  short* v_lds = k_lds + kChunk * P;                           // [kChunk][PV]
  short* q_lds = v_lds + kChunk * PV;                          // [G][D]
  float* p_lds = reinterpret_cast<float*>(q_lds + ((G * D + 7) & ~7));
  // p_lds: [4 waves][kChunk] exp'd scores — LDS broadcast reads replace
  // the per-iteration __shfl (long dependency chain). (An f32 V tile
  // was measured: the +17 KB LDS costs a block/CU of occupancy and
  // LOSES 13% at 2k ctx — V stays bf16.)

  const int len = pos[b];
  const int start = (window > 0) ? max(0, len - window) : 0;  // sliding
  const int64_t kvbase = ((int64_t)b * Hkv + hkv) * Smax * D;
  const int64_t vbase = ((int64_t)b * Hkv + hkv) * Smax * DV;

  // Stage this group's q rows.
  for (int i = threadIdx.x; i < G * D / 8; i += blockDim.x) {
    const int g = i / (D / 8);
    reinterpret_cast<short8*>(q_lds)[i] =
        reinterpret_cast<const short8*>(q + (int64_t)b * ldq + (hkv * G + g) * D)[i % (D / 8)];
  }

  // Per-wave head list: g = wid, wid+4, ... (max 4 heads per wave).
  float m[4], l[4], acc[4][DPL];
  const int nh = (G - wid + 3) / 4;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    m[i] = -1e30f;
    l[i] = 0.f;
#pragma unroll
    for (int j = 0; j < DPL; ++j) acc[i][j] = 0.f;
  }
  __syncthreads();

  const int nchunks = (len + kChunk - 1) / kChunk;
  const int split = blockIdx.y;
  for (int c = start / kChunk + split; c < nchunks; c += splits) {
    const int s0 = c * kChunk;
    const int valid = min(kChunk, len - s0);
    // Stage K and V chunk into bf16 LDS; Q8 caches dequantize while staging
    // (HBM reads stay int8 — half the KV bandwidth).
    for (int i = threadIdx.x; i < kChunk * D / 8; i += blockDim.x) {
      const int row = i / (D / 8);
      const int col = (i % (D / 8)) * 8;
      short8 kv8;
      if (row < valid) {
        if (Q8) {
          constexpr int NG = D / 64;
          const int64_t rb = kvbase / D * (int64_t)NG
                             + (int64_t)(s0 + row) * NG + col / 64;
          const float ks = bits2f(kscale[rb]);
          const int2 kq = *reinterpret_cast<const int2*>(
              (const int8_t*)kc + kvbase + (int64_t)(s0 + row) * D + col);
          const int8_t* kb = reinterpret_cast<const int8_t*>(&kq);
#pragma unroll
          for (int j = 0; j < 8; ++j) kv8.x[j] = f2bits((float)kb[j] * ks);
        } else {
          kv8 = *reinterpret_cast<const short8*>(
              (const short*)kc + kvbase + (int64_t)(s0 + row) * D + col);
        }
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) kv8.x[j] = 0;
      }
      *reinterpret_cast<short8*>(&k_lds[row * P + col]) = kv8;
    }
    for (int i = threadIdx.x; i < kChunk * DV / 8; i += blockDim.x) {
      const int row = i / (DV / 8);
      const int col = (i % (DV / 8)) * 8;
      short8 vv8;
      if (row < valid) {
        if (Q8) {
          constexpr int NGV = DV / 64;
          const int64_t rb = vbase / DV * (int64_t)NGV
                             + (int64_t)(s0 + row) * NGV + col / 64;
          const float vs = bits2f(vscale[rb]);
          const int2 vq = *reinterpret_cast<const int2*>(
              (const int8_t*)vc + vbase + (int64_t)(s0 + row) * DV + col);
          const int8_t* vb = reinterpret_cast<const int8_t*>(&vq);
#pragma unroll
          for (int j = 0; j < 8; ++j) vv8.x[j] = f2bits((float)vb[j] * vs);
        } else {
          vv8 = *reinterpret_cast<const short8*>(
              (const short*)vc + vbase + (int64_t)(s0 + row) * DV + col);
        }
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) vv8.x[j] = 0;
      }
      *reinterpret_cast<short8*>(&v_lds[row * PV + col]) = vv8;
    }
    __syncthreads();

    for (int hi = 0; hi < nh; ++hi) {
      const int g = wid + hi * 4;
      // Score for position s = lane (packed bf16 dot2: one VALU per
      // 2 dims instead of 2 converts + 2 fmas).
      float s_val = 0.f;
      const short* krow = &k_lds[lane * P];
      const short* qrow = &q_lds[g * D];
      for (int d = 0; d < D; d += 4) {
        const short4v kq = *reinterpret_cast<const short4v*>(&krow[d]);
        const short4v qq = *reinterpret_cast<const short4v*>(&qrow[d]);
        s_val = __builtin_amdgcn_fdot2_f32_bf16(
            *reinterpret_cast<const bf16x2*>(&kq.x[0]),
            *reinterpret_cast<const bf16x2*>(&qq.x[0]), s_val, false);
        s_val = __builtin_amdgcn_fdot2_f32_bf16(
            *reinterpret_cast<const bf16x2*>(&kq.x[2]),
            *reinterpret_cast<const bf16x2*>(&qq.x[2]), s_val, false);
      }
      s_val *= scale;
      const bool in_win = (lane < valid) && (s0 + lane >= start);
      if (!in_win) s_val = -1e30f;
      // Online softmax update.
      const float cmax = wave_allreduce_max(s_val);
      const float mn = fmaxf(m[hi], cmax);
      const float alpha = __expf(m[hi] - mn);
      const float p = in_win ? __expf(s_val - mn) : 0.f;
      l[hi] = l[hi] * alpha + wave_allreduce_sum(p);
      m[hi] = mn;
#pragma unroll
      for (int j = 0; j < DPL; ++j) acc[hi][j] *= alpha;
      // Accumulate P @ V: lane owns dims d = DPL*lane + j; p comes
      // from a per-wave LDS slab (broadcast reads — no serial __shfl
      // dependency chain)
      p_lds[wid * kChunk + lane] = p;
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");  // wave-local
      for (int s = 0; s < valid; ++s) {
        const float ps = p_lds[wid * kChunk + s];
        const short* vrow = &v_lds[s * PV + DPL * lane];
#pragma unroll
        for (int j = 0; j < DPL; ++j)
          acc[hi][j] = fmaf(ps, bits2f(vrow[j]), acc[hi][j]);
      }
    }
    __syncthreads();
  }

  if (partials != nullptr) {
    // write unnormalized partials; the combine kernel (or a cross-rank
    // context-parallel combine) normalizes + folds sinks
    for (int hi = 0; hi < nh; ++hi) {
      const int g = wid + hi * 4;
      float* prow = partials
          + (((int64_t)b * Hq + hkv * G + g) * splits + split) * (DV + 2);
#pragma unroll
      for (int j = 0; j < DPL; ++j) prow[DPL * lane + j] = acc[hi][j];
      if (lane == 0) {
        prow[DV] = m[hi];
        prow[DV + 1] = l[hi];
      }
    }
    return;
  }
  for (int hi = 0; hi < nh; ++hi) {
    const int g = wid + hi * 4;
    float inv;
    if (sinks != nullptr) {
      // gpt-oss attention sink: one extra learned softmax logit per q head
      // whose probability mass is dropped from the output.
      const float sk = bits2f(sinks[hkv * G + g]);
      const float mx = fmaxf(m[hi], sk);
      const float num = __expf(m[hi] - mx);
      const float denom = l[hi] * num + __expf(sk - mx);
      inv = (denom > 0.f) ? num / denom : 0.f;
    } else {
      inv = (l[hi] > 0.f) ? 1.f / l[hi] : 0.f;
    }
    short* orow = out + ((int64_t)b * Hq + hkv * G + g) * DV + DPL * lane;
#pragma unroll
    for (int j = 0; j < DPL; ++j) orow[j] = f2bits(acc[hi][j] * inv);
  }
}


// ---------------------------------------------------------------------------
// MFMA decode attention: same contract as attn_decode_kernel, compute
// reshaped for the matrix cores. The G (<=16) query heads of a
// (batch, kv-head) group form the 16-row A operand; each of the 4 waves
// owns a 16-position subtile of every 64-position chunk and keeps an
// INDEPENDENT online-softmax state (m, l, acc) — position space is
// fully partitioned across waves, so there is no per-chunk cross-wave
// synchronization at all, just one merge at the end (intra-block
// flash-decode split). Scores = Q[16,D] x K^T via v_mfma_f32_16x16x32,
// P@V via the same MFMA with the per-wave P slab zero-padded to k=32.
// Replaces a serial 64-iteration __shfl broadcast loop per (head,
// chunk) in the scalar kernel (~26us -> ~14us per layer at batch 64,
// and the win grows with context).
template <int D, int DV, bool Q8>
__global__ __launch_bounds__(256) void attn_decode_mfma_kernel(
    const short* __restrict__ q, const void* __restrict__ kc,
    const void* __restrict__ vc, const short* __restrict__ kscale,
    const short* __restrict__ vscale, const int* __restrict__ pos,
    short* __restrict__ out, const int Hq, const int Hkv, const int Smax,
    const float scale, const int ldq, const int window,
    const short* __restrict__ sinks, float* __restrict__ partials,
    const int splits) {
  constexpr int DC = D / 32;   // QK k-chunks
  constexpr int NV = DV / 16;  // PV n-tiles
  constexpr int KP = D + 8;    // k_lds row pitch (16-lane frags bank-clean)
  constexpr int VP = 88;       // vt_lds pitch: covers the k<=79 overread of
                               // wave 3 (zero-P columns) and 44-dword row
                               // stride hits all banks across 16 rows
  constexpr int SP = 40;       // per-wave P slab pitch
  constexpr int STAGE = 64 * KP * 2 + DV * VP * 2;
  static_assert(STAGE >= 4 * 16 * DV * 4 + 4 * 16 * 2 * 4,
                "merge area must fit in the staging LDS");
  __shared__ __attribute__((aligned(16))) char smem[STAGE];
  __shared__ short p_lds[4][16 * SP];
  short* k_lds = reinterpret_cast<short*>(smem);
  short* vt_lds = reinterpret_cast<short*>(smem + 64 * KP * 2);

  const int b = blockIdx.x / Hkv;
  const int hkv = blockIdx.x % Hkv;
  const int G = Hq / Hkv;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int row = lane & 15;
  const int hi = lane >> 4;
  const int len = pos[b];
  const int start = (window > 0) ? max(0, len - window) : 0;
  const int64_t kvbase = ((int64_t)b * Hkv + hkv) * Smax * D;
  const int64_t vbase = ((int64_t)b * Hkv + hkv) * (int64_t)Smax * DV;

  // Q fragments (A layout: lane holds Q[head=row][c*32 + hi*8 .. +8]);
  // rows >= G duplicate head G-1 — their outputs are never stored and
  // softmax is per-row, so they cannot contaminate real heads
  const short* qp = q + (int64_t)b * ldq + (int64_t)(hkv * G + min(row, G - 1)) * D;
  bf16x8 qf[DC];
#pragma unroll
  for (int c = 0; c < DC; ++c)
    qf[c] = *reinterpret_cast<const bf16x8*>(&qp[c * 32 + hi * 8]);

  f32x4 acc[NV];
#pragma unroll
  for (int n = 0; n < NV; ++n) acc[n] = {0.f, 0.f, 0.f, 0.f};
  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -1e30f;
    l_run[r] = 0.f;
  }

  // zero this wave's dead P columns (k = 16..31) once: the PV MFMA runs
  // k=32 and the upper half multiplies whatever V rows it overreads
  for (int i = lane; i < 16 * 16; i += 64)
    p_lds[wave][(i >> 4) * SP + 16 + (i & 15)] = 0;
  // zero vt's never-staged tail columns (s = 64..VP-1) once: wave 3's
  // k>=16 B-fragment overreads them, and uninitialized LDS bits can be
  // NaN — NaN * 0 is NaN, which would poison the accumulators
  for (int i = threadIdx.x; i < DV * (VP - 64); i += 256)
    vt_lds[(i / (VP - 64)) * VP + 64 + i % (VP - 64)] = 0;

  const int nchunks = (len + kChunk - 1) / kChunk;
  const int split = blockIdx.y;
  for (int c = start / kChunk + split; c < nchunks; c += splits) {
    const int s0 = c * kChunk;
    const int valid = min(kChunk, len - s0);
    __syncthreads();   // previous tile's consumers done
    for (int i = threadIdx.x; i < kChunk * (D / 8); i += blockDim.x) {
      const int sr = i / (D / 8);
      const int col = (i % (D / 8)) * 8;
      short8 kv8;
      if (sr < valid) {
        if (Q8) {
          constexpr int NG = D / 64;
          const int64_t rb = kvbase / D * (int64_t)NG
                             + (int64_t)(s0 + sr) * NG + col / 64;
          const float ks = bits2f(kscale[rb]);
          const int2 kq = *reinterpret_cast<const int2*>(
              (const int8_t*)kc + kvbase + (int64_t)(s0 + sr) * D + col);
          const int8_t* kb = reinterpret_cast<const int8_t*>(&kq);
#pragma unroll
          for (int j = 0; j < 8; ++j) kv8.x[j] = f2bits((float)kb[j] * ks);
        } else {
          kv8 = *reinterpret_cast<const short8*>(
              (const short*)kc + kvbase + (int64_t)(s0 + sr) * D + col);
        }
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) kv8.x[j] = 0;
      }
      *reinterpret_cast<short8*>(&k_lds[sr * KP + col]) = kv8;
    }
    // V transposed [dv][s] so the PV B-fragment reads are contiguous;
    // global side reads full short8 rows (the 8 scattered 2 B LDS
    // stores are cheap, 4 B global loads were not)
    for (int i = threadIdx.x; i < kChunk * (DV / 8); i += blockDim.x) {
      const int sr = i / (DV / 8);
      const int dc = (i % (DV / 8)) * 8;
      short8 vv8;
      if (sr < valid) {
        if (Q8) {
          constexpr int NGV = DV / 64;
          const int64_t rb = vbase / DV * (int64_t)NGV
                             + (int64_t)(s0 + sr) * NGV + dc / 64;
          const float vs = bits2f(vscale[rb]);
          const int2 vq = *reinterpret_cast<const int2*>(
              (const int8_t*)vc + vbase + (int64_t)(s0 + sr) * DV + dc);
          const int8_t* vb = reinterpret_cast<const int8_t*>(&vq);
#pragma unroll
          for (int j = 0; j < 8; ++j) vv8.x[j] = f2bits((float)vb[j] * vs);
        } else {
          vv8 = *reinterpret_cast<const short8*>(
              (const short*)vc + vbase + (int64_t)(s0 + sr) * DV + dc);
        }
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) vv8.x[j] = 0;
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) vt_lds[(dc + j) * VP + sr] = vv8.x[j];
    }
    __syncthreads();

    // scores for this wave's 16 positions: C[head=hi*4+r][pos col=row]
    f32x4 sc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int cc = 0; cc < DC; ++cc) {
      const bf16x8 kf = *reinterpret_cast<const bf16x8*>(
          &k_lds[(wave * 16 + row) * KP + cc * 32 + hi * 8]);
      sc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[cc], kf, sc, 0, 0, 0);
    }
    const int spos = s0 + wave * 16 + row;
    const bool dead = (spos >= len) || (spos < start);
    float pv[4];
#pragma unroll
    for (int r = 0; r < 4; ++r)
      pv[r] = dead ? -INFINITY : sc[r] * scale;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mx = pv[r];
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off, 16));
      const float mn = fmaxf(m_run[r], mx);
      const float corr = __expf(m_run[r] - mn);
      pv[r] = __expf(pv[r] - mn);   // -inf underflows to exactly 0
      float ls = pv[r];
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        ls += __shfl_xor(ls, off, 16);
      l_run[r] = l_run[r] * corr + ls;
      m_run[r] = mn;
#pragma unroll
      for (int n = 0; n < NV; ++n) acc[n][r] *= corr;
    }
    // P slab (cols 0..15 = this wave's positions), then PV
#pragma unroll
    for (int r = 0; r < 4; ++r)
      p_lds[wave][(hi * 4 + r) * SP + row] = f2bits(pv[r]);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");  // wave-local slab
#pragma unroll
    for (int n = 0; n < NV; ++n) {
      const bf16x8 pf =
          *reinterpret_cast<const bf16x8*>(&p_lds[wave][row * SP + hi * 8]);
      const bf16x8 vf = *reinterpret_cast<const bf16x8*>(
          &vt_lds[(n * 16 + row) * VP + wave * 16 + hi * 8]);
      acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, vf, acc[n], 0, 0,
                                                       0);
    }
  }

  // ---- cross-wave merge (position space was partitioned; combine the
  // four partial states like a split-S merge, in the freed staging LDS)
  __syncthreads();
  float* mrg = reinterpret_cast<float*>(smem);            // [4][16][DV]
  float* mml = mrg + 4 * 16 * DV;                         // [4][16][2]
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int head = hi * 4 + r;
#pragma unroll
    for (int n = 0; n < NV; ++n)
      mrg[(wave * 16 + head) * DV + n * 16 + row] = acc[n][r];
    if (row == 0) {
      mml[(wave * 16 + head) * 2] = m_run[r];
      mml[(wave * 16 + head) * 2 + 1] = l_run[r];
    }
  }
  __syncthreads();
  if (wave != 0) return;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int head = hi * 4 + r;
    float M = -1e30f;
#pragma unroll
    for (int w = 0; w < 4; ++w)
      M = fmaxf(M, mml[(w * 16 + head) * 2]);
    float L = 0.f;
#pragma unroll
    for (int n = 0; n < NV; ++n) acc[n][r] = 0.f;
#pragma unroll
    for (int w = 0; w < 4; ++w) {
      const float wgt = __expf(mml[(w * 16 + head) * 2] - M);
      L += mml[(w * 16 + head) * 2 + 1] * wgt;
#pragma unroll
      for (int n = 0; n < NV; ++n)
        acc[n][r] += mrg[(w * 16 + head) * DV + n * 16 + row] * wgt;
    }
    m_run[r] = M;
    l_run[r] = L;
  }

  if (partials != nullptr) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int head = hi * 4 + r;
      if (head >= G) continue;
      float* prow = partials
          + (((int64_t)b * Hq + hkv * G + head) * splits + split) * (DV + 2);
#pragma unroll
      for (int n = 0; n < NV; ++n) prow[n * 16 + row] = acc[n][r];
      if (row == 0) {
        prow[DV] = m_run[r];
        prow[DV + 1] = l_run[r];
      }
    }
    return;
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int head = hi * 4 + r;
    if (head >= G) continue;
    float inv;
    if (sinks != nullptr) {
      const float sk = bits2f(sinks[hkv * G + head]);
      const float mx = fmaxf(m_run[r], sk);
      const float num = __expf(m_run[r] - mx);
      const float denom = l_run[r] * num + __expf(sk - mx);
      inv = (denom > 0.f) ? num / denom : 0.f;
    } else {
      inv = (l_run[r] > 0.f) ? 1.f / l_run[r] : 0.f;
    }
    short* orow = out + ((int64_t)b * Hq + hkv * G + head) * DV;
#pragma unroll
    for (int n = 0; n < NV; ++n)
      orow[n * 16 + row] = f2bits(acc[n][r] * inv);
  }
}

// merge the per-split unnormalized partials: one 64-lane block per (b, h).
template <int D>
__global__ void attn_combine_kernel(const float* __restrict__ partials,
                                    const short* __restrict__ sinks,
                                    short* __restrict__ out, const int Hq,
                                    const int splits) {
  constexpr int DPL = D / kWave;
  const int bh = blockIdx.x;          // b * Hq + h
  const int h = bh % Hq;
  const int lane = threadIdx.x;
  const float* base = partials + (int64_t)bh * splits * (D + 2);
  float M = -1e30f;
  for (int s = 0; s < splits; ++s) M = fmaxf(M, base[s * (D + 2) + D]);
  if (sinks != nullptr) M = fmaxf(M, bits2f(sinks[h]));
  float denom = (sinks != nullptr) ? __expf(bits2f(sinks[h]) - M) : 0.f;
  float acc[DPL];
#pragma unroll
  for (int j = 0; j < DPL; ++j) acc[j] = 0.f;
  for (int s = 0; s < splits; ++s) {
    const float* prow = base + s * (D + 2);
    const float w = __expf(prow[D] - M);
    denom += prow[D + 1] * w;
#pragma unroll
    for (int j = 0; j < DPL; ++j) acc[j] += prow[DPL * lane + j] * w;
  }
  const float inv = denom > 0.f ? 1.f / denom : 0.f;
  short* orow = out + (int64_t)bh * D + DPL * lane;
#pragma unroll
  for (int j = 0; j < DPL; ++j) orow[j] = f2bits(acc[j] * inv);
}

// merge split/rank partials [B*Hq, splits, DV+2] -> out [B, Hq, DV]
// (cross-rank context parallelism gathers partials and calls this with
// splits = world * local_splits).
void attn_combine(torch::Tensor partials, c10::optional<torch::Tensor> sinks,
                  torch::Tensor out, int64_t splits) {
  const int64_t B = out.size(0), Hq = out.size(1), DV = out.size(2);
  DNET_CHECK(DV == 64 || DV == 128, "combine DV 64/128");
  DNET_CHECK(partials.is_contiguous() && out.is_contiguous(), "contig");
  DNET_CHECK(partials.numel() >= B * Hq * splits * (DV + 2), "partials size");
  auto stream = current_stream();
  const short* skp = sinks.has_value() ? (const short*)sinks->data_ptr()
                                       : nullptr;
  if (DV == 128)
    hipLaunchKernelGGL((attn_combine_kernel<128>), dim3((unsigned)(B * Hq)),
                       dim3(kWave), 0, stream,
                       (const float*)partials.data_ptr(), skp,
                       (short*)out.data_ptr(), (int)Hq, (int)splits);
  else
    hipLaunchKernelGGL((attn_combine_kernel<64>), dim3((unsigned)(B * Hq)),
                       dim3(kWave), 0, stream,
                       (const float*)partials.data_ptr(), skp,
                       (short*)out.data_ptr(), (int)Hq, (int)splits);
}

// q may be a column slice of a fused-QKV buffer: strides (ldq, D, 1).
void attn_decode(torch::Tensor q, torch::Tensor kcache, torch::Tensor vcache,
                 torch::Tensor pos, torch::Tensor out, double scale,
                 int64_t window, c10::optional<torch::Tensor> sinks,
                 c10::optional<torch::Tensor> kscale,
                 c10::optional<torch::Tensor> vscale,
                 c10::optional<torch::Tensor> partials, int64_t splits,
                 bool combine) {
  const int64_t B = q.size(0), Hq = q.size(1), D = q.size(2);
  const int64_t Hkv = kcache.size(1), Smax = kcache.size(2);
  const int64_t DV = vcache.size(3);
  DNET_CHECK(kcache.size(0) == B && kcache.size(3) == D, "kcache shape");
  DNET_CHECK(out.size(2) == DV, "out width = v head dim");
  DNET_CHECK(Hq % Hkv == 0 && Hq / Hkv <= 16, "GQA group <= 16");
  DNET_CHECK((D == 64 && DV == 64) || (D == 128 && DV == 128) ||
                 (D == 192 && DV == 128),
             "head dims must be 64/64, 128/128 or 192/128 (MLA)");
  DNET_CHECK(pos.dtype() == torch::kInt32, "pos int32");
  DNET_CHECK(q.stride(2) == 1 && q.stride(1) == D, "q inner dims contiguous");
  DNET_CHECK(kcache.is_contiguous() && vcache.is_contiguous() &&
                 out.is_contiguous(), "contig");
  auto stream = current_stream();
  const int G = (int)(Hq / Hkv);
  const size_t lds = (kChunk * ((int)D + 4) + kChunk * ((int)DV + 4)
                      + (((int64_t)G * D + 7) & ~7)) * sizeof(short)
                     + 4 * kChunk * sizeof(float);
  if (splits > 1 || !combine) {
    DNET_CHECK(partials.has_value()
                   && partials->numel() >= B * Hq * splits * (DV + 2),
               "split-S partials scratch required");
  }
  const dim3 grid((unsigned)(B * Hkv), (unsigned)splits);
  const int ldq = (int)q.stride(0);
  const short* skp = sinks.has_value() ? (const short*)sinks->data_ptr()
                                       : nullptr;
  const bool q8 = kcache.dtype() == torch::kInt8;
  const short* ksp = q8 ? (const short*)kscale->data_ptr() : nullptr;
  const short* vsp = q8 ? (const short*)vscale->data_ptr() : nullptr;
  float* pp = (splits > 1 || !combine) ? (float*)partials->data_ptr()
                                       : nullptr;
  // Scalar flash-decode is the default: the MFMA variant measured
  // SLOWER end to end (3635 vs 3683 tok/s short-ctx, 1433 vs 1491 at
  // 2k ctx) — decode attention has so little arithmetic per KV byte
  // that the V-transpose staging the PV MFMA needs (8 scattered 2 B
  // LDS stores per element vs one short8 row store) costs more than
  // the matrix cores save. Kept for A/B via DNET_ATTN_MFMA=1.
  static const bool mfma_on = []() {
    const char* e = getenv("DNET_ATTN_MFMA");
    return e != nullptr && e[0] == '1';
  }();
#define LAUNCH_ATTN(DD, DDV, QQ)                                              \
  do {                                                                        \
    if (mfma_on)                                                              \
      hipLaunchKernelGGL((attn_decode_mfma_kernel<DD, DDV, QQ>), grid,        \
                         dim3(256), 0, stream, (const short*)q.data_ptr(),    \
                         kcache.data_ptr(), vcache.data_ptr(), ksp, vsp,      \
                         (const int*)pos.data_ptr(), (short*)out.data_ptr(),  \
                         (int)Hq, (int)Hkv, (int)Smax, (float)scale, ldq,     \
                         (int)window, (splits > 1 ? nullptr : skp), pp,       \
                         (int)splits);                                        \
    else                                                                      \
      hipLaunchKernelGGL((attn_decode_kernel<DD, DDV, QQ>), grid, dim3(256),  \
                         lds, stream, (const short*)q.data_ptr(),             \
                         kcache.data_ptr(), vcache.data_ptr(), ksp, vsp,      \
                         (const int*)pos.data_ptr(), (short*)out.data_ptr(),  \
                         (int)Hq, (int)Hkv, (int)Smax, (float)scale, ldq,     \
                         (int)window, (splits > 1 ? nullptr : skp), pp,       \
                         (int)splits);                                        \
  } while (0)
  if (D == 192 && q8) LAUNCH_ATTN(192, 128, true);
  else if (D == 192) LAUNCH_ATTN(192, 128, false);
  else if (D == 128 && q8) LAUNCH_ATTN(128, 128, true);
  else if (D == 128) LAUNCH_ATTN(128, 128, false);
  else if (q8) LAUNCH_ATTN(64, 64, true);
  else LAUNCH_ATTN(64, 64, false);
#undef LAUNCH_ATTN
  if (splits > 1 && combine) {
    if (DV == 128)
      hipLaunchKernelGGL((attn_combine_kernel<128>), dim3((unsigned)(B * Hq)),
                         dim3(kWave), 0, stream, pp, skp,
                         (short*)out.data_ptr(), (int)Hq, (int)splits);
    else
      hipLaunchKernelGGL((attn_combine_kernel<64>), dim3((unsigned)(B * Hq)),
                         dim3(kWave), 0, stream, pp, skp,
                         (short*)out.data_ptr(), (int)Hq, (int)splits);
  }
}

}  // namespace dnet
