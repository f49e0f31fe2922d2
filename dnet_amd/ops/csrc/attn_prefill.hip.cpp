// MFMA flash prefill attention (gfx950), causal + sliding window + sinks.
//
// Replaces the chunked-einsum prefill path (models/base.py
// _chunked_causal_attention: hipBLASLt GEMMs + materialized fp32 score
// tensors + masked_fill + softmax — ~160 ms of the round-1 TTFT). One
// fused kernel: QK^T and PV on v_mfma_f32_16x16x32_bf16, fp32 online
// softmax in the C fragments, score tiles never touch HBM.
// Reference counterpart: src/dnet/core/models/llama.py:76-102 (MLX SDPA).
//
// Geometry: block = 4 waves; each wave owns 16 q rows of a 64-row q tile
// (grid.x = ceil(T/64), grid.y = Hq, grid.z = B). Per 32-position s tile:
//   - the 4 waves cooperatively stage K [32][D] and V-transposed [Dv][32]
//     into LDS (coalesced 16B lines; Vt transposed on the write side),
//   - QK^T: 2 x (D/32) MFMAs -> scores [16q x 16s] fp32 in C layout
//     (lane l holds col=l&15, rows (l>>4)*4+0..3),
//   - causal/window masking in registers; online softmax: row max/sum via
//     16-lane shfl_xor reductions along the col axis,
//   - P (bf16) relayed to a per-wave LDS slab -> read back as A
//     fragments; PV: (Dv/16) MFMAs per s tile accumulate O.
// Epilogue folds the gpt-oss sink logit into the denominator and writes
// O / rowsum.
#include "common.h"

namespace dnet {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

// D: qk head dim (64/128/192); DV: v head dim (64/128).
template <int D, int DV, bool WINDOW, bool SINKS>
__global__ __launch_bounds__(256) void attn_prefill_kernel(
    const short* __restrict__ q,    // [B, Hq, T, D]
    const short* __restrict__ k,    // [B, Hkv, S, D]
    const short* __restrict__ v,    // [B, Hkv, S, DV]
    const short* __restrict__ sinks,  // [Hq] or null
    short* __restrict__ out,        // [B, Hq, T, DV]
    const int B, const int Hq, const int Hkv, const int T, const int S,
    const int q_off, const int window, const float scale) {
  constexpr int ST = 32;            // s positions per tile
  constexpr int DC = D / 32;        // QK k-chunks
  constexpr int NV = DV / 16;       // PV n-tiles
  constexpr int KP = D + 8;         // k_lds row stride (pad: 16-lane
  constexpr int SP = ST + 8;        //   fragment reads hit all banks)
  __shared__ short k_lds[ST * KP];      // [32][D+8]
  __shared__ short vt_lds[DV * SP];     // [DV][32+8] (transposed)
  __shared__ short p_lds[4][16 * SP];   // per-wave P slab [16q][32+8 s]

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hk = h / (Hq / Hkv);
  const int qt = blockIdx.x * 64 + wave * 16;   // this wave's first q row
  const int row = lane & 15;
  const int hi = lane >> 4;                      // 0..3
  const int qrow = min(qt + row, T - 1);

  const short* qp = q + (((int64_t)b * Hq + h) * T + qrow) * D;
  const short* kp = k + ((int64_t)b * Hkv + hk) * S * D;
  const short* vp = v + ((int64_t)b * Hkv + hk) * S * DV;

  // Q fragments once (A layout: lane holds Q[row][c*32 + hi*8 .. +8])
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

  // causal bound for the whole 64-row block (uniform): last s needed
  const int q_hi = min((int)blockIdx.x * 64 + 63, T - 1);
  const int s_end = min(S, q_off + q_hi + 1);
  // window lower bound for the block (uniform, conservative)
  int s_begin = 0;
  if (WINDOW) {
    const int q_lo = (int)blockIdx.x * 64;
    s_begin = max(0, (q_off + q_lo) - window + 1);
    s_begin &= ~(ST - 1);  // tile-aligned; in-tile mask handles the rest
  }

  for (int s0 = s_begin; s0 < s_end; s0 += ST) {
    // ---- stage K[32][D] + Vt[DV][32] (all 4 waves cooperate) ----
    __syncthreads();   // previous tile's consumers done
    for (int idx = threadIdx.x; idx < ST * (D / 8); idx += 256) {
      const int sr = idx / (D / 8);
      const int dc = idx % (D / 8);
      const int sg = min(s0 + sr, S - 1);
      *reinterpret_cast<short8*>(&k_lds[sr * KP + dc * 8]) =
          *reinterpret_cast<const short8*>(&kp[(int64_t)sg * D + dc * 8]);
    }
    // V transposed: thread reads V[s][dv..dv+1] pairs; write [dv][s].
    // 2 elems per thread-iter keeps the global reads 4B; fine once per
    // tile (the PV reads hit LDS many times).
    for (int idx = threadIdx.x; idx < ST * (DV / 2); idx += 256) {
      const int sr = idx / (DV / 2);
      const int dc = (idx % (DV / 2)) * 2;
      const int sg = min(s0 + sr, S - 1);
      const short v0 = vp[(int64_t)sg * DV + dc];
      const short v1 = vp[(int64_t)sg * DV + dc + 1];
      vt_lds[dc * SP + sr] = v0;
      vt_lds[(dc + 1) * SP + sr] = v1;
    }
    __syncthreads();

    // ---- QK^T for two 16-s subtiles ----
    float p_val[2][4];   // [subtile][r] masked+exp'd scores (this lane)
    float m_new[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) m_new[r] = m_run[r];
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      f32x4 sc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int c = 0; c < DC; ++c) {
        // B fragment: lane holds K[scol = lane&15][c*32 + hi*8 .. +8]
        const bf16x8 kf = *reinterpret_cast<const bf16x8*>(
            &k_lds[(sub * 16 + row) * KP + c * 32 + hi * 8]);
        sc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[c], kf, sc, 0, 0, 0);
      }
      // C layout: this lane holds scores[q = hi*4 + r][s = row] (within
      // the subtile); note role swap: A rows = q, B cols = s.
      const int spos = s0 + sub * 16 + row;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qq = qt + hi * 4 + r;          // global q row of sc[r]
        const float val = sc[r] * scale;
        const int qpos = q_off + qq;
        bool dead = (spos > qpos) || (spos >= S) || (qq >= T);
        if (WINDOW) dead |= (spos <= qpos - window);
        // dead = -inf (not a finite floor): with m_run still at its
        // finite init, exp(-inf - m) underflows to exactly 0, so fully
        // masked tiles contribute nothing (a finite floor would give
        // exp(0)=1 when a whole tile is dead)
        p_val[sub][r] = dead ? -INFINITY : val;
      }
    }
    // ---- online softmax (rows live across 16 lanes of each hi group,
    // but C rows are hi*4+r — row r's 16 col values sit in lanes with
    // the SAME hi, lane&15 = col). shfl_xor over the low 4 lane bits.
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mx = fmaxf(p_val[0][r], p_val[1][r]);
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off, 16));
      m_new[r] = fmaxf(m_run[r], mx);
      const float corr = __expf(m_run[r] - m_new[r]);
      float ls = 0.f;
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
        p_val[sub][r] = __expf(p_val[sub][r] - m_new[r]);
        ls += p_val[sub][r];
      }
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        ls += __shfl_xor(ls, off, 16);
      l_run[r] = l_run[r] * corr + ls;
      m_run[r] = m_new[r];
      // rescale O accumulators for this row: row r of C fragments = acc
      // element [r] of every n-tile
#pragma unroll
      for (int n = 0; n < NV; ++n) acc[n][r] *= corr;
    }
    // ---- P -> per-wave LDS slab (bf16), then PV ----
    // this lane wrote scores for q rows hi*4+r at col `row` (per sub)
#pragma unroll
    for (int sub = 0; sub < 2; ++sub)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        p_lds[wave][(hi * 4 + r) * SP + sub * 16 + row] =
            f2bits(p_val[sub][r]);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");  // wave-local slab
#pragma unroll
    for (int n = 0; n < NV; ++n) {
      // A fragment: P[q = lane&15][hi*8 + j], k-dim = 32 s positions
      const bf16x8 pf =
          *reinterpret_cast<const bf16x8*>(&p_lds[wave][row * SP + hi * 8]);
      // B fragment: Vt[dv = n*16 + lane&15][hi*8 + j]
      const bf16x8 vf = *reinterpret_cast<const bf16x8*>(
          &vt_lds[(n * 16 + row) * SP + hi * 8]);
      acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, vf, acc[n], 0, 0,
                                                       0);
    }
  }

  // ---- epilogue: sinks + normalize + store ----
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    if (SINKS) {
      const float sk = bits2f(sinks[h]);
      const float m2 = fmaxf(m_run[r], sk);
      l_run[r] = l_run[r] * __expf(m_run[r] - m2) + __expf(sk - m2);
      const float corr = __expf(m_run[r] - m2);
#pragma unroll
      for (int n = 0; n < NV; ++n) acc[n][r] *= corr;
      m_run[r] = m2;
    }
  }
  // store: lane holds O[q = hi*4 + r][dv = n*16 + row]
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qq = qt + hi * 4 + r;
    if (qq >= T) continue;
    const float inv = 1.0f / fmaxf(l_run[r], 1e-30f);
    short* op = out + (((int64_t)b * Hq + h) * T + qq) * DV;
#pragma unroll
    for (int n = 0; n < NV; ++n)
      op[n * 16 + row] = f2bits(acc[n][r] * inv);
  }
}

void attn_prefill(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                  c10::optional<torch::Tensor> sinks, torch::Tensor out,
                  int64_t q_off, int64_t window, double scale) {
  // q [B,Hq,T,D], k [B,Hkv,S,D], v [B,Hkv,S,DV], out [B,Hq,T,DV]
  const int B = q.size(0), Hq = q.size(1), T = q.size(2), D = q.size(3);
  const int Hkv = k.size(1), S = k.size(2), DV = v.size(3);
  DNET_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous() &&
                 out.is_contiguous(),
             "contig");
  DNET_CHECK(Hq % Hkv == 0, "GQA heads");
  DNET_CHECK(out.size(2) == T && out.size(3) == DV, "out shape");
  const short* sp =
      sinks.has_value() ? (const short*)sinks->data_ptr() : nullptr;
  auto stream = current_stream();
  const dim3 grid((T + 63) / 64, Hq, B);
  const bool win = window > 0;
#define LAUNCH_AP(DD, DDV, WW, SS)                                          \
  hipLaunchKernelGGL((attn_prefill_kernel<DD, DDV, WW, SS>), grid,          \
                     dim3(256), 0, stream, (const short*)q.data_ptr(),      \
                     (const short*)k.data_ptr(), (const short*)v.data_ptr(),\
                     sp, (short*)out.data_ptr(), B, Hq, Hkv, T, S,          \
                     (int)q_off, (int)window, (float)scale)
#define LAUNCH_AP_WS(DD, DDV)                                 \
  do {                                                        \
    if (win && sp) LAUNCH_AP(DD, DDV, true, true);            \
    else if (win) LAUNCH_AP(DD, DDV, true, false);            \
    else if (sp) LAUNCH_AP(DD, DDV, false, true);             \
    else LAUNCH_AP(DD, DDV, false, false);                    \
  } while (0)
  if (D == 128 && DV == 128) LAUNCH_AP_WS(128, 128);
  else if (D == 64 && DV == 64) LAUNCH_AP_WS(64, 64);
  else if (D == 192 && DV == 128) LAUNCH_AP_WS(192, 128);
  else if (D == 96 && DV == 96) LAUNCH_AP_WS(96, 96);
  else DNET_CHECK(false, "attn_prefill: unsupported head dims");
#undef LAUNCH_AP
#undef LAUNCH_AP_WS
}

}  // namespace dnet
