// Grouped MoE expert kernels: the decode-side dense-expert path in two
// launches per layer instead of O(E) tiny per-expert GEMVs.
//
// Motivation (measured): the per-expert python loop costs ~4000 small
// kernels per token on gpt-oss-20b (24 layers x 32 experts x 3 ops) and
// runs ~350 GB/s effective; these kernels put the whole expert bank behind
// two launches with an early per-expert skip, so a single-stream decode
// reads only the routed experts' weights (top-k/E of the bank).
//
//  - moe_gateup: act[e,m,i] = glu(x[m,:] . Wg[e,i,:], x[m,:] . Wu[e,i,:])
//    with Wg/Wu the halves of the stacked gate_up weights [E, 2I, K]
//    (bf16, or grouped-int8 with fused dequant, optionally in the MFMA
//    chunk-pair packed order), glu = SwiGLU (mixtral) or the gpt-oss
//    clamped GLU. A wave owns one (e, i) pair: two K-dots amortize the x
//    read and keep the weight read a pure 16 B/lane stream.
//  - moe_down: out[m,h] += we[m,e] * (act[e,m,:] . Wd[e,h,:] + b[e,h])
//    accumulated with f32 atomics across experts.
//
// Both kernels skip expert e entirely when we[m,e] == 0 for all m in the
// tile — data-dependent work without host syncs, so the pair stays
// hipGraph-capturable (reference counterpart: the MLX MoE gather path in
// src/dnet/core/models/gpt_oss.py).
#include "common.h"

namespace dnet {

constexpr int kMoeWaves = 4;

using i32x4 = __attribute__((ext_vector_type(4))) int;

template <int M>
__device__ __forceinline__ bool expert_routed(const float* __restrict__ we,
                                              const int E, const int e,
                                              const int m0) {
  bool any = false;
#pragma unroll
  for (int m = 0; m < M; ++m) any |= (we[(int64_t)(m0 + m) * E + e] != 0.f);
  return any;
}

// MXFP4: e2m1 nibble * 2^(e8m0-127) block scale -> float, via bf16 bit
// construction with the block exponent folded into the exponent field
// (no table lookup). mag==1 is the e2m1 subnormal 0.5.
__device__ __forceinline__ float mxfp4_val(const int nib, const int ebits) {
  const int mag = nib & 7;
  const int e = mag >> 1, m = mag & 1;
  int bits = ((126 + e) << 7) | (m << 6);
  bits = (mag == 1) ? 0x3F00 : bits;
  bits = (mag == 0) ? 0 : (bits + ((ebits - 127) << 7));
  bits |= (nib >> 3) << 15;
  return bits2f((short)bits);
}

// GLU = 0: silu(g) * u   GLU = 1: gpt-oss clamped (u+1) * g * sigmoid(g*a)
template <int GLU>
__device__ __forceinline__ float apply_glu(float g, float u, const float alpha,
                                           const float limit) {
  if (GLU == 1) {
    g = fminf(g, limit);
    u = fminf(fmaxf(u, -limit), limit);
    return (u + 1.f) * g / (1.f + __expf(-g * alpha));
  }
  return g / (1.f + __expf(-g)) * u;
}

// QMODE: 0 = bf16, 1 = grouped int8, 2 = mxfp4 (nibble rows + e8m0)
// launch bounds: M<=8 keeps the tight 128-VGPR budget (measured faster
// at high occupancy); the wide tiles trade occupancy for bank-read
// amortization and need the bigger register file.
template <int M, int QMODE, bool PACKED, int GLU>
__global__ __launch_bounds__(256, M <= 8 ? 4 : 2) void moe_gateup_kernel(
    const short* __restrict__ x, const void* __restrict__ w,
    const short* __restrict__ scales, const short* __restrict__ bias,
    const float* __restrict__ we, short* __restrict__ act, const int K,
    const int I, const int E, const int Mtot, const int m0, const int G,
    const float alpha, const float limit) {
  const int e = blockIdx.z;
  if (!expert_routed<M>(we, E, e, m0)) return;
  const int wid = threadIdx.x / kWave;
  const int lane = threadIdx.x & (kWave - 1);
  const int i = blockIdx.x * kMoeWaves + wid;
  if (i >= I) return;
  const int64_t grow = (int64_t)e * 2 * I + i;      // gate row index
  const int64_t urow = grow + I;                    // up row index
  float ag[M], au[M];
#pragma unroll
  for (int m = 0; m < M; ++m) ag[m] = au[m] = 0.f;
  if (QMODE == 2) {
    const uint8_t* base = static_cast<const uint8_t*>(w);
    const int4* wg = reinterpret_cast<const int4*>(base + grow * (K / 2));
    const int4* wu = reinterpret_cast<const int4*>(base + urow * (K / 2));
    const uint8_t* sb = reinterpret_cast<const uint8_t*>(scales);
    const uint8_t* sg = sb + grow * (K / 32);
    const uint8_t* su = sb + urow * (K / 32);
    const int vecs = K / 32;   // one 16 B load = one 32-value block
    for (int v = lane; v < vecs; v += kWave) {
      i32x4 gv = *reinterpret_cast<const i32x4*>(&wg[v]);
      i32x4 uv = *reinterpret_cast<const i32x4*>(&wu[v]);
      // volatile tie: stops the software pipeliner from pre-extracting
      // the NEXT iterations' nibbles (it spilled 200+ regs otherwise);
      // the loads themselves still prefetch ahead
      asm volatile("" : "+v"(gv), "+v"(uv));
      const uint8_t* gq = reinterpret_cast<const uint8_t*>(&gv);
      const uint8_t* uq = reinterpret_cast<const uint8_t*>(&uv);
      const int ge = sg[v], ue = su[v];
      // dequant per 8-value chunk, outside the M loop (whole-block
      // arrays at gateup's 2 rows = 64 floats blew past 128 VGPR; doing
      // it inside the M loop hoisted M copies -> 1000+ scratch spills).
      // c is a REAL loop (unroll 1): only one chunk's dequant + x
      // values are live at a time
#pragma unroll 1
      for (int c = 0; c < 4; ++c) {         // 4 x 8 values per block
        float gd[8], ud[8];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const int gb = gq[c * 4 + j], ub = uq[c * 4 + j];
          gd[2 * j] = mxfp4_val(gb & 0xF, ge);
          gd[2 * j + 1] = mxfp4_val(gb >> 4, ge);
          ud[2 * j] = mxfp4_val(ub & 0xF, ue);
          ud[2 * j + 1] = mxfp4_val(ub >> 4, ue);
        }
#pragma unroll
        for (int m = 0; m < M; ++m) {
          const short8 xv = reinterpret_cast<const short8*>(
              x + (int64_t)(m0 + m) * K)[v * 4 + c];
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            const float xa = bits2f(xv.x[j]);
            ag[m] = fmaf(gd[j], xa, ag[m]);
            au[m] = fmaf(ud[j], xa, au[m]);
          }
        }
      }
    }
  } else if (QMODE == 1) {
    const int4* wg = reinterpret_cast<const int4*>(
        static_cast<const int8_t*>(w) + grow * K);
    const int4* wu = reinterpret_cast<const int4*>(
        static_cast<const int8_t*>(w) + urow * K);
    const short* sg = scales + grow * (K / G);
    const short* su = scales + urow * (K / G);
    const int vecs = K / 16;
    for (int v = lane; v < vecs; v += kWave) {
      const int4 gv = wg[v], uv = wu[v];
      const int8_t* gq = reinterpret_cast<const int8_t*>(&gv);
      const int8_t* uq = reinterpret_cast<const int8_t*>(&uv);
      const float gs = bits2f(sg[(v * 16) / G]);
      const float us = bits2f(su[(v * 16) / G]);
#pragma unroll
      for (int m = 0; m < M; ++m) {
        const short8* xr =
            reinterpret_cast<const short8*>(x + (int64_t)(m0 + m) * K);
        short8 x0, x1;
        if (PACKED) {
          const int pr = v / 4, sl = v % 4;
          x0 = xr[pr * 8 + sl];
          x1 = xr[pr * 8 + sl + 4];
        } else {
          x0 = xr[2 * v];
          x1 = xr[2 * v + 1];
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float xa = bits2f(x0.x[j]), xb = bits2f(x1.x[j]);
          ag[m] = fmaf((float)gq[j] * gs, xa, ag[m]);
          ag[m] = fmaf((float)gq[8 + j] * gs, xb, ag[m]);
          au[m] = fmaf((float)uq[j] * us, xa, au[m]);
          au[m] = fmaf((float)uq[8 + j] * us, xb, au[m]);
        }
      }
    }
  } else {
    const short8* wg =
        reinterpret_cast<const short8*>(static_cast<const short*>(w) + grow * K);
    const short8* wu =
        reinterpret_cast<const short8*>(static_cast<const short*>(w) + urow * K);
    const int vecs = K / 8;
    for (int v = lane; v < vecs; v += kWave) {
      const short8 gv = wg[v], uv = wu[v];
#pragma unroll
      for (int m = 0; m < M; ++m) {
        const short8 xv = reinterpret_cast<const short8*>(
            x + (int64_t)(m0 + m) * K)[v];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          ag[m] = fmaf(bits2f(gv.x[j]), bits2f(xv.x[j]), ag[m]);
          au[m] = fmaf(bits2f(uv.x[j]), bits2f(xv.x[j]), au[m]);
        }
      }
    }
  }
#pragma unroll
  for (int m = 0; m < M; ++m) {
    float g = wave_reduce_sum(ag[m]);
    float u = wave_reduce_sum(au[m]);
    if (lane == 0) {
      if (bias != nullptr) {
        g += bits2f(bias[grow]);
        u += bits2f(bias[urow]);
      }
      act[((int64_t)e * Mtot + m0 + m) * I + i] =
          f2bits(apply_glu<GLU>(g, u, alpha, limit));
    }
  }
}

template <int M, int QMODE, bool PACKED>
__global__ __launch_bounds__(256, 4) void moe_down_kernel(
    const short* __restrict__ act, const void* __restrict__ w,
    const short* __restrict__ scales, const short* __restrict__ bias,
    const float* __restrict__ we, float* __restrict__ out, const int I,
    const int H, const int E, const int Mtot, const int m0, const int G) {
  const int e = blockIdx.z;
  if (!expert_routed<M>(we, E, e, m0)) return;
  const int wid = threadIdx.x / kWave;
  const int lane = threadIdx.x & (kWave - 1);
  const int h = blockIdx.x * kMoeWaves + wid;
  if (h >= H) return;
  const int64_t row = (int64_t)e * H + h;
  float acc[M];
#pragma unroll
  for (int m = 0; m < M; ++m) acc[m] = 0.f;
  if (QMODE == 2) {
    const uint8_t* base = static_cast<const uint8_t*>(w);
    const int4* wr = reinterpret_cast<const int4*>(base + row * (I / 2));
    const uint8_t* sr = reinterpret_cast<const uint8_t*>(scales) +
                        row * (I / 32);
    const int vecs = I / 32;
    for (int v = lane; v < vecs; v += kWave) {
      i32x4 wv = *reinterpret_cast<const i32x4*>(&wr[v]);
      asm volatile("" : "+v"(wv));   // see gateup: stops pre-extraction
      const uint8_t* q = reinterpret_cast<const uint8_t*>(&wv);
      const int eb = sr[v];
#pragma unroll 1
      for (int c = 0; c < 4; ++c) {
        float qd[8];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const int b = q[c * 4 + j];
          qd[2 * j] = mxfp4_val(b & 0xF, eb);
          qd[2 * j + 1] = mxfp4_val(b >> 4, eb);
        }
#pragma unroll
        for (int m = 0; m < M; ++m) {
          const short8 xv = reinterpret_cast<const short8*>(
              act + ((int64_t)e * Mtot + m0 + m) * I)[v * 4 + c];
#pragma unroll
          for (int j = 0; j < 8; ++j)
            acc[m] = fmaf(qd[j], bits2f(xv.x[j]), acc[m]);
        }
      }
    }
  } else if (QMODE == 1) {
    const int4* wr = reinterpret_cast<const int4*>(
        static_cast<const int8_t*>(w) + row * I);
    const short* sr = scales + row * (I / G);
    const int vecs = I / 16;
    for (int v = lane; v < vecs; v += kWave) {
      const int4 wv = wr[v];
      const int8_t* q = reinterpret_cast<const int8_t*>(&wv);
      const float s = bits2f(sr[(v * 16) / G]);
#pragma unroll
      for (int m = 0; m < M; ++m) {
        const short8* xr = reinterpret_cast<const short8*>(
            act + ((int64_t)e * Mtot + m0 + m) * I);
        short8 x0, x1;
        if (PACKED) {
          const int pr = v / 4, sl = v % 4;
          x0 = xr[pr * 8 + sl];
          x1 = xr[pr * 8 + sl + 4];
        } else {
          x0 = xr[2 * v];
          x1 = xr[2 * v + 1];
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          acc[m] = fmaf((float)q[j] * s, bits2f(x0.x[j]), acc[m]);
          acc[m] = fmaf((float)q[8 + j] * s, bits2f(x1.x[j]), acc[m]);
        }
      }
    }
  } else {
    const short8* wr =
        reinterpret_cast<const short8*>(static_cast<const short*>(w) + row * I);
    const int vecs = I / 8;
    for (int v = lane; v < vecs; v += kWave) {
      const short8 wv = wr[v];
#pragma unroll
      for (int m = 0; m < M; ++m) {
        const short8 xv = reinterpret_cast<const short8*>(
            act + ((int64_t)e * Mtot + m0 + m) * I)[v];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[m] = fmaf(bits2f(wv.x[j]), bits2f(xv.x[j]), acc[m]);
      }
    }
  }
#pragma unroll
  for (int m = 0; m < M; ++m) {
    float r = wave_reduce_sum(acc[m]);
    if (lane == 0) {
      const float wme = we[(int64_t)(m0 + m) * E + e];
      if (wme != 0.f) {
        if (bias != nullptr) r += bits2f(bias[row]);
        atomicAdd(&out[(int64_t)(m0 + m) * H + h], r * wme);
      }
    }
  }
}

template <typename LaunchFn>
static void moe_dispatch_m(int M, LaunchFn&& fn) {
  switch (M) {
    case 1: fn(std::integral_constant<int, 1>{}); break;
    case 2: fn(std::integral_constant<int, 2>{}); break;
    case 3: fn(std::integral_constant<int, 3>{}); break;
    case 4: fn(std::integral_constant<int, 4>{}); break;
    case 6: fn(std::integral_constant<int, 6>{}); break;
    case 8: fn(std::integral_constant<int, 8>{}); break;
    case 16: fn(std::integral_constant<int, 16>{}); break;
    case 32: fn(std::integral_constant<int, 32>{}); break;
    default: TORCH_CHECK(false, "moe: unsupported M tile ", M);
  }
}

static int moe_mtile(int64_t rem, bool mx4, int cap) {
  // Wider M tiles amortize the expert-bank read (at batch 32, M=8
  // re-reads every routed bank 4x per layer) but MEASURED SLOWER:
  // gpt-oss b32 gu8/dn8 2434 tok/s vs dn32 2386 / gu16 2141 — the
  // occupancy cost (VGPR 2M accumulators) outweighs the traffic cut;
  // these kernels are latency-bound. M=16/32 tiers kept behind
  // DNET_MOE_MT_GU / DNET_MOE_MT_DN; the structural fix is an
  // LDS-staged-x MFMA expert GEMM (round-3 lever).
  if (mx4 && rem >= 6) return 6;   // M=8 mxfp4 spills ~30 regs
  if (cap >= 32 && rem >= 32) return 32;
  if (cap >= 16 && rem >= 16) return 16;
  if (rem >= 8) return 8;
  if (rem == 7 || rem == 5) return 4;
  return (int)rem;
}

static int moe_cap(const char* env, int dflt) {
  const char* e = getenv(env);
  return e ? atoi(e) : dflt;
}

void moe_gateup(torch::Tensor x, torch::Tensor w,
                c10::optional<torch::Tensor> scales,
                c10::optional<torch::Tensor> bias, torch::Tensor we,
                torch::Tensor act, int64_t group, bool packed, int64_t glu,
                double alpha, double limit) {
  const int64_t M = x.size(0), K = x.size(1);
  const int64_t E = w.size(0), I2 = w.size(1), I = I2 / 2;
  // weight dtype selects the mode: bf16, int8 (kChar), mxfp4 nibble rows
  // (kByte, [E, 2I, K/2] + e8m0 uint8 scales [E, 2I, K/32])
  const bool mx4 = w.scalar_type() == torch::kByte;
  const bool q8 = w.scalar_type() == torch::kChar;
  DNET_CHECK(w.size(2) == (mx4 ? K / 2 : K) && act.size(0) == E &&
                 act.size(1) == M && act.size(2) == I, "moe_gateup shape");
  DNET_CHECK(we.size(0) == M && we.size(1) == E, "we shape");
  DNET_CHECK(x.is_contiguous() && w.is_contiguous() && we.is_contiguous() &&
                 act.is_contiguous(), "contig");
  DNET_CHECK(K % (q8 ? 16 : (mx4 ? 32 : 8)) == 0, "K align");
  if (packed) DNET_CHECK(K % 64 == 0 && group % 64 == 0, "packed align");
  const short* sptr = nullptr;
  if (q8) {
    DNET_CHECK(scales.has_value() && group > 0 && K % group == 0, "scales");
    DNET_CHECK(scales->is_contiguous() && scales->size(0) == E &&
                   scales->size(1) == I2 && scales->size(2) == K / group,
               "scales shape");
    sptr = (const short*)scales->data_ptr();
  } else if (mx4) {
    DNET_CHECK(scales.has_value() && scales->scalar_type() == torch::kByte &&
                   scales->is_contiguous() && scales->size(0) == E &&
                   scales->size(1) == I2 && scales->size(2) == K / 32,
               "mxfp4 scales shape");
    sptr = (const short*)scales->data_ptr();
  }
  const short* bptr = bias.has_value() ? (const short*)bias->data_ptr() : nullptr;
  auto stream = current_stream();
  const dim3 grid(cdiv((int)I, kMoeWaves), 1, (unsigned)E);
  static const int gu_cap = moe_cap("DNET_MOE_MT_GU", 8);
  int64_t m0 = 0;
  while (m0 < M) {
    const int mt = moe_mtile(M - m0, mx4, gu_cap);
    moe_dispatch_m(mt, [&](auto mc) {
      constexpr int MV = decltype(mc)::value;
      auto launch = [&](auto qmc, auto pkc, auto gluc) {
        hipLaunchKernelGGL(
            (moe_gateup_kernel<MV, decltype(qmc)::value, decltype(pkc)::value,
                               decltype(gluc)::value>),
            grid, dim3(kMoeWaves * kWave), 0, stream,
            (const short*)x.data_ptr(), w.data_ptr(), sptr, bptr,
            (const float*)we.data_ptr(), (short*)act.data_ptr(), (int)K,
            (int)I, (int)E, (int)M, (int)m0, (int)std::max<int64_t>(group, 16),
            (float)alpha, (float)limit);
      };
      auto with_glu = [&](auto qmc, auto pkc) {
        if (glu == 1)
          launch(qmc, pkc, std::integral_constant<int, 1>{});
        else
          launch(qmc, pkc, std::integral_constant<int, 0>{});
      };
      if (q8) {
        if (packed)
          with_glu(std::integral_constant<int, 1>{}, std::true_type{});
        else
          with_glu(std::integral_constant<int, 1>{}, std::false_type{});
      } else if (mx4) {
        with_glu(std::integral_constant<int, 2>{}, std::false_type{});
      } else {
        with_glu(std::integral_constant<int, 0>{}, std::false_type{});
      }
    });
    m0 += mt;
  }
}

void moe_down(torch::Tensor act, torch::Tensor w,
              c10::optional<torch::Tensor> scales,
              c10::optional<torch::Tensor> bias, torch::Tensor we,
              torch::Tensor out, int64_t group, bool packed) {
  const int64_t E = w.size(0), H = w.size(1);
  const int64_t M = act.size(1), I = act.size(2);
  const bool mx4 = w.scalar_type() == torch::kByte;
  const bool q8 = w.scalar_type() == torch::kChar;
  DNET_CHECK(act.size(0) == E && w.size(2) == (mx4 ? I / 2 : I),
             "moe_down act/w shape");
  DNET_CHECK(out.size(0) == M && out.size(1) == H &&
                 out.scalar_type() == torch::kFloat, "moe_down out f32");
  DNET_CHECK(we.size(0) == M && we.size(1) == E, "we shape");
  DNET_CHECK(act.is_contiguous() && w.is_contiguous() && we.is_contiguous() &&
                 out.is_contiguous(), "contig");
  DNET_CHECK(I % (q8 ? 16 : (mx4 ? 32 : 8)) == 0, "I align");
  if (packed) DNET_CHECK(I % 64 == 0 && group % 64 == 0, "packed align");
  const short* sptr = nullptr;
  if (q8) {
    DNET_CHECK(scales.has_value() && group > 0 && I % group == 0, "scales");
    DNET_CHECK(scales->is_contiguous() && scales->size(0) == E &&
                   scales->size(1) == H && scales->size(2) == I / group,
               "scales shape");
    sptr = (const short*)scales->data_ptr();
  } else if (mx4) {
    DNET_CHECK(scales.has_value() && scales->scalar_type() == torch::kByte &&
                   scales->is_contiguous() && scales->size(0) == E &&
                   scales->size(1) == H && scales->size(2) == I / 32,
               "mxfp4 scales shape");
    sptr = (const short*)scales->data_ptr();
  }
  const short* bptr = bias.has_value() ? (const short*)bias->data_ptr() : nullptr;
  auto stream = current_stream();
  const dim3 grid(cdiv((int)H, kMoeWaves), 1, (unsigned)E);
  static const int dn_cap = moe_cap("DNET_MOE_MT_DN", 8);
  int64_t m0 = 0;
  while (m0 < M) {
    const int mt = moe_mtile(M - m0, mx4, dn_cap);
    moe_dispatch_m(mt, [&](auto mc) {
      constexpr int MV = decltype(mc)::value;
      auto launch = [&](auto qmc, auto pkc) {
        hipLaunchKernelGGL(
            (moe_down_kernel<MV, decltype(qmc)::value, decltype(pkc)::value>),
            grid, dim3(kMoeWaves * kWave), 0, stream,
            (const short*)act.data_ptr(), w.data_ptr(), sptr, bptr,
            (const float*)we.data_ptr(), (float*)out.data_ptr(), (int)I,
            (int)H, (int)E, (int)M, (int)m0,
            (int)std::max<int64_t>(group, 16));
      };
      if (q8) {
        if (packed)
          launch(std::integral_constant<int, 1>{}, std::true_type{});
        else
          launch(std::integral_constant<int, 1>{}, std::false_type{});
      } else if (mx4) {
        launch(std::integral_constant<int, 2>{}, std::false_type{});
      } else {
        launch(std::integral_constant<int, 0>{}, std::false_type{});
      }
    });
    m0 += mt;
  }
}

__global__ void dequant_mxfp4_kernel(const uint8_t* __restrict__ w,
                                     const uint8_t* __restrict__ scales,
                                     short* __restrict__ out,
                                     const int64_t nblocks, const int bpr) {
  // one thread = one 32-value block (16 B in, 64 B out)
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       i < nblocks; i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t row = i / bpr;
    const int blk = (int)(i % bpr);
    const int4 v = reinterpret_cast<const int4*>(w)[i];
    const uint8_t* q = reinterpret_cast<const uint8_t*>(&v);
    const int eb = scales[row * bpr + blk];
    short* o = out + (row * bpr + blk) * 32;
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      o[2 * j] = f2bits(mxfp4_val(q[j] & 0xF, eb));
      o[2 * j + 1] = f2bits(mxfp4_val(q[j] >> 4, eb));
    }
  }
}

void dequant_mxfp4(torch::Tensor w, torch::Tensor scales, torch::Tensor out) {
  const int64_t rows = w.size(0), kb = w.size(1);
  DNET_CHECK(w.scalar_type() == torch::kByte &&
                 scales.scalar_type() == torch::kByte, "mxfp4 dtypes");
  DNET_CHECK(kb % 16 == 0 && scales.size(1) == kb / 16 &&
                 out.size(0) == rows && out.size(1) == kb * 2, "shapes");
  DNET_CHECK(w.is_contiguous() && scales.is_contiguous() &&
                 out.is_contiguous(), "contig");
  const int64_t nblocks = rows * (kb / 16);
  const int grid = (int)std::min<int64_t>((nblocks + 255) / 256, 4096);
  hipLaunchKernelGGL(dequant_mxfp4_kernel, dim3(grid), dim3(256), 0,
                     current_stream(), (const uint8_t*)w.data_ptr(),
                     (const uint8_t*)scales.data_ptr(),
                     (short*)out.data_ptr(), nblocks, (int)(kb / 16));
}

}  // namespace dnet
