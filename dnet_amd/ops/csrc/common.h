// Common helpers for dnet_amd HIP kernels (gfx950 / CDNA4 only).
//
// All kernels in this extension are written directly for MI355X: wave64,
// vectorized bf16 loads (short4/short8 reinterpret), f32 accumulation,
// grid sized for 256 CUs in 8 XCDs. No CUDA compatibility paths.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

#define DNET_CHECK(cond, msg) TORCH_CHECK(cond, msg)
#define DNET_CHECK_HIP(call)                                              \
  do {                                                                    \
    hipError_t _e = (call);                                               \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e));  \
  } while (0)

namespace dnet {

constexpr int kWave = 64;  // CDNA wavefront

using bf16 = __hip_bfloat16;

// Vector types for wide loads (16 B = 8 bf16 per lane).
struct alignas(16) short8 { short x[8]; };
struct alignas(8) short4v { short x[4]; };

__device__ __forceinline__ float b2f(const bf16 v) { return __bfloat162float(v); }
__device__ __forceinline__ bf16 f2b(const float v) { return __float2bfloat16(v); }

// bf16 bits -> float without library calls
__device__ __forceinline__ float bits2f(const short s) {
  union { unsigned u; float f; } cvt;
  cvt.u = (unsigned)(unsigned short)s << 16;
  return cvt.f;
}
__device__ __forceinline__ short f2bits(const float f) {
  // round-to-nearest-even bf16
  union { float f; unsigned u; } cvt;
  cvt.f = f;
  unsigned u = cvt.u;
  unsigned rounding = 0x7FFF + ((u >> 16) & 1);
  return (short)((u + rounding) >> 16);
}

// Wave-wide f32 sum reduction (64 lanes).
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;  // valid in lane 0
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, 64));
  return v;  // valid in lane 0
}

// Block-wide sum using LDS scratch (scratch must hold >= blockDim.x/64 floats).
__device__ __forceinline__ float block_reduce_sum(float v, float* scratch) {
  const int lane = threadIdx.x & (kWave - 1);
  const int wid = threadIdx.x / kWave;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  const int nw = blockDim.x / kWave;
  float r = 0.f;
  if (threadIdx.x < nw) r = scratch[threadIdx.x];
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) r += __shfl_down(r, off, 64);
  r = __shfl(r, 0, 64);
  if (threadIdx.x == 0) scratch[0] = r;
  __syncthreads();
  r = scratch[0];
  __syncthreads();
  return r;
}

inline int cdiv(int64_t a, int64_t b) { return (int)((a + b - 1) / b); }

// The torch-ROCm stream for the current device (what torch kernels launch on).
inline hipStream_t current_stream() {
  return c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

}  // namespace dnet
