// Fused SwiGLU: y = silu(gate) * up, reading the fused [T, 2I] gate/up
// projection output (gate = [:, :I], up = [:, I:]).  Memory-bound
// elementwise; vectorized 16 B loads, grid-stride (CDNA4 guide G13).
//
// Reference equivalent: mlx_lm MLP silu(gate)*up (reference:
// src/dnet/core/models/llama.py TransformerBlock MLP).
#include "common.h"

namespace dnet {

__global__ void swiglu_kernel(const short* __restrict__ gu,
                              short* __restrict__ y, const int64_t I,
                              const int64_t T) {
  const int64_t nvec = T * I / 8;
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; idx < nvec;
       idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t row = idx / (I / 8);
    const int64_t col = idx % (I / 8);
    const short8 g = reinterpret_cast<const short8*>(gu + row * 2 * I)[col];
    const short8 u = reinterpret_cast<const short8*>(gu + row * 2 * I + I)[col];
    short8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float gf = bits2f(g.x[j]);
      const float uf = bits2f(u.x[j]);
      const float silu = gf / (1.f + __expf(-gf));
      o.x[j] = f2bits(silu * uf);
    }
    reinterpret_cast<short8*>(y + row * I)[col] = o;
  }
}

// Split-k-fused variant: reads the UN-COMBINED f32 split-k scratch of
// the gate/up GEMM directly (bias-free), re-zeroing it for the next
// split-k use — removes the f32->bf16 combine kernel from the decode
// dependency chain (each ~5 us launch-bound hop, 64 per step).
__global__ void swiglu_f32_kernel(float* __restrict__ gu,
                                  short* __restrict__ y, const int64_t I,
                                  const int64_t T) {
  const int64_t nvec = T * I / 4;
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       idx < nvec; idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t row = idx / (I / 4);
    const int64_t col = idx % (I / 4);
    float4 g = reinterpret_cast<float4*>(gu + row * 2 * I)[col];
    float4 u = reinterpret_cast<float4*>(gu + row * 2 * I + I)[col];
    reinterpret_cast<float4*>(gu + row * 2 * I)[col] =
        make_float4(0.f, 0.f, 0.f, 0.f);
    reinterpret_cast<float4*>(gu + row * 2 * I + I)[col] =
        make_float4(0.f, 0.f, 0.f, 0.f);
    short4v o;
    const float gf[4] = {g.x, g.y, g.z, g.w};
    const float uf[4] = {u.x, u.y, u.z, u.w};
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float silu = gf[j] / (1.f + __expf(-gf[j]));
      o.x[j] = f2bits(silu * uf[j]);
    }
    reinterpret_cast<short4v*>(y + row * I)[col] = o;
  }
}

void swiglu_f32(torch::Tensor gu_f32, torch::Tensor y, int64_t N) {
  const int64_t I = y.size(-1);
  const int64_t T = y.numel() / I;
  DNET_CHECK(N == 2 * I, "gu width must be 2*I");
  DNET_CHECK(I % 4 == 0, "I % 4");
  DNET_CHECK(gu_f32.numel() >= T * N, "scratch too small");
  DNET_CHECK(gu_f32.is_contiguous() && y.is_contiguous(), "contig");
  auto stream = current_stream();
  const int64_t nvec = T * I / 4;
  const int grid = (int)std::min<int64_t>((nvec + 255) / 256, 2048);
  hipLaunchKernelGGL(swiglu_f32_kernel, dim3(grid), dim3(256), 0, stream,
                     (float*)gu_f32.data_ptr(), (short*)y.data_ptr(), I, T);
}

void swiglu(torch::Tensor gu, torch::Tensor y) {
  const int64_t I = y.size(-1);
  const int64_t T = y.numel() / I;
  DNET_CHECK(gu.size(-1) == 2 * I, "gu last dim must be 2*I");
  DNET_CHECK(I % 8 == 0, "I % 8");
  DNET_CHECK(gu.is_contiguous() && y.is_contiguous(), "contig");
  auto stream = current_stream();
  const int64_t nvec = T * I / 8;
  const int grid = (int)std::min<int64_t>((nvec + 255) / 256, 2048);
  hipLaunchKernelGGL(swiglu_kernel, dim3(grid), dim3(256), 0, stream,
                     (const short*)gu.data_ptr(), (short*)y.data_ptr(), I, T);
}

template <bool PACKED>
__global__ void dequant_int8_kernel(const int8_t* __restrict__ w,
                                    const short* __restrict__ scales,
                                    short* __restrict__ out, const int64_t K,
                                    const int G, const int64_t N) {
  const int64_t nvec = N * K / 16;
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; idx < nvec;
       idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t n = idx / (K / 16);
    const int64_t k0 = (idx % (K / 16)) * 16;
    const int4 wv = reinterpret_cast<const int4*>(w + n * K)[k0 / 16];
    const int8_t* q = reinterpret_cast<const int8_t*>(&wv);
    const float s = bits2f(scales[n * (K / G) + k0 / G]);
    short8 o0, o1;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      o0.x[j] = f2bits((float)q[j] * s);
      o1.x[j] = f2bits((float)q[8 + j] * s);
    }
    if (PACKED) {
      // packed vec covers orig k = p*64 + s*8 (low 8) and +32 (high 8)
      const int64_t pr = k0 / 64, sl = (k0 / 16) % 4;
      const int64_t ka = pr * 64 + sl * 8;
      reinterpret_cast<short8*>(out + n * K)[ka / 8] = o0;
      reinterpret_cast<short8*>(out + n * K)[(ka + 32) / 8] = o1;
    } else {
      reinterpret_cast<short8*>(out + n * K)[k0 / 8] = o0;
      reinterpret_cast<short8*>(out + n * K)[k0 / 8 + 1] = o1;
    }
  }
}

// Dequantize grouped-int8 weights to bf16 (prefill path: the dequantized
// tile feeds a hipBLASLt GEMM via torch.matmul).
void dequant_int8(torch::Tensor w, torch::Tensor scales, torch::Tensor out,
                  int64_t group, bool packed) {
  const int64_t N = w.size(0), K = w.size(1);
  DNET_CHECK(K % 16 == 0 && group % 16 == 0, "K align");
  if (packed) DNET_CHECK(K % 64 == 0 && group % 64 == 0, "packed align");
  DNET_CHECK(out.size(0) == N && out.size(1) == K, "out shape");
  DNET_CHECK(w.is_contiguous() && scales.is_contiguous() && out.is_contiguous(), "contig");
  auto stream = current_stream();
  const int64_t nvec = N * K / 16;
  const int grid = (int)std::min<int64_t>((nvec + 255) / 256, 2048);
  if (packed)
    hipLaunchKernelGGL(dequant_int8_kernel<true>, dim3(grid), dim3(256), 0, stream,
                       (const int8_t*)w.data_ptr(), (const short*)scales.data_ptr(),
                       (short*)out.data_ptr(), K, (int)group, N);
  else
    hipLaunchKernelGGL(dequant_int8_kernel<false>, dim3(grid), dim3(256), 0, stream,
                       (const int8_t*)w.data_ptr(), (const short*)scales.data_ptr(),
                       (short*)out.data_ptr(), K, (int)group, N);
}

// int4 chunk-quad packed -> bf16 (prefill path). Packed byte at
// quad*64 + slice*16 + chunk*4 + jb holds orig k = quad*128 + chunk*32 +
// slice*8 + 2*jb (low nibble) and +1 (high nibble), offset-8.
__global__ void dequant_int4_kernel(const uint8_t* __restrict__ w,
                                    const short* __restrict__ scales,
                                    short* __restrict__ out, const int64_t K,
                                    const int G, const int64_t N) {
  const int64_t nbytes = N * K / 2;
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       idx < nbytes; idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t n = idx / (K / 2);
    const int64_t bi = idx % (K / 2);
    const int64_t quad = bi / 64;
    const int rem = (int)(bi % 64);
    const int slice = rem / 16, chunk = (rem % 16) / 4, jb = rem % 4;
    const int64_t k0 = quad * 128 + chunk * 32 + slice * 8 + 2 * jb;
    const uint8_t byte = w[idx];
    const float s = bits2f(scales[n * (K / G) + k0 / G]);
    out[n * K + k0] = f2bits((float)((byte & 0xF) - 8) * s);
    out[n * K + k0 + 1] = f2bits((float)((byte >> 4) - 8) * s);
  }
}

void dequant_int4(torch::Tensor w, torch::Tensor scales, torch::Tensor out,
                  int64_t group) {
  const int64_t N = w.size(0), K = out.size(1);
  DNET_CHECK(w.size(1) == K / 2 && K % 128 == 0 && group % 128 == 0, "shape");
  DNET_CHECK(w.is_contiguous() && scales.is_contiguous() && out.is_contiguous(), "contig");
  auto stream = current_stream();
  const int64_t nbytes = N * K / 2;
  const int grid = (int)std::min<int64_t>((nbytes + 255) / 256, 2048);
  hipLaunchKernelGGL(dequant_int4_kernel, dim3(grid), dim3(256), 0, stream,
                     (const uint8_t*)w.data_ptr(), (const short*)scales.data_ptr(),
                     (short*)out.data_ptr(), K, (int)group, N);
}

}  // namespace dnet
