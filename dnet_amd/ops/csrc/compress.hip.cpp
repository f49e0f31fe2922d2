// Column-sparsification wire-compression kernels (CDNA4).
//
// MI355X equivalents of the reference's 9 JIT Metal kernels
// (reference: src/dnet/compression/kernels.py — col_norm2, apply_mask,
// gather_cols, scatter_from_compact, ...): drop the smallest-L2-norm
// columns of an activation [R, D] before a slow wire hop. Fused into three
// kernels: per-column norms (coalesced row-major sweep), gather-pack and
// zero+scatter. Column selection (top-k) runs on host via torch.topk.
#include "common.h"

namespace dnet {

// norms[d] = sum_r x[r][d]^2 ; one thread per column, threads read
// consecutive columns so each row sweep is fully coalesced.
__global__ void col_norm2_kernel(const short* __restrict__ x,
                                 float* __restrict__ norms, const int R,
                                 const int D) {
  const int d = blockIdx.x * blockDim.x + threadIdx.x;
  if (d >= D) return;
  float acc = 0.f;
  for (int r = 0; r < R; ++r) {
    const float v = bits2f(x[(int64_t)r * D + d]);
    acc = fmaf(v, v, acc);
  }
  norms[d] = acc;
}

// out[r][j] = x[r][idx[j]]
__global__ void gather_cols_kernel(const short* __restrict__ x,
                                   const int* __restrict__ idx,
                                   short* __restrict__ out, const int R,
                                   const int D, const int K) {
  const int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  if (i >= (int64_t)R * K) return;
  const int r = (int)(i / K);
  const int j = (int)(i % K);
  out[i] = x[(int64_t)r * D + idx[j]];
}

// out zeroed except out[r][idx[j]] = in[r][j]
__global__ void scatter_cols_kernel(const short* __restrict__ in,
                                    const int* __restrict__ idx,
                                    short* __restrict__ out, const int R,
                                    const int D, const int K) {
  const int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  if (i >= (int64_t)R * K) return;
  const int r = (int)(i / K);
  const int j = (int)(i % K);
  out[(int64_t)r * D + idx[j]] = in[i];
}

void col_norm2(torch::Tensor x, torch::Tensor norms) {
  const int64_t D = x.size(-1), R = x.numel() / D;
  DNET_CHECK(norms.numel() == D && norms.dtype() == torch::kFloat32, "norms");
  auto stream = current_stream();
  hipLaunchKernelGGL(col_norm2_kernel, dim3(cdiv(D, 256)), dim3(256), 0,
                     stream, (const short*)x.data_ptr(),
                     (float*)norms.data_ptr(), (int)R, (int)D);
}

void gather_cols(torch::Tensor x, torch::Tensor idx, torch::Tensor out) {
  const int64_t D = x.size(-1), R = x.numel() / D, K = idx.numel();
  DNET_CHECK(out.numel() == R * K, "out shape");
  DNET_CHECK(idx.dtype() == torch::kInt32, "idx int32");
  auto stream = current_stream();
  const int64_t total = R * K;
  hipLaunchKernelGGL(gather_cols_kernel, dim3(cdiv(total, 256)), dim3(256), 0,
                     stream, (const short*)x.data_ptr(),
                     (const int*)idx.data_ptr(), (short*)out.data_ptr(),
                     (int)R, (int)D, (int)K);
}

void scatter_cols(torch::Tensor in, torch::Tensor idx, torch::Tensor out) {
  const int64_t D = out.size(-1), R = out.numel() / D, K = idx.numel();
  DNET_CHECK(in.numel() == R * K, "in shape");
  auto stream = current_stream();
  DNET_CHECK_HIP(hipMemsetAsync(out.data_ptr(), 0,
                                out.numel() * out.element_size(), stream));
  const int64_t total = R * K;
  hipLaunchKernelGGL(scatter_cols_kernel, dim3(cdiv(total, 256)), dim3(256),
                     0, stream, (const short*)in.data_ptr(),
                     (const int*)idx.data_ptr(), (short*)out.data_ptr(),
                     (int)R, (int)D, (int)K);
}

}  // namespace dnet
