// Fused RMSNorm (+ optional residual add) for bf16 rows, f32 accumulation.
//
// y = (x / rms(x)) * w         (residual == nullptr)
// r = r + x; y = (r / rms(r)) * w   (fused residual update, in-place on r)
//
// MI355X-native: memory-bound; vectorized short8 (16 B/lane) loads
// (cdna_hip_programming.md G13: scalar bf16 loads are ~2x slower), one
// workgroup per row, grid-stride over rows.
#include "common.h"

namespace dnet {

// F32X: x comes from the UN-COMBINED f32 split-k scratch of the
// producing GEMM (o/down projection) and is re-zeroed after the read —
// the f32->bf16 combine kernel disappears from the decode chain.
template <bool HAS_RES, bool F32X = false>
__global__ void rmsnorm_kernel(const short* __restrict__ x,
                               short* __restrict__ res,  // in/out residual
                               const short* __restrict__ w,
                               short* __restrict__ y,
                               const int H, const float eps, const int T,
                               float* __restrict__ xf = nullptr) {
  __shared__ float scratch[16];
  const int vecs = H / 8;  // H % 8 == 0 enforced on host
  for (int row = blockIdx.x; row < T; row += gridDim.x) {
    const short8* xv = reinterpret_cast<const short8*>(x + (int64_t)row * H);
    short8* rv = HAS_RES ? reinterpret_cast<short8*>(res + (int64_t)row * H) : nullptr;
    float ss = 0.f;
    for (int i = threadIdx.x; i < vecs; i += blockDim.x) {
      short8 v;
      if (F32X) {
        float4* xr = reinterpret_cast<float4*>(xf + (int64_t)row * H + i * 8);
        const float4 a = xr[0], bq = xr[1];
        xr[0] = make_float4(0.f, 0.f, 0.f, 0.f);
        xr[1] = make_float4(0.f, 0.f, 0.f, 0.f);
        const float av[8] = {a.x, a.y, a.z, a.w, bq.x, bq.y, bq.z, bq.w};
#pragma unroll
        for (int j = 0; j < 8; ++j) v.x[j] = f2bits(av[j]);
      } else {
        v = xv[i];
      }
      if (HAS_RES) {
        short8 rr = rv[i];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = bits2f(v.x[j]) + bits2f(rr.x[j]);
          v.x[j] = f2bits(f);
        }
        rv[i] = v;  // residual stream updated in bf16
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float f = bits2f(v.x[j]);
        ss += f * f;
      }
    }
    ss = block_reduce_sum(ss, scratch);
    const float scale = rsqrtf(ss / (float)H + eps);
    const short8* src = HAS_RES ? rv : xv;
    const short8* wv = reinterpret_cast<const short8*>(w);
    short8* yv = reinterpret_cast<short8*>(y + (int64_t)row * H);
    for (int i = threadIdx.x; i < vecs; i += blockDim.x) {
      short8 v = src[i];
      short8 ww = wv[i];
      short8 out;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        out.x[j] = f2bits(bits2f(v.x[j]) * scale * bits2f(ww.x[j]));
      yv[i] = out;
    }
    __syncthreads();
  }
}

// h += x where x is the f32 split-k scratch (re-zeroed): the final
// residual update of a decode window whose last down-projection was
// left deferred in scratch.
__global__ void resid_add_f32_kernel(short* __restrict__ h,
                                     float* __restrict__ xf,
                                     const int64_t total) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const float v = xf[i];
    xf[i] = 0.f;
    h[i] = f2bits(bits2f(h[i]) + v);
  }
}

void resid_add_f32(torch::Tensor h, torch::Tensor xf) {
  const int64_t total = h.numel();
  DNET_CHECK(h.is_contiguous() && xf.is_contiguous(), "contig");
  DNET_CHECK(xf.numel() >= total, "scratch too small");
  auto stream = current_stream();
  const int grid = (int)std::min<int64_t>((total + 255) / 256, 2048);
  hipLaunchKernelGGL(resid_add_f32_kernel, dim3(grid), dim3(256), 0, stream,
                     (short*)h.data_ptr(), (float*)xf.data_ptr(), total);
}

// Split-k-fused variant: x read (and re-zeroed) from the f32 scratch.
void rmsnorm_f32(torch::Tensor xf, torch::Tensor residual, torch::Tensor w,
                 torch::Tensor y, double eps) {
  const int64_t H = residual.size(-1);
  const int64_t T = residual.numel() / H;
  DNET_CHECK(H % 8 == 0, "H % 8");
  DNET_CHECK(xf.numel() >= T * H, "scratch too small");
  DNET_CHECK(residual.is_contiguous() && y.is_contiguous() &&
                 w.is_contiguous() && xf.is_contiguous(), "contig");
  auto stream = current_stream();
  const int grid = (int)std::min<int64_t>(T, 2048);
  hipLaunchKernelGGL((rmsnorm_kernel<true, true>), dim3(grid), dim3(256), 0,
                     stream, nullptr, (short*)residual.data_ptr(),
                     (const short*)w.data_ptr(), (short*)y.data_ptr(),
                     (int)H, (float)eps, (int)T, (float*)xf.data_ptr());
}

// y = rmsnorm(x) * w ; if residual is given: residual += x (in place), then
// y = rmsnorm(residual) * w.
void rmsnorm(torch::Tensor x, c10::optional<torch::Tensor> residual,
             torch::Tensor w, torch::Tensor y, double eps) {
  DNET_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16, "x must be bf16 on GPU");
  const int64_t H = x.size(-1);
  const int64_t T = x.numel() / H;
  DNET_CHECK(H % 8 == 0, "H must be a multiple of 8");
  DNET_CHECK(x.is_contiguous() && y.is_contiguous() && w.is_contiguous(), "contiguous");
  auto stream = current_stream();
  const int block = 256;
  const int grid = std::min<int64_t>(T, 2048);
  if (residual.has_value()) {
    DNET_CHECK(residual->is_contiguous(), "residual contiguous");
    hipLaunchKernelGGL(rmsnorm_kernel<true>, dim3(grid), dim3(block), 0, stream,
                       (const short*)x.data_ptr(), (short*)residual->data_ptr(),
                       (const short*)w.data_ptr(), (short*)y.data_ptr(),
                       (int)H, (float)eps, (int)T);
  } else {
    hipLaunchKernelGGL(rmsnorm_kernel<false>, dim3(grid), dim3(block), 0, stream,
                       (const short*)x.data_ptr(), nullptr,
                       (const short*)w.data_ptr(), (short*)y.data_ptr(),
                       (int)H, (float)eps, (int)T);
  }
}

}  // namespace dnet
