// Standalone probe of __builtin_amdgcn_global_load_lds semantics on
// gfx950 (no torch): verifies (a) the LDS destination mapping
// (wave-uniform base + lane*16), (b) that the DMA retires on vmcnt, and
// (c) that ds_read still works after m0 was used by the glds.
//
// Build: hipcc --offload-arch=gfx950 -O3 scripts/glds_probe.hip.cpp -o /tmp/glds_probe
#include <hip/hip_runtime.h>
#include <cstdio>

__global__ void probe_kernel(const int* __restrict__ src, int* __restrict__ dst,
                             int* __restrict__ dst2) {
  __shared__ int lds[64 * 4 * 4];  // 4 waves x 64 lanes x 16B
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  // each lane sources chunk (wave*64 + lane) with a per-lane swizzled
  // global offset: chunk c reads src chunk c ^ 3
  const int c = wave * 64 + lane;
  const int cs = c ^ 3;
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)(src + cs * 4),
      (__attribute__((address_space(3))) void*)&lds[wave * 64 * 4], 16, 0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  // read back linearly: dst[chunk c] should hold src[c ^ 3]
#pragma unroll
  for (int j = 0; j < 4; ++j) dst[c * 4 + j] = lds[c * 4 + j];
  // (c) counted-wait path: issue glds, then 2 dummy global loads, wait
  // vmcnt(2) (leaves the dummies in flight), read LDS
  __syncthreads();
  __shared__ int lds2[64 * 4 * 4];
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)(src + c * 4),
      (__attribute__((address_space(3))) void*)&lds2[wave * 64 * 4], 16, 0, 0);
  int d0, d1;
  asm volatile("global_load_dword %0, %2, off\n\t"
               "global_load_dword %1, %2, off offset:4"
               : "=v"(d0), "=v"(d1)
               : "v"(src));
  asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  asm volatile("s_barrier" ::: "memory");
#pragma unroll
  for (int j = 0; j < 4; ++j) dst2[c * 4 + j] = lds2[c * 4 + j];
  asm volatile("s_waitcnt vmcnt(0)" : "+v"(d0), "+v"(d1)::"memory");
  if (threadIdx.x == 0 && d0 != src[0]) dst2[0] = -999;
}

int main() {
  const int n = 256 * 4;
  int *src, *dst, *dst2;
  hipMalloc(&src, n * 4);
  hipMalloc(&dst, n * 4);
  hipMalloc(&dst2, n * 4);
  int host[n];
  for (int i = 0; i < n; ++i) host[i] = i;
  hipMemcpy(src, host, n * 4, hipMemcpyHostToDevice);
  hipMemset(dst, 0xff, n * 4);
  hipMemset(dst2, 0xff, n * 4);
  probe_kernel<<<1, 256>>>(src, dst, dst2);
  hipError_t e = hipDeviceSynchronize();
  printf("kernel rc=%d (%s)\n", e, hipGetErrorString(e));
  int out[n], out2[n];
  hipMemcpy(out, dst, n * 4, hipMemcpyDeviceToHost);
  hipMemcpy(out2, dst2, n * 4, hipMemcpyDeviceToHost);
  int bad = 0, bad2 = 0;
  for (int c = 0; c < 256; ++c)
    for (int j = 0; j < 4; ++j) {
      if (out[c * 4 + j] != (c ^ 3) * 4 + j) {
        if (bad < 5)
          printf("swizzle mismatch: chunk %d j %d got %d want %d\n", c, j,
                 out[c * 4 + j], (c ^ 3) * 4 + j);
        ++bad;
      }
      if (out2[c * 4 + j] != c * 4 + j) {
        if (bad2 < 5)
          printf("counted-wait mismatch: chunk %d j %d got %d want %d\n", c,
                 j, out2[c * 4 + j], c * 4 + j);
        ++bad2;
      }
    }
  printf("glds swizzle: %s (%d bad)\n", bad ? "FAIL" : "PASS", bad);
  printf("glds counted wait: %s (%d bad)\n", bad2 ? "FAIL" : "PASS", bad2);
  return bad + bad2 ? 1 : 0;
}
