#include "hip/hip_runtime.h"
// K8a — fused per-channel PReLU forward/backward for NHWC (channels_last).
//
// torch's prelu_backward on channels_last bf16 was 40% of the whole
// training step on MI355X (profiles/r01_bench_b256_kernel_stats.md).
// Each thread owns 8 consecutive channels (16-B vector I/O, guide G13),
// accumulates the weight grad in registers, one LDS + one global atomic
// per owned channel.  C power of two, 8 <= C <= 1024 (python guards).
//
//   fwd: y = x > 0 ? x : a[c] * x
//   bwd: dx = x > 0 ? g : a[c] * g;  da[c] = sum_{x<=0} x * g
#include "common.h"
#include "vec8.h"

template <typename T>
__global__ void prelu_fwd_kernel(const T* __restrict__ x,
                                 const float* __restrict__ a,
                                 T* __restrict__ y, int64_t n_pix, int C) {
  ChanMap m = chan_map8(C);
  float av[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) av[j] = a[m.c0 + j];
  float v[8], o[8];
  for (int64_t p = m.p0; p < n_pix; p += m.pstep) {
    int64_t i = p * C + m.c0;
    load8(x, i, v);
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = v[j] > 0.f ? v[j] : av[j] * v[j];
    store8(y, i, o);
  }
}

template <typename T>
__global__ void prelu_bwd_kernel(const T* __restrict__ x,
                                 const T* __restrict__ g,
                                 const float* __restrict__ a,
                                 T* __restrict__ dx,
                                 float* __restrict__ da, int64_t n_pix,
                                 int C) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* da_lds = (float*)smem_raw;
  for (int c = threadIdx.x; c < C; c += blockDim.x) da_lds[c] = 0.f;
  __syncthreads();
  ChanMap m = chan_map8(C);
  float av[8], acc[8] = {};
#pragma unroll
  for (int j = 0; j < 8; ++j) av[j] = a[m.c0 + j];
  float xv[8], gv[8], dxv[8];
  for (int64_t p = m.p0; p < n_pix; p += m.pstep) {
    int64_t i = p * C + m.c0;
    load8(x, i, xv);
    load8(g, i, gv);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      dxv[j] = xv[j] > 0.f ? gv[j] : av[j] * gv[j];
      if (xv[j] <= 0.f) acc[j] += xv[j] * gv[j];
    }
    store8(dx, i, dxv);
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) atomicAdd(&da_lds[m.c0 + j], acc[j]);
  __syncthreads();
  for (int c = threadIdx.x; c < C; c += blockDim.x)
    if (da_lds[c] != 0.f) atomicAdd(&da[c], da_lds[c]);
}

extern "C" void bdbnn_prelu_fwd(const void* x, const float* a, void* y,
                                int64_t n, int C, bool bf16,
                                hipStream_t stream) {
  int64_t n_pix = n / C;
  int grid = grid_pix8(n_pix, C);
  if (bf16)
   hipLaunchKernelGGL(( prelu_fwd_kernel<uint16_t>), dim3(grid), dim3(256), 0, stream, 
        (const uint16_t*)x, a, (uint16_t*)y, n_pix, C);
  else
   hipLaunchKernelGGL(( prelu_fwd_kernel<float>), dim3(grid), dim3(256), 0, stream, 
        (const float*)x, a, (float*)y, n_pix, C);
}

extern "C" void bdbnn_prelu_bwd(const void* x, const void* g, const float* a,
                                void* dx, float* da, int64_t n, int C,
                                bool bf16, hipStream_t stream) {
  int64_t n_pix = n / C;
  int grid = grid_pix8(n_pix, C);
  size_t lds = sizeof(float) * C;
  hipMemsetAsync(da, 0, sizeof(float) * C, stream);
  if (bf16)
   hipLaunchKernelGGL(( prelu_bwd_kernel<uint16_t>), dim3(grid), dim3(256), lds, stream, 
        (const uint16_t*)x, (const uint16_t*)g, a, (uint16_t*)dx, da, n_pix,
        C);
  else
   hipLaunchKernelGGL(( prelu_bwd_kernel<float>), dim3(grid), dim3(256), lds, stream, 
        (const float*)x, (const float*)g, a, (float*)dx, da, n_pix, C);
}
