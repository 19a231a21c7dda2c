#include "hip/hip_runtime.h"
// K8a — fused per-channel PReLU forward/backward for NHWC (channels_last).
//
// torch's prelu_backward on channels_last bf16 was 40% of the whole
// training step on MI355X (profiles/r01_bench_b256_kernel_stats.md);
// this replaces it with one memory-bound pass each way.
//
//   fwd: y = x > 0 ? x : a[c] * x
//   bwd: dx = x > 0 ? g : a[c] * g
//        da[c] = sum over pixels of (x > 0 ? 0 : x * g)
// backward is ONE pass: reads x,g once, writes dx, accumulates da in an
// LDS per-channel array (C <= 1024), one global atomicAdd per channel per
// block at the end.
#include "common.h"

template <typename T>
__global__ void prelu_fwd_kernel(const T* __restrict__ x,
                                 const float* __restrict__ a,
                                 T* __restrict__ y, int64_t n, int C) {
  GRID_STRIDE(i, n) {
    int c = int(i % C);
    float v;
    if constexpr (sizeof(T) == 2) v = bf16_to_f32(((const uint16_t*)x)[i]);
    else                          v = ((const float*)x)[i];
    float o = v > 0.f ? v : a[c] * v;
    if constexpr (sizeof(T) == 2) ((uint16_t*)y)[i] = f32_to_bf16(o);
    else                          ((float*)y)[i] = o;
  }
}

template <typename T>
__global__ void prelu_bwd_kernel(const T* __restrict__ x,
                                 const T* __restrict__ g,
                                 const float* __restrict__ a,
                                 T* __restrict__ dx,
                                 float* __restrict__ da, int64_t n, int C) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* da_lds = (float*)smem_raw;
  for (int c = threadIdx.x; c < C; c += blockDim.x) da_lds[c] = 0.f;
  __syncthreads();
  GRID_STRIDE(i, n) {
    int c = int(i % C);
    float xv, gv;
    if constexpr (sizeof(T) == 2) {
      xv = bf16_to_f32(((const uint16_t*)x)[i]);
      gv = bf16_to_f32(((const uint16_t*)g)[i]);
    } else {
      xv = ((const float*)x)[i];
      gv = ((const float*)g)[i];
    }
    float dxi = xv > 0.f ? gv : a[c] * gv;
    if constexpr (sizeof(T) == 2) ((uint16_t*)dx)[i] = f32_to_bf16(dxi);
    else                          ((float*)dx)[i] = dxi;
    if (xv <= 0.f) atomicAdd(&da_lds[c], xv * gv);
  }
  __syncthreads();
  for (int c = threadIdx.x; c < C; c += blockDim.x)
    if (da_lds[c] != 0.f) atomicAdd(&da[c], da_lds[c]);
}

extern "C" void bdbnn_prelu_fwd(const void* x, const float* a, void* y,
                                int64_t n, int C, bool bf16,
                                hipStream_t stream) {
  int block = 256;
  int grid = (int)bd_min<int64_t>((n + block - 1) / block, 2048);
  if (bf16)
   hipLaunchKernelGGL(( prelu_fwd_kernel<uint16_t>), dim3(grid), dim3(block), 0, stream, 
        (const uint16_t*)x, a, (uint16_t*)y, n, C);
  else
   hipLaunchKernelGGL(( prelu_fwd_kernel<float>), dim3(grid), dim3(block), 0, stream, 
        (const float*)x, a, (float*)y, n, C);
}

extern "C" void bdbnn_prelu_bwd(const void* x, const void* g, const float* a,
                                void* dx, float* da, int64_t n, int C,
                                bool bf16, hipStream_t stream) {
  int block = 256;
  int grid = (int)bd_min<int64_t>((n + block - 1) / block, 2048);
  size_t lds = sizeof(float) * C;
  hipMemsetAsync(da, 0, sizeof(float) * C, stream);
  if (bf16)
   hipLaunchKernelGGL(( prelu_bwd_kernel<uint16_t>), dim3(grid), dim3(block), lds, stream, 
        (const uint16_t*)x, (const uint16_t*)g, a, (uint16_t*)dx, da, n, C);
  else
   hipLaunchKernelGGL(( prelu_bwd_kernel<float>), dim3(grid), dim3(block), lds, stream, 
        (const float*)x, (const float*)g, a, (float*)dx, da, n, C);
}
