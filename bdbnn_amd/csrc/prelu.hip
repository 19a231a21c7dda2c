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

// fixed-channel ownership: register da accumulation, one LDS add + one
// global atomic per owned channel (C a power of two <= 1024).
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
  const int64_t n_pix = n / C;
  int c, r_off, rows, per;
  if (C <= 256) { rows = 256 / C; c = threadIdx.x & (C - 1);
                  r_off = threadIdx.x / C; per = 1; }
  else { rows = 1; c = threadIdx.x; r_off = 0; per = C / 256; }
  for (int j = 0; j < per; ++j) {
    int cc = c + j * 256;
    float av = a[cc];
    float acc = 0.f;
    for (int64_t p = (int64_t)blockIdx.x * rows + r_off; p < n_pix;
         p += (int64_t)gridDim.x * rows) {
      int64_t i = p * C + cc;
      float xv, gv;
      if constexpr (sizeof(T) == 2) {
        xv = bf16_to_f32(((const uint16_t*)x)[i]);
        gv = bf16_to_f32(((const uint16_t*)g)[i]);
      } else {
        xv = ((const float*)x)[i];
        gv = ((const float*)g)[i];
      }
      float dxi = xv > 0.f ? gv : av * gv;
      if constexpr (sizeof(T) == 2) ((uint16_t*)dx)[i] = f32_to_bf16(dxi);
      else                          ((float*)dx)[i] = dxi;
      if (xv <= 0.f) acc += xv * gv;
    }
    atomicAdd(&da_lds[cc], acc);
  }
  __syncthreads();
  for (int c2 = threadIdx.x; c2 < C; c2 += blockDim.x)
    if (da_lds[c2] != 0.f) atomicAdd(&da[c2], da_lds[c2]);
}

extern "C" void bdbnn_prelu_fwd(const void* x, const float* a, void* y,
                                int64_t n, int C, bool bf16,
                                hipStream_t stream) {
  int block = 256;
  int grid = (int)bd_min<int64_t>((n + block - 1) / block, 2048);
  if (bf16)
    prelu_fwd_kernel<uint16_t><<<grid, block, 0, stream>>>(
        (const uint16_t*)x, a, (uint16_t*)y, n, C);
  else
    prelu_fwd_kernel<float><<<grid, block, 0, stream>>>(
        (const float*)x, a, (float*)y, n, C);
}

extern "C" void bdbnn_prelu_bwd(const void* x, const void* g, const float* a,
                                void* dx, float* da, int64_t n, int C,
                                bool bf16, hipStream_t stream) {
  int block = 256;
  int64_t n_pix = n / C;
  int rows = C <= 256 ? 256 / C : 1;
  int grid = (int)bd_min<int64_t>((n_pix + rows - 1) / rows, 2048);
  size_t lds = sizeof(float) * C;
  hipMemsetAsync(da, 0, sizeof(float) * C, stream);
  if (bf16)
    prelu_bwd_kernel<uint16_t><<<grid, block, lds, stream>>>(
        (const uint16_t*)x, (const uint16_t*)g, a, (uint16_t*)dx, da, n, C);
  else
    prelu_bwd_kernel<float><<<grid, block, lds, stream>>>(
        (const float*)x, (const float*)g, a, (float*)dx, da, n, C);
}
