// Shared helpers for the bdbnn_amd gfx950 kernels.
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

#define BDBNN_WAVE 64

template <typename T>
__host__ __device__ __forceinline__ T bd_min(T a, T b) {
  return a < b ? a : b;
}

__device__ __forceinline__ float bf16_to_f32(uint16_t u) {
  union { uint32_t u; float f; } v;
  v.u = uint32_t(u) << 16;
  return v.f;
}

__device__ __forceinline__ uint16_t f32_to_bf16(float f) {
  union { float f; uint32_t u; } v;
  v.f = f;
  // round-to-nearest-even
  uint32_t rounding = 0x7fff + ((v.u >> 16) & 1);
  return uint16_t((v.u + rounding) >> 16);
}

// Grid-stride helper
#define GRID_STRIDE(i, n)                                            \
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;   \
       i < (n); i += (int64_t)gridDim.x * blockDim.x)

// Multi-tensor descriptor: fits in kernel args (<4 KB).
constexpr int BDBNN_MAX_TENSORS = 64;

struct TensorListArg {
  const float* ptr[BDBNN_MAX_TENSORS];
  int64_t numel[BDBNN_MAX_TENSORS];
  int n;
};
