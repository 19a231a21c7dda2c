// 8-element vector load/store helpers (bf16: one 16-B uint4; fp32: two
// float4).  Guide G13: scalar bf16 loads cost ~2-2.5x on gfx950.
#pragma once
#include "common.h"

template <typename T>
__device__ __forceinline__ void load8(const T* p, int64_t i, float v[8]) {
  if constexpr (sizeof(T) == 2) {
    uint4 r = *(const uint4*)(p + i);
    const uint16_t* u = (const uint16_t*)&r;
#pragma unroll
    for (int j = 0; j < 8; ++j) v[j] = bf16_to_f32(u[j]);
  } else {
    float4 a = *(const float4*)((const float*)p + i);
    float4 b = *(const float4*)((const float*)p + i + 4);
    v[0] = a.x; v[1] = a.y; v[2] = a.z; v[3] = a.w;
    v[4] = b.x; v[5] = b.y; v[6] = b.z; v[7] = b.w;
  }
}

template <typename T>
__device__ __forceinline__ void store8(T* p, int64_t i, const float v[8]) {
  if constexpr (sizeof(T) == 2) {
    uint4 r;
    uint16_t* u = (uint16_t*)&r;
#pragma unroll
    for (int j = 0; j < 8; ++j) u[j] = f32_to_bf16(v[j]);
    *(uint4*)(p + i) = r;
  } else {
    float4 a{v[0], v[1], v[2], v[3]}, b{v[4], v[5], v[6], v[7]};
    *(float4*)((float*)p + i) = a;
    *(float4*)((float*)p + i + 4) = b;
  }
}

// channel-group geometry for NHWC kernels: thread owns VEC=8 consecutive
// channels; C is a power of two >= 8.
struct ChanMap {
  int c0;        // first owned channel
  int64_t p0;    // first pixel
  int64_t pstep; // pixel stride
};

__device__ __forceinline__ ChanMap chan_map8(int C) {
  int tpr = C / 8;                       // threads per pixel row
  ChanMap m;
  m.c0 = (threadIdx.x % tpr) * 8;
  int rows = 256 / tpr;                  // rows covered per block iter
  m.p0 = (int64_t)blockIdx.x * rows + threadIdx.x / tpr;
  m.pstep = (int64_t)gridDim.x * rows;
  return m;
}

__host__ __device__ __forceinline__ int grid_pix8(int64_t n_pix, int C) {
  int rows = 256 / (C / 8 < 256 ? C / 8 : 256);
  if (C / 8 >= 256) rows = 1;
  int64_t g = (n_pix + rows - 1) / rows;
  return (int)(g < 2048 ? g : 2048);
}
