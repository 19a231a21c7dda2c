// K8b — fused NHWC max-pool (3x3 stride-2 pad-1 stem pool) with u8
// argmax indices; gather-based backward (no atomics).
// torch's channels_last maxpool fwd+bwd was ~1.9 ms/step at batch 512
// (profiles/r01_b512_steady_state.md).
#include "common.h"
#include "vec8.h"

struct PoolParams {
  int N, C, H, W, Ho, Wo, ks, stride, pad;
};

// thread owns 8 consecutive channels of one output pixel
template <typename T>
__global__ void maxpool_fwd_kernel(const T* __restrict__ x,
                                   T* __restrict__ out,
                                   unsigned char* __restrict__ idx,
                                   PoolParams p, int64_t n_opix) {
  ChanMap m = chan_map8(p.C);
  for (int64_t op = m.p0; op < n_opix; op += m.pstep) {
    int n = int(op / ((int64_t)p.Ho * p.Wo));
    int rem = int(op % ((int64_t)p.Ho * p.Wo));
    int oy = rem / p.Wo, ox = rem % p.Wo;
    int iy0 = oy * p.stride - p.pad;
    int ix0 = ox * p.stride - p.pad;
    float best[8];
    int bi[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) { best[j] = -3.4e38f; bi[j] = 0; }
    for (int t = 0; t < p.ks * p.ks; ++t) {
      int iy = iy0 + t / p.ks, ix = ix0 + t % p.ks;
      if (iy < 0 || iy >= p.H || ix < 0 || ix >= p.W) continue;
      float v[8];
      load8(x, (((int64_t)n * p.H + iy) * p.W + ix) * p.C + m.c0, v);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        if (v[j] > best[j]) { best[j] = v[j]; bi[j] = t; }
    }
    store8(out, op * p.C + m.c0, best);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      idx[op * p.C + m.c0 + j] = (unsigned char)bi[j];
  }
}

// thread owns 8 consecutive channels of one INPUT pixel; gathers from
// the <= ceil(ks/stride)^2 output windows that could have selected it.
template <typename T>
__global__ void maxpool_bwd_kernel(const T* __restrict__ dy,
                                   const unsigned char* __restrict__ idx,
                                   T* __restrict__ dx, PoolParams p,
                                   int64_t n_ipix) {
  ChanMap m = chan_map8(p.C);
  for (int64_t ip = m.p0; ip < n_ipix; ip += m.pstep) {
    int n = int(ip / ((int64_t)p.H * p.W));
    int rem = int(ip % ((int64_t)p.H * p.W));
    int iy = rem / p.W, ix = rem % p.W;
    float acc[8] = {};
    // windows (oy, ox) with iy0 <= iy < iy0+ks
    int oy_lo = max(0, (iy + p.pad - p.ks + p.stride) / p.stride);
    int oy_hi = min(p.Ho - 1, (iy + p.pad) / p.stride);
    int ox_lo = max(0, (ix + p.pad - p.ks + p.stride) / p.stride);
    int ox_hi = min(p.Wo - 1, (ix + p.pad) / p.stride);
    for (int oy = oy_lo; oy <= oy_hi; ++oy)
      for (int ox = ox_lo; ox <= ox_hi; ++ox) {
        int t = (iy - (oy * p.stride - p.pad)) * p.ks +
                (ix - (ox * p.stride - p.pad));
        int64_t op = ((int64_t)n * p.Ho + oy) * p.Wo + ox;
        float g[8];
        load8(dy, op * p.C + m.c0, g);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (idx[op * p.C + m.c0 + j] == t) acc[j] += g[j];
      }
    store8(dx, ip * p.C + m.c0, acc);
  }
}

extern "C" void bdbnn_maxpool_fwd(const void* x, void* out,
                                  unsigned char* idx, int N, int C, int H,
                                  int W, int Ho, int Wo, int ks, int stride,
                                  int pad, bool bf16, hipStream_t stream) {
  PoolParams p{N, C, H, W, Ho, Wo, ks, stride, pad};
  int64_t n_opix = (int64_t)N * Ho * Wo;
  int grid = grid_pix8(n_opix, C);
  if (bf16)
    maxpool_fwd_kernel<uint16_t><<<grid, 256, 0, stream>>>(
        (const uint16_t*)x, (uint16_t*)out, idx, p, n_opix);
  else
    maxpool_fwd_kernel<float><<<grid, 256, 0, stream>>>(
        (const float*)x, (float*)out, idx, p, n_opix);
}

extern "C" void bdbnn_maxpool_bwd(const void* dy, const unsigned char* idx,
                                  void* dx, int N, int C, int H, int W,
                                  int Ho, int Wo, int ks, int stride,
                                  int pad, bool bf16, hipStream_t stream) {
  PoolParams p{N, C, H, W, Ho, Wo, ks, stride, pad};
  int64_t n_ipix = (int64_t)N * H * W;
  int grid = grid_pix8(n_ipix, C);
  if (bf16)
    maxpool_bwd_kernel<uint16_t><<<grid, 256, 0, stream>>>(
        (const uint16_t*)dy, idx, (uint16_t*)dx, p, n_ipix);
  else
    maxpool_bwd_kernel<float><<<grid, 256, 0, stream>>>(
        (const float*)dy, idx, (float*)dx, p, n_ipix);
}
