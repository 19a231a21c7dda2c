#include "hip/hip_runtime.h"
// K4/K6 — fused classification losses over (B, C) logits.
//
// logit KD (ref:utils/KD_loss.py:10-43):
//   loss = mean_n( -sum_c softmax(t)_c * log_softmax(s)_c )
// cross-entropy (ref:train.py:318):
//   loss = mean_n( logsumexp(s_n) - s_n[y_n] )
//
// One block per row forward (max + logsumexp + dot in registers/LDS,
// logits read once); elementwise backward from saved row stats:
//   KD: ds = g/B * (softmax(s) - softmax(t))
//   CE: ds = g/B * (softmax(s) - onehot(y))
#include "common.h"

__device__ __forceinline__ float block_reduce(float v, float* red, int op) {
  // op 0 = max, 1 = sum
  red[threadIdx.x] = v;
  __syncthreads();
  for (int o = blockDim.x / 2; o > 0; o >>= 1) {
    if (threadIdx.x < o)
      red[threadIdx.x] = op == 0
          ? fmaxf(red[threadIdx.x], red[threadIdx.x + o])
          : red[threadIdx.x] + red[threadIdx.x + o];
    __syncthreads();
  }
  float r = red[0];
  __syncthreads();
  return r;
}

template <typename T>
__device__ __forceinline__ float ld(const T* p, int64_t i) {
  if constexpr (sizeof(T) == 2) return bf16_to_f32(((const uint16_t*)p)[i]);
  else                          return ((const float*)p)[i];
}

// stats[b] = (m_s, lseZ_s, m_t, lseZ_t); out += loss_row / B
template <typename T>
__global__ void kd_logit_fwd_kernel(const T* __restrict__ s,
                                    const T* __restrict__ t,
                                    float* __restrict__ stats,
                                    float* __restrict__ out, int B, int C) {
  __shared__ float red[256];
  int b = blockIdx.x;
  const T* sr = s + (int64_t)b * C;
  const T* tr = t + (int64_t)b * C;
  float ms = -3.4e38f, mt = -3.4e38f;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    ms = fmaxf(ms, ld(sr, c));
    mt = fmaxf(mt, ld(tr, c));
  }
  ms = block_reduce(ms, red, 0);
  mt = block_reduce(mt, red, 0);
  float zs = 0.f, zt = 0.f;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    zs += expf(ld(sr, c) - ms);
    zt += expf(ld(tr, c) - mt);
  }
  zs = block_reduce(zs, red, 1);
  zt = block_reduce(zt, red, 1);
  float lzs = logf(zs), lzt = logf(zt);
  float dot = 0.f;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    float pt = expf(ld(tr, c) - mt - lzt);
    dot += pt * (ld(sr, c) - ms - lzs);
  }
  dot = block_reduce(dot, red, 1);
  if (threadIdx.x == 0) {
    stats[b * 4 + 0] = ms + lzs;
    stats[b * 4 + 1] = mt + lzt;
    atomicAdd(out, -dot / (float)B);
  }
}

template <typename T>
__global__ void kd_logit_bwd_kernel(const T* __restrict__ s,
                                    const T* __restrict__ t,
                                    const float* __restrict__ stats,
                                    T* __restrict__ ds, float gscale,
                                    int64_t n, int C) {
  GRID_STRIDE(i, n) {
    int b = int(i / C);
    float lse_s = stats[b * 4 + 0], lse_t = stats[b * 4 + 1];
    float ps = expf(ld(s, i) - lse_s);
    float pt = expf(ld(t, i) - lse_t);
    float v = gscale * (ps - pt);
    if constexpr (sizeof(T) == 2) ((uint16_t*)ds)[i] = f32_to_bf16(v);
    else                          ((float*)ds)[i] = v;
  }
}

template <typename T>
__global__ void ce_fwd_kernel(const T* __restrict__ s,
                              const int64_t* __restrict__ y,
                              float* __restrict__ stats,
                              float* __restrict__ out, int B, int C) {
  __shared__ float red[256];
  int b = blockIdx.x;
  const T* sr = s + (int64_t)b * C;
  float ms = -3.4e38f;
  for (int c = threadIdx.x; c < C; c += blockDim.x)
    ms = fmaxf(ms, ld(sr, c));
  ms = block_reduce(ms, red, 0);
  float zs = 0.f;
  for (int c = threadIdx.x; c < C; c += blockDim.x)
    zs += expf(ld(sr, c) - ms);
  zs = block_reduce(zs, red, 1);
  if (threadIdx.x == 0) {
    float lse = ms + logf(zs);
    stats[b] = lse;
    atomicAdd(out, (lse - ld(sr, (int)y[b])) / (float)B);
  }
}

template <typename T>
__global__ void ce_bwd_kernel(const T* __restrict__ s,
                              const int64_t* __restrict__ y,
                              const float* __restrict__ stats,
                              T* __restrict__ ds, float gscale, int64_t n,
                              int C) {
  GRID_STRIDE(i, n) {
    int b = int(i / C);
    int c = int(i % C);
    float p = expf(ld(s, i) - stats[b]);
    float v = gscale * (p - (c == (int)y[b] ? 1.f : 0.f));
    if constexpr (sizeof(T) == 2) ((uint16_t*)ds)[i] = f32_to_bf16(v);
    else                          ((float*)ds)[i] = v;
  }
}

extern "C" void bdbnn_kd_logit_fwd(const void* s, const void* t,
                                   float* stats, float* out, int B, int C,
                                   bool bf16, hipStream_t stream) {
  hipMemsetAsync(out, 0, sizeof(float), stream);
  if (bf16)
   hipLaunchKernelGGL(( kd_logit_fwd_kernel<uint16_t>), dim3(B), dim3(256), 0, stream, 
        (const uint16_t*)s, (const uint16_t*)t, stats, out, B, C);
  else
   hipLaunchKernelGGL(( kd_logit_fwd_kernel<float>), dim3(B), dim3(256), 0, stream, 
        (const float*)s, (const float*)t, stats, out, B, C);
}

extern "C" void bdbnn_kd_logit_bwd(const void* s, const void* t,
                                   const float* stats, void* ds,
                                   float gscale, int B, int C, bool bf16,
                                   hipStream_t stream) {
  int64_t n = (int64_t)B * C;
  int grid = (int)bd_min<int64_t>((n + 255) / 256, 2048);
  if (bf16)
   hipLaunchKernelGGL(( kd_logit_bwd_kernel<uint16_t>), dim3(grid), dim3(256), 0, stream, 
        (const uint16_t*)s, (const uint16_t*)t, stats, (uint16_t*)ds,
        gscale, n, C);
  else
   hipLaunchKernelGGL(( kd_logit_bwd_kernel<float>), dim3(grid), dim3(256), 0, stream, 
        (const float*)s, (const float*)t, stats, (float*)ds, gscale, n, C);
}

extern "C" void bdbnn_ce_fwd(const void* s, const int64_t* y, float* stats,
                             float* out, int B, int C, bool bf16,
                             hipStream_t stream) {
  hipMemsetAsync(out, 0, sizeof(float), stream);
  if (bf16)
   hipLaunchKernelGGL(( ce_fwd_kernel<uint16_t>), dim3(B), dim3(256), 0, stream, 
        (const uint16_t*)s, y, stats, out, B, C);
  else
   hipLaunchKernelGGL(( ce_fwd_kernel<float>), dim3(B), dim3(256), 0, stream, 
        (const float*)s, y, stats, out, B, C);
}

extern "C" void bdbnn_ce_bwd(const void* s, const int64_t* y,
                             const float* stats, void* ds, float gscale,
                             int B, int C, bool bf16, hipStream_t stream) {
  int64_t n = (int64_t)B * C;
  int grid = (int)bd_min<int64_t>((n + 255) / 256, 2048);
  if (bf16)
   hipLaunchKernelGGL(( ce_bwd_kernel<uint16_t>), dim3(grid), dim3(256), 0, stream, 
        (const uint16_t*)s, y, stats, (uint16_t*)ds, gscale, n, C);
  else
   hipLaunchKernelGGL(( ce_bwd_kernel<float>), dim3(grid), dim3(256), 0, stream, 
        (const float*)s, y, stats, (float*)ds, gscale, n, C);
}
