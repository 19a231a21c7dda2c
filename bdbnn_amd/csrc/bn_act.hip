// K8 — fused BatchNorm (+ residual add) (+ PReLU/ReLU) for NHWC, training
// and eval.  Replaces the MIOpen BN pipeline + separate add + activation
// (6+ kernel launches and ~8 tensor passes per block position) with:
//   fwd: stats pass (one read) + normalize/add/act pass (reads x,skip;
//        writes out and the pre-activation z needed by backward)
//   bwd: reduce pass (reads dy,z,x -> per-channel sums + da) +
//        apply pass (reads dy,z,x -> writes dx and dskip)
//
// BN semantics match nn.BatchNorm2d: biased batch var for normalization,
// unbiased var into running_var, momentum update in the finalize step.
// act_kind: 0 = identity, 1 = per-channel PReLU, 2 = ReLU.
#include "common.h"

// ---- pass 1: per-channel sum / sumsq ----
__global__ void bn_stats_kernel(const void* __restrict__ xv, float* __restrict__ s1,
                                float* __restrict__ s2, int64_t n, int C,
                                bool bf16) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* l1 = (float*)smem_raw;
  float* l2 = l1 + C;
  for (int c = threadIdx.x; c < C; c += blockDim.x) { l1[c] = 0.f; l2[c] = 0.f; }
  __syncthreads();
  GRID_STRIDE(i, n) {
    int c = int(i % C);
    float v = bf16 ? bf16_to_f32(((const uint16_t*)xv)[i])
                   : ((const float*)xv)[i];
    atomicAdd(&l1[c], v);
    atomicAdd(&l2[c], v * v);
  }
  __syncthreads();
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    if (l1[c] != 0.f) atomicAdd(&s1[c], l1[c]);
    if (l2[c] != 0.f) atomicAdd(&s2[c], l2[c]);
  }
}

// ---- finalize: mean/invstd + running-stat update ----
__global__ void bn_finalize_kernel(const float* __restrict__ s1,
                                   const float* __restrict__ s2,
                                   float* __restrict__ mean,
                                   float* __restrict__ invstd,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   int C, float n, float momentum, float eps) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float m = s1[c] / n;
  float var = fmaxf(s2[c] / n - m * m, 0.f);
  mean[c] = m;
  invstd[c] = rsqrtf(var + eps);
  if (running_mean != nullptr) {
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * m;
    float var_unb = var * n / fmaxf(n - 1.f, 1.f);
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * var_unb;
  }
}

// ---- pass 2: normalize + add + act ----
template <typename T>
__global__ void bn_act_fwd_kernel(const T* __restrict__ x,
                                  const T* __restrict__ skip,
                                  const float* __restrict__ mean,
                                  const float* __restrict__ invstd,
                                  const float* __restrict__ gamma,
                                  const float* __restrict__ beta,
                                  const float* __restrict__ a,
                                  T* __restrict__ out, T* __restrict__ zout,
                                  int64_t n, int C, int act_kind) {
  GRID_STRIDE(i, n) {
    int c = int(i % C);
    float v;
    if constexpr (sizeof(T) == 2) v = bf16_to_f32(((const uint16_t*)x)[i]);
    else                          v = ((const float*)x)[i];
    float z = gamma[c] * (v - mean[c]) * invstd[c] + beta[c];
    if (skip != nullptr) {
      float s;
      if constexpr (sizeof(T) == 2) s = bf16_to_f32(((const uint16_t*)skip)[i]);
      else                          s = ((const float*)skip)[i];
      z += s;
    }
    float o = z;
    if (act_kind == 1) o = z > 0.f ? z : a[c] * z;
    else if (act_kind == 2) o = fmaxf(z, 0.f);
    if constexpr (sizeof(T) == 2) {
      ((uint16_t*)out)[i] = f32_to_bf16(o);
      if (zout != nullptr) ((uint16_t*)zout)[i] = f32_to_bf16(z);
    } else {
      ((float*)out)[i] = o;
      if (zout != nullptr) ((float*)zout)[i] = z;
    }
  }
}

// ---- backward pass 1: per-channel reductions ----
// sums layout: [C][3] = (sum dz, sum dz*xhat, da)
template <typename T>
__global__ void bn_act_bwd_reduce_kernel(
    const T* __restrict__ dy, const T* __restrict__ z,
    const T* __restrict__ x, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ a,
    float* __restrict__ sums, int64_t n, int C, int act_kind) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* r0 = (float*)smem_raw;      // sum dz
  float* r1 = r0 + C;                // sum dz * xhat
  float* r2 = r1 + C;                // da
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    r0[c] = 0.f; r1[c] = 0.f; r2[c] = 0.f;
  }
  __syncthreads();
  GRID_STRIDE(i, n) {
    int c = int(i % C);
    float dyi, zi, xi;
    if constexpr (sizeof(T) == 2) {
      dyi = bf16_to_f32(((const uint16_t*)dy)[i]);
      zi = bf16_to_f32(((const uint16_t*)z)[i]);
      xi = bf16_to_f32(((const uint16_t*)x)[i]);
    } else {
      dyi = ((const float*)dy)[i]; zi = ((const float*)z)[i];
      xi = ((const float*)x)[i];
    }
    float dz = dyi;
    if (act_kind == 1) {
      dz = zi > 0.f ? dyi : a[c] * dyi;
      if (zi <= 0.f) atomicAdd(&r2[c], dyi * zi);
    } else if (act_kind == 2) {
      dz = zi > 0.f ? dyi : 0.f;
    }
    float xhat = (xi - mean[c]) * invstd[c];
    atomicAdd(&r0[c], dz);
    atomicAdd(&r1[c], dz * xhat);
  }
  __syncthreads();
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    if (r0[c] != 0.f) atomicAdd(&sums[c * 3 + 0], r0[c]);
    if (r1[c] != 0.f) atomicAdd(&sums[c * 3 + 1], r1[c]);
    if (r2[c] != 0.f) atomicAdd(&sums[c * 3 + 2], r2[c]);
  }
}

// ---- backward pass 2: dx (+ dskip) ----
template <typename T>
__global__ void bn_act_bwd_apply_kernel(
    const T* __restrict__ dy, const T* __restrict__ z,
    const T* __restrict__ x, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ gamma,
    const float* __restrict__ a, const float* __restrict__ sums,
    T* __restrict__ dx, T* __restrict__ dskip, int64_t n, int C,
    int act_kind, float inv_n) {
  GRID_STRIDE(i, n) {
    int c = int(i % C);
    float dyi, zi, xi;
    if constexpr (sizeof(T) == 2) {
      dyi = bf16_to_f32(((const uint16_t*)dy)[i]);
      zi = bf16_to_f32(((const uint16_t*)z)[i]);
      xi = bf16_to_f32(((const uint16_t*)x)[i]);
    } else {
      dyi = ((const float*)dy)[i]; zi = ((const float*)z)[i];
      xi = ((const float*)x)[i];
    }
    float dz = dyi;
    if (act_kind == 1) dz = zi > 0.f ? dyi : a[c] * dyi;
    else if (act_kind == 2) dz = zi > 0.f ? dyi : 0.f;
    float xhat = (xi - mean[c]) * invstd[c];
    float sdz = sums[c * 3 + 0], sdzx = sums[c * 3 + 1];
    float dxi = gamma[c] * invstd[c] *
                (dz - sdz * inv_n - xhat * sdzx * inv_n);
    if constexpr (sizeof(T) == 2) {
      ((uint16_t*)dx)[i] = f32_to_bf16(dxi);
      if (dskip != nullptr) ((uint16_t*)dskip)[i] = f32_to_bf16(dz);
    } else {
      ((float*)dx)[i] = dxi;
      if (dskip != nullptr) ((float*)dskip)[i] = dz;
    }
  }
}

// ---- eval-mode fused normalize(+add)(+act) using running stats ----
template <typename T>
__global__ void bn_act_eval_kernel(const T* __restrict__ x,
                                   const T* __restrict__ skip,
                                   const float* __restrict__ rm,
                                   const float* __restrict__ rv,
                                   const float* __restrict__ gamma,
                                   const float* __restrict__ beta,
                                   const float* __restrict__ a,
                                   T* __restrict__ out, int64_t n, int C,
                                   int act_kind, float eps) {
  GRID_STRIDE(i, n) {
    int c = int(i % C);
    float v;
    if constexpr (sizeof(T) == 2) v = bf16_to_f32(((const uint16_t*)x)[i]);
    else                          v = ((const float*)x)[i];
    float z = gamma[c] * (v - rm[c]) * rsqrtf(rv[c] + eps) + beta[c];
    if (skip != nullptr) {
      float s;
      if constexpr (sizeof(T) == 2) s = bf16_to_f32(((const uint16_t*)skip)[i]);
      else                          s = ((const float*)skip)[i];
      z += s;
    }
    float o = z;
    if (act_kind == 1) o = z > 0.f ? z : a[c] * z;
    else if (act_kind == 2) o = fmaxf(z, 0.f);
    if constexpr (sizeof(T) == 2) ((uint16_t*)out)[i] = f32_to_bf16(o);
    else                          ((float*)out)[i] = o;
  }
}

static inline int grid_for(int64_t n) {
  return (int)bd_min<int64_t>((n + 255) / 256, 2048);
}

extern "C" void bdbnn_bn_stats(const void* x, float* s1, float* s2,
                               int64_t n, int C, bool bf16,
                               hipStream_t stream) {
  hipMemsetAsync(s1, 0, sizeof(float) * C, stream);
  hipMemsetAsync(s2, 0, sizeof(float) * C, stream);
  bn_stats_kernel<<<grid_for(n), 256, 2 * sizeof(float) * C, stream>>>(
      x, s1, s2, n, C, bf16);
}

extern "C" void bdbnn_bn_finalize(const float* s1, const float* s2,
                                  float* mean, float* invstd,
                                  float* running_mean, float* running_var,
                                  int C, float n, float momentum, float eps,
                                  hipStream_t stream) {
  bn_finalize_kernel<<<(C + 255) / 256, 256, 0, stream>>>(
      s1, s2, mean, invstd, running_mean, running_var, C, n, momentum, eps);
}

#define BN_LAUNCH(kernel, ...)                                              \
  if (bf16) kernel<uint16_t><<<grid_for(n), 256, lds, stream>>>(__VA_ARGS__); \
  else      kernel<float><<<grid_for(n), 256, lds, stream>>>(__VA_ARGS__);

extern "C" void bdbnn_bn_act_fwd(const void* x, const void* skip,
                                 const float* mean, const float* invstd,
                                 const float* gamma, const float* beta,
                                 const float* a, void* out, void* zout,
                                 int64_t n, int C, int act_kind, bool bf16,
                                 hipStream_t stream) {
  size_t lds = 0;
  if (bf16)
    bn_act_fwd_kernel<uint16_t><<<grid_for(n), 256, lds, stream>>>(
        (const uint16_t*)x, (const uint16_t*)skip, mean, invstd, gamma, beta,
        a, (uint16_t*)out, (uint16_t*)zout, n, C, act_kind);
  else
    bn_act_fwd_kernel<float><<<grid_for(n), 256, lds, stream>>>(
        (const float*)x, (const float*)skip, mean, invstd, gamma, beta, a,
        (float*)out, (float*)zout, n, C, act_kind);
}

extern "C" void bdbnn_bn_act_bwd_reduce(const void* dy, const void* z,
                                        const void* x, const float* mean,
                                        const float* invstd, const float* a,
                                        float* sums, int64_t n, int C,
                                        int act_kind, bool bf16,
                                        hipStream_t stream) {
  hipMemsetAsync(sums, 0, sizeof(float) * C * 3, stream);
  size_t lds = 3 * sizeof(float) * C;
  if (bf16)
    bn_act_bwd_reduce_kernel<uint16_t><<<grid_for(n), 256, lds, stream>>>(
        (const uint16_t*)dy, (const uint16_t*)z, (const uint16_t*)x, mean,
        invstd, a, sums, n, C, act_kind);
  else
    bn_act_bwd_reduce_kernel<float><<<grid_for(n), 256, lds, stream>>>(
        (const float*)dy, (const float*)z, (const float*)x, mean, invstd, a,
        sums, n, C, act_kind);
}

extern "C" void bdbnn_bn_act_bwd_apply(const void* dy, const void* z,
                                       const void* x, const float* mean,
                                       const float* invstd,
                                       const float* gamma, const float* a,
                                       const float* sums, void* dx,
                                       void* dskip, int64_t n, int C,
                                       int act_kind, float inv_n, bool bf16,
                                       hipStream_t stream) {
  if (bf16)
    bn_act_bwd_apply_kernel<uint16_t><<<grid_for(n), 256, 0, stream>>>(
        (const uint16_t*)dy, (const uint16_t*)z, (const uint16_t*)x, mean,
        invstd, gamma, a, sums, (uint16_t*)dx, (uint16_t*)dskip, n, C,
        act_kind, inv_n);
  else
    bn_act_bwd_apply_kernel<float><<<grid_for(n), 256, 0, stream>>>(
        (const float*)dy, (const float*)z, (const float*)x, mean, invstd,
        gamma, a, sums, (float*)dx, (float*)dskip, n, C, act_kind, inv_n);
}

extern "C" void bdbnn_bn_act_eval(const void* x, const void* skip,
                                  const float* rm, const float* rv,
                                  const float* gamma, const float* beta,
                                  const float* a, void* out, int64_t n,
                                  int C, int act_kind, float eps, bool bf16,
                                  hipStream_t stream) {
  if (bf16)
    bn_act_eval_kernel<uint16_t><<<grid_for(n), 256, 0, stream>>>(
        (const uint16_t*)x, (const uint16_t*)skip, rm, rv, gamma, beta, a,
        (uint16_t*)out, n, C, act_kind, eps);
  else
    bn_act_eval_kernel<float><<<grid_for(n), 256, 0, stream>>>(
        (const float*)x, (const float*)skip, rm, rv, gamma, beta, a,
        (float*)out, n, C, act_kind, eps);
}
