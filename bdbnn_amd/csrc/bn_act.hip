// K8 — fused BatchNorm (+ residual add) (+ PReLU/ReLU) for NHWC, training
// and eval.  Replaces the MIOpen BN pipeline + separate add + activation
// (6+ kernel launches and ~8 tensor passes per block position) with:
//   fwd: stats pass (one read) + normalize/add/act pass (reads x,skip;
//        writes out and the pre-activation z needed by backward)
//   bwd: reduce pass (reads dy,z,x -> per-channel sums + da) +
//        apply pass (reads dy,z,x -> writes dx and dskip)
//
// Channel-reduction layout: C is a power of two <= 1024 (python guards);
// every thread owns FIXED channels (c = tid mod C), accumulates in
// registers over its pixel stripe — coalesced along C, no per-element
// atomics — and lands ONE LDS add + ONE global atomic per owned channel.
//
// BN semantics match nn.BatchNorm2d: biased batch var for normalization,
// unbiased var into running_var, momentum update in the finalize step.
// act_kind: 0 = identity, 1 = per-channel PReLU, 2 = ReLU.
#include "common.h"

#define BN_MAXC_PER_THREAD 4  // supports C up to 4*256 = 1024

__device__ __forceinline__ float load_f(const void* p, int64_t i, bool bf16) {
  return bf16 ? bf16_to_f32(((const uint16_t*)p)[i]) : ((const float*)p)[i];
}

__device__ __forceinline__ void store_f(void* p, int64_t i, float v,
                                        bool bf16) {
  if (bf16) ((uint16_t*)p)[i] = f32_to_bf16(v);
  else      ((float*)p)[i] = v;
}

// ---- pass 1: per-channel sum / sumsq (fixed-channel ownership) ----
template <typename T>
__global__ void bn_stats_kernel(const T* __restrict__ x,
                                float* __restrict__ s1,
                                float* __restrict__ s2,
                                int64_t n_pix, int C) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* l1 = (float*)smem_raw;
  float* l2 = l1 + C;
  for (int c = threadIdx.x; c < C; c += blockDim.x) { l1[c] = 0.f; l2[c] = 0.f; }
  __syncthreads();
  const bool bf16 = sizeof(T) == 2;
  if (C <= 256) {
    int rows = 256 / C;
    int c = threadIdx.x & (C - 1);
    int r_off = threadIdx.x / C;
    float a1 = 0.f, a2 = 0.f;
    for (int64_t p = (int64_t)blockIdx.x * rows + r_off; p < n_pix;
         p += (int64_t)gridDim.x * rows) {
      float v = load_f(x, p * C + c, bf16);
      a1 += v; a2 += v * v;
    }
    atomicAdd(&l1[c], a1);
    atomicAdd(&l2[c], a2);
  } else {
    int per = C / 256;
    float a1[BN_MAXC_PER_THREAD] = {}, a2[BN_MAXC_PER_THREAD] = {};
    for (int64_t p = blockIdx.x; p < n_pix; p += gridDim.x)
      for (int j = 0; j < per; ++j) {
        float v = load_f(x, p * C + threadIdx.x + j * 256, bf16);
        a1[j] += v; a2[j] += v * v;
      }
    for (int j = 0; j < per; ++j) {
      l1[threadIdx.x + j * 256] = a1[j];
      l2[threadIdx.x + j * 256] = a2[j];
    }
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

// ---- pass 2: normalize + add + act (fixed-channel, scale/shift hoisted) ----
template <typename T, int ACT>
__global__ void bn_act_fwd_kernel(const T* __restrict__ x,
                                  const T* __restrict__ skip,
                                  const float* __restrict__ mean,
                                  const float* __restrict__ invstd,
                                  const float* __restrict__ gamma,
                                  const float* __restrict__ beta,
                                  const float* __restrict__ a,
                                  T* __restrict__ out, T* __restrict__ zout,
                                  int64_t n_pix, int C) {
  const bool bf16 = sizeof(T) == 2;
  int c, r_off, rows, per;
  if (C <= 256) { rows = 256 / C; c = threadIdx.x & (C - 1);
                  r_off = threadIdx.x / C; per = 1; }
  else { rows = 1; c = threadIdx.x; r_off = 0; per = C / 256; }
  for (int j = 0; j < per; ++j) {
    int cc = c + j * 256;
    float sc = gamma[cc] * invstd[cc];
    float sh = beta[cc] - mean[cc] * sc;
    float av = (ACT == 1) ? a[cc] : 0.f;
    for (int64_t p = (int64_t)blockIdx.x * rows + r_off; p < n_pix;
         p += (int64_t)gridDim.x * rows) {
      int64_t i = p * C + cc;
      float z = sc * load_f(x, i, bf16) + sh;
      if (skip != nullptr) z += load_f(skip, i, bf16);
      float o = z;
      if (ACT == 1) o = z > 0.f ? z : av * z;
      else if (ACT == 2) o = fmaxf(z, 0.f);
      store_f(out, i, o, bf16);
      if (zout != nullptr) store_f(zout, i, z, bf16);
    }
  }
}

// ---- backward pass 1: per-channel reductions ----
// sums layout: [C][3] = (sum dz, sum dz*xhat, da)
template <typename T, int ACT>
__global__ void bn_act_bwd_reduce_kernel(
    const T* __restrict__ dy, const T* __restrict__ z,
    const T* __restrict__ x, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ a,
    float* __restrict__ sums, int64_t n_pix, int C) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* r0 = (float*)smem_raw;
  float* r1 = r0 + C;
  float* r2 = r1 + C;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    r0[c] = 0.f; r1[c] = 0.f; r2[c] = 0.f;
  }
  __syncthreads();
  const bool bf16 = sizeof(T) == 2;
  int c, r_off, rows, per;
  if (C <= 256) { rows = 256 / C; c = threadIdx.x & (C - 1);
                  r_off = threadIdx.x / C; per = 1; }
  else { rows = 1; c = threadIdx.x; r_off = 0; per = C / 256; }
  for (int j = 0; j < per; ++j) {
    int cc = c + j * 256;
    float mu = mean[cc], is = invstd[cc];
    float av = (ACT == 1) ? a[cc] : 0.f;
    float s0 = 0.f, s1 = 0.f, s2 = 0.f;
    for (int64_t p = (int64_t)blockIdx.x * rows + r_off; p < n_pix;
         p += (int64_t)gridDim.x * rows) {
      int64_t i = p * C + cc;
      float dyi = load_f(dy, i, bf16);
      float dz = dyi;
      if (ACT == 1) {
        float zi = load_f(z, i, bf16);
        dz = zi > 0.f ? dyi : av * dyi;
        if (zi <= 0.f) s2 += dyi * zi;
      } else if (ACT == 2) {
        float zi = load_f(z, i, bf16);
        dz = zi > 0.f ? dyi : 0.f;
      }
      float xhat = (load_f(x, i, bf16) - mu) * is;
      s0 += dz;
      s1 += dz * xhat;
    }
    atomicAdd(&r0[cc], s0);
    atomicAdd(&r1[cc], s1);
    if (ACT == 1) atomicAdd(&r2[cc], s2);
  }
  __syncthreads();
  for (int cc = threadIdx.x; cc < C; cc += blockDim.x) {
    if (r0[cc] != 0.f) atomicAdd(&sums[cc * 3 + 0], r0[cc]);
    if (r1[cc] != 0.f) atomicAdd(&sums[cc * 3 + 1], r1[cc]);
    if (r2[cc] != 0.f) atomicAdd(&sums[cc * 3 + 2], r2[cc]);
  }
}

// ---- backward pass 2: dx (+ dskip) ----
template <typename T, int ACT>
__global__ void bn_act_bwd_apply_kernel(
    const T* __restrict__ dy, const T* __restrict__ z,
    const T* __restrict__ x, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ gamma,
    const float* __restrict__ a, const float* __restrict__ sums,
    T* __restrict__ dx, T* __restrict__ dskip, int64_t n_pix, int C,
    float inv_n) {
  const bool bf16 = sizeof(T) == 2;
  int c, r_off, rows, per;
  if (C <= 256) { rows = 256 / C; c = threadIdx.x & (C - 1);
                  r_off = threadIdx.x / C; per = 1; }
  else { rows = 1; c = threadIdx.x; r_off = 0; per = C / 256; }
  for (int j = 0; j < per; ++j) {
    int cc = c + j * 256;
    float mu = mean[cc], is = invstd[cc];
    float gis = gamma[cc] * is;
    float av = (ACT == 1) ? a[cc] : 0.f;
    float sdz_n = sums[cc * 3 + 0] * inv_n;
    float sdzx_n = sums[cc * 3 + 1] * inv_n;
    for (int64_t p = (int64_t)blockIdx.x * rows + r_off; p < n_pix;
         p += (int64_t)gridDim.x * rows) {
      int64_t i = p * C + cc;
      float dyi = load_f(dy, i, bf16);
      float dz = dyi;
      if (ACT == 1) {
        float zi = load_f(z, i, bf16);
        dz = zi > 0.f ? dyi : av * dyi;
      } else if (ACT == 2) {
        float zi = load_f(z, i, bf16);
        dz = zi > 0.f ? dyi : 0.f;
      }
      float xhat = (load_f(x, i, bf16) - mu) * is;
      float dxi = gis * (dz - sdz_n - xhat * sdzx_n);
      store_f(dx, i, dxi, bf16);
      if (dskip != nullptr) store_f(dskip, i, dz, bf16);
    }
  }
}

// ---- eval-mode fused normalize(+add)(+act) using running stats ----
template <typename T, int ACT>
__global__ void bn_act_eval_kernel(const T* __restrict__ x,
                                   const T* __restrict__ skip,
                                   const float* __restrict__ rm,
                                   const float* __restrict__ rv,
                                   const float* __restrict__ gamma,
                                   const float* __restrict__ beta,
                                   const float* __restrict__ a,
                                   T* __restrict__ out, int64_t n_pix, int C,
                                   float eps) {
  const bool bf16 = sizeof(T) == 2;
  int c, r_off, rows, per;
  if (C <= 256) { rows = 256 / C; c = threadIdx.x & (C - 1);
                  r_off = threadIdx.x / C; per = 1; }
  else { rows = 1; c = threadIdx.x; r_off = 0; per = C / 256; }
  for (int j = 0; j < per; ++j) {
    int cc = c + j * 256;
    float is = rsqrtf(rv[cc] + eps);
    float sc = gamma[cc] * is;
    float sh = beta[cc] - rm[cc] * sc;
    float av = (ACT == 1) ? a[cc] : 0.f;
    for (int64_t p = (int64_t)blockIdx.x * rows + r_off; p < n_pix;
         p += (int64_t)gridDim.x * rows) {
      int64_t i = p * C + cc;
      float zv = sc * load_f(x, i, bf16) + sh;
      if (skip != nullptr) zv += load_f(skip, i, bf16);
      float o = zv;
      if (ACT == 1) o = zv > 0.f ? zv : av * zv;
      else if (ACT == 2) o = fmaxf(zv, 0.f);
      store_f(out, i, o, bf16);
    }
  }
}

static inline int grid_for_pix(int64_t n_pix, int C) {
  int rows = C <= 256 ? 256 / C : 1;
  return (int)bd_min<int64_t>((n_pix + rows - 1) / rows, 2048);
}

template <typename T>
static void launch_fwd(const void* x, const void* skip, const float* mean,
                       const float* invstd, const float* gamma,
                       const float* beta, const float* a, void* out,
                       void* zout, int64_t n_pix, int C, int act_kind,
                       int grid, hipStream_t stream) {
  auto X = (const T*)x; auto S = (const T*)skip;
  auto O = (T*)out; auto Z = (T*)zout;
  if (act_kind == 1)
    bn_act_fwd_kernel<T, 1><<<grid, 256, 0, stream>>>(
        X, S, mean, invstd, gamma, beta, a, O, Z, n_pix, C);
  else if (act_kind == 2)
    bn_act_fwd_kernel<T, 2><<<grid, 256, 0, stream>>>(
        X, S, mean, invstd, gamma, beta, a, O, Z, n_pix, C);
  else
    bn_act_fwd_kernel<T, 0><<<grid, 256, 0, stream>>>(
        X, S, mean, invstd, gamma, beta, a, O, Z, n_pix, C);
}

template <typename T>
static void launch_bwd_reduce(const void* dy, const void* z, const void* x,
                              const float* mean, const float* invstd,
                              const float* a, float* sums, int64_t n_pix,
                              int C, int act_kind, int grid, size_t lds,
                              hipStream_t stream) {
  auto DY = (const T*)dy; auto ZZ = (const T*)z; auto X = (const T*)x;
  if (act_kind == 1)
    bn_act_bwd_reduce_kernel<T, 1><<<grid, 256, lds, stream>>>(
        DY, ZZ, X, mean, invstd, a, sums, n_pix, C);
  else if (act_kind == 2)
    bn_act_bwd_reduce_kernel<T, 2><<<grid, 256, lds, stream>>>(
        DY, ZZ, X, mean, invstd, a, sums, n_pix, C);
  else
    bn_act_bwd_reduce_kernel<T, 0><<<grid, 256, lds, stream>>>(
        DY, ZZ, X, mean, invstd, a, sums, n_pix, C);
}

template <typename T>
static void launch_bwd_apply(const void* dy, const void* z, const void* x,
                             const float* mean, const float* invstd,
                             const float* gamma, const float* a,
                             const float* sums, void* dx, void* dskip,
                             int64_t n_pix, int C, int act_kind, float inv_n,
                             int grid, hipStream_t stream) {
  auto DY = (const T*)dy; auto ZZ = (const T*)z; auto X = (const T*)x;
  auto DX = (T*)dx; auto DS = (T*)dskip;
  if (act_kind == 1)
    bn_act_bwd_apply_kernel<T, 1><<<grid, 256, 0, stream>>>(
        DY, ZZ, X, mean, invstd, gamma, a, sums, DX, DS, n_pix, C, inv_n);
  else if (act_kind == 2)
    bn_act_bwd_apply_kernel<T, 2><<<grid, 256, 0, stream>>>(
        DY, ZZ, X, mean, invstd, gamma, a, sums, DX, DS, n_pix, C, inv_n);
  else
    bn_act_bwd_apply_kernel<T, 0><<<grid, 256, 0, stream>>>(
        DY, ZZ, X, mean, invstd, gamma, a, sums, DX, DS, n_pix, C, inv_n);
}

template <typename T>
static void launch_eval(const void* x, const void* skip, const float* rm,
                        const float* rv, const float* gamma,
                        const float* beta, const float* a, void* out,
                        int64_t n_pix, int C, int act_kind, float eps,
                        int grid, hipStream_t stream) {
  auto X = (const T*)x; auto S = (const T*)skip; auto O = (T*)out;
  if (act_kind == 1)
    bn_act_eval_kernel<T, 1><<<grid, 256, 0, stream>>>(
        X, S, rm, rv, gamma, beta, a, O, n_pix, C, eps);
  else if (act_kind == 2)
    bn_act_eval_kernel<T, 2><<<grid, 256, 0, stream>>>(
        X, S, rm, rv, gamma, beta, a, O, n_pix, C, eps);
  else
    bn_act_eval_kernel<T, 0><<<grid, 256, 0, stream>>>(
        X, S, rm, rv, gamma, beta, a, O, n_pix, C, eps);
}

extern "C" void bdbnn_bn_stats(const void* x, float* s1, float* s2,
                               int64_t n, int C, bool bf16,
                               hipStream_t stream) {
  hipMemsetAsync(s1, 0, sizeof(float) * C, stream);
  hipMemsetAsync(s2, 0, sizeof(float) * C, stream);
  int64_t n_pix = n / C;
  int grid = grid_for_pix(n_pix, C);
  size_t lds = 2 * sizeof(float) * C;
  if (bf16)
    bn_stats_kernel<uint16_t><<<grid, 256, lds, stream>>>(
        (const uint16_t*)x, s1, s2, n_pix, C);
  else
    bn_stats_kernel<float><<<grid, 256, lds, stream>>>(
        (const float*)x, s1, s2, n_pix, C);
}

extern "C" void bdbnn_bn_finalize(const float* s1, const float* s2,
                                  float* mean, float* invstd,
                                  float* running_mean, float* running_var,
                                  int C, float n, float momentum, float eps,
                                  hipStream_t stream) {
  bn_finalize_kernel<<<(C + 255) / 256, 256, 0, stream>>>(
      s1, s2, mean, invstd, running_mean, running_var, C, n, momentum, eps);
}

extern "C" void bdbnn_bn_act_fwd(const void* x, const void* skip,
                                 const float* mean, const float* invstd,
                                 const float* gamma, const float* beta,
                                 const float* a, void* out, void* zout,
                                 int64_t n, int C, int act_kind, bool bf16,
                                 hipStream_t stream) {
  int64_t n_pix = n / C;
  int grid = grid_for_pix(n_pix, C);
  if (bf16)
    launch_fwd<uint16_t>(x, skip, mean, invstd, gamma, beta, a, out, zout,
                         n_pix, C, act_kind, grid, stream);
  else
    launch_fwd<float>(x, skip, mean, invstd, gamma, beta, a, out, zout,
                      n_pix, C, act_kind, grid, stream);
}

extern "C" void bdbnn_bn_act_bwd_reduce(const void* dy, const void* z,
                                        const void* x, const float* mean,
                                        const float* invstd, const float* a,
                                        float* sums, int64_t n, int C,
                                        int act_kind, bool bf16,
                                        hipStream_t stream) {
  hipMemsetAsync(sums, 0, sizeof(float) * C * 3, stream);
  int64_t n_pix = n / C;
  int grid = grid_for_pix(n_pix, C);
  size_t lds = 3 * sizeof(float) * C;
  if (bf16)
    launch_bwd_reduce<uint16_t>(dy, z, x, mean, invstd, a, sums, n_pix, C,
                                act_kind, grid, lds, stream);
  else
    launch_bwd_reduce<float>(dy, z, x, mean, invstd, a, sums, n_pix, C,
                             act_kind, grid, lds, stream);
}

extern "C" void bdbnn_bn_act_bwd_apply(const void* dy, const void* z,
                                       const void* x, const float* mean,
                                       const float* invstd,
                                       const float* gamma, const float* a,
                                       const float* sums, void* dx,
                                       void* dskip, int64_t n, int C,
                                       int act_kind, float inv_n, bool bf16,
                                       hipStream_t stream) {
  int64_t n_pix = n / C;
  int grid = grid_for_pix(n_pix, C);
  if (bf16)
    launch_bwd_apply<uint16_t>(dy, z, x, mean, invstd, gamma, a, sums, dx,
                               dskip, n_pix, C, act_kind, inv_n, grid,
                               stream);
  else
    launch_bwd_apply<float>(dy, z, x, mean, invstd, gamma, a, sums, dx,
                            dskip, n_pix, C, act_kind, inv_n, grid, stream);
}

extern "C" void bdbnn_bn_act_eval(const void* x, const void* skip,
                                  const float* rm, const float* rv,
                                  const float* gamma, const float* beta,
                                  const float* a, void* out, int64_t n,
                                  int C, int act_kind, float eps, bool bf16,
                                  hipStream_t stream) {
  int64_t n_pix = n / C;
  int grid = grid_for_pix(n_pix, C);
  if (bf16)
    launch_eval<uint16_t>(x, skip, rm, rv, gamma, beta, a, out, n_pix, C,
                          act_kind, eps, grid, stream);
  else
    launch_eval<float>(x, skip, rm, rv, gamma, beta, a, out, n_pix, C,
                       act_kind, eps, grid, stream);
}
