// K8 — fused BatchNorm (+ residual add) (+ PReLU/ReLU) for NHWC, training
// and eval.  Replaces the MIOpen BN pipeline + separate add + activation
// (6+ kernel launches and ~8 tensor passes per block position) with:
//   fwd: stats pass (one read) + normalize/add/act pass (reads x,skip;
//        writes out and the pre-activation z needed by backward)
//   bwd: reduce pass (reads dy,z,x -> per-channel sums + da) +
//        apply pass (reads dy,z,x -> writes dx and dskip)
//
// Memory layout: every thread owns 8 consecutive channels (16-B vector
// loads, guide G13), accumulates channel sums in registers over its pixel
// stripe, and lands one LDS add + one global atomic per owned channel.
// C must be a power of two, 8 <= C <= 1024 (python guards; fallback is
// the composition path).
//
// BN semantics match nn.BatchNorm2d: biased batch var for normalization,
// unbiased var into running_var, momentum update in the finalize step.
// act_kind: 0 = identity, 1 = per-channel PReLU, 2 = ReLU.
#include "common.h"
#include "vec8.h"

// ---- pass 1: per-channel sum / sumsq ----
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
  ChanMap m = chan_map8(C);
  float a1[8] = {}, a2[8] = {};
  // 4 independent pixel loads in flight per wave: one blocking load per
  // iteration leaves the wave latency-bound at ~0.6 B/cyc/CU (34 VGPR,
  // plenty of headroom for the extra 48 registers)
  float v0[8], v1[8], v2[8], v3[8];
  int64_t p = m.p0;
  for (; p + 3 * m.pstep < n_pix; p += 4 * m.pstep) {
    load8(x, p * C + m.c0, v0);
    load8(x, (p + m.pstep) * C + m.c0, v1);
    load8(x, (p + 2 * m.pstep) * C + m.c0, v2);
    load8(x, (p + 3 * m.pstep) * C + m.c0, v3);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      a1[j] += v0[j] + v1[j] + v2[j] + v3[j];
      a2[j] += v0[j] * v0[j] + v1[j] * v1[j] + v2[j] * v2[j] +
               v3[j] * v3[j];
    }
  }
  for (; p < n_pix; p += m.pstep) {
    load8(x, p * C + m.c0, v0);
#pragma unroll
    for (int j = 0; j < 8; ++j) { a1[j] += v0[j]; a2[j] += v0[j] * v0[j]; }
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    atomicAdd(&l1[m.c0 + j], a1[j]);
    atomicAdd(&l2[m.c0 + j], a2[j]);
  }
  __syncthreads();
  // 32-way sliced output ([32][C], folded by bn_finalize): full-grid
  // atomics onto C words serialize ~2048-deep otherwise
  const int off = (blockIdx.x & 31) * C;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    if (l1[c] != 0.f) atomicAdd(&s1[off + c], l1[c]);
    if (l2[c] != 0.f) atomicAdd(&s2[off + c], l2[c]);
  }
}

// ---- finalize: mean/invstd + running-stat update ----
// nslice > 1: s1/s2 are [nslice][C] partial sums (the conv epilogue's
// 32-way sliced accumulators) — summed here, where it costs nothing.
__global__ void bn_finalize_kernel(const float* __restrict__ s1,
                                   const float* __restrict__ s2,
                                   float* __restrict__ mean,
                                   float* __restrict__ invstd,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   int C, float n, float momentum, float eps,
                                   int nslice) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float a1 = 0.f, a2 = 0.f;
  for (int j = 0; j < nslice; ++j) {
    a1 += s1[j * C + c];
    a2 += s2[j * C + c];
  }
  float m = a1 / n;
  float var = fmaxf(a2 / n - m * m, 0.f);
  mean[c] = m;
  invstd[c] = rsqrtf(var + eps);
  if (running_mean != nullptr) {
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * m;
    float var_unb = var * n / fmaxf(n - 1.f, 1.f);
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * var_unb;
  }
}

// ---- pass 2: normalize + add + act (+ next conv's sign/mask pack) ----
// PACK: the consuming binary conv's sign + clip-STE-mask bitplanes are
// assembled in the epilogue from the ROUNDED stored values (bit-exact
// with csrc/pack.hip sign_mask_pack_kernel on the written tensor), so
// that conv never re-reads the activation to pack it.  Each thread owns
// 8 consecutive channels -> one byte of each plane; the 4 lanes of a
// 32-channel word are consecutive (C % 32 == 0 => threads-per-row % 4
// == 0) and combine with intra-wave shuffles, lane (tid&3)==0 stores.
template <typename T, int ACT, bool PACK>
__global__ void bn_act_fwd_kernel(const T* __restrict__ x,
                                  const T* __restrict__ skip,
                                  const float* __restrict__ mean,
                                  const float* __restrict__ invstd,
                                  const float* __restrict__ gamma,
                                  const float* __restrict__ beta,
                                  const float* __restrict__ a,
                                  T* __restrict__ out, T* __restrict__ zout,
                                  int64_t n_pix, int C,
                                  uint32_t* __restrict__ xpk,
                                  uint32_t* __restrict__ mpk) {
  ChanMap m = chan_map8(C);
  float sc[8], sh[8], av[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int cc = m.c0 + j;
    sc[j] = gamma[cc] * invstd[cc];
    sh[j] = beta[cc] - mean[cc] * sc[j];
    av[j] = (ACT == 1) ? a[cc] : 0.f;
  }
  const int sub = threadIdx.x & 3;         // byte slot within the word
  const int64_t cw = m.c0 >> 5;            // word column (lane sub == 0)
  const int CW = C >> 5;
  float v[8], s[8], z[8], o[8];
  for (int64_t p = m.p0; p < n_pix; p += m.pstep) {
    int64_t i = p * C + m.c0;
    load8(x, i, v);
    if (skip != nullptr) load8(skip, i, s);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      z[j] = sc[j] * v[j] + sh[j];
      if (skip != nullptr) z[j] += s[j];
      o[j] = z[j];
      if (ACT == 1) o[j] = z[j] > 0.f ? z[j] : av[j] * z[j];
      else if (ACT == 2) o[j] = fmaxf(z[j], 0.f);
    }
    store8(out, i, o);
    if (zout != nullptr) store8(zout, i, z);
    if constexpr (PACK) {
      unsigned both = 0;   // sign byte | mask byte << 8
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float r = (sizeof(T) == 2) ? bf16_to_f32(f32_to_bf16(o[j])) : o[j];
        both |= (r >= 0.f ? 1u : 0u) << j;
        both |= (fabsf(r) <= 1.f ? 1u : 0u) << (8 + j);
      }
      unsigned b1 = __shfl_down(both, 1);
      unsigned b2 = __shfl_down(both, 2);
      unsigned b3 = __shfl_down(both, 3);
      if (sub == 0) {
        uint32_t sw = (both & 0xffu) | ((b1 & 0xffu) << 8) |
                      ((b2 & 0xffu) << 16) | ((b3 & 0xffu) << 24);
        uint32_t mw = ((both >> 8) & 0xffu) | (((b1 >> 8) & 0xffu) << 8) |
                      (((b2 >> 8) & 0xffu) << 16) |
                      (((b3 >> 8) & 0xffu) << 24);
        xpk[p * CW + cw] = sw;
        mpk[p * CW + cw] = mw;
      }
    }
  }
}

// ---- backward pass 1: per-channel reductions ----
// sums layout: [32 slices][C][3] = (sum dz, sum dz*xhat, da) partials —
// 32-way sliced so the end-of-kernel global atomics see 1/32 of the
// per-word contention (2048 blocks onto 3C words measured as a >20 us
// serial tail); bn_fold_sums_kernel folds the slices.
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
  ChanMap m = chan_map8(C);
  float mu[8], is[8], av[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int cc = m.c0 + j;
    mu[j] = mean[cc]; is[j] = invstd[cc];
    av[j] = (ACT == 1) ? a[cc] : 0.f;
  }
  float s0[8] = {}, s1[8] = {}, s2[8] = {};
  // 2 pixel iterations in flight (6 independent loads): one blocking
  // triple per iteration leaves the wave latency-bound.  (The r1
  // attempt unrolled with a COMBINED body and regressed on register
  // pressure; issuing the loads up front then reducing keeps liveness
  // to the 6 vector buffers.)
  float dyv[8], zv[8], xv[8], dyw[8], zw[8], xw[8];
#define BN_RED_BODY(DYV, ZV, XV)                                          \
  {                                                                       \
    _Pragma("unroll")                                                     \
    for (int j = 0; j < 8; ++j) {                                         \
      float dz = DYV[j];                                                  \
      if (ACT == 1) {                                                     \
        dz = ZV[j] > 0.f ? DYV[j] : av[j] * DYV[j];                       \
        if (ZV[j] <= 0.f) s2[j] += DYV[j] * ZV[j];                        \
      } else if (ACT == 2) {                                              \
        dz = ZV[j] > 0.f ? DYV[j] : 0.f;                                  \
      }                                                                   \
      float xhat = (XV[j] - mu[j]) * is[j];                               \
      s0[j] += dz;                                                        \
      s1[j] += dz * xhat;                                                 \
    }                                                                     \
  }
  int64_t p = m.p0;
  for (; p + m.pstep < n_pix; p += 2 * m.pstep) {
    int64_t i = p * C + m.c0;
    int64_t i2 = (p + m.pstep) * C + m.c0;
    load8(dy, i, dyv);
    load8(x, i, xv);
    if (ACT != 0) load8(z, i, zv);
    load8(dy, i2, dyw);
    load8(x, i2, xw);
    if (ACT != 0) load8(z, i2, zw);
    BN_RED_BODY(dyv, zv, xv)
    BN_RED_BODY(dyw, zw, xw)
  }
  for (; p < n_pix; p += m.pstep) {
    int64_t i = p * C + m.c0;
    load8(dy, i, dyv);
    load8(x, i, xv);
    if (ACT != 0) load8(z, i, zv);
    BN_RED_BODY(dyv, zv, xv)
  }
#undef BN_RED_BODY
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    atomicAdd(&r0[m.c0 + j], s0[j]);
    atomicAdd(&r1[m.c0 + j], s1[j]);
    if (ACT == 1) atomicAdd(&r2[m.c0 + j], s2[j]);
  }
  __syncthreads();
  float* my = sums + (blockIdx.x & 31) * (C * 3);
  for (int cc = threadIdx.x; cc < C; cc += blockDim.x) {
    if (r0[cc] != 0.f) atomicAdd(&my[cc * 3 + 0], r0[cc]);
    if (r1[cc] != 0.f) atomicAdd(&my[cc * 3 + 1], r1[cc]);
    if (r2[cc] != 0.f) atomicAdd(&my[cc * 3 + 2], r2[cc]);
  }
}

// fold the 32 slices -> [C][3] (tiny; launched right after the reduce)
__global__ void bn_fold_sums_kernel(const float* __restrict__ sums32,
                                    float* __restrict__ out, int n) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  float v = 0.f;
#pragma unroll
  for (int s = 0; s < 32; ++s) v += sums32[s * n + i];
  out[i] = v;
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
  ChanMap m = chan_map8(C);
  float mu[8], is[8], gis[8], av[8], sdz_n[8], sdzx_n[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int cc = m.c0 + j;
    mu[j] = mean[cc]; is[j] = invstd[cc];
    gis[j] = gamma[cc] * is[j];
    av[j] = (ACT == 1) ? a[cc] : 0.f;
    sdz_n[j] = sums[cc * 3 + 0] * inv_n;
    sdzx_n[j] = sums[cc * 3 + 1] * inv_n;
  }
  // 2 pixel iterations in flight (up to 6 independent loads): one
  // blocking triple per iteration leaves the wave latency-bound
  float dyv[8], zv[8], xv[8], dxv[8], dzv[8];
  float dyw[8], zw[8], xw[8];
#define BN_APPLY_BODY(DYV, ZV, XV, I)                                     \
  {                                                                       \
    _Pragma("unroll")                                                     \
    for (int j = 0; j < 8; ++j) {                                         \
      float dz = DYV[j];                                                  \
      if (ACT == 1) dz = ZV[j] > 0.f ? DYV[j] : av[j] * DYV[j];           \
      else if (ACT == 2) dz = ZV[j] > 0.f ? DYV[j] : 0.f;                 \
      float xhat = (XV[j] - mu[j]) * is[j];                               \
      dxv[j] = gis[j] * (dz - sdz_n[j] - xhat * sdzx_n[j]);               \
      dzv[j] = dz;                                                        \
    }                                                                     \
    store8(dx, I, dxv);                                                   \
    if (dskip != nullptr) store8(dskip, I, dzv);                          \
  }
  int64_t p = m.p0;
  for (; p + m.pstep < n_pix; p += 2 * m.pstep) {
    int64_t i = p * C + m.c0;
    int64_t i2 = (p + m.pstep) * C + m.c0;
    load8(dy, i, dyv);
    load8(x, i, xv);
    if (ACT != 0) load8(z, i, zv);
    load8(dy, i2, dyw);
    load8(x, i2, xw);
    if (ACT != 0) load8(z, i2, zw);
    BN_APPLY_BODY(dyv, zv, xv, i)
    BN_APPLY_BODY(dyw, zw, xw, i2)
  }
  for (; p < n_pix; p += m.pstep) {
    int64_t i = p * C + m.c0;
    load8(dy, i, dyv);
    load8(x, i, xv);
    if (ACT != 0) load8(z, i, zv);
    BN_APPLY_BODY(dyv, zv, xv, i)
  }
#undef BN_APPLY_BODY
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
  ChanMap m = chan_map8(C);
  float sc[8], sh[8], av[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int cc = m.c0 + j;
    float is = rsqrtf(rv[cc] + eps);
    sc[j] = gamma[cc] * is;
    sh[j] = beta[cc] - rm[cc] * sc[j];
    av[j] = (ACT == 1) ? a[cc] : 0.f;
  }
  float v[8], s[8], o[8];
  for (int64_t p = m.p0; p < n_pix; p += m.pstep) {
    int64_t i = p * C + m.c0;
    load8(x, i, v);
    if (skip != nullptr) load8(skip, i, s);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float zv = sc[j] * v[j] + sh[j];
      if (skip != nullptr) zv += s[j];
      o[j] = zv;
      if (ACT == 1) o[j] = zv > 0.f ? zv : av[j] * zv;
      else if (ACT == 2) o[j] = fmaxf(zv, 0.f);
    }
    store8(out, i, o);
  }
}

// ---------------- launchers ----------------

template <typename T>
static void launch_fwd(const void* x, const void* skip, const float* mean,
                       const float* invstd, const float* gamma,
                       const float* beta, const float* a, void* out,
                       void* zout, int64_t n_pix, int C, int act_kind,
                       uint32_t* xpk, uint32_t* mpk, int grid,
                       hipStream_t stream) {
  auto X = (const T*)x; auto S = (const T*)skip;
  auto O = (T*)out; auto Z = (T*)zout;
#define BN_FWD(ACT)                                                       \
  {                                                                       \
    if (xpk != nullptr)                                                   \
      bn_act_fwd_kernel<T, ACT, true><<<grid, 256, 0, stream>>>(          \
          X, S, mean, invstd, gamma, beta, a, O, Z, n_pix, C, xpk, mpk);  \
    else                                                                  \
      bn_act_fwd_kernel<T, ACT, false><<<grid, 256, 0, stream>>>(         \
          X, S, mean, invstd, gamma, beta, a, O, Z, n_pix, C, xpk, mpk);  \
  }
  if (act_kind == 1) BN_FWD(1)
  else if (act_kind == 2) BN_FWD(2)
  else BN_FWD(0)
#undef BN_FWD
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
  hipMemsetAsync(s1, 0, sizeof(float) * 32 * C, stream);
  hipMemsetAsync(s2, 0, sizeof(float) * 32 * C, stream);
  int64_t n_pix = n / C;
  int grid = grid_pix8(n_pix, C);
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
                                  int nslice, hipStream_t stream) {
  bn_finalize_kernel<<<(C + 255) / 256, 256, 0, stream>>>(
      s1, s2, mean, invstd, running_mean, running_var, C, n, momentum, eps,
      nslice);
}

extern "C" void bdbnn_bn_act_fwd(const void* x, const void* skip,
                                 const float* mean, const float* invstd,
                                 const float* gamma, const float* beta,
                                 const float* a, void* out, void* zout,
                                 int64_t n, int C, int act_kind,
                                 uint32_t* xpk, uint32_t* mpk, bool bf16,
                                 hipStream_t stream) {
  int64_t n_pix = n / C;
  int grid = grid_pix8(n_pix, C);
  if (bf16)
    launch_fwd<uint16_t>(x, skip, mean, invstd, gamma, beta, a, out, zout,
                         n_pix, C, act_kind, xpk, mpk, grid, stream);
  else
    launch_fwd<float>(x, skip, mean, invstd, gamma, beta, a, out, zout,
                      n_pix, C, act_kind, xpk, mpk, grid, stream);
}

extern "C" void bdbnn_bn_act_bwd_reduce(const void* dy, const void* z,
                                        const void* x, const float* mean,
                                        const float* invstd, const float* a,
                                        float* sums32, float* sums,
                                        int64_t n, int C, int act_kind,
                                        bool bf16, hipStream_t stream) {
  hipMemsetAsync(sums32, 0, sizeof(float) * 32 * C * 3, stream);
  int64_t n_pix = n / C;
  int grid = grid_pix8(n_pix, C);
  size_t lds = 3 * sizeof(float) * C;
  if (bf16)
    launch_bwd_reduce<uint16_t>(dy, z, x, mean, invstd, a, sums32, n_pix, C,
                                act_kind, grid, lds, stream);
  else
    launch_bwd_reduce<float>(dy, z, x, mean, invstd, a, sums32, n_pix, C,
                             act_kind, grid, lds, stream);
  bn_fold_sums_kernel<<<(C * 3 + 255) / 256, 256, 0, stream>>>(sums32, sums,
                                                               C * 3);
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
  int grid = grid_pix8(n_pix, C);
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
  int grid = grid_pix8(n_pix, C);
  if (bf16)
    launch_eval<uint16_t>(x, skip, rm, rv, gamma, beta, a, out, n_pix, C,
                          act_kind, eps, grid, stream);
  else
    launch_eval<float>(x, skip, rm, rv, gamma, beta, a, out, n_pix, C,
                       act_kind, eps, grid, stream);
}
