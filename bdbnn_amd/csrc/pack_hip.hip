#include "hip/hip_runtime.h"
// K2 — quantizer kernels: sign+bitpack, +-1 decode, STE/EDE backward masks,
// weight pack (+ per-channel alpha and the per-tap padding-correction table).
//
// Bit conventions (the framework-wide contract, see ops/binary_conv.py):
//   activation pack: bit c of word (p, cw) = 1  iff  x[p, 32*cw+c] >= 0
//   weight pack:     bit c = 1               iff  w < 0        (INVERTED)
// so that a_word XOR w_word == XNOR(sign_a, sign_w) and the conv inner loop
// is two VALU ops per 32 channels.  Unused tail bits (C % 32 != 0): a-pack
// stores 0, w-pack stores 1 — every garbage bit then contributes exactly 1
// to the popcount sum, a compile-known constant the epilogue subtracts.
#include "common.h"
#include "vec8.h"

// ---------------- activation sign+pack ----------------
// x: NHWC-contiguous (channels_last torch tensor), P = N*H*W pixels,
// out: uint32 [P][CW].  One thread per output word.
template <typename T>
__global__ void sign_pack_kernel(const T* __restrict__ x,
                                 uint32_t* __restrict__ out,
                                 int64_t n_words, int C, int CW) {
  GRID_STRIDE(i, n_words) {
    int cw = int(i % CW);
    int64_t p = i / CW;
    const T* px = x + p * C + cw * 32;
    int nbits = min(32, C - cw * 32);
    uint32_t bits = 0;
    if (nbits == 32) {
#pragma unroll
      for (int c = 0; c < 32; ++c) {
        float v;
        if constexpr (sizeof(T) == 2) v = bf16_to_f32(((const uint16_t*)px)[c]);
        else                          v = ((const float*)px)[c];
        bits |= (v >= 0.f ? 1u : 0u) << c;
      }
    } else {
      for (int c = 0; c < nbits; ++c) {
        float v;
        if constexpr (sizeof(T) == 2) v = bf16_to_f32(((const uint16_t*)px)[c]);
        else                          v = ((const float*)px)[c];
        bits |= (v >= 0.f ? 1u : 0u) << c;
      }
    }
    out[i] = bits;
  }
}

extern "C" void bdbnn_sign_pack(const void* x, uint32_t* out, int64_t pixels,
                                int C, int CW, bool bf16,
                                hipStream_t stream) {
  int64_t n_words = pixels * CW;
  int block = 256;
  int grid = (int)bd_min<int64_t>((n_words + block - 1) / block, 65535 * 8);
  if (bf16)
   hipLaunchKernelGGL(( sign_pack_kernel<uint16_t>), dim3(grid), dim3(block), 0, stream, 
        (const uint16_t*)x, out, n_words, C, CW);
  else
   hipLaunchKernelGGL(( sign_pack_kernel<float>), dim3(grid), dim3(block), 0, stream, 
        (const float*)x, out, n_words, C, CW);
}

// ---------------- +-1 decode (for the dense MFMA backward) ----------------
// 8-wide vectorized (n % 8 == 0 guaranteed by callers: numel is a
// multiple of C >= 8).
template <typename TI, typename TO>
__global__ void binsign_decode_kernel(const TI* __restrict__ x,
                                      TO* __restrict__ y, int64_t n8) {
  GRID_STRIDE(q, n8) {
    int64_t i = q * 8;
    float v[8], s[8];
    load8(x, i, v);
#pragma unroll
    for (int j = 0; j < 8; ++j) s[j] = v[j] >= 0.f ? 1.f : -1.f;
    store8(y, i, s);
  }
}

extern "C" void bdbnn_binsign_decode(const void* x, void* y, int64_t n,
                                     bool in_bf16, bool out_bf16,
                                     hipStream_t stream) {
  int block = 256;
  int64_t n8 = n / 8;   // callers guarantee n % 8 == 0
  int grid = (int)bd_min<int64_t>((n8 + block - 1) / block, 2048);
  n = n8;
#define CASE(IB, OB, TI, TO)                                              \
  if (in_bf16 == IB && out_bf16 == OB)                                    \
   hipLaunchKernelGGL(( binsign_decode_kernel<TI, TO>), dim3(grid), dim3(block), 0, stream,             \
        (const TI*)x, (TO*)y, n);
  CASE(false, false, float, float)
  CASE(false, true, float, uint16_t)
  CASE(true, false, uint16_t, float)
  CASE(true, true, uint16_t, uint16_t)
#undef CASE
}

// ---------------- STE/EDE mask multiply (quantizer backward) ----------------
// y = g * mask(x); mode 0 = clip-STE 1(|x|<=1), 1 = ReActNet polynomial,
// 2 = EDE k*t*(1-tanh^2(t*x)).  Output dtype = x dtype.
template <typename TG, typename TX>
__global__ void ste_mask_mul_kernel(const TG* __restrict__ g,
                                    const TX* __restrict__ x,
                                    TX* __restrict__ y, int64_t n8,
                                    int mode, float t, float k) {
  GRID_STRIDE(q, n8) {
    int64_t i = q * 8;
    float gv[8], xv[8], o[8];
    load8(g, i, gv);
    load8(x, i, xv);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float m;
      if (mode == 0) {
        m = fabsf(xv[j]) <= 1.f ? 1.f : 0.f;
      } else if (mode == 1) {
        m = (xv[j] >= -1.f && xv[j] < 0.f) ? 2.f + 2.f * xv[j]
          : (xv[j] >= 0.f && xv[j] < 1.f) ? 2.f - 2.f * xv[j] : 0.f;
      } else {
        float th = tanhf(t * xv[j]);
        m = k * t * (1.f - th * th);
      }
      o[j] = gv[j] * m;
    }
    store8(y, i, o);
  }
}

extern "C" void bdbnn_ste_mask_mul(const void* g, const void* x, void* y,
                                   int64_t n, bool g_bf16, bool x_bf16,
                                   int mode, float t, float k,
                                   hipStream_t stream) {
  int block = 256;
  int64_t n8 = n / 8;   // callers guarantee n % 8 == 0
  int grid = (int)bd_min<int64_t>((n8 + block - 1) / block, 2048);
  n = n8;
#define CASE(GB, XB, TG, TX)                                              \
  if (g_bf16 == GB && x_bf16 == XB)                                       \
   hipLaunchKernelGGL(( ste_mask_mul_kernel<TG, TX>), dim3(grid), dim3(block), 0, stream,               \
        (const TG*)g, (const TX*)x, (TX*)y, n, mode, t, k);
  CASE(false, false, float, float)
  CASE(false, true, float, uint16_t)
  CASE(true, false, uint16_t, float)
  CASE(true, true, uint16_t, uint16_t)
#undef CASE
}

// ---------------- weight pack ----------------
// w: fp32 contiguous [K][C][KH][KW].
// wp: uint32 [K][KH][KW][CW]   (bit = 1 iff w < 0; garbage tail bits = 1)
// alpha: fp32 [K]              (mean |w[k]|)
// stab:  fp32 [K][T]           (T = KH*KW; S[k][t] = C - 2*popc_real(t))
//                              the zero-pad epilogue correction table.
__global__ void weight_alpha_kernel(const float* __restrict__ w,
                                    float* __restrict__ alpha,
                                    int K, int64_t per_k) {
  __shared__ float red[256];
  int k = blockIdx.x;
  const float* wk = w + (int64_t)k * per_k;
  float s = 0.f;
  for (int64_t i = threadIdx.x; i < per_k; i += blockDim.x)
    s += fabsf(wk[i]);
  red[threadIdx.x] = s;
  __syncthreads();
  for (int off = 128; off > 0; off >>= 1) {
    if (threadIdx.x < off) red[threadIdx.x] += red[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) alpha[k] = red[0] / float(per_k);
}

__global__ void weight_pack_kernel(const float* __restrict__ w,
                                   uint32_t* __restrict__ wp,
                                   int K, int C, int T, int CW) {
  // one thread per packed word (k, t, cw); w layout stride of c is T
  int64_t n_words = (int64_t)K * T * CW;
  GRID_STRIDE(i, n_words) {
    int cw = int(i % CW);
    int t = int((i / CW) % T);
    int k = int(i / CW / T);
    int nbits = min(32, C - cw * 32);
    const float* wk = w + ((int64_t)k * C) * T + t;
    uint32_t bits = 0;
    for (int c = 0; c < nbits; ++c)
      if (wk[(int64_t)(cw * 32 + c) * T] < 0.f)
        bits |= 1u << c;                  // inverted convention
    // garbage tail bits = 1 (XOR with a-pack's 0 counts exactly 1)
    if (nbits < 32) bits |= ~((1u << nbits) - 1u);
    wp[i] = bits;
  }
}

// pad-correction table from the packed (inverted) bits:
// S[k][t] = C - 2*q_t = 2*popc(inverted real bits) - C.
__global__ void weight_stab_kernel(const uint32_t* __restrict__ wp,
                                   float* __restrict__ stab, int K, int C,
                                   int T, int CW) {
  int64_t n_taps = (int64_t)K * T;
  int garbage = 32 * CW - C;   // garbage bits are 1 in every tap
  GRID_STRIDE(i, n_taps) {
    int pop = 0;
    for (int cw = 0; cw < CW; ++cw) pop += __popc(wp[i * CW + cw]);
    stab[i] = float(2 * (pop - garbage) - C);
  }
}

extern "C" void bdbnn_weight_pack(const float* w, uint32_t* wp, float* alpha,
                                  float* stab, int K, int C, int KH, int KW,
                                  int CW, hipStream_t stream) {
  int64_t per_k = (int64_t)C * KH * KW;
 hipLaunchKernelGGL(( weight_alpha_kernel), dim3(K), dim3(256), 0, stream, w, alpha, K, per_k);
  int T = KH * KW;
  int64_t n_words = (int64_t)K * T * CW;
  int block = 256;
  int grid = (int)bd_min<int64_t>((n_words + block - 1) / block, 4096);
 hipLaunchKernelGGL(( weight_pack_kernel), dim3(grid), dim3(block), 0, stream, w, wp, K, C, T, CW);
  int64_t n_taps = (int64_t)K * T;
  int grid2 = (int)bd_min<int64_t>((n_taps + block - 1) / block, 4096);
 hipLaunchKernelGGL(( weight_stab_kernel), dim3(grid2), dim3(block), 0, stream, wp, stab, K, C, T, CW);
}

// ---------------- sign+mask pack (one pass) ----------------
// Emits BOTH bitplanes the 'ste' backward needs: sign bits (for the
// XNOR conv and the +-1 decode) and |x| <= 1 clip-STE mask bits.
// Backward then never touches the fp activations, and the forward does
// not save them.
template <typename T>
__global__ void sign_mask_pack_kernel(const T* __restrict__ x,
                                      uint32_t* __restrict__ sp,
                                      uint32_t* __restrict__ mp,
                                      int64_t n_words, int C, int CW) {
  GRID_STRIDE(i, n_words) {
    int cw = int(i % CW);
    int64_t p = i / CW;
    const T* px = x + p * C + cw * 32;
    int nbits = min(32, C - cw * 32);
    uint32_t sbits = 0, mbits = 0;
    if (nbits == 32) {
#pragma unroll
      for (int c = 0; c < 32; ++c) {
        float v;
        if constexpr (sizeof(T) == 2) v = bf16_to_f32(((const uint16_t*)px)[c]);
        else                          v = ((const float*)px)[c];
        sbits |= (v >= 0.f ? 1u : 0u) << c;
        mbits |= (fabsf(v) <= 1.f ? 1u : 0u) << c;
      }
    } else {
      for (int c = 0; c < nbits; ++c) {
        float v;
        if constexpr (sizeof(T) == 2) v = bf16_to_f32(((const uint16_t*)px)[c]);
        else                          v = ((const float*)px)[c];
        sbits |= (v >= 0.f ? 1u : 0u) << c;
        mbits |= (fabsf(v) <= 1.f ? 1u : 0u) << c;
      }
    }
    sp[i] = sbits;
    mp[i] = mbits;
  }
}

extern "C" void bdbnn_sign_mask_pack(const void* x, uint32_t* sp,
                                     uint32_t* mp, int64_t pixels, int C,
                                     int CW, bool bf16, hipStream_t stream) {
  int64_t n_words = pixels * CW;
  int block = 256;
  int grid = (int)bd_min<int64_t>((n_words + block - 1) / block, 4096);
  if (bf16)
   hipLaunchKernelGGL(( sign_mask_pack_kernel<uint16_t>), dim3(grid), dim3(block), 0, stream, 
        (const uint16_t*)x, sp, mp, n_words, C, CW);
  else
   hipLaunchKernelGGL(( sign_mask_pack_kernel<float>), dim3(grid), dim3(block), 0, stream, 
        (const float*)x, sp, mp, n_words, C, CW);
}

// ---------------- decode packed signs -> +-1 NHWC tensor ----------------
// one thread per 32-channel word; writes 32 consecutive elements.
template <typename TO>
__global__ void decode_packed_kernel(const uint32_t* __restrict__ sp,
                                     TO* __restrict__ y, int64_t n_words,
                                     int C, int CW) {
  GRID_STRIDE(i, n_words) {
    int cw = int(i % CW);
    int64_t p = i / CW;
    int nbits = min(32, C - cw * 32);
    uint32_t bits = sp[i];
    TO* py = y + p * C + cw * 32;
    if (nbits == 32) {
#pragma unroll
      for (int c = 0; c < 32; ++c) {
        float s = (bits >> c) & 1 ? 1.f : -1.f;
        if constexpr (sizeof(TO) == 2) ((uint16_t*)py)[c] = f32_to_bf16(s);
        else                           ((float*)py)[c] = s;
      }
    } else {
      for (int c = 0; c < nbits; ++c) {
        float s = (bits >> c) & 1 ? 1.f : -1.f;
        if constexpr (sizeof(TO) == 2) ((uint16_t*)py)[c] = f32_to_bf16(s);
        else                           ((float*)py)[c] = s;
      }
    }
  }
}

extern "C" void bdbnn_decode_packed(const uint32_t* sp, void* y,
                                    int64_t pixels, int C, int CW,
                                    bool out_bf16, hipStream_t stream) {
  int64_t n_words = pixels * CW;
  int block = 256;
  int grid = (int)bd_min<int64_t>((n_words + block - 1) / block, 4096);
  if (out_bf16)
   hipLaunchKernelGGL(( decode_packed_kernel<uint16_t>), dim3(grid), dim3(block), 0, stream, 
        sp, (uint16_t*)y, n_words, C, CW);
  else
   hipLaunchKernelGGL(( decode_packed_kernel<float>), dim3(grid), dim3(block), 0, stream, 
        sp, (float*)y, n_words, C, CW);
}

// ---------------- masked grad: dx = mask_bit ? g : 0 ----------------
template <typename TG, typename TO>
__global__ void mask_mul_packed_kernel(const TG* __restrict__ g,
                                       const uint32_t* __restrict__ mp,
                                       TO* __restrict__ dx, int64_t n_words,
                                       int C, int CW) {
  GRID_STRIDE(i, n_words) {
    int cw = int(i % CW);
    int64_t p = i / CW;
    int nbits = min(32, C - cw * 32);
    uint32_t bits = mp[i];
    const TG* pg = g + p * C + cw * 32;
    TO* pd = dx + p * C + cw * 32;
    if (nbits == 32) {
#pragma unroll
      for (int c = 0; c < 32; ++c) {
        float gv;
        if constexpr (sizeof(TG) == 2) gv = bf16_to_f32(((const uint16_t*)pg)[c]);
        else                           gv = ((const float*)pg)[c];
        float o = (bits >> c) & 1 ? gv : 0.f;
        if constexpr (sizeof(TO) == 2) ((uint16_t*)pd)[c] = f32_to_bf16(o);
        else                           ((float*)pd)[c] = o;
      }
    } else {
      for (int c = 0; c < nbits; ++c) {
        float gv;
        if constexpr (sizeof(TG) == 2) gv = bf16_to_f32(((const uint16_t*)pg)[c]);
        else                           gv = ((const float*)pg)[c];
        float o = (bits >> c) & 1 ? gv : 0.f;
        if constexpr (sizeof(TO) == 2) ((uint16_t*)pd)[c] = f32_to_bf16(o);
        else                           ((float*)pd)[c] = o;
      }
    }
  }
}

extern "C" void bdbnn_mask_mul_packed(const void* g, const uint32_t* mp,
                                      void* dx, int64_t pixels, int C,
                                      int CW, bool g_bf16, bool out_bf16,
                                      hipStream_t stream) {
  int64_t n_words = pixels * CW;
  int block = 256;
  int grid = (int)bd_min<int64_t>((n_words + block - 1) / block, 4096);
#define MCASE(GB, OB, TG, TO)                                             \
  if (g_bf16 == GB && out_bf16 == OB)                                     \
   hipLaunchKernelGGL(( mask_mul_packed_kernel<TG, TO>), dim3(grid), dim3(block), 0, stream,            \
        (const TG*)g, mp, (TO*)dx, n_words, C, CW);
  MCASE(false, false, float, float)
  MCASE(false, true, float, uint16_t)
  MCASE(true, false, uint16_t, float)
  MCASE(true, true, uint16_t, uint16_t)
#undef MCASE
}

// ---------------- decode packed weights -> alpha * (+-1), NCHW ----------
// wp: [K][KH][KW][CW] inverted-bit convention (bit 1 <=> w < 0).
template <typename TO>
__global__ void weight_decode_kernel(const uint32_t* __restrict__ wp,
                                     const float* __restrict__ alpha,
                                     TO* __restrict__ w, int K, int C,
                                     int T, int CW) {
  int64_t n_words = (int64_t)K * T * CW;
  GRID_STRIDE(i, n_words) {
    int cw = int(i % CW);
    int t = int((i / CW) % T);
    int k = int(i / CW / T);
    int nbits = min(32, C - cw * 32);
    uint32_t bits = wp[i];
    float al = alpha[k];
    // w layout [K][C][KH][KW]: stride of c is T
    TO* pw = w + ((int64_t)k * C + cw * 32) * T + t;
    for (int c = 0; c < nbits; ++c) {
      float v = (bits >> c) & 1 ? -al : al;  // inverted convention
      if constexpr (sizeof(TO) == 2) ((uint16_t*)pw)[(int64_t)c * T] = f32_to_bf16(v);
      else                           ((float*)pw)[(int64_t)c * T] = v;
    }
  }
}

extern "C" void bdbnn_weight_decode(const uint32_t* wp, const float* alpha,
                                    void* w, int K, int C, int T, int CW,
                                    bool out_bf16, hipStream_t stream) {
  int64_t n_words = (int64_t)K * T * CW;
  int block = 256;
  int grid = (int)bd_min<int64_t>((n_words + block - 1) / block, 4096);
  if (out_bf16)
   hipLaunchKernelGGL(( weight_decode_kernel<uint16_t>), dim3(grid), dim3(block), 0, stream, 
        wp, alpha, (uint16_t*)w, K, C, T, CW);
  else
   hipLaunchKernelGGL(( weight_decode_kernel<float>), dim3(grid), dim3(block), 0, stream, 
        wp, alpha, (float*)w, K, C, T, CW);
}
