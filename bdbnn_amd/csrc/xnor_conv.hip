// K1 — bit-packed XNOR+popcount binary convolution forward for gfx950.
//
// out[n,oy,ox,k] = alpha_k * dot( sign(x[patch]), sign(w[k]) )
// with zero-padding semantics (pad taps contribute 0).
//
// Implicit-GEMM formulation: M = N*Ho*Wo output pixels, N-dim = K output
// channels, K-dim = T*CW packed words (T = KH*KW taps, CW = ceil(C/32)).
// The +-1 dot product over one 32-channel word is
//     dot_w = 2*popc(a XOR b_inv) - 32
// because the weight pack stores INVERTED sign bits (csrc/pack.hip), so the
// inner loop is exactly {v_xor_b32, v_bcnt_u32_b32} per word = 64 binary ops
// per 2 VALU instructions.
//
// Padding is handled OUTSIDE the hot loop: pad taps load a = 0 and the
// epilogue subtracts the per-(k,tap) correction S[k][t] = C - 2*popc(sign_w)
// for each invalid tap (see pack.hip for the garbage-bit convention; every
// garbage tail bit contributes exactly 1 to popc, a constant G).
//
//   dot(sp,k) = 2*POP + BASE - sum_{t invalid(sp)} S[k][t],
//   BASE = -2*G - C*T,  G = (32*CW - C) * T.
//
// Tiling: 256-thread block computes a 128(spatial) x 64(channel) tile;
// both operands staged in LDS in 8-word chunks; each thread owns an
// 8(spatial) x 4(channel) register tile read via ds_read_b128 (a-words
// broadcast within a 16-lane group, b4 conflict-free across the 256-B
// bank row): 1024 binary MACs per ~70 instructions.
#include "common.h"

#define TILE_M 128
#define TILE_K 64
#define CHUNK 8

struct XnorConvParams {
  int N, H, W, C, K, KH, KW, stride, pad, Ho, Wo, CW;
  int WORDS;   // KH*KW*CW
  int T;       // KH*KW
  int base;    // -2*G - C*T
};

template <typename TO>
__global__ __launch_bounds__(256) void xnor_conv_kernel(
    const uint32_t* __restrict__ xp, const uint32_t* __restrict__ wp,
    const float* __restrict__ alpha, const float* __restrict__ stab,
    TO* __restrict__ out, XnorConvParams p, int grid_m) {
  // XCD-aware block remap (8 XCDs, private L2s): give each XCD a
  // contiguous run of spatial tiles so neighbouring tiles (sharing input
  // rows) land on one L2.  bijective for any grid size.
  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, idx = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  int m_blk = wg % grid_m;        // spatial tile
  int k_blk = wg / grid_m;        // channel tile
  const int tid = threadIdx.x;
  const int64_t M = (int64_t)p.N * p.Ho * p.Wo;
  const int64_t m0 = (int64_t)m_blk * TILE_M;
  const int k0_blk = k_blk * TILE_K;

  __shared__ uint32_t a_lds[CHUNK][TILE_M];
  __shared__ uint32_t w_lds[CHUNK][TILE_K];
  __shared__ int row_base[TILE_M];    // input pixel index of tap (0,0)
  __shared__ unsigned short row_inv[TILE_M];  // invalid-tap bitmask (T<=9)

  // ---- per-row metadata (once per block) ----
  for (int r = tid; r < TILE_M; r += blockDim.x) {
    int64_t sp = m0 + r;
    if (sp >= M) { row_base[r] = 0; row_inv[r] = 0xffff; continue; }
    int n = int(sp / ((int64_t)p.Ho * p.Wo));
    int rem = int(sp % ((int64_t)p.Ho * p.Wo));
    int oy = rem / p.Wo, ox = rem % p.Wo;
    int iy0 = oy * p.stride - p.pad;
    int ix0 = ox * p.stride - p.pad;
    row_base[r] = (n * p.H + iy0) * p.W + ix0;
    unsigned short inv = 0;
    for (int t = 0; t < p.T; ++t) {
      int kh = t / p.KW, kw = t % p.KW;
      int iy = iy0 + kh, ix = ix0 + kw;
      if (iy < 0 || iy >= p.H || ix < 0 || ix >= p.W) inv |= 1u << t;
    }
    row_inv[r] = inv;
  }
  __syncthreads();

  // thread's register tile: rows r0..r0+7, channels kq..kq+3
  const int r0 = (tid / 16) * 8;
  const int kq = (tid % 16) * 4;
  int acc[8][4] = {};

  const int n_chunks = (p.WORDS + CHUNK - 1) / CHUNK;
  for (int ch = 0; ch < n_chunks; ++ch) {
    const int w0 = ch * CHUNK;
    // ---- stage: 4 a-words + 2 w-words per thread ----
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      int j = tid + it * 256;
      int c = j / TILE_M, r = j % TILE_M;
      int widx = w0 + c;
      uint32_t av = 0;
      if (widx < p.WORDS) {
        int t = widx / p.CW, cw = widx - t * p.CW;
        if (!((row_inv[r] >> t) & 1)) {
          int kh = t / p.KW, kw = t - kh * p.KW;
          int64_t pix = (int64_t)row_base[r] + kh * p.W + kw;
          av = xp[pix * p.CW + cw];
        }
      }
      a_lds[c][r] = av;
    }
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      int j = tid + it * 256;
      int c = j / TILE_K, kk = j % TILE_K;
      int widx = w0 + c;
      int kg = k0_blk + kk;
      uint32_t wv = 0;
      if (widx < p.WORDS && kg < p.K)
        wv = wp[(int64_t)kg * p.WORDS + widx];
      w_lds[c][kk] = wv;
    }
    __syncthreads();
    // ---- compute ----
#pragma unroll
    for (int c = 0; c < CHUNK; ++c) {
      uint32_t a8[8], b4[4];
      *(uint4*)&a8[0] = *(const uint4*)&a_lds[c][r0];
      *(uint4*)&a8[4] = *(const uint4*)&a_lds[c][r0 + 4];
      *(uint4*)b4 = *(const uint4*)&w_lds[c][kq];
#pragma unroll
      for (int i = 0; i < 8; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] += __popc(a8[i] ^ b4[j]);
    }
    __syncthreads();
  }

  // ---- epilogue: scale, pad-correction, store ----
  float al[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    int kg = k0_blk + kq + j;
    al[j] = (kg < p.K) ? alpha[kg] : 0.f;
  }
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    int64_t sp = m0 + r0 + i;
    if (sp >= M) continue;
    unsigned inv = row_inv[r0 + i];
    float corr[4] = {0.f, 0.f, 0.f, 0.f};
    if (inv) {
      for (int t = 0; t < p.T; ++t)
        if ((inv >> t) & 1) {
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            int kg = k0_blk + kq + j;
            if (kg < p.K) corr[j] += stab[(int64_t)kg * p.T + t];
          }
        }
    }
    if constexpr (sizeof(TO) == 2) {
      uint16_t vals[4];
#pragma unroll
      for (int j = 0; j < 4; ++j)
        vals[j] = f32_to_bf16(al[j] * (2.f * acc[i][j] + p.base - corr[j]));
      int kg = k0_blk + kq;
      if (kg + 3 < p.K)
        *(uint2*)&out[sp * p.K + kg] = *(uint2*)vals;
      else
        for (int j = 0; j < 4 && kg + j < p.K; ++j)
          out[sp * p.K + kg + j] = vals[j];
    } else {
      float vals[4];
#pragma unroll
      for (int j = 0; j < 4; ++j)
        vals[j] = al[j] * (2.f * acc[i][j] + p.base - corr[j]);
      int kg = k0_blk + kq;
      if (kg + 3 < p.K)
        *(float4*)&out[sp * p.K + kg] = *(float4*)vals;
      else
        for (int j = 0; j < 4 && kg + j < p.K; ++j)
          ((float*)out)[sp * p.K + kg + j] = vals[j];
    }
  }
}

extern "C" void bdbnn_xnor_conv_fwd(
    const uint32_t* xp, const uint32_t* wp, const float* alpha,
    const float* stab, void* out, bool out_bf16,
    int N, int H, int W, int C, int K, int KH, int KW, int stride, int pad,
    int Ho, int Wo, hipStream_t stream) {
  XnorConvParams p;
  p.N = N; p.H = H; p.W = W; p.C = C; p.K = K; p.KH = KH; p.KW = KW;
  p.stride = stride; p.pad = pad; p.Ho = Ho; p.Wo = Wo;
  p.CW = (C + 31) / 32;
  p.T = KH * KW;
  p.WORDS = p.T * p.CW;
  int G = (32 * p.CW - C) * p.T;
  p.base = -2 * G - C * p.T;
  int64_t M = (int64_t)N * Ho * Wo;
  int grid_m = int((M + TILE_M - 1) / TILE_M);
  int grid_k = (K + TILE_K - 1) / TILE_K;
  dim3 grid(grid_m * grid_k);
  if (out_bf16)
    xnor_conv_kernel<uint16_t><<<grid, 256, 0, stream>>>(
        xp, wp, alpha, stab, (uint16_t*)out, p, grid_m);
  else
    xnor_conv_kernel<float><<<grid, 256, 0, stream>>>(
        xp, wp, alpha, stab, (float*)out, p, grid_m);
}
