// SUPERSEDED on the hot path by csrc/conv_bwd.hip (wgrad2): kept as an
// independent rocWMMA implementation — a second oracle the v2 kernel's
// GPU numerics tests cross-check against, and the minimal readable
// form of the algorithm.  MFMA bf16 wgrad for the
// 3x3/stride-1/pad-1 binary convs.
//
//   dW[k,c,kh,kw] = sum_{n,oy,ox} g[n,oy,ox,k] * xb[n,oy-1+kh,ox-1+kw,c]
//
// Per tap t it is a (K x M)(M x C) GEMM with M = N*Ho*Wo; the +-1 xb
// operand is decoded IN-KERNEL from the packed sign bits (zero at the
// borders).  Split-K over pixels; fp32 atomicAdd combine into dW
// (output is tiny: K*C*9 words).  One wave per block, 32x32 dW tile.
#include "common.h"
#include <rocwmma/rocwmma.hpp>

using rocwmma::fragment;
using rocwmma::matrix_a;
using rocwmma::matrix_b;
using rocwmma::accumulator;
using rocwmma::row_major;
using rocwmma::col_major;
using bf16_t = rocwmma::bfloat16_t;

#define WG_TK 32   // k tile
#define WG_TC 32   // c tile
#define WG_BM 16   // pixels per MFMA step

struct WgradParams {
  int N, H, W, C, K, CW;
  int64_t M;            // N*H*W
  int64_t chunk;        // pixels per block
};

__global__ __launch_bounds__(64) void conv_wgrad_kernel(
    const bf16_t* __restrict__ g, const uint32_t* __restrict__ xp,
    float* __restrict__ dw, WgradParams p, int grid_k, int grid_c,
    int grid_t) {
  // block -> (k tile, c tile, tap, pixel split)
  int b = blockIdx.x;
  const int kt = b % grid_k; b /= grid_k;
  const int ct = b % grid_c; b /= grid_c;
  const int t = b % grid_t;  b /= grid_t;
  const int split = b;
  const int kh = t / 3, kw = t % 3;
  const int lane = threadIdx.x;
  const int64_t HW = (int64_t)p.H * p.W;

  __shared__ bf16_t b_lds[WG_BM][WG_TC];
  __shared__ float out_st[32][32];

  fragment<accumulator, 32, 32, 16, float> acc;
  rocwmma::fill_fragment(acc, 0.f);

  const int64_t m0 = split * p.chunk;
  const int64_t m1 = (m0 + p.chunk < p.M) ? m0 + p.chunk : p.M;
  // B staging: each of 64 lanes decodes 8 (pix, c) values: lane covers
  // pix = lane/4 (16 rows x 4 lanes), c = (lane%4)*8 + 0..7
  const int s_pix = lane >> 2;
  const int s_c0 = (lane & 3) * 8;

  for (int64_t m = m0; m + WG_BM <= m1; m += WG_BM) {
    // decode xb chunk [16 pixels][32 c] for this tap (0 when padded)
    {
      int64_t pix = m + s_pix;
      int n = int(pix / HW);
      int rem = int(pix % HW);
      int iy = rem / p.W - 1 + kh;
      int ix = rem % p.W - 1 + kw;
      bf16_t v[8];
      if (iy >= 0 && iy < p.H && ix >= 0 && ix < p.W) {
        int c = ct * WG_TC + s_c0;
        uint32_t word =
            xp[(((int64_t)n * p.H + iy) * p.W + ix) * p.CW + (c >> 5)];
        int sh = c & 31;
#pragma unroll
        for (int i = 0; i < 8; ++i)
          v[i] = bf16_t(((word >> (sh + i)) & 1) ? 1.f : -1.f);
      } else {
#pragma unroll
        for (int i = 0; i < 8; ++i) v[i] = bf16_t(0.f);
      }
      *(uint4*)&b_lds[s_pix][s_c0] = *(uint4*)v;
    }
    __builtin_amdgcn_s_barrier();   // single wave: order LDS writes/reads
    // A = g^T: col_major load straight from global (ld = K)
    fragment<matrix_a, 32, 32, 16, bf16_t, col_major> fa;
    fragment<matrix_b, 32, 32, 16, bf16_t, row_major> fb;
    rocwmma::load_matrix_sync(fa, g + m * p.K + kt * WG_TK, p.K);
    rocwmma::load_matrix_sync(fb, &b_lds[0][0], WG_TC);
    rocwmma::mma_sync(acc, fa, fb, acc);
    __builtin_amdgcn_s_barrier();
  }
  // tail pixels (m1 - m not multiple of 16): handled by the LAST split
  // only when chunk is not 16-aligned; python guards chunk % 16 == 0 and
  // M % 16 == 0 so there is no tail.

  rocwmma::store_matrix_sync(&out_st[0][0], acc, 32, rocwmma::mem_row_major);
  __builtin_amdgcn_s_barrier();
  // atomicAdd the 32x32 tile into dW[k][c][kh][kw] (strided: c stride 9)
  for (int e = lane; e < 32 * 32; e += 64) {
    int kk = e >> 5, cc = e & 31;
    float v = out_st[kk][cc];
    if (v != 0.f) {
      int64_t k = kt * WG_TK + kk;
      int64_t c = ct * WG_TC + cc;
      atomicAdd(&dw[((k * p.C + c) * 3 + kh) * 3 + kw], v);
    }
  }
}

extern "C" void bdbnn_conv_wgrad(const void* g, const uint32_t* xp,
                                 float* dw, int N, int H, int W, int C,
                                 int K, int CW, hipStream_t stream) {
  WgradParams p{N, H, W, C, K, CW, (int64_t)N * H * W, 0};
  // pick a split count that fills the chip (>= 2048 blocks total)
  int grid_k = K / WG_TK, grid_c = C / WG_TC, grid_t = 9;
  int base = grid_k * grid_c * grid_t;
  int split = 1;
  while (base * split < 2048 && (p.M / (split * 2)) >= 64) split *= 2;
  int64_t chunk = (p.M + split - 1) / split;
  chunk = ((chunk + WG_BM - 1) / WG_BM) * WG_BM;   // 16-aligned chunks
  p.chunk = chunk;
  split = int((p.M + chunk - 1) / chunk);
  dim3 grid(base * split);
  conv_wgrad_kernel<<<grid, 64, 0, stream>>>(
      (const bf16_t*)g, xp, dw, p, grid_k, grid_c, grid_t);
}
