// SUPERSEDED on the hot path by csrc/conv_bwd.hip (dgrad2): kept as an
// independent rocWMMA implementation — a second oracle the v2 kernel's
// GPU numerics tests cross-check against, and the minimal readable
// form of the algorithm.  Hand-written MFMA bf16 dgrad
// for the 3x3/stride-1/pad-1 binary convs.
//
//   dx[n,iy,ix,c] = sum_{kh,kw,k} g[n, iy+1-kh, ix+1-kw, k] * wb[k,c,kh,kw]
//
// Implicit GEMM: M = N*H*W input pixels, N-dim = C, K-dim = 9*K.  The
// +-alpha weight operand is decoded IN-KERNEL from the packed bits
// (csrc/pack.hip inverted convention) — the B operand never exists in
// memory as a dense tensor.  Fragments via rocWMMA (layout-safe);
// correctness-first single-buffered 2-barrier ladder (pipelining is
// round-2 work, see docs/R2_MFMA_CONV_BACKWARD.md).
//
// Constraints (python falls back to the MIOpen path otherwise):
//   KH = KW = 3, stride = 1, pad = 1, (N*H*W) % 128 == 0,
//   C % 64 == 0, K % 16 == 0, g and dx bf16 channels_last.
#include "common.h"
#include <rocwmma/rocwmma.hpp>

using rocwmma::fragment;
using rocwmma::matrix_a;
using rocwmma::matrix_b;
using rocwmma::accumulator;
using rocwmma::row_major;
using bf16_t = rocwmma::bfloat16_t;  // rocWMMA's bf16 (same 16-bit storage)

#define DG_BM 128
#define DG_BN 64
#define DG_BK 16

struct DgradParams {
  int N, H, W, C, K, CW;
};

__global__ __launch_bounds__(256) void conv_dgrad_kernel(
    const bf16_t* __restrict__ g, const uint32_t* __restrict__ wp,
    const float* __restrict__ alpha, bf16_t* __restrict__ dx,
    DgradParams p, int grid_m) {
  const int m_blk = blockIdx.x % grid_m;
  const int c_blk = blockIdx.x / grid_m;
  const int tid = threadIdx.x;
  const int64_t HW = (int64_t)p.H * p.W;

  __shared__ bf16_t a_lds[DG_BM][DG_BK];
  __shared__ bf16_t b_lds[DG_BK][DG_BN];
  __shared__ float out_st[4][32][32];   // per-wave fp32 bounce for stores
  __shared__ int row_n[DG_BM], row_y[DG_BM], row_x[DG_BM];

  // pixel decomposition of this block's 128 M-rows (once)
  for (int r = tid; r < DG_BM; r += blockDim.x) {
    int64_t m = (int64_t)m_blk * DG_BM + r;
    row_n[r] = int(m / HW);
    int rem = int(m % HW);
    row_y[r] = rem / p.W;
    row_x[r] = rem % p.W;
  }
  __syncthreads();

  // wave -> (m-half, c-half): 4 waves as 2 (M) x 2 (N)
  const int wid = tid / 64;
  const int wm = (wid >> 1) * 64;   // 0 or 64 within the M tile
  const int wc = (wid & 1) * 32;    // 0 or 32 within the C tile

  fragment<accumulator, 32, 32, 16, float> acc0, acc1;
  rocwmma::fill_fragment(acc0, 0.f);
  rocwmma::fill_fragment(acc1, 0.f);

  const int a_row = tid >> 1;          // A staging: 2 threads per row
  const int a_half = (tid & 1) * 8;    // 8 bf16 = 16 B each
  const int b_kk = tid & 15;           // B staging: 4 c-values per thread
  const int b_c0 = (tid >> 4) * 4;

  for (int t = 0; t < 9; ++t) {
    const int kh = t / 3, kw = t % 3;
    for (int k0 = 0; k0 < p.K; k0 += DG_BK) {
      // ---- stage A: g at the mirrored tap (zeros when out of bounds) ----
      {
        int y = row_y[a_row] + 1 - kh;
        int x = row_x[a_row] + 1 - kw;
        if (y >= 0 && y < p.H && x >= 0 && x < p.W) {
          const bf16_t* src =
              g + (((int64_t)row_n[a_row] * p.H + y) * p.W + x) * p.K + k0 +
              a_half;
          *(uint4*)&a_lds[a_row][a_half] = *(const uint4*)src;
        } else {
          uint4 z{0, 0, 0, 0};
          *(uint4*)&a_lds[a_row][a_half] = z;
        }
      }
      // ---- stage B: decode +-alpha from packed bits ----
      {
        int k = k0 + b_kk;
        int c = c_blk * DG_BN + b_c0;
        uint32_t word = wp[((int64_t)k * 9 + t) * p.CW + (c >> 5)];
        float al = alpha[k];
        int sh = c & 31;
        bf16_t v[4];
#pragma unroll
        for (int i = 0; i < 4; ++i)
          v[i] = bf16_t(((word >> (sh + i)) & 1) ? -al : al);
        *(uint2*)&b_lds[b_kk][b_c0] = *(uint2*)v;
      }
      __syncthreads();
      // ---- MFMA ----
      fragment<matrix_a, 32, 32, 16, bf16_t, row_major> fa;
      fragment<matrix_b, 32, 32, 16, bf16_t, row_major> fb;
      rocwmma::load_matrix_sync(fb, &b_lds[0][wc], DG_BN);
      rocwmma::load_matrix_sync(fa, &a_lds[wm][0], DG_BK);
      rocwmma::mma_sync(acc0, fa, fb, acc0);
      rocwmma::load_matrix_sync(fa, &a_lds[wm + 32][0], DG_BK);
      rocwmma::mma_sync(acc1, fa, fb, acc1);
      __syncthreads();
    }
  }

  // ---- store: fp32 fragment -> per-wave LDS bounce -> bf16 global ----
  // (rocWMMA stores the fp32 accumulator; the bf16 convert + coalesced
  // 16-B stores happen from the bounce buffer)
  const int lane = tid & 63;
  bf16_t* base = dx + ((int64_t)m_blk * DG_BM + wm) * p.C +
                         c_blk * DG_BN + wc;
#pragma unroll
  for (int frag = 0; frag < 2; ++frag) {
    rocwmma::store_matrix_sync(&out_st[wid][0][0],
                               frag == 0 ? acc0 : acc1, 32,
                               rocwmma::mem_row_major);
    __builtin_amdgcn_s_barrier();  // wave-local: LDS writes visible (wave64)
    const int r = lane >> 1;
    const int c0 = (lane & 1) * 16;
    bf16_t v[16];
#pragma unroll
    for (int i = 0; i < 16; ++i)
      v[i] = bf16_t(out_st[wid][r][c0 + i]);
    bf16_t* dst = base + ((int64_t)frag * 32 + r) * p.C + c0;
    *(uint4*)dst = *(uint4*)&v[0];
    *(uint4*)(dst + 8) = *(uint4*)&v[8];
  }
}

extern "C" void bdbnn_conv_dgrad(const void* g, const uint32_t* wp,
                                 const float* alpha, void* dx, int N, int H,
                                 int W, int C, int K, int CW,
                                 hipStream_t stream) {
  DgradParams p{N, H, W, C, K, CW};
  int64_t M = (int64_t)N * H * W;
  int grid_m = int(M / DG_BM);
  dim3 grid(grid_m * (C / DG_BN));
  conv_dgrad_kernel<<<grid, 256, 0, stream>>>(
      (const bf16_t*)g, wp, alpha, (bf16_t*)dx, p, grid_m);
}
