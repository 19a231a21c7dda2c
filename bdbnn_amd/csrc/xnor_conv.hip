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
// Structure (guide T14 "issue-early / write-late" + one barrier/chunk):
// 256-thread block computes a 128(spatial) x 64(channel) tile; per
// 8-word chunk each thread issues its next-chunk global loads FIRST,
// computes the current LDS buffers (8x4 register tile, ds_read_b128,
// 1024 binary MACs per ~70 instructions), then writes the prefetched
// words into the alternate LDS buffer — HBM latency hides under the
// popcount work.  Word->(tap, offset) address tables are precomputed
// per block (no integer div/mod in the staging loop).
#include "common.h"
#include <cstdlib>

#define TILE_K 64
#define CHUNK 8
#define W_PER_THREAD 2  // CHUNK*TILE_K / 256
#define MAX_WORDS 160   // KH*KW*CW <= 9*16 (C<=512, 3x3); guarded on host
// TM (tile rows): 128 everywhere.  The TM=256 variant (meant to
// amortize the short 2-3 chunk main loop of C<=128 layers) measured
// WORSE on every shape it applied to — its ~100 extra VGPRs cost more
// occupancy than the amortization saved (profiles/r02_stem_tm_ab.md);
// it stays compiled for BDBNN_XNOR_TM=256 A/B on future shapes.

struct XnorConvParams {
  int N, H, W, C, K, KH, KW, stride, pad, Ho, Wo, CW;
  int WORDS;   // KH*KW*CW
  int T;       // KH*KW
  int base;    // -2*G - C*T
};

// s1/s2 (nullable): per-out-channel sum / sum-of-squares of the STORED
// (dtype-rounded) outputs, accumulated in the epilogue — feeds the fused
// BN directly so BN never re-reads the conv output for its stats pass.
template <typename TO, bool STATS, int TM>
__global__ __launch_bounds__(256) void xnor_conv_kernel(
    const uint32_t* __restrict__ xp, const uint32_t* __restrict__ wp,
    const float* __restrict__ alpha, const float* __restrict__ stab,
    TO* __restrict__ out, float* __restrict__ s1, float* __restrict__ s2,
    XnorConvParams p, int grid_m) {
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
  const int64_t m0 = (int64_t)m_blk * TM;
  const int k0_blk = k_blk * TILE_K;

  constexpr int APT = CHUNK * TM / 256;   // a-words staged per thread
  constexpr int RPT = TM / 16;            // rows per thread tile
  __shared__ uint32_t a_lds[2][CHUNK][TM];
  __shared__ uint32_t w_lds[2][CHUNK][TILE_K];
  __shared__ int row_basecw[TM];      // pixel index of tap (0,0), x CW
  __shared__ unsigned short row_inv[TM];      // invalid-tap bitmask (T<=9)
  __shared__ int off_tab[MAX_WORDS];          // (kh*W+kw)*CW + cw per word
  __shared__ unsigned char tap_tab[MAX_WORDS];
  __shared__ float csum[2][TILE_K];           // per-channel stats partials

  // ---- per-block metadata ----
  for (int r = tid; r < TM; r += blockDim.x) {
    int64_t sp = m0 + r;
    if (sp >= M) { row_basecw[r] = 0; row_inv[r] = 0xffff; continue; }
    int n = int(sp / ((int64_t)p.Ho * p.Wo));
    int rem = int(sp % ((int64_t)p.Ho * p.Wo));
    int oy = rem / p.Wo, ox = rem % p.Wo;
    int iy0 = oy * p.stride - p.pad;
    int ix0 = ox * p.stride - p.pad;
    row_basecw[r] = ((n * p.H + iy0) * p.W + ix0) * p.CW;
    unsigned short inv = 0;
    for (int t = 0; t < p.T; ++t) {
      int kh = t / p.KW, kw = t % p.KW;
      int iy = iy0 + kh, ix = ix0 + kw;
      if (iy < 0 || iy >= p.H || ix < 0 || ix >= p.W) inv |= 1u << t;
    }
    row_inv[r] = inv;
  }
  for (int wdx = tid; wdx < p.WORDS; wdx += blockDim.x) {
    int t = wdx / p.CW, cw = wdx - t * p.CW;
    int kh = t / p.KW, kw = t - kh * p.KW;
    off_tab[wdx] = (kh * p.W + kw) * p.CW + cw;
    tap_tab[wdx] = (unsigned char)t;
  }
  if constexpr (STATS)
    for (int c = tid; c < TILE_K; c += blockDim.x) {
      csum[0][c] = 0.f; csum[1][c] = 0.f;
    }
  __syncthreads();

  // thread's register tile: rows r0..r0+RPT-1, channels kq..kq+3
  const int r0 = (tid / 16) * RPT;
  const int kq = (tid % 16) * 4;
  int acc[RPT][4] = {};

  // staging coordinates (fixed per thread)
  int a_c[APT], a_r[APT];
#pragma unroll
  for (int it = 0; it < APT; ++it) {
    int j = tid + it * 256;
    a_c[it] = j / TM;
    a_r[it] = j % TM;
  }
  int w_c[W_PER_THREAD], w_k[W_PER_THREAD];
#pragma unroll
  for (int it = 0; it < W_PER_THREAD; ++it) {
    int j = tid + it * 256;
    w_c[it] = j / TILE_K;
    w_k[it] = j % TILE_K;
  }

  const int n_chunks = (p.WORDS + CHUNK - 1) / CHUNK;
  uint32_t av[APT], wv[W_PER_THREAD];

#define STAGE_LOAD(w0)                                                    \
  {                                                                       \
    _Pragma("unroll")                                                     \
    for (int it = 0; it < APT; ++it) {                                    \
      int widx = (w0) + a_c[it];                                          \
      int r = a_r[it];                                                    \
      uint32_t v = 0;                                                     \
      if (widx < p.WORDS && !((row_inv[r] >> tap_tab[widx]) & 1))         \
        v = xp[(int64_t)row_basecw[r] + off_tab[widx]];                   \
      av[it] = v;                                                         \
    }                                                                     \
    _Pragma("unroll")                                                     \
    for (int it = 0; it < W_PER_THREAD; ++it) {                           \
      int widx = (w0) + w_c[it];                                          \
      int kg = k0_blk + w_k[it];                                          \
      uint32_t v = 0;                                                     \
      if (widx < p.WORDS && kg < p.K)                                     \
        v = wp[(int64_t)kg * p.WORDS + widx];                             \
      wv[it] = v;                                                         \
    }                                                                     \
  }

#define STAGE_WRITE(buf)                                                  \
  {                                                                       \
    _Pragma("unroll")                                                     \
    for (int it = 0; it < APT; ++it)                                      \
      a_lds[buf][a_c[it]][a_r[it]] = av[it];                              \
    _Pragma("unroll")                                                     \
    for (int it = 0; it < W_PER_THREAD; ++it)                             \
      w_lds[buf][w_c[it]][w_k[it]] = wv[it];                              \
  }

  STAGE_LOAD(0);
  STAGE_WRITE(0);
  __syncthreads();

  for (int ch = 0; ch < n_chunks; ++ch) {
    const int buf = ch & 1;
    const bool more = ch + 1 < n_chunks;
    if (more) STAGE_LOAD((ch + 1) * CHUNK);          // issue early
#pragma unroll
    for (int c = 0; c < CHUNK; ++c) {
      uint32_t a8[RPT], b4[4];
#pragma unroll
      for (int q = 0; q < RPT; q += 4)
        *(uint4*)&a8[q] = *(const uint4*)&a_lds[buf][c][r0 + q];
      *(uint4*)b4 = *(const uint4*)&w_lds[buf][c][kq];
#pragma unroll
      for (int i = 0; i < RPT; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] += __popc(a8[i] ^ b4[j]);
    }
    if (more) STAGE_WRITE(buf ^ 1);                  // write late
    __syncthreads();
  }
#undef STAGE_LOAD
#undef STAGE_WRITE

  // ---- epilogue: scale, pad-correction, store ----
  float al[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    int kg = k0_blk + kq + j;
    al[j] = (kg < p.K) ? alpha[kg] : 0.f;
  }
  float st1[STATS ? 4 : 1] = {}, st2[STATS ? 4 : 1] = {};
#pragma unroll
  for (int i = 0; i < RPT; ++i) {
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
      if constexpr (STATS)
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          float v = bf16_to_f32(vals[j]);   // stats of the ROUNDED value
          st1[j] += v; st2[j] += v * v;
        }
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
      if constexpr (STATS)
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          st1[j] += vals[j]; st2[j] += vals[j] * vals[j];
        }
    }
  }
  if constexpr (STATS) {
    __syncthreads();  // csum init visible
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      atomicAdd(&csum[0][kq + j], st1[j]);
      atomicAdd(&csum[1][kq + j], st2[j]);
    }
    __syncthreads();
    // 32-way sliced accumulators ([32][K], summed by bn_finalize): with
    // one flat [K] buffer ~12k blocks contend on K words and the whole
    // kernel slows ~15% e2e (r1 optimization log); 32 slices cut the
    // per-word contention 32x for 128 KB of scratch.
    const int slice = blockIdx.x & 31;
    for (int c = tid; c < TILE_K; c += blockDim.x) {
      int kg = k0_blk + c;
      if (kg < p.K) {
        if (csum[0][c] != 0.f) atomicAdd(&s1[slice * p.K + kg], csum[0][c]);
        if (csum[1][c] != 0.f) atomicAdd(&s2[slice * p.K + kg], csum[1][c]);
      }
    }
  }
}

extern "C" void bdbnn_xnor_conv_fwd(
    const uint32_t* xp, const uint32_t* wp, const float* alpha,
    const float* stab, void* out, float* s1, float* s2, bool out_bf16,
    int N, int H, int W, int C, int K, int KH, int KW, int stride, int pad,
    int Ho, int Wo, hipStream_t stream) {
  XnorConvParams p;
  p.N = N; p.H = H; p.W = W; p.C = C; p.K = K; p.KH = KH; p.KW = KW;
  p.stride = stride; p.pad = pad; p.Ho = Ho; p.Wo = Wo;
  p.CW = (C + 31) / 32;
  p.T = KH * KW;
  p.WORDS = p.T * p.CW;
  // shapes beyond the table (C > 512 at 3x3) are rejected by the binding
  int G = (32 * p.CW - C) * p.T;
  p.base = -2 * G - C * p.T;
  int64_t M = (int64_t)N * Ho * Wo;
  // TM=256 ("tall tiles amortize the short-K epilogue") was MEASURED
  // WORSE everywhere it applied (C=64: 171 vs 201 TbinMAC/s, C=128:
  // 200 vs 256 — gpurun kernel_bench A/B): the ~100 extra VGPR cost
  // more occupancy than the longer main loop saved.  Default stays 128;
  // the template + env override remain for re-testing on new shapes
  // (BDBNN_XNOR_TM={128,256}).
  static const int tm_env = [] {
    const char* e = getenv("BDBNN_XNOR_TM");
    return e ? atoi(e) : 0;
  }();
  int TM = (tm_env == 256) ? 256 : 128;
  int grid_m = int((M + TM - 1) / TM);
  int grid_k = (K + TILE_K - 1) / TILE_K;
  dim3 grid(grid_m * grid_k);
  if (s1 != nullptr) {
    hipMemsetAsync(s1, 0, sizeof(float) * 32 * K, stream);
    hipMemsetAsync(s2, 0, sizeof(float) * 32 * K, stream);
  }
#define XLAUNCH(TO, ST, TMV)                                              \
  xnor_conv_kernel<TO, ST, TMV><<<grid, 256, 0, stream>>>(                \
      xp, wp, alpha, stab, (TO*)out, s1, s2, p, grid_m)
  if (s1 != nullptr) {
    if (out_bf16) { if (TM == 256) XLAUNCH(uint16_t, true, 256);
                    else XLAUNCH(uint16_t, true, 128); }
    else          { if (TM == 256) XLAUNCH(float, true, 256);
                    else XLAUNCH(float, true, 128); }
  } else {
    if (out_bf16) { if (TM == 256) XLAUNCH(uint16_t, false, 256);
                    else XLAUNCH(uint16_t, false, 128); }
    else          { if (TM == 256) XLAUNCH(float, false, 256);
                    else XLAUNCH(float, false, 128); }
  }
#undef XLAUNCH
}
