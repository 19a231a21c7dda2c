// MFMA bf16 backward for the 3x3/stride-1/pad-1 binary convolutions —
// the training hot path's dgrad (v2), replacing both MIOpen igemm and
// the separate decode/mask passes (SURVEY.md hard-part #1).
//
//   dx[n,y,x,c] = mask(n,y,x,c) * sum_{dy,dx,k} g[n, y+dy-1, x+dx-1, k]
//                                  * Wd[t=(dy,dx)][c][k]
//
// where Wd[t][c][k] = alpha_k * sign(w[k,c,2-dy,2-dx]) is the mirrored,
// transposed, pre-decoded weight (tiny tensor, produced once per
// backward by dgrad_weight_decode below), and mask is the clip-STE
// bitplane packed by the forward (csrc/pack.hip) — applied in the
// epilogue, so the old mask_mul_packed pass over the full dx tensor
// disappears.
//
// Why v2 is fast where v1 (csrc/conv_dgrad.hip) was 4-6x behind MIOpen:
//   * HALO staging: the block's g tile is staged ONCE per 64-channel
//     chunk as a 2D zero-padded halo band, and all 9 taps read shifted
//     windows of it from LDS — v1 re-staged the same g rows 9x.
//   * 576-deep K per halo stage (9 taps x 64 ch) with ONE barrier per
//     tap (write-late double buffering, the XNOR-forward structure) —
//     v1 ran 2 barriers per 16-deep K step.
//   * XOR-swizzled LDS (guide T2) — conflict-free ds_read_b128.
//   * raw __builtin_amdgcn_mfma_f32_32x32x16_bf16 with 256x64 block
//     tile, 8 waves as 4(M)x2(N).
//
// Geometry: the 256 M-rows of a block are image-row BANDS in padded
// x-coordinates (Wp = W rounded up to 8/16/32/64); pad taps read
// explicit zeros in the halo, dummy columns (x >= W) produce rows that
// are simply not stored.  One template instantiation per Wp class.
//
// Constraints (python falls back to the aten/MIOpen path otherwise):
//   KH = KW = 3, stride = 1, pad = 1, C % 64 == 0, K % 64 == 0,
//   W <= 64 (and H <= 8 when W <= 8), g bf16 NHWC, dx bf16 NHWC.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

#define DG2_BM 256       // M rows (padded pixels) per block
#define DG2_BN 64        // dx channels per block
#define DG2_BK 64        // g channels per K-chunk

struct Dgrad2Params {
  int N, H, W, C, K;
  int bands_per_image;   // ceil(H / RB)
  int total_slots;       // N * bands_per_image
  int CW;                // C/32 (mask words per pixel)
};

// One template instantiation per padded-width class.
//   WP: padded row width; RB: image rows per band; GB: bands per block.
//   GB*(RB)*WP == 256.
//
// Each block is quasi-persistent: it loops over a CONTIGUOUS range of
// m-tiles (and all g-channel chunks of each), double-buffering the
// halo, the B tile AND the per-tile tables across the whole
// (tile, k-chunk) stream — the per-tile startup serialization of the
// naive one-tile-per-block version (measured 8x off the MFMA floor at
// 1 block/CU occupancy) disappears, and consecutive tiles' overlapping
// halo rows stay L2/L1-warm on the same CU.
// ACC: fuse the residual-skip gradient (same-shape bf16 NHWC tensor)
// into the epilogue store — dx = mask*acc + skip_grad — replacing the
// separate autograd accumulation pass (one full read+write of dx).
template <int WP, int RB, int GB, bool ACC>
__global__ __launch_bounds__(512, 2) void conv_dgrad2_kernel(
    const __bf16* __restrict__ g, const __bf16* __restrict__ wd,
    const uint32_t* __restrict__ mp, __bf16* __restrict__ dx,
    const __bf16* __restrict__ accp,
    Dgrad2Params p, int grid_m, int nb_m, int tiles_per_block) {
  constexpr int WH = WP + 2;                  // halo row width
  constexpr int NHE = GB * (RB + 2) * WH;     // halo entries (128 B each)
  constexpr int PIECES = NHE * 8;             // 16-B staging pieces
  constexpr int PPT = (PIECES + 511) / 512;   // pieces per thread

  // XCD-aware bijective remap (8 XCDs with private L2s)
  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, idx = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int rb_m = wg % nb_m;               // which m-tile RANGE
  const int c_blk = wg / nb_m;
  const int c0 = c_blk * DG2_BN;
  const int tile0 = rb_m * tiles_per_block;
  const int tile_end = bd_min(tile0 + tiles_per_block, grid_m);
  const int my_tiles = tile_end - tile0;
  if (my_tiles <= 0) return;
  const int tid = threadIdx.x;

  __shared__ __align__(16) __bf16 halo[2][NHE * DG2_BK];
  __shared__ __align__(16) __bf16 blds[2][DG2_BN * DG2_BK];
  __shared__ int he_base[2][NHE];   // g pixel index of halo entry, or -1
  __shared__ int tab_he[2][DG2_BM];   // he00 of m-row (tap dy=dx=0)
  __shared__ int tab_out[2][DG2_BM];  // output pixel index, or -1
  __shared__ unsigned char tab_x7[2][DG2_BM];  // x & 7 (swizzle key base)

#define BUILD_TABLES(tile, tb)                                            \
  {                                                                       \
    const int slot0 = (tile)*GB;                                          \
    for (int he = tid; he < NHE; he += 512) {                             \
      int band = he / ((RB + 2) * WH);                                    \
      int rem = he - band * ((RB + 2) * WH);                              \
      int hr = rem / WH, hx = rem - hr * WH;                              \
      int slot = slot0 + band;                                            \
      int base = -1;                                                      \
      if (slot < p.total_slots) {                                         \
        int n = slot / p.bands_per_image;                                 \
        int r0 = (slot - n * p.bands_per_image) * RB;                     \
        int y = r0 + hr - 1, x = hx - 1;                                  \
        if (y >= 0 && y < p.H && x >= 0 && x < p.W)                       \
          base = (n * p.H + y) * p.W + x;                                 \
      }                                                                   \
      he_base[tb][he] = base;                                             \
    }                                                                     \
    for (int m = tid; m < DG2_BM; m += 512) {                             \
      int band = m / (RB * WP);                                           \
      int rem = m - band * (RB * WP);                                     \
      int rloc = rem / WP, x = rem - rloc * WP;                           \
      int slot = slot0 + band;                                            \
      tab_he[tb][m] = band * (RB + 2) * WH + rloc * WH + x;               \
      tab_x7[tb][m] = (unsigned char)(x & 7);                             \
      int out = -1;                                                       \
      if (slot < p.total_slots) {                                         \
        int n = slot / p.bands_per_image;                                 \
        int y = (slot - n * p.bands_per_image) * RB + rloc;               \
        if (y < p.H && x < p.W) out = (n * p.H + y) * p.W + x;            \
      }                                                                   \
      tab_out[tb][m] = out;                                               \
    }                                                                     \
  }

  // ---- wave decomposition: 8 waves as 4(M) x 2(N) ----
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int wm0 = (wid >> 1) * 64;       // wave's M offset (64 rows)
  const int wc = (wid & 1) * 32;         // wave's c offset (32 cols)
  const int lrow = lane & 31;
  const int lk8 = lane >> 5;             // which 8-element k-half

  const int mA0 = wm0 + lrow, mA1 = wm0 + 32 + lrow;
  const int cB = wc + lrow;

  // per-lane A addressing (reloaded at each tile switch)
  int heA0, heA1, x7A0, x7A1;
#define LOAD_LANE_TABS(tb)                                                \
  {                                                                       \
    heA0 = tab_he[tb][mA0];                                               \
    heA1 = tab_he[tb][mA1];                                               \
    x7A0 = tab_x7[tb][mA0];                                               \
    x7A1 = tab_x7[tb][mA1];                                               \
  }

  f32x16 acc0, acc1;
#pragma unroll
  for (int i = 0; i < 16; ++i) { acc0[i] = 0.f; acc1[i] = 0.f; }

  uint4 hreg[PPT];
  uint4 breg;

#define HALO_LOAD(tb, k0)                                                 \
  {                                                                       \
    _Pragma("unroll")                                                     \
    for (int it = 0; it < PPT; ++it) {                                    \
      int i = tid + it * 512;                                             \
      uint4 v{0, 0, 0, 0};                                                \
      if (i < PIECES) {                                                   \
        int he = i >> 3, k8 = i & 7;                                      \
        int base = he_base[tb][he];                                       \
        if (base >= 0)                                                    \
          v = *(const uint4*)(g + (int64_t)base * p.K + (k0) + k8 * 8);   \
      }                                                                   \
      hreg[it] = v;                                                       \
    }                                                                     \
  }

#define HALO_WRITE(buf)                                                   \
  {                                                                       \
    _Pragma("unroll")                                                     \
    for (int it = 0; it < PPT; ++it) {                                    \
      int i = tid + it * 512;                                             \
      if (i < PIECES) {                                                   \
        int he = i >> 3, k8 = i & 7;                                      \
        int rem = he - (he / WH) * WH;  /* hx */                          \
        *(uint4*)&halo[buf][he * DG2_BK + ((k8 ^ (rem & 7)) << 3)] =      \
            hreg[it];                                                     \
      }                                                                   \
    }                                                                     \
  }

  const int b_c = tid >> 3;        // 0..63
  const int b_k8 = tid & 7;        // 0..7
#define B_LOAD(t, k0)                                                     \
  breg = *(const uint4*)(wd + (((t) * p.C + c0 + b_c) * (int64_t)p.K) +   \
                         (k0) + b_k8 * 8);
#define B_WRITE(buf)                                                      \
  *(uint4*)&blds[buf][b_c * DG2_BK + ((b_k8 ^ (b_c & 7)) << 3)] = breg;

  // epilogue: clip-STE mask from the packed bitplane, bf16 store.
  // C/D layout: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5).
  const int ccol = c0 + wc + lrow;
  const int cw_word = ccol >> 5;
  const int cbit = ccol & 31;
#define EPILOGUE(tb)                                                      \
  {                                                                       \
    _Pragma("unroll")                                                     \
    for (int half = 0; half < 2; ++half) {                                \
      const f32x16& acc = half ? acc1 : acc0;                             \
      const int mbase = wm0 + half * 32 + 4 * lk8;                        \
      int pixr[16];                                                       \
      uint32_t mpw[16];                                                   \
      uint16_t av[16];                                                    \
      _Pragma("unroll")                                                   \
      for (int reg = 0; reg < 16; ++reg)                                  \
        pixr[reg] = tab_out[tb][mbase + (reg & 3) + 8 * (reg >> 2)];      \
      /* batch-issue the mask/skip-grad loads (one waitcnt for all,   */ \
      /* instead of a load-use pair per element that serializes the   */ \
      /* whole epilogue behind HBM latency once per tile).  pix<0     */ \
      /* lanes read a safe dummy address and are masked below.        */ \
      _Pragma("unroll")                                                   \
      for (int reg = 0; reg < 16; ++reg) {                                \
        int64_t pz = pixr[reg] < 0 ? 0 : (int64_t)pixr[reg];              \
        mpw[reg] = mp[pz * p.CW + cw_word];                               \
        if (ACC) av[reg] = *(const uint16_t*)(accp + pz * p.C + ccol);    \
      }                                                                   \
      _Pragma("unroll")                                                   \
      for (int reg = 0; reg < 16; ++reg) {                                \
        int pix = pixr[reg];                                              \
        if (pix < 0) continue;                                            \
        float v = (mpw[reg] >> cbit) & 1 ? acc[reg] : 0.f;                \
        if (ACC) v += bf16_to_f32(av[reg]);                               \
        uint16_t h = f32_to_bf16(v);                                      \
        *(uint16_t*)(dx + (int64_t)pix * p.C + ccol) = h;                 \
      }                                                                   \
    }                                                                     \
  }

  // ---- (tile, k-chunk) stream with cross-tile prefetch ----
  const int n_k0 = p.K / DG2_BK;
  const int S = my_tiles * n_k0;

  BUILD_TABLES(tile0, 0);
  __syncthreads();
  LOAD_LANE_TABS(0);
  HALO_LOAD(0, 0);
  HALO_WRITE(0);
  B_LOAD(0, 0);
  B_WRITE(0);
  __syncthreads();

  int hb = 0, bb = 0, tb = 0;
  for (int s = 0; s < S; ++s) {
    const int ki = s % n_k0;
    const int k0 = ki * DG2_BK;
    const bool more = s + 1 < S;
    const bool tile_switch = (ki == n_k0 - 1);  // next s starts a new tile
    const int nki = tile_switch ? 0 : ki + 1;
#pragma unroll
    for (int t = 0; t < 9; ++t) {
      const int dy = t / 3, dxt = t - dy * 3;
      if (t < 8) {
        B_LOAD(t + 1, k0);
      } else if (more) {
        B_LOAD(0, nki * DG2_BK);
      }
      if (t == 0 && more && tile_switch) {
        // build the NEXT tile's tables into the other buffer; the tap-0
        // barrier below publishes them before HALO_LOAD reads them
        BUILD_TABLES(tile0 + (s + 1) / n_k0, tb ^ 1);
      }
      if (t == 1 && more) {
        HALO_LOAD(tile_switch ? tb ^ 1 : tb, nki * DG2_BK);
      }
      const int heT0 = (heA0 + dy * WH + dxt) * DG2_BK;
      const int heT1 = (heA1 + dy * WH + dxt) * DG2_BK;
      const int keyA0 = ((x7A0 + dxt) & 7) ^ lk8;
      const int keyA1 = ((x7A1 + dxt) & 7) ^ lk8;
      const int keyB = (cB & 7) ^ lk8;
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {
        bf16x8 a0 = *(const bf16x8*)&halo[hb][heT0 + (((kk << 1) ^ keyA0) << 3)];
        bf16x8 a1 = *(const bf16x8*)&halo[hb][heT1 + (((kk << 1) ^ keyA1) << 3)];
        bf16x8 b = *(const bf16x8*)&blds[bb][cB * DG2_BK + (((kk << 1) ^ keyB) << 3)];
        acc0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b, acc0, 0, 0, 0);
        acc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b, acc1, 0, 0, 0);
      }
      if (t == 8 && more) HALO_WRITE(hb ^ 1);
      if (t < 8 || more) {
        B_WRITE(bb ^ 1);
        __syncthreads();
        bb ^= 1;
      }
    }
    hb ^= 1;
    if (tile_switch) {
      EPILOGUE(tb);
#pragma unroll
      for (int i = 0; i < 16; ++i) { acc0[i] = 0.f; acc1[i] = 0.f; }
      tb ^= 1;
      LOAD_LANE_TABS(tb);
    }
  }
#undef BUILD_TABLES
#undef LOAD_LANE_TABS
#undef HALO_LOAD
#undef HALO_WRITE
#undef B_LOAD
#undef B_WRITE
#undef EPILOGUE
}

// ---------------- host launcher ----------------

extern "C" int bdbnn_conv_dgrad2(const void* g, const void* wd,
                                 const uint32_t* mp, void* dx,
                                 const void* accp, int N, int H,
                                 int W, int C, int K, hipStream_t stream) {
  if (C % 64 || K % 64 || W > 64) return -1;
  Dgrad2Params p;
  p.N = N; p.H = H; p.W = W; p.C = C; p.K = K;
  p.CW = C / 32;
  int grid_m;
#define LAUNCH(WPV, RBV, GBV)                                             \
  {                                                                       \
    p.bands_per_image = (H + (RBV)-1) / (RBV);                            \
    p.total_slots = N * p.bands_per_image;                                \
    grid_m = (p.total_slots + (GBV)-1) / (GBV);                           \
    int n_ctile = C / DG2_BN;                                             \
    /* 1 block/CU resident (LDS-bound): ~2 ranges per CU for tail      */ \
    /* balance, contiguous m-tiles per block for halo L2 reuse         */ \
    int nb_m = bd_min(grid_m, bd_min(512, 768 / n_ctile));                \
    int tpb = (grid_m + nb_m - 1) / nb_m;                                 \
    nb_m = (grid_m + tpb - 1) / tpb;                                      \
    dim3 grid(nb_m * n_ctile);                                            \
    if (accp)                                                             \
      conv_dgrad2_kernel<WPV, RBV, GBV, true><<<grid, 512, 0, stream>>>(  \
          (const __bf16*)g, (const __bf16*)wd, mp, (__bf16*)dx,           \
          (const __bf16*)accp, p, grid_m, nb_m, tpb);                     \
    else                                                                  \
      conv_dgrad2_kernel<WPV, RBV, GBV, false><<<grid, 512, 0, stream>>>( \
          (const __bf16*)g, (const __bf16*)wd, mp, (__bf16*)dx, nullptr,  \
          p, grid_m, nb_m, tpb);                                          \
    return 0;                                                             \
  }
  if (W <= 8) {
    if (H > 8) return -1;
    LAUNCH(8, 8, 4);
  } else if (W <= 16) {
    LAUNCH(16, 16, 1);
  } else if (W <= 32) {
    LAUNCH(32, 8, 1);
  } else {
    LAUNCH(64, 4, 1);
  }
#undef LAUNCH
}

// ---------------- mirrored/transposed weight decode ----------------
// wp: uint32 [K][9][CW] inverted bits (csrc/pack.hip: bit 1 <=> w < 0),
// alpha: fp32 [K]  ->  wd bf16 [9][C][K], wd[t][c][k] = +-alpha_k with
// the tap MIRRORED (t reads w[k,c,2-dy,2-dx]).
__global__ void dgrad_wdec_kernel(const uint32_t* __restrict__ wp,
                                  const float* __restrict__ alpha,
                                  __bf16* __restrict__ wd, int C, int K,
                                  int CW) {
  // one thread per 8 outputs along k at fixed (t, c)
  int64_t total = (int64_t)9 * C * (K / 8);
  GRID_STRIDE(i, total) {
    int k8 = int(i % (K / 8));
    int64_t rem = i / (K / 8);
    int c = int(rem % C);
    int t = int(rem / C);
    int tm = 8 - t;                       // mirrored tap
    int cw = c >> 5, cb = c & 31;
    __bf16 v[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int k = k8 * 8 + j;
      uint32_t bits = wp[((int64_t)k * 9 + tm) * CW + cw];
      float al = alpha[k];
      v[j] = (__bf16)((bits >> cb) & 1 ? -al : al);
    }
    *(uint4*)&wd[((int64_t)t * C + c) * K + k8 * 8] = *(uint4*)v;
  }
}

extern "C" void bdbnn_dgrad_wdec(const uint32_t* wp, const float* alpha,
                                 void* wd, int C, int K,
                                 hipStream_t stream) {
  int64_t total = (int64_t)9 * C * (K / 8);
  int blocks = (int)bd_min<int64_t>((total + 255) / 256, 4096);
  dgrad_wdec_kernel<<<blocks, 256, 0, stream>>>(wp, alpha, (__bf16*)wd, C,
                                                K, C / 32);
}

// ======================= wgrad v2 =======================
//
//   dwT[t][c][k] = sum_m xb[pix(m) + Delta_t][c] * g[pix(m)][k]
//
// computed TRANSPOSED (D[c][k] per tap) so both MFMA operands read
// contiguous along the reduction (m) axis:
//   A[c][m] = +-1 decoded from the PADDED C-PLANE bitplanes (xcp, built
//             by repack_cplane below) via a 256-entry byte->8xbf16 LDS
//             LUT, into three kw-SHIFTED copies (X0/X1/X2) so every tap
//             read is aligned;  pad taps are explicit zeros.
//   B[m][k] = g staged TRANSPOSED [k][m] in LDS.
// All 9 taps accumulate in one block pass over the chunk (g and xb are
// read from HBM exactly once; xb moves as BITS - 32x less traffic than
// the decoded tensor the MIOpen path reads).  fp32 atomicAdd combine
// into dwT[9][C][K]; wgrad_finish transposes to [K][C][3][3] and
// applies the |w|<=1 STE mask in the same pass.
//
// A chunk is GB image-BANDS of RB rows each (GB*RB*WP == 128), so small
// images (7x7 at WP=8) fill a whole band instead of 49/128 of a flat
// chunk — each band has its own 2-row halo in X.
//
// Constraints: 3x3/s1/p1, C % 64 == 0, K % 64 == 0, W <= 64, g bf16
// channels_last.

#define WG2_CHUNK 128       // m entries per chunk
#define WG2_BC 64           // c tile
#define WG2_BK 64           // k tile

struct Wgrad2Params {
  int N, H, W, C, K;
  int bands_per_image;      // ceil(H / RB)
  int total_slots;          // N * bands_per_image
  int total_chunks;         // ceil(total_slots / GB)
  int split;                // chunk stride (== blocks per (c,k) tile)
  int nh_rows;              // N * H  (xcp row count)
};

// Block geometry: each (c,k) tile is m-split over `split` blocks; a
// block owns a CONTIGUOUS chunk range, so (for GB==1) consecutive
// chunks are consecutive row-bands of one image and the X decode only
// produces the RB NEW rows per chunk — the 2 halo rows ride a row RING
// (plane = (y+1) mod XROWS) decoded by the previous chunk.  Partials go
// to a per-block slab dwT[split][9][C][K] (plain coalesced stores);
// wgrad_finish sums the slabs — no end-of-kernel atomic storm on the
// small dwT (C=K=64: 256 blocks x 37k words of atomicAdd measured as a
// multi-10us tail).
template <int WP, int RB, int GB>
__global__ __launch_bounds__(512, 2) void conv_wgrad2_kernel(
    const __bf16* __restrict__ g, const uint64_t* __restrict__ xcp,
    float* __restrict__ dwT, Wgrad2Params p, int grid_ck) {
  constexpr int XROWS = RB + 2;           // ring planes (halo rows) per band
  constexpr int XSTRIDE = GB * XROWS * WP + 8;  // +16 B pad: conflict-free
  constexpr int GSTRIDE = WG2_CHUNK + 8;  // gT row stride (elements)
  constexpr int LGWP = WP == 8 ? 3 : WP == 16 ? 4 : WP == 32 ? 5 : 6;
  constexpr bool RING = (GB == 1);        // GB>1: bands are whole images

  const int tid = threadIdx.x;
  int wg = blockIdx.x;
  const int ck = wg % grid_ck;            // (c,k) tile
  const int split_id = wg / grid_ck;
  const int c_blk = ck % (p.C / WG2_BC);
  const int k_blk = ck / (p.C / WG2_BC);
  const int c0 = c_blk * WG2_BC, k0 = k_blk * WG2_BK;

  // X: [3 copies][64 c] rows of GB*XROWS*WP (+pad) elements
  __shared__ __align__(16) __bf16 X[3 * WG2_BC * XSTRIDE];
  __shared__ __align__(16) __bf16 gT[2][WG2_BK * GSTRIDE];
  __shared__ __align__(16) __bf16 lut[256][8];
  __shared__ float red[4][32][32];        // m-split combine scratch
  // raw xcp row bits: the global u64 loads prefetch into registers
  // during the previous chunk's MFMA phase and land here at the chunk
  // boundary, so X_DECODE reads them at LDS latency
  constexpr int NBITS = WG2_BC * GB * XROWS;
  __shared__ uint64_t xbits[NBITS];
  __shared__ unsigned char xvalid[NBITS];  // 0 = pad row (decode to 0)

  // ---- byte -> 8 x (+-1) LUT (bit 1 <=> x >= 0 <=> +1) ----
  for (int b = tid; b < 256; b += 512) {
#pragma unroll
    for (int j = 0; j < 8; ++j)
      lut[b][j] = (__bf16)(((b >> j) & 1) ? 1.f : -1.f);
  }

  // ---- wave decomposition: quad (c-half, k-half) x m-split ----
  const int wid = tid >> 6, lane = tid & 63;
  const int qc = (wid >> 1) & 1, qk = wid & 1, ms_grp = wid >> 2;
  const int lrow = lane & 31, lhalf = lane >> 5;

  f32x16 acc[9];
#pragma unroll
  for (int t = 0; t < 9; ++t)
#pragma unroll
    for (int i = 0; i < 16; ++i) acc[t][i] = 0.f;

  // contiguous chunk range of this split block
  const int cpb = (p.total_chunks + p.split - 1) / p.split;
  const int ch_lo = split_id * cpb;
  const int ch_hi = bd_min(ch_lo + cpb, p.total_chunks);
  const int n_chunks = ch_hi - ch_lo;   // may be 0: slab still written
  int ch = ch_lo;

  // staging assignments (fixed per thread)
  //   gT: thread loads 16 k-elements of ONE pixel (32 B contiguous; a
  //   4-lane group covers one pixel's 64-ch row = 128 B coalesced) and
  //   scatter-writes them transposed.  (A 2-pixel/8-ch variant halved
  //   the LDS stores but scattered the GLOBAL loads to 64 cache lines
  //   per wave — measured 30-60% slower on the WP<=16 classes.)
  const int sg_m = tid >> 2;              // 0..127
  const int sg_k16 = (tid & 3) * 16;      // k offset of its 16 elements
  uint4 greg[2];

// (uses the CHUNK_INFO of the same chunk — call order guarantees it)
#define G_LOAD(chunk)                                                     \
  {                                                                       \
    int band = sg_m / (RB * WP);                                          \
    int rem = sg_m - band * (RB * WP);                                    \
    int x = rem & (WP - 1);                                               \
    uint4 z{0, 0, 0, 0};                                                  \
    greg[0] = z; greg[1] = z;                                             \
    if (u_vs[band] && x < p.W) {                                          \
      int y = u_y0[band] + (rem >> LGWP);                                 \
      if (y < p.H) {                                                      \
        const __bf16* src =                                               \
            g + ((int64_t)(u_n[band] * p.H + y) * p.W + x) * p.K + k0 +   \
            sg_k16;                                                       \
        greg[0] = *(const uint4*)src;                                     \
        greg[1] = *(const uint4*)(src + 8);                               \
      }                                                                   \
    }                                                                     \
  }

#define G_WRITE(buf)                                                      \
  {                                                                       \
    _Pragma("unroll")                                                     \
    for (int j = 0; j < 16; ++j) {                                        \
      __bf16 v = ((const __bf16*)greg)[j];                                \
      gT[buf][(sg_k16 + j) * GSTRIDE + sg_m] = v;                         \
    }                                                                     \
  }

  // ---- raw-bits prefetch ----
  // rows are addressed by WINDOW position w in [r0, r0+NR): global row
  // y = y0 + w - 1, ring plane = RING ? (y0+w) % XROWS : w.  A fresh
  // chunk decodes the whole window (r0=0, NR=XROWS); a continuation
  // chunk (same image, next band) only the RB new bottom rows (r0=2).
  constexpr int BPT = (NBITS + 511) / 512;
  uint64_t bits_reg[BPT];
  unsigned char valid_reg[BPT];
  int nbits_cur = 0;
  // block-uniform per-band facts of a chunk (one divide per band, not
  // one per decode target)
  int u_n[GB], u_y0[GB], u_pb[GB];
  bool u_vs[GB];
#define CHUNK_INFO(chunk)                                                 \
  {                                                                       \
    _Pragma("unroll")                                                     \
    for (int b = 0; b < GB; ++b) {                                        \
      int slot = (chunk)*GB + b;                                          \
      u_vs[b] = slot < p.total_slots;                                     \
      int n = slot / p.bands_per_image;                                   \
      u_n[b] = n;                                                         \
      u_y0[b] = (slot - n * p.bands_per_image) * RB;                      \
      u_pb[b] = RING ? u_y0[b] % XROWS : 0;                               \
    }                                                                     \
  }
#define BITS_LOAD(chunk, r0, NR)                                          \
  {                                                                       \
    nbits_cur = WG2_BC * GB * (NR);                                       \
    _Pragma("unroll")                                                     \
    for (int it = 0; it < BPT; ++it) {                                    \
      int i = tid + it * 512;                                             \
      uint64_t b = 0;                                                     \
      unsigned char v = 0;                                                \
      if (i < nbits_cur) {                                                \
        int w = (r0) + i % (NR);                                          \
        int rem = i / (NR);                                               \
        int band = rem % GB;                                              \
        int c = rem / GB;                                                 \
        if (u_vs[band]) {                                                 \
          int y = u_y0[band] + w - 1;                                     \
          if (y >= 0 && y < p.H) {                                        \
            b = xcp[(int64_t)(c0 + c) * p.nh_rows + u_n[band] * p.H + y]; \
            v = 1;                                                        \
          }                                                               \
        }                                                                 \
      }                                                                   \
      bits_reg[it] = b;                                                   \
      valid_reg[it] = v;                                                  \
    }                                                                     \
  }
#define BITS_WRITE(chunk, r0, NR)                                         \
  {                                                                       \
    _Pragma("unroll")                                                     \
    for (int it = 0; it < BPT; ++it) {                                    \
      int i = tid + it * 512;                                             \
      if (i < nbits_cur) {                                                \
        int w = (r0) + i % (NR);                                          \
        int rem = i / (NR);                                               \
        int band = rem % GB;                                              \
        int c = rem / GB;                                                 \
        int plane = RING ? u_pb[band] + w : w;                            \
        if (RING && plane >= XROWS) plane -= XROWS;                       \
        int bi = (c * GB + band) * XROWS + plane;                         \
        xbits[bi] = bits_reg[it];                                         \
        xvalid[bi] = valid_reg[it];                                       \
      }                                                                   \
    }                                                                     \
  }

  // ---- X decode for one chunk from the LDS-resident bits; the x-pad
  // edge zeros (X0[x=0], X2[x=W-1]) are folded into the target write ----
  const int eb8 = (p.W - 1) >> 3, eel = (p.W - 1) & 7;
#define X_DECODE(chunk, r0, NR)                                           \
  {                                                                       \
    const int xtgt = 3 * WG2_BC * GB * (NR) * (WP / 8);                   \
    _Pragma("unroll 2")                                                   \
    for (int it = 0; it < (3 * WG2_BC * GB * XROWS * (WP / 8) + 511) /    \
                              512; ++it) {                                \
      int i = tid + it * 512;                                             \
      if (i < xtgt) {                                                     \
        int xb8 = i % (WP / 8);                                           \
        int rem = i / (WP / 8);                                           \
        int w = (r0) + rem % (NR);                                        \
        rem /= (NR);                                                      \
        int band = rem % GB;                                              \
        rem /= GB;                                                        \
        int c = rem % WG2_BC;                                             \
        int dxs = rem / WG2_BC;                                           \
        int plane = RING ? u_pb[band] + w : w;                            \
        if (RING && plane >= XROWS) plane -= XROWS;                       \
        int bi = (c * GB + band) * XROWS + plane;                         \
        uint64_t bits = xbits[bi];                                        \
        /* kw shift: X_dxs[x] = xb[x + dxs - 1] */                        \
        uint64_t sh = dxs == 0 ? (bits << 1) : (bits >> (dxs - 1));       \
        unsigned byte = (unsigned)(sh >> (8 * xb8)) & 0xffu;              \
        uint4 vv{0, 0, 0, 0};                                             \
        if (xvalid[bi]) {                                                 \
          vv = *(const uint4*)&lut[byte][0];                              \
          if (dxs == 0 && xb8 == 0) ((uint16_t*)&vv)[0] = 0;              \
          if (dxs == 2 && xb8 == eb8) ((uint16_t*)&vv)[eel] = 0;          \
        }                                                                 \
        *(uint4*)&X[(dxs * WG2_BC + c) * XSTRIDE +                        \
                    (band * XROWS + plane) * WP + xb8 * 8] = vv;          \
      }                                                                   \
    }                                                                     \
  }

  // continuation = same image's next band (GB==1 ranges walk bands in
  // order; a new image or the range start re-decodes the full window)
#define IS_CONT(chunk) \
  (RING && (chunk) > ch_lo && ((chunk) % p.bands_per_image) != 0)

  if (n_chunks > 0) {
    CHUNK_INFO(ch);
    BITS_LOAD(ch, 0, XROWS);
    G_LOAD(ch);
    BITS_WRITE(ch, 0, XROWS);
    G_WRITE(0);
  }
  int gb = 0;

  // per-lane fragment bases
  // A (X): lane row c = qc*32 + lrow; m-run start offset = lhalf*8
  const int a_c = qc * 32 + lrow;
  // B (gT): lane row k = qk*32 + lrow
  const int b_k = qk * 32 + lrow;

  for (int ci = 0; ci < n_chunks; ++ci) {
    const bool more = ci + 1 < n_chunks;
    const bool cont = IS_CONT(ch);
    const bool cont_n = IS_CONT(ch + 1);
    CHUNK_INFO(ch);
    // y0 of this chunk modulo the ring (0 when !RING: y0 == 0 there)
    const int y0m = RING ? u_pb[0] : 0;
    __syncthreads();                  // xbits + gT visible; X free
    if (cont) X_DECODE(ch, 2, RB)
    else X_DECODE(ch, 0, XROWS)
    if (more) {
      CHUNK_INFO(ch + 1);             // (also serves BITS_WRITE below)
      if (cont_n) BITS_LOAD(ch + 1, 2, RB)
      else BITS_LOAD(ch + 1, 0, XROWS)
      G_LOAD(ch + 1);                 // lands during the MFMA phase
    }
    __syncthreads();                  // X ready for all waves

#pragma unroll
    for (int msl = 0; msl < 4; ++msl) {   // 4 x 16-m steps = its 64-m half
      const int m16 = ms_grp * 64 + msl * 16;
      // B-frag: 8 m at fixed k from gT
      bf16x8 bfrag =
          *(const bf16x8*)&gT[gb][b_k * GSTRIDE + m16 + lhalf * 8];
      // A-frags per tap from X: ring plane (rloc + dy), x-run
      const int mstart = m16 + lhalf * 8;
      const int rl = mstart >> LGWP;
      const int band = rl / RB, rloc = rl - band * RB;
      const int xs = mstart & (WP - 1);
      int pl[3];
#pragma unroll
      for (int dy = 0; dy < 3; ++dy) {
        int v = RING ? y0m + rloc + dy : rloc + dy;
        if (RING && v >= XROWS) v -= XROWS;
        if (RING && v >= XROWS) v -= XROWS;
        pl[dy] = band * XROWS + v;
      }
#pragma unroll
      for (int t = 0; t < 9; ++t) {
        const int dy = t / 3, dxs = t - dy * 3;
        bf16x8 afrag = *(const bf16x8*)&X[(dxs * WG2_BC + a_c) * XSTRIDE +
                                          pl[dy] * WP + xs];
        acc[t] =
            __builtin_amdgcn_mfma_f32_32x32x16_bf16(afrag, bfrag, acc[t],
                                                    0, 0, 0);
      }
    }
    __syncthreads();                  // all reads of X/gT[gb] done
    if (more) {
      if (cont_n) BITS_WRITE(ch + 1, 2, RB)
      else BITS_WRITE(ch + 1, 0, XROWS)
      G_WRITE(gb ^ 1);
    }
    gb ^= 1;
    ++ch;
  }

  // ---- m-split combine + plain coalesced slab store ----
  // C/D layout: col j (k) = lane&31, row (c) = (reg&3)+8*(reg>>2)+4*lhalf
  const int quad = wid & 3;
  float* slab = dwT + (int64_t)split_id * 9 * p.C * p.K;
#pragma unroll
  for (int t = 0; t < 9; ++t) {
    __syncthreads();
    if (ms_grp == 1) {
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        int row = (reg & 3) + 8 * (reg >> 2) + 4 * lhalf;
        red[quad][row][lrow] = acc[t][reg];
      }
    }
    __syncthreads();
    if (ms_grp == 0) {
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        int row = (reg & 3) + 8 * (reg >> 2) + 4 * lhalf;
        float v = acc[t][reg] + red[quad][row][lrow];
        slab[((int64_t)t * p.C + c0 + qc * 32 + row) * p.K + k0 +
             qk * 32 + lrow] = v;
      }
    }
  }
}

// split (= slab count) mirrored for the host-side dwT allocation
extern "C" int bdbnn_wgrad2_nslab(int N, int H, int W, int C, int K) {
  if (C % 64 || K % 64 || W > 64) return -1;
  int grid_ck = (C / WG2_BC) * (K / WG2_BK);
  int split = (256 + grid_ck - 1) / grid_ck;
  int bands, total_chunks;
  if (W <= 8 && H <= 8) {
    bands = 1; total_chunks = (N + 1) / 2;
  } else if (W <= 16) {
    bands = (H + 7) / 8; total_chunks = N * bands;
  } else if (W <= 32) {
    bands = (H + 3) / 4; total_chunks = N * bands;
  } else {
    bands = (H + 1) / 2; total_chunks = N * bands;
  }
  return split < total_chunks ? split : total_chunks;
}

extern "C" int bdbnn_conv_wgrad2(const void* g, const uint64_t* xcp,
                                 float* dwT, int N, int H, int W, int C,
                                 int K, hipStream_t stream) {
  if (C % 64 || K % 64 || W > 64) return -1;
  Wgrad2Params p;
  p.N = N; p.H = H; p.W = W; p.C = C; p.K = K;
  p.nh_rows = N * H;
  int grid_ck = (C / WG2_BC) * (K / WG2_BK);
  // LDS (~110-145 KB/block) admits ONE 512-thread block per CU: round
  // the m-split so the grid is a whole multiple of 256 CUs (384 blocks
  // = 1.5 dispatch rounds would idle half the chip for half the time)
  int split = (256 + grid_ck - 1) / grid_ck;
  p.split = split;
#define WLAUNCH(WPV, RBV, GBV)                                            \
  {                                                                       \
    p.bands_per_image = (H + (RBV)-1) / (RBV);                            \
    p.total_slots = N * p.bands_per_image;                                \
    p.total_chunks = (p.total_slots + (GBV)-1) / (GBV);                   \
    if (p.split > p.total_chunks) p.split = p.total_chunks;               \
    dim3 grid(grid_ck * p.split);                                         \
    conv_wgrad2_kernel<WPV, RBV, GBV><<<grid, 512, 0, stream>>>(          \
        (const __bf16*)g, xcp, dwT, p, grid_ck);                          \
    return 0;                                                             \
  }
  if (W <= 8 && H <= 8) WLAUNCH(8, 8, 2)
  else if (W <= 16) WLAUNCH(16, 8, 1)
  else if (W <= 32) WLAUNCH(32, 4, 1)
  else WLAUNCH(64, 2, 1)
#undef WLAUNCH
}

// ---------------- padded c-plane repack ----------------
// xp [P][CW] (bit c of word = sign of x[p, 32cw+c])  ->  xcp u64 rows
// [C][N*H], row = W bits of one image row zero-padded to 64.
__global__ void repack_cplane_kernel(const uint32_t* __restrict__ xp,
                                     uint64_t* __restrict__ xcp, int NH,
                                     int W, int C, int CW) {
  // one thread per (c-word cw, image row): builds 32 u64 rows bit by bit
  GRID_STRIDE(i, (int64_t)NH * CW) {
    int cw = int(i % CW);
    int64_t row = i / CW;
    const uint32_t* src = xp + (row * W) * CW + cw;
    uint64_t acc[32];
#pragma unroll
    for (int c = 0; c < 32; ++c) acc[c] = 0;
    for (int x = 0; x < W; ++x) {
      uint32_t word = src[(int64_t)x * CW];
#pragma unroll
      for (int c = 0; c < 32; ++c)
        acc[c] |= (uint64_t)((word >> c) & 1) << x;
    }
#pragma unroll
    for (int c = 0; c < 32; ++c) {
      int cc = cw * 32 + c;
      if (cc < C) xcp[(int64_t)cc * NH + row] = acc[c];
    }
  }
}

extern "C" void bdbnn_repack_cplane(const uint32_t* xp, uint64_t* xcp,
                                    int NH, int W, int C, int CW,
                                    hipStream_t stream) {
  int64_t total = (int64_t)NH * CW;
  int blocks = (int)bd_min<int64_t>((total + 255) / 256, 8192);
  repack_cplane_kernel<<<blocks, 256, 0, stream>>>(xp, xcp, NH, W, C, CW);
}

// ---------------- wgrad finish: slab sum + transpose + STE mask ----
// dw[k][c][t] = sum_s dwT[s][t][c][k] * 1(|w[k][c][t]| <= 1)
// (the slabs are conv_wgrad2's per-m-split-block partials; summing them
// here replaces an end-of-kernel atomic storm on the tiny dwT)
__global__ void wgrad_finish_kernel(const float* __restrict__ dwT,
                                    const float* __restrict__ w,
                                    float* __restrict__ dw, int C, int K,
                                    int nslab) {
  int64_t stride = (int64_t)9 * C * K;
  GRID_STRIDE(i, stride) {
    int t = int(i % 9);
    int64_t rem = i / 9;
    int c = int(rem % C);
    int k = int(rem / C);
    float wv = w[i];
    int64_t j = ((int64_t)t * C + c) * K + k;
    float v = 0.f;
    for (int s = 0; s < nslab; ++s) v += dwT[s * stride + j];
    dw[i] = (wv <= 1.f && wv >= -1.f) ? v : 0.f;
  }
}

extern "C" void bdbnn_wgrad_finish(const float* dwT, const float* w,
                                   float* dw, int C, int K, int nslab,
                                   hipStream_t stream) {
  int64_t total = (int64_t)9 * C * K;
  int blocks = (int)bd_min<int64_t>((total + 255) / 256, 8192);
  wgrad_finish_kernel<<<blocks, 256, 0, stream>>>(dwT, w, dw, C, K, nslab);
}
