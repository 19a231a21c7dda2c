// K7b — dedicated MFMA kernels for the real-valued STEM convolution
// (7x7, stride 2, pad 3, C=3 -> K=64, e.g. 3x224x224 -> 64x112x112).
//
// MIOpen's igemm treats C=3 as the GEMM K-dim and starves the matrix
// cores (measured 2.9 ms fwd + 2.8 ms wrw per b2048 step ~ 15x off the
// MFMA floor).  Here the GEMM K-dim is the whole patch: 7 tap-rows x
// (7 taps x 4-padded channels -> 32 slots) = 224, so every bf16x8
// A-fragment is a 16-byte-aligned run of the LDS-staged input row:
//   slot t = dy*32 + dx*4 + c   (c: 0..2 real, 3 zero-pad; dx*4+c < 28,
//   slots 28..31 zero)
//   A[m][t] = x4[iy(m)+dy][ix(m) + dx][c],  x4 = input padded to 4 ch
//   B[t][k] = w4[k][c][dy][dx]  (pre-padded once per step, tiny)
//
// fwd:  out[m][k] = sum_t A[m][t] * B[t][k]      (one MFMA acc / wave)
// wrw:  dw4[t][k] = sum_m A[m][t] * g[m][k]      (pixel-split + slabs)
//
// Both stage the input in its natural [row][4ch-col] layout (a 2 x 64
// output tile needs 9 x 133 input cols = 4.7 K elements) and build A
// fragments as plain aligned bf16x8 LDS reads — the 4-channel pad is
// what makes (2*mx)*4 + 8j always 16-B aligned.
//
// The input is consumed as an NHWC bf16 tensor padded to 4 channels
// (x4, produced by stem_pack_x4 below: one cheap pass per step);
// weights via stem_pack_w4 / grads back via stem_unpack_dw4.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

#define ST_KD 224          // GEMM K-dim: 7 rows x 32 slots
#define ST_ROWS 9          // input rows per 2-output-row tile
#define ST_ICOLS 133       // input cols per 64-output-col tile (2*64+5)
#define ST_M 128           // output pixels per tile: 2 rows x 64 cols
#define ST_K 64            // output channels

struct StemParams {
  int N, H, W;             // input size (e.g. 224x224)
  int Ho, Wo;              // output size (H/2)
  int bands_per_image;     // ceil(Ho / 2)
  int total_tiles;         // N * bands_per_image * ceil(Wo/64)
  int tiles_x;             // ceil(Wo / 64)
};

// ---------------- forward ----------------
// grid: 512 blocks, each walking a contiguous tile range; weights
// (224x64 bf16 = 28 KB) staged once per block.
__global__ __launch_bounds__(512, 2) void stem_fwd_kernel(
    const __bf16* __restrict__ x4, const __bf16* __restrict__ w4,
    __bf16* __restrict__ out, StemParams p, int n_blocks) {
  const int tid = threadIdx.x;

  // wlds[k][t] rows padded to 232 elems (116 dwords, gcd(116,64)=4:
  // the 32 b128 B-frag lanes land on distinct banks)
  constexpr int WSTRIDE = ST_KD + 8;
  constexpr int XP8 = (ST_ICOLS * 4 + 7) / 8;   // 67 16-B pieces per row
  constexpr int XSTR = XP8 * 8;                 // staged row stride (536)
  __shared__ __align__(16) __bf16 wlds[ST_K * WSTRIDE];      // 29.7 KB
  __shared__ __align__(16) __bf16 xin[2][ST_ROWS * XSTR];

  // stage weights (w4T [k][224] -> wlds [k][232], once per block)
  for (int i = tid * 8; i < ST_K * ST_KD; i += 512 * 8) {
    int k = i / ST_KD, t = i - k * ST_KD;
    *(uint4*)&wlds[k * WSTRIDE + t] = *(const uint4*)&w4[i];
  }

  // wave decomposition: 4(M) x 2(K)
  const int wid = tid >> 6, lane = tid & 63;
  const int wm0 = (wid >> 1) * 32;      // wave's 32 output pixels
  const int wk0 = (wid & 1) * 32;       // wave's 32 output channels
  const int lrow = lane & 31, lk8 = lane >> 5;

  const int cpb = (p.total_tiles + n_blocks - 1) / n_blocks;
  const int t_lo = blockIdx.x * cpb;
  const int t_hi = bd_min(t_lo + cpb, p.total_tiles);

  // input staging: 9 x 67 16-B pieces
  constexpr int PIECES = ST_ROWS * XP8;          // 603
  constexpr int PPT = (PIECES + 511) / 512;      // 2
  uint4 xreg[PPT];

#define ST_LOAD(tile)                                                     \
  {                                                                       \
    int tx = (tile) % p.tiles_x;                                          \
    int rem = (tile) / p.tiles_x;                                         \
    int band = rem % p.bands_per_image;                                   \
    int n = rem / p.bands_per_image;                                      \
    int iy0 = band * 4 - 3;                                               \
    int ix0 = tx * 128 - 3;                                               \
    _Pragma("unroll")                                                     \
    for (int it = 0; it < PPT; ++it) {                                    \
      int i = tid + it * 512;                                             \
      uint4 v{0, 0, 0, 0};                                                \
      if (i < PIECES) {                                                   \
        int r = i / XP8, cpix = (i - r * XP8) * 2;  /* 2 pixels / 16 B */ \
        int iy = iy0 + r;                                                 \
        int ix = ix0 + cpix;                                              \
        if (iy >= 0 && iy < p.H) {                                        \
          const __bf16* src =                                             \
              x4 + (((int64_t)n * p.H + iy) * p.W + ix) * 4;              \
          if (ix >= 0 && ix + 1 < p.W) v = *(const uint4*)src;            \
          else {                                                          \
            if (ix >= 0 && ix < p.W)                                      \
              *(uint2*)&v = *(const uint2*)src;                           \
            if (ix + 1 >= 0 && ix + 1 < p.W)                              \
              *((uint2*)&v + 1) = *(const uint2*)(src + 4);               \
          }                                                               \
        }                                                                 \
      }                                                                   \
      xreg[it] = v;                                                       \
    }                                                                     \
  }
#define ST_WRITE(buf)                                                     \
  {                                                                       \
    _Pragma("unroll")                                                     \
    for (int it = 0; it < PPT; ++it) {                                    \
      int i = tid + it * 512;                                             \
      if (i < PIECES) *(uint4*)&xin[buf][i * 8] = xreg[it];               \
    }                                                                     \
  }

  if (t_lo >= t_hi) return;
  ST_LOAD(t_lo);
  ST_WRITE(0);
  __syncthreads();

  // per-lane A base: output pixel m = wm0 + lrow (row-major in the
  // 2 x 64 tile): my = m >> 6, mx = m & 63; input col = 2*mx
  const int m = wm0 + lrow;
  const int my = m >> 6, mx = m & 63;
  const int a_base = (my * 2) * XSTR + mx * 8;  // elem offset

  int xb = 0;
  for (int t = t_lo; t < t_hi; ++t) {
    const bool more = t + 1 < t_hi;
    if (more) ST_LOAD(t + 1);
    f32x16 acc;
#pragma unroll
    for (int i = 0; i < 16; ++i) acc[i] = 0.f;
#pragma unroll
    for (int kc = 0; kc < 14; ++kc) {   // 14 x 16-deep K chunks
      // k-dim slice [16kc, 16kc+16): rows dy = kc/2, slot halves
      const int dy = kc >> 1;
      const int so = (kc & 1) * 16 + lk8 * 8;     // slot offset 0/8/16/24
      // slots with dx >= 7 or c == 3 read arbitrary staged values, but
      // the matching B weights are ZERO, so no A-side guard is needed
      bf16x8 afrag = *(const bf16x8*)&xin[xb][a_base + dy * XSTR + so];
      bf16x8 bfrag = *(const bf16x8*)&wlds[(wk0 + lrow) * WSTRIDE +
                                           kc * 16 + lk8 * 8];
      acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(afrag, bfrag, acc, 0,
                                                    0, 0);
    }
    // epilogue: C layout col=lane&31 (k), row=(reg&3)+8*(reg>>2)+4*lk8
    {
      int tx = t % p.tiles_x;
      int rem = t / p.tiles_x;
      int band = rem % p.bands_per_image;
      int n = rem / p.bands_per_image;
      int oy0 = band * 2, ox0 = tx * 64;
      const int kcol = wk0 + lrow;
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        int mm = wm0 + (reg & 3) + 8 * (reg >> 2) + 4 * lk8;
        int oy = oy0 + (mm >> 6), ox = ox0 + (mm & 63);
        if (oy < p.Ho && ox < p.Wo) {
          __bf16 v = (__bf16)acc[reg];
          out[(((int64_t)n * p.Ho + oy) * p.Wo + ox) * ST_K + kcol] = v;
        }
      }
    }
    if (more) {
      ST_WRITE(xb ^ 1);   // other buffer: prior reads of it finished
      __syncthreads();    // before THIS tile's MFMA began
      xb ^= 1;
    }
  }
}

// ---------------- operand pack / unpack helpers ----------------

// x (N,3,H,W any layout via strides handled host-side: expects NHWC
// contiguous bf16/f32) -> x4 (N,H,W,4) bf16 with channel 3 zeroed
template <typename TI>
__global__ void stem_pack_x4_kernel(const TI* __restrict__ x,
                                    __bf16* __restrict__ x4,
                                    int64_t npix) {
  GRID_STRIDE(i, npix) {
    const TI* s = x + i * 3;
    __bf16 v[4];
    if constexpr (sizeof(TI) == 2) {
      v[0] = ((const __bf16*)s)[0];
      v[1] = ((const __bf16*)s)[1];
      v[2] = ((const __bf16*)s)[2];
    } else {
      v[0] = (__bf16)((const float*)s)[0];
      v[1] = (__bf16)((const float*)s)[1];
      v[2] = (__bf16)((const float*)s)[2];
    }
    v[3] = (__bf16)0.f;
    *(uint2*)&x4[i * 4] = *(uint2*)v;
  }
}

extern "C" void bdbnn_stem_pack_x4(const void* x, void* x4, int64_t npix,
                                   bool bf16, hipStream_t stream) {
  int blocks = (int)bd_min<int64_t>((npix + 255) / 256, 8192);
  if (bf16)
    stem_pack_x4_kernel<uint16_t><<<blocks, 256, 0, stream>>>(
        (const uint16_t*)x, (__bf16*)x4, npix);
  else
    stem_pack_x4_kernel<float><<<blocks, 256, 0, stream>>>(
        (const float*)x, (__bf16*)x4, npix);
}

// w (64,3,7,7 fp32) -> w4T bf16 [64][224]:  w4T[k][dy*32+dx*4+c]
__global__ void stem_pack_w4_kernel(const float* __restrict__ w,
                                    __bf16* __restrict__ w4T) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;   // k*224 + t
  if (i >= ST_K * ST_KD) return;
  int k = i / ST_KD, t = i - k * ST_KD;
  int dy = t >> 5, rem = t & 31;
  int dx = rem >> 2, c = rem & 3;
  float v = 0.f;
  if (dx < 7 && c < 3) v = w[((k * 3 + c) * 7 + dy) * 7 + dx];
  w4T[i] = (__bf16)v;
}

extern "C" void bdbnn_stem_pack_w4(const float* w, void* w4T,
                                   hipStream_t stream) {
  stem_pack_w4_kernel<<<(ST_K * ST_KD + 255) / 256, 256, 0, stream>>>(
      w, (__bf16*)w4T);
}

// dw4T fp32 [64][224] -> dw (64,3,7,7 fp32)
__global__ void stem_unpack_dw4_kernel(const float* __restrict__ dw4T,
                                       float* __restrict__ dw) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;   // ((k*3+c)*7+dy)*7+dx
  if (i >= ST_K * 3 * 49) return;
  int dx = i % 7;
  int rem = i / 7;
  int dy = rem % 7;
  rem /= 7;
  int c = rem % 3;
  int k = rem / 3;
  dw[i] = dw4T[k * ST_KD + dy * 32 + dx * 4 + c];
}

extern "C" void bdbnn_stem_unpack_dw4(const float* dw4T, float* dw,
                                      hipStream_t stream) {
  stem_unpack_dw4_kernel<<<(ST_K * 3 * 49 + 255) / 256, 256, 0, stream>>>(
      dw4T, dw);
}

extern "C" void bdbnn_stem_fwd(const void* x4, const void* w4T, void* out,
                               int N, int H, int W, hipStream_t stream) {
  StemParams p;
  p.N = N; p.H = H; p.W = W;
  p.Ho = H / 2; p.Wo = W / 2;
  p.bands_per_image = (p.Ho + 1) / 2;
  p.tiles_x = (p.Wo + 63) / 64;
  p.total_tiles = N * p.bands_per_image * p.tiles_x;
  int n_blocks = bd_min(512, p.total_tiles);
  stem_fwd_kernel<<<n_blocks, 512, 0, stream>>>(
      (const __bf16*)x4, (const __bf16*)w4T, (__bf16*)out, p, n_blocks);
}

// ---------------- weight gradient (wrw) ----------------
// dw4T[k][t] = sum_pix patch[pix][t] * g[pix][k], computed per 2x64
// output tile as D[k][t] = gT[k][m] x pT[t][m] MFMA products with both
// operands m-contiguous in LDS:
//   gT: the tile's NHWC g slab transposed (b16 scatter, 16/thread),
//   pT: the im2col expansion of the staged input rows (8-gather b128
//       writes; invalid slots are written as zeros).
// Each block walks a contiguous tile range and stores its fp32 partial
// slab [64][224]; stem_unpack_dw4 folds the slabs (no atomics).
__global__ __launch_bounds__(512, 2) void stem_wrw_kernel(
    const __bf16* __restrict__ x4, const __bf16* __restrict__ g,
    float* __restrict__ dwslab, StemParams p, int n_blocks) {
  const int tid = threadIdx.x;
  constexpr int XP8 = (ST_ICOLS * 4 + 7) / 8;
  constexpr int XSTR = XP8 * 8;
  constexpr int MSTR = ST_M + 8;                 // 136
  __shared__ __align__(16) __bf16 xin[2][ST_ROWS * XSTR];   // 19.3 KB
  __shared__ __align__(16) __bf16 gT[ST_K * MSTR];          // 17.4 KB
  __shared__ __align__(16) __bf16 pT[ST_KD * MSTR];         // 61 KB

  const int wid = tid >> 6, lane = tid & 63;
  const int kb = wid & 1;               // k block (2 x 32)
  const int tg = wid >> 1;              // t group: blocks {tg, tg+4}
  const int lrow = lane & 31, lk8 = lane >> 5;

  const int cpb = (p.total_tiles + n_blocks - 1) / n_blocks;
  const int t_lo = blockIdx.x * cpb;
  const int t_hi = bd_min(t_lo + cpb, p.total_tiles);

  constexpr int PIECES = ST_ROWS * XP8;
  constexpr int PPT = (PIECES + 511) / 512;
  uint4 xreg[PPT];

  // g staging coordinates: pixel m = tid>>2, 16 k at (tid&3)*16
  const int sg_m = tid >> 2;
  const int sg_k16 = (tid & 3) * 16;
  uint4 greg[2];

#define SW_GLOAD(tile)                                                    \
  {                                                                       \
    int tx = (tile) % p.tiles_x;                                          \
    int rem = (tile) / p.tiles_x;                                         \
    int band = rem % p.bands_per_image;                                   \
    int n = rem / p.bands_per_image;                                      \
    int oy = band * 2 + (sg_m >> 6);                                      \
    int ox = tx * 64 + (sg_m & 63);                                       \
    uint4 z{0, 0, 0, 0};                                                  \
    greg[0] = z; greg[1] = z;                                             \
    if (oy < p.Ho && ox < p.Wo) {                                         \
      const __bf16* src =                                                 \
          g + (((int64_t)n * p.Ho + oy) * p.Wo + ox) * ST_K + sg_k16;     \
      greg[0] = *(const uint4*)src;                                       \
      greg[1] = *(const uint4*)(src + 8);                                 \
    }                                                                     \
  }
#define SW_GWRITE()                                                       \
  {                                                                       \
    _Pragma("unroll")                                                     \
    for (int j = 0; j < 16; ++j)                                          \
      gT[(sg_k16 + j) * MSTR + sg_m] = ((const __bf16*)greg)[j];          \
  }

  // pT build: 3584 b128 targets, id -> (m8 = id & 15, t = id >> 4)
#define PT_BUILD(buf)                                                     \
  {                                                                       \
    _Pragma("unroll")                                                     \
    for (int it = 0; it < 7; ++it) {                                      \
      int id = tid + it * 512;                                            \
      int m8 = id & 15, t = id >> 4;                                      \
      int dy = t >> 5, sl = t & 31;                                       \
      int dx = sl >> 2, c = sl & 3;                                       \
      bf16x8 v = {};                                                      \
      if (dx < 7 && c < 3) {                                              \
        int m0 = m8 * 8;                                                  \
        int my = m0 >> 6, mx0 = m0 & 63;                                  \
        const __bf16* srow = &xin[buf][(my * 2 + dy) * XSTR];             \
        _Pragma("unroll")                                                 \
        for (int j = 0; j < 8; ++j)                                       \
          v[j] = srow[(2 * (mx0 + j) + dx) * 4 + c];                      \
      }                                                                   \
      *(bf16x8*)&pT[t * MSTR + m8 * 8] = v;                               \
    }                                                                     \
  }

  if (t_lo >= t_hi) {
    // still must emit a (zero) slab: wgrad_finish-style fold reads all
    float* slab = dwslab + (int64_t)blockIdx.x * ST_K * ST_KD;
    for (int i = tid; i < ST_K * ST_KD; i += 512) slab[i] = 0.f;
    return;
  }

  f32x16 acc[2];
#pragma unroll
  for (int a = 0; a < 2; ++a)
#pragma unroll
    for (int i = 0; i < 16; ++i) acc[a][i] = 0.f;

  ST_LOAD(t_lo);
  SW_GLOAD(t_lo);
  ST_WRITE(0);
  __syncthreads();      // xin[0] visible for PT_BUILD
  PT_BUILD(0);
  SW_GWRITE();

  int xb = 0;
  for (int t = t_lo; t < t_hi; ++t) {
    const bool more = t + 1 < t_hi;
    if (more) {
      ST_LOAD(t + 1);
      SW_GLOAD(t + 1);
    }
    __syncthreads();    // pT + gT ready
#pragma unroll
    for (int ms = 0; ms < 8; ++ms) {    // 8 x 16-m reduction steps
      const int moff = ms * 16 + lk8 * 8;
      bf16x8 afrag = *(const bf16x8*)&gT[(kb * 32 + lrow) * MSTR + moff];
#pragma unroll
      for (int a = 0; a < 2; ++a) {
        int tb = tg + a * 4;
        if (tb >= 7) continue;          // 224 = 7 t-blocks only
        bf16x8 bfrag =
            *(const bf16x8*)&pT[(tb * 32 + lrow) * MSTR + moff];
        acc[a] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(afrag, bfrag,
                                                         acc[a], 0, 0, 0);
      }
    }
    __syncthreads();    // reads done; safe to restage
    if (more) {
      ST_WRITE(xb ^ 1);
      __syncthreads();  // xin[xb^1] visible
      PT_BUILD(xb ^ 1);
      SW_GWRITE();
      xb ^= 1;
    }
  }

  // ---- slab store: D rows = k, cols = t ----
  float* slab = dwslab + (int64_t)blockIdx.x * ST_K * ST_KD;
  // zero-init (waves only cover 7 of 8 t-blocks and 224 of 224 slots,
  // but write below covers every (k, t) except none — still zero first
  // for the tg==3 second-accumulator slot symmetry)
  __syncthreads();
#pragma unroll
  for (int a = 0; a < 2; ++a) {
    int tb = tg + a * 4;
    if (tb >= 7) continue;
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      int krow = kb * 32 + (reg & 3) + 8 * (reg >> 2) + 4 * lk8;
      slab[krow * ST_KD + tb * 32 + lrow] = acc[a][reg];
    }
  }
}

extern "C" int bdbnn_stem_wrw_nslab(int N, int H, int W) {
  int Ho = H / 2, Wo = W / 2;
  int total = N * ((Ho + 1) / 2) * ((Wo + 63) / 64);
  return total < 512 ? total : 512;
}

extern "C" void bdbnn_stem_wrw(const void* x4, const void* g,
                               float* dwslab, int N, int H, int W,
                               hipStream_t stream) {
  StemParams p;
  p.N = N; p.H = H; p.W = W;
  p.Ho = H / 2; p.Wo = W / 2;
  p.bands_per_image = (p.Ho + 1) / 2;
  p.tiles_x = (p.Wo + 63) / 64;
  p.total_tiles = N * p.bands_per_image * p.tiles_x;
  int n_blocks = bd_min(512, p.total_tiles);
  stem_wrw_kernel<<<n_blocks, 512, 0, stream>>>(
      (const __bf16*)x4, (const __bf16*)g, dwslab, p, n_blocks);
}

// fold slabs + unpack [64][224] -> dw (64,3,7,7 fp32)
__global__ void stem_fold_dw4_kernel(const float* __restrict__ slabs,
                                     float* __restrict__ dw, int nslab) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;  // ((k*3+c)*7+dy)*7+dx
  if (i >= ST_K * 3 * 49) return;
  int dx = i % 7;
  int rem = i / 7;
  int dy = rem % 7;
  rem /= 7;
  int c = rem % 3;
  int k = rem / 3;
  int64_t j = k * ST_KD + dy * 32 + dx * 4 + c;
  float v = 0.f;
  for (int s = 0; s < nslab; ++s)
    v += slabs[(int64_t)s * ST_K * ST_KD + j];
  dw[i] = v;
}

extern "C" void bdbnn_stem_fold_dw4(const float* slabs, float* dw,
                                    int nslab, hipStream_t stream) {
  stem_fold_dw4_kernel<<<(ST_K * 3 * 49 + 255) / 256, 256, 0, stream>>>(
      slabs, dw, nslab);
}
