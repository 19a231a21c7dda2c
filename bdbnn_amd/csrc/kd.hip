// K5 — fused multi-tensor weight-space KD loss.
//
// Reference (ref:utils/KD_loss.py:56-67): per matched conv pair,
// KLDivLoss(log_target=True)(W_s, W_t) = mean_e(exp(W_t) * (W_t - W_s)),
// summed over pairs; the reference walks both models' named_modules every
// batch.  Here one launch reduces every pair; backward is analytic:
// d/dW_s = -exp(W_t)/numel.
#include "common.h"

struct PtrList2 { float* ptr[BDBNN_MAX_TENSORS]; };

constexpr int64_t KD_CHUNK_ELEMS = 32 * 1024;

__global__ void weight_kd_fwd_kernel(TensorListArg ws_meta, PtrList2 ws,
                                     PtrList2 wt,
                                     const int* __restrict__ block_tensor,
                                     const int64_t* __restrict__ block_off,
                                     double* __restrict__ out) {
  int l = block_tensor[blockIdx.x];
  int64_t n = ws_meta.numel[l];
  int64_t off = block_off[blockIdx.x];
  int64_t end = bd_min(n, off + KD_CHUNK_ELEMS);
  const float* S = ws.ptr[l];
  const float* T = wt.ptr[l];
  double s = 0;
  for (int64_t i = off + threadIdx.x; i < end; i += blockDim.x) {
    float t = T[i];
    s += (double)(expf(t) * (t - S[i]));
  }
  __shared__ double red[256];
  red[threadIdx.x] = s;
  __syncthreads();
  for (int o = 128; o > 0; o >>= 1) {
    if (threadIdx.x < o) red[threadIdx.x] += red[threadIdx.x + o];
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd(out, red[0] / (double)n);
}

__global__ void weight_kd_bwd_kernel(TensorListArg wt_meta, PtrList2 wt,
                                     PtrList2 grads,
                                     const int* __restrict__ block_tensor,
                                     const int64_t* __restrict__ block_off,
                                     const float* __restrict__ gscale) {
  int l = block_tensor[blockIdx.x];
  int64_t n = wt_meta.numel[l];
  int64_t off = block_off[blockIdx.x];
  int64_t end = bd_min(n, off + KD_CHUNK_ELEMS);
  const float* T = wt.ptr[l];
  float* G = grads.ptr[l];
  float coef = -gscale[0] / (float)n;
  for (int64_t i = off + threadIdx.x; i < end; i += blockDim.x)
    G[i] = coef * expf(T[i]);
}

extern "C" void bdbnn_weight_kd_fwd(const TensorListArg* meta,
                                    const PtrList2* ws, const PtrList2* wt,
                                    const int* bt, const int64_t* bo,
                                    int n_blocks, double* out,
                                    hipStream_t stream) {
  hipMemsetAsync(out, 0, sizeof(double), stream);
  weight_kd_fwd_kernel<<<n_blocks, 256, 0, stream>>>(
      *meta, *ws, *wt, bt, bo, out);
}

extern "C" void bdbnn_weight_kd_bwd(const TensorListArg* meta,
                                    const PtrList2* wt, const PtrList2* grads,
                                    const int* bt, const int64_t* bo,
                                    int n_blocks, const float* gscale,
                                    hipStream_t stream) {
  weight_kd_bwd_kernel<<<n_blocks, 256, 0, stream>>>(
      *meta, *wt, *grads, bt, bo, gscale);
}
