// K9 — fused multi-tensor optimizer steps (SGD-momentum, Adam).
//
// Reference uses per-tensor torch SGD/Adam (ref:train.py:319,332); here one
// launch updates every parameter of a group.  Block -> (tensor, offset)
// schedule is built on the host (cached per optimizer in Python).
#include "common.h"

struct PtrList { float* ptr[BDBNN_MAX_TENSORS]; };

constexpr int64_t OPT_CHUNK_ELEMS = 32 * 1024;

__global__ void fused_sgd_kernel(TensorListArg params_meta, PtrList p,
                                 PtrList g, PtrList buf,
                                 const int* __restrict__ block_tensor,
                                 const int64_t* __restrict__ block_off,
                                 float lr, float momentum, float wd) {
  int l = block_tensor[blockIdx.x];
  int64_t n = params_meta.numel[l];
  int64_t off = block_off[blockIdx.x];
  int64_t end = bd_min(n, off + OPT_CHUNK_ELEMS);
  float* P = p.ptr[l];
  const float* G = g.ptr[l];
  float* B = buf.ptr[l];
  for (int64_t i = off + threadIdx.x; i < end; i += blockDim.x) {
    float gi = G[i] + wd * P[i];
    float bi = B[i] * momentum + gi;
    B[i] = bi;
    P[i] -= lr * bi;
  }
}

__global__ void fused_adam_kernel(TensorListArg params_meta, PtrList p,
                                  PtrList g, PtrList m1, PtrList m2,
                                  const int* __restrict__ block_tensor,
                                  const int64_t* __restrict__ block_off,
                                  float lr, float beta1, float beta2,
                                  float eps, float wd, float bc1, float bc2) {
  int l = block_tensor[blockIdx.x];
  int64_t n = params_meta.numel[l];
  int64_t off = block_off[blockIdx.x];
  int64_t end = bd_min(n, off + OPT_CHUNK_ELEMS);
  float* P = p.ptr[l];
  const float* G = g.ptr[l];
  float* M = m1.ptr[l];
  float* V = m2.ptr[l];
  float inv_bc1 = 1.f / bc1;
  float inv_bc2 = 1.f / bc2;
  for (int64_t i = off + threadIdx.x; i < end; i += blockDim.x) {
    float gi = G[i] + wd * P[i];
    float mi = M[i] * beta1 + (1.f - beta1) * gi;
    float vi = V[i] * beta2 + (1.f - beta2) * gi * gi;
    M[i] = mi; V[i] = vi;
    float denom = sqrtf(vi * inv_bc2) + eps;
    P[i] -= lr * (mi * inv_bc1) / denom;
  }
}

extern "C" void bdbnn_fused_sgd(const TensorListArg* meta, const PtrList* p,
                                const PtrList* g, const PtrList* buf,
                                const int* bt, const int64_t* bo,
                                int n_blocks, float lr, float momentum,
                                float wd, hipStream_t stream) {
  fused_sgd_kernel<<<n_blocks, 256, 0, stream>>>(
      *meta, *p, *g, *buf, bt, bo, lr, momentum, wd);
}

extern "C" void bdbnn_fused_adam(const TensorListArg* meta, const PtrList* p,
                                 const PtrList* g, const PtrList* m1,
                                 const PtrList* m2, const int* bt,
                                 const int64_t* bo, int n_blocks, float lr,
                                 float beta1, float beta2, float eps,
                                 float wd, float bc1, float bc2,
                                 hipStream_t stream) {
  fused_adam_kernel<<<n_blocks, 256, 0, stream>>>(
      *meta, *p, *g, *m1, *m2, bt, bo, lr, beta1, beta2, eps, wd, bc1, bc2);
}
