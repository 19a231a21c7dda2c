// K3 — fused multi-tensor kurtosis regularizer.
//
// Reference semantics (ref:kurtosis.py:23-28):
//   kurt = mean(z^4), z = (w - mu)/sigma, sigma UNBIASED; loss = (kurt-tgt)^2
// The reference launches this per layer per step (19 small launches +
// Python object churn, ref:train.py:461-512); here: ONE forward launch
// computes raw moments of ALL tensors (fp64 accumulation, per-block
// partials + device atomics), a tiny finalize kernel derives
// (mu, sigma, kurt, z3mean, loss) per tensor, and ONE backward launch
// applies the analytic gradient
//   dloss/dw_i = 2(kurt-tgt) * (4/(n*sigma)) *
//                (z_i^3 - mean(z^3) - z_i*kurt*n/(n-1)).
//
// Work split: a host-built schedule maps each block to (tensor l, offset);
// every block covers CHUNK_ELEMS contiguous elements of its tensor.
#include "common.h"

constexpr int64_t KURT_CHUNK_ELEMS = 32 * 1024;

struct GradPtrs { float* ptr[BDBNN_MAX_TENSORS]; };

__global__ void kurt_moments_kernel(TensorListArg lists,
                                    const int* __restrict__ block_tensor,
                                    const int64_t* __restrict__ block_off,
                                    double* __restrict__ mom) {
  int l = block_tensor[blockIdx.x];
  const float* w = lists.ptr[l];
  int64_t n = lists.numel[l];
  int64_t off = block_off[blockIdx.x];
  int64_t end = bd_min(n, off + KURT_CHUNK_ELEMS);
  double s1 = 0, s2 = 0, s3 = 0, s4 = 0;
  for (int64_t i = off + threadIdx.x; i < end; i += blockDim.x) {
    double v = (double)w[i];
    double v2 = v * v;
    s1 += v; s2 += v2; s3 += v2 * v; s4 += v2 * v2;
  }
  __shared__ double red[4][256];
  red[0][threadIdx.x] = s1; red[1][threadIdx.x] = s2;
  red[2][threadIdx.x] = s3; red[3][threadIdx.x] = s4;
  __syncthreads();
  for (int o = 128; o > 0; o >>= 1) {
    if (threadIdx.x < o)
      for (int m = 0; m < 4; ++m)
        red[m][threadIdx.x] += red[m][threadIdx.x + o];
    __syncthreads();
  }
  if (threadIdx.x < 4)
    atomicAdd(&mom[l * 4 + threadIdx.x], red[threadIdx.x][0]);
}

__global__ void kurt_finalize_kernel(TensorListArg lists,
                                     const double* __restrict__ mom,
                                     const float* __restrict__ targets,
                                     float* __restrict__ stats,
                                     float* __restrict__ losses,
                                     float* __restrict__ kurts) {
  int l = blockIdx.x * blockDim.x + threadIdx.x;
  if (l >= lists.n) return;
  double n = (double)lists.numel[l];
  double m1 = mom[l * 4 + 0] / n;
  double m2 = mom[l * 4 + 1] / n;
  double m3 = mom[l * 4 + 2] / n;
  double m4 = mom[l * 4 + 3] / n;
  double c2 = m2 - m1 * m1;
  double c3 = m3 - 3 * m1 * m2 + 2 * m1 * m1 * m1;
  double c4 = m4 - 4 * m1 * m3 + 6 * m1 * m1 * m2 - 3 * m1 * m1 * m1 * m1;
  double var_unb = c2 * n / (n - 1.0);   // torch.std is unbiased (ref:25)
  double sigma = sqrt(var_unb);
  double kurt = c4 / (var_unb * var_unb);
  double z3m = c3 / (var_unb * sigma);
  double d = kurt - (double)targets[l];
  stats[l * 4 + 0] = (float)m1;
  stats[l * 4 + 1] = (float)sigma;
  stats[l * 4 + 2] = (float)kurt;
  stats[l * 4 + 3] = (float)z3m;
  losses[l] = (float)(d * d);
  kurts[l] = (float)kurt;
}

__global__ void kurt_bwd_kernel(TensorListArg lists, GradPtrs gp,
                                const int* __restrict__ block_tensor,
                                const int64_t* __restrict__ block_off,
                                const float* __restrict__ stats,
                                const float* __restrict__ targets,
                                const float* __restrict__ gscale) {
  int l = block_tensor[blockIdx.x];
  const float* w = lists.ptr[l];
  float* g = gp.ptr[l];
  int64_t n = lists.numel[l];
  float mu = stats[l * 4 + 0], sigma = stats[l * 4 + 1];
  float kurt = stats[l * 4 + 2], z3m = stats[l * 4 + 3];
  float coef = gscale[0] * 2.f * (kurt - targets[l]) * 4.f / (float(n) * sigma);
  float knn = kurt * float(n) / float(n - 1);
  int64_t off = block_off[blockIdx.x];
  int64_t end = bd_min(n, off + KURT_CHUNK_ELEMS);
  for (int64_t i = off + threadIdx.x; i < end; i += blockDim.x) {
    float z = (w[i] - mu) / sigma;
    g[i] = coef * (z * z * z - z3m - z * knn);
  }
}

extern "C" void bdbnn_kurtosis_fwd(const TensorListArg* lists,
                                   const int* block_tensor_dev,
                                   const int64_t* block_off_dev,
                                   int n_blocks, double* mom_dev,
                                   const float* targets, float* stats,
                                   float* losses, float* kurts,
                                   hipStream_t stream) {
  hipMemsetAsync(mom_dev, 0, sizeof(double) * 4 * lists->n, stream);
  kurt_moments_kernel<<<n_blocks, 256, 0, stream>>>(
      *lists, block_tensor_dev, block_off_dev, mom_dev);
  int fin_threads = 64;
  kurt_finalize_kernel<<<(lists->n + fin_threads - 1) / fin_threads,
                         fin_threads, 0, stream>>>(
      *lists, mom_dev, targets, stats, losses, kurts);
}

extern "C" void bdbnn_kurtosis_bwd(const TensorListArg* lists,
                                   const GradPtrs* gp,
                                   const int* block_tensor_dev,
                                   const int64_t* block_off_dev,
                                   int n_blocks, const float* stats,
                                   const float* targets, const float* gscale,
                                   hipStream_t stream) {
  kurt_bwd_kernel<<<n_blocks, 256, 0, stream>>>(
      *lists, *gp, block_tensor_dev, block_off_dev, stats, targets, gscale);
}
