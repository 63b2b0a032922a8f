// BatchNorm over NHWC (spatial) / [M][C] (feed-forward) bf16 tensors.
// Stats, affine params and gradients are fp32 (SURVEY hard-part #3:
// "BN in bf16 - keep stats in fp32").
//
// Training forward: batch mean/var (biased, like torch) + running-stat
// update (running_var uses the unbiased estimator, matching torch).
// Backward: the standard two-reduction formulation.

#include "common.h"

// pass 1: per-channel sum and sum-of-squares.
// Row-major coalesced with row-lanes (same scheme as col_sum_bf16):
// a block covers a <=256-wide channel window; blockDim.x/colsW row-lanes
// walk rows in parallel, LDS-reduce, one atomic pair per channel.
__global__ void bn_stats(const unsigned short* __restrict__ x, long m, int c,
                         float* __restrict__ sum, float* __restrict__ sumsq) {
  __shared__ float ls[512];
  int c0 = blockIdx.y * 256;
  int colsW = min(256, c - c0);
  int lanes = (int)blockDim.x / colsW;
  int sub = (int)threadIdx.x / colsW;
  int col = c0 + (int)threadIdx.x % colsW;
  float s = 0.f, ss = 0.f;
  if (sub < lanes) {
    for (long r = (long)blockIdx.x * lanes + sub; r < m;
         r += (long)gridDim.x * lanes) {
      float v = bf2f(x[r * c + col]);
      s += v;
      ss += v * v;
    }
  }
  ls[threadIdx.x] = s;
  ls[256 + threadIdx.x] = ss;
  __syncthreads();
  if (sub == 0) {
    for (int q = 1; q < lanes; ++q) {
      s += ls[q * colsW + col - c0];
      ss += ls[256 + q * colsW + col - c0];
    }
    atomicAdd(&sum[col], s);
    atomicAdd(&sumsq[col], ss);
  }
}

// finalize: mean/istd from sums; update running stats in-place (fp32)
__global__ void bn_finalize(const float* __restrict__ sum,
                            const float* __restrict__ sumsq, long m, int c,
                            float eps, float momentum,
                            float* __restrict__ mean,
                            float* __restrict__ istd,
                            float* __restrict__ running_mean,
                            float* __restrict__ running_var) {
  int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= c) return;
  float mu = sum[col] / (float)m;
  float var = fmaxf(sumsq[col] / (float)m - mu * mu, 0.f);
  mean[col] = mu;
  istd[col] = rsqrtf(var + eps);
  if (running_mean != nullptr) {
    float unbiased = m > 1 ? var * (float)m / (float)(m - 1) : var;
    running_mean[col] = (1.f - momentum) * running_mean[col] + momentum * mu;
    running_var[col] = (1.f - momentum) * running_var[col] + momentum * unbiased;
  }
}

// y = gamma * (x - mean) * istd + beta
__global__ void bn_apply(const unsigned short* __restrict__ x,
                         unsigned short* __restrict__ y, long m, int c,
                         const float* __restrict__ mean,
                         const float* __restrict__ istd,
                         const float* __restrict__ gamma,
                         const float* __restrict__ beta) {
  long total = m * c;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < total; i += (long)gridDim.x * blockDim.x) {
    int col = (int)(i % c);
    float v = (bf2f(x[i]) - mean[col]) * istd[col];
    y[i] = f2bf(gamma[col] * v + beta[col]);
  }
}

// eval-mode apply from running stats
__global__ void bn_apply_eval(const unsigned short* __restrict__ x,
                              unsigned short* __restrict__ y, long m, int c,
                              const float* __restrict__ running_mean,
                              const float* __restrict__ running_var,
                              const float* __restrict__ gamma,
                              const float* __restrict__ beta, float eps) {
  long total = m * c;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < total; i += (long)gridDim.x * blockDim.x) {
    int col = (int)(i % c);
    float is = rsqrtf(running_var[col] + eps);
    float v = (bf2f(x[i]) - running_mean[col]) * is;
    y[i] = f2bf(gamma[col] * v + beta[col]);
  }
}

// backward pass 1: dgamma = sum dy*xhat, dbeta = sum dy (coalesced scheme)
__global__ void bn_bwd_reduce(const unsigned short* __restrict__ x,
                              const unsigned short* __restrict__ dy, long m,
                              int c, const float* __restrict__ mean,
                              const float* __restrict__ istd,
                              float* __restrict__ dgamma,
                              float* __restrict__ dbeta) {
  __shared__ float ls[512];
  int c0 = blockIdx.y * 256;
  int colsW = min(256, c - c0);
  int lanes = (int)blockDim.x / colsW;
  int sub = (int)threadIdx.x / colsW;
  int col = c0 + (int)threadIdx.x % colsW;
  float dg = 0.f, db = 0.f;
  if (sub < lanes) {
    float mu = mean[col], is = istd[col];
    for (long r = (long)blockIdx.x * lanes + sub; r < m;
         r += (long)gridDim.x * lanes) {
      float g = bf2f(dy[r * c + col]);
      float xh = (bf2f(x[r * c + col]) - mu) * is;
      dg += g * xh;
      db += g;
    }
  }
  ls[threadIdx.x] = dg;
  ls[256 + threadIdx.x] = db;
  __syncthreads();
  if (sub == 0) {
    for (int q = 1; q < lanes; ++q) {
      dg += ls[q * colsW + col - c0];
      db += ls[256 + q * colsW + col - c0];
    }
    atomicAdd(&dgamma[col], dg);
    atomicAdd(&dbeta[col], db);
  }
}

// backward pass 2 (training):
// dx = gamma*istd * (dy - dbeta/m - xhat * dgamma/m)
__global__ void bn_bwd_apply(const unsigned short* __restrict__ x,
                             const unsigned short* __restrict__ dy,
                             unsigned short* __restrict__ dx, long m, int c,
                             const float* __restrict__ mean,
                             const float* __restrict__ istd,
                             const float* __restrict__ gamma,
                             const float* __restrict__ dgamma,
                             const float* __restrict__ dbeta) {
  long total = m * c;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  float inv_m = 1.f / (float)m;
  for (; i < total; i += (long)gridDim.x * blockDim.x) {
    int col = (int)(i % c);
    float xh = (bf2f(x[i]) - mean[col]) * istd[col];
    float g = bf2f(dy[i]);
    dx[i] = f2bf(gamma[col] * istd[col] *
                 (g - dbeta[col] * inv_m - xh * dgamma[col] * inv_m));
  }
}

extern "C" {

static dim3 _colgrid(long m, int c) {
  int colsW = c < 256 ? c : 256;
  int lanes = 256 / colsW;
  long chunks = (m + lanes - 1) / lanes;
  return dim3((unsigned)min((long)1024, max((long)1, chunks)),
              (unsigned)((c + 255) / 256));
}

void launch_bn_stats(const void* x, long m, int c, float* sum, float* sumsq,
                     hipStream_t s) {
  hipLaunchKernelGGL(bn_stats, _colgrid(m, c), dim3(256), 0, s,
                     (const unsigned short*)x, m, c, sum, sumsq);
}

void launch_bn_finalize(const float* sum, const float* sumsq, long m, int c,
                        float eps, float momentum, float* mean, float* istd,
                        float* running_mean, float* running_var,
                        hipStream_t s) {
  hipLaunchKernelGGL(bn_finalize, dim3((c + 255) / 256), dim3(256), 0, s, sum,
                     sumsq, m, c, eps, momentum, mean, istd, running_mean,
                     running_var);
}

void launch_bn_apply(const void* x, void* y, long m, int c, const float* mean,
                     const float* istd, const float* gamma, const float* beta,
                     hipStream_t s) {
  long total = m * c;
  int grid = (int)min((long)2048, (total + 255) / 256 + 1);
  hipLaunchKernelGGL(bn_apply, dim3(grid), dim3(256), 0, s,
                     (const unsigned short*)x, (unsigned short*)y, m, c, mean,
                     istd, gamma, beta);
}

void launch_bn_apply_eval(const void* x, void* y, long m, int c,
                          const float* rm, const float* rv, const float* gamma,
                          const float* beta, float eps, hipStream_t s) {
  long total = m * c;
  int grid = (int)min((long)2048, (total + 255) / 256 + 1);
  hipLaunchKernelGGL(bn_apply_eval, dim3(grid), dim3(256), 0, s,
                     (const unsigned short*)x, (unsigned short*)y, m, c, rm,
                     rv, gamma, beta, eps);
}

void launch_bn_bwd_reduce(const void* x, const void* dy, long m, int c,
                          const float* mean, const float* istd, float* dgamma,
                          float* dbeta, hipStream_t s) {
  hipLaunchKernelGGL(bn_bwd_reduce, _colgrid(m, c), dim3(256), 0, s,
                     (const unsigned short*)x, (const unsigned short*)dy, m, c,
                     mean, istd, dgamma, dbeta);
}

void launch_bn_bwd_apply(const void* x, const void* dy, void* dx, long m,
                         int c, const float* mean, const float* istd,
                         const float* gamma, const float* dgamma,
                         const float* dbeta, hipStream_t s) {
  long total = m * c;
  int grid = (int)min((long)2048, (total + 255) / 256 + 1);
  hipLaunchKernelGGL(bn_bwd_apply, dim3(grid), dim3(256), 0, s,
                     (const unsigned short*)x, (const unsigned short*)dy,
                     (unsigned short*)dx, m, c, mean, istd, gamma, dgamma,
                     dbeta);
}

}  // extern "C"
