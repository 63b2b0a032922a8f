// BatchNorm over NHWC (spatial) / [M][C] (feed-forward) bf16 tensors.
// Stats, affine params and gradients are fp32 (SURVEY hard-part #3:
// "BN in bf16 - keep stats in fp32").
//
// All kernels are 8-channel vectorized (s16x8 = 16B/lane loads, G13) when
// C % 8 == 0, with a scalar fallback for narrow feature dims (C < 8, e.g.
// the z=2 generator input BN). Reductions use the row-lane scheme: a block
// covers a channel window, blockDim/width row-lanes walk rows coalesced,
// LDS-reduce, one atomic per channel.

#include "common.h"

// ------------------------------------------------------------- vector path
// pass 1: per-channel sum / sumsq. Channel groups of 8; colsW8 = number of
// 8-channel groups covered by one block (<= 32 so lanes >= 8).
__global__ void bn_stats_v8(const s16x8* __restrict__ x, long m, int c,
                            float* __restrict__ sum,
                            float* __restrict__ sumsq) {
  // each thread accumulates 8 channel-partials in registers; row-lanes are
  // reduced channel-slot by channel-slot through one LDS array
  int c8 = c / 8;
  int g0 = blockIdx.y * 32;                 // first 8-group of this block
  int groups = min(32, c8 - g0);            // groups covered (<=32)
  int lanes = (int)blockDim.x / groups;     // row-lanes
  int sub = (int)threadIdx.x / groups;
  int g = g0 + (int)threadIdx.x % groups;
  float s[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  float ss[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  if (sub < lanes) {
    for (long r = (long)blockIdx.x * lanes + sub; r < m;
         r += (long)gridDim.x * lanes) {
      s16x8 v = x[r * c8 + g];
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f((unsigned short)v[j]);
        s[j] += f;
        ss[j] += f * f;
      }
    }
  }
  // reduce row-lanes through LDS; ONE partial row per block, no atomics
  // (global-atomic contention across 1024 blocks was 4x slower)
  __shared__ float red[256];
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    red[threadIdx.x] = s[j];
    __syncthreads();
    if (sub == 0) {
      float acc = 0.f;
      for (int q = 0; q < lanes; ++q) acc += red[q * groups + g - g0];
      sum[(long)blockIdx.x * 2 * c + g * 8 + j] = acc;   // partials [gx][2c]
    }
    __syncthreads();
    red[threadIdx.x] = ss[j];
    __syncthreads();
    if (sub == 0) {
      float acc = 0.f;
      for (int q = 0; q < lanes; ++q) acc += red[q * groups + g - g0];
      sum[(long)blockIdx.x * 2 * c + c + g * 8 + j] = acc;
    }
    __syncthreads();
  }
}

// second pass: partials [gx][2c] -> sum[c], sumsq[c]. One wave per
// channel (a single 256-thread block serializes on load latency).
__global__ void bn_stats_sum2(const float* __restrict__ part, int gx, int c,
                              float* __restrict__ sum,
                              float* __restrict__ sumsq) {
  int col = blockIdx.x * (blockDim.x >> 6) + ((int)threadIdx.x >> 6);
  int lane = (int)threadIdx.x & 63;
  if (col >= c) return;
  float a = 0.f, b = 0.f;
  for (int r = lane; r < gx; r += 64) {
    a += part[(long)r * 2 * c + col];
    b += part[(long)r * 2 * c + c + col];
  }
  a = wave_reduce_sum(a);
  b = wave_reduce_sum(b);
  if (lane == 0) {
    sum[col] = a;
    sumsq[col] = b;
  }
}

__global__ void bn_apply_v8(const s16x8* __restrict__ x,
                            s16x8* __restrict__ y, long m, int c,
                            const float* __restrict__ mean,
                            const float* __restrict__ istd,
                            const float* __restrict__ gamma,
                            const float* __restrict__ beta) {
  int c8 = c / 8;
  long total = m * c8;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < total; i += (long)gridDim.x * blockDim.x) {
    int g = (int)(i % c8) * 8;
    s16x8 v = x[i];
    s16x8 o;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = (bf2f((unsigned short)v[j]) - mean[g + j]) * istd[g + j];
      o[j] = (short)f2bf(gamma[g + j] * f + beta[g + j]);
    }
    y[i] = o;
  }
}

__global__ void bn_apply_eval_v8(const s16x8* __restrict__ x,
                                 s16x8* __restrict__ y, long m, int c,
                                 const float* __restrict__ rm,
                                 const float* __restrict__ rv,
                                 const float* __restrict__ gamma,
                                 const float* __restrict__ beta, float eps) {
  int c8 = c / 8;
  long total = m * c8;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < total; i += (long)gridDim.x * blockDim.x) {
    int g = (int)(i % c8) * 8;
    s16x8 v = x[i];
    s16x8 o;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float is = rsqrtf(rv[g + j] + eps);
      float f = (bf2f((unsigned short)v[j]) - rm[g + j]) * is;
      o[j] = (short)f2bf(gamma[g + j] * f + beta[g + j]);
    }
    y[i] = o;
  }
}

__global__ void bn_bwd_reduce_v8(const s16x8* __restrict__ x,
                                 const s16x8* __restrict__ dy, long m, int c,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ istd,
                                 float* __restrict__ dgamma,
                                 float* __restrict__ dbeta) {
  int c8 = c / 8;
  int g0 = blockIdx.y * 32;
  int groups = min(32, c8 - g0);
  int lanes = (int)blockDim.x / groups;
  int sub = (int)threadIdx.x / groups;
  int g = g0 + (int)threadIdx.x % groups;
  float dg[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  float db[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  if (sub < lanes) {
    float mu[8], is[8];
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      mu[j] = mean[g * 8 + j];
      is[j] = istd[g * 8 + j];
    }
    for (long r = (long)blockIdx.x * lanes + sub; r < m;
         r += (long)gridDim.x * lanes) {
      s16x8 vx = x[r * c8 + g];
      s16x8 vg = dy[r * c8 + g];
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float gg = bf2f((unsigned short)vg[j]);
        float xh = (bf2f((unsigned short)vx[j]) - mu[j]) * is[j];
        dg[j] += gg * xh;
        db[j] += gg;
      }
    }
  }
  __shared__ float red[256];
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    red[threadIdx.x] = dg[j];
    __syncthreads();
    if (sub == 0) {
      float acc = 0.f;
      for (int q = 0; q < lanes; ++q) acc += red[q * groups + g - g0];
      dgamma[(long)blockIdx.x * 2 * c + g * 8 + j] = acc;  // partials
    }
    __syncthreads();
    red[threadIdx.x] = db[j];
    __syncthreads();
    if (sub == 0) {
      float acc = 0.f;
      for (int q = 0; q < lanes; ++q) acc += red[q * groups + g - g0];
      dgamma[(long)blockIdx.x * 2 * c + c + g * 8 + j] = acc;
    }
    __syncthreads();
  }
}

__global__ void bn_bwd_sum2(const float* __restrict__ part, int gx, int c,
                            float* __restrict__ dgamma,
                            float* __restrict__ dbeta) {
  int col = blockIdx.x * (blockDim.x >> 6) + ((int)threadIdx.x >> 6);
  int lane = (int)threadIdx.x & 63;
  if (col >= c) return;
  float a = 0.f, b = 0.f;
  for (int r = lane; r < gx; r += 64) {
    a += part[(long)r * 2 * c + col];
    b += part[(long)r * 2 * c + c + col];
  }
  a = wave_reduce_sum(a);
  b = wave_reduce_sum(b);
  if (lane == 0) {
    dgamma[col] = a;
    dbeta[col] = b;
  }
}

__global__ void bn_bwd_apply_v8(const s16x8* __restrict__ x,
                                const s16x8* __restrict__ dy,
                                s16x8* __restrict__ dx, long m, int c,
                                const float* __restrict__ mean,
                                const float* __restrict__ istd,
                                const float* __restrict__ gamma,
                                const float* __restrict__ dgamma,
                                const float* __restrict__ dbeta) {
  int c8 = c / 8;
  long total = m * c8;
  float inv_m = 1.f / (float)m;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < total; i += (long)gridDim.x * blockDim.x) {
    int g = (int)(i % c8) * 8;
    s16x8 vx = x[i], vg = dy[i];
    s16x8 o;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float xh = (bf2f((unsigned short)vx[j]) - mean[g + j]) * istd[g + j];
      float gg = bf2f((unsigned short)vg[j]);
      o[j] = (short)f2bf(gamma[g + j] * istd[g + j] *
                         (gg - dbeta[g + j] * inv_m -
                          xh * dgamma[g + j] * inv_m));
    }
    dx[i] = o;
  }
}

// BN backward apply with the PRODUCER activation's backward fused in.
// x (the BatchNorm input) IS the producing conv's activation output, so
// dpre = dxbn * act'(x) costs zero extra memory traffic, and the conv's
// bias gradient (column sums of dpre) accumulates into per-block
// partials on the way out — this removes the conv's standalone
// act_bwd_bias pass (3 full tensor streams) from the backward.
// Col-group structure (fixed channels per thread) like bn_bwd_reduce_v8.
__global__ void bn_bwd_apply_act_v8(
    const s16x8* __restrict__ x, const s16x8* __restrict__ dy,
    s16x8* __restrict__ dx, long m, int c, const float* __restrict__ mean,
    const float* __restrict__ istd, const float* __restrict__ gamma,
    const float* __restrict__ dgamma, const float* __restrict__ dbeta,
    int act, float slope, float* __restrict__ bias_part) {
  int c8 = c / 8;
  int g0 = blockIdx.y * 32;
  int groups = min(32, c8 - g0);
  int lanes = (int)blockDim.x / groups;
  int sub = (int)threadIdx.x / groups;
  int g = g0 + (int)threadIdx.x % groups;
  float inv_m = 1.f / (float)m;
  float db[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  if (sub < lanes) {
    float mu[8], is[8], gis[8], dgs[8], dbs[8];
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      mu[j] = mean[g * 8 + j];
      is[j] = istd[g * 8 + j];
      gis[j] = gamma[g * 8 + j] * is[j];
      dgs[j] = dgamma[g * 8 + j] * inv_m;
      dbs[j] = dbeta[g * 8 + j] * inv_m;
    }
    for (long r = (long)blockIdx.x * lanes + sub; r < m;
         r += (long)gridDim.x * lanes) {
      s16x8 vx = x[r * c8 + g], vg = dy[r * c8 + g];
      s16x8 o;
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float y = bf2f((unsigned short)vx[j]);
        float xh = (y - mu[j]) * is[j];
        float gg = bf2f((unsigned short)vg[j]);
        float dbn = gis[j] * (gg - dbs[j] - xh * dgs[j]);
        float dpre = dbn * act_bwd_from_y(y, act, slope);
        o[j] = (short)f2bf(dpre);
        db[j] += dpre;
      }
      dx[r * c8 + g] = o;
    }
  }
  if (bias_part == nullptr) return;
  __shared__ float red[256];
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    red[threadIdx.x] = db[j];
    __syncthreads();
    if (sub == 0) {
      float acc = 0.f;
      for (int q = 0; q < lanes; ++q) acc += red[q * groups + g - g0];
      bias_part[(long)blockIdx.x * c + g * 8 + j] = acc;
    }
    __syncthreads();
  }
}

// ------------------------------------------------------------- scalar path
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

static dim3 _colgrid_v8(long m, int c) {
  int c8 = c / 8;
  int groups = c8 < 32 ? c8 : 32;
  int lanes = 256 / groups;
  long chunks = (m + lanes - 1) / lanes;
  return dim3((unsigned)min((long)1024, max((long)1, chunks)),
              (unsigned)((c8 + 31) / 32));
}

static int _egrid(long total) {
  return (int)min((long)2048, total / 256 + 1);
}

// v8 two-pass (partials in scratch [gx][2c], no atomics); returns gx.
int launch_bn_stats_part(const void* x, long m, int c, float* scratch,
                         hipStream_t s) {
  dim3 g = _colgrid_v8(m, c);
  if (g.x > 2048) g.x = 2048;
  hipLaunchKernelGGL(bn_stats_v8, g, dim3(256), 0, s, (const s16x8*)x, m, c,
                     scratch, nullptr);
  return (int)g.x;
}

void launch_bn_stats_sum2(const float* scratch, int gx, int c, float* sum,
                          float* sumsq, hipStream_t s) {
  hipLaunchKernelGGL(bn_stats_sum2, dim3((c + 3) / 4), dim3(256), 0, s,
                     scratch, gx, c, sum, sumsq);
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
  if (c % 8 == 0) {
    hipLaunchKernelGGL(bn_apply_v8, dim3(_egrid(m * c / 8)), dim3(256), 0, s,
                       (const s16x8*)x, (s16x8*)y, m, c, mean, istd, gamma,
                       beta);
    return;
  }
  hipLaunchKernelGGL(bn_apply, dim3(_egrid(m * c)), dim3(256), 0, s,
                     (const unsigned short*)x, (unsigned short*)y, m, c, mean,
                     istd, gamma, beta);
}

void launch_bn_apply_eval(const void* x, void* y, long m, int c,
                          const float* rm, const float* rv, const float* gamma,
                          const float* beta, float eps, hipStream_t s) {
  if (c % 8 == 0) {
    hipLaunchKernelGGL(bn_apply_eval_v8, dim3(_egrid(m * c / 8)), dim3(256),
                       0, s, (const s16x8*)x, (s16x8*)y, m, c, rm, rv, gamma,
                       beta, eps);
    return;
  }
  hipLaunchKernelGGL(bn_apply_eval, dim3(_egrid(m * c)), dim3(256), 0, s,
                     (const unsigned short*)x, (unsigned short*)y, m, c, rm,
                     rv, gamma, beta, eps);
}

int launch_bn_bwd_reduce_part(const void* x, const void* dy, long m, int c,
                              const float* mean, const float* istd,
                              float* scratch, hipStream_t s) {
  dim3 g = _colgrid_v8(m, c);
  if (g.x > 2048) g.x = 2048;
  hipLaunchKernelGGL(bn_bwd_reduce_v8, g, dim3(256), 0, s, (const s16x8*)x,
                     (const s16x8*)dy, m, c, mean, istd, scratch, nullptr);
  return (int)g.x;
}

void launch_bn_bwd_sum2(const float* scratch, int gx, int c, float* dgamma,
                        float* dbeta, hipStream_t s) {
  hipLaunchKernelGGL(bn_bwd_sum2, dim3((c + 3) / 4), dim3(256), 0, s,
                     scratch, gx, c, dgamma, dbeta);
}

void launch_bn_bwd_reduce(const void* x, const void* dy, long m, int c,
                          const float* mean, const float* istd, float* dgamma,
                          float* dbeta, hipStream_t s) {
  hipLaunchKernelGGL(bn_bwd_reduce, _colgrid(m, c), dim3(256), 0, s,
                     (const unsigned short*)x, (const unsigned short*)dy, m, c,
                     mean, istd, dgamma, dbeta);
}

int launch_bn_bwd_apply_act(const void* x, const void* dy, void* dx, long m,
                            int c, const float* mean, const float* istd,
                            const float* gamma, const float* dgamma,
                            const float* dbeta, int act, float slope,
                            float* bias_part, hipStream_t s) {
  dim3 g = _colgrid_v8(m, c);
  if (g.x > 2048) g.x = 2048;
  hipLaunchKernelGGL(bn_bwd_apply_act_v8, g, dim3(256), 0, s, (const s16x8*)x,
                     (const s16x8*)dy, (s16x8*)dx, m, c, mean, istd, gamma,
                     dgamma, dbeta, act, slope, bias_part);
  return (int)g.x;
}

void launch_bn_bwd_apply(const void* x, const void* dy, void* dx, long m,
                         int c, const float* mean, const float* istd,
                         const float* gamma, const float* dgamma,
                         const float* dbeta, hipStream_t s) {
  if (c % 8 == 0) {
    hipLaunchKernelGGL(bn_bwd_apply_v8, dim3(_egrid(m * c / 8)), dim3(256), 0,
                       s, (const s16x8*)x, (const s16x8*)dy, (s16x8*)dx, m, c,
                       mean, istd, gamma, dgamma, dbeta);
    return;
  }
  hipLaunchKernelGGL(bn_bwd_apply, dim3(_egrid(m * c)), dim3(256), 0, s,
                     (const unsigned short*)x, (const unsigned short*)dy,
                     (unsigned short*)dx, m, c, mean, istd, gamma, dgamma,
                     dbeta);
}

}  // extern "C"
