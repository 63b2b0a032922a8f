// Elementwise kernels: activations fwd/bwd, bias-column ops, losses.
// All memory-bound: bf16 I/O vectorized 8-wide (G13: scalar bf16 is 2-2.5x
// slower), grid-stride loops capped so the scheduler has room (G11).

#include "common.h"

// ---------------------------------------------------------------- act fwd
// y = act(x), bf16, vectorized by 8.
__global__ void act_fwd_bf16(const s16x8* __restrict__ x, s16x8* __restrict__ y,
                             long n8, int act, float slope) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n8; i += stride) {
    s16x8 v = x[i];
    s16x8 o;
    #pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = (short)f2bf(act_fwd(bf2f((unsigned short)v[j]), act, slope));
    y[i] = o;
  }
}

// dx = dy * act'(y)  (derivative from the output, see common.h)
__global__ void act_bwd_bf16(const s16x8* __restrict__ dy,
                             const s16x8* __restrict__ y,
                             s16x8* __restrict__ dx,
                             long n8, int act, float slope) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n8; i += stride) {
    s16x8 g = dy[i], v = y[i];
    s16x8 o;
    #pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = (short)f2bf(bf2f((unsigned short)g[j]) *
                         act_bwd_from_y(bf2f((unsigned short)v[j]), act, slope));
    dx[i] = o;
  }
}

// scalar tails (n % 8 != 0)
__global__ void act_fwd_tail(const unsigned short* __restrict__ x,
                             unsigned short* __restrict__ y, long start,
                             long n, int act, float slope) {
  long i = start + threadIdx.x;
  if (i < n) y[i] = f2bf(act_fwd(bf2f(x[i]), act, slope));
}

__global__ void act_bwd_tail(const unsigned short* __restrict__ dy,
                             const unsigned short* __restrict__ y,
                             unsigned short* __restrict__ dx, long start,
                             long n, int act, float slope) {
  long i = start + threadIdx.x;
  if (i < n)
    dx[i] = f2bf(bf2f(dy[i]) * act_bwd_from_y(bf2f(y[i]), act, slope));
}

// --------------------------------------------------------------- bias ops
// column sum over C[M][N] -> out[N] fp32 (bias gradient).
// Row-major coalesced: a block covers a 256-wide column window and walks
// rows with blockDim.x/colsW row-lanes in parallel, then LDS-reduces the
// row-lanes and atomically adds one partial per column. N is typically
// small (64..1024) and M huge (batch x spatial), so the row axis carries
// the parallelism (grid.x row-chunks).
// 8-wide path (n % 8 == 0): 16B loads, 8 register partials per thread
__global__ void col_sum_v8(const s16x8* __restrict__ a,
                           float* __restrict__ out, long m, int n) {
  int n8 = n / 8;
  int g0 = blockIdx.y * 32;
  int groups = min(32, n8 - g0);
  int lanes = (int)blockDim.x / groups;
  int sub = (int)threadIdx.x / groups;
  int g = g0 + (int)threadIdx.x % groups;
  float s[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  if (sub < lanes) {
    for (long r = (long)blockIdx.x * lanes + sub; r < m;
         r += (long)gridDim.x * lanes) {
      s16x8 v = a[r * n8 + g];
      #pragma unroll
      for (int j = 0; j < 8; ++j) s[j] += bf2f((unsigned short)v[j]);
    }
  }
  __shared__ float red[256];
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    red[threadIdx.x] = s[j];
    __syncthreads();
    if (sub == 0) {
      float acc = 0.f;
      for (int q = 0; q < lanes; ++q) acc += red[q * groups + g - g0];
      out[(long)blockIdx.x * n + g * 8 + j] = acc;   // partials [gx][n]
    }
    __syncthreads();
  }
}

__global__ void col_sum_sum2(const float* __restrict__ part, int gx, int n,
                             float* __restrict__ out) {
  int col = blockIdx.x * (blockDim.x >> 6) + ((int)threadIdx.x >> 6);
  int lane = (int)threadIdx.x & 63;
  if (col >= n) return;
  float a = 0.f;
  for (int r = lane; r < gx; r += 64) a += part[(long)r * n + col];
  a = wave_reduce_sum(a);
  if (lane == 0) out[col] = a;
}

// fused activation-backward + bias-grad partials over [M][N] (N % 8 == 0):
// dx = dy * act'(y) is written AND per-column partial sums of dx land in
// partials [gx][n] (same second pass as col_sum). Saves a full re-read of
// dpre for the bias gradient.
__global__ void act_bwd_bias_v8(const s16x8* __restrict__ dy,
                                const s16x8* __restrict__ y,
                                s16x8* __restrict__ dx,
                                float* __restrict__ partials, long m, int n,
                                int act, float slope) {
  int n8 = n / 8;
  int g0 = blockIdx.y * 32;
  int groups = min(32, n8 - g0);
  int lanes = (int)blockDim.x / groups;
  int sub = (int)threadIdx.x / groups;
  int g = g0 + (int)threadIdx.x % groups;
  float s[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  if (sub < lanes) {
    for (long r = (long)blockIdx.x * lanes + sub; r < m;
         r += (long)gridDim.x * lanes) {
      s16x8 vg = dy[r * n8 + g];
      s16x8 o;
      if (act != 0) {
        s16x8 vy = y[r * n8 + g];
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
          float d = bf2f((unsigned short)vg[j]) *
                    act_bwd_from_y(bf2f((unsigned short)vy[j]), act, slope);
          o[j] = (short)f2bf(d);
          s[j] += d;
        }
      } else {
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
          float d = bf2f((unsigned short)vg[j]);
          o[j] = (short)f2bf(d);
          s[j] += d;
        }
      }
      dx[r * n8 + g] = o;
    }
  }
  __shared__ float red[256];
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    red[threadIdx.x] = s[j];
    __syncthreads();
    if (sub == 0) {
      float acc = 0.f;
      for (int q = 0; q < lanes; ++q) acc += red[q * groups + g - g0];
      partials[(long)blockIdx.x * n + g * 8 + j] = acc;
    }
    __syncthreads();
  }
}

__global__ void col_sum_bf16(const unsigned short* __restrict__ a,
                             float* __restrict__ out, long m, int n) {
  __shared__ float ls[256];
  int c0 = blockIdx.y * 256;
  int colsW = min(256, n - c0);
  int lanes = (int)blockDim.x / colsW;          // row-lanes per block
  int sub = (int)threadIdx.x / colsW;
  int c = c0 + (int)threadIdx.x % colsW;
  float acc = 0.f;
  if (sub < lanes) {
    for (long r = (long)blockIdx.x * lanes + sub; r < m;
         r += (long)gridDim.x * lanes)
      acc += bf2f(a[r * n + c]);
  }
  ls[threadIdx.x] = acc;
  __syncthreads();
  if (sub == 0) {
    for (int s = 1; s < lanes; ++s) acc += ls[s * colsW + c - c0];
    atomicAdd(&out[c], acc);
  }
}

// ------------------------------------------------------------------ losses
// BCE-with-logits: loss_i = max(x,0) - x*y + log(1+exp(-|x|)); mean-reduced.
// (reference D7 sigmoid+XENT fused, Java:159-164)
__global__ void bce_logits_fwd(const unsigned short* __restrict__ logits,
                               const unsigned short* __restrict__ labels,
                               float* __restrict__ loss_sum, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  float acc = 0.f;
  for (; i < n; i += (long)gridDim.x * blockDim.x) {
    float x = bf2f(logits[i]), y = bf2f(labels[i]);
    acc += fmaxf(x, 0.f) - x * y + log1pf(__expf(-fabsf(x)));
  }
  acc = wave_reduce_sum(acc);
  __shared__ float ws[16];
  int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  if (lane == 0) ws[wid] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = 0.f;
    for (int w = 0; w < (int)(blockDim.x >> 6); ++w) s += ws[w];
    atomicAdd(loss_sum, s);
  }
}

// dlogits = (sigmoid(x) - y) * gscale   (gscale = dL/dloss / n)
__global__ void bce_logits_bwd(const unsigned short* __restrict__ logits,
                               const unsigned short* __restrict__ labels,
                               unsigned short* __restrict__ dlogits,
                               float gscale, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n; i += (long)gridDim.x * blockDim.x) {
    float x = bf2f(logits[i]), y = bf2f(labels[i]);
    float s = 1.0f / (1.0f + __expf(-x));
    dlogits[i] = f2bf((s - y) * gscale);
  }
}

// Softmax cross-entropy vs one-hot (MCXENT; reference classifier head).
// One wave per row (C <= a few thousand).
__global__ void softmax_xent_fwd(const unsigned short* __restrict__ logits,
                                 const unsigned short* __restrict__ onehot,
                                 float* __restrict__ loss_sum,
                                 unsigned short* __restrict__ probs,
                                 int rows, int cols) {
  int row = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  int lane = threadIdx.x & 63;
  if (row >= rows) return;
  const unsigned short* xr = logits + (long)row * cols;
  float mx = -1e30f;
  for (int c = lane; c < cols; c += 64) mx = fmaxf(mx, bf2f(xr[c]));
  mx = wave_reduce_max(mx);
  mx = __shfl(mx, 0, 64);
  float se = 0.f;
  for (int c = lane; c < cols; c += 64) se += __expf(bf2f(xr[c]) - mx);
  se = wave_reduce_sum(se);
  se = __shfl(se, 0, 64);
  float lse = __logf(se) + mx;
  float l = 0.f;
  for (int c = lane; c < cols; c += 64) {
    float p = __expf(bf2f(xr[c]) - lse);
    probs[(long)row * cols + c] = f2bf(p);
    l += bf2f(onehot[(long)row * cols + c]) * (lse - bf2f(xr[c]));
  }
  l = wave_reduce_sum(l);
  if (lane == 0) atomicAdd(loss_sum, l);
}

// dlogits = (probs - onehot) * gscale
__global__ void softmax_xent_bwd(const unsigned short* __restrict__ probs,
                                 const unsigned short* __restrict__ onehot,
                                 unsigned short* __restrict__ dlogits,
                                 float gscale, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n; i += (long)gridDim.x * blockDim.x)
    dlogits[i] = f2bf((bf2f(probs[i]) - bf2f(onehot[i])) * gscale);
}

// ------------------------------------------------------------- launchers
extern "C" {

void launch_act_fwd(const void* x, void* y, long n, int act, float slope,
                    hipStream_t s) {
  long n8 = n / 8;
  int grid = (int)min((long)2048, ceil_div((int)min(n8, (long)1 << 30), 256));
  if (grid < 1) grid = 1;
  if (n8 > 0)
    hipLaunchKernelGGL(act_fwd_bf16, dim3(grid), dim3(256), 0, s,
                       (const s16x8*)x, (s16x8*)y, n8, act, slope);
  if (n % 8)
    hipLaunchKernelGGL(act_fwd_tail, dim3(1), dim3(64), 0, s,
                       (const unsigned short*)x, (unsigned short*)y, n8 * 8, n,
                       act, slope);
}

void launch_act_bwd(const void* dy, const void* y, void* dx, long n, int act,
                    float slope, hipStream_t s) {
  long n8 = n / 8;
  int grid = (int)min((long)2048, ceil_div((int)min(n8, (long)1 << 30), 256));
  if (grid < 1) grid = 1;
  if (n8 > 0)
    hipLaunchKernelGGL(act_bwd_bf16, dim3(grid), dim3(256), 0, s,
                       (const s16x8*)dy, (const s16x8*)y, (s16x8*)dx, n8, act,
                       slope);
  if (n % 8)
    hipLaunchKernelGGL(act_bwd_tail, dim3(1), dim3(64), 0, s,
                       (const unsigned short*)dy, (const unsigned short*)y,
                       (unsigned short*)dx, n8 * 8, n, act, slope);
}

int launch_act_bwd_bias(const void* dy, const void* y, void* dx,
                        float* scratch, long m, int n, int act, float slope,
                        hipStream_t s) {
  int n8 = n / 8;
  int groups = n8 < 32 ? n8 : 32;
  int lanes = 256 / groups;
  long chunks = (m + lanes - 1) / lanes;
  dim3 grid((unsigned)min((long)2048, max((long)1, chunks)),
            (unsigned)ceil_div(n8, 32));
  hipLaunchKernelGGL(act_bwd_bias_v8, grid, dim3(256), 0, s,
                     (const s16x8*)dy, (const s16x8*)y, (s16x8*)dx, scratch,
                     m, n, act, slope);
  return (int)grid.x;
}

int launch_col_sum_part(const void* a, float* scratch, long m, int n,
                        hipStream_t s) {
  int n8 = n / 8;
  int groups = n8 < 32 ? n8 : 32;
  int lanes = 256 / groups;
  long chunks = (m + lanes - 1) / lanes;
  dim3 grid((unsigned)min((long)2048, max((long)1, chunks)),
            (unsigned)ceil_div(n8, 32));
  hipLaunchKernelGGL(col_sum_v8, grid, dim3(256), 0, s, (const s16x8*)a,
                     scratch, m, n);
  return (int)grid.x;
}

void launch_col_sum_sum2(const float* scratch, int gx, int n, float* out,
                         hipStream_t s) {
  hipLaunchKernelGGL(col_sum_sum2, dim3(ceil_div(n, 4)), dim3(256), 0, s,
                     scratch, gx, n, out);
}

void launch_col_sum(const void* a, float* out, long m, int n, hipStream_t s) {
  int colsW = min(256, n);
  int lanes = 256 / colsW;
  long chunks = (m + lanes - 1) / lanes;
  dim3 grid((unsigned)min((long)1024, max((long)1, chunks)),
            (unsigned)ceil_div(n, 256));
  hipLaunchKernelGGL(col_sum_bf16, grid, dim3(256), 0, s,
                     (const unsigned short*)a, out, m, n);
}

void launch_bce_fwd(const void* logits, const void* labels, float* loss_sum,
                    long n, hipStream_t s) {
  int grid = min(1024, ceil_div((int)min(n, (long)1 << 30), 256));
  hipLaunchKernelGGL(bce_logits_fwd, dim3(grid), dim3(256), 0, s,
                     (const unsigned short*)logits,
                     (const unsigned short*)labels, loss_sum, n);
}

void launch_bce_bwd(const void* logits, const void* labels, void* dlogits,
                    float gscale, long n, hipStream_t s) {
  int grid = min(1024, ceil_div((int)min(n, (long)1 << 30), 256));
  hipLaunchKernelGGL(bce_logits_bwd, dim3(grid), dim3(256), 0, s,
                     (const unsigned short*)logits,
                     (const unsigned short*)labels, (unsigned short*)dlogits,
                     gscale, n);
}

void launch_softmax_xent_fwd(const void* logits, const void* onehot,
                             float* loss_sum, void* probs, int rows, int cols,
                             hipStream_t s) {
  int wpb = 4;  // 4 waves/block
  int grid = ceil_div(rows, wpb);
  hipLaunchKernelGGL(softmax_xent_fwd, dim3(grid), dim3(wpb * 64), 0, s,
                     (const unsigned short*)logits,
                     (const unsigned short*)onehot, loss_sum,
                     (unsigned short*)probs, rows, cols);
}

void launch_softmax_xent_bwd(const void* probs, const void* onehot,
                             void* dlogits, float gscale, long n,
                             hipStream_t s) {
  int grid = min(1024, ceil_div((int)min(n, (long)1 << 30), 256));
  hipLaunchKernelGGL(softmax_xent_bwd, dim3(grid), dim3(256), 0, s,
                     (const unsigned short*)probs,
                     (const unsigned short*)onehot,
                     (unsigned short*)dlogits, gscale, n);
}

}  // extern "C"
