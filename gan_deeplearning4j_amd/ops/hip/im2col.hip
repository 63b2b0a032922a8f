// NHWC im2col / col2im for the implicit-GEMM convolution path.
//
// im2col:  col[np][(r*S+s)*C + c] = in[n][hi][wi][c],  hi = ho*stride - pad + r
//          np = (n*Ho + ho)*Wo + wo; zero where out of bounds.
//          col K dim padded to kpad (zero tail) so the MFMA GEMM sees K%64==0.
// col2im:  gather-sum form (no atomics): for each input pixel, sum the col
//          entries of every patch that covers it. Optionally fused
//          bias + activation (used as the conv-transpose forward epilogue).
//
// Data is bf16, channels innermost (both sides C-contiguous => coalesced).

#include "common.h"

struct ConvGeom {
  int N, H, W, C;       // image dims (the im2col SOURCE / col2im TARGET)
  int Ho, Wo;           // patch-grid dims
  int R, S;             // kernel
  int stride, pad;
  int kpad;             // padded K = round_up(R*S*C, 64)
};

__global__ void im2col_nhwc(const unsigned short* __restrict__ in,
                            unsigned short* __restrict__ col, ConvGeom g) {
  // one thread per col element (vectorized x2 over c when C % 2 == 0 is a
  // later optimization; kept scalar-simple: each thread does 4 elements)
  long total = (long)g.N * g.Ho * g.Wo * g.kpad;
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  long stride = (long)gridDim.x * blockDim.x * 4;
  int rsc = g.R * g.S * g.C;
  for (; i < total; i += stride) {
    #pragma unroll
    for (int u = 0; u < 4; ++u) {
      long idx = i + u;
      if (idx >= total) break;
      int k = (int)(idx % g.kpad);
      long np = idx / g.kpad;
      unsigned short v = 0;
      if (k < rsc) {
        int c = k % g.C;
        int rs = k / g.C;
        int s_ = rs % g.S, r = rs / g.S;
        int wo = (int)(np % g.Wo);
        long t = np / g.Wo;
        int ho = (int)(t % g.Ho);
        int n = (int)(t / g.Ho);
        int hi = ho * g.stride - g.pad + r;
        int wi = wo * g.stride - g.pad + s_;
        if (hi >= 0 && hi < g.H && wi >= 0 && wi < g.W)
          v = in[(((long)n * g.H + hi) * g.W + wi) * g.C + c];
      }
      col[idx] = v;
    }
  }
}

// fast path: C % 8 == 0 -> move 8 bf16 (16B) per thread-iteration
__global__ void im2col_nhwc_v8(const s16x8* __restrict__ in,
                               s16x8* __restrict__ col, ConvGeom g) {
  int kp8 = g.kpad / 8;
  long total8 = (long)g.N * g.Ho * g.Wo * kp8;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  int rsc = g.R * g.S * g.C;
  int c8 = g.C / 8;
  for (; i < total8; i += stride) {
    int k8 = (int)(i % kp8);
    long np = i / kp8;
    int k = k8 * 8;
    s16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
    if (k < rsc) {
      int c = k % g.C;
      int rs = k / g.C;
      int s_ = rs % g.S, r = rs / g.S;
      int wo = (int)(np % g.Wo);
      long t = np / g.Wo;
      int ho = (int)(t % g.Ho);
      int n = (int)(t / g.Ho);
      int hi = ho * g.stride - g.pad + r;
      int wi = wo * g.stride - g.pad + s_;
      if (hi >= 0 && hi < g.H && wi >= 0 && wi < g.W)
        v = in[(((long)n * g.H + hi) * g.W + wi) * c8 + c / 8];
    }
    col[i] = v;
  }
}

// dIn[n][h][w][c] = sum over (r,s) with ho = (h + pad - r)/stride integral
// and in range of dcol[np(ho,wo)][(r*S+s)*C + c]
// With fuse_bias_act != 0 this is the convT forward epilogue:
// out = act(gather + bias[c]).
__global__ void col2im_nhwc(const unsigned short* __restrict__ dcol,
                            unsigned short* __restrict__ din, ConvGeom g,
                            const float* __restrict__ bias, int act,
                            float slope) {
  long total = (long)g.N * g.H * g.W * g.C;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    int c = (int)(i % g.C);
    long t = i / g.C;
    int w = (int)(t % g.W);
    t /= g.W;
    int h = (int)(t % g.H);
    int n = (int)(t / g.H);
    float acc = 0.f;
    for (int r = 0; r < g.R; ++r) {
      int hop = h + g.pad - r;
      if (hop < 0 || hop % g.stride) continue;
      int ho = hop / g.stride;
      if (ho >= g.Ho) continue;
      for (int s_ = 0; s_ < g.S; ++s_) {
        int wop = w + g.pad - s_;
        if (wop < 0 || wop % g.stride) continue;
        int wo = wop / g.stride;
        if (wo >= g.Wo) continue;
        long np = ((long)n * g.Ho + ho) * g.Wo + wo;
        acc += bf2f(dcol[np * g.kpad + (r * g.S + s_) * g.C + c]);
      }
    }
    if (bias != nullptr) acc += bias[c];
    din[i] = f2bf(act_fwd(acc, act, slope));
  }
}

// fast col2im: C % 8 == 0. SS: compile-time stride (1, 2) or 0 = generic
// (the stride div/mod chain was a measurable cost in the gather loop).
template <int SS>
__global__ void col2im_nhwc_v8(const unsigned short* __restrict__ dcol,
                               s16x8* __restrict__ din, ConvGeom g,
                               const float* __restrict__ bias, int act,
                               float slope) {
  long total8 = (long)g.N * g.H * g.W * (g.C / 8);
  int c8 = g.C / 8;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < total8; i += stride) {
    int cg = (int)(i % c8);
    long t = i / c8;
    int w = (int)(t % g.W);
    t /= g.W;
    int h = (int)(t % g.H);
    int n = (int)(t / g.H);
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int r = 0; r < g.R; ++r) {
      int hop = h + g.pad - r;
      int st = SS ? SS : g.stride;
      if (hop < 0 || (SS == 2 ? (hop & 1) : (SS == 1 ? 0 : hop % st)))
        continue;
      int ho = SS == 2 ? (hop >> 1) : (SS == 1 ? hop : hop / st);
      if (ho >= g.Ho) continue;
      for (int s_ = 0; s_ < g.S; ++s_) {
        int wop = w + g.pad - s_;
        if (wop < 0 || (SS == 2 ? (wop & 1) : (SS == 1 ? 0 : wop % st)))
          continue;
        int wo = SS == 2 ? (wop >> 1) : (SS == 1 ? wop : wop / st);
        if (wo >= g.Wo) continue;
        long np = ((long)n * g.Ho + ho) * g.Wo + wo;
        s16x8 v = *(const s16x8*)(&dcol[np * g.kpad +
                                        (r * g.S + s_) * g.C + cg * 8]);
        #pragma unroll
        for (int j = 0; j < 8; ++j) acc[j] += bf2f((unsigned short)v[j]);
      }
    }
    s16x8 o;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float b = bias != nullptr ? bias[cg * 8 + j] : 0.f;
      o[j] = (short)f2bf(act_fwd(acc[j] + b, act, slope));
    }
    din[i] = o;
  }
}

// col2im with fused BN-statistics: col-group/row-lane structure (like the
// reduction kernels) so each thread owns 8 channels, accumulating
// sum/sumsq of the post-activation output into partials [gx][2C].
template <int SS>
__global__ void col2im_stats_v8(const unsigned short* __restrict__ dcol,
                                s16x8* __restrict__ din, ConvGeom g,
                                const float* __restrict__ bias, int act,
                                float slope, float* __restrict__ part) {
  int c8 = g.C / 8;
  long rows = (long)g.N * g.H * g.W;          // pixels
  int g0 = blockIdx.y * 32;
  int groups = min(32, c8 - g0);
  int lanes = (int)blockDim.x / groups;
  int sub = (int)threadIdx.x / groups;
  int cg = g0 + (int)threadIdx.x % groups;
  float s[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  float ss[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  if (sub < lanes) {
    for (long px = (long)blockIdx.x * lanes + sub; px < rows;
         px += (long)gridDim.x * lanes) {
      long t = px;
      int w = (int)(t % g.W);
      t /= g.W;
      int h = (int)(t % g.H);
      int n = (int)(t / g.H);
      float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
      for (int r = 0; r < g.R; ++r) {
        int hop = h + g.pad - r;
        int st = SS ? SS : g.stride;
        if (hop < 0 || (SS == 2 ? (hop & 1) : (SS == 1 ? 0 : hop % st)))
          continue;
        int ho = SS == 2 ? (hop >> 1) : (SS == 1 ? hop : hop / st);
        if (ho >= g.Ho) continue;
        for (int s_ = 0; s_ < g.S; ++s_) {
          int wop = w + g.pad - s_;
          if (wop < 0 || (SS == 2 ? (wop & 1) : (SS == 1 ? 0 : wop % st)))
            continue;
          int wo = SS == 2 ? (wop >> 1) : (SS == 1 ? wop : wop / st);
          if (wo >= g.Wo) continue;
          long np = ((long)n * g.Ho + ho) * g.Wo + wo;
          s16x8 v = *(const s16x8*)(&dcol[np * g.kpad +
                                          (r * g.S + s_) * g.C + cg * 8]);
          #pragma unroll
          for (int j = 0; j < 8; ++j) acc[j] += bf2f((unsigned short)v[j]);
        }
      }
      s16x8 o;
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float b = bias != nullptr ? bias[cg * 8 + j] : 0.f;
        float y = act_fwd(acc[j] + b, act, slope);
        o[j] = (short)f2bf(y);
        s[j] += y;
        ss[j] += y * y;
      }
      din[px * c8 + cg] = o;
    }
  }
  __shared__ float red[256];
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    red[threadIdx.x] = s[j];
    __syncthreads();
    if (sub == 0) {
      float acc = 0.f;
      for (int q = 0; q < lanes; ++q) acc += red[q * groups + cg - g0];
      part[(long)blockIdx.x * 2 * g.C + cg * 8 + j] = acc;
    }
    __syncthreads();
    red[threadIdx.x] = ss[j];
    __syncthreads();
    if (sub == 0) {
      float acc = 0.f;
      for (int q = 0; q < lanes; ++q) acc += red[q * groups + cg - g0];
      part[(long)blockIdx.x * 2 * g.C + g.C + cg * 8 + j] = acc;
    }
    __syncthreads();
  }
}

// col2im with the PRODUCER's activation backward fused: din here is the
// gradient at the producer conv's activation output, and y0 is that
// activation output itself ([N,H,W,C], elementwise-aligned) — so
// dpre = col_accum * act'(y0) costs one extra read instead of the
// producer's whole standalone act_bwd_bias pass, and the producer's
// bias gradient (column sums of dpre) reduces into partials [gx][C] on
// the way out. Col-group structure like col2im_stats_v8.
template <int SS>
__global__ void col2im_dact_v8(const unsigned short* __restrict__ dcol,
                               s16x8* __restrict__ din, ConvGeom g,
                               const s16x8* __restrict__ y0, int act,
                               float slope, float* __restrict__ part) {
  int c8 = g.C / 8;
  long rows = (long)g.N * g.H * g.W;          // pixels
  int g0 = blockIdx.y * 32;
  int groups = min(32, c8 - g0);
  int lanes = (int)blockDim.x / groups;
  int sub = (int)threadIdx.x / groups;
  int cg = g0 + (int)threadIdx.x % groups;
  float db[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  if (sub < lanes) {
    for (long px = (long)blockIdx.x * lanes + sub; px < rows;
         px += (long)gridDim.x * lanes) {
      long t = px;
      int w = (int)(t % g.W);
      t /= g.W;
      int h = (int)(t % g.H);
      int n = (int)(t / g.H);
      float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
      for (int r = 0; r < g.R; ++r) {
        int hop = h + g.pad - r;
        int st = SS ? SS : g.stride;
        if (hop < 0 || (SS == 2 ? (hop & 1) : (SS == 1 ? 0 : hop % st)))
          continue;
        int ho = SS == 2 ? (hop >> 1) : (SS == 1 ? hop : hop / st);
        if (ho >= g.Ho) continue;
        for (int s_ = 0; s_ < g.S; ++s_) {
          int wop = w + g.pad - s_;
          if (wop < 0 || (SS == 2 ? (wop & 1) : (SS == 1 ? 0 : wop % st)))
            continue;
          int wo = SS == 2 ? (wop >> 1) : (SS == 1 ? wop : wop / st);
          if (wo >= g.Wo) continue;
          long np = ((long)n * g.Ho + ho) * g.Wo + wo;
          s16x8 v = *(const s16x8*)(&dcol[np * g.kpad +
                                          (r * g.S + s_) * g.C + cg * 8]);
          #pragma unroll
          for (int j = 0; j < 8; ++j) acc[j] += bf2f((unsigned short)v[j]);
        }
      }
      s16x8 vy = y0[px * c8 + cg];
      s16x8 o;
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float y = bf2f((unsigned short)vy[j]);
        float dpre = acc[j] * act_bwd_from_y(y, act, slope);
        o[j] = (short)f2bf(dpre);
        db[j] += dpre;
      }
      din[px * c8 + cg] = o;
    }
  }
  if (part == nullptr) return;
  __shared__ float red[256];
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    red[threadIdx.x] = db[j];
    __syncthreads();
    if (sub == 0) {
      float acc = 0.f;
      for (int q = 0; q < lanes; ++q) acc += red[q * groups + cg - g0];
      part[(long)blockIdx.x * g.C + cg * 8 + j] = acc;
    }
    __syncthreads();
  }
}

extern "C" {

// returns gx (partials rows)
int launch_col2im_stats(const void* dcol, void* din, ConvGeom g,
                        const float* bias, int act, float slope, float* part,
                        hipStream_t s) {
  int c8 = g.C / 8;
  int groups = c8 < 32 ? c8 : 32;
  int lanes = 256 / groups;
  long rows = (long)g.N * g.H * g.W;
  long chunks = (rows + lanes - 1) / lanes;
  dim3 grid((unsigned)min((long)2048, max((long)1, chunks)),
            (unsigned)ceil_div(c8, 32));
  if (g.stride == 2)
    hipLaunchKernelGGL((col2im_stats_v8<2>), grid, dim3(256), 0, s,
                       (const unsigned short*)dcol, (s16x8*)din, g, bias,
                       act, slope, part);
  else if (g.stride == 1)
    hipLaunchKernelGGL((col2im_stats_v8<1>), grid, dim3(256), 0, s,
                       (const unsigned short*)dcol, (s16x8*)din, g, bias,
                       act, slope, part);
  else
    hipLaunchKernelGGL((col2im_stats_v8<0>), grid, dim3(256), 0, s,
                       (const unsigned short*)dcol, (s16x8*)din, g, bias,
                       act, slope, part);
  return (int)grid.x;
}

// returns gx (bias-partials rows)
int launch_col2im_dact(const void* dcol, void* din, ConvGeom g,
                       const void* y0, int act, float slope, float* part,
                       hipStream_t s) {
  int c8 = g.C / 8;
  int groups = c8 < 32 ? c8 : 32;
  int lanes = 256 / groups;
  long rows = (long)g.N * g.H * g.W;
  long chunks = (rows + lanes - 1) / lanes;
  // col2im gathers R*S taps per pixel: it needs a near-full grid (a
  // 256-block cap measured 4x slower at DCGAN-64 dgrad sizes); partials
  // rows scale with grid.x (buffer sized 2048 in the binding)
  dim3 grid((unsigned)min((long)2048, max((long)1, chunks)),
            (unsigned)ceil_div(c8, 32));
  if (g.stride == 2)
    hipLaunchKernelGGL((col2im_dact_v8<2>), grid, dim3(256), 0, s,
                       (const unsigned short*)dcol, (s16x8*)din, g,
                       (const s16x8*)y0, act, slope, part);
  else if (g.stride == 1)
    hipLaunchKernelGGL((col2im_dact_v8<1>), grid, dim3(256), 0, s,
                       (const unsigned short*)dcol, (s16x8*)din, g,
                       (const s16x8*)y0, act, slope, part);
  else
    hipLaunchKernelGGL((col2im_dact_v8<0>), grid, dim3(256), 0, s,
                       (const unsigned short*)dcol, (s16x8*)din, g,
                       (const s16x8*)y0, act, slope, part);
  return (int)grid.x;
}

void launch_im2col(const void* in, void* col, ConvGeom g, hipStream_t s) {
  if (g.C % 8 == 0) {
    long total8 = (long)g.N * g.Ho * g.Wo * (g.kpad / 8);
    int grid = (int)min((long)2048, (total8 + 255) / 256 + 1);
    hipLaunchKernelGGL(im2col_nhwc_v8, dim3(grid), dim3(256), 0, s,
                       (const s16x8*)in, (s16x8*)col, g);
    return;
  }
  long total = (long)g.N * g.Ho * g.Wo * g.kpad;
  int grid = (int)min((long)2048, (total / 4 + 255) / 256 + 1);
  hipLaunchKernelGGL(im2col_nhwc, dim3(grid), dim3(256), 0, s,
                     (const unsigned short*)in, (unsigned short*)col, g);
}

void launch_col2im(const void* dcol, void* din, ConvGeom g, const float* bias,
                   int act, float slope, hipStream_t s) {
  // v8 path needs 16B-aligned channel groups in dcol rows: kpad and the
  // (r*S+s)*C offsets must be multiples of 8 -> C % 8 == 0 && kpad % 8 == 0
  if (g.C % 8 == 0 && g.kpad % 8 == 0) {
    long total8 = (long)g.N * g.H * g.W * (g.C / 8);
    int grid = (int)min((long)2048, (total8 + 255) / 256 + 1);
    if (g.stride == 2)
      hipLaunchKernelGGL((col2im_nhwc_v8<2>), dim3(grid), dim3(256), 0, s,
                         (const unsigned short*)dcol, (s16x8*)din, g, bias,
                         act, slope);
    else if (g.stride == 1)
      hipLaunchKernelGGL((col2im_nhwc_v8<1>), dim3(grid), dim3(256), 0, s,
                         (const unsigned short*)dcol, (s16x8*)din, g, bias,
                         act, slope);
    else
      hipLaunchKernelGGL((col2im_nhwc_v8<0>), dim3(grid), dim3(256), 0, s,
                         (const unsigned short*)dcol, (s16x8*)din, g, bias,
                         act, slope);
    return;
  }
  long total = (long)g.N * g.H * g.W * g.C;
  int grid = (int)min((long)2048, (total + 255) / 256 + 1);
  hipLaunchKernelGGL(col2im_nhwc, dim3(grid), dim3(256), 0, s,
                     (const unsigned short*)dcol, (unsigned short*)din, g,
                     bias, act, slope);
}

}  // extern "C"
