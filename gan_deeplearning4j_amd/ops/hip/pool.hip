// NHWC max-pool (with argmax for backward) and nearest-neighbour upsample.
// Covers the reference's SubsamplingLayer(MAX, 2x2 s1) (Java:141-154) and
// Upsampling2D(2) (Java:201-211), plus their gradients.

#include "common.h"

struct PoolGeom {
  int N, H, W, C;   // input dims
  int Ho, Wo;       // output dims
  int k, stride;    // kernel, stride (square)
};

__global__ void maxpool_fwd_nhwc(const unsigned short* __restrict__ in,
                                 unsigned short* __restrict__ out,
                                 unsigned char* __restrict__ argmax,
                                 PoolGeom g) {
  long total = (long)g.N * g.Ho * g.Wo * g.C;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < total; i += (long)gridDim.x * blockDim.x) {
    int c = (int)(i % g.C);
    long t = i / g.C;
    int wo = (int)(t % g.Wo);
    t /= g.Wo;
    int ho = (int)(t % g.Ho);
    int n = (int)(t / g.Ho);
    float best = -1e30f;
    int besti = 0;
    for (int r = 0; r < g.k; ++r) {
      int h = ho * g.stride + r;
      if (h >= g.H) continue;
      for (int s_ = 0; s_ < g.k; ++s_) {
        int w = wo * g.stride + s_;
        if (w >= g.W) continue;
        float v = bf2f(in[(((long)n * g.H + h) * g.W + w) * g.C + c]);
        if (v > best) { best = v; besti = r * g.k + s_; }
      }
    }
    out[i] = f2bf(best);
    argmax[i] = (unsigned char)besti;
  }
}

// gather form: din[h][w] = sum of dout over windows whose argmax picked it
__global__ void maxpool_bwd_nhwc(const unsigned short* __restrict__ dout,
                                 const unsigned char* __restrict__ argmax,
                                 unsigned short* __restrict__ din,
                                 PoolGeom g) {
  long total = (long)g.N * g.H * g.W * g.C;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < total; i += (long)gridDim.x * blockDim.x) {
    int c = (int)(i % g.C);
    long t = i / g.C;
    int w = (int)(t % g.W);
    t /= g.W;
    int h = (int)(t % g.H);
    int n = (int)(t / g.H);
    float acc = 0.f;
    for (int r = 0; r < g.k; ++r) {
      int hop = h - r;
      if (hop < 0 || hop % g.stride) continue;
      int ho = hop / g.stride;
      if (ho >= g.Ho) continue;
      for (int s_ = 0; s_ < g.k; ++s_) {
        int wop = w - s_;
        if (wop < 0 || wop % g.stride) continue;
        int wo = wop / g.stride;
        if (wo >= g.Wo) continue;
        long o = (((long)n * g.Ho + ho) * g.Wo + wo) * g.C + c;
        if (argmax[o] == (unsigned char)(r * g.k + s_)) acc += bf2f(dout[o]);
      }
    }
    din[i] = f2bf(acc);
  }
}

// nearest upsample x scale: out[n][h][w][c] = in[n][h/s][w/s][c]
__global__ void upsample_fwd_nhwc(const unsigned short* __restrict__ in,
                                  unsigned short* __restrict__ out,
                                  int n_, int h, int w, int c, int scale) {
  int Ho = h * scale, Wo = w * scale;
  long total = (long)n_ * Ho * Wo * c;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < total; i += (long)gridDim.x * blockDim.x) {
    int cc = (int)(i % c);
    long t = i / c;
    int wo = (int)(t % Wo);
    t /= Wo;
    int ho = (int)(t % Ho);
    int n = (int)(t / Ho);
    out[i] = in[(((long)n * h + ho / scale) * w + wo / scale) * c + cc];
  }
}

// backward: din = sum over the scale x scale block of dout
__global__ void upsample_bwd_nhwc(const unsigned short* __restrict__ dout,
                                  unsigned short* __restrict__ din,
                                  int n_, int h, int w, int c, int scale) {
  int Ho = h * scale, Wo = w * scale;
  long total = (long)n_ * h * w * c;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < total; i += (long)gridDim.x * blockDim.x) {
    int cc = (int)(i % c);
    long t = i / c;
    int wi = (int)(t % w);
    t /= w;
    int hi = (int)(t % h);
    int n = (int)(t / h);
    float acc = 0.f;
    for (int r = 0; r < scale; ++r)
      for (int s_ = 0; s_ < scale; ++s_)
        acc += bf2f(dout[(((long)n * Ho + hi * scale + r) * Wo +
                          wi * scale + s_) * c + cc]);
    din[i] = f2bf(acc);
  }
}

extern "C" {

static int _grid_for(long total) {
  long g = (total + 255) / 256;
  return (int)min((long)2048, max((long)1, g));
}

void launch_maxpool_fwd(const void* in, void* out, void* argmax, PoolGeom g,
                        hipStream_t s) {
  hipLaunchKernelGGL(maxpool_fwd_nhwc,
                     dim3(_grid_for((long)g.N * g.Ho * g.Wo * g.C)), dim3(256),
                     0, s, (const unsigned short*)in, (unsigned short*)out,
                     (unsigned char*)argmax, g);
}

void launch_maxpool_bwd(const void* dout, const void* argmax, void* din,
                        PoolGeom g, hipStream_t s) {
  hipLaunchKernelGGL(maxpool_bwd_nhwc,
                     dim3(_grid_for((long)g.N * g.H * g.W * g.C)), dim3(256),
                     0, s, (const unsigned short*)dout,
                     (const unsigned char*)argmax, (unsigned short*)din, g);
}

void launch_upsample_fwd(const void* in, void* out, int n, int h, int w,
                         int c, int scale, hipStream_t s) {
  hipLaunchKernelGGL(upsample_fwd_nhwc,
                     dim3(_grid_for((long)n * h * w * c * scale * scale)),
                     dim3(256), 0, s, (const unsigned short*)in,
                     (unsigned short*)out, n, h, w, c, scale);
}

void launch_upsample_bwd(const void* dout, void* din, int n, int h, int w,
                         int c, int scale, hipStream_t s) {
  hipLaunchKernelGGL(upsample_bwd_nhwc, dim3(_grid_for((long)n * h * w * c)),
                     dim3(256), 0, s, (const unsigned short*)dout,
                     (unsigned short*)din, n, h, w, c, scale);
}

}  // extern "C"
