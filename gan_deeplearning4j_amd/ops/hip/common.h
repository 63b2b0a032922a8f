// Common helpers for the gfx950 (CDNA4) kernel library.
// Wavefront = 64; block sizes are multiples of 64 throughout.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define DEV_INLINE __device__ __forceinline__

// ---------------------------------------------------------------- vectors
typedef __attribute__((ext_vector_type(2))) float    f32x2;
typedef __attribute__((ext_vector_type(4))) float    f32x4;
typedef __attribute__((ext_vector_type(16))) float   f32x16;
typedef __attribute__((ext_vector_type(4))) short    s16x4;
typedef __attribute__((ext_vector_type(8))) short    s16x8;
typedef __attribute__((ext_vector_type(4))) int      i32x4;

using bf16 = __hip_bfloat16;

DEV_INLINE float bf2f(unsigned short u) {
  union { unsigned int i; float f; } v;
  v.i = (unsigned int)u << 16;
  return v.f;
}

DEV_INLINE unsigned short f2bf(float f) {
  union { float f; unsigned int i; } v;
  v.f = f;
  // round-to-nearest-even
  unsigned int lsb = (v.i >> 16) & 1;
  v.i += 0x7fffu + lsb;
  return (unsigned short)(v.i >> 16);
}

// -------------------------------------------------------------- activations
// act codes: 0 identity, 1 tanh, 2 sigmoid, 3 leaky-relu, 4 relu
DEV_INLINE float act_fwd(float x, int act, float slope) {
  switch (act) {
    case 1: return tanhf(x);
    case 2: return 1.0f / (1.0f + __expf(-x));
    case 3: return x > 0.0f ? x : slope * x;
    case 4: return x > 0.0f ? x : 0.0f;
    default: return x;
  }
}

// derivative expressed in terms of the OUTPUT y (valid for all five)
DEV_INLINE float act_bwd_from_y(float y, int act, float slope) {
  switch (act) {
    case 1: return 1.0f - y * y;
    case 2: return y * (1.0f - y);
    case 3: return y > 0.0f ? 1.0f : slope;
    case 4: return y > 0.0f ? 1.0f : 0.0f;
    default: return 1.0f;
  }
}

// ------------------------------------------------------------- reductions
DEV_INLINE float wave_reduce_sum(float v) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_down(v, off, 64);
  return v;  // valid in lane 0
}

DEV_INLINE float wave_reduce_max(float v) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_down(v, off, 64));
  return v;
}

constexpr int ceil_div(int a, int b) { return (a + b - 1) / b; }

// ------------------------------------------------- fast integer division
// Granlund-Montgomery round-up magic; exact for 0 <= n < 2^30, 1 <= d < 2^16.
// mul is 64-bit: for power-of-two (and small) divisors ceil(2^(32+l)/d)
// exceeds 32 bits.
struct FastDiv {
  unsigned long long mul;
  int shift;  // = 32 + ceil(log2 d)
  int d;
};

inline FastDiv make_fastdiv(int d) {
  int l = 0;
  while ((1 << l) < d) ++l;
  FastDiv f;
  f.shift = 32 + l;
  f.mul = ((1ULL << (32 + l)) + (unsigned long long)d - 1) /
          (unsigned long long)d;
  f.d = d;
  return f;
}

DEV_INLINE unsigned fdiv(unsigned n, FastDiv f) {
  return (unsigned)(((unsigned long long)n * f.mul) >> f.shift);
}

DEV_INLINE unsigned fmod_(unsigned n, FastDiv f, unsigned q) {
  return n - q * (unsigned)f.d;
}

// Conv gather geometry for implicit-GEMM staging: maps an im2col
// coordinate (np, k) to an input-image address without materializing col.
// mode 0 (forward):    src pixel = grid_pos * stride - pad + (r,s)
// mode 1 (transposed): src pixel = (grid_pos + pad - (r,s)) / stride,
//                      valid only when divisible (conv dgrad / convT fwd)
// mode 2 (parity class of a stride-2 transposed conv): the GEMM covers one
//   (qh,qw) output-parity class; R,S are the per-class tap counts and
//   (r,s) taps map to src pixel (grid_pos + off - tap); output rows
//   scatter back to the full image via (oH,oW,oqh,oqw) in the epilogue.
//   Every tap is valid (no 4x zero-padding like mode 1 at stride 2).
struct ConvGather {
  int N, H, W, C;        // source image dims (NHWC)
  int Ho, Wo;            // patch grid (mode 2: the class grid)
  int R, S, stride, pad;
  int rsc;               // R*S*C (valid k range; >= rsc is zero padding)
  int mode;
  int off_h, off_w;      // mode 2: ho = hi2 + off - r2
  int oH, oW, oqh, oqw;  // mode 2: output scatter (full dims + class)
  FastDiv fC, fS, fWo, fHo, fStride;
};

#define HIP_CHECK_LAST()                                                     \
  do {                                                                       \
    hipError_t e = hipGetLastError();                                        \
    if (e != hipSuccess) {                                                   \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(e), __FILE__,      \
             __LINE__);                                                      \
    }                                                                        \
  } while (0)
