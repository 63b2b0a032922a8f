// FP8 (OCP e4m3fn — gfx950 native, NOT the MI300X fnuz variant) conv path.
//
// Mixed-precision scheme: bf16 master activations/weights are quantized
// per-tensor (dynamic amax scaling) to e4m3, the conv forward runs on
// mfma_f32_16x16x32_fp8_fp8 with fp32 accumulation, and the epilogue
// de-scales + bias + activation back to bf16. Backward stays bf16.
//
// Same implicit-GEMM structure as the bf16 TN kernel: 128x128x64 tiles,
// double-buffered LDS staged by 16B global_load_lds with the im2col gather
// fused into the source address (zero-page redirect for OOB/pad).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) float f32x8;

#define GLDS16(gsrc, ldst)                                                    \
  __builtin_amdgcn_global_load_lds(                                          \
      (const __attribute__((address_space(1))) unsigned int*)(gsrc),          \
      (__attribute__((address_space(3))) unsigned int*)(ldst), 16, 0, 0)

// ----------------------------------------------------------- quantization
// amax via atomicMax on the positive-float bit pattern
__global__ void amax_bf16(const s16x8* __restrict__ x, long n8,
                          unsigned* __restrict__ amax_bits) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  float m = 0.f;
  for (; i < n8; i += (long)gridDim.x * blockDim.x) {
    s16x8 v = x[i];
    #pragma unroll
    for (int j = 0; j < 8; ++j) m = fmaxf(m, fabsf(bf2f((unsigned short)v[j])));
  }
  m = wave_reduce_max(m);
  if ((threadIdx.x & 63) == 0) {
    union { float f; unsigned u; } c;
    c.f = m;
    atomicMax(amax_bits, c.u);  // positive floats order like their bits
  }
}

// scale = 448/amax (e4m3 max normal 448); inv = 1/scale
__global__ void fp8_make_scale(const unsigned* __restrict__ amax_bits,
                               float* __restrict__ scale,
                               float* __restrict__ inv) {
  union { float f; unsigned u; } c;
  c.u = *amax_bits;
  float a = c.f;
  float s = (a > 1e-20f) ? 448.f / a : 1.f;
  *scale = s;
  *inv = 1.f / s;
}

// bf16 -> e4m3 with scale (device scalar), vectorized by 8
__global__ void quant_fp8(const s16x8* __restrict__ x,
                          unsigned long long* __restrict__ y, long n8,
                          const float* __restrict__ scale) {
  float s = *scale;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n8; i += (long)gridDim.x * blockDim.x) {
    s16x8 v = x[i];
    unsigned lo = 0, hi = 0;
    lo = __builtin_amdgcn_cvt_pk_fp8_f32(bf2f((unsigned short)v[0]) * s,
                                         bf2f((unsigned short)v[1]) * s, lo,
                                         false);
    lo = __builtin_amdgcn_cvt_pk_fp8_f32(bf2f((unsigned short)v[2]) * s,
                                         bf2f((unsigned short)v[3]) * s, lo,
                                         true);
    hi = __builtin_amdgcn_cvt_pk_fp8_f32(bf2f((unsigned short)v[4]) * s,
                                         bf2f((unsigned short)v[5]) * s, hi,
                                         false);
    hi = __builtin_amdgcn_cvt_pk_fp8_f32(bf2f((unsigned short)v[6]) * s,
                                         bf2f((unsigned short)v[7]) * s, hi,
                                         true);
    y[i] = ((unsigned long long)hi << 32) | lo;
  }
}

// bf16 -> e4m3 with the previous step's scale (delayed scaling), fused
// with this tensor's amax reduction so the standalone full-read amax
// pass disappears.  v_cvt_pk_fp8_f32 does NOT saturate — a scaled value
// past +-448 encodes as e4m3fn NaN (0x7f) and poisons the GEMM (measured:
// the second D forward of a GAN step overflowed its first-step scale) —
// so the scaled input is clamped to +-448 explicitly; the updated scale
// applies from the next step (TransformerEngine-style recipe).
__global__ void quant_fp8_delayed(const s16x8* __restrict__ x,
                                  unsigned long long* __restrict__ y, long n8,
                                  const float* __restrict__ scale,
                                  unsigned* __restrict__ amax_bits) {
  float s = *scale;
  float m = 0.f;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n8; i += (long)gridDim.x * blockDim.x) {
    s16x8 v = x[i];
    float f[8];
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float raw = bf2f((unsigned short)v[j]);
      m = fmaxf(m, fabsf(raw));
      f[j] = fminf(fmaxf(raw * s, -448.f), 448.f);  // saturate, never NaN
    }
    unsigned lo = 0, hi = 0;
    lo = __builtin_amdgcn_cvt_pk_fp8_f32(f[0], f[1], lo, false);
    lo = __builtin_amdgcn_cvt_pk_fp8_f32(f[2], f[3], lo, true);
    hi = __builtin_amdgcn_cvt_pk_fp8_f32(f[4], f[5], hi, false);
    hi = __builtin_amdgcn_cvt_pk_fp8_f32(f[6], f[7], hi, true);
    y[i] = ((unsigned long long)hi << 32) | lo;
  }
  m = wave_reduce_max(m);
  if ((threadIdx.x & 63) == 0) {
    union { float f; unsigned u; } c;
    c.f = m;
    atomicMax(amax_bits, c.u);
  }
}

// scale/inv for the NEXT step from the accumulated amax; resets amax
__global__ void fp8_roll_scale(unsigned* __restrict__ amax_bits,
                               float* __restrict__ scale,
                               float* __restrict__ inv) {
  union { float f; unsigned u; } c;
  c.u = *amax_bits;
  float a = c.f;
  float sc = (a > 1e-20f) ? 448.f / a : 1.f;
  *scale = sc;
  *inv = 1.f / sc;
  *amax_bits = 0;
}

// ------------------------------------------------------------ fp8 TN GEMM
// A gathered from an fp8 NHWC image (implicit conv) or plain [M][K];
// B plain fp8 [N][K]. LDS tiles [128 rows][128 k] fp8 = 16 KiB each.
// K depth is 128 to feed gfx950's DOUBLE-RATE fp8 MFMA
// (mfma_scale_f32_16x16x128_f8f6f4 with unit e8m0 scales — there is no
// non-scaled large-K fp8 form; the CDNA3-era 16x16x32_fp8 runs at the
// bf16 FLOP rate, i.e. half the 5 PF fp8 peak).
constexpr int F8_BM = 128, F8_BN = 128, F8_BK = 128;
constexpr int F8_TILE_B = F8_BM * F8_BK;  // bytes (1B/elem)

DEV_INLINE void f8_stage(const unsigned char* __restrict__ g, int row0,
                         int nrows, long ldk, int k0, char* lds) {
  const int t = threadIdx.x;
  const int wid = t >> 6;
  #pragma unroll
  for (int i = 0; i < 4; ++i) {
    int chunk = i * 256 + t;       // 1024 chunks = row*8 + slot16
    int row = chunk >> 3;
    int slot = chunk & 7;
    int gslot = slot ^ (row & 7);
    int grow = min(row0 + row, nrows - 1);
    const unsigned char* src = g + (long)grow * ldk + k0 + gslot * 16;
    char* dst = lds + (i * 256 + wid * 64) * 16;
    GLDS16(src, dst);
  }
}

DEV_INLINE void k_decode_f8(const ConvGather& g, unsigned k, int& r, int& s,
                            int& c) {
  unsigned rs = fdiv(k, g.fC);
  c = (int)(k - rs * g.C);
  unsigned rr = fdiv(rs, g.fS);
  s = (int)(rs - rr * g.S);
  r = (int)rr;
}

// NOTE: 16-fp8 chunks must not straddle (r,s) boundaries -> requires
// C % 16 == 0 (enforced by the binding).
// Row decode hoisted out of the K-loop (same scheme as TnGatherStager in
// gemm.hip): each thread stages the same 4 rows every K-step.
struct F8GatherStager {
  long base[4];      // (long)n * H*W*C element offset per chunk
  int h0[4], w0[4];  // mode-adjusted spatial bases per chunk

  DEV_INLINE void init(const ConvGather& g, int row0, int nrows) {
    const int t = threadIdx.x;
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      int row = (i * 256 + t) >> 3;
      int np = min(row0 + row, nrows - 1);
      unsigned q1 = fdiv((unsigned)np, g.fWo);
      int wo = (int)((unsigned)np - q1 * g.Wo);
      unsigned q2 = fdiv(q1, g.fHo);
      int ho = (int)(q1 - q2 * g.Ho);
      base[i] = (long)(int)q2 * g.H * g.W * g.C;
      if (g.mode == 0) {
        h0[i] = ho * g.stride - g.pad;
        w0[i] = wo * g.stride - g.pad;
      } else if (g.mode == 2) {
        h0[i] = ho + g.off_h;
        w0[i] = wo + g.off_w;
      } else {
        h0[i] = ho + g.pad;
        w0[i] = wo + g.pad;
      }
    }
  }

  DEV_INLINE void stage(const unsigned char* __restrict__ img,
                        const ConvGather& g,
                        const unsigned char* __restrict__ zp, int k0,
                        char* lds) const {
    const int t = threadIdx.x;
    const int wid = t >> 6;
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      int chunk = i * 256 + t;
      int row = chunk >> 3;
      int slot = chunk & 7;
      int gslot = slot ^ (row & 7);
      int k = k0 + gslot * 16;  // 16 fp8 channels per chunk
      const unsigned char* src = zp;
      if (k < g.rsc) {
        int r, s, c;
        k_decode_f8(g, (unsigned)k, r, s, c);
        int hi, wi;
        bool valid;
        if (g.mode == 0) {
          hi = h0[i] + r;
          wi = w0[i] + s;
          valid = hi >= 0 && hi < g.H && wi >= 0 && wi < g.W;
        } else if (g.mode == 2) {
          hi = h0[i] - r;
          wi = w0[i] - s;
          valid = hi >= 0 && hi < g.H && wi >= 0 && wi < g.W;
        } else {
          int hop = h0[i] - r;
          int wop = w0[i] - s;
          if (hop < 0 || wop < 0) {
            valid = false;
            hi = wi = 0;
          } else {
            unsigned qh = fdiv((unsigned)hop, g.fStride);
            unsigned qw = fdiv((unsigned)wop, g.fStride);
            valid = (hop == (int)(qh * g.stride)) &&
                    (wop == (int)(qw * g.stride)) && (int)qh < g.H &&
                    (int)qw < g.W;
            hi = (int)qh;
            wi = (int)qw;
          }
        }
        if (valid) src = img + base[i] + (long)(hi * g.W + wi) * g.C + c;
      }
      char* dst = lds + (i * 256 + wid * 64) * 16;
      GLDS16(src, dst);
    }
  }
};

typedef __attribute__((ext_vector_type(8))) int i32x8;
typedef __attribute__((ext_vector_type(4))) int i32x4;

// fragment = 32 consecutive fp8 (lane's K window kq*32) as two XOR-swizzled
// 16B LDS reads; rows are 128B (8 slots of 16B)
DEV_INLINE i32x8 f8_frag32(const char* lds, int row, int kq) {
  int s0 = (2 * kq) ^ (row & 7);
  int s1 = (2 * kq + 1) ^ (row & 7);
  i32x4 lo = *(const i32x4*)(lds + row * 128 + s0 * 16);
  i32x4 hi = *(const i32x4*)(lds + row * 128 + s1 * 16);
  i32x8 r;
  r[0] = lo[0]; r[1] = lo[1]; r[2] = lo[2]; r[3] = lo[3];
  r[4] = hi[0]; r[5] = hi[1]; r[6] = hi[2]; r[7] = hi[3];
  return r;
}

template <bool GATHER_A>
__global__ __launch_bounds__(256, 2) void gemm_tn_fp8_core(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ B,
    unsigned short* __restrict__ C, const float* __restrict__ bias,
    const float* __restrict__ inv_qa, const float* __restrict__ inv_qb,
    int M, int N, int K, long lda, long ldb, int act, float slope,
    ConvGather ga, const unsigned char* __restrict__ zp) {
  __shared__ __attribute__((aligned(128))) char lds[4 * F8_TILE_B];
  auto abuf = [&](int i) -> char* { return lds + (i ? 2 * F8_TILE_B : 0); };
  auto bbuf = [&](int i) -> char* {
    return lds + F8_TILE_B + (i ? 2 * F8_TILE_B : 0);
  };

  int nwgx = gridDim.x;
  int bidx = blockIdx.x;
  if (nwgx >= 8) {
    int q = nwgx / 8, r = nwgx % 8;
    int xcd = bidx % 8, idx = bidx / 8;
    bidx = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int m0 = bidx * F8_BM;
  const int n0 = blockIdx.y * F8_BN;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wr = wid >> 1, wc = wid & 1;
  const int fr = lane & 15, fq = lane >> 4;

  f32x4 acc[4][4];
  #pragma unroll
  for (int i = 0; i < 4; ++i)
    #pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int ntiles = K / F8_BK;
  F8GatherStager gs;
  if (GATHER_A) {
    gs.init(ga, m0, M);
    gs.stage(A, ga, zp, 0, abuf(0));
  } else {
    f8_stage(A, m0, M, lda, 0, abuf(0));
  }
  f8_stage(B, n0, N, ldb, 0, bbuf(0));
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  for (int t = 0; t < ntiles; ++t) {
    int cur = t & 1;
    if (t + 1 < ntiles) {
      if (GATHER_A)
        gs.stage(A, ga, zp, (t + 1) * F8_BK, abuf(cur ^ 1));
      else
        f8_stage(A, m0, M, lda, (t + 1) * F8_BK, abuf(cur ^ 1));
      f8_stage(B, n0, N, ldb, (t + 1) * F8_BK, bbuf(cur ^ 1));
    }
    const char* Al = abuf(cur);
    const char* Bl = bbuf(cur);
    {
      i32x8 a[4], b[4];
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        a[mi] = f8_frag32(Al, wr * 64 + mi * 16 + fr, fq);
      #pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        b[ni] = f8_frag32(Bl, wc * 64 + ni * 16 + fr, fq);
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          // cbsz=0/blgp=0: both operands e4m3 fp8; e8m0 scale byte 127
          // = x1.0 (per-tensor scaling stays in the epilogue de-scale)
          acc[mi][ni] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
              a[mi], b[ni], acc[mi][ni], 0, 0, 0, 127, 0, 127);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
  }

  // de-scale + bias + act, LDS-staged coalesced stores (N % 8 == 0 assumed
  // for the fast path; scalar fallback otherwise)
  float descale = (*inv_qa) * (*inv_qb);
  if ((N & 7) == 0) {
    unsigned short* ctile = (unsigned short*)lds;
    __syncthreads();
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      #pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        int lc = wc * 64 + ni * 16 + fr;
        float bv = bias != nullptr ? bias[min(n0 + lc, N - 1)] : 0.f;
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          int lr = wr * 64 + mi * 16 + fq * 4 + r;
          ctile[lr * 128 + lc] =
              f2bf(act_fwd(acc[mi][ni][r] * descale + bv, act, slope));
        }
      }
    }
    __syncthreads();
    const int t = threadIdx.x;
    #pragma unroll
    for (int i = 0; i < 8; ++i) {
      int piece = i * 256 + t;
      int row = piece >> 4;
      int seg = piece & 15;
      int grow = m0 + row;
      int gcol = n0 + seg * 8;
      if (grow < M && gcol + 8 <= N) {
        long crow = grow;
        if (GATHER_A && ga.mode == 2) {
          int n2, h2, w2;
          unsigned q1 = fdiv((unsigned)grow, ga.fWo);
          w2 = (int)((unsigned)grow - q1 * ga.Wo);
          unsigned q2 = fdiv(q1, ga.fHo);
          h2 = (int)(q1 - q2 * ga.Ho);
          n2 = (int)q2;
          crow = ((long)n2 * ga.oH + h2 * ga.stride + ga.oqh) * ga.oW +
                 w2 * ga.stride + ga.oqw;
        }
        *(s16x8*)(&C[crow * N + gcol]) =
            *(const s16x8*)(ctile + row * 128 + seg * 8);
      }
    }
    return;
  }
  #pragma unroll
  for (int mi = 0; mi < 4; ++mi)
    #pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int col = n0 + wc * 64 + ni * 16 + fr;
      if (col >= N) continue;
      float bv = bias != nullptr ? bias[col] : 0.f;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wr * 64 + mi * 16 + fq * 4 + r;
        if (row >= M) continue;
        C[(long)row * N + col] =
            f2bf(act_fwd(acc[mi][ni][r] * descale + bv, act, slope));
      }
    }
}

extern "C" {

// 8-phase fp8 pipeline (gemm8p_fp8.hip) — default for large shapes
int gemm_tn_8p_fp8_eligible(int M, int N, int K);
int launch_gemm_tn_8p_fp8(const void* A, const void* B, void* C,
                          const float* bias, const float* inv_qa,
                          const float* inv_qb, int M, int N, int K, long lda,
                          long ldb, int act, float slope, int gather,
                          ConvGather ga, const void* zp, hipStream_t s);

void launch_amax(const void* x, long n, unsigned* amax_bits, hipStream_t s) {
  long n8 = n / 8;
  int grid = (int)min((long)1024, n8 / 256 + 1);
  hipLaunchKernelGGL(amax_bf16, dim3(grid), dim3(256), 0, s, (const s16x8*)x,
                     n8, amax_bits);
}

void launch_fp8_make_scale(const unsigned* amax_bits, float* scale,
                           float* inv, hipStream_t s) {
  hipLaunchKernelGGL(fp8_make_scale, dim3(1), dim3(1), 0, s, amax_bits,
                     scale, inv);
}

void launch_quant_fp8(const void* x, void* y, long n, const float* scale,
                      hipStream_t s) {
  long n8 = n / 8;
  int grid = (int)min((long)2048, n8 / 256 + 1);
  hipLaunchKernelGGL(quant_fp8, dim3(grid), dim3(256), 0, s, (const s16x8*)x,
                     (unsigned long long*)y, n8, scale);
}

void launch_quant_fp8_delayed(const void* x, void* y, long n,
                              const float* scale, unsigned* amax_bits,
                              hipStream_t s) {
  long n8 = n / 8;
  int grid = (int)min((long)2048, n8 / 256 + 1);
  hipLaunchKernelGGL(quant_fp8_delayed, dim3(grid), dim3(256), 0, s,
                     (const s16x8*)x, (unsigned long long*)y, n8, scale,
                     amax_bits);
}

void launch_fp8_roll_scale(unsigned* amax_bits, float* scale, float* inv,
                           hipStream_t s) {
  hipLaunchKernelGGL(fp8_roll_scale, dim3(1), dim3(1), 0, s, amax_bits,
                     scale, inv);
}

void launch_gemm_tn_fp8(const void* A, const void* B, void* C,
                        const float* bias, const float* inv_qa,
                        const float* inv_qb, int M, int N, int K, long lda,
                        long ldb, int act, float slope, int gather,
                        ConvGather ga, const void* zp, hipStream_t s) {
  if (gemm_tn_8p_fp8_eligible(M, N, K)) {
    launch_gemm_tn_8p_fp8(A, B, C, bias, inv_qa, inv_qb, M, N, K, lda, ldb,
                          act, slope, gather, ga, zp, s);
    return;
  }
  dim3 grid(ceil_div(M, F8_BM), ceil_div(N, F8_BN));
  if (gather)
    hipLaunchKernelGGL((gemm_tn_fp8_core<true>), grid, dim3(256), 0, s,
                       (const unsigned char*)A, (const unsigned char*)B,
                       (unsigned short*)C, bias, inv_qa, inv_qb, M, N, K, lda,
                       ldb, act, slope, ga, (const unsigned char*)zp);
  else
    hipLaunchKernelGGL((gemm_tn_fp8_core<false>), grid, dim3(256), 0, s,
                       (const unsigned char*)A, (const unsigned char*)B,
                       (unsigned short*)C, bias, inv_qa, inv_qb, M, N, K, lda,
                       ldb, act, slope, ga, (const unsigned char*)zp);
}

}  // extern "C"
