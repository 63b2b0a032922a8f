// 256x256x128 8-phase deep-pipelined e4m3 MFMA TN GEMM for gfx950.
//
// The fp8 port of gemm8p.hip's counted-vmcnt pipeline: identical LDS
// byte-geometry (an operand K-tile is 256 rows x 128 fp8 = 32 KiB = the
// bf16 kernel's 256 x 64 x 2B; same 32 subtiles of [16 r][4 slot][16 B],
// same staging regions, same glds schedule, same single vmcnt(4) per
// K-tile), with:
//   - fragments of 32 consecutive fp8 per lane = TWO swizzled
//     ds_read_b128 at adjacent slots (slot pair (fq&1)*2 in kblk fq>>1);
//   - 8x mfma_scale_f32_16x16x128_f8f6f4 per phase (one K=128-deep MFMA
//     per fragment pair; unit e8m0 scales, per-tensor descale in the
//     epilogue) — the double-rate fp8 path;
//   - gather granularity of 16 channels per slot (C % 16 == 0).
// See gemm8p.hip for the schedule derivation and correctness argument.

#include "common.h"
#include <stdlib.h>

typedef __attribute__((ext_vector_type(8))) int i32x8;

#define GLDS16(gsrc, ldst)                                                    \
  __builtin_amdgcn_global_load_lds(                                          \
      (const __attribute__((address_space(1))) unsigned int*)(gsrc),          \
      (__attribute__((address_space(3))) unsigned int*)(ldst), 16, 0, 0)

namespace p8f {

constexpr int BM = 256, BN = 256, BK = 128;
constexpr int OPTILE = BM * BK;       // 32 KiB per operand per K-tile
constexpr int BUF = 2 * OPTILE;
constexpr int LDS_B = 2 * BUF;        // 128 KiB

DEV_INLINE int swz(int r) { return (-(r >> 2)) & 3; }

DEV_INLINE int region_blk(int region, int j) {
  if (region & 1)
    return ((j >> 1) << 2) + (j & 1) + ((region >> 1) << 1);
  return ((region & 2) << 1) + (j & 3) + ((j >> 2) << 3);
}

// Stage one region (16 subtiles, 2 glds/thread) of a plain fp8 operand.
// k addresses are in ELEMENTS (1 B each); a slot is 16 elements.
DEV_INLINE void stage(const unsigned char* __restrict__ g, int row0,
                      int nrows, long ldk, int k0, int region,
                      char* op_lds) {
  const int w = threadIdx.x >> 6;
  const int rsub = (threadIdx.x & 63) >> 2;
  const int pslot = threadIdx.x & 3;
  #pragma unroll
  for (int i = 0; i < 2; ++i) {
    int kb = i;
    int blk = region_blk(region, w);
    int row = blk * 16 + rsub;
    int grow = min(row0 + row, nrows - 1);
    int k = k0 + kb * 64 + ((pslot ^ swz(rsub)) << 4);
    const unsigned char* src = g + (long)grow * ldk + k;
    char* dst = op_lds + (((blk << 1) + kb) << 10);
    GLDS16(src, dst);
  }
}

// Implicit-GEMM A staging from an fp8 NHWC image (modes 0/1/2).
struct GatherA {
  long base[2];
  int h0[2], w0[2];

  DEV_INLINE void init(const ConvGather& g, int m0, int M) {
    const int w = threadIdx.x >> 6;
    const int rsub = (threadIdx.x & 63) >> 2;
    #pragma unroll
    for (int q = 0; q < 2; ++q) {
      int blk = region_blk(q ? 0 : 2, w);
      int np = min(m0 + blk * 16 + rsub, M - 1);
      unsigned q1 = fdiv((unsigned)np, g.fWo);
      int wo = (int)((unsigned)np - q1 * g.Wo);
      unsigned q2 = fdiv(q1, g.fHo);
      int ho = (int)(q1 - q2 * g.Ho);
      base[q] = (long)(int)q2 * g.H * g.W * g.C;
      if (g.mode == 0) {
        h0[q] = ho * g.stride - g.pad;
        w0[q] = wo * g.stride - g.pad;
      } else if (g.mode == 2) {
        h0[q] = ho + g.off_h;
        w0[q] = wo + g.off_w;
      } else {
        h0[q] = ho + g.pad;
        w0[q] = wo + g.pad;
      }
    }
  }

  DEV_INLINE void stage(const unsigned char* __restrict__ img,
                        const ConvGather& g,
                        const unsigned char* __restrict__ zp, int k0, int q,
                        char* op_lds) const {
    const int w = threadIdx.x >> 6;
    const int rsub = (threadIdx.x & 63) >> 2;
    const int pslot = threadIdx.x & 3;
    const int blk = region_blk(q ? 0 : 2, w);
    #pragma unroll
    for (int i = 0; i < 2; ++i) {
      int k = k0 + i * 64 + ((pslot ^ swz(rsub)) << 4);
      const unsigned char* src = zp;
      if (k < g.rsc) {
        int r, s, c;
        {
          unsigned rs = fdiv((unsigned)k, g.fC);
          c = (int)((unsigned)k - rs * g.C);
          unsigned rr = fdiv(rs, g.fS);
          s = (int)(rs - rr * g.S);
          r = (int)rr;
        }
        int hi, wi;
        bool valid;
        if (g.mode == 0) {
          hi = h0[q] + r;
          wi = w0[q] + s;
          valid = hi >= 0 && hi < g.H && wi >= 0 && wi < g.W;
        } else if (g.mode == 2) {
          hi = h0[q] - r;
          wi = w0[q] - s;
          valid = hi >= 0 && hi < g.H && wi >= 0 && wi < g.W;
        } else {
          int hop = h0[q] - r;
          int wop = w0[q] - s;
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
        if (valid) src = img + base[q] + (long)(hi * g.W + wi) * g.C + c;
      }
      char* dst = op_lds + (((blk << 1) + i) << 10);
      GLDS16(src, dst);
    }
  }
};

// 32-fp8 fragment: row (rowblk*16 + fr), k = fq*32 .. +32 (within BK=128)
DEV_INLINE i32x8 frag(const char* opb, int rowblk, int fr, int fq) {
  int kb = fq >> 1;
  int s0 = ((fq & 1) << 1) ^ swz(fr);
  int s1 = (((fq & 1) << 1) + 1) ^ swz(fr);
  const char* base = opb + (((rowblk << 1) + kb) << 10) + (fr << 6);
  i32x4 lo = *(const i32x4*)(base + (s0 << 4));
  i32x4 hi = *(const i32x4*)(base + (s1 << 4));
  i32x8 r;
  r[0] = lo[0]; r[1] = lo[1]; r[2] = lo[2]; r[3] = lo[3];
  r[4] = hi[0]; r[5] = hi[1]; r[6] = hi[2]; r[7] = hi[3];
  return r;
}

}  // namespace p8f

template <bool GATHER_A>
__global__ __launch_bounds__(512, 2) void gemm_tn_8p_fp8(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ B,
    unsigned short* __restrict__ C, const float* __restrict__ bias,
    const float* __restrict__ inv_qa, const float* __restrict__ inv_qb,
    int M, int N, int K, long lda, long ldb, int act, float slope,
    ConvGather ga, const unsigned char* __restrict__ zp) {
  using namespace p8f;
  extern __shared__ __attribute__((aligned(16))) char lds[];

  int nwg = gridDim.x * gridDim.y;
  int bid = blockIdx.y * gridDim.x + blockIdx.x;
  if (nwg >= 8) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = bid % 8, idx = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int m0 = (bid % gridDim.x) * BM;
  const int n0 = (bid / gridDim.x) * BN;

  // static young-half priority (measured +4.9% on the bf16 8p template
  // vs per-cluster setprio flips; tools/ab_gemm8p.py)
  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wr = wid >> 2, wc = wid & 3;
  const int fr = lane & 15, fq = lane >> 4;

  f32x4 acc[8][4];
  #pragma unroll
  for (int i = 0; i < 8; ++i)
    #pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int ntiles = K / BK;
  auto abuf = [&](int tt) -> char* { return lds + (tt & 1) * BUF; };
  auto bbuf = [&](int tt) -> char* { return lds + (tt & 1) * BUF + OPTILE; };

  GatherA gs;
  if (GATHER_A) gs.init(ga, m0, M);

  auto stage_a = [&](int tt, int region) {
    int tc = min(tt, ntiles - 1);
    if (GATHER_A)
      gs.stage(A, ga, zp, tc * BK, region == 0 ? 1 : 0, abuf(tc));
    else
      stage(A, m0, M, lda, tc * BK, region, abuf(tc));
  };
  auto stage_b = [&](int tt, int region) {
    int tc = min(tt, ntiles - 1);
    stage(B, n0, N, ldb, tc * BK, region, bbuf(tc));
  };

  stage_a(0, 0);
  stage_b(0, 1);
  stage_a(0, 2);
  stage_b(0, 3);
  asm volatile("" ::: "memory");
  stage_a(1, 0);
  stage_b(1, 1);
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  i32x8 a[4], blo[2], bhi[2];

#define P8F_QUAD(MIH, NIH, BREG)                                             \
  _Pragma("unroll") for (int mi = 0; mi < 4; ++mi)                           \
      _Pragma("unroll") for (int ni = 0; ni < 2; ++ni)                       \
      acc[(MIH)*4 + mi][(NIH)*2 + ni] =                                      \
      __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(                      \
          a[mi], BREG[ni], acc[(MIH)*4 + mi][(NIH)*2 + ni], 0, 0, 0, 127,    \
          0, 127)

#define P8F_BAR_MFMA(MIH, NIH, BREG)                                         \
  __builtin_amdgcn_s_barrier();                                              \
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");                         \
  __builtin_amdgcn_sched_barrier(0);                                         \
  P8F_QUAD(MIH, NIH, BREG);                                                  \
  __builtin_amdgcn_s_barrier()

  for (int t = 0; t < ntiles; ++t) {
    const char* Ab = abuf(t);
    const char* Bb = bbuf(t);
    // phase 1: A(mi0-3) + B(ni0-1); stage R2(t+1)
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi)
      a[mi] = frag(Ab, wr * 8 + mi, fr, fq);
    #pragma unroll
    for (int ni = 0; ni < 2; ++ni)
      blo[ni] = frag(Bb, wc * 4 + ni, fr, fq);
    stage_a(t + 1, 2);
    P8F_BAR_MFMA(0, 0, blo);

    // phase 2: B(ni2-3); stage R3(t+1)
    #pragma unroll
    for (int ni = 0; ni < 2; ++ni)
      bhi[ni] = frag(Bb, wc * 4 + 2 + ni, fr, fq);
    stage_b(t + 1, 3);
    P8F_BAR_MFMA(0, 1, bhi);

    // phase 3: A(mi4-7); stage R0(t+2)
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi)
      a[mi] = frag(Ab, wr * 8 + 4 + mi, fr, fq);
    stage_a(t + 2, 0);
    P8F_BAR_MFMA(1, 0, blo);

    // phase 4: stage R1(t+2); the tile's single counted wait
    stage_b(t + 2, 1);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    P8F_QUAD(1, 1, bhi);
    __builtin_amdgcn_s_barrier();
  }
#undef P8F_BAR_MFMA
#undef P8F_QUAD

  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  // epilogue: de-scale + bias + activation, ctile-staged stores
  float descale = (*inv_qa) * (*inv_qb);
  unsigned short* ctile = (unsigned short*)lds;
  #pragma unroll
  for (int mi = 0; mi < 8; ++mi) {
    #pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int lc = wc * 64 + ni * 16 + fr;
      float bv = bias != nullptr ? bias[min(n0 + lc, N - 1)] : 0.f;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        int lr = wr * 128 + mi * 16 + fq * 4 + r;
        ctile[lr * 256 + lc] =
            f2bf(act_fwd(acc[mi][ni][r] * descale + bv, act, slope));
      }
    }
  }
  __syncthreads();
  const int t = threadIdx.x;
  #pragma unroll
  for (int i = 0; i < 16; ++i) {
    int piece = i * 512 + t;  // 8192 16B pieces = 256 rows x 32 segs
    int row = piece >> 5;
    int seg = piece & 31;
    int grow = m0 + row;
    int gcol = n0 + seg * 8;
    if (grow < M && gcol < N) {
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
      s16x8 v = *(const s16x8*)(ctile + row * 256 + seg * 8);
      if (gcol + 8 <= N) {
        *(s16x8*)(&C[crow * N + gcol]) = v;
      } else {
        for (int j = 0; j < 8 && gcol + j < N; ++j)
          C[crow * N + gcol + j] = (unsigned short)v[j];
      }
    }
  }
}

// ---------------------------------------------------------------------------
extern "C" {

static int p8f_enabled() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("GDLJ_8P_FP8");
    v = (e != nullptr && e[0] == '0') ? 0 : 1;
    if (v) {
      (void)hipFuncSetAttribute(
          reinterpret_cast<const void*>(&gemm_tn_8p_fp8<false>),
          hipFuncAttributeMaxDynamicSharedMemorySize, p8f::LDS_B);
      (void)hipFuncSetAttribute(
          reinterpret_cast<const void*>(&gemm_tn_8p_fp8<true>),
          hipFuncAttributeMaxDynamicSharedMemorySize, p8f::LDS_B);
    }
  }
  return v;
}

int gemm_tn_8p_fp8_eligible(int M, int N, int K) {
  if (!p8f_enabled()) return 0;
  if (M < 32768) return 0;  // large-M regime only (see bf16 8p note)
  if (K % p8f::BK != 0 || K < 2 * p8f::BK) return 0;
  if (N < 192) return 0;
  long blocks = (long)ceil_div(M, p8f::BM) * ceil_div(N, p8f::BN);
  if (blocks < 256) return 0;
  return 1;
}

int launch_gemm_tn_8p_fp8(const void* A, const void* B, void* C,
                          const float* bias, const float* inv_qa,
                          const float* inv_qb, int M, int N, int K, long lda,
                          long ldb, int act, float slope, int gather,
                          ConvGather ga, const void* zp, hipStream_t s) {
  dim3 grid(ceil_div(M, p8f::BM), ceil_div(N, p8f::BN));
  if (gather)
    hipLaunchKernelGGL((gemm_tn_8p_fp8<true>), grid, dim3(512), p8f::LDS_B,
                       s, (const unsigned char*)A, (const unsigned char*)B,
                       (unsigned short*)C, bias, inv_qa, inv_qb, M, N, K,
                       lda, ldb, act, slope, ga, (const unsigned char*)zp);
  else
    hipLaunchKernelGGL((gemm_tn_8p_fp8<false>), grid, dim3(512), p8f::LDS_B,
                       s, (const unsigned char*)A, (const unsigned char*)B,
                       (unsigned short*)C, bias, inv_qa, inv_qb, M, N, K,
                       lda, ldb, act, slope, ga, (const unsigned char*)zp);
  return (int)grid.x;
}

}  // extern "C"
