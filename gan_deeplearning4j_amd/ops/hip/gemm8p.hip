// 256x256x64 8-phase deep-pipelined bf16 MFMA TN GEMM for gfx950 (CDNA4).
//
// This is the round-2 rewrite of the GEMM inner loop: the 128x128 2-barrier
// K-loop (gemm.hip) ceilings at ~45% MfmaUtil because its stage drains
// vmcnt(0) at every barrier.  Here the K-step is split into 4 phases per
// K-tile (8 per double-tile iteration), each phase
//   { ds_read register subtile | issue one 16-subtile staging region via
//     global_load_lds | raw s_barrier | s_waitcnt lgkmcnt(0) | setprio(1)
//     16x mfma_f32_16x16x32_bf16 | setprio(0) | raw s_barrier }
// with a single counted s_waitcnt vmcnt(4) per K-tile (phase 4), so staging
// loads stay in flight ACROSS barriers (guide T3+T4).  Raw
// __builtin_amdgcn_s_barrier() everywhere in the main loop: __syncthreads()
// would drain the in-flight LDS-DMA (vmcnt(0)) and serialize the pipeline.
//
// Geometry: BM=BN=256, BK=64, 512 threads = 8 waves as 2(M) x 4(N); each
// wave owns a 128x64 C panel = 8x4 fragments of 16x16, acc = 128 VGPRs.
// LDS = 128 KiB: two K-tile buffers x (A 32 KiB + B 32 KiB); one block/CU.
//
// LDS image: per operand tile, 32 subtiles [rowblk 0..15][kblk 0..1] of
// 1 KiB = [16 rows][4 slots][16 B]; a slot holds 8 consecutive bf16 of k.
// Swizzle (T2, both-sides rule 21): physical slot = kslot ^ swz(row&15),
// applied to the glds SOURCE address (LDS destination stays lane-linear,
// dst byte = lane*16) and to the ds_read_b128 fragment address.
// swz(r) = (-(r>>2))&3 spreads each of ds_read_b128's 16-lane groups
// ({0-3,12-15,20-27} etc., MI355X_MICROARCH §LDS) over 16 distinct 16-B
// bank slots: conflict-free by construction (SWZ=2); SWZ=1 is the guide's
// st_16x32 single-bit variant, SWZ=0 linear (A/B-measurable).
//
// Staging schedule (regions of 16 subtiles = 16 KiB, 2 glds/thread each):
//   R0 = A rowblks {0-3,8-11}   (the mi0-3 half of both wave rows)
//   R1 = B rowblks {0,1,4,5,8,9,12,13}  (the ni0-1 half of all wave cols)
//   R2 = A rowblks {4-7,12-15}, R3 = B complement.
// Phase p of K-tile t computes quadrant  p1:(mi0-3,ni0-1) p2:(mi0-3,ni2-3)
// p3:(mi4-7,ni0-1) p4:(mi4-7,ni2-3), so R0 is dead after p2, R1 after p3,
// R2/R3 after p4; stages issue at the earliest legal phase:
//   p1 -> R2(t+1)   p2 -> R3(t+1)   p3 -> R0(t+2)   p4 -> R1(t+2)
// (a region staged at phase p overwrites data last ds_read in phase p-1;
// every wave's phase-(p-1) reads completed before its own lgkmcnt(0) ->
// MFMA -> trailing barrier, so the write cannot race a read).  The single
// vmcnt(4) at p4 leaves exactly R0(t+2)+R1(t+2) (4 glds) in flight and
// guarantees all of tile t+1 has landed before its first ds_read.  Tail
// tiles clamp to ntiles-1 and overwrite dead regions so the glds count per
// phase is constant and the vmcnt immediates stay exact.
//
// GATHER_A = implicit-GEMM convolution (same ConvGather modes as gemm.hip:
// 0 fwd, 1 transposed, 2 parity-class with output scatter); B (weights) is
// always a plain K-contiguous operand.
//
// Replaces the libnd4j/cuDNN GEMM-conv dependency surface of the reference
// (SURVEY.md §2.2-2.3); no reference counterpart file exists.

#include "common.h"
#include <stdlib.h>
#include <stdio.h>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

#define GLDS16(gsrc, ldst)                                                    \
  __builtin_amdgcn_global_load_lds(                                          \
      (const __attribute__((address_space(1))) unsigned int*)(gsrc),          \
      (__attribute__((address_space(3))) unsigned int*)(ldst), 16, 0, 0)

namespace p8 {

constexpr int BM = 256, BN = 256, BK = 64;
constexpr int OPTILE = BM * BK * 2;   // 32 KiB per operand per K-tile
constexpr int BUF = 2 * OPTILE;       // A+B per K-tile
constexpr int LDS_B = 2 * BUF;        // 128 KiB total

template <int SWZ>
DEV_INLINE int swz(int r) {
  if (SWZ == 0) return 0;
  if (SWZ == 1) return ((r >> 3) & 1) << 1;  // st_16x32
  return (-(r >> 2)) & 3;                    // b128-group conflict-free
}

// rowblk of subtile j (0..7) within a staging region
DEV_INLINE int region_blk(int region, int j) {
  if (region & 1)  // B regions: pairs {0,1},{4,5},... (+2 for R3)
    return ((j >> 1) << 2) + (j & 1) + ((region >> 1) << 1);
  // A regions: {0-3, 8-11} (+4 for R2)
  return ((region & 2) << 1) + (j & 3) + ((j >> 2) << 3);
}

// Stage one region (16 subtiles, 2 glds per thread) of a plain operand.
template <int SWZ>
DEV_INLINE void stage(const unsigned short* __restrict__ g, int row0,
                      int nrows, long ldk, int k0, int region,
                      char* op_lds) {
  const int w = threadIdx.x >> 6;
  const int rsub = (threadIdx.x & 63) >> 2;
  const int pslot = threadIdx.x & 3;
  #pragma unroll
  for (int i = 0; i < 2; ++i) {
    int kb = i;                       // subtile s = w*2+i: j = w, kb = i
    int blk = region_blk(region, w);
    int row = blk * 16 + rsub;
    int grow = min(row0 + row, nrows - 1);
    int k = k0 + kb * 32 + ((pslot ^ swz<SWZ>(rsub)) << 3);
    const unsigned short* src = g + (long)grow * ldk + k;
    char* dst = op_lds + (((blk << 1) + kb) << 10);
    GLDS16(src, dst);
  }
}

// Implicit-GEMM A-staging: rows are im2col rows gathered from the NHWC
// image.  Each thread stages the SAME two rows every K-step (one per
// region 2 / region 0), so the np->(n,ho,wo) decode is hoisted once.
template <int SWZ>
struct GatherA {
  long base[2];      // n-plane element offset       [q: 0 = R2, 1 = R0]
  int h0[2], w0[2];  // mode-adjusted spatial bases

  DEV_INLINE void init(const ConvGather& g, int m0, int M) {
    const int w = threadIdx.x >> 6;
    const int rsub = (threadIdx.x & 63) >> 2;
    #pragma unroll
    for (int q = 0; q < 2; ++q) {
      int blk = region_blk(q ? 0 : 2, w);
      int np = min(m0 + blk * 16 + rsub, M - 1);
      int n, ho, wo;
      {
        unsigned q1 = fdiv((unsigned)np, g.fWo);
        wo = (int)((unsigned)np - q1 * g.Wo);
        unsigned q2 = fdiv(q1, g.fHo);
        ho = (int)(q1 - q2 * g.Ho);
        n = (int)q2;
      }
      base[q] = (long)n * g.H * g.W * g.C;
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

  // q: 0 when staging region 2, 1 when staging region 0
  DEV_INLINE void stage(const unsigned short* __restrict__ img,
                        const ConvGather& g,
                        const unsigned short* __restrict__ zp, int k0, int q,
                        char* op_lds) const {
    const int w = threadIdx.x >> 6;
    const int rsub = (threadIdx.x & 63) >> 2;
    const int pslot = threadIdx.x & 3;
    const int blk = region_blk(q ? 0 : 2, w);
    #pragma unroll
    for (int i = 0; i < 2; ++i) {
      int k = k0 + i * 32 + ((pslot ^ swz<SWZ>(rsub)) << 3);
      const unsigned short* src = zp;
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

// MFMA fragment read: row (rowblk*16 + fr), k = kc*32 + fq*8 .. +8
template <int SWZ>
DEV_INLINE bf16x8 frag(const char* opb, int rowblk, int kc, int fr, int fq) {
  return *(const bf16x8*)(opb + (((rowblk << 1) + kc) << 10) + (fr << 6) +
                          (((fq ^ swz<SWZ>(fr)) & 3) << 4));
}

}  // namespace p8

// DEEP: two counted waits per K-tile (phase-1 tail + phase 4) at depth
// vmcnt(6) = 3 regions in flight, instead of one vmcnt(4) wait at depth 2.
// The extra wait is where each next-needed region's landing is forced at
// the latest legal point, so 50% more staging latency is hidden.
template <bool GATHER_A, int SWZ, bool DEEP>
__global__ __launch_bounds__(512, 2) void gemm_tn_8p(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ B,
    unsigned short* __restrict__ C, const float* __restrict__ bias, int M,
    int N, int K, long lda, long ldb, int act, float slope, ConvGather ga,
    const unsigned short* __restrict__ zp) {
  using namespace p8;
  extern __shared__ __attribute__((aligned(16))) char lds[];

  // XCD-aware bijective remap over the whole grid (T1)
  int nwg = gridDim.x * gridDim.y;
  int bid = blockIdx.y * gridDim.x + blockIdx.x;
  if (nwg >= 8) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = bid % 8, idx = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int m0 = (bid % gridDim.x) * BM;
  const int n0 = (bid / gridDim.x) * BN;

  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wr = wid >> 2, wc = wid & 3;  // 2(m) x 4(n) wave grid
  const int fr = lane & 15, fq = lane >> 4;

  f32x4 acc[8][4];
  #pragma unroll
  for (int i = 0; i < 8; ++i)
    #pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int ntiles = K / BK;
  auto abuf = [&](int tt) -> char* { return lds + (tt & 1) * BUF; };
  auto bbuf = [&](int tt) -> char* { return lds + (tt & 1) * BUF + OPTILE; };

  GatherA<SWZ> gs;
  if (GATHER_A) gs.init(ga, m0, M);

  auto stage_a = [&](int tt, int region) {
    int tc = min(tt, ntiles - 1);
    if (GATHER_A)
      gs.stage(A, ga, zp, tc * BK, region == 0 ? 1 : 0, abuf(tc));
    else
      stage<SWZ>(A, m0, M, lda, tc * BK, region, abuf(tc));
  };
  auto stage_b = [&](int tt, int region) {
    int tc = min(tt, ntiles - 1);
    stage<SWZ>(B, n0, N, ldb, tc * BK, region, bbuf(tc));
  };

  // prologue: all of tile 0, then R0+R1 of tile 1 (12 glds / thread);
  // wait leaves R0(1)+R1(1) = 4 glds in flight (the steady-state queue).
  stage_a(0, 0);
  stage_b(0, 1);
  stage_a(0, 2);
  stage_b(0, 3);
  // ordering fence: tile-0 glds must be OLDER than tile-1's so the
  // counted vmcnt below guards exactly the tile-1 pair
  asm volatile("" ::: "memory");
  stage_a(1, 0);
  stage_b(1, 1);
  if (DEEP)
    asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  else
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  bf16x8 a[4][2], blo[2][2], bhi[2][2];

#define P8_QUAD(MIH, NIH, BREG)                                              \
  _Pragma("unroll") for (int kc = 0; kc < 2; ++kc)                           \
      _Pragma("unroll") for (int mi = 0; mi < 4; ++mi)                       \
      _Pragma("unroll") for (int ni = 0; ni < 2; ++ni)                       \
      acc[(MIH)*4 + mi][(NIH)*2 + ni] =                                      \
      __builtin_amdgcn_mfma_f32_16x16x32_bf16(                               \
          a[mi][kc], BREG[ni][kc], acc[(MIH)*4 + mi][(NIH)*2 + ni], 0, 0, 0)

#define P8_BAR_MFMA(MIH, NIH, BREG, TAILWAIT)                                \
  __builtin_amdgcn_s_barrier();                                              \
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");                         \
  __builtin_amdgcn_sched_barrier(0);                                         \
  __builtin_amdgcn_s_setprio(1);                                             \
  P8_QUAD(MIH, NIH, BREG);                                                   \
  __builtin_amdgcn_s_setprio(0);                                             \
  TAILWAIT;                                                                  \
  __builtin_amdgcn_s_barrier()

  for (int t = 0; t < ntiles; ++t) {
    const char* Ab = abuf(t);
    const char* Bb = bbuf(t);
    // phase 1: A(mi0-3) + B(ni0-1) reads; stage R2(t+1)
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      a[mi][0] = frag<SWZ>(Ab, wr * 8 + mi, 0, fr, fq);
      a[mi][1] = frag<SWZ>(Ab, wr * 8 + mi, 1, fr, fq);
    }
    #pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      blo[ni][0] = frag<SWZ>(Bb, wc * 4 + ni, 0, fr, fq);
      blo[ni][1] = frag<SWZ>(Bb, wc * 4 + ni, 1, fr, fq);
    }
    stage_a(t + 1, 2);
    // DEEP: force R3(t) (phase-2's bhi source) landed at this phase's tail
    P8_BAR_MFMA(0, 0, blo,
                if (DEEP) asm volatile("s_waitcnt vmcnt(6)" ::: "memory"));

    // phase 2: B(ni2-3) reads; stage R3(t+1)
    #pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      bhi[ni][0] = frag<SWZ>(Bb, wc * 4 + 2 + ni, 0, fr, fq);
      bhi[ni][1] = frag<SWZ>(Bb, wc * 4 + 2 + ni, 1, fr, fq);
    }
    stage_b(t + 1, 3);
    P8_BAR_MFMA(0, 1, bhi, );

    // phase 3: A(mi4-7) reads; stage R0(t+2)
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      a[mi][0] = frag<SWZ>(Ab, wr * 8 + 4 + mi, 0, fr, fq);
      a[mi][1] = frag<SWZ>(Ab, wr * 8 + 4 + mi, 1, fr, fq);
    }
    stage_a(t + 2, 0);
    P8_BAR_MFMA(1, 0, blo, );

    // phase 4: no reads; stage R1(t+2); the tile's single counted wait
    stage_b(t + 2, 1);
    if (DEEP)
      asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
    P8_QUAD(1, 1, bhi);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
  }
#undef P8_BAR_MFMA
#undef P8_QUAD

  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  // epilogue: bias + activation, LDS-staged coalesced stores
  // (ctile [256][256] bf16 = the whole 128 KiB LDS)
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
        ctile[lr * 256 + lc] = f2bf(act_fwd(acc[mi][ni][r] + bv, act, slope));
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
        int n, h2, w2;
        unsigned q1 = fdiv((unsigned)grow, ga.fWo);
        w2 = (int)((unsigned)grow - q1 * ga.Wo);
        unsigned q2 = fdiv(q1, ga.fHo);
        h2 = (int)(q1 - q2 * ga.Ho);
        n = (int)q2;
        crow = ((long)n * ga.oH + h2 * ga.stride + ga.oqh) * ga.oW +
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

static int p8_swz_mode() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("GDLJ_8P_SWZ");
    v = (e != nullptr) ? atoi(e) : 2;
    if (v < 0 || v > 2) v = 2;
  }
  return v;
}

static int p8_deep() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("GDLJ_8P_DEEP");
    v = (e != nullptr && e[0] == '1') ? 1 : 0;
  }
  return v;
}

static int p8_enabled() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("GDLJ_8P");
    v = (e != nullptr && e[0] == '0') ? 0 : 1;
    if (v) {
      // 128 KiB dynamic LDS needs an explicit opt-in per kernel
      #define P8_SETATTR(KF)                                                 \
        (void)hipFuncSetAttribute(reinterpret_cast<const void*>(&KF),        \
                                  hipFuncAttributeMaxDynamicSharedMemorySize,\
                                  p8::LDS_B)
      P8_SETATTR((gemm_tn_8p<false, 0, false>));
      P8_SETATTR((gemm_tn_8p<false, 1, false>));
      P8_SETATTR((gemm_tn_8p<false, 2, false>));
      P8_SETATTR((gemm_tn_8p<true, 0, false>));
      P8_SETATTR((gemm_tn_8p<true, 1, false>));
      P8_SETATTR((gemm_tn_8p<true, 2, false>));
      P8_SETATTR((gemm_tn_8p<false, 2, true>));
      P8_SETATTR((gemm_tn_8p<true, 2, true>));
      #undef P8_SETATTR
    }
  }
  return v;
}

// Eligibility: bf16 out, K deep enough to amortize the pipeline prologue
// (>= 4 K-tiles; at K = 128 the 2-block/CU 128-tile kernel wins, measured
// profiles/gemm8p_ab.md), N wide enough that the 256-col tile isn't mostly
// padding, and >= 256 blocks so every CU gets work at one block/CU.
int gemm_tn_8p_eligible(int M, int N, int K) {
  if (!p8_enabled()) return 0;
  if (K % p8::BK != 0 || K < 4 * p8::BK) return 0;
  if (N < 192) return 0;
  long blocks = (long)ceil_div(M, p8::BM) * ceil_div(N, p8::BN);
  if (blocks < 256) return 0;
  return 1;
}

int launch_gemm_tn_8p(const void* A, const void* B, void* C,
                      const float* bias, int M, int N, int K, long lda,
                      long ldb, int act, float slope, int gather,
                      ConvGather ga, const void* zp, hipStream_t s) {
  dim3 grid(ceil_div(M, p8::BM), ceil_div(N, p8::BN));
  dim3 blk(512);
  int swzm = p8_swz_mode();
  #define P8_LAUNCH(G, S, D)                                                 \
    hipLaunchKernelGGL((gemm_tn_8p<G, S, D>), grid, blk, p8::LDS_B, s,       \
                       (const unsigned short*)A, (const unsigned short*)B,   \
                       (unsigned short*)C, bias, M, N, K, lda, ldb, act,     \
                       slope, ga, (const unsigned short*)zp)
  if (p8_deep()) {
    if (gather) P8_LAUNCH(true, 2, true);
    else P8_LAUNCH(false, 2, true);
  } else if (gather) {
    if (swzm == 0) P8_LAUNCH(true, 0, false);
    else if (swzm == 1) P8_LAUNCH(true, 1, false);
    else P8_LAUNCH(true, 2, false);
  } else {
    if (swzm == 0) P8_LAUNCH(false, 0, false);
    else if (swzm == 1) P8_LAUNCH(false, 1, false);
    else P8_LAUNCH(false, 2, false);
  }
  #undef P8_LAUNCH
  return (int)grid.x;
}

}  // extern "C"
