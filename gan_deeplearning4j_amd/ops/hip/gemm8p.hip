// 8-phase deep-pipelined bf16 MFMA TN GEMM for gfx950 (CDNA4).
//
// This is the round-2 rewrite of the GEMM inner loop: the 128x128 2-barrier
// K-loop (gemm.hip) ceilings at ~45% MfmaUtil because its stage drains
// vmcnt(0) at every barrier.  Here the K-step is split into 4 phases per
// K-tile, each phase
//   { ds_read register subtile | issue one staging region via
//     global_load_lds | raw s_barrier | s_waitcnt lgkmcnt(0) | setprio(1)
//     MFMA quadrant | setprio(0) | raw s_barrier }
// with a single counted s_waitcnt per K-tile (phase 4), so staging loads
// stay in flight ACROSS barriers (guide T3+T4).  Raw
// __builtin_amdgcn_s_barrier() everywhere in the main loop: __syncthreads()
// would drain the in-flight LDS-DMA (vmcnt(0)) and serialize the pipeline.
//
// Two tile shapes share the template (BNT = tile N):
//   BNT=256: 8 waves as 2(M) x 4(N); wave = 128x64 C panel, 8x4 fragments,
//            acc 128 VGPRs; LDS 128 KiB; 16 MFMA / phase; vmcnt(4).
//   BNT=128: 8 waves as 4(M) x 2(N); wave = 64x64, 4x4 fragments, acc 64
//            VGPRs; LDS 96 KiB; 8 MFMA / phase; vmcnt(3) (B regions are
//            half-size: 1 glds per thread). Covers the N=128 conv family
//            (conv2-class fwd, G-side dgrads) the 256-wide tile wastes.
//
// LDS image: per operand tile, subtiles [rowblk][kblk 0..1] of 1 KiB =
// [16 rows][4 slots][16 B]; a slot holds 8 consecutive bf16 of k.
// Swizzle (T2, both-sides rule 21): physical slot = kslot ^ swz(row&15),
// applied to the glds SOURCE address (LDS destination stays lane-linear)
// and to the ds_read_b128 fragment address. swz(r) = (-(r>>2))&3 spreads
// each of ds_read_b128's 16-lane groups over 16 distinct 16-B bank slots:
// conflict-free by construction (SWZ=2); SWZ=1 is the guide's st_16x32
// single-bit variant, SWZ=0 linear (A/B-measured in profiles/gemm8p_ab.md).
//
// Staging schedule: per K-tile, 4 regions — R0/R2 = the A rows quadrant
// phases 1-2 / 3-4 read, R1/R3 = the B rows the ni-halves read.  Phase p
// of K-tile t computes quadrant p1:(miH0,niH0) p2:(miH0,niH1) p3:(miH1,
// niH0) p4:(miH1,niH1), so R0 is dead after p2, R1 after p3, R2/R3 after
// p4; stages issue at the earliest legal phase:
//   p1 -> R2(t+1)   p2 -> R3(t+1)   p3 -> R0(t+2)   p4 -> R1(t+2)
// (a region staged at phase p overwrites data last ds_read in phase p-1;
// every wave's phase-(p-1) reads completed before its own lgkmcnt(0) ->
// MFMA -> trailing barrier, so the write cannot race a read).  The single
// counted wait at p4 leaves exactly R0(t+2)+R1(t+2) glds in flight and
// guarantees all of tile t+1 has landed before its first ds_read.  Tail
// tiles clamp to ntiles-1 and overwrite dead regions with identical data
// so the glds count per phase is constant and the vmcnt immediates exact.
//
// GATHER_A = implicit-GEMM convolution (ConvGather modes 0 fwd /
// 1 transposed / 2 parity-class with output scatter); B (weights) is
// always a plain K-contiguous operand.
//
// DEEP (BNT=256 only): two counted waits per K-tile at depth 3 regions —
// measured neutral-to-negative (profiles/gemm8p_ab.md), kept env-gated.
//
// Replaces the libnd4j/cuDNN GEMM-conv dependency surface of the reference
// (SURVEY.md §2.2-2.3); no reference counterpart file exists.

#include "common.h"
#include <stdlib.h>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

#define GLDS16(gsrc, ldst)                                                    \
  __builtin_amdgcn_global_load_lds(                                          \
      (const __attribute__((address_space(1))) unsigned int*)(gsrc),          \
      (__attribute__((address_space(3))) unsigned int*)(ldst), 16, 0, 0)

namespace p8 {

constexpr int BM = 256, BK = 64;
constexpr int OPTA = BM * BK * 2;  // 32 KiB A per K-tile

template <int SWZ>
DEV_INLINE int swz(int r) {
  if (SWZ == 0) return 0;
  if (SWZ == 1) return ((r >> 3) & 1) << 1;  // st_16x32
  return (-(r >> 2)) & 3;                    // b128-group conflict-free
}

// A-region rowblk: the 8 blks quadrant half MIH reads across all waves.
// BNT=256 (2 M-waves, 128 rows each): blk = wr*8 + MIH*4 + idx
// BNT=128 (4 M-waves,  64 rows each): blk = wr*4 + MIH*2 + idx
template <int BNT>
DEV_INLINE int ablk(int mih, int j) {
  constexpr int h = (BNT == 256) ? 4 : 2;  // blks per wave-half
  return (j / h) * (2 * h) + mih * h + (j % h);
}

// B-region rowblk: every wave owns 4 blks (64 cols); a region is the
// ni-half = 2 blks per wave: blk = wc*4 + NIH*2 + idx.
DEV_INLINE int bblk(int nih, int j) {
  return (j >> 1) * 4 + nih * 2 + (j & 1);
}

// Stage one A region (8 subtiles x 2 kblk, 2 glds per thread).
template <int SWZ, int BNT>
DEV_INLINE void stage_a_plain(const unsigned short* __restrict__ g, int row0,
                              int nrows, long ldk, int k0, int mih,
                              char* op_lds) {
  const int w = threadIdx.x >> 6;
  const int rsub = (threadIdx.x & 63) >> 2;
  const int pslot = threadIdx.x & 3;
  #pragma unroll
  for (int i = 0; i < 2; ++i) {
    int blk = ablk<BNT>(mih, w);
    int row = blk * 16 + rsub;
    int grow = min(row0 + row, nrows - 1);
    int k = k0 + i * 32 + ((pslot ^ swz<SWZ>(rsub)) << 3);
    const unsigned short* src = g + (long)grow * ldk + k;
    char* dst = op_lds + (((blk << 1) + i) << 10);
    GLDS16(src, dst);
  }
}

// Stage one B region: BNT=256 -> 8 blks (2 glds/thread);
// BNT=128 -> 4 blks (1 glds/thread: wave w covers (blk j=w>>1, kb=w&1)).
template <int SWZ, int BNT>
DEV_INLINE void stage_b_plain(const unsigned short* __restrict__ g, int row0,
                              int nrows, long ldk, int k0, int nih,
                              char* op_lds) {
  const int w = threadIdx.x >> 6;
  const int rsub = (threadIdx.x & 63) >> 2;
  const int pslot = threadIdx.x & 3;
  if (BNT == 256) {
    #pragma unroll
    for (int i = 0; i < 2; ++i) {
      int blk = bblk(nih, w);
      int row = blk * 16 + rsub;
      int grow = min(row0 + row, nrows - 1);
      int k = k0 + i * 32 + ((pslot ^ swz<SWZ>(rsub)) << 3);
      const unsigned short* src = g + (long)grow * ldk + k;
      char* dst = op_lds + (((blk << 1) + i) << 10);
      GLDS16(src, dst);
    }
  } else {
    int blk = bblk(nih, w >> 1);
    int kb = w & 1;
    int row = blk * 16 + rsub;
    int grow = min(row0 + row, nrows - 1);
    int k = k0 + kb * 32 + ((pslot ^ swz<SWZ>(rsub)) << 3);
    const unsigned short* src = g + (long)grow * ldk + k;
    char* dst = op_lds + (((blk << 1) + kb) << 10);
    GLDS16(src, dst);
  }
}

// Implicit-GEMM A-staging: rows gathered from the NHWC image.  Each
// thread stages the SAME two rows every K-step (one per A region), so
// the np->(n,ho,wo) decode is hoisted once.
template <int SWZ, int BNT>
struct GatherA {
  long base[2];      // n-plane element offset       [q = region MIH]
  int h0[2], w0[2];  // mode-adjusted spatial bases

  DEV_INLINE void init(const ConvGather& g, int m0, int M) {
    const int w = threadIdx.x >> 6;
    const int rsub = (threadIdx.x & 63) >> 2;
    #pragma unroll
    for (int q = 0; q < 2; ++q) {
      int blk = ablk<BNT>(q, w);
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

  DEV_INLINE void stage(const unsigned short* __restrict__ img,
                        const ConvGather& g,
                        const unsigned short* __restrict__ zp, int k0, int q,
                        char* op_lds) const {
    const int w = threadIdx.x >> 6;
    const int rsub = (threadIdx.x & 63) >> 2;
    const int pslot = threadIdx.x & 3;
    const int blk = ablk<BNT>(q, w);
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

// TWEAK bits (within-probe A/B candidates, tools/ab_gemm8p.py):
//   1 = partial s_waitcnt lgkmcnt(8) before phase-1's first barrier
//   2 = static young-half setprio instead of per-cluster flips (T5 static)
//   4 = n-major XCD grid decomposition (B-panel L2 affinity)
//   16 = prefetch the B ni-half-1 fragments in phase 1 (16/0/8/0 reads)
// TWEAK=2 (static young-half priority, no per-cluster flips) measured
// +4.9% within-probe over the flip form at square-4k (tools/ab_gemm8p.py:
// 1188 -> 1247 TF median of 12 interleaved rounds; n-major decomposition
// -2.2%, partial lgkm +0.5% noise) — it is the production default below.
template <bool GATHER_A, int SWZ, bool DEEP, int BNT = 256, int TWEAK = 2>
__global__ __launch_bounds__(512, 2) void gemm_tn_8p(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ B,
    unsigned short* __restrict__ C, const float* __restrict__ bias, int M,
    int N, int K, long lda, long ldb, int act, float slope, ConvGather ga,
    const unsigned short* __restrict__ zp) {
  using namespace p8;
  constexpr int OPTB = BNT * BK * 2;
  constexpr int BUF = OPTA + OPTB;
  constexpr int MI = (BNT == 256) ? 8 : 4;   // per-wave M fragments
  constexpr int WCN = (BNT == 256) ? 4 : 2;  // waves along N
  extern __shared__ __attribute__((aligned(16))) char lds[];

  // XCD-aware bijective remap over the whole grid (T1)
  int nwg = gridDim.x * gridDim.y;
  int bid = blockIdx.y * gridDim.x + blockIdx.x;
  if (nwg >= 8) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = bid % 8, idx = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int m0 = (TWEAK & 4) ? (bid / gridDim.y) * BM
                             : (bid % gridDim.x) * BM;
  const int n0 = (TWEAK & 4) ? (bid % gridDim.y) * BNT
                             : (bid / gridDim.x) * BNT;

  if (TWEAK & 2) {
    // one-time priority for the younger dispatch half (waves 4-7);
    // condition must be provably wave-uniform (readfirstlane) or the
    // s_setprio is emitted unconditionally under exec masking
    if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
      __builtin_amdgcn_s_setprio(1);
  }

  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wr = wid / WCN, wc = wid % WCN;
  const int fr = lane & 15, fq = lane >> 4;

  f32x4 acc[MI][4];
  #pragma unroll
  for (int i = 0; i < MI; ++i)
    #pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int ntiles = K / BK;
  auto abuf = [&](int tt) -> char* { return lds + (tt & 1) * BUF; };
  auto bbuf = [&](int tt) -> char* { return lds + (tt & 1) * BUF + OPTA; };

  GatherA<SWZ, BNT> gs;
  if (GATHER_A) gs.init(ga, m0, M);

  auto stage_a = [&](int tt, int mih) {
    int tc = min(tt, ntiles - 1);
    if (GATHER_A)
      gs.stage(A, ga, zp, tc * BK, mih, abuf(tc));
    else
      stage_a_plain<SWZ, BNT>(A, m0, M, lda, tc * BK, mih, abuf(tc));
  };
  auto stage_b = [&](int tt, int nih) {
    int tc = min(tt, ntiles - 1);
    stage_b_plain<SWZ, BNT>(B, n0, N, ldb, tc * BK, nih, bbuf(tc));
  };

  // prologue: all of tile 0, then R0+R1 of tile 1; wait leaves the
  // tile-1 pair in flight (the steady-state queue).
  stage_a(0, 0);
  stage_b(0, 0);
  stage_a(0, 1);
  stage_b(0, 1);
  // ordering fence: tile-0 glds must be OLDER than tile-1's so the
  // counted vmcnt below guards exactly the tile-1 pair
  asm volatile("" ::: "memory");
  stage_a(1, 0);
  stage_b(1, 0);
  if (BNT == 256) {
    if (DEEP)
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();

  bf16x8 a[MI / 2][2], blo[2][2], bhi[2][2];

#define P8_QUAD(MIH, NIH, BREG)                                              \
  _Pragma("unroll") for (int kc = 0; kc < 2; ++kc)                           \
      _Pragma("unroll") for (int mi = 0; mi < MI / 2; ++mi)                  \
      _Pragma("unroll") for (int ni = 0; ni < 2; ++ni)                       \
      acc[(MIH) * (MI / 2) + mi][(NIH)*2 + ni] =                             \
      __builtin_amdgcn_mfma_f32_16x16x32_bf16(                               \
          a[mi][kc], BREG[ni][kc], acc[(MIH) * (MI / 2) + mi][(NIH)*2 + ni], \
          0, 0, 0)

#define P8_BAR_MFMA(MIH, NIH, BREG, TAILWAIT)                                \
  __builtin_amdgcn_s_barrier();                                              \
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");                         \
  __builtin_amdgcn_sched_barrier(0);                                         \
  if (!(TWEAK & 2)) __builtin_amdgcn_s_setprio(1);                           \
  P8_QUAD(MIH, NIH, BREG);                                                   \
  if (!(TWEAK & 2)) __builtin_amdgcn_s_setprio(0);                           \
  TAILWAIT;                                                                  \
  __builtin_amdgcn_s_barrier()

  for (int t = 0; t < ntiles; ++t) {
    const char* Ab = abuf(t);
    const char* Bb = bbuf(t);
    // phase 1: A(mi half 0) + B(ni half 0) reads; stage R2(t+1)
    #pragma unroll
    for (int mi = 0; mi < MI / 2; ++mi) {
      a[mi][0] = frag<SWZ>(Ab, wr * MI + mi, 0, fr, fq);
      a[mi][1] = frag<SWZ>(Ab, wr * MI + mi, 1, fr, fq);
    }
    #pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      blo[ni][0] = frag<SWZ>(Bb, wc * 4 + ni, 0, fr, fq);
      blo[ni][1] = frag<SWZ>(Bb, wc * 4 + ni, 1, fr, fq);
    }
    if (TWEAK & 16) {
      #pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        bhi[ni][0] = frag<SWZ>(Bb, wc * 4 + 2 + ni, 0, fr, fq);
        bhi[ni][1] = frag<SWZ>(Bb, wc * 4 + 2 + ni, 1, fr, fq);
      }
    }
    stage_a(t + 1, 1);
    if (TWEAK & 1)  // drain most of the 12-read burst before the barrier
      asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");
    // DEEP: force R3(t) (phase-2's bhi source) landed at this phase's tail
    P8_BAR_MFMA(0, 0, blo,
                if (DEEP && BNT == 256)
                    asm volatile("s_waitcnt vmcnt(6)" ::: "memory"));

    // phase 2: B(ni half 1) reads (unless phase-1 prefetched); stage R3
    if (!(TWEAK & 16)) {
      #pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        bhi[ni][0] = frag<SWZ>(Bb, wc * 4 + 2 + ni, 0, fr, fq);
        bhi[ni][1] = frag<SWZ>(Bb, wc * 4 + 2 + ni, 1, fr, fq);
      }
    }
    stage_b(t + 1, 1);
    P8_BAR_MFMA(0, 1, bhi, );

    // phase 3: A(mi half 1) reads; stage R0(t+2)
    #pragma unroll
    for (int mi = 0; mi < MI / 2; ++mi) {
      a[mi][0] = frag<SWZ>(Ab, wr * MI + MI / 2 + mi, 0, fr, fq);
      a[mi][1] = frag<SWZ>(Ab, wr * MI + MI / 2 + mi, 1, fr, fq);
    }
    stage_a(t + 2, 0);
    P8_BAR_MFMA(1, 0, blo, );

    // phase 4: no reads; stage R1(t+2); the tile's single counted wait
    stage_b(t + 2, 0);
    if (BNT == 256) {
      if (DEEP)
        asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    if (!(TWEAK & 2)) __builtin_amdgcn_s_setprio(1);
    P8_QUAD(1, 1, bhi);
    if (!(TWEAK & 2)) __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
  }
#undef P8_BAR_MFMA
#undef P8_QUAD

  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  // epilogue: bias + activation, LDS-staged coalesced stores
  // (ctile [256][BNT] bf16)
  unsigned short* ctile = (unsigned short*)lds;
  #pragma unroll
  for (int mi = 0; mi < MI; ++mi) {
    #pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int lc = wc * 64 + ni * 16 + fr;
      float bv = bias != nullptr ? bias[min(n0 + lc, N - 1)] : 0.f;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        int lr = wr * (MI * 16) + mi * 16 + fq * 4 + r;
        ctile[lr * BNT + lc] = f2bf(act_fwd(acc[mi][ni][r] + bv, act, slope));
      }
    }
  }
  __syncthreads();
  const int t = threadIdx.x;
  constexpr int SEGS = BNT / 8;  // 16B pieces per row
  #pragma unroll
  for (int i = 0; i < 256 * SEGS / 512; ++i) {
    int piece = i * 512 + t;
    int row = piece / SEGS;
    int seg = piece % SEGS;
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
      s16x8 v = *(const s16x8*)(ctile + row * BNT + seg * 8);
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

// 256x128 tile: measured neutral on dcgan64's N=128 family (+1% conv2
// microbench, -0.4% e2e) and -1.5% on dcgan28's — the N<=128 gathers are
// staging-latency-bound, so the tile shape doesn't matter and the
// 2-block/CU 128-tile kernel's co-residency wins slightly.  Opt-in.
static int p8_n128() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("GDLJ_8P_N128");
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
      // >64 KiB dynamic LDS needs an explicit opt-in per kernel
      #define P8_SETATTR(KF, B)                                              \
        (void)hipFuncSetAttribute(reinterpret_cast<const void*>(&KF),        \
                                  hipFuncAttributeMaxDynamicSharedMemorySize,\
                                  B)
      constexpr int L256 = 2 * (p8::OPTA + 256 * p8::BK * 2);
      constexpr int L128 = 2 * (p8::OPTA + 128 * p8::BK * 2);
      P8_SETATTR((gemm_tn_8p<false, 0, false>), L256);
      P8_SETATTR((gemm_tn_8p<false, 1, false>), L256);
      P8_SETATTR((gemm_tn_8p<false, 2, false>), L256);
      P8_SETATTR((gemm_tn_8p<true, 0, false>), L256);
      P8_SETATTR((gemm_tn_8p<true, 1, false>), L256);
      P8_SETATTR((gemm_tn_8p<true, 2, false>), L256);
      P8_SETATTR((gemm_tn_8p<false, 2, true>), L256);
      P8_SETATTR((gemm_tn_8p<true, 2, true>), L256);
      P8_SETATTR((gemm_tn_8p<false, 2, false, 128, 2>), L128);
      P8_SETATTR((gemm_tn_8p<true, 2, false, 128, 2>), L128);
      #undef P8_SETATTR
    }
  }
  return v;
}

// Tile-N selection: 0 = not eligible, else the BNT to launch.
// Both shapes need K deep enough to amortize the pipeline prologue
// (>= 4 K-tiles; at K = 128 the 2-block/CU 128-tile kernel wins),
// >= 256 blocks to fill the chip at one block/CU, and a large-M regime
// (dcgan28's M=8192 dense family measured -2.6% under 8p routing:
// 812.8k -> 791.5k img/s; every shape the 8p wins on has M >= 131k).
// GDLJ_8P_MINM overrides (0 for microbenches). profiles/gemm8p_ab.md.
static int p8_min_m() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("GDLJ_8P_MINM");
    v = (e != nullptr) ? atoi(e) : 32768;
  }
  return v;
}

static int p8_pick_bnt(int M, int N, int K) {
  if (!p8_enabled()) return 0;
  if (M < p8_min_m()) return 0;
  if (K % p8::BK != 0 || K < 4 * p8::BK) return 0;
  if (N >= 192) {
    long blocks = (long)ceil_div(M, p8::BM) * ceil_div(N, 256);
    if (blocks >= 256) return 256;
    return 0;
  }
  if (N >= 96 && p8_n128()) {
    long blocks = (long)ceil_div(M, p8::BM) * ceil_div(N, 128);
    if (blocks >= 256) return 128;
  }
  return 0;
}

int gemm_tn_8p_eligible(int M, int N, int K) {
  return p8_pick_bnt(M, N, K) != 0;
}

int launch_gemm_tn_8p(const void* A, const void* B, void* C,
                      const float* bias, int M, int N, int K, long lda,
                      long ldb, int act, float slope, int gather,
                      ConvGather ga, const void* zp, hipStream_t s) {
  int bnt = p8_pick_bnt(M, N, K);
  dim3 grid(ceil_div(M, p8::BM), ceil_div(N, bnt == 128 ? 128 : 256));
  dim3 blk(512);
  int lds_b = 2 * (p8::OPTA + (bnt == 128 ? 128 : 256) * p8::BK * 2);
  int swzm = p8_swz_mode();
  #define P8_LAUNCH(G, S, D, BT)                                             \
    hipLaunchKernelGGL((gemm_tn_8p<G, S, D, BT>), grid, blk, lds_b, s,       \
                       (const unsigned short*)A, (const unsigned short*)B,   \
                       (unsigned short*)C, bias, M, N, K, lda, ldb, act,     \
                       slope, ga, (const unsigned short*)zp)
  if (bnt == 128) {
    if (gather) P8_LAUNCH(true, 2, false, 128);
    else P8_LAUNCH(false, 2, false, 128);  // TWEAK default = 2
  } else if (p8_deep()) {
    if (gather) P8_LAUNCH(true, 2, true, 256);
    else P8_LAUNCH(false, 2, true, 256);
  } else if (gather) {
    if (swzm == 0) P8_LAUNCH(true, 0, false, 256);
    else if (swzm == 1) P8_LAUNCH(true, 1, false, 256);
    else P8_LAUNCH(true, 2, false, 256);
  } else {
    if (swzm == 0) P8_LAUNCH(false, 0, false, 256);
    else if (swzm == 1) P8_LAUNCH(false, 1, false, 256);
    else P8_LAUNCH(false, 2, false, 256);
  }
  #undef P8_LAUNCH
  return (int)grid.x;
}

// Within-probe A/B entry (tools/ab_gemm8p.py): plain bf16 TN at a fixed
// variant id; co-compiled instantiations keep codegen context shared so
// interleaved deltas are reliable (guide rule 19/24).
int launch_gemm_tn_8p_tweak(int tweak, const void* A, const void* B, void* C,
                            int M, int N, int K, long lda, long ldb,
                            hipStream_t s) {
  (void)p8_enabled();  // ensure LDS attributes are set
  constexpr int L256 = 2 * (p8::OPTA + 256 * p8::BK * 2);
  dim3 grid(ceil_div(M, p8::BM), ceil_div(N, 256));
  dim3 blk(512);
  ConvGather dummy{};
  #define P8_TW(T)                                                           \
    do {                                                                     \
      (void)hipFuncSetAttribute(                                             \
          reinterpret_cast<const void*>(&gemm_tn_8p<false, 2, false, 256,    \
                                                    T>),                     \
          hipFuncAttributeMaxDynamicSharedMemorySize, L256);                 \
      hipLaunchKernelGGL((gemm_tn_8p<false, 2, false, 256, T>), grid, blk,   \
                         L256, s, (const unsigned short*)A,                  \
                         (const unsigned short*)B, (unsigned short*)C,       \
                         nullptr, M, N, K, lda, ldb, 0, 0.0f, dummy,         \
                         nullptr);                                           \
    } while (0)
  switch (tweak) {
    case 1: P8_TW(1); break;
    case 2: P8_TW(2); break;
    case 3: P8_TW(3); break;
    case 4: P8_TW(4); break;
    case 5: P8_TW(5); break;
    case 7: P8_TW(7); break;
    case 18: P8_TW(18); break;  // 16|2: bhi prefetch + static priority
    case 8:  // DEEP pipeline re-test under the static-priority regime
      (void)hipFuncSetAttribute(
          reinterpret_cast<const void*>(&gemm_tn_8p<false, 2, true, 256, 2>),
          hipFuncAttributeMaxDynamicSharedMemorySize, L256);
      hipLaunchKernelGGL((gemm_tn_8p<false, 2, true, 256, 2>), grid, blk,
                         L256, s, (const unsigned short*)A,
                         (const unsigned short*)B, (unsigned short*)C,
                         nullptr, M, N, K, lda, ldb, 0, 0.0f, dummy,
                         nullptr);
      break;
    default: P8_TW(0); break;
  }
  #undef P8_TW
  return (int)grid.x;
}

}  // extern "C"
