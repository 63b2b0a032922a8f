// bf16 MFMA GEMM kernels for gfx950 (CDNA4) — the conv/dense compute core.
//
// Two flavors cover every GEMM in the framework:
//
//  gemm_tn:  C[M][N] = act(A[M][K] . B[N][K]^T + bias)
//            both operands K-contiguous (TN form): conv/convT/dense forward
//            (A=im2col, B=weights) and data-grad (B=W^T).
//            128x128x64 tile, 4 waves, double-buffered LDS staged with
//            global_load_lds (16B), XOR-swizzled via the SOURCE address
//            (guide rule 21), mfma_f32_16x16x32_bf16 inner loop, LDS-staged
//            coalesced epilogue with fused bias+activation.
//            GATHER_A variant = implicit-GEMM convolution: the A operand is
//            gathered straight from the NHWC image (im2col fused into the
//            staging address computation; out-of-bounds/padded chunks are
//            redirected to a 16B zero page so global_load_lds stays
//            unconditional).
//
//  gemm_nt:  C[M][N] (fp32) += sum_k A[k][M] * B[k][N]
//            contraction over ROWS of both operands: weight-grad
//            (A=dOut, B=im2col). 128x128x64 tile, 8 waves, double-buffered
//            register staging (loads for tile t+1 issued before tile t's
//            MFMAs), split-K over blockIdx.z with fp32 atomics. Either
//            operand may be gathered from an NHWC image (implicit wgrad).
//
// K must be a multiple of 64 for gemm_tn (callers pad; gathered A pads by
// zero-page); gemm_nt handles arbitrary K by zero-fill.

#include "common.h"
#include <stdlib.h>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

#define GLDS16(gsrc, ldst)                                                    \
  __builtin_amdgcn_global_load_lds(                                          \
      (const __attribute__((address_space(1))) unsigned int*)(gsrc),          \
      (__attribute__((address_space(3))) unsigned int*)(ldst), 16, 0, 0)

// decode np -> (n, ho, wo)
DEV_INLINE void np_decode(const ConvGather& g, unsigned np, int& n, int& ho,
                          int& wo) {
  unsigned q1 = fdiv(np, g.fWo);
  wo = (int)(np - q1 * g.Wo);
  unsigned q2 = fdiv(q1, g.fHo);
  ho = (int)(q1 - q2 * g.Ho);
  n = (int)q2;
}

// decode k -> (r, s, c); caller checks k < rsc
DEV_INLINE void k_decode(const ConvGather& g, unsigned k, int& r, int& s,
                         int& c) {
  unsigned rs = fdiv(k, g.fC);
  c = (int)(k - rs * g.C);
  unsigned rr = fdiv(rs, g.fS);
  s = (int)(rs - rr * g.S);
  r = (int)rr;
}

// source address for grid position (ho,wo) + kernel offset (r,s), channel c;
// returns nullptr-sentinel via 'valid'
DEV_INLINE const unsigned short* gather_addr(
    const ConvGather& g, const unsigned short* img, int n, int ho, int wo,
    int r, int s, int c, bool& valid) {
  int hi, wi;
  if (g.mode == 0) {
    hi = ho * g.stride - g.pad + r;
    wi = wo * g.stride - g.pad + s;
    valid = hi >= 0 && hi < g.H && wi >= 0 && wi < g.W;
  } else if (g.mode == 2) {
    hi = ho + g.off_h - r;
    wi = wo + g.off_w - s;
    valid = hi >= 0 && hi < g.H && wi >= 0 && wi < g.W;
  } else {
    int hop = ho + g.pad - r;
    int wop = wo + g.pad - s;
    if (hop < 0 || wop < 0) {
      valid = false;
      hi = wi = 0;
    } else {
      unsigned qh = fdiv((unsigned)hop, g.fStride);
      unsigned qw = fdiv((unsigned)wop, g.fStride);
      valid = (hop == (int)(qh * g.stride)) &&
              (wop == (int)(qw * g.stride)) && (int)qh < g.H && (int)qw < g.W;
      hi = (int)qh;
      wi = (int)qw;
    }
  }
  return img + (((long)n * g.H + hi) * g.W + wi) * g.C + c;
}

// ---------------------------------------------------------------------------
// TN kernel
// ---------------------------------------------------------------------------
constexpr int TN_BM = 128, TN_BN = 128, TN_BK = 64;
constexpr int TN_TILE_B = TN_BM * TN_BK * 2;  // 16 KiB per operand tile

// stage one plain operand tile: 1024 16B chunks, 4 glds per thread
DEV_INLINE void tn_stage(const unsigned short* __restrict__ g, int row0,
                         int nrows, long ldk, int k0, char* lds) {
  const int t = threadIdx.x;
  const int wid = t >> 6;
  #pragma unroll
  for (int i = 0; i < 4; ++i) {
    int chunk = i * 256 + t;         // = row*8 + slot
    int row = chunk >> 3;
    int slot = chunk & 7;
    int gslot = slot ^ (row & 7);    // inverse swizzle on the source
    int grow = min(row0 + row, nrows - 1);
    const unsigned short* src = g + (long)grow * ldk + k0 + gslot * 8;
    char* dst = lds + (i * 256 + wid * 64) * 16;  // wave-uniform, lane-linear
    GLDS16(src, dst);
  }
}

// implicit-GEMM stage: A rows are im2col rows gathered from the NHWC image.
// The (np -> n,ho,wo) row decode and the n-plane base offset are invariant
// across the whole K-loop (each thread stages the same 4 rows every step),
// so they are hoisted into registers once (init) and only the cheap k part
// (tap + channel) is decoded per chunk per step (stage).
template <int NTH>  // block thread count: 4 chunks staged per thread
struct TnGatherStagerT {
  long base[4];        // (long)n * H*W*C element offset per chunk
  int h0[4], w0[4];    // mode-adjusted spatial bases per chunk

  DEV_INLINE void init(const ConvGather& g, int row0, int nrows) {
    const int t = threadIdx.x;
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      int row = (i * NTH + t) >> 3;
      int np = min(row0 + row, nrows - 1);
      int n, ho, wo;
      np_decode(g, (unsigned)np, n, ho, wo);
      base[i] = (long)n * g.H * g.W * g.C;
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

  DEV_INLINE void stage(const unsigned short* __restrict__ img,
                        const ConvGather& g,
                        const unsigned short* __restrict__ zp, int k0,
                        char* lds) const {
    const int t = threadIdx.x;
    const int wid = t >> 6;
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      int chunk = i * NTH + t;
      int row = chunk >> 3;
      int slot = chunk & 7;
      int gslot = slot ^ (row & 7);
      int k = k0 + gslot * 8;
      const unsigned short* src = zp;
      if (k < g.rsc) {
        int r, s, c;
        k_decode(g, (unsigned)k, r, s, c);
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
      char* dst = lds + (i * NTH + wid * 64) * 16;
      GLDS16(src, dst);
    }
  }
};
using TnGatherStager = TnGatherStagerT<256>;

DEV_INLINE bf16x8 tn_frag(const char* lds, int row, int kslot) {
  int byte = row * 128 + ((kslot ^ (row & 7)) * 16);
  return *(const bf16x8*)(lds + byte);
}

template <bool GATHER_A>
__global__ __launch_bounds__(256, 2) void gemm_tn_core(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ B,
    unsigned short* __restrict__ C, float* __restrict__ Cf,
    const float* __restrict__ bias, int M, int N, int K, long lda, long ldb,
    int act, float slope, ConvGather ga,
    const unsigned short* __restrict__ zp,
    float* __restrict__ bn_part) {
  __shared__ __attribute__((aligned(128))) char lds[4 * TN_TILE_B];
  auto abuf = [&](int i) -> char* { return lds + (i ? 2 * TN_TILE_B : 0); };
  auto bbuf = [&](int i) -> char* {
    return lds + TN_TILE_B + (i ? 2 * TN_TILE_B : 0);
  };

  // XCD-aware swizzle (T1, bijective variant) over M-tiles
  int nwgx = gridDim.x;
  int bidx = blockIdx.x;
  if (nwgx >= 8) {
    int q = nwgx / 8, r = nwgx % 8;
    int xcd = bidx % 8, idx = bidx / 8;
    bidx = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int m0 = bidx * TN_BM;
  const int n0 = blockIdx.y * TN_BN;

  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wr = wid >> 1, wc = wid & 1;
  const int fr = lane & 15, fq = lane >> 4;

  f32x4 acc[4][4];
  #pragma unroll
  for (int i = 0; i < 4; ++i)
    #pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int ntiles = K / TN_BK;
  TnGatherStager gs;
  if (GATHER_A) {
    gs.init(ga, m0, M);
    gs.stage(A, ga, zp, 0, abuf(0));
  } else {
    tn_stage(A, m0, M, lda, 0, abuf(0));
  }
  tn_stage(B, n0, N, ldb, 0, bbuf(0));
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  for (int t = 0; t < ntiles; ++t) {
    int cur = t & 1;
    if (t + 1 < ntiles) {
      if (GATHER_A)
        gs.stage(A, ga, zp, (t + 1) * TN_BK, abuf(cur ^ 1));
      else
        tn_stage(A, m0, M, lda, (t + 1) * TN_BK, abuf(cur ^ 1));
      tn_stage(B, n0, N, ldb, (t + 1) * TN_BK, bbuf(cur ^ 1));
    }
    const char* Al = abuf(cur);
    const char* Bl = bbuf(cur);
    #pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      bf16x8 a[4], b[4];
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        a[mi] = tn_frag(Al, wr * 64 + mi * 16 + fr, kc * 4 + fq);
      #pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        b[ni] = tn_frag(Bl, wc * 64 + ni * 16 + fr, kc * 4 + fq);
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[mi], b[ni], acc[mi][ni], 0, 0, 0);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
  }

  // ---- epilogue: bias + activation, LDS-staged coalesced stores ------
  if (C != nullptr && (N & 7) == 0) {
    unsigned short* ctile = (unsigned short*)lds;  // [128][128] bf16
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
          ctile[lr * 128 + lc] = f2bf(act_fwd(acc[mi][ni][r] + bv, act,
                                              slope));
        }
      }
    }
    __syncthreads();
    const int t = threadIdx.x;
    #pragma unroll
    for (int i = 0; i < 8; ++i) {
      int piece = i * 256 + t;       // 2048 16B pieces = 128 rows x 16
      int row = piece >> 4;
      int seg = piece & 15;
      int grow = m0 + row;
      int gcol = n0 + seg * 8;
      if (grow < M && gcol < N) {
        long crow = grow;
        if (GATHER_A && ga.mode == 2) {
          int n, h2, w2;
          np_decode(ga, (unsigned)grow, n, h2, w2);
          crow = ((long)n * ga.oH + h2 * ga.stride + ga.oqh) * ga.oW +
                 w2 * ga.stride + ga.oqw;
        }
        s16x8 v = *(const s16x8*)(ctile + row * 128 + seg * 8);
        if (gcol + 8 <= N) {
          *(s16x8*)(&C[crow * N + gcol]) = v;
        } else {
          for (int j = 0; j < 8 && gcol + j < N; ++j)
            C[crow * N + gcol + j] = (unsigned short)v[j];
        }
      }
    }
    // optional fused BN-statistics: per-channel (tile-col) sum/sumsq of
    // the post-activation tile -> partials [gridDim.x][2N] (second pass =
    // bn_stats_sum2). Rows beyond M are masked out.
    if (bn_part != nullptr) {
      float* red = (float*)(lds + 2 * TN_TILE_B);  // past the 32KB ctile
      const int col = t & 127;
      const int rl = t >> 7;  // 2 row-lanes x 64 rows
      float s = 0.f, ss = 0.f;
      for (int row = rl; row < 128; row += 2) {
        if (m0 + row < M) {
          float v = bf2f(ctile[row * 128 + col]);
          s += v;
          ss += v * v;
        }
      }
      red[t] = s;
      red[256 + t] = ss;
      __syncthreads();
      if (rl == 0 && n0 + col < N) {
        s += red[128 + col];
        ss += red[256 + 128 + col];
        bn_part[(long)blockIdx.x * 2 * N + n0 + col] = s;
        bn_part[(long)blockIdx.x * 2 * N + N + n0 + col] = ss;
      }
    }
    return;
  }
  // fallback (fp32 out or N not a multiple of 8): direct predicated stores
  #pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
    #pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int col = n0 + wc * 64 + ni * 16 + fr;
      if (col >= N) continue;
      float bv = bias != nullptr ? bias[col] : 0.f;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wr * 64 + mi * 16 + fq * 4 + r;
        if (row >= M) continue;
        long crow = row;
        if (GATHER_A && ga.mode == 2) {
          int n, h2, w2;
          np_decode(ga, (unsigned)row, n, h2, w2);
          crow = ((long)n * ga.oH + h2 * ga.stride + ga.oqh) * ga.oW +
                 w2 * ga.stride + ga.oqw;
        }
        float v = act_fwd(acc[mi][ni][r] + bv, act, slope);
        if (C != nullptr)
          C[crow * N + col] = f2bf(v);
        else
          Cf[crow * N + col] = v;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Single-shot TN for K in {64, 128} (the K-thin dcol/dense family):
// every K-tile is staged ONCE up front (no double-buffer loop, no
// per-tile vmcnt(0)+barrier drain pairs) and the block runs
// stage -> one wait -> MFMA -> epilogue.  Same 128x128 tile / 4 waves /
// 2 blocks per CU geometry as gemm_tn_core; these shapes are bound by
// the C-write stream + per-block overhead, so the win is structural
// overhead removal, not MFMA scheduling.
// ---------------------------------------------------------------------------
template <int KT, bool GATHER_A = false>  // KT = 64 or 128
__global__ __launch_bounds__(256, 2) void gemm_tn_kshort(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ B,
    unsigned short* __restrict__ C, const float* __restrict__ bias, int M,
    int N, long lda, long ldb, int act, float slope, int direct_epi,
    ConvGather ga, const unsigned short* __restrict__ zp) {
  constexpr int NT = KT / TN_BK;  // K-tiles
  __shared__ __attribute__((aligned(128))) char lds[2 * NT * TN_TILE_B];
  auto abuf = [&](int t) -> char* { return lds + t * TN_TILE_B; };
  auto bbuf = [&](int t) -> char* { return lds + (NT + t) * TN_TILE_B; };

  int nwgx = gridDim.x;
  int bidx = blockIdx.x;
  if (nwgx >= 8) {
    int q = nwgx / 8, r = nwgx % 8;
    int xcd = bidx % 8, idx = bidx / 8;
    bidx = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int m0 = bidx * TN_BM;
  const int n0 = blockIdx.y * TN_BN;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wr = wid >> 1, wc = wid & 1;
  const int fr = lane & 15, fq = lane >> 4;

  f32x4 acc[4][4];
  #pragma unroll
  for (int i = 0; i < 4; ++i)
    #pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  TnGatherStager gs;
  if (GATHER_A) gs.init(ga, m0, M);
  #pragma unroll
  for (int t = 0; t < NT; ++t) {
    if (GATHER_A)
      gs.stage(A, ga, zp, t * TN_BK, abuf(t));
    else
      tn_stage(A, m0, M, lda, t * TN_BK, abuf(t));
    tn_stage(B, n0, N, ldb, t * TN_BK, bbuf(t));
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  #pragma unroll
  for (int t = 0; t < NT; ++t) {
    const char* Al = abuf(t);
    const char* Bl = bbuf(t);
    #pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      bf16x8 a[4], b[4];
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        a[mi] = tn_frag(Al, wr * 64 + mi * 16 + fr, kc * 4 + fq);
      #pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        b[ni] = tn_frag(Bl, wc * 64 + ni * 16 + fr, kc * 4 + fq);
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[mi], b[ni], acc[mi][ni], 0, 0, 0);
    }
  }

  if ((N & 7) == 0 && direct_epi != 1) {
    unsigned short* ctile = (unsigned short*)lds;  // [128][128] bf16
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
          ctile[lr * 128 + lc] = f2bf(act_fwd(acc[mi][ni][r] + bv, act,
                                              slope));
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
          np_decode(ga, (unsigned)grow, n2, h2, w2);
          crow = ((long)n2 * ga.oH + h2 * ga.stride + ga.oqh) * ga.oW +
                 w2 * ga.stride + ga.oqw;
        }
        s16x8 v = *(const s16x8*)(ctile + row * 128 + seg * 8);
        if (direct_epi == 2)  // probe: nontemporal C stream
          __builtin_nontemporal_store(v, (s16x8*)(&C[crow * N + gcol]));
        else
          *(s16x8*)(&C[crow * N + gcol]) = v;
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
        long crow = row;
        if (GATHER_A && ga.mode == 2) {
          int n2, h2, w2;
          np_decode(ga, (unsigned)row, n2, h2, w2);
          crow = ((long)n2 * ga.oH + h2 * ga.stride + ga.oqh) * ga.oW +
                 w2 * ga.stride + ga.oqw;
        }
        C[crow * N + col] = f2bf(act_fwd(acc[mi][ni][r] + bv, act,
                                              slope));
      }
    }
}

// 256-row stager for the wide K-short tile (2048 chunks per 64-K tile)
DEV_INLINE void tn_stage256(const unsigned short* __restrict__ g, int row0,
                            int nrows, long ldk, int k0, char* lds) {
  const int t = threadIdx.x;
  const int wid = t >> 6;
  #pragma unroll
  for (int i = 0; i < 8; ++i) {
    int chunk = i * 256 + t;
    int row = chunk >> 3;
    int slot = chunk & 7;
    int gslot = slot ^ (row & 7);
    int grow = min(row0 + row, nrows - 1);
    const unsigned short* src = g + (long)grow * ldk + k0 + gslot * 8;
    char* dst = lds + (i * 256 + wid * 64) * 16;
    GLDS16(src, dst);
  }
}

// Wide K=128 single-shot tile: 256(M) x 64(N), 4 waves each owning a
// 64x64 panel, 2 blocks/CU (A 64 KiB + B 16 KiB LDS) — half the block
// count of the 128x128 kshort at the same per-thread register budget,
// so the per-block stage-latency chain amortizes over 2x the MACs.
__global__ __launch_bounds__(256, 2) void gemm_tn_kshort_wide(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ B,
    unsigned short* __restrict__ C, const float* __restrict__ bias, int M,
    int N, long lda, long ldb, int act, float slope) {
  // 80 KiB dynamic LDS (static __shared__ caps at 64 KiB; the launcher
  // raises MaxDynamicSharedMemorySize): 2 blocks/CU at exactly 160 KiB
  extern __shared__ __attribute__((aligned(16))) char lds[];
  auto abuf = [&](int t) -> char* { return lds + t * 32768; };       // 2x32K
  auto bbuf = [&](int t) -> char* { return lds + 65536 + t * 8192; };

  int nwgx = gridDim.x;
  int bidx = blockIdx.x;
  if (nwgx >= 8) {
    int q = nwgx / 8, r = nwgx % 8;
    int xcd = bidx % 8, idx = bidx / 8;
    bidx = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int m0 = bidx * 256;
  const int n0 = blockIdx.y * 64;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int fr = lane & 15, fq = lane >> 4;

  f32x4 acc[4][4];
  #pragma unroll
  for (int i = 0; i < 4; ++i)
    #pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  #pragma unroll
  for (int t = 0; t < 2; ++t) {
    tn_stage256(A, m0, M, lda, t * 64, abuf(t));
    // B: 64 rows x 64 k = 512 chunks (2 per thread)
    const int th = threadIdx.x;
    #pragma unroll
    for (int i = 0; i < 2; ++i) {
      int chunk = i * 256 + th;
      int row = chunk >> 3;
      int slot = chunk & 7;
      int gslot = slot ^ (row & 7);
      int grow = min(n0 + row, N - 1);
      const unsigned short* src = B + (long)grow * ldb + t * 64 + gslot * 8;
      char* dst = bbuf(t) + (i * 256 + wid * 64) * 16;
      GLDS16(src, dst);
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  #pragma unroll
  for (int t = 0; t < 2; ++t) {
    const char* Al = abuf(t);
    const char* Bl = bbuf(t);
    #pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      bf16x8 a[4], b[4];
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        a[mi] = tn_frag(Al, wid * 64 + mi * 16 + fr, kc * 4 + fq);
      #pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        b[ni] = tn_frag(Bl, ni * 16 + fr, kc * 4 + fq);
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[mi], b[ni], acc[mi][ni], 0, 0, 0);
    }
  }

  // epilogue: ctile [256][64] bf16 = 32 KiB
  unsigned short* ctile = (unsigned short*)lds;
  __syncthreads();
  #pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
    #pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int lc = ni * 16 + fr;
      float bv = bias != nullptr ? bias[min(n0 + lc, N - 1)] : 0.f;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        int lr = wid * 64 + mi * 16 + fq * 4 + r;
        ctile[lr * 64 + lc] = f2bf(act_fwd(acc[mi][ni][r] + bv, act, slope));
      }
    }
  }
  __syncthreads();
  const int t = threadIdx.x;
  #pragma unroll
  for (int i = 0; i < 8; ++i) {
    int piece = i * 256 + t;  // 2048 pieces = 256 rows x 8 segs
    int row = piece >> 3;
    int seg = piece & 7;
    int grow = m0 + row;
    int gcol = n0 + seg * 8;
    if (grow < M && gcol + 8 <= N)
      *(s16x8*)(&C[(long)grow * N + gcol]) =
          *(const s16x8*)(ctile + row * 64 + seg * 8);
  }
}

// ---------------------------------------------------------------------------
// 256x128x64 TN variant (GDLJ_TN256): +37% arithmetic intensity per staged
// byte (2M MACs per 48 KiB vs 1M per 32 KiB) at the SAME 8-wave/CU
// occupancy — one 512-thread block per CU with 96 KiB dynamic LDS
// (2x32 KiB A + 2x16 KiB B double buffers; the 64 KiB epilogue ctile
// reuses the A region). No mode-2 scatter / fp32-out / bn_part here —
// the launcher falls back to the 128 tile for those.
// ---------------------------------------------------------------------------
constexpr int T2_BM = 256, T2_BN = 128, T2_BK = 64;
constexpr int T2_AT = T2_BM * T2_BK * 2;  // 32 KiB
constexpr int T2_BT = T2_BN * T2_BK * 2;  // 16 KiB

DEV_INLINE void t2_stage_a(const unsigned short* __restrict__ g, int row0,
                           int nrows, long ldk, int k0, char* lds) {
  const int t = threadIdx.x;
  const int wid = t >> 6;
  #pragma unroll
  for (int i = 0; i < 4; ++i) {           // 2048 chunks
    int chunk = i * 512 + t;
    int row = chunk >> 3;
    int slot = chunk & 7;
    int gslot = slot ^ (row & 7);
    int grow = min(row0 + row, nrows - 1);
    const unsigned short* src = g + (long)grow * ldk + k0 + gslot * 8;
    char* dst = lds + (i * 512 + wid * 64) * 16;
    GLDS16(src, dst);
  }
}

DEV_INLINE void t2_stage_b(const unsigned short* __restrict__ g, int row0,
                           int nrows, long ldk, int k0, char* lds) {
  const int t = threadIdx.x;
  const int wid = t >> 6;
  #pragma unroll
  for (int i = 0; i < 2; ++i) {           // 1024 chunks
    int chunk = i * 512 + t;
    int row = chunk >> 3;
    int slot = chunk & 7;
    int gslot = slot ^ (row & 7);
    int grow = min(row0 + row, nrows - 1);
    const unsigned short* src = g + (long)grow * ldk + k0 + gslot * 8;
    char* dst = lds + (i * 512 + wid * 64) * 16;
    GLDS16(src, dst);
  }
}

template <bool GATHER_A>
__global__ __launch_bounds__(512, 1) void gemm_tn_core256(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ B,
    unsigned short* __restrict__ C, const float* __restrict__ bias, int M,
    int N, int K, long lda, long ldb, int act, float slope, ConvGather ga,
    const unsigned short* __restrict__ zp) {
  extern __shared__ __attribute__((aligned(128))) char lds[];
  auto abuf = [&](int i) -> char* { return lds + (i ? T2_AT : 0); };
  auto bbuf = [&](int i) -> char* {
    return lds + 2 * T2_AT + (i ? T2_BT : 0);
  };

  int nwgx = gridDim.x;
  int bidx = blockIdx.x;
  if (nwgx >= 8) {
    int q = nwgx / 8, r = nwgx % 8;
    int xcd = bidx % 8, idx = bidx / 8;
    bidx = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int m0 = bidx * T2_BM;
  const int n0 = blockIdx.y * T2_BN;

  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wr = wid >> 1, wc = wid & 1;  // 4(m) x 2(n) wave grid
  const int fr = lane & 15, fq = lane >> 4;

  f32x4 acc[4][4];
  #pragma unroll
  for (int i = 0; i < 4; ++i)
    #pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int ntiles = K / T2_BK;
  TnGatherStagerT<512> gs;
  if (GATHER_A) {
    gs.init(ga, m0, M);
    gs.stage(A, ga, zp, 0, abuf(0));
  } else {
    t2_stage_a(A, m0, M, lda, 0, abuf(0));
  }
  t2_stage_b(B, n0, N, ldb, 0, bbuf(0));
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  for (int t = 0; t < ntiles; ++t) {
    int cur = t & 1;
    if (t + 1 < ntiles) {
      if (GATHER_A)
        gs.stage(A, ga, zp, (t + 1) * T2_BK, abuf(cur ^ 1));
      else
        t2_stage_a(A, m0, M, lda, (t + 1) * T2_BK, abuf(cur ^ 1));
      t2_stage_b(B, n0, N, ldb, (t + 1) * T2_BK, bbuf(cur ^ 1));
    }
    const char* Al = abuf(cur);
    const char* Bl = bbuf(cur);
    #pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      bf16x8 a[4], b[4];
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        a[mi] = tn_frag(Al, wr * 64 + mi * 16 + fr, kc * 4 + fq);
      #pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        b[ni] = tn_frag(Bl, wc * 64 + ni * 16 + fr, kc * 4 + fq);
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[mi], b[ni], acc[mi][ni], 0, 0, 0);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
  }

  // epilogue: bias + act, LDS-staged coalesced stores (ctile 256x128 bf16
  // = 64 KiB in the A-buffer region)
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
          ctile[lr * 128 + lc] = f2bf(act_fwd(acc[mi][ni][r] + bv, act,
                                              slope));
        }
      }
    }
    __syncthreads();
    const int t = threadIdx.x;
    #pragma unroll
    for (int i = 0; i < 8; ++i) {
      int piece = i * 512 + t;       // 4096 16B pieces = 256 rows x 16
      int row = piece >> 4;
      int seg = piece & 15;
      int grow = m0 + row;
      int gcol = n0 + seg * 8;
      if (grow < M && gcol < N) {
        s16x8 v = *(const s16x8*)(ctile + row * 128 + seg * 8);
        if (gcol + 8 <= N) {
          *(s16x8*)(&C[(long)grow * N + gcol]) = v;
        } else {
          for (int j = 0; j < 8 && gcol + j < N; ++j)
            C[(long)grow * N + gcol + j] = (unsigned short)v[j];
        }
      }
    }
    return;
  }
  // N % 8 != 0 fallback: direct predicated stores
  #pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
    #pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int col = n0 + wc * 64 + ni * 16 + fr;
      if (col >= N) continue;
      float bv = bias != nullptr ? bias[col] : 0.f;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wr * 64 + mi * 16 + fq * 4 + r;
        if (row >= M) continue;
        C[(long)row * N + col] = f2bf(act_fwd(acc[mi][ni][r] + bv, act,
                                              slope));
      }
    }
  }
}

// ---------------------------------------------------------------------------
// NT kernel (weight grad): C[M][N] += sum_k A[k][M]*B[k][N], fp32 out.
//
// Contraction runs over operand ROWS, so rows stage straight into LDS with
// global_load_lds (fully coalesced) in a [kb][mb][4k][16m] window layout,
// and MFMA fragments come out via ds_read_b64_tr_b16 (gfx950 hardware
// transpose read: per 16-lane group, lane l receives column l of a [4][16]
// row-major window — verified by tools/probe_tr16.hip).
// ---------------------------------------------------------------------------
constexpr int NT_BM = 128, NT_BN = 128, NT_BK = 64;
constexpr int NT_TILE_E = NT_BM * NT_BK;     // elements per operand tile
constexpr int NT_THREADS = 512;              // 8 waves: 2(m) x 4(n)

typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;

// chunk c (16B = 8 elems) of the window-linear LDS image:
//   window w = c>>3 (64 elems: [4 rows k][16 cols m]), j = (c>>1)&3, h = c&1
//   w = kb*(BM/16) + mb
// source = operand row (k0 + kb*4 + j), cols (c0 + mb*16 + h*8 .. +8)
DEV_INLINE void nt_stage(const unsigned short* __restrict__ g, int k0, int K,
                         int c0, int L, long ldl,
                         const unsigned short* __restrict__ zp, char* lds) {
  const int t = threadIdx.x;
  const int wid = t >> 6;
  #pragma unroll
  for (int i = 0; i < 2; ++i) {
    int c = i * 512 + t;
    int w = c >> 3;
    int j = (c >> 1) & 3;
    int h = c & 1;
    int kb = w >> 3;
    int mb = w & 7;
    int row = k0 + kb * 4 + j;
    const unsigned short* src = zp;
    if (row < K) {
      int col = c0 + mb * 16 + h * 8;
      col = min(col, max(0, L - 8));  // M-edge clamp (rows >= M not stored)
      src = g + (long)row * ldl + col;
    }
    char* dst = lds + (i * 512 + wid * 64) * 16;
    GLDS16(src, dst);
  }
}

// gathered flavor: operand is the im2col of an NHWC image
// (contraction row = np; tile column = rsc coordinate).  The column-side
// k_decode (tap r,s + channel) is K-loop-invariant per thread — each
// thread stages the same two rsc coordinates every K-step — so it is
// hoisted into registers once (the row side np varies with k0 and is
// decoded per step).
struct NtGatherCols {
  int r[2], s[2], cc[2];
  bool in_rsc[2];

  DEV_INLINE void init(const ConvGather& g, int c0) {
    const int t = threadIdx.x;
    #pragma unroll
    for (int i = 0; i < 2; ++i) {
      int c = i * 512 + t;
      int w = c >> 3;
      int h = c & 1;
      int mb = w & 7;
      int kk = c0 + mb * 16 + h * 8;
      in_rsc[i] = kk < g.rsc;
      if (in_rsc[i])
        k_decode(g, (unsigned)kk, r[i], s[i], cc[i]);
      else
        r[i] = s[i] = cc[i] = 0;
    }
  }
};

DEV_INLINE void nt_stage_gather(const unsigned short* __restrict__ img,
                                const ConvGather& g, const NtGatherCols& gc,
                                int k0, int K,
                                const unsigned short* __restrict__ zp,
                                char* lds) {
  const int t = threadIdx.x;
  const int wid = t >> 6;
  #pragma unroll
  for (int i = 0; i < 2; ++i) {
    int c = i * 512 + t;
    int w = c >> 3;
    int j = (c >> 1) & 3;
    int kb = w >> 3;
    int np = k0 + kb * 4 + j;
    const unsigned short* src = zp;
    if (np < K && gc.in_rsc[i]) {
      int n, ho, wo;
      np_decode(g, (unsigned)np, n, ho, wo);
      bool valid;
      const unsigned short* p = gather_addr(g, img, n, ho, wo, gc.r[i],
                                            gc.s[i], gc.cc[i], valid);
      if (valid) src = p;
    }
    char* dst = lds + (i * 512 + wid * 64) * 16;
    GLDS16(src, dst);
  }
}

// fragment: lane holds col m = (mb*16 + fr), k = kc*32 + fq*8 + (0..7)
DEV_INLINE bf16x8 nt_tr_frag(const unsigned short* lds, int kc, int mb) {
  const int l = threadIdx.x & 63;
  const int g = l >> 4;
  int kb = kc * 8 + 2 * g;
  const unsigned short* a0 = lds + ((kb * 8 + mb) * 64 + (l & 15) * 4);
  bf16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (__attribute__((address_space(3))) bf16x4*)a0);
  bf16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (__attribute__((address_space(3))) bf16x4*)(a0 + 512));
  bf16x8 r;
  #pragma unroll
  for (int j = 0; j < 4; ++j) {
    r[j] = lo[j];
    r[4 + j] = hi[j];
  }
  return r;
}

// GMODE: 0 = plain/plain, 1 = A gathered, 2 = B gathered
template <int GMODE>
__global__ __launch_bounds__(NT_THREADS, 2) void gemm_nt_core(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ B,
    float* __restrict__ C, int M, int N, int K, long lda, long ldb,
    int kchunks_per_block, int use_atomic, ConvGather gg,
    const unsigned short* __restrict__ zp) {
  __shared__ __attribute__((aligned(128))) unsigned short lds[4 * NT_TILE_E];
  auto albuf = [&](int i) -> unsigned short* {
    return lds + (i ? 2 * NT_TILE_E : 0);
  };
  auto blbuf = [&](int i) -> unsigned short* {
    return lds + NT_TILE_E + (i ? 2 * NT_TILE_E : 0);
  };

  const int m0 = blockIdx.x * NT_BM;
  const int n0 = blockIdx.y * NT_BN;
  // (static young-half wave priority measured NEUTRAL here — at 2
  // blocks/CU the SIMDs arbitrate across blocks, unlike the 1-block/CU
  // 8p template where it is +4.9-8.9%; tools/ab_gemm8p.py)
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wr = wid >> 2, wc = wid & 3;
  const int fr = lane & 15, fq = lane >> 4;

  f32x4 acc[4][2];
  #pragma unroll
  for (int i = 0; i < 4; ++i)
    #pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int kchunks = (K + NT_BK - 1) / NT_BK;
  int kt0 = blockIdx.z * kchunks_per_block;
  int kt1 = min(kt0 + kchunks_per_block, kchunks);
  if (kt0 >= kt1) return;

  NtGatherCols gca, gcb;
  if (GMODE == 1) gca.init(gg, m0);
  if (GMODE == 2) gcb.init(gg, n0);
  auto stage_a = [&](int kt, char* buf) {
    if (GMODE == 1)
      nt_stage_gather(A, gg, gca, kt * NT_BK, K, zp, buf);
    else
      nt_stage(A, kt * NT_BK, K, m0, M, lda, zp, buf);
  };
  auto stage_b = [&](int kt, char* buf) {
    if (GMODE == 2)
      nt_stage_gather(B, gg, gcb, kt * NT_BK, K, zp, buf);
    else
      nt_stage(B, kt * NT_BK, K, n0, N, ldb, zp, buf);
  };

  stage_a(kt0, (char*)albuf(0));
  stage_b(kt0, (char*)blbuf(0));
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  for (int kt = kt0; kt < kt1; ++kt) {
    int cur = (kt - kt0) & 1;
    if (kt + 1 < kt1) {
      stage_a(kt + 1, (char*)albuf(cur ^ 1));
      stage_b(kt + 1, (char*)blbuf(cur ^ 1));
    }
    const unsigned short* Al = albuf(cur);
    const unsigned short* Bl = blbuf(cur);
    #pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      bf16x8 a[4], b[2];
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        a[mi] = nt_tr_frag(Al, kc, wr * 4 + mi);
      #pragma unroll
      for (int ni = 0; ni < 2; ++ni)
        b[ni] = nt_tr_frag(Bl, kc, wc * 2 + ni);
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        #pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[mi], b[ni], acc[mi][ni], 0, 0, 0);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
  }

  #pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
    #pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      int col = n0 + wc * 32 + ni * 16 + fr;
      if (col >= N) continue;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wr * 64 + mi * 16 + fq * 4 + r;
        if (row >= M) continue;
        if (use_atomic)
          atomicAdd(&C[(long)row * N + col], acc[mi][ni][r]);
        else
          C[(long)row * N + col] = acc[mi][ni][r];
      }
    }
  }
}

// ---------------------------------------------------------------------------
extern "C" {

// 8-phase deep-pipelined 256x256 TN core (gemm8p.hip) — the default for
// large bf16 TN shapes; this file's 128-tile kernels remain the fallback
// (fp32 out, bn_part fusion, small N/K/M).
int gemm_tn_8p_eligible(int M, int N, int K);
int launch_gemm_tn_8p(const void* A, const void* B, void* C,
                      const float* bias, int M, int N, int K, long lda,
                      long ldb, int act, float slope, int gather,
                      ConvGather ga, const void* zp, hipStream_t s);

// env-gated 256-row TN tile (see gemm_tn_core256). Opted in with
// GDLJ_TN256=1; requires bf16 output, no bn_part, gather mode != 2,
// and enough rows to fill the chip at 1 block/CU.
static int t2_enabled() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("GDLJ_TN256");
    v = (e != nullptr && e[0] == '1') ? 1 : 0;
    if (v) {
      (void)hipFuncSetAttribute(
          reinterpret_cast<const void*>(&gemm_tn_core256<false>),
          hipFuncAttributeMaxDynamicSharedMemorySize, 2 * T2_AT + 2 * T2_BT);
      (void)hipFuncSetAttribute(
          reinterpret_cast<const void*>(&gemm_tn_core256<true>),
          hipFuncAttributeMaxDynamicSharedMemorySize, 2 * T2_AT + 2 * T2_BT);
    }
  }
  return v;
}

static int kshort_enabled() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("GDLJ_KSHORT");
    v = (e != nullptr && e[0] == '0') ? 0 : 1;
  }
  return v;
}

int launch_gemm_tn(const void* A, const void* B, void* C_bf16, float* C_f32,
                   const float* bias, int M, int N, int K, long lda, long ldb,
                   int act, float slope, float* bn_part, hipStream_t s) {
  if (C_bf16 != nullptr && bn_part == nullptr && (K == 64 || K == 128) &&
      kshort_enabled()) {
    // 256x64 wide tile measured -10% vs the 128x128 kshort (fewer
    // blocks did NOT amortize: the wide tile halves B-reuse per byte of
    // LDS and the 80 KiB footprint stiffens scheduling) — opt-in probe.
    static int kwide = -1;
    if (kwide < 0) {
      const char* e = getenv("GDLJ_KSHORT_WIDE");
      kwide = (e != nullptr && e[0] == '1') ? 1 : 0;
    }
    if (kwide && K == 128 && (N & 63) == 0 && M >= 16 * 256) {
      static int attr_done = 0;
      if (!attr_done) {
        (void)hipFuncSetAttribute(
            reinterpret_cast<const void*>(&gemm_tn_kshort_wide),
            hipFuncAttributeMaxDynamicSharedMemorySize, 5 * 16384);
        attr_done = 1;
      }
      dim3 gridw(ceil_div(M, 256), N / 64);
      hipLaunchKernelGGL(gemm_tn_kshort_wide, gridw, dim3(256), 5 * 16384,
                         s, (const unsigned short*)A,
                         (const unsigned short*)B,
                         (unsigned short*)C_bf16, bias, M, N, lda, ldb, act,
                         slope);
      return (int)gridw.x;
    }
    dim3 grid(ceil_div(M, TN_BM), ceil_div(N, TN_BN));
    static int depi = -1;
    if (depi < 0) {
      const char* e = getenv("GDLJ_KSHORT_EPI");
      depi = (e != nullptr && e[0] == '1') ? 1 : 0;  // probe: direct stores
    }
    ConvGather kdummy{};
    if (K == 64)
      hipLaunchKernelGGL((gemm_tn_kshort<64>), grid, dim3(256), 0, s,
                         (const unsigned short*)A, (const unsigned short*)B,
                         (unsigned short*)C_bf16, bias, M, N, lda, ldb, act,
                         slope, depi, kdummy, nullptr);
    else
      hipLaunchKernelGGL((gemm_tn_kshort<128>), grid, dim3(256), 0, s,
                         (const unsigned short*)A, (const unsigned short*)B,
                         (unsigned short*)C_bf16, bias, M, N, lda, ldb, act,
                         slope, depi, kdummy, nullptr);
    return (int)grid.x;
  }
  if (C_bf16 != nullptr && bn_part == nullptr &&
      gemm_tn_8p_eligible(M, N, K)) {
    ConvGather dummy{};
    return launch_gemm_tn_8p(A, B, C_bf16, bias, M, N, K, lda, ldb, act,
                             slope, 0, dummy, nullptr, s);
  }
  if (t2_enabled() && C_bf16 != nullptr && bn_part == nullptr &&
      M >= 8 * T2_BM) {
    dim3 grid(ceil_div(M, T2_BM), ceil_div(N, T2_BN));
    ConvGather dummy{};
    hipLaunchKernelGGL((gemm_tn_core256<false>), grid, dim3(512),
                       2 * T2_AT + 2 * T2_BT, s, (const unsigned short*)A,
                       (const unsigned short*)B, (unsigned short*)C_bf16,
                       bias, M, N, K, lda, ldb, act, slope, dummy, nullptr);
    return (int)grid.x;
  }
  dim3 grid(ceil_div(M, TN_BM), ceil_div(N, TN_BN));
  ConvGather dummy{};
  hipLaunchKernelGGL((gemm_tn_core<false>), grid, dim3(256), 0, s,
                     (const unsigned short*)A, (const unsigned short*)B,
                     (unsigned short*)C_bf16, C_f32, bias, M, N, K, lda, ldb,
                     act, slope, dummy, nullptr, bn_part);
  return (int)grid.x;
}

// conv_direct.hip: LDS-image direct conv for the small-C forward family
extern "C" int launch_conv_direct_smallc(const void*, const void*, void*,
                                         const float*, const void*, int, int,
                                         int, int, int, int, int, int, int,
                                         int, int, int, int, float,
                                         hipStream_t);

static int direct_conv_enabled() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("GDLJ_DIRECT_CONV");
    v = (e != nullptr && e[0] == '0') ? 0 : 1;
  }
  return v;
}

// implicit-GEMM conv forward / gathered-A TN: A is an NHWC image; M = the
// number of patch positions; K = kpad (zero page covers k >= rsc).
int launch_gemm_tn_gather(const void* img, const void* B, void* C_bf16,
                          const float* bias, int M, int N, int K, long ldb,
                          int act, float slope, ConvGather ga,
                          const void* zero_page, float* bn_part,
                          hipStream_t s) {
  if (bn_part == nullptr && ga.mode == 0 && ldb == K &&
      direct_conv_enabled() &&
      launch_conv_direct_smallc(img, B, C_bf16, bias, zero_page, ga.N, ga.H,
                                ga.W, ga.C, ga.Ho, ga.Wo, ga.R, ga.S,
                                ga.stride, ga.pad, N, K, act, slope, s))
    return ceil_div(M, 128);
  if (bn_part == nullptr && (K == 64 || K == 128) && kshort_enabled()) {
    dim3 grid(ceil_div(M, TN_BM), ceil_div(N, TN_BN));
    if (K == 64)
      hipLaunchKernelGGL((gemm_tn_kshort<64, true>), grid, dim3(256), 0, s,
                         (const unsigned short*)img,
                         (const unsigned short*)B, (unsigned short*)C_bf16,
                         bias, M, N, 0, ldb, act, slope, 0, ga,
                         (const unsigned short*)zero_page);
    else
      hipLaunchKernelGGL((gemm_tn_kshort<128, true>), grid, dim3(256), 0, s,
                         (const unsigned short*)img,
                         (const unsigned short*)B, (unsigned short*)C_bf16,
                         bias, M, N, 0, ldb, act, slope, 0, ga,
                         (const unsigned short*)zero_page);
    return (int)grid.x;
  }
  if (bn_part == nullptr && gemm_tn_8p_eligible(M, N, K)) {
    return launch_gemm_tn_8p(img, B, C_bf16, bias, M, N, K, 0, ldb, act,
                             slope, 1, ga, zero_page, s);
  }
  if (t2_enabled() && bn_part == nullptr && ga.mode != 2 && M >= 8 * T2_BM) {
    dim3 grid(ceil_div(M, T2_BM), ceil_div(N, T2_BN));
    hipLaunchKernelGGL((gemm_tn_core256<true>), grid, dim3(512),
                       2 * T2_AT + 2 * T2_BT, s, (const unsigned short*)img,
                       (const unsigned short*)B, (unsigned short*)C_bf16,
                       bias, M, N, K, 0, ldb, act, slope, ga,
                       (const unsigned short*)zero_page);
    return (int)grid.x;
  }
  dim3 grid(ceil_div(M, TN_BM), ceil_div(N, TN_BN));
  hipLaunchKernelGGL((gemm_tn_core<true>), grid, dim3(256), 0, s,
                     (const unsigned short*)img, (const unsigned short*)B,
                     (unsigned short*)C_bf16, nullptr, bias, M, N, K, 0, ldb,
                     act, slope, ga, (const unsigned short*)zero_page,
                     bn_part);
  return (int)grid.x;
}

void launch_gemm_nt(const void* A, const void* B, float* C, int M, int N,
                    int K, long lda, long ldb, int splitk, int gmode,
                    ConvGather gg, const void* zp, hipStream_t s) {
  int kchunks = (K + NT_BK - 1) / NT_BK;
  if (splitk > kchunks) splitk = kchunks;
  int per_block = (kchunks + splitk - 1) / splitk;
  dim3 grid(ceil_div(M, NT_BM), ceil_div(N, NT_BN), splitk);
  int ua = splitk > 1 ? 1 : 0;
  const unsigned short* z = (const unsigned short*)zp;
  if (gmode == 1)
    hipLaunchKernelGGL((gemm_nt_core<1>), grid, dim3(NT_THREADS), 0, s,
                       (const unsigned short*)A, (const unsigned short*)B, C,
                       M, N, K, lda, ldb, per_block, ua, gg, z);
  else if (gmode == 2)
    hipLaunchKernelGGL((gemm_nt_core<2>), grid, dim3(NT_THREADS), 0, s,
                       (const unsigned short*)A, (const unsigned short*)B, C,
                       M, N, K, lda, ldb, per_block, ua, gg, z);
  else
    hipLaunchKernelGGL((gemm_nt_core<0>), grid, dim3(NT_THREADS), 0, s,
                       (const unsigned short*)A, (const unsigned short*)B, C,
                       M, N, K, lda, ldb, per_block, ua, gg, z);
}

}  // extern "C"
