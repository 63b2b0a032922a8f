// Direct strided conv dgrad for gfx950 — fuses the dcol GEMM + col2im
// round trip into one parity-decomposed kernel.
//
// The default strided dgrad materializes dcol[np][R*S*C8] (13.9 GB at
// the DCGAN-64 conv2 shape) and re-reads it in col2im.  Here each
// block owns a tile of dx pixels of ONE output-parity class
// (hi%stride, wi%stride) EXCLUSIVELY — so no atomics on dx, and every
// tap of the class's reduced tap set is valid for every pixel (no
// stride^2 MFMA waste, unlike a naive direct transposed conv).  The
// block stages the contributing dy region into LDS once (coalesced,
// zero-page halo cells absorb image borders), streams the transposed
// weight slices tap-by-tap (double-buffered, counted vmcnt), and
// accumulates complete per-pixel sums in MFMA registers.  Because the
// sums are complete, the epilogue also applies the producer's
// activation backward (col2im_dact's fusion) and emits its
// bias-gradient partials.  Each class pixel's C8-channel row is a
// whole 128-byte line, so the scattered stores stay line-coalesced.
//
// GEMM view per class tap (r,s): dx[px][c] += dy_rs[px][co] *
// Wt[(r*S+s)*C8 + c][co], Wt = the [R*S*C8][Ko8] transposed weight
// pack the dcol path already caches (gpu_ops "wt" cache).
//
// Tile: BM class-pixels x 64 c (BM 128 or 64), 4 waves; LDS = dy
// region + 32 KiB W double-buffer (ctile reuses the W area).
// Eligibility (launcher): stride == 2, Wc | BM, (Hc*Wc) % BM == 0,
// C8 % 64 == 0, Ko8 % 128 == 0 == ldw, region + 32 KiB <= 120 KiB.
// Covers conv2/conv3-class dgrads of the DCGAN family; everything
// else falls back to dcol+col2im (gemm.hip / im2col.hip).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

#define GLDS16(gsrc, ldst)                                                    \
  __builtin_amdgcn_global_load_lds(                                          \
      (const __attribute__((address_space(1))) unsigned int*)(gsrc),          \
      (__attribute__((address_space(3))) unsigned int*)(ldst), 16, 0, 0)

namespace cdd {
constexpr int BN = 64;
constexpr int WSLICE_B = 16 * 1024;      // one [64c][128co] slice (2 subs)
constexpr int WBUF_B = 2 * WSLICE_B;     // double buffer; >= ctile bytes
}  // namespace cdd

// 64-row x 64-k staged sub-tile, same swizzled layout/frag read as the
// proven tn layout (gemm.hip tn_stage / tn_frag geometry).
DEV_INLINE void cdd_stage_w64(const unsigned short* __restrict__ g,
                              long ldk, int k0, char* lds) {
  const int t = threadIdx.x;
  const int wid = t >> 6;
  #pragma unroll
  for (int i = 0; i < 2; ++i) {
    int chunk = i * 256 + t;  // 512 chunks = row*8 + slot
    int row = chunk >> 3;
    int slot = chunk & 7;
    int gslot = slot ^ (row & 7);
    const unsigned short* src = g + (long)row * ldk + k0 + gslot * 8;
    char* dst = lds + (i * 256 + wid * 64) * 16;
    GLDS16(src, dst);
  }
}

DEV_INLINE bf16x8 cdd_wfrag(const char* lds, int row, int kslot) {
  int byte = row * 128 + ((kslot ^ (row & 7)) * 16);
  return *(const bf16x8*)(lds + byte);
}

// MI = pixel fragments per wave (BM = MI * 64).
//
// The same kernel serves the strided TRANSPOSED-conv forward (G side):
// y[ho,wo,co] = sum over parity-valid taps of x[(ho+p-r)/2][ci] *
// W[ci,co,r,s] is the identical gather; pass x as `dy`, the
// [R*S*Co8][Cin] "w2a" pack as `Wt`, bias_fwd + act for the fused
// epilogue, and stats_part ([2048][2*C8]) for fused BN statistics.
template <int MI>
__global__ __launch_bounds__(256, 2) void conv_dgrad_direct(
    const unsigned short* __restrict__ dy,   // [N][Ho][Wo][Ko8]
    const unsigned short* __restrict__ Wt,   // [R*S*C8][ldw], k = co
    unsigned short* __restrict__ dx,         // [N][H][W][C8]
    const unsigned short* __restrict__ y0,   // producer act out (or null)
    float* __restrict__ part,                // [2048][C8] f32 (or null)
    const float* __restrict__ bias_fwd,      // convT fwd bias (or null)
    float* __restrict__ stats_part,          // [2048][2*C8] f32 (or null)
    const unsigned short* __restrict__ zp,   // 16B zero page
    int Nb, int H, int W, int C8, int Ho, int Wo, int Ko8, long ldw,
    int R, int S, int pad, int act, float slope,
    int Hc, int Wc, int rows_c, int rgn_rows, int rgn_cols,
    FastDiv fWc) {
  using namespace cdd;
  constexpr int BM = MI * 64;
  extern __shared__ __attribute__((aligned(16))) char lds[];
  char* rgn = lds;                                  // dy region
  const int rgn_b = rgn_rows * rgn_cols * Ko8 * 2;
  auto wbuf = [&](int i) -> char* { return lds + rgn_b + i * WSLICE_B; };

  // XCD-aware swizzle over class-pixel tiles
  int nwgx = gridDim.x;
  int bidx = blockIdx.x;
  if (nwgx >= 8) {
    int q = nwgx / 8, r = nwgx % 8;
    int xcd = bidx % 8, idx = bidx / 8;
    bidx = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int np0 = bidx * BM;
  const int hwc = Hc * Wc;
  const int n = np0 / hwc;         // once per block: plain div is fine
  const int cp0 = np0 - n * hwc;
  const int hc0 = cp0 / Wc;        // Wc | BM => whole class-row start
  const int cy = blockIdx.y;       // c-tile: channels [cy*64, cy*64+64)
  const int qh = blockIdx.z >> 1;  // parity class
  const int qw = blockIdx.z & 1;

  // class tap set: r = r0 + 2*rc (rc < Rc), s = s0 + 2*sc (sc < Sc)
  const int r0 = (qh + pad) & 1;
  const int s0 = (qw + pad) & 1;
  const int Rc = (R - r0 + 1) >> 1;
  const int Sc = (S - s0 + 1) >> 1;
  const int rb0 = (qh + pad - r0) >> 1;   // dy row offset of tap rc=0
  const int wb0 = (qw + pad - s0) >> 1;
  const int lo_h = hc0 + rb0 - (Rc - 1);  // dy row of region row 0
  const int lo_w = wb0 - (Sc - 1);        // dy col of region col 0

  // ---- stage the dy region: [rgn_rows][rgn_cols][Ko8], zero halo ----
  // Source-side chunk swizzle (slot = j ^ (cc & mask)): without it the
  // A-fragment reads of 16 consecutive-cc pixels land 256B apart =
  // same LDS bank (16-way conflict); the xor spreads them 8-wide.
  const int co_chunks = Ko8 >> 3;
  const int cmask = co_chunks - 1;
  const int ncell = rgn_rows * rgn_cols * co_chunks;
  for (int cell = threadIdx.x; cell < ncell; cell += 256) {
    int rc_ = cell / co_chunks;
    int slot = cell - rc_ * co_chunks;
    int rr = rc_ / rgn_cols;
    int cc = rc_ - rr * rgn_cols;
    int j = slot ^ (cc & cmask);
    int hig = lo_h + rr;
    int wig = lo_w + cc;
    const unsigned short* src = zp;
    if (hig >= 0 && hig < Ho && wig >= 0 && wig < Wo)
      src = dy + (((long)n * Ho + hig) * Wo + wig) * Ko8 + j * 8;
    char* dst = rgn + (cell & ~63) * 16;
    GLDS16(src, dst);
  }

  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int fr = lane & 15, fq = lane >> 4;

  // this lane's A-side class pixels: px_l = wid*(MI*16) + mi*16 + fr
  int pxh[MI], pxw[MI];
  #pragma unroll
  for (int mi = 0; mi < MI; ++mi) {
    int pxl = wid * (MI * 16) + mi * 16 + fr;
    int hl = (int)fdiv((unsigned)pxl, fWc);
    pxh[mi] = hl;                    // class-row within tile
    pxw[mi] = pxl - hl * Wc;         // class-col (global: full width)
  }

  const int nj = Ko8 >> 7;           // 128-co chunks per tap
  const int nslice = Rc * Sc * nj;
  const unsigned short* wt_cy = Wt + (long)cy * 64 * ldw;
  auto wsrc_for = [&](int sl) {
    int tapj = sl / nj, j = sl - tapj * nj;
    int rc_ = tapj / Sc, sc_ = tapj - rc_ * Sc;
    int tap = (r0 + 2 * rc_) * S + (s0 + 2 * sc_);
    return wt_cy + (long)tap * C8 * ldw + j * 128;
  };

  // first W slice, then one wait covers region + slice 0
  cdd_stage_w64(wsrc_for(0), ldw, 0, wbuf(0));
  cdd_stage_w64(wsrc_for(0), ldw, 64, wbuf(0) + 8 * 1024);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  f32x4 acc[MI][4];
  #pragma unroll
  for (int i = 0; i < MI; ++i)
    #pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  for (int sl = 0; sl < nslice; ++sl) {
    int tapj = sl / nj;
    int j = sl - tapj * nj;
    int rc_ = tapj / Sc, sc_ = tapj - rc_ * Sc;
    int cur = sl & 1;
    if (sl + 1 < nslice) {
      const unsigned short* ws = wsrc_for(sl + 1);
      cdd_stage_w64(ws, ldw, 0, wbuf(cur ^ 1));
      cdd_stage_w64(ws, ldw, 64, wbuf(cur ^ 1) + 8 * 1024);
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __syncthreads();
    // region coordinates for this tap: rr = pxh + (Rc-1) - rc,
    // cc = pxw + (Sc-1) - sc  (always in-grid; halo cells are zero)
    long abase[MI];
    int axor[MI];
    #pragma unroll
    for (int mi = 0; mi < MI; ++mi) {
      int rr = pxh[mi] + (Rc - 1) - rc_;
      int cc = pxw[mi] + (Sc - 1) - sc_;
      abase[mi] = ((long)rr * rgn_cols + cc) * Ko8 * 2;
      axor[mi] = cc & cmask;
    }
    const char* Wl = wbuf(cur);
    #pragma unroll
    for (int kc = 0; kc < 4; ++kc) {
      int jc = j * 16 + kc * 4 + fq;       // 16B co-chunk within Ko8
      bf16x8 a[MI], b[4];
      #pragma unroll
      for (int mi = 0; mi < MI; ++mi)
        a[mi] = *(const bf16x8*)(rgn + abase[mi] +
                                 ((jc ^ axor[mi]) << 4));
      int sub = kc >> 1, kslot = (kc & 1) * 4 + fq;
      #pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        b[ni] = cdd_wfrag(Wl + sub * 8 * 1024, ni * 16 + fr, kslot);
      #pragma unroll
      for (int mi = 0; mi < MI; ++mi)
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[mi], b[ni], acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();  // all waves done with wbuf(cur) before restage
  }

  // ---- epilogue: ctile in the W-buffer area; dgrad mode applies the
  // producer act' downstream, convT-fwd mode applies bias+act here ----
  unsigned short* ctile = (unsigned short*)wbuf(0);  // [BM][64]
  #pragma unroll
  for (int mi = 0; mi < MI; ++mi) {
    #pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int lc = ni * 16 + fr;
      float bv = bias_fwd != nullptr ? bias_fwd[cy * 64 + lc] : 0.f;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        int lr = wid * (MI * 16) + mi * 16 + fq * 4 + r;
        float v = acc[mi][ni][r];
        if (bias_fwd != nullptr) v = act_fwd(v + bv, act, slope);
        ctile[lr * 64 + lc] = f2bf(v);
      }
    }
  }
  __syncthreads();
  const int t2 = threadIdx.x;
  const int seg = t2 & 7;            // fixed c-block per thread
  float bsum[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  float ssum[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  float ssq[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  #pragma unroll
  for (int i = 0; i < BM / 32; ++i) {
    int piece = i * 256 + t2;        // BM*8 pieces = BM rows x 8 segs
    int row = piece >> 3;
    int hl = (int)fdiv((unsigned)row, fWc);
    int wcl = row - hl * Wc;
    int hi = qh + 2 * (hc0 + hl);
    int wi = qw + 2 * wcl;
    long addr = (((long)n * H + hi) * W + wi) * C8 + cy * 64 + seg * 8;
    s16x8 v = *(const s16x8*)(ctile + row * 64 + seg * 8);
    if (y0 != nullptr) {
      s16x8 yv = *(const s16x8*)(y0 + addr);
      #pragma unroll
      for (int jj = 0; jj < 8; ++jj) {
        float d = bf2f((unsigned short)v[jj]) *
                  act_bwd_from_y(bf2f((unsigned short)yv[jj]), act, slope);
        v[jj] = (short)f2bf(d);
        bsum[jj] += d;
      }
    }
    if (stats_part != nullptr) {
      #pragma unroll
      for (int jj = 0; jj < 8; ++jj) {
        float yv = bf2f((unsigned short)v[jj]);
        ssum[jj] += yv;
        ssq[jj] += yv * yv;
      }
    }
    *(s16x8*)(dx + addr) = v;
  }
  // Block-level seg reductions in LDS before the global atomics: naive
  // per-lane atomics (2048 lane-ops per block, 32-way duplicate
  // addresses) serialized across ALL concurrent blocks and cost
  // ~12 ms/call at the DCGAN-64 conv2 shape.  All conditions are
  // block-uniform, so the inner barriers are safe.
  float* red = (float*)rgn;  // region is dead after the MFMA loop
  const int rrow = t2 >> 3;  // 32 contributor rows per seg
  auto reduce_and_add = [&](const float* vec, float* dst) {
    #pragma unroll
    for (int jj = 0; jj < 8; ++jj)
      red[(rrow * 8 + seg) * 8 + jj] = vec[jj];
    __syncthreads();
    for (int off = 16; off > 0; off >>= 1) {
      if (rrow < off) {
        #pragma unroll
        for (int jj = 0; jj < 8; ++jj)
          red[(rrow * 8 + seg) * 8 + jj] +=
              red[((rrow + off) * 8 + seg) * 8 + jj];
      }
      __syncthreads();
    }
    if (rrow == 0) {
      #pragma unroll
      for (int jj = 0; jj < 8; ++jj)
        atomicAdd(&dst[seg * 8 + jj], red[seg * 8 + jj]);
    }
    __syncthreads();  // red is reused by the next reduction
  };
  if (y0 != nullptr && part != nullptr)
    reduce_and_add(bsum,
                   part + (long)(blockIdx.x & 2047) * C8 + cy * 64);
  if (stats_part != nullptr) {
    float* srow = stats_part + (long)(blockIdx.x & 2047) * 2 * C8;
    reduce_and_add(ssum, srow + cy * 64);
    reduce_and_add(ssq, srow + C8 + cy * 64);
  }
}

extern "C" {

// Returns the LDS bytes needed when eligible (sign flags BM: positive
// => BM=128, negative magnitude => BM=64), else 0.
int conv_dgrad_direct_eligible(int H, int W, int C8, int Ko8, long ldw,
                               int R, int S, int stride, int pad) {
  using namespace cdd;
  if (stride != 2 || C8 % 64 != 0 || Ko8 % 128 != 0 || ldw != Ko8)
    return 0;
  // R,S >= 2 keeps every parity class's tap set non-empty (Rc,Sc >= 1)
  if (R < 2 || S < 2 || R > 8 || S > 8 || (H & 1) || (W & 1)) return 0;
  int Hc = H / 2, Wc = W / 2;
  for (int bm : {128, 64}) {
    if (Wc > bm || bm % Wc != 0 || (Hc * Wc) % bm != 0) continue;
    int rows_c = bm / Wc;
    int rgn_rows = rows_c + ((R + 1) >> 1) - 1;
    int rgn_cols = Wc + ((S + 1) >> 1) - 1;
    // cap at 2 blocks/CU (160 KiB LDS): 1-block/CU shapes measured
    // slower than their dcol path (stage latency unhidden)
    int lds_b = rgn_rows * rgn_cols * Ko8 * 2 + WBUF_B;
    if (lds_b > 80 * 1024) continue;
    return bm == 128 ? lds_b : -lds_b;
  }
  return 0;
}

void launch_conv_dgrad_direct(const void* dy, const void* Wt, void* dx,
                              const void* y0, float* part,
                              const float* bias_fwd, float* stats_part,
                              const void* zp, int Nb, int H, int W, int C8,
                              int Ho, int Wo, int Ko8, long ldw, int R,
                              int S, int pad, int act, float slope,
                              int lds_flag, hipStream_t s) {
  using namespace cdd;
  int bm = lds_flag > 0 ? 128 : 64;
  int lds_b = lds_flag > 0 ? lds_flag : -lds_flag;
  int Hc = H / 2, Wc = W / 2;
  int rows_c = bm / Wc;
  int rgn_rows = rows_c + ((R + 1) >> 1) - 1;
  int rgn_cols = Wc + ((S + 1) >> 1) - 1;
  static int attr_done = 0;
  if (!attr_done) {
    (void)hipFuncSetAttribute(
        reinterpret_cast<const void*>(&conv_dgrad_direct<2>),
        hipFuncAttributeMaxDynamicSharedMemorySize, 120 * 1024);
    (void)hipFuncSetAttribute(
        reinterpret_cast<const void*>(&conv_dgrad_direct<1>),
        hipFuncAttributeMaxDynamicSharedMemorySize, 120 * 1024);
    attr_done = 1;
  }
  dim3 grid((long)Nb * Hc * Wc / bm, C8 / 64, 4);
  if (bm == 128)
    hipLaunchKernelGGL((conv_dgrad_direct<2>), grid, dim3(256), lds_b, s,
                       (const unsigned short*)dy, (const unsigned short*)Wt,
                       (unsigned short*)dx, (const unsigned short*)y0, part,
                       bias_fwd, stats_part,
                       (const unsigned short*)zp, Nb, H, W, C8, Ho, Wo, Ko8,
                       ldw, R, S, pad, act, slope, Hc, Wc, rows_c, rgn_rows,
                       rgn_cols, make_fastdiv(Wc));
  else
    hipLaunchKernelGGL((conv_dgrad_direct<1>), grid, dim3(256), lds_b, s,
                       (const unsigned short*)dy, (const unsigned short*)Wt,
                       (unsigned short*)dx, (const unsigned short*)y0, part,
                       bias_fwd, stats_part,
                       (const unsigned short*)zp, Nb, H, W, C8, Ho, Wo, Ko8,
                       ldw, R, S, pad, act, slope, Hc, Wc, rows_c, rgn_rows,
                       rgn_cols, make_fastdiv(Wc));
}

}  // extern "C"
