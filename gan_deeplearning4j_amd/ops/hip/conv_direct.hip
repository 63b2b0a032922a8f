// Direct small-C convolution forward for gfx950 — the conv1 family.
//
// The implicit-GEMM gather is TA-address-bound when C8 = 8: a staged 16B
// chunk is one tap's 8 channels, so every global_load_lds issues 64
// SCATTERED addresses (measured 74-125 TF on the first conv regardless
// of GEMM structure; profiles/gemm8p_ab.md probes).  Here the block
// stages the padded INPUT IMAGE REGION for its 128 output positions into
// LDS once — fully coalesced 16B chunks, zero-page redirect for the halo
// — and the MFMA A-fragments are ds_read_b128 DIRECTLY from that image
// (one tap's 8 channels per fragment, address = a few VALU ops).  The
// whole K (R*S*8 padded to <=256) is consumed in one shot: no K-loop
// staging at all.
//
// Tile: 128 np x 64 Kout, 4 waves (4M x 1N), 2 blocks/CU.
// Eligibility (launcher): mode-0 forward, C8 == 8, Kout <= 64,
// K = rup64(R*S*8) <= 256, (Ho*Wo) % 128 == 0 (a block never crosses a
// sample), image region rows*(W+2*pad)*16B <= 24 KiB.
// Fallbacks cover everything else (gemm.hip / gemm8p.hip).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

#define GLDS16(gsrc, ldst)                                                    \
  __builtin_amdgcn_global_load_lds(                                          \
      (const __attribute__((address_space(1))) unsigned int*)(gsrc),          \
      (__attribute__((address_space(3))) unsigned int*)(ldst), 16, 0, 0)

namespace cdir {
constexpr int BM = 128, BN = 64, BK = 64;
constexpr int IMG_B = 24 * 1024;   // padded image region budget
constexpr int BT_B = 8 * 1024;     // one 64-row x 64-k B tile
// LDS: image 24K + B 4 x 8K = 56K -> 2 blocks/CU; ctile (16K) reuses B.
constexpr int LDS_B = IMG_B + 4 * BT_B;
}  // namespace cdir

// B tile: 64 rows x 64 k per K-tile, same swizzled [row][8 slot] image
// and fragment read as gemm.hip's tn layout.
DEV_INLINE void cdir_stage_b64(const unsigned short* __restrict__ g,
                               int nrows, long ldk, int k0, char* lds) {
  const int t = threadIdx.x;
  const int wid = t >> 6;
  #pragma unroll
  for (int i = 0; i < 2; ++i) {
    int chunk = i * 256 + t;  // 512 chunks = row*8 + slot
    int row = chunk >> 3;
    int slot = chunk & 7;
    int gslot = slot ^ (row & 7);
    int grow = min(row, nrows - 1);
    const unsigned short* src = g + (long)grow * ldk + k0 + gslot * 8;
    char* dst = lds + (i * 256 + wid * 64) * 16;
    GLDS16(src, dst);
  }
}

DEV_INLINE bf16x8 cdir_bfrag(const char* lds, int row, int kslot) {
  int byte = row * 128 + ((kslot ^ (row & 7)) * 16);
  return *(const bf16x8*)(lds + byte);
}

// geometry: Wp = W + 2*pad (padded width), rows = (ho_span-1)*stride + R
__global__ __launch_bounds__(256, 2) void conv_direct_smallc(
    const unsigned short* __restrict__ img,   // [N][H][W][8] bf16
    const unsigned short* __restrict__ Wp,    // [Kout][kpad] packed weights
    unsigned short* __restrict__ C,           // [N*Ho*Wo][Kout] bf16
    const float* __restrict__ bias,
    const unsigned short* __restrict__ zp,    // 16B zero page
    int Nb, int H, int W, int Ho, int Wo, int R, int S, int stride, int pad,
    int Kout, int kpad, int M, int act, float slope, int rows,
    FastDiv fWo, FastDiv fS) {
  using namespace cdir;
  extern __shared__ __attribute__((aligned(16))) char lds[];
  char* img_lds = lds;                       // [rows][Wp][8ch] 16B cells
  auto bbuf = [&](int t) -> char* { return lds + IMG_B + t * BT_B; };

  const int Wp_ = W + 2 * pad;
  // np tile -> (n, ho0): blocks walk np linearly; (Ho*Wo)%128==0 keeps a
  // block inside one sample
  int nwgx = gridDim.x;
  int bidx = blockIdx.x;
  if (nwgx >= 8) {
    int q = nwgx / 8, r = nwgx % 8;
    int xcd = bidx % 8, idx = bidx / 8;
    bidx = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int np0 = bidx * BM;
  // (Ho*Wo) % 128 == 0 and Wo | 128 => np0 is a whole-output-row boundary
  // within one sample; once per block, plain integer div is fine.
  const int orow = (int)(fdiv((unsigned)np0, fWo));  // global output row
  const int n = orow / Ho;
  const int ho0 = orow - n * Ho;

  // ---- stage the padded image region (zero-page halo) -----------------
  // cells = rows x Wp_ 16B chunks; top row of the region maps to input
  // row hi0 = ho0*stride - pad
  const int hi0 = ho0 * stride - pad;
  const int ncell = rows * Wp_;
  for (int cell = threadIdx.x; cell < ncell; cell += 256) {
    int rr = cell / Wp_;
    int cc = cell - rr * Wp_;
    int hi = hi0 + rr;
    int wi = cc - pad;
    const unsigned short* src = zp;
    if (hi >= 0 && hi < H && wi >= 0 && wi < W)
      src = img + (((long)n * H + hi) * W + wi) * 8;
    // glds dest is wave-uniform + lane*16: cells are assigned so that a
    // wave's 64 lanes write 64 consecutive cells
    char* dst = img_lds + (cell & ~63) * 16;
    GLDS16(src, dst);
  }
  // ---- stage all B K-tiles -------------------------------------------
  const int nt = kpad / BK;
  for (int t = 0; t < nt; ++t)
    cdir_stage_b64(Wp, Kout, kpad, t * BK, bbuf(t));
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int fr = lane & 15, fq = lane >> 4;

  f32x4 acc[2][4];
  #pragma unroll
  for (int i = 0; i < 2; ++i)
    #pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // this lane's A rows: np_local = wid*32 + mi*16 + fr
  for (int t = 0; t < nt; ++t) {
    const char* Bl = bbuf(t);
    #pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      int k = t * BK + kc * 32 + fq * 8;  // one tap's 8 channels
      bf16x8 a[2];
      int tap = k >> 3;  // (r, s) index; k < R*S*8 checked via tap
      bool valid = tap < R * S;
      int tapc = valid ? tap : R * S - 1;  // clamp: keep reads in-region
      unsigned rr_ = fdiv((unsigned)tapc, fS);
      int s_ = tapc - (int)rr_ * S;
      int r_ = (int)rr_;
      #pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        int npl = wid * 32 + mi * 16 + fr;
        int ho_l = (int)fdiv((unsigned)npl, fWo);
        int wo = npl - ho_l * Wo;
        // region-local address (halo built in): row = ho_l*stride + r,
        // col = wo*stride + s
        int cell = (ho_l * stride + r_) * Wp_ + wo * stride + s_;
        bf16x8 v = *(const bf16x8*)(img_lds + cell * 16);
        if (!valid) v = bf16x8{};
        a[mi] = v;
      }
      bf16x8 b[4];
      #pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        b[ni] = cdir_bfrag(Bl, ni * 16 + fr, kc * 4 + fq);
      #pragma unroll
      for (int mi = 0; mi < 2; ++mi)
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[mi], b[ni], acc[mi][ni], 0, 0, 0);
    }
  }

  // ---- epilogue: bias + act, ctile [128][64] in the B region ----------
  unsigned short* ctile = (unsigned short*)bbuf(0);
  __syncthreads();
  #pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
    #pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int lc = ni * 16 + fr;
      float bv = bias != nullptr ? bias[min(lc, Kout - 1)] : 0.f;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        int lr = wid * 32 + mi * 16 + fq * 4 + r;
        ctile[lr * 64 + lc] = f2bf(act_fwd(acc[mi][ni][r] + bv, act, slope));
      }
    }
  }
  __syncthreads();
  const int t2 = threadIdx.x;
  #pragma unroll
  for (int i = 0; i < 4; ++i) {
    int piece = i * 256 + t2;  // 1024 pieces = 128 rows x 8 segs
    int row = piece >> 3;
    int seg = piece & 7;
    int grow = np0 + row;
    int gcol = seg * 8;
    if (grow < M && gcol + 8 <= Kout)
      *(s16x8*)(&C[(long)grow * Kout + gcol]) =
          *(const s16x8*)(ctile + row * 64 + seg * 8);
    else if (grow < M)
      for (int j = 0; j < 8 && gcol + j < Kout; ++j)
        C[(long)grow * Kout + gcol + j] = ctile[row * 64 + gcol + j];
  }
}

extern "C" {

// Returns 1 and launches when eligible, else 0 (caller falls back).
int launch_conv_direct_smallc(const void* img, const void* Wp, void* C,
                              const float* bias, const void* zp, int Nb,
                              int H, int W, int C8, int Ho, int Wo, int R,
                              int S, int stride, int pad, int Kout, int kpad,
                              int act, float slope, hipStream_t s) {
  using namespace cdir;
  if (C8 != 8 || Kout > 64 || (Kout & 7) != 0 || kpad > 256 ||
      (kpad & 63) != 0)
    return 0;
  long hw = (long)Ho * Wo;
  if (hw % BM != 0 || Wo > BM) return 0;
  int hoperblk = BM / Wo;                   // whole output rows per block
  if (hoperblk * Wo != BM) return 0;        // Wo must divide 128
  int rows = (hoperblk - 1) * stride + R;
  int Wp_ = W + 2 * pad;
  if (rows * Wp_ * 16 > IMG_B) return 0;
  int M = (int)((long)Nb * hw);
  static int attr_done = 0;
  if (!attr_done) {
    (void)hipFuncSetAttribute(
        reinterpret_cast<const void*>(&conv_direct_smallc),
        hipFuncAttributeMaxDynamicSharedMemorySize, LDS_B);
    attr_done = 1;
  }
  dim3 grid(M / BM);
  (void)hoperblk;
  hipLaunchKernelGGL(conv_direct_smallc, grid, dim3(256), LDS_B, s,
                     (const unsigned short*)img, (const unsigned short*)Wp,
                     (unsigned short*)C, bias, (const unsigned short*)zp,
                     Nb, H, W, Ho, Wo, R, S, stride, pad, Kout, kpad, M,
                     act, slope, rows, make_fastdiv(Wo), make_fastdiv(S));
  return 1;
}

}  // extern "C"
