// bf16 MFMA GEMM kernels for gfx950 (CDNA4) — the conv/dense compute core.
//
// Two flavors cover every GEMM in the framework:
//
//  gemm_tn:  C[M][N] = act(A[M][K] . B[N][K]^T + bias)
//            both operands K-contiguous (TN form): conv/convT/dense forward
//            (A=im2col, B=weights) and data-grad (B=W^T).
//            128x128x64 tile, 4 waves, double-buffered LDS staged with
//            global_load_lds (16B), XOR-swizzled via the SOURCE address
//            (guide rule 21), mfma_f32_16x16x32_bf16 inner loop.
//
//  gemm_nt:  C[M][N] (fp32) += sum_k A[k][M] * B[k][N]
//            contraction over ROWS of both operands: weight-grad
//            (A=dOut, B=im2col). 64x64x64 tile, register-staged LDS
//            transpose, split-K over blockIdx.z with fp32 atomics.
//
// K (contraction) must be a multiple of 64 for gemm_tn (callers pad via
// im2col kpad / weight padding); gemm_nt handles arbitrary K by zero-fill.

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

#define GLDS16(gsrc, ldst)                                                    \
  __builtin_amdgcn_global_load_lds(                                          \
      (const __attribute__((address_space(1))) unsigned int*)(gsrc),          \
      (__attribute__((address_space(3))) unsigned int*)(ldst), 16, 0, 0)

// ---------------------------------------------------------------------------
// TN kernel
// ---------------------------------------------------------------------------
// LDS image per operand tile: [128 rows][64 k] bf16, rows 128 B, stored
// lane-linear by glds; the XOR swizzle slot' = slot ^ (row&7) lives on the
// global SOURCE address and on the ds_read byte offset (both-sides rule).

constexpr int TN_BM = 128, TN_BN = 128, TN_BK = 64;
constexpr int TN_TILE_B = TN_BM * TN_BK * 2;  // 16 KiB per operand tile

// stage one operand tile (A or B): 8 chunks of 16B per thread? ->
// tile = 128*64*2B = 16KB = 1024 chunks; 256 threads -> 4 glds each.
DEV_INLINE void tn_stage(const unsigned short* __restrict__ g, int row0,
                         int nrows, long ldk, int k0, char* lds) {
  const int t = threadIdx.x;
  const int wid = t >> 6;   // wave id (LDS dst base must be wave-uniform)
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

// fragment read: lane holds rows (l&15), k-octet (l>>4) of a 16x32 subtile
DEV_INLINE bf16x8 tn_frag(const char* lds, int row, int kslot) {
  int byte = row * 128 + ((kslot ^ (row & 7)) * 16);
  return *(const bf16x8*)(lds + byte);
}

__global__ __launch_bounds__(256, 2) void gemm_tn_bf16(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ B,
    unsigned short* __restrict__ C, float* __restrict__ Cf,
    const float* __restrict__ bias, int M, int N, int K, long lda, long ldb,
    int act, float slope) {
  __shared__ __attribute__((aligned(128))) char lds[4 * TN_TILE_B];
  // buffer offsets (avoid LDS pointer arrays: static-initializer limitation)
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
  const int wr = wid >> 1, wc = wid & 1;      // wave -> 64x64 sub-tile
  const int fr = lane & 15, fq = lane >> 4;   // fragment row / k-octet

  f32x4 acc[4][4];
  #pragma unroll
  for (int i = 0; i < 4; ++i)
    #pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int ntiles = K / TN_BK;
  tn_stage(A, m0, M, lda, 0, abuf(0));
  tn_stage(B, n0, N, ldb, 0, bbuf(0));
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  for (int t = 0; t < ntiles; ++t) {
    int cur = t & 1;
    if (t + 1 < ntiles) {
      tn_stage(A, m0, M, lda, (t + 1) * TN_BK, abuf(cur ^ 1));
      tn_stage(B, n0, N, ldb, (t + 1) * TN_BK, bbuf(cur ^ 1));
    }
    const char* Al = abuf(cur);
    const char* Bl = bbuf(cur);
    #pragma unroll
    for (int kc = 0; kc < 2; ++kc) {      // two K=32 chunks per BK=64
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

  // ---- epilogue: bias + activation --------------------------------
  // MFMA C-fragments are (16-col x 4-row) slivers per store: direct global
  // stores touch 32 B per 128 B line (25% write efficiency). Stage the
  // whole 128x128 tile through LDS (reusing the operand buffers) and store
  // full 256 B rows, 16 B per lane, fully coalesced.
  if (C != nullptr && (N & 7) == 0) {
    unsigned short* ctile = (unsigned short*)lds;  // [128][128] bf16, 32 KiB
    __syncthreads();  // K-loop LDS reads are done (post-barrier), reuse lds
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
        float v = act_fwd(acc[mi][ni][r] + bv, act, slope);
        if (C != nullptr)
          C[(long)row * N + col] = f2bf(v);
        else
          Cf[(long)row * N + col] = v;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// NT kernel (weight grad): C[M][N] += sum_k A[k][M]*B[k][N], fp32 out.
// ---------------------------------------------------------------------------
// LDS: transposed tiles [64 rows][72 k-pitch] bf16 (pitch 144 B: 16B-aligned
// b128 fragment reads, conflict-free across the 16-lane read groups).

constexpr int NT_BM = 128, NT_BN = 128, NT_BK = 64, NT_PITCH = 72;
constexpr int NT_TILE_E = NT_BM * NT_PITCH;  // elements per LDS tile
constexpr int NT_THREADS = 512;              // 8 waves: 2(m) x 4(n)

// Register-staged transpose load of [k0..k0+63] x [c0..c0+127] of src[K][L]
// into regs (zero-filled outside K), written later as ldsT[c][k] b32 pairs.
// Thread t owns column c = t&127 and k-pairs {kq + 8p}, p=0..7.
struct NtStageRegs {
  ushort2 v[8];
};

DEV_INLINE void nt_load(const unsigned short* __restrict__ g, int k0, int K,
                        int c0, int L, long ldl, NtStageRegs& r) {
  const int t = threadIdx.x;
  const int c = t & 127;
  const int kq = (t >> 7) * 2;      // 4 k-pair groups across 512 threads
  int gc = min(c0 + c, L - 1);
  #pragma unroll
  for (int p = 0; p < 8; ++p) {
    int k = kq + p * 8;
    unsigned short v0 = 0, v1 = 0;
    if (k0 + k < K) v0 = g[(long)(k0 + k) * ldl + gc];
    if (k0 + k + 1 < K) v1 = g[(long)(k0 + k + 1) * ldl + gc];
    r.v[p] = make_ushort2(v0, v1);
  }
}

DEV_INLINE void nt_write(const NtStageRegs& r,
                         unsigned short* __restrict__ lds) {
  const int t = threadIdx.x;
  const int c = t & 127;
  const int kq = (t >> 7) * 2;
  #pragma unroll
  for (int p = 0; p < 8; ++p)
    *(ushort2*)(&lds[c * NT_PITCH + kq + p * 8]) = r.v[p];
}

__global__ __launch_bounds__(NT_THREADS, 1) void gemm_nt_bf16(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ B,
    float* __restrict__ C, int M, int N, int K, long lda, long ldb,
    int kchunks_per_block, int use_atomic) {
  __shared__ __attribute__((aligned(128))) unsigned short lds[4 * NT_TILE_E];
  auto albuf = [&](int i) -> unsigned short* {
    return lds + (i ? 2 * NT_TILE_E : 0);
  };
  auto blbuf = [&](int i) -> unsigned short* {
    return lds + NT_TILE_E + (i ? 2 * NT_TILE_E : 0);
  };

  const int m0 = blockIdx.x * NT_BM;
  const int n0 = blockIdx.y * NT_BN;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wr = wid >> 2, wc = wid & 3;    // wave -> 64(m) x 32(n) sub-tile
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

  NtStageRegs ra, rb;
  nt_load(A, kt0 * NT_BK, K, m0, M, lda, ra);
  nt_load(B, kt0 * NT_BK, K, n0, N, ldb, rb);
  nt_write(ra, albuf(0));
  nt_write(rb, blbuf(0));
  __syncthreads();

  int cur = 0;
  for (int kt = kt0; kt < kt1; ++kt) {
    // issue next tile's loads early: HBM latency hides under the MFMAs
    if (kt + 1 < kt1) {
      nt_load(A, (kt + 1) * NT_BK, K, m0, M, lda, ra);
      nt_load(B, (kt + 1) * NT_BK, K, n0, N, ldb, rb);
    }
    const unsigned short* Al = albuf(cur);
    const unsigned short* Bl = blbuf(cur);
    #pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      bf16x8 a[4], b[2];
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        a[mi] = *(const bf16x8*)(&Al[(wr * 64 + mi * 16 + fr) * NT_PITCH +
                                     kc * 32 + fq * 8]);
      #pragma unroll
      for (int ni = 0; ni < 2; ++ni)
        b[ni] = *(const bf16x8*)(&Bl[(wc * 32 + ni * 16 + fr) * NT_PITCH +
                                     kc * 32 + fq * 8]);
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        #pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[mi], b[ni], acc[mi][ni], 0, 0, 0);
    }
    if (kt + 1 < kt1) {
      // disjoint buffer: no barrier needed before the write pass
      nt_write(ra, albuf(cur ^ 1));
      nt_write(rb, blbuf(cur ^ 1));
    }
    __syncthreads();
    cur ^= 1;
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

void launch_gemm_tn(const void* A, const void* B, void* C_bf16, float* C_f32,
                    const float* bias, int M, int N, int K, long lda, long ldb,
                    int act, float slope, hipStream_t s) {
  dim3 grid(ceil_div(M, TN_BM), ceil_div(N, TN_BN));
  hipLaunchKernelGGL(gemm_tn_bf16, grid, dim3(256), 0, s,
                     (const unsigned short*)A, (const unsigned short*)B,
                     (unsigned short*)C_bf16, C_f32, bias, M, N, K, lda, ldb,
                     act, slope);
}

void launch_gemm_nt(const void* A, const void* B, float* C, int M, int N,
                    int K, long lda, long ldb, int splitk, hipStream_t s) {
  int kchunks = (K + NT_BK - 1) / NT_BK;
  if (splitk > kchunks) splitk = kchunks;
  int per_block = (kchunks + splitk - 1) / splitk;
  dim3 grid(ceil_div(M, NT_BM), ceil_div(N, NT_BN), splitk);
  hipLaunchKernelGGL(gemm_nt_bf16, grid, dim3(NT_THREADS), 0, s,
                     (const unsigned short*)A, (const unsigned short*)B, C, M,
                     N, K, lda, ldb, per_block, splitk > 1 ? 1 : 0);
}

}  // extern "C"
