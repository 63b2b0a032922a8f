// Torch bindings for the gfx950 kernel library (_C extension).
// Thin layer: checks, allocation, stream plumbing; all compute is in the
// .hip kernels. Built in-tree by setup.py (hipcc, --offload-arch=gfx950).

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

// csv_loader.cpp (native data loader)
torch::Tensor csv_load(const std::string& path, int64_t skip_rows);

namespace {

struct ConvGeom {
  int N, H, W, C, Ho, Wo, R, S, stride, pad, kpad;
};
struct PoolGeom {
  int N, H, W, C, Ho, Wo, k, stride;
};
struct FastDiv {
  unsigned long long mul;  // 64-bit: pow2 divisors overflow 32-bit magic
  int shift;
  int d;
};
struct ConvGather {
  int N, H, W, C, Ho, Wo, R, S, stride, pad, rsc, mode;
  int off_h, off_w, oH, oW, oqh, oqw;
  FastDiv fC, fS, fWo, fHo, fStride;
};

static FastDiv make_fastdiv_h(int d) {
  int l = 0;
  while ((1 << l) < d) ++l;
  FastDiv f;
  f.shift = 32 + l;
  f.mul = ((1ULL << (32 + l)) + (unsigned long long)d - 1) /
          (unsigned long long)d;
  f.d = d;
  return f;
}

static ConvGather make_gather(int N, int H, int W, int C, int Ho, int Wo,
                              int R, int S, int stride, int pad,
                              int mode = 0) {
  ConvGather g;
  g.N = N; g.H = H; g.W = W; g.C = C; g.Ho = Ho; g.Wo = Wo;
  g.R = R; g.S = S; g.stride = stride; g.pad = pad;
  g.rsc = R * S * C;
  g.mode = mode;
  g.off_h = g.off_w = 0;
  g.oH = g.oW = g.oqh = g.oqw = 0;
  g.fC = make_fastdiv_h(C);
  g.fS = make_fastdiv_h(S);
  g.fWo = make_fastdiv_h(Wo);
  g.fHo = make_fastdiv_h(Ho);
  g.fStride = make_fastdiv_h(stride);
  return g;
}

extern "C" {
int launch_gemm_tn(const void*, const void*, void*, float*, const float*,
                   int, int, int, long, long, int, float, float*,
                   hipStream_t);
int launch_gemm_tn_8p_tweak(int, const void*, const void*, void*, int, int,
                            int, long, long, hipStream_t);
int launch_gemm_tn_gather(const void*, const void*, void*, const float*,
                          int, int, int, long, int, float, ConvGather,
                          const void*, float*, hipStream_t);
int launch_col2im_dact(const void*, void*, ConvGeom, const void*, int,
                       float, float*, hipStream_t);
int conv_dgrad_direct_eligible(int, int, int, int, long, int, int, int,
                               int);
void launch_conv_dgrad_direct(const void*, const void*, void*, const void*,
                              float*, const float*, float*, const void*,
                              int, int, int, int, int, int, int, long, int,
                              int, int, int, float, int, hipStream_t);
int launch_col2im_stats(const void*, void*, ConvGeom, const float*, int,
                        float, float*, hipStream_t);
void launch_gemm_nt(const void*, const void*, float*, int, int, int, long,
                    long, int, int, ConvGather, const void*, hipStream_t);
void launch_im2col(const void*, void*, ConvGeom, hipStream_t);
void launch_col2im(const void*, void*, ConvGeom, const float*, int, float,
                   hipStream_t);
void launch_maxpool_fwd(const void*, void*, void*, PoolGeom, hipStream_t);
void launch_maxpool_bwd(const void*, const void*, void*, PoolGeom,
                        hipStream_t);
void launch_upsample_fwd(const void*, void*, int, int, int, int, int,
                         hipStream_t);
void launch_upsample_bwd(const void*, void*, int, int, int, int, int,
                         hipStream_t);
void launch_act_fwd(const void*, void*, long, int, float, hipStream_t);
void launch_act_bwd(const void*, const void*, void*, long, int, float,
                    hipStream_t);
void launch_col_sum(const void*, float*, long, int, hipStream_t);
void launch_bce_fwd(const void*, const void*, float*, long, hipStream_t);
void launch_bce_bwd(const void*, const void*, void*, float, long, hipStream_t);
void launch_softmax_xent_fwd(const void*, const void*, float*, void*, int,
                             int, hipStream_t);
void launch_softmax_xent_bwd(const void*, const void*, void*, float, long,
                             hipStream_t);
void launch_bn_stats(const void*, long, int, float*, float*, hipStream_t);
int launch_bn_stats_part(const void*, long, int, float*, hipStream_t);
void launch_bn_stats_sum2(const float*, int, int, float*, float*,
                          hipStream_t);
int launch_bn_bwd_reduce_part(const void*, const void*, long, int,
                              const float*, const float*, float*,
                              hipStream_t);
void launch_bn_bwd_sum2(const float*, int, int, float*, float*, hipStream_t);
int launch_col_sum_part(const void*, float*, long, int, hipStream_t);
void launch_col_sum_sum2(const float*, int, int, float*, hipStream_t);
int launch_act_bwd_bias(const void*, const void*, void*, float*, long, int,
                        int, float, hipStream_t);
void launch_bn_finalize(const float*, const float*, long, int, float, float,
                        float*, float*, float*, float*, hipStream_t);
void launch_bn_apply(const void*, void*, long, int, const float*,
                     const float*, const float*, const float*, hipStream_t);
void launch_bn_apply_eval(const void*, void*, long, int, const float*,
                          const float*, const float*, const float*, float,
                          hipStream_t);
void launch_bn_bwd_reduce(const void*, const void*, long, int, const float*,
                          const float*, float*, float*, hipStream_t);
int launch_bn_bwd_apply_act(const void*, const void*, void*, long, int,
                            const float*, const float*, const float*,
                            const float*, const float*, int, float, float*,
                            hipStream_t);
void launch_bn_bwd_apply(const void*, const void*, void*, long, int,
                         const float*, const float*, const float*,
                         const float*, const float*, hipStream_t);
void launch_fused_adam(void*, const void*, float*, float*, float*, long, int,
                       int, float, float, float, float, float, float, int,
                       const int*, hipStream_t);
void launch_amax(const void*, long, unsigned*, hipStream_t);
void launch_fp8_make_scale(const unsigned*, float*, float*, hipStream_t);
void launch_quant_fp8(const void*, void*, long, const float*, hipStream_t);
void launch_quant_fp8_delayed(const void*, void*, long, const float*,
                              unsigned*, hipStream_t);
void launch_fp8_roll_scale(unsigned*, float*, float*, hipStream_t);
void launch_gemm_tn_fp8(const void*, const void*, void*, const float*,
                        const float*, const float*, int, int, int, long,
                        long, int, float, int, ConvGather, const void*,
                        hipStream_t);
void launch_fused_rmsprop(void*, const void*, float*, float*, long, int, int,
                          float, float, float, float, float, hipStream_t);
}

hipStream_t cur_stream() {
  return (hipStream_t)at::cuda::getCurrentCUDAStream().stream();
}

void check_bf16(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

void check_f32(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be fp32");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

// ------------------------------------------------------------------ GEMM
torch::Tensor gemm_tn(torch::Tensor A, torch::Tensor B,
                      c10::optional<torch::Tensor> bias, int64_t act,
                      double slope, bool out_f32) {
  check_bf16(A, "A");
  check_bf16(B, "B");
  int64_t M = A.size(0), K = A.size(1), N = B.size(0);
  TORCH_CHECK(B.size(1) == K, "K mismatch");
  TORCH_CHECK(K % 64 == 0, "K must be padded to a multiple of 64, got ", K);
  const float* bias_p = nullptr;
  if (bias.has_value() && bias->defined() && bias->numel() > 0) {
    check_f32(*bias, "bias");
    TORCH_CHECK(bias->numel() == N, "bias size");
    bias_p = bias->data_ptr<float>();
  }
  auto opts = A.options();
  torch::Tensor C = torch::empty(
      {M, N}, out_f32 ? opts.dtype(torch::kFloat32) : opts);
  launch_gemm_tn(A.data_ptr(), B.data_ptr(),
                 out_f32 ? nullptr : C.data_ptr(),
                 out_f32 ? C.data_ptr<float>() : nullptr, bias_p, (int)M,
                 (int)N, (int)K, K, K, (int)act, (float)slope, nullptr,
                 cur_stream());
  return C;
}

// TN GEMM with fused BN statistics of the output: returns {y, sum, sumsq}
std::vector<torch::Tensor> gemm_tn_stats(torch::Tensor A, torch::Tensor B,
                                         c10::optional<torch::Tensor> bias,
                                         int64_t act, double slope) {
  check_bf16(A, "A");
  check_bf16(B, "B");
  int64_t M = A.size(0), K = A.size(1), N = B.size(0);
  TORCH_CHECK(B.size(1) == K && K % 64 == 0 && N % 8 == 0);
  const float* bias_p = nullptr;
  if (bias.has_value() && bias->defined() && bias->numel() > 0) {
    check_f32(*bias, "bias");
    bias_p = bias->data_ptr<float>();
  }
  auto f32 = A.options().dtype(torch::kFloat32);
  torch::Tensor C = torch::empty({M, N}, A.options());
  int64_t gx = (M + 127) / 128;
  torch::Tensor part = torch::empty({gx, 2 * N}, f32);
  torch::Tensor sum = torch::empty({N}, f32), sumsq = torch::empty({N}, f32);
  auto st = cur_stream();
  launch_gemm_tn(A.data_ptr(), B.data_ptr(), C.data_ptr(), nullptr, bias_p,
                 (int)M, (int)N, (int)K, K, K, (int)act, (float)slope,
                 part.data_ptr<float>(), st);
  launch_bn_stats_sum2(part.data_ptr<float>(), (int)gx, (int)N,
                       sum.data_ptr<float>(), sumsq.data_ptr<float>(), st);
  return {C, sum, sumsq};
}

torch::Tensor gemm_nt(torch::Tensor A, torch::Tensor B, int64_t splitk,
                      c10::optional<torch::Tensor> zero_page) {
  // C[M][N] = sum_k A[k][M]*B[k][N], fp32 out.
  // Operand row pitches must be multiples of 8 elements (16B-aligned rows
  // for global_load_lds staging).
  check_bf16(A, "A");
  check_bf16(B, "B");
  int64_t K = A.size(0), M = A.size(1), N = B.size(1);
  TORCH_CHECK(B.size(0) == K, "K mismatch");
  TORCH_CHECK(M % 8 == 0 && N % 8 == 0,
              "gemm_nt operand widths must be multiples of 8 (pad)");
  torch::Tensor zp = zero_page.has_value() && zero_page->defined()
                         ? *zero_page
                         : torch::zeros({16}, A.options());
  // splitk==1 writes every element exactly once: skip the zero-fill pass
  torch::Tensor C =
      splitk > 1 ? torch::zeros({M, N}, A.options().dtype(torch::kFloat32))
                 : torch::empty({M, N}, A.options().dtype(torch::kFloat32));
  ConvGather dummy{};
  launch_gemm_nt(A.data_ptr(), B.data_ptr(), C.data_ptr<float>(), (int)M,
                 (int)N, (int)K, M, N, (int)splitk, 0, dummy, zp.data_ptr(),
                 cur_stream());
  return C;
}

// Implicit-GEMM conv forward: y2d[NP][Kout] = act(im2col(x).Wp^T + bias)
// without materializing col. x NHWC [Nb,H,W,C] (C % 8 == 0), Wp [Kout][kpad].
std::vector<torch::Tensor> conv_fwd_implicit(torch::Tensor x, torch::Tensor Wp,
                                c10::optional<torch::Tensor> bias,
                                torch::Tensor zero_page, int64_t Nb,
                                int64_t H, int64_t W, int64_t C, int64_t Ho,
                                int64_t Wo, int64_t R, int64_t S,
                                int64_t stride, int64_t pad, int64_t act,
                                double slope, int64_t mode,
                                int64_t want_stats) {
  check_bf16(x, "x");
  check_bf16(Wp, "Wp");
  TORCH_CHECK(C % 8 == 0, "implicit conv needs C % 8 == 0");
  int64_t Kout = Wp.size(0), kpad = Wp.size(1);
  int64_t M = Nb * Ho * Wo;
  const float* bias_p = nullptr;
  if (bias.has_value() && bias->defined() && bias->numel() > 0) {
    check_f32(*bias, "bias");
    bias_p = bias->data_ptr<float>();
  }
  torch::Tensor y = torch::empty({M, Kout}, x.options());
  ConvGather g = make_gather((int)Nb, (int)H, (int)W, (int)C, (int)Ho,
                             (int)Wo, (int)R, (int)S, (int)stride, (int)pad,
                             (int)mode);
  auto st = cur_stream();
  if (want_stats && Kout % 8 == 0) {
    auto f32 = x.options().dtype(torch::kFloat32);
    int64_t gx = (M + 127) / 128;
    torch::Tensor part = torch::empty({gx, 2 * Kout}, f32);
    torch::Tensor sum = torch::empty({Kout}, f32);
    torch::Tensor sumsq = torch::empty({Kout}, f32);
    launch_gemm_tn_gather(x.data_ptr(), Wp.data_ptr(), y.data_ptr(), bias_p,
                          (int)M, (int)Kout, (int)kpad, kpad, (int)act,
                          (float)slope, g, zero_page.data_ptr(),
                          part.data_ptr<float>(), st);
    launch_bn_stats_sum2(part.data_ptr<float>(), (int)((M + 127) / 128),
                         (int)Kout, sum.data_ptr<float>(),
                         sumsq.data_ptr<float>(), st);
    return std::vector<torch::Tensor>{y, sum, sumsq};
  }
  launch_gemm_tn_gather(x.data_ptr(), Wp.data_ptr(), y.data_ptr(), bias_p,
                        (int)M, (int)Kout, (int)kpad, kpad, (int)act,
                        (float)slope, g, zero_page.data_ptr(), nullptr, st);
  return std::vector<torch::Tensor>{y};
}

// Implicit weight-grad: C[M][N] += sum_np A'[np][M] * B'[np][N] where the
// operand selected by gmode (1=A, 2=B) is the im2col of an NHWC image.
torch::Tensor gemm_nt_implicit(torch::Tensor A, torch::Tensor B,
                               int64_t gmode, int64_t Mdim, int64_t Ndim,
                               int64_t Kdim, torch::Tensor img_dims,
                               int64_t splitk, torch::Tensor zero_page) {
  check_bf16(A, "A");
  check_bf16(B, "B");
  auto d = img_dims.cpu().contiguous();
  TORCH_CHECK(d.numel() == 10 || d.numel() == 11,
              "img_dims = [N,H,W,C,Ho,Wo,R,S,stride,pad(,mode)]");
  const int64_t* p = d.data_ptr<int64_t>();
  ConvGather g = make_gather((int)p[0], (int)p[1], (int)p[2], (int)p[3],
                             (int)p[4], (int)p[5], (int)p[6], (int)p[7],
                             (int)p[8], (int)p[9],
                             d.numel() == 11 ? (int)p[10] : 0);
  torch::Tensor C =
      splitk > 1
          ? torch::zeros({Mdim, Ndim}, A.options().dtype(torch::kFloat32))
          : torch::empty({Mdim, Ndim}, A.options().dtype(torch::kFloat32));
  long lda = gmode == 1 ? 0 : Mdim;
  long ldb = gmode == 2 ? 0 : Ndim;
  launch_gemm_nt(A.data_ptr(), B.data_ptr(), C.data_ptr<float>(), (int)Mdim,
                 (int)Ndim, (int)Kdim, lda, ldb, (int)splitk, (int)gmode, g,
                 zero_page.data_ptr(), cur_stream());
  return C;
}

// Parity-decomposed transposed conv (stride s): the output splits into
// s*s parity classes; each class is a dense gathered GEMM over its valid
// (r2,s2) taps (no 4x zero-tap inflation, no dcol/col2im round trip).
// Covers conv data-grad AND convT forward. wcls: one [Nout][kpad_class]
// weight pack per class, ordered qh-major.
torch::Tensor conv_parity_implicit(
    torch::Tensor img, std::vector<torch::Tensor> wcls,
    c10::optional<torch::Tensor> bias, torch::Tensor zero_page, int64_t Nb,
    int64_t Hs, int64_t Ws, int64_t Cs, int64_t oHd, int64_t oWd,
    int64_t Nout, int64_t R, int64_t S, int64_t stride, int64_t pad,
    int64_t act, double slope) {
  check_bf16(img, "img");
  TORCH_CHECK(Cs % 8 == 0, "parity conv needs source C % 8 == 0");
  TORCH_CHECK((int64_t)wcls.size() == stride * stride, "one pack per class");
  const float* bias_p = nullptr;
  if (bias.has_value() && bias->defined() && bias->numel() > 0) {
    check_f32(*bias, "bias");
    bias_p = bias->data_ptr<float>();
  }
  torch::Tensor y = torch::empty({Nb * oHd * oWd, Nout}, img.options());
  int s_ = (int)stride;
  for (int qh = 0; qh < s_; ++qh) {
    for (int qw = 0; qw < s_; ++qw) {
      torch::Tensor& wq = wcls[qh * s_ + qw];
      check_bf16(wq, "wcls");
      int prh = (int)((qh + pad) % stride);
      int prw = (int)((qw + pad) % stride);
      int R2 = (int)((R - prh + stride - 1) / stride);
      int S2 = (int)((S - prw + stride - 1) / stride);
      int Hq = (int)((oHd - qh + stride - 1) / stride);
      int Wq = (int)((oWd - qw + stride - 1) / stride);
      if (Hq <= 0 || Wq <= 0 || R2 <= 0 || S2 <= 0) continue;
      TORCH_CHECK(wq.size(0) == Nout, "pack Nout");
      ConvGather g = make_gather((int)Nb, (int)Hs, (int)Ws, (int)Cs, Hq, Wq,
                                 R2, S2, (int)stride, (int)pad, 2);
      g.off_h = (int)((qh + pad - prh) / stride);
      g.off_w = (int)((qw + pad - prw) / stride);
      g.oH = (int)oHd;
      g.oW = (int)oWd;
      g.oqh = qh;
      g.oqw = qw;
      int M = (int)(Nb * Hq * Wq);
      int K = (int)wq.size(1);
      launch_gemm_tn_gather(img.data_ptr(), wq.data_ptr(), y.data_ptr(),
                            bias_p, M, (int)Nout, K, K, (int)act,
                            (float)slope, g, zero_page.data_ptr(), nullptr,
                            cur_stream());
    }
  }
  return y;
}

// ------------------------------------------------------------------ conv
torch::Tensor im2col(torch::Tensor x, int64_t N, int64_t H, int64_t W,
                     int64_t C, int64_t Ho, int64_t Wo, int64_t R, int64_t S,
                     int64_t stride, int64_t pad, int64_t kpad) {
  check_bf16(x, "x");
  ConvGeom g{(int)N, (int)H, (int)W, (int)C, (int)Ho, (int)Wo,
             (int)R, (int)S, (int)stride, (int)pad, (int)kpad};
  torch::Tensor col = torch::empty({N * Ho * Wo, kpad}, x.options());
  launch_im2col(x.data_ptr(), col.data_ptr(), g, cur_stream());
  return col;
}

torch::Tensor col2im(torch::Tensor dcol, int64_t N, int64_t H, int64_t W,
                     int64_t C, int64_t Ho, int64_t Wo, int64_t R, int64_t S,
                     int64_t stride, int64_t pad, int64_t kpad,
                     c10::optional<torch::Tensor> bias, int64_t act,
                     double slope) {
  check_bf16(dcol, "dcol");
  ConvGeom g{(int)N, (int)H, (int)W, (int)C, (int)Ho, (int)Wo,
             (int)R, (int)S, (int)stride, (int)pad, (int)kpad};
  const float* bias_p = nullptr;
  if (bias.has_value() && bias->defined() && bias->numel() > 0) {
    check_f32(*bias, "bias");
    bias_p = bias->data_ptr<float>();
  }
  torch::Tensor out = torch::empty({N, H, W, C}, dcol.options());
  launch_col2im(dcol.data_ptr(), out.data_ptr(), g, bias_p, (int)act,
                (float)slope, cur_stream());
  return out;
}

// col2im with the producer conv's activation backward fused (strided
// dgrad): y0 is the producer's activation output [N,H,W,C]; returns
// {dpre_nhwc, db} where db is the producer's bias gradient.
std::vector<torch::Tensor> col2im_dact(
    torch::Tensor dcol, int64_t N, int64_t H, int64_t W, int64_t C,
    int64_t Ho, int64_t Wo, int64_t R, int64_t S, int64_t stride,
    int64_t pad, int64_t kpad, torch::Tensor y0, int64_t act, double slope,
    bool want_bias) {
  check_bf16(dcol, "dcol");
  check_bf16(y0, "y0");
  TORCH_CHECK(C % 8 == 0 && kpad % 8 == 0, "col2im_dact needs C,kpad %8==0");
  TORCH_CHECK(y0.numel() == N * H * W * C, "y0 shape");
  ConvGeom g{(int)N, (int)H, (int)W, (int)C, (int)Ho, (int)Wo,
             (int)R, (int)S, (int)stride, (int)pad, (int)kpad};
  auto f32 = dcol.options().dtype(torch::kFloat32);
  torch::Tensor out = torch::empty({N, H, W, C}, dcol.options());
  torch::Tensor db = torch::empty({want_bias ? C : 0}, f32);
  torch::Tensor part;
  float* part_p = nullptr;
  auto s = cur_stream();
  if (want_bias) {
    part = torch::empty({2048, C}, f32);
    part_p = part.data_ptr<float>();
  }
  int gx = launch_col2im_dact(dcol.data_ptr(), out.data_ptr(), g,
                              y0.data_ptr(), (int)act, (float)slope, part_p,
                              s);
  if (want_bias)
    launch_col_sum_sum2(part.data_ptr<float>(), gx, (int)C,
                        db.data_ptr<float>(), s);
  return {out, db};
}

// Direct parity-decomposed strided dgrad (conv_dgrad_direct.hip):
// replaces dcol GEMM + col2im(_dact) when eligible.  dpre is the 2D
// post-act-bwd gradient [N*Ho*Wo][Ko8]; wt the [R*S*C8][ldw] transposed
// weight pack; y0 (optional) the producer's activation output for the
// fused act backward.  Returns {} when ineligible (caller falls back),
// else {dx_nhwc} or {dx_nhwc, producer_db}.
std::vector<torch::Tensor> conv_dgrad_direct(
    torch::Tensor dpre, torch::Tensor wt, c10::optional<torch::Tensor> y0,
    torch::Tensor zero_page, int64_t N, int64_t H, int64_t W, int64_t C8,
    int64_t Ho, int64_t Wo, int64_t R, int64_t S, int64_t stride,
    int64_t pad, int64_t act, double slope, bool want_bias) {
  check_bf16(dpre, "dpre");
  check_bf16(wt, "wt");
  int64_t Ko8 = dpre.size(1);
  long ldw = (long)wt.size(1);
  int el = conv_dgrad_direct_eligible((int)H, (int)W, (int)C8, (int)Ko8,
                                      ldw, (int)R, (int)S, (int)stride,
                                      (int)pad);
  if (el == 0) return {};
  TORCH_CHECK(dpre.size(0) == N * Ho * Wo, "dpre rows");
  TORCH_CHECK(wt.size(0) >= R * S * C8, "wt rows");
  auto st = cur_stream();
  torch::Tensor out = torch::empty({N, H, W, C8}, dpre.options());
  const void* y0p = nullptr;
  float* part_p = nullptr;
  torch::Tensor part, db;
  if (y0.has_value() && y0->defined()) {
    check_bf16(*y0, "y0");
    TORCH_CHECK(y0->numel() == N * H * W * C8, "y0 shape");
    y0p = y0->data_ptr();
    if (want_bias) {
      auto f32 = dpre.options().dtype(torch::kFloat32);
      part = torch::zeros({2048, C8}, f32);
      part_p = part.data_ptr<float>();
      db = torch::empty({C8}, f32);
    }
  }
  launch_conv_dgrad_direct(dpre.data_ptr(), wt.data_ptr(), out.data_ptr(),
                           y0p, part_p, nullptr, nullptr,
                           zero_page.data_ptr(), (int)N, (int)H, (int)W,
                           (int)C8, (int)Ho, (int)Wo, (int)Ko8, ldw,
                           (int)R, (int)S, (int)pad, (int)act,
                           (float)slope, el, st);
  if (part_p != nullptr) {
    launch_col_sum_sum2(part_p, 2048, (int)C8, db.data_ptr<float>(), st);
    return {out, db};
  }
  return {out};
}

// Strided transposed-conv FORWARD through the same parity-direct
// kernel: x2d is the flattened input [N*Hi*Wi][Cin-pad], w2a the
// [R*S*Co8][Cin-pad] pack (gpu_ops "w2a" cache), bias f32 [Co8].
// Returns {} when ineligible, {y_nhwc} or {y_nhwc, sum, sumsq}
// (fused BN statistics) otherwise.
std::vector<torch::Tensor> conv_transpose_fwd_direct(
    torch::Tensor x2d, torch::Tensor w2a, c10::optional<torch::Tensor> bias,
    torch::Tensor zero_page, int64_t N, int64_t Hi, int64_t Wi,
    int64_t Ho, int64_t Wo, int64_t Co8, int64_t R, int64_t S,
    int64_t stride, int64_t pad, int64_t act, double slope,
    bool want_stats) {
  check_bf16(x2d, "x2d");
  check_bf16(w2a, "w2a");
  int64_t Kin = x2d.size(1);       // padded Cin = contraction K
  long ldw = (long)w2a.size(1);
  int el = conv_dgrad_direct_eligible((int)Ho, (int)Wo, (int)Co8, (int)Kin,
                                      ldw, (int)R, (int)S, (int)stride,
                                      (int)pad);
  if (el == 0) return {};
  TORCH_CHECK(x2d.size(0) == N * Hi * Wi, "x2d rows");
  TORCH_CHECK(w2a.size(0) >= R * S * Co8, "w2a rows");
  auto st = cur_stream();
  torch::Tensor y = torch::empty({N, Ho, Wo, Co8}, x2d.options());
  const float* bias_p = nullptr;
  if (bias.has_value() && bias->defined() && bias->numel() > 0) {
    check_f32(*bias, "bias");
    TORCH_CHECK(bias->numel() >= Co8, "bias length");
    bias_p = bias->data_ptr<float>();
  }
  torch::Tensor zero_bias;
  if (bias_p == nullptr) {
    // the kernel keys "apply act_fwd" off the bias pointer
    zero_bias = torch::zeros({Co8},
                             x2d.options().dtype(torch::kFloat32));
    bias_p = zero_bias.data_ptr<float>();
  }
  float* stats_p = nullptr;
  torch::Tensor part, ssum, ssq;
  if (want_stats) {
    auto f32 = x2d.options().dtype(torch::kFloat32);
    part = torch::zeros({2048, 2 * Co8}, f32);
    stats_p = part.data_ptr<float>();
    ssum = torch::empty({Co8}, f32);
    ssq = torch::empty({Co8}, f32);
  }
  launch_conv_dgrad_direct(x2d.data_ptr(), w2a.data_ptr(), y.data_ptr(),
                           nullptr, nullptr, bias_p, stats_p,
                           zero_page.data_ptr(), (int)N, (int)Ho, (int)Wo,
                           (int)Co8, (int)Hi, (int)Wi, (int)Kin, ldw,
                           (int)R, (int)S, (int)pad, (int)act,
                           (float)slope, el, st);
  if (stats_p != nullptr) {
    launch_bn_stats_sum2(stats_p, 2048, (int)Co8, ssum.data_ptr<float>(),
                         ssq.data_ptr<float>(), st);
    return {y, ssum, ssq};
  }
  return {y};
}

// col2im with fused BN stats: returns {y_nhwc, sum, sumsq}
std::vector<torch::Tensor> col2im_stats(
    torch::Tensor dcol, int64_t N, int64_t H, int64_t W, int64_t C,
    int64_t Ho, int64_t Wo, int64_t R, int64_t S, int64_t stride,
    int64_t pad, int64_t kpad, c10::optional<torch::Tensor> bias,
    int64_t act, double slope) {
  check_bf16(dcol, "dcol");
  TORCH_CHECK(C % 8 == 0 && kpad % 8 == 0);
  ConvGeom g{(int)N, (int)H, (int)W, (int)C, (int)Ho, (int)Wo,
             (int)R, (int)S, (int)stride, (int)pad, (int)kpad};
  const float* bias_p = nullptr;
  if (bias.has_value() && bias->defined() && bias->numel() > 0) {
    check_f32(*bias, "bias");
    bias_p = bias->data_ptr<float>();
  }
  auto f32 = dcol.options().dtype(torch::kFloat32);
  torch::Tensor out = torch::empty({N, H, W, C}, dcol.options());
  torch::Tensor part = torch::empty({2048, 2 * C}, f32);
  torch::Tensor sum = torch::empty({C}, f32), sumsq = torch::empty({C}, f32);
  auto st = cur_stream();
  int gx = launch_col2im_stats(dcol.data_ptr(), out.data_ptr(), g, bias_p,
                               (int)act, (float)slope,
                               part.data_ptr<float>(), st);
  launch_bn_stats_sum2(part.data_ptr<float>(), gx, (int)C,
                       sum.data_ptr<float>(), sumsq.data_ptr<float>(), st);
  return {out, sum, sumsq};
}

// ------------------------------------------------------------------ pool
std::vector<torch::Tensor> maxpool_fwd(torch::Tensor x, int64_t N, int64_t H,
                                       int64_t W, int64_t C, int64_t k,
                                       int64_t stride) {
  check_bf16(x, "x");
  int Ho = (int)((H - k) / stride + 1), Wo = (int)((W - k) / stride + 1);
  PoolGeom g{(int)N, (int)H, (int)W, (int)C, Ho, Wo, (int)k, (int)stride};
  torch::Tensor out = torch::empty({N, Ho, Wo, C}, x.options());
  torch::Tensor am = torch::empty({N, Ho, Wo, C},
                                  x.options().dtype(torch::kUInt8));
  launch_maxpool_fwd(x.data_ptr(), out.data_ptr(), am.data_ptr(), g,
                     cur_stream());
  return {out, am};
}

torch::Tensor maxpool_bwd(torch::Tensor dout, torch::Tensor argmax, int64_t N,
                          int64_t H, int64_t W, int64_t C, int64_t k,
                          int64_t stride) {
  check_bf16(dout, "dout");
  int Ho = (int)((H - k) / stride + 1), Wo = (int)((W - k) / stride + 1);
  PoolGeom g{(int)N, (int)H, (int)W, (int)C, Ho, Wo, (int)k, (int)stride};
  torch::Tensor din = torch::empty({N, H, W, C}, dout.options());
  launch_maxpool_bwd(dout.data_ptr(), argmax.data_ptr(), din.data_ptr(), g,
                     cur_stream());
  return din;
}

torch::Tensor upsample_fwd(torch::Tensor x, int64_t N, int64_t H, int64_t W,
                           int64_t C, int64_t scale) {
  check_bf16(x, "x");
  torch::Tensor out = torch::empty({N, H * scale, W * scale, C}, x.options());
  launch_upsample_fwd(x.data_ptr(), out.data_ptr(), (int)N, (int)H, (int)W,
                      (int)C, (int)scale, cur_stream());
  return out;
}

torch::Tensor upsample_bwd(torch::Tensor dout, int64_t N, int64_t H,
                           int64_t W, int64_t C, int64_t scale) {
  check_bf16(dout, "dout");
  torch::Tensor din = torch::empty({N, H, W, C}, dout.options());
  launch_upsample_bwd(dout.data_ptr(), din.data_ptr(), (int)N, (int)H, (int)W,
                      (int)C, (int)scale, cur_stream());
  return din;
}

// ----------------------------------------------------------- elementwise
torch::Tensor act_fwd(torch::Tensor x, int64_t act, double slope) {
  check_bf16(x, "x");
  torch::Tensor y = torch::empty_like(x);
  launch_act_fwd(x.data_ptr(), y.data_ptr(), x.numel(), (int)act,
                 (float)slope, cur_stream());
  return y;
}

// fused act-backward + bias grad: returns (dpre, dbias_f32)
std::vector<torch::Tensor> act_bwd_bias(torch::Tensor dy,
                                        c10::optional<torch::Tensor> y,
                                        int64_t act, double slope) {
  check_bf16(dy, "dy");
  int64_t m = dy.size(0), n = dy.size(1);
  TORCH_CHECK(n % 8 == 0, "act_bwd_bias needs N % 8 == 0");
  auto f32 = dy.options().dtype(torch::kFloat32);
  torch::Tensor dx = torch::empty_like(dy);
  torch::Tensor scratch = torch::empty({2048, n}, f32);
  torch::Tensor db = torch::empty({n}, f32);
  const void* y_p = nullptr;
  if (act != 0) {
    TORCH_CHECK(y.has_value(), "non-identity act needs y");
    check_bf16(*y, "y");
    y_p = y->data_ptr();
  }
  auto s = cur_stream();
  int gx = launch_act_bwd_bias(dy.data_ptr(), y_p, dx.data_ptr(),
                               scratch.data_ptr<float>(), (long)m, (int)n,
                               (int)act, (float)slope, s);
  launch_col_sum_sum2(scratch.data_ptr<float>(), gx, (int)n,
                      db.data_ptr<float>(), s);
  return {dx, db};
}

torch::Tensor act_bwd(torch::Tensor dy, torch::Tensor y, int64_t act,
                      double slope) {
  check_bf16(dy, "dy");
  check_bf16(y, "y");
  torch::Tensor dx = torch::empty_like(dy);
  launch_act_bwd(dy.data_ptr(), y.data_ptr(), dx.data_ptr(), dy.numel(),
                 (int)act, (float)slope, cur_stream());
  return dx;
}

torch::Tensor col_sum(torch::Tensor a) {
  check_bf16(a, "a");
  int64_t m = a.size(0), n = a.size(1);
  auto f32 = a.options().dtype(torch::kFloat32);
  if (n % 8 == 0) {
    torch::Tensor out = torch::empty({n}, f32);
    torch::Tensor scratch = torch::empty({2048, n}, f32);
    int gx = launch_col_sum_part(a.data_ptr(), scratch.data_ptr<float>(),
                                 (long)m, (int)n, cur_stream());
    launch_col_sum_sum2(scratch.data_ptr<float>(), gx, (int)n,
                        out.data_ptr<float>(), cur_stream());
    return out;
  }
  torch::Tensor out = torch::zeros({n}, f32);
  launch_col_sum(a.data_ptr(), out.data_ptr<float>(), (long)m, (int)n,
                 cur_stream());
  return out;
}

// ----------------------------------------------------------------- losses
torch::Tensor bce_fwd(torch::Tensor logits, torch::Tensor labels) {
  check_bf16(logits, "logits");
  check_bf16(labels, "labels");
  torch::Tensor s = torch::zeros({1}, logits.options().dtype(torch::kFloat32));
  launch_bce_fwd(logits.data_ptr(), labels.data_ptr(), s.data_ptr<float>(),
                 logits.numel(), cur_stream());
  return s;
}

torch::Tensor bce_bwd(torch::Tensor logits, torch::Tensor labels,
                      double gscale) {
  check_bf16(logits, "logits");
  torch::Tensor d = torch::empty_like(logits);
  launch_bce_bwd(logits.data_ptr(), labels.data_ptr(), d.data_ptr(),
                 (float)gscale, logits.numel(), cur_stream());
  return d;
}

std::vector<torch::Tensor> softmax_xent_fwd(torch::Tensor logits,
                                            torch::Tensor onehot) {
  check_bf16(logits, "logits");
  check_bf16(onehot, "onehot");
  int64_t rows = logits.size(0), cols = logits.size(1);
  torch::Tensor s = torch::zeros({1}, logits.options().dtype(torch::kFloat32));
  torch::Tensor probs = torch::empty_like(logits);
  launch_softmax_xent_fwd(logits.data_ptr(), onehot.data_ptr(),
                          s.data_ptr<float>(), probs.data_ptr(), (int)rows,
                          (int)cols, cur_stream());
  return {s, probs};
}

torch::Tensor softmax_xent_bwd(torch::Tensor probs, torch::Tensor onehot,
                               double gscale) {
  check_bf16(probs, "probs");
  torch::Tensor d = torch::empty_like(probs);
  launch_softmax_xent_bwd(probs.data_ptr(), onehot.data_ptr(), d.data_ptr(),
                          (float)gscale, probs.numel(), cur_stream());
  return d;
}

// -------------------------------------------------------------------- BN
std::vector<torch::Tensor> bn_fwd_train(torch::Tensor x, torch::Tensor gamma,
                                        torch::Tensor beta,
                                        torch::Tensor running_mean,
                                        torch::Tensor running_var,
                                        double momentum, double eps) {
  check_bf16(x, "x");
  check_f32(gamma, "gamma");
  int64_t c = x.size(-1);
  long m = x.numel() / c;
  auto f32 = x.options().dtype(torch::kFloat32);
  torch::Tensor mean = torch::empty({c}, f32), istd = torch::empty({c}, f32);
  torch::Tensor y = torch::empty_like(x);
  auto s = cur_stream();
  torch::Tensor sum, sumsq;
  if (c % 8 == 0) {
    sum = torch::empty({c}, f32);
    sumsq = torch::empty({c}, f32);
    torch::Tensor scratch = torch::empty({2048, 2 * c}, f32);
    int gx = launch_bn_stats_part(x.data_ptr(), m, (int)c,
                                  scratch.data_ptr<float>(), s);
    launch_bn_stats_sum2(scratch.data_ptr<float>(), gx, (int)c,
                         sum.data_ptr<float>(), sumsq.data_ptr<float>(), s);
  } else {
    sum = torch::zeros({c}, f32);
    sumsq = torch::zeros({c}, f32);
    launch_bn_stats(x.data_ptr(), m, (int)c, sum.data_ptr<float>(),
                    sumsq.data_ptr<float>(), s);
  }
  launch_bn_finalize(sum.data_ptr<float>(), sumsq.data_ptr<float>(), m,
                     (int)c, (float)eps, (float)momentum,
                     mean.data_ptr<float>(), istd.data_ptr<float>(),
                     running_mean.defined() && running_mean.numel() > 0
                         ? running_mean.data_ptr<float>()
                         : nullptr,
                     running_var.defined() && running_var.numel() > 0
                         ? running_var.data_ptr<float>()
                         : nullptr,
                     s);
  launch_bn_apply(x.data_ptr(), y.data_ptr(), m, (int)c,
                  mean.data_ptr<float>(), istd.data_ptr<float>(),
                  gamma.data_ptr<float>(), beta.data_ptr<float>(), s);
  return {y, mean, istd};
}

// training forward from PRE-COMPUTED sums (fused producer statistics)
std::vector<torch::Tensor> bn_fwd_train_pre(
    torch::Tensor x, torch::Tensor sum, torch::Tensor sumsq,
    torch::Tensor gamma, torch::Tensor beta, torch::Tensor running_mean,
    torch::Tensor running_var, double momentum, double eps) {
  check_bf16(x, "x");
  check_f32(sum, "sum");
  check_f32(sumsq, "sumsq");
  int64_t c = x.size(-1);
  long m = x.numel() / c;
  auto f32 = x.options().dtype(torch::kFloat32);
  torch::Tensor mean = torch::empty({c}, f32), istd = torch::empty({c}, f32);
  torch::Tensor y = torch::empty_like(x);
  auto s = cur_stream();
  launch_bn_finalize(sum.data_ptr<float>(), sumsq.data_ptr<float>(), m,
                     (int)c, (float)eps, (float)momentum,
                     mean.data_ptr<float>(), istd.data_ptr<float>(),
                     running_mean.defined() && running_mean.numel() > 0
                         ? running_mean.data_ptr<float>()
                         : nullptr,
                     running_var.defined() && running_var.numel() > 0
                         ? running_var.data_ptr<float>()
                         : nullptr,
                     s);
  launch_bn_apply(x.data_ptr(), y.data_ptr(), m, (int)c,
                  mean.data_ptr<float>(), istd.data_ptr<float>(),
                  gamma.data_ptr<float>(), beta.data_ptr<float>(), s);
  return {y, mean, istd};
}

torch::Tensor bn_fwd_eval(torch::Tensor x, torch::Tensor gamma,
                          torch::Tensor beta, torch::Tensor rm,
                          torch::Tensor rv, double eps) {
  check_bf16(x, "x");
  int64_t c = x.size(-1);
  long m = x.numel() / c;
  torch::Tensor y = torch::empty_like(x);
  launch_bn_apply_eval(x.data_ptr(), y.data_ptr(), m, (int)c,
                       rm.data_ptr<float>(), rv.data_ptr<float>(),
                       gamma.data_ptr<float>(), beta.data_ptr<float>(),
                       (float)eps, cur_stream());
  return y;
}

std::vector<torch::Tensor> bn_bwd(torch::Tensor x, torch::Tensor dy,
                                  torch::Tensor mean, torch::Tensor istd,
                                  torch::Tensor gamma) {
  check_bf16(x, "x");
  check_bf16(dy, "dy");
  int64_t c = x.size(-1);
  long m = x.numel() / c;
  auto f32 = x.options().dtype(torch::kFloat32);
  torch::Tensor dx = torch::empty_like(x);
  auto s = cur_stream();
  torch::Tensor dgamma, dbeta;
  if (c % 8 == 0) {
    dgamma = torch::empty({c}, f32);
    dbeta = torch::empty({c}, f32);
    torch::Tensor scratch = torch::empty({2048, 2 * c}, f32);
    int gx = launch_bn_bwd_reduce_part(x.data_ptr(), dy.data_ptr(), m,
                                       (int)c, mean.data_ptr<float>(),
                                       istd.data_ptr<float>(),
                                       scratch.data_ptr<float>(), s);
    launch_bn_bwd_sum2(scratch.data_ptr<float>(), gx, (int)c,
                       dgamma.data_ptr<float>(), dbeta.data_ptr<float>(), s);
  } else {
    dgamma = torch::zeros({c}, f32);
    dbeta = torch::zeros({c}, f32);
    launch_bn_bwd_reduce(x.data_ptr(), dy.data_ptr(), m, (int)c,
                         mean.data_ptr<float>(), istd.data_ptr<float>(),
                         dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                         s);
  }
  launch_bn_bwd_apply(x.data_ptr(), dy.data_ptr(), dx.data_ptr(), m, (int)c,
                      mean.data_ptr<float>(), istd.data_ptr<float>(),
                      gamma.data_ptr<float>(), dgamma.data_ptr<float>(),
                      dbeta.data_ptr<float>(), s);
  return {dx, dgamma, dbeta};
}

// BN backward with the producer activation's backward fused into the
// apply pass: returns {dpre, dgamma, dbeta, db_bias}. x must be the
// producer's ACTIVATION OUTPUT (= BN input); dpre is the gradient at the
// producer's pre-activation, db_bias its bias gradient (column sums of
// dpre). Requires C % 8 == 0.
std::vector<torch::Tensor> bn_bwd_act(torch::Tensor x, torch::Tensor dy,
                                      torch::Tensor mean, torch::Tensor istd,
                                      torch::Tensor gamma, int64_t act,
                                      double slope, bool want_bias) {
  check_bf16(x, "x");
  check_bf16(dy, "dy");
  int64_t c = x.size(-1);
  long m = x.numel() / c;
  TORCH_CHECK(c % 8 == 0, "bn_bwd_act needs C % 8 == 0");
  auto f32 = x.options().dtype(torch::kFloat32);
  torch::Tensor dx = torch::empty_like(x);
  auto s = cur_stream();
  torch::Tensor dgamma = torch::empty({c}, f32);
  torch::Tensor dbeta = torch::empty({c}, f32);
  torch::Tensor scratch = torch::empty({2048, 2 * c}, f32);
  int gx = launch_bn_bwd_reduce_part(x.data_ptr(), dy.data_ptr(), m, (int)c,
                                     mean.data_ptr<float>(),
                                     istd.data_ptr<float>(),
                                     scratch.data_ptr<float>(), s);
  launch_bn_bwd_sum2(scratch.data_ptr<float>(), gx, (int)c,
                     dgamma.data_ptr<float>(), dbeta.data_ptr<float>(), s);
  torch::Tensor db = torch::empty({want_bias ? c : 0}, f32);
  torch::Tensor bpart;
  float* bpart_p = nullptr;
  if (want_bias) {
    bpart = torch::empty({2048, c}, f32);
    bpart_p = bpart.data_ptr<float>();
  }
  int gx2 = launch_bn_bwd_apply_act(
      x.data_ptr(), dy.data_ptr(), dx.data_ptr(), m, (int)c,
      mean.data_ptr<float>(), istd.data_ptr<float>(),
      gamma.data_ptr<float>(), dgamma.data_ptr<float>(),
      dbeta.data_ptr<float>(), (int)act, (float)slope, bpart_p, s);
  if (want_bias)
    launch_col_sum_sum2(bpart.data_ptr<float>(), gx2, (int)c,
                        db.data_ptr<float>(), s);
  return {dx, dgamma, dbeta, db};
}

// -------------------------------------------------------------------- fp8
std::vector<torch::Tensor> fp8_quantize(torch::Tensor x) {
  // bf16 -> e4m3 with per-tensor dynamic amax scaling.
  // Returns {u8 data, scale, inv_scale} (scale/inv stay on device).
  check_bf16(x, "x");
  TORCH_CHECK(x.numel() % 8 == 0, "fp8 quantize needs numel % 8 == 0");
  auto f32 = x.options().dtype(torch::kFloat32);
  torch::Tensor amax = torch::zeros({1}, x.options().dtype(torch::kInt32));
  torch::Tensor scale = torch::empty({1}, f32), inv = torch::empty({1}, f32);
  torch::Tensor y = torch::empty(x.sizes(), x.options().dtype(torch::kUInt8));
  auto s = cur_stream();
  launch_amax(x.data_ptr(), x.numel(), (unsigned*)amax.data_ptr(), s);
  launch_fp8_make_scale((const unsigned*)amax.data_ptr(),
                        scale.data_ptr<float>(), inv.data_ptr<float>(), s);
  launch_quant_fp8(x.data_ptr(), y.data_ptr(), x.numel(),
                   scale.data_ptr<float>(), s);
  return {y, scale, inv};
}

torch::Tensor fp8_quantize_delayed(torch::Tensor x, torch::Tensor scale,
                                   torch::Tensor inv, torch::Tensor amax,
                                   bool first) {
  // Delayed-scaling quantize (TransformerEngine-style): quantize with the
  // scale rolled from the PREVIOUS step's amax (cvt saturates at +-448 so
  // a stale scale clips, never overflows) while accumulating this
  // tensor's amax for the next roll.  first=true bootstraps with an
  // exact two-pass amax.
  check_bf16(x, "x");
  TORCH_CHECK(x.numel() % 8 == 0, "fp8 quantize needs numel % 8 == 0");
  torch::Tensor y = torch::empty(x.sizes(), x.options().dtype(torch::kUInt8));
  auto s = cur_stream();
  if (first) {
    launch_amax(x.data_ptr(), x.numel(), (unsigned*)amax.data_ptr(), s);
    launch_fp8_roll_scale((unsigned*)amax.data_ptr(),
                          scale.data_ptr<float>(), inv.data_ptr<float>(), s);
    launch_quant_fp8_delayed(x.data_ptr(), y.data_ptr(), x.numel(),
                             scale.data_ptr<float>(),
                             (unsigned*)amax.data_ptr(), s);
  } else {
    launch_fp8_roll_scale((unsigned*)amax.data_ptr(),
                          scale.data_ptr<float>(), inv.data_ptr<float>(), s);
    launch_quant_fp8_delayed(x.data_ptr(), y.data_ptr(), x.numel(),
                             scale.data_ptr<float>(),
                             (unsigned*)amax.data_ptr(), s);
  }
  return y;
}

torch::Tensor gemm_tn_fp8(torch::Tensor Aq, torch::Tensor Bq,
                          torch::Tensor inv_qa, torch::Tensor inv_qb,
                          c10::optional<torch::Tensor> bias, int64_t act,
                          double slope) {
  // C[M][N] bf16 = descale * (Aq . Bq^T) — both operands e4m3, K-major,
  // K padded to 128 (the double-rate fp8 MFMA's K depth)
  TORCH_CHECK(Aq.scalar_type() == torch::kUInt8 && Aq.is_contiguous());
  TORCH_CHECK(Bq.scalar_type() == torch::kUInt8 && Bq.is_contiguous());
  TORCH_CHECK(Aq.size(1) == Bq.size(1), "K mismatch");
  TORCH_CHECK(Aq.size(1) % 128 == 0, "fp8 K must be padded to 128");
  int64_t M = Aq.size(0), N = Bq.size(0), K = Aq.size(1);
  const float* bias_p = nullptr;
  if (bias.has_value() && bias->defined() && bias->numel() > 0) {
    check_f32(*bias, "bias");
    bias_p = bias->data_ptr<float>();
  }
  torch::Tensor C = torch::empty({M, N},
                                 Aq.options().dtype(torch::kBFloat16));
  ConvGather dummy{};
  launch_gemm_tn_fp8(Aq.data_ptr(), Bq.data_ptr(), C.data_ptr(), bias_p,
                     inv_qa.data_ptr<float>(), inv_qb.data_ptr<float>(),
                     (int)M, (int)N, (int)K, K, K, (int)act, (float)slope,
                     0, dummy, nullptr, cur_stream());
  return C;
}

torch::Tensor conv_fwd_implicit_fp8(
    torch::Tensor xq, torch::Tensor wq, c10::optional<torch::Tensor> bias,
    torch::Tensor inv_qx, torch::Tensor inv_qw, torch::Tensor zero_page,
    int64_t Nb, int64_t H, int64_t W, int64_t C, int64_t Ho, int64_t Wo,
    int64_t R, int64_t S, int64_t stride, int64_t pad, int64_t act,
    double slope, int64_t mode) {
  TORCH_CHECK(xq.scalar_type() == torch::kUInt8 && xq.is_contiguous());
  TORCH_CHECK(wq.scalar_type() == torch::kUInt8 && wq.is_contiguous());
  TORCH_CHECK(C % 16 == 0, "fp8 implicit conv needs C % 16 == 0");
  int64_t Kout = wq.size(0), kpad = wq.size(1);
  TORCH_CHECK(kpad % 128 == 0, "fp8 pack K must be padded to 128");
  int64_t M = Nb * Ho * Wo;
  const float* bias_p = nullptr;
  if (bias.has_value() && bias->defined() && bias->numel() > 0) {
    check_f32(*bias, "bias");
    bias_p = bias->data_ptr<float>();
  }
  torch::Tensor y = torch::empty({M, Kout},
                                 xq.options().dtype(torch::kBFloat16));
  ConvGather g = make_gather((int)Nb, (int)H, (int)W, (int)C, (int)Ho,
                             (int)Wo, (int)R, (int)S, (int)stride, (int)pad,
                             (int)mode);
  launch_gemm_tn_fp8(xq.data_ptr(), wq.data_ptr(), y.data_ptr(), bias_p,
                     inv_qx.data_ptr<float>(), inv_qw.data_ptr<float>(),
                     (int)M, (int)Kout, (int)kpad, 0, kpad, (int)act,
                     (float)slope, 1, g, zero_page.data_ptr(), cur_stream());
  return y;
}

torch::Tensor gemm_tn_8p_tweak(torch::Tensor A, torch::Tensor B,
                               int64_t tweak) {
  check_bf16(A, "A");
  check_bf16(B, "B");
  int64_t M = A.size(0), N = B.size(0), K = A.size(1);
  TORCH_CHECK(B.size(1) == K && K % 64 == 0);
  torch::Tensor C = torch::empty({M, N}, A.options());
  launch_gemm_tn_8p_tweak((int)tweak, A.data_ptr(), B.data_ptr(),
                          C.data_ptr(), (int)M, (int)N, (int)K, K, K,
                          cur_stream());
  return C;
}

// ------------------------------------------------------------------ optim
void fused_adam(torch::Tensor param, torch::Tensor grad,
                c10::optional<torch::Tensor> master, torch::Tensor m,
                torch::Tensor v, double lr, double b1, double b2, double eps,
                double clip, double l2, int64_t t,
                c10::optional<torch::Tensor> t_dev) {
  TORCH_CHECK(param.is_cuda() && param.is_contiguous());
  int param_bf16 = param.scalar_type() == torch::kBFloat16;
  int grad_bf16 = grad.scalar_type() == torch::kBFloat16;
  float* master_p = nullptr;
  if (param_bf16) {
    TORCH_CHECK(master.has_value(), "bf16 param needs fp32 master");
    master_p = master->data_ptr<float>();
  }
  const int* t_p = nullptr;
  if (t_dev.has_value() && t_dev->defined()) {
    TORCH_CHECK(t_dev->scalar_type() == torch::kInt32, "t_dev must be int32");
    t_p = t_dev->data_ptr<int>();
  }
  launch_fused_adam(param.data_ptr(), grad.data_ptr(), master_p,
                    m.data_ptr<float>(), v.data_ptr<float>(), param.numel(),
                    grad_bf16, param_bf16, (float)lr, (float)b1, (float)b2,
                    (float)eps, (float)clip, (float)l2, (int)t, t_p,
                    cur_stream());
}

void fused_rmsprop(torch::Tensor param, torch::Tensor grad,
                   c10::optional<torch::Tensor> master, torch::Tensor v,
                   double lr, double decay, double eps, double clip,
                   double l2) {
  TORCH_CHECK(param.is_cuda() && param.is_contiguous());
  int param_bf16 = param.scalar_type() == torch::kBFloat16;
  int grad_bf16 = grad.scalar_type() == torch::kBFloat16;
  float* master_p = nullptr;
  if (param_bf16) {
    TORCH_CHECK(master.has_value(), "bf16 param needs fp32 master");
    master_p = master->data_ptr<float>();
  }
  launch_fused_rmsprop(param.data_ptr(), grad.data_ptr(), master_p,
                       v.data_ptr<float>(), param.numel(), grad_bf16,
                       param_bf16, (float)lr, (float)decay, (float)eps,
                       (float)clip, (float)l2, cur_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("gemm_tn", &gemm_tn, "C = act(A.B^T + bias) bf16 MFMA");
  mod.def("gemm_tn_8p_tweak", &gemm_tn_8p_tweak,
          "within-probe A/B entry for 8p kernel variants");
  mod.def("gemm_nt", &gemm_nt, "C = A^T.B (contraction over rows) fp32 out");
  mod.def("conv_fwd_implicit", &conv_fwd_implicit,
          "implicit-GEMM conv forward (gathered im2col A)");
  mod.def("conv_parity_implicit", &conv_parity_implicit,
          "parity-decomposed strided transposed conv / conv dgrad");
  mod.def("gemm_nt_implicit", &gemm_nt_implicit,
          "weight-grad GEMM with one operand gathered as im2col");
  mod.def("fp8_quantize_delayed", &fp8_quantize_delayed,
          "bf16 -> e4m3 with delayed per-tensor scaling (fused amax)");
  mod.def("gemm_tn_fp8", &gemm_tn_fp8,
          "C = descale*(Aq.Bq^T) e4m3 MFMA, bf16 out");
  mod.def("fp8_quantize", &fp8_quantize,
          "bf16 -> e4m3 (OCP) with dynamic per-tensor scale");
  mod.def("conv_fwd_implicit_fp8", &conv_fwd_implicit_fp8,
          "implicit-GEMM conv forward on fp8 MFMA");
  mod.def("im2col", &im2col);
  mod.def("col2im", &col2im);
  mod.def("col2im_stats", &col2im_stats,
          "col2im with fused bias+act+BN statistics");
  mod.def("gemm_tn_stats", &gemm_tn_stats,
          "TN GEMM with fused output BN statistics");
  mod.def("maxpool_fwd", &maxpool_fwd);
  mod.def("maxpool_bwd", &maxpool_bwd);
  mod.def("upsample_fwd", &upsample_fwd);
  mod.def("upsample_bwd", &upsample_bwd);
  mod.def("act_fwd", &act_fwd);
  mod.def("act_bwd", &act_bwd);
  mod.def("act_bwd_bias", &act_bwd_bias,
          "fused activation backward + bias gradient");
  mod.def("col_sum", &col_sum);
  mod.def("bce_fwd", &bce_fwd);
  mod.def("bce_bwd", &bce_bwd);
  mod.def("softmax_xent_fwd", &softmax_xent_fwd);
  mod.def("softmax_xent_bwd", &softmax_xent_bwd);
  mod.def("bn_fwd_train", &bn_fwd_train);
  mod.def("bn_fwd_train_pre", &bn_fwd_train_pre,
          "BN training forward from producer-fused statistics");
  mod.def("bn_fwd_eval", &bn_fwd_eval);
  mod.def("bn_bwd", &bn_bwd);
  mod.def("bn_bwd_act", &bn_bwd_act);
  mod.def("col2im_dact", &col2im_dact);
  mod.def("conv_dgrad_direct", &conv_dgrad_direct,
          "direct parity-decomposed strided dgrad ({} when ineligible)");
  mod.def("conv_transpose_fwd_direct", &conv_transpose_fwd_direct,
          "strided convT forward via the parity-direct kernel");
  mod.def("fused_adam", &fused_adam);
  mod.def("fused_rmsprop", &fused_rmsprop);
  mod.def("csv_load", &csv_load, "multithreaded CSV -> fp32 tensor");
  mod.attr("arch") = "gfx950";
}
