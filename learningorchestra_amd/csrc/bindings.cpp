// PyTorch bindings for the learningorchestra_amd gfx950 kernel library.
// Thin argument-checking wrappers; all compute is in the .hip translation
// units. Built in-tree (build_ext.py) as learningorchestra_amd/_lo_C.so.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <hip/hip_runtime.h>

namespace lo {
struct ConvGeom {
  int B, H, W, C;
  int KH, KW, SH, SW, PH, PW;
  int OH, OW;
  unsigned mgOW = 0, mgOH = 0;
  int sOW = 0, sOH = 0;
};

// magic-division constants (see gemm.hip fdiv): exact for n < 2^31
static inline void fill_fastdiv(lo::ConvGeom& g) {
  auto mk = [](unsigned d, unsigned& mg, int& s) {
    if (d <= 1) { mg = 0; s = 0; return; }
    s = 0;
    while ((1u << s) < d) ++s;
    if ((1u << s) == d) { mg = 1; return; }          // power of two: q = n>>s
    // mg = ceil(2^(31+s)/d) fits u32 (2^s/d in (1,2)); device computes
    // q = mulhi(n, mg) >> (s-1) == (n*mg) >> (31+s), exact for n < 2^31
    const unsigned long long L = 1ull << (31 + s);
    mg = (unsigned)((L + d - 1) / d);
    s = s - 1;
  };
  mk((unsigned)g.OW, g.mgOW, g.sOW);
  mk((unsigned)g.OH, g.mgOH, g.sOH);
}
struct GemmArgs {
  const void *A, *B;
  void* C;
  const float* bias;
  long lda, ldb, ldc;
  int M, N, K;
  bool ta, tb;
  int epi;
  bool out_f32;
  int splits;
  int gather;
  ConvGeom geom;
  float* stats_sum = nullptr;
  float* stats_sumsq = nullptr;
  const void* addend = nullptr;
};
bool gemm_dispatch(const GemmArgs& g, hipStream_t s);
void launch_mfma_probe(const void* A, const void* B, float* D, hipStream_t s);
void launch_im2col(const void* in, void* col, int B, int H, int W, int C,
                   int KH, int KW, int SH, int SW, int PH, int PW,
                   int OH, int OW, int Kpad, hipStream_t s);
bool launch_conv_fwd_small(const void* x, const void* w, long ldw,
                           const void* bias, void* y, long ldy, int B, int H,
                           int W, int C, int KH, int KW, int SH, int SW,
                           int PH, int PW, int OH, int OW, int outC, int relu,
                           hipStream_t s);
bool launch_conv1d_fwd(const void* x, const void* w, long ldw,
                       const void* bias, void* y, long ldy, int B, int H,
                       int C, int KH, int PH, int OH, int outC, int relu,
                       hipStream_t s);
bool launch_conv1d_dx(const void* dy2, long ldy, const void* wt, long ldw,
                      void* dx, int B, int H, int C, int KH, int PH, int OH,
                      int outC, int accumulate, hipStream_t s);
bool launch_conv_dw_c1(const void* dy2, long ldy, const void* x, void* dw,
                       long ldw, int B, int H, int W, int KH, int KW, int SH,
                       int SW, int PH, int PW, int OH, int OW, int outC,
                       hipStream_t s);
bool launch_conv_dw_smallc(const void* dy2, long ldy, const void* x, void* dw,
                           long ldw, int B, int H, int W, int C, int KH,
                           int KW, int SH, int SW, int PH, int PW, int OH,
                           int OW, int outC, hipStream_t s);
bool launch_conv_dx(const void* dy2, long ldy, const void* wt, long ldw,
                    void* dx, int B, int H, int W, int C, int KH, int KW,
                    int SH, int SW, int PH, int PW, int OH, int OW, int outC,
                    hipStream_t s);
void launch_col2im(const void* dcol, void* dx, int B, int H, int W, int C,
                   int KH, int KW, int SH, int SW, int PH, int PW, int OH,
                   int OW, int Kpad, hipStream_t s);
void launch_maxpool_fwd(const void* in, void* out, void* idx, int B, int H, int W,
                        int C, int KH, int KW, int SH, int SW, int PH, int PW,
                        int OH, int OW, int relu_sentinel, hipStream_t s);
void launch_maxpool_bwd(const void* dy, const void* idx, void* dx,
                        const void* relu_y, int B, int H,
                        int W, int C, int KH, int KW, int SH, int SW, int PH,
                        int PW, int OH, int OW, hipStream_t s);
void launch_bn_stats(const void* x, void* sum, void* sumsq, long M, int C,
                     hipStream_t s);
void launch_bn_fwd(const void* x, void* y, const void* mean, const void* invstd,
                   const void* gamma, const void* beta, const void* residual,
                   long M, int C, int relu, hipStream_t s);
void launch_bn_bwd_reduce(const void* dy, const void* y, const void* x,
                          const void* mean, const void* invstd, void* dbeta,
                          void* dgamma, long M, int C, int relu, hipStream_t s);
void launch_bn_finalize_stats(const void* scratch, void* mean, void* invstd,
                              void* rmean, void* rvar, float invM,
                              float momentum, float eps, int C,
                              hipStream_t s);
void launch_bn_bwd_dx(const void* dy, const void* y, const void* x, void* dx,
                      const void* mean, const void* invstd, const void* gamma,
                      const void* dbeta, const void* dgamma, long M, int C,
                      int relu, hipStream_t s);
void launch_add_relu(const void* a, const void* b, void* z, long n, int relu,
                     hipStream_t s);
void launch_avgpool_global(const void* x, void* out, int B, int HW, int C,
                           hipStream_t s);
void launch_avgpool_global_bwd(const void* dy, void* dx, int B, int HW, int C,
                               hipStream_t s);
void launch_relu_bwd(const void* dy, const void* y, void* dx, long n, hipStream_t s);
void launch_sgd(void* master, const void* grad, void* mom, void* mirror, long n,
                float lr, float mu, float wd, float gscale, hipStream_t s);
void launch_adam(void* master, const void* grad, void* m1, void* m2, void* mirror,
                 const void* step_dev, long n, float lr, float b1, float b2,
                 float eps, float wd, float gscale, hipStream_t s);
void launch_colsum(const void* dy, void* out, long M, int N, long ldy, hipStream_t s);
void launch_colsum_masked(const void* dy, const void* mask, void* out, long M,
                          int N, long ldy, hipStream_t s);
void launch_argmax_rows(const void* x, void* out, long M, int C, int Cvalid,
                        long ldx, hipStream_t s);
void launch_accuracy_count(const void* pred, const void* label, void* out, long M,
                           hipStream_t s);
void launch_softmax_ce(const void* logits, const void* labels, void* dlogits,
                       void* loss_sum, void* correct, long M, int C, int Cvalid,
                       float gscale, hipStream_t s);
void launch_tree_hist(const void* binned, const void* node_of, const void* grad,
                      const void* hess, void* hist, long N, int F, int n_nodes,
                      int B, double gbound, double hbound, hipStream_t s);
void launch_embedding_fwd(const void* ids, const void* table, void* out,
                          long n, int dim, hipStream_t s);
void launch_embedding_bwd(const void* ids, const void* dy, void* gtable,
                          long n, int dim, hipStream_t s);
}  // namespace lo

namespace {

hipStream_t stream() { return at::hip::getCurrentHIPStream().stream(); }

void check_bf16(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == at::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

void check_f32(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == at::kFloat, name, " must be fp32");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

// gemm: C = op_ta(A) @ op_tb(B) (+bias)(+relu). Returns false if no native
// config covers the shape (python falls back to a library GEMM).
bool gemm(at::Tensor A, at::Tensor B, at::Tensor C,
          c10::optional<at::Tensor> bias, bool ta, bool tb, int64_t epi,
          int64_t splits, c10::optional<at::Tensor> stats = c10::nullopt,
          c10::optional<at::Tensor> addend = c10::nullopt) {
  check_bf16(A, "A");
  check_bf16(B, "B");
  TORCH_CHECK(C.is_cuda() && C.is_contiguous(), "C must be contiguous GPU");
  const bool out_f32 = C.scalar_type() == at::kFloat;
  TORCH_CHECK(out_f32 || C.scalar_type() == at::kBFloat16, "C must be bf16 or fp32");
  const int M = (int)C.size(0), N = (int)C.size(1);
  const int K = (int)(ta ? A.size(0) : A.size(1));
  TORCH_CHECK((int)(ta ? A.size(1) : A.size(0)) == M, "A/M mismatch");
  TORCH_CHECK((int)(tb ? B.size(1) : B.size(0)) == K, "B/K mismatch");
  TORCH_CHECK((int)(tb ? B.size(0) : B.size(1)) == N, "B/N mismatch");
  const float* bias_p = nullptr;
  if (bias.has_value()) {
    check_f32(*bias, "bias");
    TORCH_CHECK(bias->numel() == N, "bias/N mismatch");
    bias_p = bias->data_ptr<float>();
  }
  lo::GemmArgs g{A.data_ptr(), B.data_ptr(), C.data_ptr(), bias_p,
                 A.size(1), B.size(1), C.size(1), M, N, K, ta, tb,
                 (int)epi, out_f32, (int)splits, 0, {}};
  if (addend.has_value()) {
    check_bf16(*addend, "addend");
    TORCH_CHECK(!out_f32 && splits == 1, "addend needs bf16 non-split-K out");
    TORCH_CHECK(addend->sizes() == C.sizes(), "addend/C shape mismatch");
    g.addend = addend->data_ptr();
  }
  if (stats.has_value()) {
    check_f32(*stats, "stats");
    TORCH_CHECK(stats->numel() == 2 * N, "stats must be [2, N]");
    g.stats_sum = stats->data_ptr<float>();
    g.stats_sumsq = g.stats_sum + N;
  }
  return lo::gemm_dispatch(g, stream());
}

// implicit-GEMM conv forward: y2d[B*OH*OW, outC] = im2col(x) @ W^T (+bias,
// +relu) without materializing col. W is [outC, kpad].
bool gemm_conv_fwd(at::Tensor x, at::Tensor Wt, at::Tensor C,
                   c10::optional<at::Tensor> bias, bool relu,
                   int64_t KH, int64_t KW, int64_t SH, int64_t SW,
                   int64_t PH, int64_t PW) {
  check_bf16(x, "x");
  check_bf16(Wt, "W");
  check_bf16(C, "C");
  const int B = (int)x.size(0), H = (int)x.size(1), W = (int)x.size(2),
            Ci = (int)x.size(3);
  const int OH = (int)((H + 2 * PH - KH) / SH) + 1;
  const int OW = (int)((W + 2 * PW - KW) / SW) + 1;
  const int M = (int)C.size(0), N = (int)C.size(1);
  const int K = (int)Wt.size(1);  // kpad
  TORCH_CHECK(M == (long)B * OH * OW && N == Wt.size(0), "conv shapes");
  const float* bias_p = nullptr;
  if (bias.has_value()) { check_f32(*bias, "bias"); bias_p = bias->data_ptr<float>(); }
  lo::ConvGeom geom{B, H, W, Ci, (int)KH, (int)KW, (int)SH, (int)SW,
                    (int)PH, (int)PW, OH, OW};
  fill_fastdiv(geom);
  lo::GemmArgs g{x.data_ptr(), Wt.data_ptr(), C.data_ptr(), bias_p,
                 K, Wt.size(1), C.size(1), M, N, K, false, true,
                 relu ? 1 : 0, false, 1, 1, geom};
  return lo::gemm_dispatch(g, stream());
}

// implicit-GEMM conv dW: gw[outC, kpad] = dY^T @ im2col(x), fp32 split-K
bool gemm_conv_dw(at::Tensor dy2, at::Tensor x, at::Tensor gw, int64_t splits,
                  int64_t KH, int64_t KW, int64_t SH, int64_t SW,
                  int64_t PH, int64_t PW) {
  check_bf16(dy2, "dy2");
  check_bf16(x, "x");
  check_f32(gw, "gw");
  const int B = (int)x.size(0), H = (int)x.size(1), W = (int)x.size(2),
            Ci = (int)x.size(3);
  const int OH = (int)((H + 2 * PH - KH) / SH) + 1;
  const int OW = (int)((W + 2 * PW - KW) / SW) + 1;
  const int M = (int)gw.size(0), N = (int)gw.size(1);   // outC, kpad
  const long K = dy2.size(0);                           // B*OH*OW
  TORCH_CHECK(K == (long)B * OH * OW && M == dy2.size(1), "conv dW shapes");
  lo::ConvGeom geom{B, H, W, Ci, (int)KH, (int)KW, (int)SH, (int)SW,
                    (int)PH, (int)PW, OH, OW};
  fill_fastdiv(geom);
  lo::GemmArgs g{dy2.data_ptr(), x.data_ptr(), gw.data_ptr(), nullptr,
                 dy2.size(1), N, gw.size(1), M, N, (int)K, true, false,
                 0, true, (int)std::max<int64_t>(splits, 2), 2, geom};
  return lo::gemm_dispatch(g, stream());
}

at::Tensor mfma_probe(at::Tensor A, at::Tensor B) {
  check_bf16(A, "A");
  check_bf16(B, "B");
  TORCH_CHECK(A.size(0) == 16 && A.size(1) == 32 && B.size(0) == 32 && B.size(1) == 16);
  auto D = at::zeros({16, 16}, A.options().dtype(at::kFloat));
  lo::launch_mfma_probe(A.data_ptr(), B.data_ptr(), D.data_ptr<float>(), stream());
  return D;
}

at::Tensor im2col(at::Tensor in, int64_t KH, int64_t KW, int64_t SH, int64_t SW,
                  int64_t PH, int64_t PW, int64_t Kpad, at::Tensor col) {
  check_bf16(in, "in");
  check_bf16(col, "col");
  const int B = (int)in.size(0), H = (int)in.size(1), W = (int)in.size(2),
            C = (int)in.size(3);
  const int OH = (H + 2 * (int)PH - (int)KH) / (int)SH + 1;
  const int OW = (W + 2 * (int)PW - (int)KW) / (int)SW + 1;
  TORCH_CHECK(col.size(0) == (long)B * OH * OW && col.size(1) == Kpad, "col shape");
  TORCH_CHECK(Kpad >= KH * KW * C && Kpad % 8 == 0, "Kpad");
  lo::launch_im2col(in.data_ptr(), col.data_ptr(), B, H, W, C, (int)KH, (int)KW,
                    (int)SH, (int)SW, (int)PH, (int)PW, OH, OW, (int)Kpad, stream());
  return col;
}

at::Tensor col2im(at::Tensor dcol, int64_t B, int64_t H, int64_t W, int64_t C,
                  int64_t KH, int64_t KW, int64_t SH, int64_t SW, int64_t PH,
                  int64_t PW, at::Tensor dx) {
  check_bf16(dcol, "dcol");
  check_bf16(dx, "dx");
  const int OH = (int)((H + 2 * PH - KH) / SH) + 1;
  const int OW = (int)((W + 2 * PW - KW) / SW) + 1;
  TORCH_CHECK(dcol.size(0) == B * OH * OW, "dcol rows");
  lo::launch_col2im(dcol.data_ptr(), dx.data_ptr(), (int)B, (int)H, (int)W, (int)C,
                    (int)KH, (int)KW, (int)SH, (int)SW, (int)PH, (int)PW, OH, OW,
                    (int)dcol.size(1), stream());
  return dx;
}

// small-image fused conv forward: x image staged in LDS, w tiles
// double-buffered, y written once.  Returns false if shape not eligible.
bool conv_fwd_small(at::Tensor x, at::Tensor w,
                    c10::optional<at::Tensor> bias, at::Tensor y, int64_t KH,
                    int64_t KW, int64_t SH, int64_t SW, int64_t PH,
                    int64_t PW, bool relu) {
  check_bf16(x, "x");
  check_bf16(w, "w");
  check_bf16(y, "y");
  const int B = (int)x.size(0), H = (int)x.size(1), W = (int)x.size(2),
            C = (int)x.size(3);
  const int OH = (H + 2 * (int)PH - (int)KH) / (int)SH + 1;
  const int OW = (W + 2 * (int)PW - (int)KW) / (int)SW + 1;
  const int outC = (int)y.size(1);
  TORCH_CHECK(y.size(0) == (long)B * OH * OW, "y rows");
  TORCH_CHECK(w.size(0) == outC && w.size(1) >= (long)KH * KW * C, "w shape");
  const void* bp = nullptr;
  if (bias) {
    TORCH_CHECK(bias->scalar_type() == at::kFloat && bias->numel() >= outC);
    bp = bias->data_ptr();
  }
  return lo::launch_conv_fwd_small(
      x.data_ptr(), w.data_ptr(), w.stride(0), bp, y.data_ptr(), y.stride(0),
      B, H, W, C, (int)KH, (int)KW, (int)SH, (int)SW, (int)PH, (int)PW, OH,
      OW, outC, relu ? 1 : 0, stream());
}

// 1-D (sequence) conv forward: W==1, KW==1, stride 1 — h-tiled x window
// in LDS, outC in 64-wide slices.
bool conv1d_fwd(at::Tensor x, at::Tensor w, c10::optional<at::Tensor> bias,
                at::Tensor y, int64_t KH, int64_t PH, bool relu) {
  check_bf16(x, "x");
  check_bf16(w, "w");
  check_bf16(y, "y");
  const int B = (int)x.size(0), H = (int)x.size(1), C = (int)x.size(3);
  TORCH_CHECK(x.size(2) == 1, "conv1d needs W==1");
  const int OH = H + 2 * (int)PH - (int)KH + 1;
  const int outC = (int)y.size(1);
  TORCH_CHECK(y.size(0) == (long)B * OH, "y rows");
  const void* bp = nullptr;
  if (bias) bp = bias->data_ptr();
  return lo::launch_conv1d_fwd(x.data_ptr(), w.data_ptr(), w.stride(0), bp,
                               y.data_ptr(), y.stride(0), B, H, C, (int)KH,
                               (int)PH, OH, outC, relu ? 1 : 0, stream());
}

// 1-D conv dX: h-tiled LDS fp32 accumulator, non-atomic RMW scatter.
bool conv1d_dx(at::Tensor dy2, at::Tensor wt, at::Tensor dx, int64_t KH,
               int64_t PH, bool accumulate) {
  check_bf16(dy2, "dy2");
  check_bf16(wt, "wt");
  check_bf16(dx, "dx");
  const int B = (int)dx.size(0), H = (int)dx.size(1), C = (int)dx.size(3);
  TORCH_CHECK(dx.size(2) == 1, "conv1d needs W==1");
  const int OH = H + 2 * (int)PH - (int)KH + 1;
  const int outC = (int)dy2.size(1);
  TORCH_CHECK(dy2.size(0) == (long)B * OH, "dy2 rows");
  return lo::launch_conv1d_dx(dy2.data_ptr(), dy2.stride(0), wt.data_ptr(),
                              wt.stride(0), dx.data_ptr(), B, H, C, (int)KH,
                              (int)PH, OH, outC, accumulate ? 1 : 0,
                              stream());
}

// C=1 conv dW: dY^T @ im2col(x) with x images LDS-resident; dw fp32
// accumulated with one atomicAdd per element per block (dw zeroed here).
bool conv_dw_small(at::Tensor dy2, at::Tensor x, at::Tensor dw, int64_t KH,
                   int64_t KW, int64_t SH, int64_t SW, int64_t PH, int64_t PW) {
  check_bf16(dy2, "dy2");
  check_bf16(x, "x");
  TORCH_CHECK(dw.scalar_type() == at::kFloat, "dw must be fp32");
  const int B = (int)x.size(0), H = (int)x.size(1), W = (int)x.size(2),
            C = (int)x.size(3);
  const int OH = (H + 2 * (int)PH - (int)KH) / (int)SH + 1;
  const int OW = (W + 2 * (int)PW - (int)KW) / (int)SW + 1;
  const int outC = (int)dy2.size(1);
  TORCH_CHECK(dy2.size(0) == (long)B * OH * OW, "dy2 rows");
  TORCH_CHECK(dw.size(0) == outC && dw.size(1) >= KH * KW * C, "dw shape");
  dw.zero_();  // harmless extra zero when ineligible (fallback re-zeroes)
  return lo::launch_conv_dw_smallc(dy2.data_ptr(), dy2.stride(0), x.data_ptr(),
                                   dw.data_ptr(), dw.stride(0), B, H, W, C,
                                   (int)KH, (int)KW, (int)SH, (int)SW,
                                   (int)PH, (int)PW, OH, OW, outC, stream());
}

bool conv_dw_c1(at::Tensor dy2, at::Tensor x, at::Tensor dw, int64_t KH,
                int64_t KW, int64_t SH, int64_t SW, int64_t PH, int64_t PW) {
  check_bf16(dy2, "dy2");
  check_bf16(x, "x");
  TORCH_CHECK(dw.scalar_type() == at::kFloat, "dw must be fp32");
  const int B = (int)x.size(0), H = (int)x.size(1), W = (int)x.size(2);
  TORCH_CHECK(x.size(3) == 1, "conv_dw_c1 needs C=1");
  const int OH = (H + 2 * (int)PH - (int)KH) / (int)SH + 1;
  const int OW = (W + 2 * (int)PW - (int)KW) / (int)SW + 1;
  const int outC = (int)dy2.size(1);
  TORCH_CHECK(dy2.size(0) == (long)B * OH * OW, "dy2 rows");
  TORCH_CHECK(dw.size(0) == outC && dw.size(1) >= KH * KW, "dw shape");
  dw.zero_();
  return lo::launch_conv_dw_c1(dy2.data_ptr(), dy2.stride(0), x.data_ptr(),
                               dw.data_ptr(), dw.stride(0), B, H, W, (int)KH,
                               (int)KW, (int)SH, (int)SW, (int)PH, (int)PW,
                               OH, OW, outC, stream());
}

// fused conv dX: dy2 [B*OH*OW, outC] @ wt[kpad, outC]^T scattered into
// dx [B,H,W,C] through an LDS fp32 accumulator (no dcol matrix).
// Returns false when the shape is not eligible (caller falls back).
bool conv_dx(at::Tensor dy2, at::Tensor wt, at::Tensor dx, int64_t KH,
             int64_t KW, int64_t SH, int64_t SW, int64_t PH, int64_t PW) {
  check_bf16(dy2, "dy2");
  check_bf16(wt, "wt");
  check_bf16(dx, "dx");
  const int B = (int)dx.size(0), H = (int)dx.size(1), W = (int)dx.size(2),
            C = (int)dx.size(3);
  const int OH = (H + 2 * (int)PH - (int)KH) / (int)SH + 1;
  const int OW = (W + 2 * (int)PW - (int)KW) / (int)SW + 1;
  const int outC = (int)dy2.size(1);
  TORCH_CHECK(dy2.size(0) == (long)B * OH * OW, "dy2 rows");
  TORCH_CHECK(wt.size(0) >= (long)KH * KW * C && wt.size(1) == outC, "wt shape");
  return lo::launch_conv_dx(dy2.data_ptr(), dy2.stride(0), wt.data_ptr(),
                            wt.stride(0), dx.data_ptr(), B, H, W, C, (int)KH,
                            (int)KW, (int)SH, (int)SW, (int)PH, (int)PW, OH,
                            OW, outC, stream());
}

std::vector<at::Tensor> maxpool_fwd(at::Tensor in, int64_t KH, int64_t KW,
                                    int64_t SH, int64_t SW, int64_t PH, int64_t PW,
                                    bool relu_sentinel = false) {
  check_bf16(in, "in");
  const int B = (int)in.size(0), H = (int)in.size(1), W = (int)in.size(2),
            C = (int)in.size(3);
  const int OH = (H + 2 * (int)PH - (int)KH) / (int)SH + 1;
  const int OW = (W + 2 * (int)PW - (int)KW) / (int)SW + 1;
  auto out = at::empty({B, OH, OW, C}, in.options());
  auto idx = at::empty({B, OH, OW, C}, in.options().dtype(at::kByte));
  lo::launch_maxpool_fwd(in.data_ptr(), out.data_ptr(), idx.data_ptr(), B, H, W, C,
                         (int)KH, (int)KW, (int)SH, (int)SW, (int)PH, (int)PW,
                         OH, OW, relu_sentinel ? 1 : 0, stream());
  return {out, idx};
}

at::Tensor maxpool_bwd(at::Tensor dy, at::Tensor idx, int64_t H, int64_t W,
                       int64_t KH, int64_t KW, int64_t SH, int64_t SW,
                       int64_t PH, int64_t PW, at::Tensor dx,
                       c10::optional<at::Tensor> relu_y) {
  check_bf16(dy, "dy");
  check_bf16(dx, "dx");
  const int B = (int)dy.size(0), OH = (int)dy.size(1), OW = (int)dy.size(2),
            C = (int)dy.size(3);
  const void* yp = nullptr;
  if (relu_y) {
    check_bf16(*relu_y, "relu_y");
    TORCH_CHECK(relu_y->numel() == dx.numel(), "relu_y shape");
    yp = relu_y->data_ptr();
  }
  lo::launch_maxpool_bwd(dy.data_ptr(), idx.data_ptr(), dx.data_ptr(), yp, B, (int)H,
                         (int)W, C, (int)KH, (int)KW, (int)SH, (int)SW,
                         (int)PH, (int)PW, OH, OW, stream());
  return dx;
}

// ----------------------------------------------------------- batchnorm ----
void bn_stats(at::Tensor x, at::Tensor sum, at::Tensor sumsq) {
  check_bf16(x, "x");
  check_f32(sum, "sum");
  check_f32(sumsq, "sumsq");
  const int C = (int)x.size(-1);
  TORCH_CHECK(C % 8 == 0 && sum.numel() == C && sumsq.numel() == C);
  lo::launch_bn_stats(x.data_ptr(), sum.data_ptr(), sumsq.data_ptr(),
                      x.numel() / C, C, stream());
}

void bn_fwd(at::Tensor x, at::Tensor y, at::Tensor mean, at::Tensor invstd,
            at::Tensor gamma, at::Tensor beta, bool relu,
            c10::optional<at::Tensor> residual = c10::nullopt) {
  check_bf16(x, "x");
  check_bf16(y, "y");
  const int C = (int)x.size(-1);
  const void* res = nullptr;
  if (residual.has_value()) {
    check_bf16(*residual, "residual");
    TORCH_CHECK(residual->numel() == x.numel(), "residual shape");
    res = residual->data_ptr();
  }
  lo::launch_bn_fwd(x.data_ptr(), y.data_ptr(), mean.data_ptr(),
                    invstd.data_ptr(), gamma.data_ptr(), beta.data_ptr(),
                    res, x.numel() / C, C, relu ? 1 : 0, stream());
}

void bn_bwd_reduce(at::Tensor dy, at::Tensor y, at::Tensor x, at::Tensor mean,
                   at::Tensor invstd, at::Tensor dbeta, at::Tensor dgamma,
                   bool relu) {
  check_bf16(dy, "dy");
  const int C = (int)x.size(-1);
  lo::launch_bn_bwd_reduce(dy.data_ptr(), y.data_ptr(), x.data_ptr(),
                           mean.data_ptr(), invstd.data_ptr(), dbeta.data_ptr(),
                           dgamma.data_ptr(), x.numel() / C, C, relu ? 1 : 0,
                           stream());
}

void bn_bwd_dx(at::Tensor dy, at::Tensor y, at::Tensor x, at::Tensor dx,
               at::Tensor mean, at::Tensor invstd, at::Tensor gamma,
               at::Tensor dbeta, at::Tensor dgamma, bool relu) {
  check_bf16(dx, "dx");
  const int C = (int)x.size(-1);
  lo::launch_bn_bwd_dx(dy.data_ptr(), y.data_ptr(), x.data_ptr(), dx.data_ptr(),
                       mean.data_ptr(), invstd.data_ptr(), gamma.data_ptr(),
                       dbeta.data_ptr(), dgamma.data_ptr(), x.numel() / C, C,
                       relu ? 1 : 0, stream());
}

void bn_finalize_stats(at::Tensor scratch, at::Tensor mean, at::Tensor invstd,
                       int64_t M,
                       c10::optional<at::Tensor> rmean,
                       c10::optional<at::Tensor> rvar,
                       double momentum, double eps) {
  check_f32(scratch, "scratch");
  check_f32(mean, "mean");
  lo::launch_bn_finalize_stats(
      scratch.data_ptr(), mean.data_ptr(), invstd.data_ptr(),
      rmean ? rmean->data_ptr() : nullptr,
      rvar ? rvar->data_ptr() : nullptr,
      1.f / (float)M, (float)momentum, (float)eps,
      (int)mean.numel(), stream());
}

void add_relu(at::Tensor a, at::Tensor b, at::Tensor z, bool relu) {
  check_bf16(a, "a");
  check_bf16(b, "b");
  check_bf16(z, "z");
  TORCH_CHECK(a.numel() == b.numel() && a.numel() == z.numel() &&
              a.numel() % 8 == 0);
  lo::launch_add_relu(a.data_ptr(), b.data_ptr(), z.data_ptr(), a.numel(),
                      relu ? 1 : 0, stream());
}

at::Tensor avgpool_global(at::Tensor x, at::Tensor out) {
  check_bf16(x, "x");
  check_bf16(out, "out");
  const int B = (int)x.size(0);
  const int C = (int)x.size(-1);
  const int HW = (int)(x.numel() / ((long)B * C));
  lo::launch_avgpool_global(x.data_ptr(), out.data_ptr(), B, HW, C, stream());
  return out;
}

at::Tensor avgpool_global_bwd(at::Tensor dy, at::Tensor dx) {
  check_bf16(dy, "dy");
  check_bf16(dx, "dx");
  const int B = (int)dy.size(0);
  const int C = (int)dy.size(-1);
  const int HW = (int)(dx.numel() / ((long)B * C));
  lo::launch_avgpool_global_bwd(dy.data_ptr(), dx.data_ptr(), B, HW, C, stream());
  return dx;
}

at::Tensor relu_bwd(at::Tensor dy, at::Tensor y, at::Tensor dx) {
  check_bf16(dy, "dy");
  check_bf16(y, "y");
  check_bf16(dx, "dx");
  TORCH_CHECK(dy.numel() == y.numel() && dy.numel() == dx.numel());
  TORCH_CHECK(dy.numel() % 8 == 0, "relu_bwd needs 8-aligned numel");
  lo::launch_relu_bwd(dy.data_ptr(), y.data_ptr(), dx.data_ptr(), dy.numel(),
                      stream());
  return dx;
}

void sgd_step(at::Tensor master, at::Tensor grad, at::Tensor mom, at::Tensor mirror,
              double lr, double mu, double wd, double gscale) {
  check_f32(master, "master");
  check_f32(grad, "grad");
  check_f32(mom, "mom");
  check_bf16(mirror, "mirror");
  TORCH_CHECK(master.numel() == grad.numel() && master.numel() == mom.numel() &&
              master.numel() == mirror.numel());
  TORCH_CHECK(master.numel() % 4 == 0, "arena must be 4-aligned");
  lo::launch_sgd(master.data_ptr(), grad.data_ptr(), mom.data_ptr(),
                 mirror.data_ptr(), master.numel(), (float)lr, (float)mu,
                 (float)wd, (float)gscale, stream());
}

void adam_step(at::Tensor master, at::Tensor grad, at::Tensor m1, at::Tensor m2,
               at::Tensor mirror, at::Tensor step_dev, double lr, double b1,
               double b2, double eps, double wd, double gscale) {
  check_f32(master, "master");
  check_f32(grad, "grad");
  check_f32(m1, "m1");
  check_f32(m2, "m2");
  check_bf16(mirror, "mirror");
  TORCH_CHECK(step_dev.scalar_type() == at::kInt && step_dev.is_cuda(),
              "step_dev must be int32 on GPU");
  TORCH_CHECK(master.numel() % 4 == 0, "arena must be 4-aligned");
  lo::launch_adam(master.data_ptr(), grad.data_ptr(), m1.data_ptr(), m2.data_ptr(),
                  mirror.data_ptr(), step_dev.data_ptr(), master.numel(),
                  (float)lr, (float)b1, (float)b2, (float)eps, (float)wd,
                  (float)gscale, stream());
}

at::Tensor colsum(at::Tensor dy, at::Tensor out,
                  c10::optional<at::Tensor> mask = c10::nullopt) {
  check_bf16(dy, "dy");
  check_f32(out, "out");
  TORCH_CHECK(dy.dim() == 2 && out.numel() == dy.size(1));
  if (mask) {
    TORCH_CHECK(mask->scalar_type() == at::kByte && mask->is_contiguous() &&
                mask->numel() == dy.numel(), "mask: u8, same shape");
    TORCH_CHECK(dy.size(1) <= 256, "masked colsum: N <= 256");
    lo::launch_colsum_masked(dy.data_ptr(), mask->data_ptr(), out.data_ptr(),
                             dy.size(0), (int)dy.size(1), dy.size(1),
                             stream());
    return out;
  }
  lo::launch_colsum(dy.data_ptr(), out.data_ptr(), dy.size(0), (int)dy.size(1),
                    dy.size(1), stream());
  return out;
}

at::Tensor argmax_rows(at::Tensor x, int64_t cvalid) {
  check_bf16(x, "x");
  TORCH_CHECK(x.dim() == 2);
  auto out = at::empty({x.size(0)}, x.options().dtype(at::kInt));
  lo::launch_argmax_rows(x.data_ptr(), out.data_ptr(), x.size(0), (int)x.size(1),
                         (int)cvalid, x.size(1), stream());
  return out;
}

at::Tensor accuracy_count(at::Tensor pred, at::Tensor label) {
  TORCH_CHECK(pred.scalar_type() == at::kInt && label.scalar_type() == at::kLong);
  auto out = at::zeros({1}, pred.options());
  lo::launch_accuracy_count(pred.data_ptr(), label.data_ptr(), out.data_ptr(),
                            pred.numel(), stream());
  return out;
}

// fused softmax-CE: writes dlogits; returns nothing (loss_sum/correct are
// caller-provided accumulators so one graph-captured step reuses them)
void softmax_ce(at::Tensor logits, at::Tensor labels, at::Tensor dlogits,
                c10::optional<at::Tensor> loss_sum, c10::optional<at::Tensor> correct,
                int64_t cvalid, double gscale) {
  check_bf16(logits, "logits");
  check_bf16(dlogits, "dlogits");
  TORCH_CHECK(labels.scalar_type() == at::kLong && labels.is_cuda());
  TORCH_CHECK(logits.dim() == 2 && logits.sizes() == dlogits.sizes());
  TORCH_CHECK(cvalid <= logits.size(1));
  TORCH_CHECK(cvalid <= 32 || logits.size(1) % 8 == 0,
              "wave path needs 8-aligned C");
  void* lp = nullptr;
  void* cp = nullptr;
  if (loss_sum.has_value()) { check_f32(*loss_sum, "loss_sum"); lp = loss_sum->data_ptr(); }
  if (correct.has_value()) { cp = correct->data_ptr(); }
  lo::launch_softmax_ce(logits.data_ptr(), labels.data_ptr(), dlogits.data_ptr(),
                        lp, cp, logits.size(0), (int)logits.size(1), (int)cvalid,
                        (float)gscale, stream());
}

void tree_hist(at::Tensor binned, at::Tensor node_of, at::Tensor grad,
               at::Tensor hess, at::Tensor hist, int64_t n_nodes, int64_t n_bins,
               double gbound = 0.0, double hbound = 0.0) {
  TORCH_CHECK(binned.is_cuda() && binned.scalar_type() == at::kByte &&
              binned.is_contiguous(), "binned must be contiguous u8 GPU");
  TORCH_CHECK(node_of.scalar_type() == at::kInt && node_of.is_contiguous());
  check_f32(grad, "grad");
  check_f32(hess, "hess");
  check_f32(hist, "hist");
  const long N = binned.size(0);
  const int F = (int)binned.size(1);
  TORCH_CHECK(node_of.numel() == N && grad.numel() == N && hess.numel() == N);
  TORCH_CHECK(hist.numel() == n_nodes * F * n_bins * 2, "hist size");
  lo::launch_tree_hist(binned.data_ptr(), node_of.data_ptr(), grad.data_ptr(),
                       hess.data_ptr(), hist.data_ptr(), N, F, (int)n_nodes,
                       (int)n_bins, gbound, hbound, stream());
}

at::Tensor embedding_fwd(at::Tensor ids, at::Tensor table, at::Tensor out) {
  TORCH_CHECK(ids.scalar_type() == at::kLong && ids.is_cuda() && ids.is_contiguous());
  check_bf16(table, "table");
  check_bf16(out, "out");
  const int dim = (int)table.size(1);
  TORCH_CHECK(dim % 8 == 0, "embedding dim must be 8-aligned");
  TORCH_CHECK(out.numel() == ids.numel() * dim, "out shape");
  lo::launch_embedding_fwd(ids.data_ptr(), table.data_ptr(), out.data_ptr(),
                           ids.numel(), dim, stream());
  return out;
}

void embedding_bwd(at::Tensor ids, at::Tensor dy, at::Tensor gtable) {
  TORCH_CHECK(ids.scalar_type() == at::kLong && ids.is_cuda() && ids.is_contiguous());
  check_bf16(dy, "dy");
  check_f32(gtable, "gtable");
  const int dim = (int)gtable.size(1);
  TORCH_CHECK(dy.numel() == ids.numel() * dim, "dy shape");
  lo::launch_embedding_bwd(ids.data_ptr(), dy.data_ptr(), gtable.data_ptr(),
                           ids.numel(), dim, stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("gemm", &gemm, "bf16 MFMA GEMM (gfx950)",
        py::arg("A"), py::arg("B"), py::arg("C"), py::arg("bias") = py::none(),
        py::arg("ta") = false, py::arg("tb") = false, py::arg("epi") = 0,
        py::arg("splits") = 1, py::arg("stats") = py::none(),
        py::arg("addend") = py::none());
  m.def("gemm_conv_fwd", &gemm_conv_fwd);
  m.def("gemm_conv_dw", &gemm_conv_dw);
  m.def("mfma_probe", &mfma_probe);
  m.def("im2col", &im2col);
  m.def("conv_fwd_small", &conv_fwd_small, "small-image fused conv fwd");
  m.def("conv1d_fwd", &conv1d_fwd, "1-D conv fwd (h-tiled LDS window)");
  m.def("conv1d_dx", &conv1d_dx, "1-D conv dX (h-tiled LDS RMW)");
  m.def("conv_dw_c1", &conv_dw_c1, "C=1 conv dW (x LDS-resident)");
  m.def("conv_dw_small", &conv_dw_small,
        "small-image C%8==0 conv dW (x+dyT LDS-resident per image)");
  m.def("conv_dx", &conv_dx, "fused conv dX (LDS-accumulated scatter)");
  m.def("col2im", &col2im);
  m.def("maxpool_fwd", &maxpool_fwd, py::arg("in"), py::arg("KH"),
        py::arg("KW"), py::arg("SH"), py::arg("SW"), py::arg("PH"),
        py::arg("PW"), py::arg("relu_sentinel") = false);
  m.def("maxpool_bwd", &maxpool_bwd);
  m.def("relu_bwd", &relu_bwd);
  m.def("sgd_step", &sgd_step);
  m.def("adam_step", &adam_step);
  m.def("colsum", &colsum, py::arg("dy"), py::arg("out"),
        py::arg("mask") = py::none());
  m.def("argmax_rows", &argmax_rows);
  m.def("accuracy_count", &accuracy_count);
  m.def("tree_hist", &tree_hist, py::arg("binned"), py::arg("node_of"),
        py::arg("grad"), py::arg("hess"), py::arg("hist"),
        py::arg("n_nodes"), py::arg("n_bins"),
        py::arg("gbound") = 0.0, py::arg("hbound") = 0.0);
  m.def("embedding_fwd", &embedding_fwd);
  m.def("bn_stats", &bn_stats);
  m.def("bn_fwd", &bn_fwd, py::arg("x"), py::arg("y"), py::arg("mean"),
        py::arg("invstd"), py::arg("gamma"), py::arg("beta"),
        py::arg("relu"), py::arg("residual") = py::none());
  m.def("bn_bwd_reduce", &bn_bwd_reduce);
  m.def("bn_bwd_dx", &bn_bwd_dx);
  m.def("bn_finalize_stats", &bn_finalize_stats, py::arg("scratch"),
        py::arg("mean"), py::arg("invstd"), py::arg("M"),
        py::arg("rmean") = py::none(), py::arg("rvar") = py::none(),
        py::arg("momentum") = 0.0, py::arg("eps") = 1e-5);
  m.def("add_relu", &add_relu);
  m.def("avgpool_global", &avgpool_global);
  m.def("avgpool_global_bwd", &avgpool_global_bwd);
  m.def("embedding_bwd", &embedding_bwd);
  m.def("softmax_ce", &softmax_ce,
        py::arg("logits"), py::arg("labels"), py::arg("dlogits"),
        py::arg("loss_sum") = py::none(), py::arg("correct") = py::none(),
        py::arg("cvalid") = 0, py::arg("gscale") = 1.0);
}
