// Python bindings for the gfx950 gossip kernels (HIP-native; uses the
// c10::hip stream API directly — no CUDA-compat layer).

#include <hip/hip_runtime.h>

#include <algorithm>
#include <torch/extension.h>

#include <c10/hip/HIPStream.h>

typedef unsigned short ushort_t;

extern "C" {
int bn_reduce_nblocks(int64_t M, int C);
void bn_fwd_reduce(const ushort_t* x, float* partials, int64_t M, int C,
                   hipStream_t s);
void bn_fwd_reduce_finalize(const ushort_t* x, float* partials,
                            unsigned int* counter, const float* gamma,
                            const float* beta, float* rmean, float* rvar,
                            float* smean, float* sinvstd,
                            float* scale_shift, double momentum, double eps,
                            int64_t M, int C, bool update_running,
                            hipStream_t s);
void bn_bwd_reduce_finalize(const ushort_t* x, const ushort_t* dy,
                            const ushort_t* y, const float* smean,
                            const float* sinvstd, float* partials,
                            unsigned int* counter, const float* gamma,
                            float* dgamma, float* dbeta, float* coef,
                            int64_t M, int C, bool relu, bool training,
                            hipStream_t s);
void bn_fwd_finalize(const float* shadows, const float* gamma,
                     const float* beta, float* rmean, float* rvar,
                     float* smean, float* sinvstd, float* scale_shift,
                     double momentum, double eps, int64_t M, int C,
                     bool update_running, hipStream_t s);
void bn_eval_prep(const float* rmean, const float* rvar, const float* gamma,
                  const float* beta, float* scale_shift, double eps, int C,
                  hipStream_t s);
void bn_fwd_apply(const ushort_t* x, const ushort_t* res, ushort_t* y,
                  const float* scale_shift, int64_t M, int C, bool relu,
                  hipStream_t s);
void bn_bwd_reduce(const ushort_t* x, const ushort_t* dy, const ushort_t* y,
                   const float* smean, const float* sinvstd, float* shadows,
                   int64_t M, int C, bool relu, hipStream_t s);
void bn_bwd_finalize(const float* shadows, const float* gamma,
                     const float* smean, const float* sinvstd, float* dgamma,
                     float* dbeta, float* coef, int64_t M, int C,
                     bool training, hipStream_t s);
void bn_bwd_apply(const ushort_t* x, const ushort_t* dy, const ushort_t* y,
                  ushort_t* dx, ushort_t* dres, const float* coef, int64_t M,
                  int C, bool relu, hipStream_t s);
void sgp_scale(float* x, const float* a, int64_t n, hipStream_t stream);
void sgp_add_scale(float* x, const float* r, const float* a, int64_t n,
                   hipStream_t stream);
void sgp_pack_mix(float* x, float* out, const float* a, int64_t n,
                  hipStream_t stream);
void sgp_average(float* x, const float* y, int64_t n, hipStream_t stream);
void sgp_pack_mix_bf16(float* x, unsigned short* out, const float* a,
                       int64_t n, hipStream_t stream);
void sgp_add_scale_bf16(float* x, const unsigned short* r, const float* a,
                        int64_t n, hipStream_t stream);
void sgp_sgd_step_bf16gs(float* p, const unsigned short* g, float* buf,
                         unsigned short* shadow, const float* lr_ptr,
                         double mu, double wd, double damp, bool nesterov,
                         bool first, int64_t n, hipStream_t stream);
void sgp_cast_shadow(const float* p, unsigned short* shadow, int64_t n,
                     hipStream_t stream);
void sgp_gather_multi_f32(const float* const* srcs, const int64_t* offsets,
                          float* out, int nparams, int64_t total,
                          hipStream_t stream);
void sgp_gather_multi_bf16(const unsigned short* const* srcs,
                           const int64_t* offsets, unsigned short* out,
                           int nparams, int64_t total, hipStream_t stream);
void sgp_sgd_step(float* p, const float* g, float* buf, const float* lr_ptr,
                  double mu, double wd, double damp, bool nesterov,
                  bool first, int64_t n, hipStream_t stream);
void sgp_mfma_probe(const ushort_t* A, const ushort_t* B, float* C,
                    hipStream_t s);
void sgp_gemm_nt_bf16(const ushort_t* A, const ushort_t* B, ushort_t* C,
                      int64_t M, int N, int K, hipStream_t s);
void sgp_gemm_nt_bf16_v2(const ushort_t* A, const ushort_t* B, ushort_t* C,
                         int64_t M, int N, int K, hipStream_t s);
void sgp_gemm_nt_bf16_v3(const ushort_t* A, const ushort_t* B, ushort_t* C,
                         int64_t M, int N, int K, hipStream_t s);
void sgp_gemm_nt_bf16_v4(const ushort_t* A, const ushort_t* B, ushort_t* C,
                         int64_t M, int N, int K, hipStream_t s);
void sgp_gemm_nt_bf16_v5(const ushort_t* A, const ushort_t* B, ushort_t* C,
                         int64_t M, int N, int K, hipStream_t s);
void sgp_gemm_nt_bf16_v6(const ushort_t* A, const ushort_t* B, ushort_t* C,
                         int64_t M, int N, int K, int span, hipStream_t s);
void sgp_gemm_nt_bf16_v7(const ushort_t* A, const ushort_t* B, ushort_t* C,
                         int64_t M, int N, int K, int span, hipStream_t s);
void sgp_gemm_nt_splitk_bf16(const ushort_t* A, const ushort_t* B,
                             float* P, ushort_t* C, int64_t M, int N,
                             int K, int split, hipStream_t s);
void sgp_gemm_tn_wgrad_bf16(const ushort_t* dy, const ushort_t* x,
                            float* partials, float* dw, int64_t M, int Co,
                            int Ci, int split, hipStream_t s);
void sgp_conv3x3_nhwc_bf16(const ushort_t* X, const ushort_t* Wt,
                           ushort_t* Y, float* P, int Nb, int H, int W,
                           int Ci, int Co, int Ho, int Wo, int stride,
                           int split, hipStream_t s);
void sgp_conv3x3_wgrad_bf16(const ushort_t* dy, const ushort_t* X,
                            float* partials, float* dw, int Nb, int H,
                            int W, int Ci, int Co, int Ho, int Wo,
                            int stride, int split, hipStream_t s);
}

namespace {

hipStream_t current_stream(const torch::Tensor& t) {
  return c10::hip::getCurrentHIPStream(t.device().index()).stream();
}

void check_flat(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be a device tensor");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be fp32");
}

void check_scalar(const torch::Tensor& a, const torch::Tensor& like) {
  TORCH_CHECK(a.is_cuda(), "scalar must be a device tensor");
  TORCH_CHECK(a.numel() == 1, "scalar must have exactly one element");
  TORCH_CHECK(a.scalar_type() == torch::kFloat32, "scalar must be fp32");
  TORCH_CHECK(a.device() == like.device(), "scalar on wrong device");
}

void scale_(torch::Tensor x, torch::Tensor a) {
  check_flat(x, "x");
  check_scalar(a, x);
  sgp_scale(x.data_ptr<float>(), a.data_ptr<float>(), x.numel(),
            current_stream(x));
}

void add_scale_(torch::Tensor x, torch::Tensor r, torch::Tensor a) {
  check_flat(x, "x");
  check_flat(r, "r");
  check_scalar(a, x);
  TORCH_CHECK(x.numel() == r.numel(), "size mismatch");
  sgp_add_scale(x.data_ptr<float>(), r.data_ptr<float>(), a.data_ptr<float>(),
                x.numel(), current_stream(x));
}

void pack_mix_(torch::Tensor x, torch::Tensor out, torch::Tensor a) {
  check_flat(x, "x");
  check_flat(out, "out");
  check_scalar(a, x);
  TORCH_CHECK(x.numel() == out.numel(), "size mismatch");
  sgp_pack_mix(x.data_ptr<float>(), out.data_ptr<float>(), a.data_ptr<float>(),
               x.numel(), current_stream(x));
}

void average_(torch::Tensor x, torch::Tensor y) {
  check_flat(x, "x");
  check_flat(y, "y");
  TORCH_CHECK(x.numel() == y.numel(), "size mismatch");
  sgp_average(x.data_ptr<float>(), y.data_ptr<float>(), x.numel(),
              current_stream(x));
}

void sgd_step_(torch::Tensor p, torch::Tensor g, torch::Tensor buf,
               torch::Tensor lr, double momentum, double weight_decay,
               double dampening, bool nesterov, bool first_step) {
  check_flat(p, "p");
  check_flat(g, "g");
  check_flat(buf, "buf");
  check_scalar(lr, p);
  TORCH_CHECK(p.numel() == g.numel() && p.numel() == buf.numel(),
              "size mismatch");
  sgp_sgd_step(p.data_ptr<float>(), g.data_ptr<float>(), buf.data_ptr<float>(),
               lr.data_ptr<float>(), momentum, weight_decay, dampening,
               nesterov, first_step, p.numel(), current_stream(p));
}

void sgd_step_bf16gs_(torch::Tensor p, torch::Tensor g, torch::Tensor buf,
                      torch::Tensor shadow, torch::Tensor lr,
                      double momentum, double weight_decay,
                      double dampening, bool nesterov, bool first_step) {
  check_flat(p, "p");
  check_flat(buf, "buf");
  check_scalar(lr, p);
  TORCH_CHECK(g.is_cuda() && g.is_contiguous()
              && g.scalar_type() == torch::kBFloat16,
              "g must be contiguous bf16");
  TORCH_CHECK(shadow.is_cuda() && shadow.is_contiguous()
              && shadow.scalar_type() == torch::kBFloat16,
              "shadow must be contiguous bf16");
  TORCH_CHECK(p.numel() == g.numel() && p.numel() == buf.numel()
              && p.numel() == shadow.numel(), "size mismatch");
  sgp_sgd_step_bf16gs(
      p.data_ptr<float>(),
      reinterpret_cast<const unsigned short*>(g.data_ptr()),
      buf.data_ptr<float>(),
      reinterpret_cast<unsigned short*>(shadow.data_ptr()),
      lr.data_ptr<float>(), momentum, weight_decay, dampening, nesterov,
      first_step, p.numel(), current_stream(p));
}

// ptr_table: int64 DEVICE tensor of source addresses (0 = missing ->
// zero-fill); offsets: int64 DEVICE tensor [nparams+1] of flat starts.
void gather_multi_(torch::Tensor ptr_table, torch::Tensor offsets,
                   torch::Tensor out) {
  TORCH_CHECK(ptr_table.is_cuda() && ptr_table.is_contiguous()
              && ptr_table.scalar_type() == torch::kInt64);
  TORCH_CHECK(offsets.is_cuda() && offsets.is_contiguous()
              && offsets.scalar_type() == torch::kInt64);
  TORCH_CHECK(out.is_cuda() && out.is_contiguous());
  const int nparams = (int)ptr_table.numel();
  TORCH_CHECK(offsets.numel() == nparams + 1);
  if (out.scalar_type() == torch::kFloat32) {
    sgp_gather_multi_f32(
        reinterpret_cast<const float* const*>(ptr_table.data_ptr()),
        offsets.data_ptr<int64_t>(), out.data_ptr<float>(), nparams,
        out.numel(), current_stream(out));
  } else if (out.scalar_type() == torch::kBFloat16) {
    sgp_gather_multi_bf16(
        reinterpret_cast<const unsigned short* const*>(
            ptr_table.data_ptr()),
        offsets.data_ptr<int64_t>(),
        reinterpret_cast<unsigned short*>(out.data_ptr()), nparams,
        out.numel(), current_stream(out));
  } else {
    TORCH_CHECK(false, "gather_multi_: fp32 or bf16 only");
  }
}

void cast_shadow_(torch::Tensor p, torch::Tensor shadow) {
  check_flat(p, "p");
  TORCH_CHECK(shadow.is_cuda() && shadow.is_contiguous()
              && shadow.scalar_type() == torch::kBFloat16,
              "shadow must be contiguous bf16");
  TORCH_CHECK(p.numel() == shadow.numel(), "size mismatch");
  sgp_cast_shadow(p.data_ptr<float>(),
                  reinterpret_cast<unsigned short*>(shadow.data_ptr()),
                  p.numel(), current_stream(p));
}

void pack_mix_bf16_(torch::Tensor x, torch::Tensor out, torch::Tensor a) {
  check_flat(x, "x");
  TORCH_CHECK(out.is_cuda() && out.is_contiguous()
              && out.scalar_type() == torch::kBFloat16,
              "out must be contiguous bf16 device tensor");
  check_scalar(a, x);
  TORCH_CHECK(x.numel() == out.numel(), "size mismatch");
  sgp_pack_mix_bf16(x.data_ptr<float>(),
                    reinterpret_cast<unsigned short*>(out.data_ptr()),
                    a.data_ptr<float>(), x.numel(), current_stream(x));
}

void add_scale_bf16_(torch::Tensor x, torch::Tensor r, torch::Tensor a) {
  check_flat(x, "x");
  TORCH_CHECK(r.is_cuda() && r.is_contiguous()
              && r.scalar_type() == torch::kBFloat16,
              "r must be contiguous bf16 device tensor");
  check_scalar(a, x);
  TORCH_CHECK(x.numel() == r.numel(), "size mismatch");
  sgp_add_scale_bf16(x.data_ptr<float>(),
                     reinterpret_cast<const unsigned short*>(r.data_ptr()),
                     a.data_ptr<float>(), x.numel(), current_stream(x));
}

void mfma_probe(torch::Tensor A, torch::Tensor B, torch::Tensor C) {
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == torch::kBFloat16
              && A.numel() == 16 * 32);
  TORCH_CHECK(B.is_cuda() && B.scalar_type() == torch::kBFloat16
              && B.numel() == 16 * 32);
  TORCH_CHECK(C.is_cuda() && C.scalar_type() == torch::kFloat32
              && C.numel() == 16 * 16);
  sgp_mfma_probe(reinterpret_cast<const ushort_t*>(A.data_ptr()),
                 reinterpret_cast<const ushort_t*>(B.data_ptr()),
                 C.data_ptr<float>(), current_stream(A));
}

void gemm_nt_check(const torch::Tensor& A, const torch::Tensor& B,
                   const torch::Tensor& C) {
  TORCH_CHECK(A.is_cuda() && A.is_contiguous()
              && A.scalar_type() == torch::kBFloat16, "A must be bf16");
  TORCH_CHECK(B.is_cuda() && B.is_contiguous()
              && B.scalar_type() == torch::kBFloat16, "B must be bf16");
  TORCH_CHECK(C.is_cuda() && C.is_contiguous()
              && C.scalar_type() == torch::kBFloat16, "C must be bf16");
  TORCH_CHECK(B.size(1) == A.size(1) && C.size(0) == A.size(0)
              && C.size(1) == B.size(0), "shape mismatch");
  TORCH_CHECK(A.size(1) % 32 == 0, "K must be a multiple of 32");
}

void gemm_nt_bf16_v2(torch::Tensor A, torch::Tensor B, torch::Tensor C) {
  gemm_nt_check(A, B, C);
  sgp_gemm_nt_bf16_v2(reinterpret_cast<const ushort_t*>(A.data_ptr()),
                      reinterpret_cast<const ushort_t*>(B.data_ptr()),
                      reinterpret_cast<ushort_t*>(C.data_ptr()), A.size(0),
                      (int)B.size(0), (int)A.size(1), current_stream(A));
}

void gemm_nt_bf16_v3(torch::Tensor A, torch::Tensor B, torch::Tensor C) {
  gemm_nt_check(A, B, C);
  TORCH_CHECK(A.size(1) % 64 == 0, "v3 requires K %% 64 == 0");
  sgp_gemm_nt_bf16_v3(reinterpret_cast<const ushort_t*>(A.data_ptr()),
                      reinterpret_cast<const ushort_t*>(B.data_ptr()),
                      reinterpret_cast<ushort_t*>(C.data_ptr()), A.size(0),
                      (int)B.size(0), (int)A.size(1), current_stream(A));
}

void gemm_nt_bf16_v5(torch::Tensor A, torch::Tensor B, torch::Tensor C) {
  gemm_nt_check(A, B, C);
  TORCH_CHECK(A.size(1) % 64 == 0, "v5 requires K %% 64 == 0");
  sgp_gemm_nt_bf16_v5(reinterpret_cast<const ushort_t*>(A.data_ptr()),
                      reinterpret_cast<const ushort_t*>(B.data_ptr()),
                      reinterpret_cast<ushort_t*>(C.data_ptr()), A.size(0),
                      (int)B.size(0), (int)A.size(1), current_stream(A));
}

void gemm_nt_bf16_v6(torch::Tensor A, torch::Tensor B, torch::Tensor C,
                     bool span) {
  gemm_nt_check(A, B, C);
  TORCH_CHECK(A.size(0) % 256 == 0 && B.size(0) % 128 == 0
              && A.size(1) % 64 == 0,
              "v6 requires M %% 256 == 0, N %% 128 == 0, K %% 64 == 0");
  sgp_gemm_nt_bf16_v6(reinterpret_cast<const ushort_t*>(A.data_ptr()),
                      reinterpret_cast<const ushort_t*>(B.data_ptr()),
                      reinterpret_cast<ushort_t*>(C.data_ptr()), A.size(0),
                      (int)B.size(0), (int)A.size(1), span ? 1 : 0,
                      current_stream(A));
}

void gemm_nt_bf16_v7(torch::Tensor A, torch::Tensor B, torch::Tensor C,
                     bool span) {
  gemm_nt_check(A, B, C);
  TORCH_CHECK(A.size(0) % 256 == 0 && B.size(0) % 128 == 0
              && A.size(1) % 64 == 0,
              "v7 requires M %% 256 == 0, N %% 128 == 0, K %% 64 == 0");
  sgp_gemm_nt_bf16_v7(reinterpret_cast<const ushort_t*>(A.data_ptr()),
                      reinterpret_cast<const ushort_t*>(B.data_ptr()),
                      reinterpret_cast<ushort_t*>(C.data_ptr()), A.size(0),
                      (int)B.size(0), (int)A.size(1), span ? 1 : 0,
                      current_stream(A));
}

void gemm_nt_splitk_bf16(torch::Tensor A, torch::Tensor B,
                         torch::Tensor C, int64_t split) {
  gemm_nt_check(A, B, C);
  TORCH_CHECK(A.size(1) % 64 == 0, "split-K requires K %% 64 == 0");
  TORCH_CHECK(split >= 1 && split <= 64);
  const int64_t M = A.size(0), N = B.size(0);
  torch::Tensor partials = torch::empty(
      {split * M * N}, A.options().dtype(torch::kFloat32));
  sgp_gemm_nt_splitk_bf16(
      reinterpret_cast<const ushort_t*>(A.data_ptr()),
      reinterpret_cast<const ushort_t*>(B.data_ptr()),
      partials.data_ptr<float>(),
      reinterpret_cast<ushort_t*>(C.data_ptr()), M, (int)N,
      (int)A.size(1), (int)split, current_stream(A));
}

void gemm_nt_bf16_v4(torch::Tensor A, torch::Tensor B, torch::Tensor C) {
  gemm_nt_check(A, B, C);
  TORCH_CHECK(A.size(0) % 128 == 0 && B.size(0) % 128 == 0
              && A.size(1) % 64 == 0,
              "v4 requires M,N %% 128 == 0 and K %% 64 == 0");
  sgp_gemm_nt_bf16_v4(reinterpret_cast<const ushort_t*>(A.data_ptr()),
                      reinterpret_cast<const ushort_t*>(B.data_ptr()),
                      reinterpret_cast<ushort_t*>(C.data_ptr()), A.size(0),
                      (int)B.size(0), (int)A.size(1), current_stream(A));
}

void gemm_tn_wgrad_bf16(torch::Tensor dy, torch::Tensor x,
                        torch::Tensor partials, torch::Tensor dw,
                        int64_t split) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous()
              && dy.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(x.is_cuda() && x.is_contiguous()
              && x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(x.size(0) == dy.size(0), "M mismatch");
  const int64_t Co = dy.size(1), Ci = x.size(1);
  TORCH_CHECK(dw.is_cuda() && dw.is_contiguous()
              && dw.scalar_type() == torch::kFloat32
              && dw.numel() == Co * Ci, "dw must be fp32 [Co*Ci]");
  TORCH_CHECK(partials.is_cuda() && partials.is_contiguous()
              && partials.scalar_type() == torch::kFloat32
              && partials.numel() == split * Co * Ci,
              "partials must be fp32 [split*Co*Ci]");
  TORCH_CHECK(dy.size(0) >= 1, "M must be positive");
  sgp_gemm_tn_wgrad_bf16(
      reinterpret_cast<const ushort_t*>(dy.data_ptr()),
      reinterpret_cast<const ushort_t*>(x.data_ptr()),
      partials.data_ptr<float>(), dw.data_ptr<float>(), dy.size(0),
      (int)Co, (int)Ci, (int)split, current_stream(dy));
}

void gemm_nt_bf16(torch::Tensor A, torch::Tensor B, torch::Tensor C) {
  TORCH_CHECK(A.is_cuda() && A.is_contiguous()
              && A.scalar_type() == torch::kBFloat16, "A must be bf16");
  TORCH_CHECK(B.is_cuda() && B.is_contiguous()
              && B.scalar_type() == torch::kBFloat16, "B must be bf16");
  TORCH_CHECK(C.is_cuda() && C.is_contiguous()
              && C.scalar_type() == torch::kBFloat16, "C must be bf16");
  const int64_t M = A.size(0);
  const int64_t K = A.size(1);
  const int64_t N = B.size(0);
  TORCH_CHECK(B.size(1) == K && C.size(0) == M && C.size(1) == N,
              "shape mismatch");
  TORCH_CHECK(K % 32 == 0, "K must be a multiple of 32");
  sgp_gemm_nt_bf16(reinterpret_cast<const ushort_t*>(A.data_ptr()),
                   reinterpret_cast<const ushort_t*>(B.data_ptr()),
                   reinterpret_cast<ushort_t*>(C.data_ptr()), M, (int)N,
                   (int)K, current_stream(A));
}

// MFMA implicit-GEMM 3x3 conv (NHWC bf16, pad=1).  `x`/`y` are
// channels_last 4-D activations (their memory IS NHWC); `w` is the
// materialized [Co,3,3,Ci] weight.
void conv3x3_nhwc_bf16(torch::Tensor x, torch::Tensor w, torch::Tensor y,
                       int64_t stride) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16
              && x.is_contiguous(torch::MemoryFormat::ChannelsLast),
              "x must be bf16 channels_last");
  TORCH_CHECK(y.is_cuda() && y.scalar_type() == torch::kBFloat16
              && y.is_contiguous(torch::MemoryFormat::ChannelsLast),
              "y must be bf16 channels_last");
  TORCH_CHECK(w.is_cuda() && w.is_contiguous()
              && w.scalar_type() == torch::kBFloat16
              && w.dim() == 4 && w.size(1) == 3 && w.size(2) == 3,
              "w must be bf16 [Co,3,3,Ci] contiguous");
  const int Nb = (int)x.size(0), Ci = (int)x.size(1);
  const int H = (int)x.size(2), W = (int)x.size(3);
  const int Co = (int)y.size(1), Ho = (int)y.size(2), Wo = (int)y.size(3);
  TORCH_CHECK(w.size(0) == Co && w.size(3) == Ci, "weight shape mismatch");
  TORCH_CHECK(Ci % 64 == 0, "Ci must be a multiple of 64");
  TORCH_CHECK(stride == 1 || stride == 2, "stride must be 1 or 2");
  TORCH_CHECK(Ho == (H + 2 - 3) / stride + 1 && Wo == (W + 2 - 3) / stride + 1,
              "output spatial mismatch (pad=1, 3x3)");

  // split-K to fill the chip on small-M shapes (e.g. the 7x7 stage has
  // 52 tiles for ~512 resident-block slots).  Target ~768 blocks, keep
  // >= 2 k-steps per split, and cap the fp32 partials workspace at
  // 64 MB — the partial write+read traffic must stay small relative to
  // the MACs or split-K loses (measured trade-off, profiles/r02_*)
  const int64_t M = (int64_t)Nb * Ho * Wo;
  const int64_t tiles = ((M + 127) / 128) * ((Co + 127) / 128);
  const int KT = 9 * Ci / 64;
  // only under-filled launches with small outputs benefit: the fp32
  // partial write+read traffic scales with split*M*Co (measured: 28x28
  // shape 239 -> 176 TF with split, M*Co = 3.2M; 7x7 shape 54 -> 187
  // TF, M*Co = 0.8M)
  const bool want_split = tiles < 384 && M * Co <= (2ll << 20);
  int64_t split = want_split ? 512 / (tiles > 0 ? tiles : 1) : 1;
  if (split > KT / 2) split = KT / 2;
  const int64_t max_mem_split = (64ll << 20) / (M * Co * 4);
  if (split > max_mem_split) split = max_mem_split;
  if (split < 1) split = 1;

  torch::Tensor partials;
  float* pptr = nullptr;
  if (split > 1) {
    partials = torch::empty({split * M * Co},
                            x.options().dtype(torch::kFloat32));
    pptr = partials.data_ptr<float>();
  }
  sgp_conv3x3_nhwc_bf16(
      reinterpret_cast<const ushort_t*>(x.data_ptr()),
      reinterpret_cast<const ushort_t*>(w.data_ptr()),
      reinterpret_cast<ushort_t*>(y.data_ptr()), pptr, Nb, H, W, Ci, Co,
      Ho, Wo, (int)stride, (int)split, current_stream(x));
}

// 3x3 wgrad: dW (fp32, [Co*3*3*Ci] = channels_last memory order of the
// [Co,Ci,3,3] gradient) from channels_last x and dy.
void conv3x3_wgrad_bf16(torch::Tensor x, torch::Tensor dy,
                        torch::Tensor dw, int64_t stride) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16
              && x.is_contiguous(torch::MemoryFormat::ChannelsLast),
              "x must be bf16 channels_last");
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == torch::kBFloat16
              && dy.is_contiguous(torch::MemoryFormat::ChannelsLast),
              "dy must be bf16 channels_last");
  const int Nb = (int)x.size(0), Ci = (int)x.size(1);
  const int H = (int)x.size(2), W = (int)x.size(3);
  const int Co = (int)dy.size(1), Ho = (int)dy.size(2),
            Wo = (int)dy.size(3);
  TORCH_CHECK(dy.size(0) == Nb);
  TORCH_CHECK(Ci % 64 == 0, "Ci must be a multiple of 64");
  TORCH_CHECK(stride == 1 || stride == 2);
  TORCH_CHECK(dw.is_cuda() && dw.is_contiguous()
              && dw.scalar_type() == torch::kFloat32
              && dw.numel() == (int64_t)Co * 9 * Ci,
              "dw must be fp32 [Co*9*Ci]");
  const int64_t M = (int64_t)Nb * Ho * Wo;
  const int K9 = 9 * Ci;
  const int64_t tiles = ((Co + 127) / 128) * (int64_t)((K9 + 127) / 128);
  int64_t split = 512 / (tiles > 0 ? tiles : 1);
  split = std::min<int64_t>(split, 64);
  split = std::min<int64_t>(split, std::max<int64_t>(1, M / 128));
  const int64_t max_mem = (128ll << 20) / ((int64_t)Co * K9 * 4);
  split = std::max<int64_t>(1, std::min(split, max_mem));
  torch::Tensor partials = torch::empty(
      {split * Co * (int64_t)K9}, x.options().dtype(torch::kFloat32));
  sgp_conv3x3_wgrad_bf16(
      reinterpret_cast<const ushort_t*>(dy.data_ptr()),
      reinterpret_cast<const ushort_t*>(x.data_ptr()),
      partials.data_ptr<float>(), dw.data_ptr<float>(), Nb, H, W, Ci, Co,
      Ho, Wo, (int)stride, (int)split, current_stream(x));
}

// ---------------------------------------------------------------- BN ops

const ushort_t* bf16_ptr(const torch::Tensor& t) {
  return reinterpret_cast<const ushort_t*>(t.data_ptr());
}
ushort_t* bf16_mut(torch::Tensor& t) {
  return reinterpret_cast<ushort_t*>(t.data_ptr());
}

void check_act(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on device");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
}

void check_f32(const torch::Tensor& t, const char* name, int64_t numel) {
  TORCH_CHECK(t.is_cuda() && t.is_contiguous(), name,
              " must be contiguous device tensor");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be fp32");
  TORCH_CHECK(t.numel() == numel, name, " wrong size");
}

int64_t bn_partials_numel(int64_t M, int64_t C) {
  return (int64_t)bn_reduce_nblocks(M, (int)C) * 2 * C;
}

void bn_fwd_reduce_py(torch::Tensor x, torch::Tensor partials, int64_t M,
                      int64_t C) {
  check_act(x, "x");
  check_f32(partials, "partials", bn_partials_numel(M, C));
  bn_fwd_reduce(bf16_ptr(x), partials.data_ptr<float>(), M, (int)C,
                current_stream(x));
}

void bn_fwd_finalize_py(torch::Tensor scratch, torch::Tensor gamma,
                        torch::Tensor beta, torch::Tensor rmean,
                        torch::Tensor rvar, torch::Tensor smean,
                        torch::Tensor sinvstd, torch::Tensor scale_shift,
                        double momentum, double eps, int64_t M, int64_t C,
                        bool update_running) {
  check_f32(scratch, "partials", bn_partials_numel(M, C));
  check_f32(gamma, "gamma", C);
  check_f32(beta, "beta", C);
  check_f32(rmean, "rmean", C);
  check_f32(rvar, "rvar", C);
  check_f32(smean, "smean", C);
  check_f32(sinvstd, "sinvstd", C);
  check_f32(scale_shift, "scale_shift", 2 * C);
  bn_fwd_finalize(scratch.data_ptr<float>(), gamma.data_ptr<float>(),
                  beta.data_ptr<float>(), rmean.data_ptr<float>(),
                  rvar.data_ptr<float>(), smean.data_ptr<float>(),
                  sinvstd.data_ptr<float>(), scale_shift.data_ptr<float>(),
                  momentum, eps, M, (int)C, update_running,
                  current_stream(scratch));
}

void bn_fwd_reduce_finalize_py(
    torch::Tensor x, torch::Tensor scratch, torch::Tensor counter,
    torch::Tensor gamma, torch::Tensor beta, torch::Tensor rmean,
    torch::Tensor rvar, torch::Tensor smean, torch::Tensor sinvstd,
    torch::Tensor scale_shift, double momentum, double eps, int64_t M,
    int64_t C, bool update_running) {
  check_act(x, "x");
  check_f32(scratch, "partials", bn_partials_numel(M, C));
  TORCH_CHECK(counter.is_cuda() && counter.scalar_type() == torch::kInt32
              && counter.numel() == 1, "counter must be int32[1] device");
  check_f32(gamma, "gamma", C);
  check_f32(beta, "beta", C);
  check_f32(scale_shift, "scale_shift", 2 * C);
  bn_fwd_reduce_finalize(
      bf16_ptr(x), scratch.data_ptr<float>(),
      reinterpret_cast<unsigned int*>(counter.data_ptr()),
      gamma.data_ptr<float>(), beta.data_ptr<float>(),
      rmean.data_ptr<float>(), rvar.data_ptr<float>(),
      smean.data_ptr<float>(), sinvstd.data_ptr<float>(),
      scale_shift.data_ptr<float>(), momentum, eps, M, (int)C,
      update_running, current_stream(x));
}

void bn_bwd_reduce_finalize_py(
    torch::Tensor x, torch::Tensor dy, torch::optional<torch::Tensor> y,
    torch::Tensor smean, torch::Tensor sinvstd, torch::Tensor scratch,
    torch::Tensor counter, torch::Tensor gamma, torch::Tensor dgamma,
    torch::Tensor dbeta, torch::Tensor coef, int64_t M, int64_t C,
    bool relu, bool training) {
  check_act(x, "x");
  check_act(dy, "dy");
  check_f32(scratch, "partials", bn_partials_numel(M, C));
  TORCH_CHECK(counter.is_cuda() && counter.scalar_type() == torch::kInt32
              && counter.numel() == 1, "counter must be int32[1] device");
  check_f32(gamma, "gamma", C);
  check_f32(dgamma, "dgamma", C);
  check_f32(dbeta, "dbeta", C);
  check_f32(coef, "coef", 3 * C);
  const ushort_t* yp = nullptr;
  if (relu) {
    TORCH_CHECK(y.has_value(), "y required for relu backward");
    check_act(y.value(), "y");
    yp = bf16_ptr(y.value());
  }
  bn_bwd_reduce_finalize(
      bf16_ptr(x), bf16_ptr(dy), yp, smean.data_ptr<float>(),
      sinvstd.data_ptr<float>(), scratch.data_ptr<float>(),
      reinterpret_cast<unsigned int*>(counter.data_ptr()),
      gamma.data_ptr<float>(), dgamma.data_ptr<float>(),
      dbeta.data_ptr<float>(), coef.data_ptr<float>(), M, (int)C, relu,
      training, current_stream(x));
}

void bn_eval_prep_py(torch::Tensor rmean, torch::Tensor rvar,
                     torch::Tensor gamma, torch::Tensor beta,
                     torch::Tensor scale_shift, double eps, int64_t C) {
  check_f32(scale_shift, "scale_shift", 2 * C);
  bn_eval_prep(rmean.data_ptr<float>(), rvar.data_ptr<float>(),
               gamma.data_ptr<float>(), beta.data_ptr<float>(),
               scale_shift.data_ptr<float>(), eps, (int)C,
               current_stream(scale_shift));
}

void bn_fwd_apply_py(torch::Tensor x, torch::optional<torch::Tensor> res,
                     torch::Tensor y, torch::Tensor scale_shift, int64_t M,
                     int64_t C, bool relu) {
  check_act(x, "x");
  check_act(y, "y");
  check_f32(scale_shift, "scale_shift", 2 * C);
  const ushort_t* rp = nullptr;
  if (res.has_value()) {
    check_act(res.value(), "res");
    rp = bf16_ptr(res.value());
  }
  bn_fwd_apply(bf16_ptr(x), rp, bf16_mut(y), scale_shift.data_ptr<float>(),
               M, (int)C, relu, current_stream(x));
}

void bn_bwd_reduce_py(torch::Tensor x, torch::Tensor dy,
                      torch::optional<torch::Tensor> y, torch::Tensor smean,
                      torch::Tensor sinvstd, torch::Tensor scratch,
                      int64_t M, int64_t C, bool relu) {
  check_act(x, "x");
  check_act(dy, "dy");
  check_f32(scratch, "partials", bn_partials_numel(M, C));
  const ushort_t* yp = nullptr;
  if (relu) {
    TORCH_CHECK(y.has_value(), "y required for relu backward");
    check_act(y.value(), "y");
    yp = bf16_ptr(y.value());
  }
  bn_bwd_reduce(bf16_ptr(x), bf16_ptr(dy), yp, smean.data_ptr<float>(),
                sinvstd.data_ptr<float>(), scratch.data_ptr<float>(), M,
                (int)C, relu, current_stream(x));
}

void bn_bwd_finalize_py(torch::Tensor partials, torch::Tensor gamma,
                        torch::Tensor smean, torch::Tensor sinvstd,
                        torch::Tensor dgamma, torch::Tensor dbeta,
                        torch::Tensor coef, int64_t M, int64_t C,
                        bool training) {
  check_f32(partials, "partials", bn_partials_numel(M, C));
  check_f32(dgamma, "dgamma", C);
  check_f32(dbeta, "dbeta", C);
  check_f32(coef, "coef", 3 * C);
  bn_bwd_finalize(partials.data_ptr<float>(), gamma.data_ptr<float>(),
                  smean.data_ptr<float>(), sinvstd.data_ptr<float>(),
                  dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                  coef.data_ptr<float>(), M, (int)C, training,
                  current_stream(partials));
}

void bn_bwd_apply_py(torch::Tensor x, torch::Tensor dy,
                     torch::optional<torch::Tensor> y, torch::Tensor dx,
                     torch::optional<torch::Tensor> dres, torch::Tensor coef,
                     int64_t M, int64_t C, bool relu) {
  check_act(x, "x");
  check_act(dy, "dy");
  check_act(dx, "dx");
  check_f32(coef, "coef", 3 * C);
  const ushort_t* yp = nullptr;
  ushort_t* drp = nullptr;
  if (relu) {
    TORCH_CHECK(y.has_value(), "y required for relu backward");
    yp = bf16_ptr(y.value());
  }
  if (dres.has_value()) drp = bf16_mut(dres.value());
  bn_bwd_apply(bf16_ptr(x), bf16_ptr(dy), yp, bf16_mut(dx), drp,
               coef.data_ptr<float>(), M, (int)C, relu, current_stream(x));
}

}  // namespace

void init_comm_core(py::module_& m);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  init_comm_core(m);
  m.doc() = "gfx950 fused gossip + batchnorm kernels";
  m.def("scale_", &scale_, "x *= a (in place, fused over flat buffer)");
  m.def("add_scale_", &add_scale_, "x = (x + r) * a");
  m.def("pack_mix_", &pack_mix_, "x *= a; out = x");
  m.def("average_", &average_, "x = (x + y) / 2");
  m.def("sgd_step_", &sgd_step_, "fused momentum-SGD step");
  m.def("sgd_step_bf16gs_", &sgd_step_bf16gs_,
        "fused momentum-SGD step, bf16 grads + bf16 shadow write-back");
  m.def("cast_shadow_", &cast_shadow_, "fp32 master -> bf16 shadow");
  m.def("gather_multi_", &gather_multi_,
        "gather scattered grad tensors into the flat buffer (one launch)");
  m.def("mfma_probe", &mfma_probe, "single 16x16x32 MFMA layout probe");
  m.def("gemm_nt_bf16", &gemm_nt_bf16,
        "C[M,N] = A[M,K] @ B[N,K]^T, bf16 MFMA, fp32 accumulate");
  m.def("gemm_nt_bf16_v2", &gemm_nt_bf16_v2,
        "pipelined (register-staged double-buffer) variant");
  m.def("gemm_nt_bf16_v3", &gemm_nt_bf16_v3,
        "global_load_lds + st_16x32 swizzle variant (K % 64 == 0)");
  m.def("gemm_nt_bf16_v5", &gemm_nt_bf16_v5,
        "NT bf16 MFMA GEMM v5 (v3 + XCD-aware tile remap)");
  m.def("gemm_nt_bf16_v6", &gemm_nt_bf16_v6, py::arg("A"), py::arg("B"),
        py::arg("C"), py::arg("span") = false,
        "NT bf16 MFMA GEMM v6 (256x128 tile, 8 waves; span=3-buf "
        "barrier-crossing glds)");
  m.def("gemm_nt_bf16_v7", &gemm_nt_bf16_v7, py::arg("A"), py::arg("B"),
        py::arg("C"), py::arg("span") = true,
        "v6 with LDS-staged vectorized epilogue");
  m.def("gemm_nt_splitk_bf16", &gemm_nt_splitk_bf16, py::arg("A"),
        py::arg("B"), py::arg("C"), py::arg("split"),
        "NT bf16 MFMA GEMM, K split over block groups (small-M shapes)");
  m.def("gemm_nt_bf16_v4", &gemm_nt_bf16_v4,
        "3-buffer glds, raw barrier + counted vmcnt (full tiles only)");
  m.def("conv3x3_nhwc_bf16", &conv3x3_nhwc_bf16, py::arg("x"),
        py::arg("w"), py::arg("y"), py::arg("stride") = 1,
        "MFMA implicit-GEMM 3x3 conv, NHWC bf16, pad=1");
  m.def("conv3x3_wgrad_bf16", &conv3x3_wgrad_bf16, py::arg("x"),
        py::arg("dy"), py::arg("dw"), py::arg("stride") = 1,
        "MFMA implicit-TN 3x3 wgrad (split-M partials + reduce)");
  m.def("gemm_tn_wgrad_bf16", &gemm_tn_wgrad_bf16,
        "EXPERIMENTAL: dW = dy^T @ x with split-M partials (round-2 "
        "validation pending)");
  m.def("pack_mix_bf16_", &pack_mix_bf16_,
        "x *= a; out_bf16 = bf16(x) (wire-format pack)");
  m.def("add_scale_bf16_", &add_scale_bf16_,
        "x = (x + float(r_bf16)) * a (wire-format accumulate)");
  m.def("bn_partials_numel", &bn_partials_numel);
  m.def("bn_fwd_reduce_finalize", &bn_fwd_reduce_finalize_py,
        "fused BN fwd reduce + last-block finalize (one launch)");
  m.def("bn_bwd_reduce_finalize", &bn_bwd_reduce_finalize_py,
        "fused BN bwd reduce + last-block finalize (one launch)");
  m.def("bn_fwd_reduce", &bn_fwd_reduce_py);
  m.def("bn_fwd_finalize", &bn_fwd_finalize_py);
  m.def("bn_eval_prep", &bn_eval_prep_py);
  m.def("bn_fwd_apply", &bn_fwd_apply_py);
  m.def("bn_bwd_reduce", &bn_bwd_reduce_py);
  m.def("bn_bwd_finalize", &bn_bwd_finalize_py);
  m.def("bn_bwd_apply", &bn_bwd_apply_py);
}
