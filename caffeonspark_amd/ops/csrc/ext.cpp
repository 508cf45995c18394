// Torch-facing bindings for the gfx950 kernel library.  Shape logic lives in
// Python (ops/gpu.py); this layer validates tensors and launches kernels on
// the current HIP stream.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <cstdint>

typedef struct ihipStream_t* hipStream_t;

namespace cosamd {

void gemm_bf16(const void* A, const void* B, void* C, const float* bias,
               int M, int N, int K, int lda, int ldb, int ldc,
               bool trans_a, bool trans_b, int store_mode, int splitk,
               bool relu, float alpha, int m_alloc, int n_alloc,
               hipStream_t stream);
void gemm_conv_fwd(const void* X, const void* B, void* C,
                   const float* bias, const void* zpage, int M, int N,
                   int K, int ldb, int ldc, bool relu, bool accum,
                   const int* geom, hipStream_t stream);
void gemm_conv_dw(const void* A, const void* X, float* C, int M, int N,
                  int K, int lda, int ldc, int store_mode, int splitk,
                  float alpha, float* db, const int* geom,
                  hipStream_t stream);
void gemm_conv_fwd_sc(const void* X, const void* B, void* C,
                      const float* bias, int M, int N, int K, int ldb,
                      int ldc, bool relu, const int* geom,
                      hipStream_t stream);
void gemm_conv_dw_sc(const void* A, const void* X, float* C, int M,
                     int N, int K, int lda, int ldc, int store_mode,
                     int splitk, float alpha, float* db, const int* geom,
                     hipStream_t stream);
void tr16_probe(float* out, int mode, hipStream_t stream);
void repack_weights(const int64_t* table, int ndesc, int64_t max_total,
                    hipStream_t stream);
void dw_unpack_acc(const float* dwp, float* arena, int Kout, int Cg,
                   int R, int S, int Kpad, bool accumulate,
                   hipStream_t stream);
int lstm_persist_fwd(const void* xg, const void* w_hc, const void* cont,
                     void* h, float* c, float* act, void* h_in, float* hg,
                     int T, int N, int H, void* bar, hipStream_t stream);
int lstm_persist_bwd(const void* dy, const void* w_hcT, const void* cont,
                     const void* h, const float* c, const float* act,
                     void* dxg, float* dh_acc, float* dc_a, float* dc_b,
                     int T, int N, int H, void* bar, hipStream_t stream);
void wino_conv(const void* x, const float* w, const float* bias, void* y,
               void* U, void* V, void* Mbuf, int N, int H, int W, int Cin,
               int K, int wK, int wC, int u_rows_alloc, bool flip,
               bool relu, hipStream_t stream);
void bn_stats(const void* x, float* sum, float* sumsq, int64_t rows, int C,
              hipStream_t stream);
void bn_bwd_sums(const void* dy, const void* xhat, float* s1, float* s2,
                 int64_t rows, int C, hipStream_t stream);
void bn_norm(const void* x, void* y, const float* mean, const float* invstd,
             int64_t rows, int C, hipStream_t stream);
void bn_bwd(const void* xhat, const void* dy, void* dx,
            const float* invstd, const float* s1, const float* s2,
            float inv_m, int64_t rows, int C, hipStream_t stream);
void im2col_t(const void* x, void* colT, int N, int H, int W, int C,
              int P, int Q, int R, int S, int sh, int sw, int ph, int pw,
              int dil, int Kpad, int c0, int Cg, hipStream_t stream);
void im2col_nhwc(const void* x, void* col, int N, int H, int W, int C,
                 int P, int Q, int R, int S, int sh, int sw, int ph, int pw,
                 int dil, int Kpad, int c0, int Ct, hipStream_t stream);
void col2im_nhwc(const void* dcol, void* dx, int N, int H, int W, int C,
                 int P, int Q, int R, int S, int sh, int sw, int ph, int pw,
                 int dil, int Kpad, int c0, int Ct, hipStream_t stream);
void maxpool_fwd(const void* x, void* y, void* idx, bool idx16, int N,
                 int H, int W, int C, int P, int Q, int kh, int kw, int sh,
                 int sw, int ph, int pw, hipStream_t stream);
void maxpool_bwd(const void* dy, const void* idx, bool idx16, void* dx,
                 int N, int H, int W, int C, int P, int Q, int kh, int kw,
                 int sh, int sw, int ph, int pw, hipStream_t stream);
void avgpool_fwd(const void* x, void* y, int N, int H, int W, int C,
                 int P, int Q, int kh, int kw, int sh, int sw, int ph, int pw,
                 hipStream_t stream);
void avgpool_bwd(const void* dy, void* dx, int N, int H, int W, int C,
                 int P, int Q, int kh, int kw, int sh, int sw, int ph, int pw,
                 hipStream_t stream);
void lrn_fwd(const void* x, void* y, float* scale, int64_t npix, int C,
             int local_size, float alpha, float beta, float k,
             hipStream_t stream);
void lrn_bwd(const void* x, const void* y, const float* scale, const void* dy,
             void* dx, void* ratio, int64_t npix, int C, int local_size,
             float alpha, float beta, hipStream_t stream);
void relu_fwd(const void* x, void* y, float slope, int64_t n,
              hipStream_t stream);
void relu_bwd(const void* y, const void* dy, void* dx, float slope, int64_t n,
              hipStream_t stream);
void relu_bwd_strided(const void* y, const void* dy, void* dx, float slope,
                      int64_t rows, int C, int ldy, int lddy,
                      hipStream_t stream);
void relu_colsum_bwd(const void* y, const void* dy, void* dx, float* db,
                     float slope, int64_t rows, int cols, int ldy,
                     int lddy, hipStream_t stream);
void dropout_fwd(const void* x, void* y, void* mask, float ratio,
                 const void* seed, int64_t n, hipStream_t stream);
void seed_bump(void* s, hipStream_t stream);
void mul_bf16(const void* a, const void* b, void* y, int64_t n,
              hipStream_t stream);
void sgd_update(float* p, const float* g, float* v, float lr, float mu,
                float wd, int64_t n, hipStream_t stream);
void nesterov_update_multi(float* p, const float* g, float* v,
                           const int64_t* seg_off, const float* seg_lr,
                           const float* seg_wd, int nseg, float mu,
                           int64_t total, void* sh, hipStream_t stream);
void adam_update_multi(float* p, const float* g, float* m, float* v,
                       const int64_t* seg_off, const float* seg_lr,
                       const float* seg_wd, int nseg, float b1, float b2,
                       float eps, int64_t total, void* sh,
                       hipStream_t stream);
void sgd_update_multi(float* p, const float* g, float* v,
                      const int64_t* seg_off, const float* seg_lr,
                      const float* seg_wd, int nseg, float mu,
                      int64_t total, void* sh, hipStream_t stream);
void colsum(const void* in, float* out, int64_t rows, int cols, int ld,
            hipStream_t stream);
void transpose_bf16(const void* in, void* out, int64_t R, int64_t C,
                    hipStream_t stream);
void bias_act_cast(const float* in, const float* bias, void* out,
                   int64_t rows, int cols, bool relu, hipStream_t stream);
void lstm_unit_fwd(const float* c_prev, const void* gates, const void* cont,
                   float* c_out, void* h_out, float* act, int64_t n, int H,
                   hipStream_t stream);
void lstm_unit_bwd(const float* c_prev, const float* c_out, const float* act,
                   const void* cont, const float* dc_next, const void* dh,
                   float* dc_prev, void* dgates, int64_t n, int H,
                   hipStream_t stream);
void embed_fwd(const float* idx, const void* w, void* y, int64_t n, int E,
               int V, hipStream_t stream);
void embed_bwd(const float* idx, const void* dy, float* dw, int64_t n, int E,
               int V, hipStream_t stream);
void lstm_seq_fwd(const void* xg, const void* w_hc, const void* cont,
                  void* h, float* c, float* act, void* h_in, float* hg,
                  int T, int N, int H, int n_alloc_whc, hipStream_t stream);
void lstm_seq_bwd(const void* dy, const void* w_hcT, const void* cont,
                  const void* h, const float* c, const float* act,
                  void* dxg, void* dh_rec, float* dh_f, float* dc_a,
                  float* dc_b, int T, int N, int H, int n_alloc_whcT,
                  hipStream_t stream);
void softmax_loss_fwd(const void* x, const float* label, float* prob,
                      float* loss, int* count, int64_t nrows, int C,
                      int ignore, bool has_ignore, hipStream_t stream);
void softmax_loss_bwd(const float* prob, const float* label, void* dx,
                      float scale, int64_t nrows, int C, int ignore,
                      bool has_ignore, hipStream_t stream);

}  // namespace cosamd

namespace {

using at::Tensor;

hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

#define CHECK_BF16(t) \
  TORCH_CHECK((t).scalar_type() == at::kBFloat16, #t " must be bf16")
#define CHECK_F32(t) \
  TORCH_CHECK((t).scalar_type() == at::kFloat, #t " must be fp32")
#define CHECK_CUDA(t) TORCH_CHECK((t).is_cuda(), #t " must be on GPU")

void py_gemm(Tensor A, Tensor B, Tensor C, c10::optional<Tensor> bias,
             int64_t M, int64_t N, int64_t K, int64_t lda, int64_t ldb,
             int64_t ldc, bool trans_a, bool trans_b, int64_t store_mode,
             int64_t splitk, bool relu, double alpha, int64_t m_alloc,
             int64_t n_alloc) {
  CHECK_CUDA(A); CHECK_BF16(A); CHECK_BF16(B);
  const float* bptr = nullptr;
  if (bias.has_value()) {
    CHECK_F32(bias.value());
    bptr = bias.value().data_ptr<float>();
  }
  cosamd::gemm_bf16(
      A.data_ptr(), B.data_ptr(), C.data_ptr(), bptr, (int)M, (int)N, (int)K, (int)lda, (int)ldb, (int)ldc,
      trans_a, trans_b, (int)store_mode, (int)splitk, relu, (float)alpha,
      (int)m_alloc, (int)n_alloc, cur_stream());
}

void py_gemm_conv_fwd(Tensor X, Tensor B, Tensor C,
                      c10::optional<Tensor> bias, Tensor zpage,
                      int64_t M, int64_t N, int64_t K, int64_t ldb,
                      int64_t ldc, bool relu, bool accum,
                      std::vector<int64_t> geom) {
  CHECK_CUDA(X); CHECK_BF16(X); CHECK_BF16(B); CHECK_BF16(C);
  const float* bptr = nullptr;
  if (bias.has_value()) {
    CHECK_F32(bias.value());
    bptr = bias.value().data_ptr<float>();
  }
  int g[14];
  TORCH_CHECK(geom.size() == 14, "geom must have 14 ints");
  for (int i = 0; i < 14; ++i) g[i] = (int)geom[i];
  cosamd::gemm_conv_fwd(X.data_ptr(), B.data_ptr(), C.data_ptr(), bptr,
                        zpage.data_ptr(), (int)M, (int)N, (int)K,
                        (int)ldb, (int)ldc, relu, accum, g,
                        cur_stream());
}

void py_gemm_conv_dw(Tensor A, Tensor X, Tensor C,
                     c10::optional<Tensor> db, int64_t M, int64_t N,
                     int64_t K, int64_t lda, int64_t ldc,
                     int64_t store_mode, int64_t splitk, double alpha,
                     std::vector<int64_t> geom) {
  CHECK_CUDA(A); CHECK_BF16(A); CHECK_BF16(X); CHECK_F32(C);
  float* dbp = nullptr;
  if (db.has_value()) {
    CHECK_F32(db.value());
    dbp = db.value().data_ptr<float>();
  }
  int g[14];
  TORCH_CHECK(geom.size() == 14, "geom must have 14 ints");
  for (int i = 0; i < 14; ++i) g[i] = (int)geom[i];
  cosamd::gemm_conv_dw(A.data_ptr(), X.data_ptr(), C.data_ptr<float>(),
                       (int)M, (int)N, (int)K, (int)lda, (int)ldc,
                       (int)store_mode, (int)splitk, (float)alpha, dbp,
                       g, cur_stream());
}

void py_relu_colsum_bwd(Tensor y, Tensor dy, Tensor dx, Tensor db,
                        double slope, int64_t rows, int64_t cols,
                        int64_t ldy, int64_t lddy) {
  CHECK_CUDA(y); CHECK_BF16(y); CHECK_BF16(dy); CHECK_BF16(dx);
  CHECK_F32(db);
  cosamd::relu_colsum_bwd(y.data_ptr(), dy.data_ptr(), dx.data_ptr(),
                          db.data_ptr<float>(), (float)slope, rows,
                          (int)cols, (int)ldy, (int)lddy, cur_stream());
}

void py_gemm_conv_fwd_sc(Tensor X, Tensor B, Tensor C,
                         c10::optional<Tensor> bias, int64_t M,
                         int64_t N, int64_t K, int64_t ldb, int64_t ldc,
                         bool relu, std::vector<int64_t> geom) {
  CHECK_CUDA(X); CHECK_BF16(X); CHECK_BF16(B); CHECK_BF16(C);
  const float* bptr = nullptr;
  if (bias.has_value()) {
    CHECK_F32(bias.value());
    bptr = bias.value().data_ptr<float>();
  }
  int g[14];
  TORCH_CHECK(geom.size() == 14, "geom must have 14 ints");
  for (int i = 0; i < 14; ++i) g[i] = (int)geom[i];
  cosamd::gemm_conv_fwd_sc(X.data_ptr(), B.data_ptr(), C.data_ptr(),
                           bptr, (int)M, (int)N, (int)K, (int)ldb,
                           (int)ldc, relu, g, cur_stream());
}

void py_gemm_conv_dw_sc(Tensor A, Tensor X, Tensor C,
                        c10::optional<Tensor> db, int64_t M, int64_t N,
                        int64_t K, int64_t lda, int64_t ldc,
                        int64_t store_mode, int64_t splitk, double alpha,
                        std::vector<int64_t> geom) {
  CHECK_CUDA(A); CHECK_BF16(A); CHECK_BF16(X); CHECK_F32(C);
  float* dbp = nullptr;
  if (db.has_value()) {
    CHECK_F32(db.value());
    dbp = db.value().data_ptr<float>();
  }
  int g[14];
  TORCH_CHECK(geom.size() == 14, "geom must have 14 ints");
  for (int i = 0; i < 14; ++i) g[i] = (int)geom[i];
  cosamd::gemm_conv_dw_sc(A.data_ptr(), X.data_ptr(),
                          C.data_ptr<float>(), (int)M, (int)N, (int)K,
                          (int)lda, (int)ldc, (int)store_mode,
                          (int)splitk, (float)alpha, dbp, g,
                          cur_stream());
}

void py_im2col(Tensor x, Tensor col, int64_t N, int64_t H, int64_t W,
               int64_t C, int64_t P, int64_t Q, int64_t R, int64_t S,
               int64_t sh, int64_t sw, int64_t ph, int64_t pw, int64_t dil,
               int64_t Kpad, int64_t c0, int64_t Ct) {
  CHECK_CUDA(x); CHECK_BF16(x); CHECK_BF16(col);
  cosamd::im2col_nhwc(x.data_ptr(), col.data_ptr(), N, H, W, C, P, Q, R, S,
                      sh, sw, ph, pw, dil, Kpad, c0, Ct, cur_stream());
}

void py_col2im(Tensor dcol, Tensor dx, int64_t N, int64_t H, int64_t W,
               int64_t C, int64_t P, int64_t Q, int64_t R, int64_t S,
               int64_t sh, int64_t sw, int64_t ph, int64_t pw, int64_t dil,
               int64_t Kpad, int64_t c0, int64_t Ct) {
  CHECK_CUDA(dcol); CHECK_BF16(dcol); CHECK_BF16(dx);
  cosamd::col2im_nhwc(dcol.data_ptr(), dx.data_ptr(), N, H, W, C, P, Q, R, S,
                      sh, sw, ph, pw, dil, Kpad, c0, Ct, cur_stream());
}

void py_maxpool_fwd(Tensor x, Tensor y, Tensor idx, int64_t N, int64_t H,
                    int64_t W, int64_t C, int64_t P, int64_t Q, int64_t kh,
                    int64_t kw, int64_t sh, int64_t sw, int64_t ph,
                    int64_t pw) {
  CHECK_CUDA(x); CHECK_BF16(x);
  cosamd::maxpool_fwd(x.data_ptr(), y.data_ptr(), idx.data_ptr(),
                      idx.scalar_type() == at::kChar, N, H,
                      W, C, P, Q, kh, kw, sh, sw, ph, pw, cur_stream());
}

void py_maxpool_bwd(Tensor dy, Tensor idx, Tensor dx, int64_t N, int64_t H,
                    int64_t W, int64_t C, int64_t P, int64_t Q, int64_t kh,
                    int64_t kw, int64_t sh, int64_t sw, int64_t ph,
                    int64_t pw) {
  cosamd::maxpool_bwd(dy.data_ptr(), idx.data_ptr(),
                      idx.scalar_type() == at::kChar, dx.data_ptr(), N, H,
                      W, C, P, Q, kh, kw, sh, sw, ph, pw, cur_stream());
}

void py_avgpool_fwd(Tensor x, Tensor y, int64_t N, int64_t H, int64_t W,
                    int64_t C, int64_t P, int64_t Q, int64_t kh, int64_t kw,
                    int64_t sh, int64_t sw, int64_t ph, int64_t pw) {
  cosamd::avgpool_fwd(x.data_ptr(), y.data_ptr(), N, H, W, C, P, Q, kh, kw,
                      sh, sw, ph, pw, cur_stream());
}

void py_avgpool_bwd(Tensor dy, Tensor dx, int64_t N, int64_t H, int64_t W,
                    int64_t C, int64_t P, int64_t Q, int64_t kh, int64_t kw,
                    int64_t sh, int64_t sw, int64_t ph, int64_t pw) {
  cosamd::avgpool_bwd(dy.data_ptr(), dx.data_ptr(), N, H, W, C, P, Q, kh, kw,
                      sh, sw, ph, pw, cur_stream());
}

void py_lrn_fwd(Tensor x, Tensor y, Tensor scale, int64_t npix, int64_t C,
                int64_t local_size, double alpha, double beta, double k) {
  CHECK_F32(scale);
  cosamd::lrn_fwd(x.data_ptr(), y.data_ptr(), scale.data_ptr<float>(), npix,
                  C, local_size, alpha, beta, k, cur_stream());
}

void py_lrn_bwd(Tensor x, Tensor y, Tensor scale, Tensor dy, Tensor dx,
                Tensor ratio, int64_t npix, int64_t C, int64_t local_size,
                double alpha, double beta) {
  cosamd::lrn_bwd(x.data_ptr(), y.data_ptr(), scale.data_ptr<float>(),
                  dy.data_ptr(), dx.data_ptr(), ratio.data_ptr(), npix, C,
                  local_size, alpha, beta, cur_stream());
}

void py_relu_fwd(Tensor x, Tensor y, double slope) {
  CHECK_BF16(x);
  cosamd::relu_fwd(x.data_ptr(), y.data_ptr(), (float)slope, x.numel(),
                   cur_stream());
}

void py_relu_bwd(Tensor y, Tensor dy, Tensor dx, double slope) {
  cosamd::relu_bwd(y.data_ptr(), dy.data_ptr(), dx.data_ptr(), (float)slope,
                   y.numel(), cur_stream());
}

void py_dropout_fwd(Tensor x, Tensor y, Tensor mask, double ratio,
                    Tensor seed) {
  cosamd::dropout_fwd(x.data_ptr(), y.data_ptr(), mask.data_ptr(), ratio,
                      seed.data_ptr(), x.numel(), cur_stream());
}

void py_mul(Tensor a, Tensor b, Tensor y) {
  cosamd::mul_bf16(a.data_ptr(), b.data_ptr(), y.data_ptr(), a.numel(),
                   cur_stream());
}

void py_sgd_update_multi(Tensor p, Tensor g, Tensor v, Tensor seg_off,
                         Tensor seg_lr, Tensor seg_wd, double mu,
                         c10::optional<Tensor> sh) {
  CHECK_F32(p); CHECK_F32(g); CHECK_F32(v);
  void* shp = nullptr;
  if (sh.has_value()) { CHECK_BF16(sh.value()); shp = sh->data_ptr(); }
  cosamd::sgd_update_multi(p.data_ptr<float>(), g.data_ptr<float>(),
                           v.data_ptr<float>(),
                           seg_off.data_ptr<int64_t>(),
                           seg_lr.data_ptr<float>(),
                           seg_wd.data_ptr<float>(),
                           (int)seg_lr.numel(), mu, p.numel(), shp,
                           cur_stream());
}

void py_nesterov_update_multi(Tensor p, Tensor g, Tensor v, Tensor seg_off,
                              Tensor seg_lr, Tensor seg_wd, double mu,
                              c10::optional<Tensor> sh) {
  CHECK_F32(p); CHECK_F32(g); CHECK_F32(v);
  void* shp = nullptr;
  if (sh.has_value()) { CHECK_BF16(sh.value()); shp = sh->data_ptr(); }
  cosamd::nesterov_update_multi(p.data_ptr<float>(), g.data_ptr<float>(),
                                v.data_ptr<float>(),
                                seg_off.data_ptr<int64_t>(),
                                seg_lr.data_ptr<float>(),
                                seg_wd.data_ptr<float>(),
                                (int)seg_lr.numel(), mu, p.numel(), shp,
                                cur_stream());
}

void py_adam_update_multi(Tensor p, Tensor g, Tensor m, Tensor v,
                          Tensor seg_off, Tensor seg_lr, Tensor seg_wd,
                          double b1, double b2, double eps,
                          c10::optional<Tensor> sh) {
  CHECK_F32(p); CHECK_F32(g); CHECK_F32(m); CHECK_F32(v);
  void* shp = nullptr;
  if (sh.has_value()) { CHECK_BF16(sh.value()); shp = sh->data_ptr(); }
  cosamd::adam_update_multi(p.data_ptr<float>(), g.data_ptr<float>(),
                            m.data_ptr<float>(), v.data_ptr<float>(),
                            seg_off.data_ptr<int64_t>(),
                            seg_lr.data_ptr<float>(),
                            seg_wd.data_ptr<float>(),
                            (int)seg_lr.numel(), b1, b2, eps, p.numel(),
                            shp, cur_stream());
}

void py_sgd_update(Tensor p, Tensor g, Tensor v, double lr, double mu,
                   double wd) {
  CHECK_F32(p); CHECK_F32(g); CHECK_F32(v);
  cosamd::sgd_update(p.data_ptr<float>(), g.data_ptr<float>(),
                     v.data_ptr<float>(), lr, mu, wd, p.numel(),
                     cur_stream());
}

void py_bias_act_cast(Tensor in, c10::optional<Tensor> bias, Tensor out,
                      bool relu) {
  CHECK_F32(in);
  const float* bptr = bias.has_value() ? bias.value().data_ptr<float>()
                                       : nullptr;
  cosamd::bias_act_cast(in.data_ptr<float>(), bptr, out.data_ptr(),
                        in.size(0), in.size(1), relu, cur_stream());
}

void py_dw_unpack_acc(Tensor dwp, Tensor arena, int64_t Kout, int64_t Cg,
                      int64_t R, int64_t S, int64_t Kpad,
                      bool accumulate) {
  cosamd::dw_unpack_acc(dwp.data_ptr<float>(), arena.data_ptr<float>(),
                        (int)Kout, (int)Cg, (int)R, (int)S, (int)Kpad,
                        accumulate, cur_stream());
}

void py_repack_weights(Tensor table, int64_t ndesc, int64_t max_total) {
  cosamd::repack_weights(table.data_ptr<int64_t>(), (int)ndesc, max_total,
                         cur_stream());
}

int64_t py_lstm_persist_fwd(Tensor xg, Tensor w_hc, Tensor cont,
                            Tensor h, Tensor c, Tensor act, Tensor h_in,
                            Tensor hg, int64_t T, int64_t N, int64_t H,
                            Tensor bar) {
  return cosamd::lstm_persist_fwd(
      xg.data_ptr(), w_hc.data_ptr(), cont.data_ptr(), h.data_ptr(),
      c.data_ptr<float>(), act.data_ptr<float>(), h_in.data_ptr(),
      hg.data_ptr<float>(), (int)T, (int)N, (int)H, bar.data_ptr(),
      cur_stream());
}

int64_t py_lstm_persist_bwd(Tensor dy, Tensor w_hcT, Tensor cont,
                            Tensor h, Tensor c, Tensor act, Tensor dxg,
                            Tensor dh_acc, Tensor dc_a, Tensor dc_b,
                            int64_t T, int64_t N, int64_t H, Tensor bar) {
  return cosamd::lstm_persist_bwd(
      dy.data_ptr(), w_hcT.data_ptr(), cont.data_ptr(), h.data_ptr(),
      c.data_ptr<float>(), act.data_ptr<float>(), dxg.data_ptr(),
      dh_acc.data_ptr<float>(), dc_a.data_ptr<float>(),
      dc_b.data_ptr<float>(), (int)T, (int)N, (int)H, bar.data_ptr(),
      cur_stream());
}

void py_tr16_probe(Tensor out, int64_t mode) {
  cosamd::tr16_probe(out.data_ptr<float>(), (int)mode, cur_stream());
}

void py_transpose(Tensor in, Tensor out, int64_t R, int64_t C) {
  CHECK_BF16(in); CHECK_BF16(out);
  cosamd::transpose_bf16(in.data_ptr(), out.data_ptr(), R, C, cur_stream());
}

void py_colsum(Tensor in, Tensor out, int64_t rows, int64_t cols,
               int64_t ld) {
  CHECK_F32(out);
  cosamd::colsum(in.data_ptr(), out.data_ptr<float>(), rows, cols, ld,
                 cur_stream());
}

void py_lstm_unit_fwd(Tensor c_prev, Tensor gates, Tensor cont, Tensor c_out,
                      Tensor h_out, Tensor act) {
  CHECK_F32(c_prev); CHECK_BF16(gates);
  int64_t n = c_prev.numel();
  int H = c_prev.size(-1);
  cosamd::lstm_unit_fwd(c_prev.data_ptr<float>(), gates.data_ptr(),
                        cont.data_ptr(), c_out.data_ptr<float>(),
                        h_out.data_ptr(), act.data_ptr<float>(), n, H,
                        cur_stream());
}

void py_lstm_unit_bwd(Tensor c_prev, Tensor c_out, Tensor act, Tensor cont,
                      Tensor dc_next, Tensor dh, Tensor dc_prev,
                      Tensor dgates) {
  int64_t n = c_prev.numel();
  int H = c_prev.size(-1);
  cosamd::lstm_unit_bwd(c_prev.data_ptr<float>(), c_out.data_ptr<float>(),
                        act.data_ptr<float>(), cont.data_ptr(),
                        dc_next.data_ptr<float>(), dh.data_ptr(),
                        dc_prev.data_ptr<float>(), dgates.data_ptr(), n, H,
                        cur_stream());
}

void py_embed_fwd(Tensor idx, Tensor w, Tensor y, int64_t V) {
  CHECK_F32(idx); CHECK_BF16(w);
  cosamd::embed_fwd(idx.data_ptr<float>(), w.data_ptr(), y.data_ptr(),
                    idx.numel(), w.size(1), V, cur_stream());
}

void py_embed_bwd(Tensor idx, Tensor dy, Tensor dw, int64_t V) {
  CHECK_F32(dw);
  cosamd::embed_bwd(idx.data_ptr<float>(), dy.data_ptr(),
                    dw.data_ptr<float>(), idx.numel(), dw.size(1), V,
                    cur_stream());
}

void py_lstm_seq_fwd(Tensor xg, Tensor w_hc, Tensor cont, Tensor h,
                     Tensor c, Tensor act, Tensor h_in, Tensor hg,
                     int64_t T, int64_t N, int64_t H, int64_t n_alloc) {
  CHECK_BF16(xg); CHECK_BF16(w_hc); CHECK_F32(c); CHECK_F32(act);
  CHECK_F32(hg);
  cosamd::lstm_seq_fwd(xg.data_ptr(), w_hc.data_ptr(), cont.data_ptr(),
                       h.data_ptr(), c.data_ptr<float>(),
                       act.data_ptr<float>(), h_in.data_ptr(),
                       hg.data_ptr<float>(), T, N, H, n_alloc,
                       cur_stream());
}

void py_lstm_seq_bwd(Tensor dy, Tensor w_hcT, Tensor cont, Tensor h,
                     Tensor c, Tensor act, Tensor dxg, Tensor dh_rec,
                     Tensor dh_f, Tensor dc_a, Tensor dc_b, int64_t T,
                     int64_t N, int64_t H, int64_t n_alloc) {
  CHECK_BF16(dy); CHECK_BF16(w_hcT); CHECK_F32(dh_f);
  cosamd::lstm_seq_bwd(dy.data_ptr(), w_hcT.data_ptr(), cont.data_ptr(),
                       h.data_ptr(), c.data_ptr<float>(),
                       act.data_ptr<float>(), dxg.data_ptr(),
                       dh_rec.data_ptr(), dh_f.data_ptr<float>(),
                       dc_a.data_ptr<float>(), dc_b.data_ptr<float>(),
                       T, N, H, n_alloc, cur_stream());
}

void py_softmax_loss_fwd(Tensor x, Tensor label, Tensor prob, Tensor loss,
                         Tensor count, int64_t ignore, bool has_ignore) {
  CHECK_BF16(x); CHECK_F32(label); CHECK_F32(prob);
  cosamd::softmax_loss_fwd(x.data_ptr(), label.data_ptr<float>(),
                           prob.data_ptr<float>(), loss.data_ptr<float>(),
                           count.data_ptr<int>(), x.size(0), x.size(1),
                           (int)ignore, has_ignore, cur_stream());
}

void py_softmax_loss_bwd(Tensor prob, Tensor label, Tensor dx, double scale,
                         int64_t ignore, bool has_ignore) {
  cosamd::softmax_loss_bwd(prob.data_ptr<float>(), label.data_ptr<float>(),
                           dx.data_ptr(), scale, prob.size(0), prob.size(1),
                           (int)ignore, has_ignore, cur_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("gemm", &py_gemm);
  m.def("relu_colsum_bwd", &py_relu_colsum_bwd);
  m.def("gemm_conv_fwd", &py_gemm_conv_fwd);
  m.def("gemm_conv_dw", &py_gemm_conv_dw);
  m.def("gemm_conv_fwd_sc", &py_gemm_conv_fwd_sc);
  m.def("gemm_conv_dw_sc", &py_gemm_conv_dw_sc);
  m.def("im2col", &py_im2col);
  m.def("bn_stats", [](Tensor x, Tensor sum, Tensor sq, int64_t rows,
                       int64_t C) {
    cosamd::bn_stats(x.data_ptr(), sum.data_ptr<float>(),
                     sq.data_ptr<float>(), rows, C, cur_stream());
  });
  m.def("bn_bwd_sums", [](Tensor dy, Tensor xhat, Tensor s1, Tensor s2,
                          int64_t rows, int64_t C) {
    cosamd::bn_bwd_sums(dy.data_ptr(), xhat.data_ptr(),
                        s1.data_ptr<float>(), s2.data_ptr<float>(), rows,
                        C, cur_stream());
  });
  m.def("bn_norm", [](Tensor x, Tensor y, Tensor mean, Tensor invstd,
                      int64_t rows, int64_t C) {
    cosamd::bn_norm(x.data_ptr(), y.data_ptr(), mean.data_ptr<float>(),
                    invstd.data_ptr<float>(), rows, C, cur_stream());
  });
  m.def("bn_bwd", [](Tensor xhat, Tensor dy, Tensor dx, Tensor invstd,
                     Tensor s1, Tensor s2, double inv_m, int64_t rows,
                     int64_t C) {
    cosamd::bn_bwd(xhat.data_ptr(), dy.data_ptr(), dx.data_ptr(),
                   invstd.data_ptr<float>(), s1.data_ptr<float>(),
                   s2.data_ptr<float>(), (float)inv_m, rows, C,
                   cur_stream());
  });
  m.def("wino_conv", [](Tensor x, Tensor w, c10::optional<Tensor> bias,
                        Tensor y, Tensor U, Tensor V, Tensor M, int64_t N,
                        int64_t H, int64_t W, int64_t Cin, int64_t K,
                        int64_t wK, int64_t wC, int64_t u_rows_alloc,
                        bool flip, bool relu) {
    cosamd::wino_conv(x.data_ptr(), w.data_ptr<float>(),
                      bias ? bias->data_ptr<float>() : nullptr,
                      y.data_ptr(), U.data_ptr(), V.data_ptr(),
                      M.data_ptr(), N, H, W, Cin, K, wK, wC, u_rows_alloc,
                      flip, relu, cur_stream());
  });
  m.def("im2col_t", [](Tensor x, Tensor colT, int64_t N, int64_t H,
                       int64_t W, int64_t C, int64_t P, int64_t Q,
                       int64_t R, int64_t S, int64_t sh, int64_t sw,
                       int64_t ph, int64_t pw, int64_t dil, int64_t Kpad,
                       int64_t c0, int64_t Cg) {
    cosamd::im2col_t(x.data_ptr(), colT.data_ptr(), N, H, W, C, P, Q, R,
                     S, sh, sw, ph, pw, dil, Kpad, c0, Cg, cur_stream());
  });
  m.def("col2im", &py_col2im);
  m.def("maxpool_fwd", &py_maxpool_fwd);
  m.def("maxpool_bwd", &py_maxpool_bwd);
  m.def("avgpool_fwd", &py_avgpool_fwd);
  m.def("avgpool_bwd", &py_avgpool_bwd);
  m.def("lrn_fwd", &py_lrn_fwd);
  m.def("lrn_bwd", &py_lrn_bwd);
  m.def("relu_fwd", &py_relu_fwd);
  m.def("relu_bwd", &py_relu_bwd);
  m.def("dropout_fwd", &py_dropout_fwd);
  m.def("relu_bwd_strided", [](Tensor y, Tensor dy, Tensor dx, double slope,
                               int64_t rows, int64_t C, int64_t ldy,
                               int64_t lddy) {
    cosamd::relu_bwd_strided(y.data_ptr(), dy.data_ptr(), dx.data_ptr(),
                             (float)slope, rows, C, ldy, lddy,
                             cur_stream());
  });
  m.def("seed_bump", [](Tensor s) {
    cosamd::seed_bump(s.data_ptr(), cur_stream());
  });
  m.def("mul", &py_mul);
  m.def("sgd_update", &py_sgd_update);
  m.def("sgd_update_multi", &py_sgd_update_multi);
  m.def("nesterov_update_multi", &py_nesterov_update_multi);
  m.def("adam_update_multi", &py_adam_update_multi);
  m.def("colsum", &py_colsum);
  m.def("transpose", &py_transpose);
  m.def("tr16_probe", &py_tr16_probe);
  m.def("repack_weights", &py_repack_weights);
  m.def("dw_unpack_acc", &py_dw_unpack_acc);
  m.def("lstm_persist_fwd", &py_lstm_persist_fwd);
  m.def("lstm_persist_bwd", &py_lstm_persist_bwd);
  m.def("bias_act_cast", &py_bias_act_cast);
  m.def("lstm_seq_fwd", &py_lstm_seq_fwd);
  m.def("lstm_seq_bwd", &py_lstm_seq_bwd);
  m.def("lstm_unit_fwd", &py_lstm_unit_fwd);
  m.def("lstm_unit_bwd", &py_lstm_unit_bwd);
  m.def("embed_fwd", &py_embed_fwd);
  m.def("embed_bwd", &py_embed_bwd);
  m.def("softmax_loss_fwd", &py_softmax_loss_fwd);
  m.def("softmax_loss_bwd", &py_softmax_loss_bwd);
}
