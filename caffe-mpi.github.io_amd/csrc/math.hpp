// math.hpp — CPU math for the engine's CPU mode and launcher declarations
// for the hand-written gfx950 HIP kernels (implemented in kernels/*.hip).
// Replaces the reference's util/math_functions.{cpp,cu} + util/im2col.* —
// note we deliberately do NOT reproduce the reference's per-call stream
// synchronization (math_functions.cu:26); all launches are async on the
// caller's stream.
#pragma once

#include "core.hpp"

namespace camd {

// ------------------------------------------------------------- CPU math
namespace cpu {
// Row-major C[M][N] = alpha*op(A)*op(B) + beta*C (fp32 accumulation, blocked)
void gemm(bool transA, bool transB, long M, long N, long K, float alpha,
          const float* A, const float* B, float beta, float* C);
void axpy(long n, float a, const float* x, float* y);
void axpby(long n, float a, const float* x, float b, float* y);
void scal(long n, float a, float* x);
void im2col(const float* im, int C, int H, int W, int kh, int kw, int ph,
            int pw, int sh, int sw, int dh, int dw, float* col);
void col2im(const float* col, int C, int H, int W, int kh, int kw, int ph,
            int pw, int sh, int sw, int dh, int dw, float* im);
}  // namespace cpu

inline int conv_out_dim(int in, int k, int pad, int stride, int dil) {
  return (in + 2 * pad - (dil * (k - 1) + 1)) / stride + 1;
}
void pool_out_dim(int H, int W, int kh, int kw, int ph, int pw, int sh,
                  int sw, int* OH, int* OW);  // ceil + clip, pooling_layer.cpp:86

// ------------------------------------------------------------ GPU launchers
namespace gpu {

// Epilogue for the MFMA GEMM (bias/ReLU fusion + the conv NCHW scatter that
// replaces a separate bias/copy pass).
struct GemmEpi {
  // when spad > 0: C column c maps to image n = c / spad, pixel s = c % spad;
  // columns with s >= S are padding and are dropped; output element is
  // C[n * n_stride + row * S + s] (NCHW scatter; the caller pre-offsets C by
  // the group's channel base and passes n_stride = Cout_total * S).
  long spad = 0;
  long S = 0;
  long n_stride = 0;
  const float* bias = nullptr;
  bool bias_per_col = false;  // IP: bias indexed by column, conv: by row
  bool relu = false;
  // strided scatter (1x1/s2 dgrad without a col2im pass): column pixel
  // sp -> (oh, ow) over OWo maps to destination pixel oh*osh*Wd + ow*osw;
  // per-row stride is Srow (= dest H*W).  OWo == 0 -> plain NCHW scatter.
  long Srow = 0;
  int OWo = 0, osh = 1, osw = 1, Wd = 0;
};

// NCHW-view operand: the GEMM axis that is contiguous in the stored tensor
// is an image axis q = n*spad + sp (sp < S valid, rest padding); element
// (r, q) lives at base[(n*chan + r)*S + sp].  Lets conv GEMMs read x / dY
// straight out of NCHW with no transpose/materialization pass.
//
// With kh > 0 the view becomes an IMPLICIT-IM2COL view (the north_star's
// implicit-GEMM conv): r indexes a col row (c, ki, kj), sp an output pixel
// (oh, ow); the element is x[n][c][oh*sh-ph+ki][ow*sw-pw+kj] (0 outside) —
// the col matrix is never materialized.  stride/dilation 1 only (every
// ResNet/GoogLeNet/AlexNet conv that needs backward-data is s1/d1; strided
// convs use the explicit col path).
struct GemmView {
  long spad = 0;  // 0 = plain operand
  long S = 0;     // valid pixels per image (OH*OW)
  long chan = 0;  // channels of the viewed tensor (address stride term)
  int kh = 0, kw = 0, ph = 0, pw = 0;  // kh>0 => implicit im2col
  int H = 0, W = 0, OW = 0;            // input dims / output row width
  int sh = 1, sw = 1;  // conv strides (round 2: strided implicit im2col —
                       // conv1 7x7s2, the 1x1s2 projections, AlexNet s4)
};

// Wt[ci][co*kh*kw + ki*kw + kj] = W[co][ci][kh-1-ki][kw-1-kj] — the
// flipped/transposed weights that turn backward-data into a forward
// convolution over dY (s1 only)
void weight_flip(hipStream_t s, const float* w, int Cout, int Cin, int kh,
                 int kw, float* wt);
void weight_flip_grouped(hipStream_t s, const float* w, int Cout, int Cin_g,
                         int kh, int kw, int groups, float* wt);

void gemm(hipStream_t s, bool transA, bool transB, long M, long N, long K,
          float alpha, const float* A, long lda, const float* B, long ldb,
          float beta, float* C, long ldc, const GemmEpi* epi = nullptr,
          const GemmView* aview = nullptr, const GemmView* bview = nullptr);

// db[c] = Σ_n Σ_s dy[n][c][s] (deterministic)
void bias_grad(hipStream_t s, const float* dy, int N, int C, long S,
               float* db);

// col[K][Nimg*Spad] from x[Nimg][C][H][W]; pad columns zero-filled.
void im2col_batched(hipStream_t s, const float* x, int Nimg, int C, int H,
                    int W, int kh, int kw, int ph, int pw, int sh, int sw,
                    int dh, int dw, int OH, int OW, long Spad, float* col);
// dx[Nimg][C][H][W] from dcol[K][Nimg*Spad] (gather, deterministic)
void col2im_batched(hipStream_t s, const float* dcol, int Nimg, int C, int H,
                    int W, int kh, int kw, int ph, int pw, int sh, int sw,
                    int dh, int dw, int OH, int OW, long Spad, float* dx);

// y += a*x (iter_size diff accumulation over the padded arena)
void axpy(hipStream_t s, long n, float a, const float* x, float* y);

// y = x * a[c] (+ b[c]) — Scale layer fwd / bwd-data
void chan_affine(hipStream_t s, const float* x, const float* a,
                 const float* b, int N, int C, long S, float* y);

void relu_fwd(hipStream_t s, const float* x, long n, float slope, float* y);
void relu_bwd(hipStream_t s, const float* x, const float* dy, long n,
              float slope, float* dx);

void pool_max_fwd(hipStream_t s, const float* x, int N, int C, int H, int W,
                  int kh, int kw, int ph, int pw, int sh, int sw, int OH,
                  int OW, float* y, int* mask);
void pool_max_bwd(hipStream_t s, const float* dy, const int* mask, int N,
                  int C, int H, int W, int kh, int kw, int ph, int pw,
                  int sh, int sw, int OH, int OW, float* dx);
void pool_ave_fwd(hipStream_t s, const float* x, int N, int C, int H, int W,
                  int kh, int kw, int ph, int pw, int sh, int sw, int OH,
                  int OW, float* y);
void pool_ave_bwd(hipStream_t s, const float* dy, int N, int C, int H, int W,
                  int kh, int kw, int ph, int pw, int sh, int sw, int OH,
                  int OW, float* dx);

// Fused BatchNorm (replaces the reference's ~10-launch GEMV chain,
// batch_norm_layer.hpp:96-120, with 3 kernels fwd / 3 bwd).
// partials: double2[nb*C] workspace. mean/var/inv_std: float[C] device.
int bn_blocks_per_channel(int N, int C);
void bn_fwd_stats(hipStream_t s, const float* x, int N, int C, long S,
                  int nb, void* partials);
void bn_fwd_finalize(hipStream_t s, const void* partials, int nb, int C,
                     long NS, float eps, float* mean, float* var,
                     float* inv_std);
void bn_fwd_norm(hipStream_t s, const float* x, const float* mean,
                 const float* inv_std, const float* scale, const float* bias,
                 int scale_bias, int N, int C, long S, float* y,
                 int fuse_relu = 0, const float* add = nullptr);
void bn_moving_avg(hipStream_t s, const float* mean, const float* var, int C,
                   float maf, int copy_only, float* gmean, float* gvar);
void bn_fwd_test(hipStream_t s, const float* x, const float* gmean,
                 const float* gvar, const float* scale, const float* bias,
                 int scale_bias, int N, int C, long S, float eps, float* y);
void bn_bwd_stats(hipStream_t s, const float* x, const float* dy,
                  const float* mean, const float* inv_std, int N, int C,
                  long S, int nb, const float* scale, const float* bias,
                  int fuse_relu, void* partials);
// writes dscale/dbias (if scale_bias) and the per-channel m_dy/m_dyxn terms
void bn_bwd_finalize(hipStream_t s, const void* partials, int nb, int C,
                     long NS, const float* scale, int scale_bias,
                     float* dscale, float* dbias, float* m_dy,
                     float* m_dyxn);
void bn_bwd_apply(hipStream_t s, const float* x, const float* dy,
                  const float* mean, const float* inv_std,
                  const float* scale, int scale_bias, const float* m_dy,
                  const float* m_dyxn, int N, int C, long S,
                  const float* bias, int fuse_relu, float* dx);

void lrn_fwd(hipStream_t s, const float* x, int N, int C, int H, int W,
             int size, float alpha, float beta, float k, float* scale,
             float* y);
void lrn_bwd(hipStream_t s, const float* x, const float* y, const float* dy,
             const float* scale, int N, int C, int H, int W, int size,
             float alpha, float beta, float* dx);

void softmax_fwd(hipStream_t s, const float* x, int outer, int C, int inner,
                 float* prob);
// loss_out: device float[1]; norm = outer*inner (VALID, no ignore label)
void softmaxloss_fwd(hipStream_t s, const float* prob, const float* label,
                     int outer, int C, int inner, float* loss_out);
void softmaxloss_bwd(hipStream_t s, const float* prob, const float* label,
                     int outer, int C, int inner, float scale, float* dx);

// out[n] = Σ_m A[m][n] — IP bias grad (deterministic)
void colsum(hipStream_t s, const float* A, long M, long N, float* out);

void axpby(hipStream_t s, long n, float a, const float* x, float b, float* y);
void copy(hipStream_t s, long n, const float* x, float* y);
void set_const(hipStream_t s, long n, float v, float* y);
void add3(hipStream_t s, long n, const float* a, const float* b, float* y,
          int fuse_relu = 0);
// y[n0..] accumulate: y += x
void acc(hipStream_t s, long n, const float* x, float* y);

// channel-block copy for Concat: src[N][Cs][S] <-> dst[N][Cd][S] at offset
void concat_fwd(hipStream_t s, const float* x, int N, int Cs, long S,
                int Cd, int c_off, float* y);
void concat_bwd(hipStream_t s, const float* dy, int N, int Cs, long S,
                int Cd, int c_off, float* dx);

void dropout_fwd(hipStream_t s, const float* x, long n, uint64_t seed,
                 uint64_t counter, float threshold, float scale, float* y,
                 uint8_t* mask);
void dropout_bwd(hipStream_t s, const float* dy, const uint8_t* mask, long n,
                 float scale, float* dx);

// fused SGD update (reference sgd_solver.cu:10-20 + the 1/nranks scale of
// net.cpp:910): g = g*gscale + decay*w; h = mom*h + lr*g; w -= h; g = 0.
void sgd_update(hipStream_t s, long n, float* g, float* w, float* h,
                float mom, float lr, float decay, float gscale);

// segmented fused SGD over a flat arena range (one launch per bucket);
// lrs/decays already folded with the per-param multipliers by the caller
void sgd_update_segmented(hipStream_t s, long lo, long hi, float* g_arena,
                          float* h_arena, const long* seg_off,
                          float* const* w_ptrs, const float* lr_mults,
                          const float* decay_mults, int nseg, float mom,
                          float lr, float decay, float gscale);

// LMDB batch transform: uint8 -> fp32 crop/mirror/mean/scale
// (data_transformer.cu:14-100 analog); geo = per-image (ho, wo, mirror),
// mean_mode 0 = none, 1 = per-channel, 2 = per-pixel (mean_file)
void transform_u8(hipStream_t s, const uint8_t* in, int N, int C, int inH,
                  int inW, int outH, int outW, const int* geo,
                  const float* mean, int mean_mode, float scale,
                  float* out);

void fill_uniform(hipStream_t s, long n, uint64_t seed, uint64_t counter,
                  float lo, float hi, float* y);
void fill_labels(hipStream_t s, long n, uint64_t seed, uint64_t counter,
                 int classes, float* y);

}  // namespace gpu
}  // namespace camd
