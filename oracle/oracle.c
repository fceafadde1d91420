/* oracle.c — CPU restatement of the Caffe-MPI hot-path algorithms.
 *
 * TEST INFRASTRUCTURE ONLY.  Per the build contract, only tests/,
 * __graft_entry__.smoke() and bench.py's cpu_baseline leg may call into this
 * library.  The product (caffe-mpi.github.io_amd) never links or loads it;
 * the HIP engine fails loudly if its own extension is missing.
 *
 * Every function restates, function-for-function, the reference CPU
 * semantics (citations are file:line into /root/reference):
 *   im2col/col2im            src/caffe/util/im2col.cpp (im2col_cpu, col2im_cpu)
 *   conv fwd/bwd             src/caffe/layers/conv_layer.cpp:24-77 +
 *                            include/caffe/layers/base_conv_layer.hpp:36-100
 *   pooling                  src/caffe/layers/pooling_layer.cpp:86-320
 *   batchnorm                src/caffe/layers/batch_norm_layer.cpp:143-293 +
 *                            include/caffe/layers/batch_norm_layer.hpp:85-130
 *   relu                     src/caffe/layers/relu_layer.cpp
 *   inner product            src/caffe/layers/inner_product_layer.cpp
 *   lrn (ACROSS_CHANNELS)    src/caffe/layers/lrn_layer.cpp:108-233
 *   softmax                  src/caffe/layers/softmax_layer.cpp:26-60
 *   softmax loss             src/caffe/layers/softmax_loss_layer.cpp:94-160
 *   eltwise SUM              src/caffe/layers/eltwise_layer.cpp
 *   dropout                  src/caffe/layers/dropout_layer.cpp:30-65
 *   sgd update               src/caffe/solvers/sgd_solver.cpp:143-252
 *                            (fused form: sgd_solver.cu:10-20)
 *   accuracy                 src/caffe/layers/accuracy_layer.cpp
 *
 * Reductions accumulate in double so the oracle is strictly more accurate
 * than any fp32 summation order; parity tests compare at 1e-4 relative fp32
 * (the reference's own EXPECT_NEAR class, test_convolution_layer.cpp:247).
 *
 * Build: gcc -O2 -fopenmp -shared -fPIC oracle.c -o liboracle.so -lm
 */
#include <float.h>
#include <math.h>
#include <stdint.h>
#include <string.h>

#define EXPORT __attribute__((visibility("default")))

/* ----------------------------------------------------------------------- */
/* im2col / col2im — reference util/im2col.cpp semantics (zero padding,     */
/* dilation, stride); col is [C*kh*kw][OH*OW] row-major for ONE image.      */

static inline int out_dim(int in, int k, int pad, int stride, int dil) {
  int kext = dil * (k - 1) + 1; /* conv_layer.cpp compute_output_shape */
  return (in + 2 * pad - kext) / stride + 1;
}

EXPORT int orc_conv_out_dim(int in, int k, int pad, int stride, int dil) {
  return out_dim(in, k, pad, stride, dil);
}

EXPORT void orc_im2col(const float *im, int C, int H, int W, int kh, int kw,
                       int ph, int pw, int sh, int sw, int dh, int dw,
                       float *col) {
  const int OH = out_dim(H, kh, ph, sh, dh);
  const int OW = out_dim(W, kw, pw, sw, dw);
#pragma omp parallel for collapse(2) schedule(static)
  for (int c = 0; c < C; ++c) {
    for (int ki = 0; ki < kh * kw; ++ki) {
      const int i = ki / kw, j = ki % kw;
      float *dst = col + ((long)(c * kh * kw + ki)) * OH * OW;
      const float *src = im + (long)c * H * W;
      for (int oh = 0; oh < OH; ++oh) {
        const int h = oh * sh - ph + i * dh;
        for (int ow = 0; ow < OW; ++ow) {
          const int w = ow * sw - pw + j * dw;
          dst[oh * OW + ow] =
              (h >= 0 && h < H && w >= 0 && w < W) ? src[h * W + w] : 0.f;
        }
      }
    }
  }
}

EXPORT void orc_col2im(const float *col, int C, int H, int W, int kh, int kw,
                       int ph, int pw, int sh, int sw, int dh, int dw,
                       float *im) {
  const int OH = out_dim(H, kh, ph, sh, dh);
  const int OW = out_dim(W, kw, pw, sw, dw);
  /* one thread per input element, gather (reference col2im_cpu via its GPU
   * twin im2col.cu:256-295 — same math, no atomics) */
#pragma omp parallel for schedule(static)
  for (int c = 0; c < C; ++c) {
    for (int h = 0; h < H; ++h) {
      for (int w = 0; w < W; ++w) {
        double acc = 0.0;
        for (int i = 0; i < kh; ++i) {
          int hk = h + ph - i * dh;
          if (hk < 0 || hk % sh) continue;
          hk /= sh;
          if (hk >= OH) continue;
          for (int j = 0; j < kw; ++j) {
            int wk = w + pw - j * dw;
            if (wk < 0 || wk % sw) continue;
            wk /= sw;
            if (wk >= OW) continue;
            acc += col[(((long)(c * kh + i) * kw + j) * OH + hk) * OW + wk];
          }
        }
        im[((long)c * H + h) * W + w] = (float)acc;
      }
    }
  }
}

/* ----------------------------------------------------------------------- */
/* GEMM, row-major: C = alpha * op(A)[MxK] * op(B)[KxN] + beta * C[MxN].   */
/* Double accumulation; OpenMP over rows. transX: 0 = N, 1 = T.            */

EXPORT void orc_gemm(int transA, int transB, int M, int N, int K, float alpha,
                     const float *A, const float *B, float beta, float *C) {
  const long lda = transA ? M : K; /* leading dim of stored A */
  const long ldb = transB ? K : N;
  /* collapse(2): conv GEMMs have M as small as 64 (stage-2 Cout) — row-only
   * parallelism starves a 256-thread host.  Each output element keeps its
   * serial k-order, so results are bit-identical to the row-parallel form. */
#pragma omp parallel for collapse(2) schedule(static)
  for (int m = 0; m < M; ++m) {
    for (int n = 0; n < N; ++n) {
      double acc = 0.0;
      if (!transA && !transB) {
        const float *a = A + (long)m * lda;
        for (int k = 0; k < K; ++k) acc += (double)a[k] * B[(long)k * ldb + n];
      } else if (!transA && transB) {
        const float *a = A + (long)m * lda;
        const float *b = B + (long)n * ldb;
        for (int k = 0; k < K; ++k) acc += (double)a[k] * b[k];
      } else if (transA && !transB) {
        for (int k = 0; k < K; ++k)
          acc += (double)A[(long)k * lda + m] * B[(long)k * ldb + n];
      } else {
        for (int k = 0; k < K; ++k)
          acc += (double)A[(long)k * lda + m] * B[(long)n * ldb + k];
      }
      float *c = C + (long)m * N + n;
      *c = (float)(alpha * acc + (beta ? (double)beta * *c : 0.0));
    }
  }
}

/* ----------------------------------------------------------------------- */
/* Convolution via im2col+GEMM (base_conv_layer.hpp:36-100 forward_cpu_gemm */
/* / backward_cpu_gemm / weight_cpu_gemm semantics, incl. groups).          */
/* Layouts: x[N][C][H][W]; w[Cout][Cin/g][kh][kw]; y[N][Cout][OH][OW].      */

EXPORT void orc_conv_fwd(const float *x, const float *w, const float *bias,
                         int has_bias, int N, int C, int H, int W, int Cout,
                         int kh, int kw, int ph, int pw, int sh, int sw,
                         int dh, int dw, int group, float *y, float *colbuf) {
  const int OH = out_dim(H, kh, ph, sh, dh);
  const int OW = out_dim(W, kw, pw, sw, dw);
  const long S = (long)OH * OW;
  const int K = C / group * kh * kw; /* rows of col per group */
  for (int n = 0; n < N; ++n) {
    orc_im2col(x + (long)n * C * H * W, C, H, W, kh, kw, ph, pw, sh, sw, dh,
               dw, colbuf);
    for (int g = 0; g < group; ++g) {
      orc_gemm(0, 0, Cout / group, (int)S, K, 1.f,
               w + (long)g * (Cout / group) * K, colbuf + (long)g * K * S,
               0.f, y + ((long)n * Cout + (long)g * (Cout / group)) * S);
    }
    if (has_bias) {
      float *yn = y + (long)n * Cout * S;
#pragma omp parallel for schedule(static)
      for (int co = 0; co < Cout; ++co)
        for (long s = 0; s < S; ++s) yn[co * S + s] += bias[co];
    }
  }
}

EXPORT void orc_conv_bwd(const float *x, const float *w, const float *dy,
                         int N, int C, int H, int W, int Cout, int kh, int kw,
                         int ph, int pw, int sh, int sw, int dlh, int dlw,
                         int group, int want_dx, int want_dw, int want_db,
                         float *dx, float *dw, float *db, float *colbuf,
                         float *colbuf2) {
  const int OH = out_dim(H, kh, ph, sh, dlh);
  const int OW = out_dim(W, kw, pw, sw, dlw);
  const long S = (long)OH * OW;
  const int K = C / group * kh * kw;
  if (want_dw) memset(dw, 0, sizeof(float) * (long)Cout * K);
  if (want_db) memset(db, 0, sizeof(float) * Cout);
  for (int n = 0; n < N; ++n) {
    const float *dyn = dy + (long)n * Cout * S;
    if (want_db) { /* backward_cpu_bias: db += dy · 1 */
#pragma omp parallel for schedule(static)
      for (int co = 0; co < Cout; ++co) {
        double acc = 0.0;
        for (long s = 0; s < S; ++s) acc += dyn[co * S + s];
        db[co] += (float)acc;
      }
    }
    if (want_dw) { /* weight_cpu_gemm: dW += dY · colᵀ (accumulate over n) */
      orc_im2col(x + (long)n * C * H * W, C, H, W, kh, kw, ph, pw, sh, sw,
                 dlh, dlw, colbuf);
    }
    if (want_dw) {
      for (int g = 0; g < group; ++g)
        orc_gemm(0, 1, Cout / group, K, (int)S, 1.f,
                 dyn + (long)g * (Cout / group) * S, colbuf + (long)g * K * S,
                 1.f, dw + (long)g * (Cout / group) * K);
    }
    if (want_dx) { /* backward_cpu_gemm: dcol = Wᵀ · dY ; dx = col2im */
      for (int g = 0; g < group; ++g)
        orc_gemm(1, 0, K, (int)S, Cout / group, 1.f,
                 w + (long)g * (Cout / group) * K,
                 dyn + (long)g * (Cout / group) * S, 0.f,
                 colbuf2 + (long)g * K * S);
      orc_col2im(colbuf2, C, H, W, kh, kw, ph, pw, sh, sw, dlh, dlw,
                 dx + (long)n * C * H * W);
    }
  }
}

/* ----------------------------------------------------------------------- */
/* Pooling — pooling_layer.cpp. Output dims use ceil + clip (:86-107).     */

EXPORT void orc_pool_out_dim(int H, int W, int kh, int kw, int ph, int pw,
                             int sh, int sw, int *OH, int *OW) {
  int oh = (int)ceilf((float)(H + 2 * ph - kh) / sh) + 1;
  int ow = (int)ceilf((float)(W + 2 * pw - kw) / sw) + 1;
  if (ph || pw) {
    if ((oh - 1) * sh >= H + ph) --oh;
    if ((ow - 1) * sw >= W + pw) --ow;
  }
  *OH = oh;
  *OW = ow;
}

EXPORT void orc_pool_max_fwd(const float *x, int N, int C, int H, int W,
                             int kh, int kw, int ph, int pw, int sh, int sw,
                             float *y, int *mask) {
  int OH, OW;
  orc_pool_out_dim(H, W, kh, kw, ph, pw, sh, sw, &OH, &OW);
#pragma omp parallel for collapse(2) schedule(static)
  for (int n = 0; n < N; ++n) {
    for (int c = 0; c < C; ++c) {
      const float *xp = x + ((long)n * C + c) * H * W;
      float *yp = y + ((long)n * C + c) * OH * OW;
      int *mp = mask + ((long)n * C + c) * OH * OW;
      for (int oh = 0; oh < OH; ++oh) {
        for (int ow = 0; ow < OW; ++ow) {
          int hs = oh * sh - ph, ws = ow * sw - pw;
          int he = hs + kh < H ? hs + kh : H;
          int we = ws + kw < W ? ws + kw : W;
          if (hs < 0) hs = 0;
          if (ws < 0) ws = 0;
          float best = -FLT_MAX;
          int bidx = -1;
          for (int h = hs; h < he; ++h)
            for (int w = ws; w < we; ++w) {
              const int idx = h * W + w;
              if (xp[idx] > best) { /* strict >, first max wins (:166) */
                best = xp[idx];
                bidx = idx;
              }
            }
          yp[oh * OW + ow] = best;
          mp[oh * OW + ow] = bidx;
        }
      }
    }
  }
}

EXPORT void orc_pool_max_bwd(const float *dy, const int *mask, int N, int C,
                             int H, int W, int OH, int OW, float *dx) {
  memset(dx, 0, sizeof(float) * (long)N * C * H * W);
  for (long nc = 0; nc < (long)N * C; ++nc) {
    const float *dyp = dy + nc * OH * OW;
    const int *mp = mask + nc * OH * OW;
    float *dxp = dx + nc * H * W;
    for (long i = 0; i < (long)OH * OW; ++i) dxp[mp[i]] += dyp[i];
  }
}

EXPORT void orc_pool_ave_fwd(const float *x, int N, int C, int H, int W,
                             int kh, int kw, int ph, int pw, int sh, int sw,
                             float *y) {
  int OH, OW;
  orc_pool_out_dim(H, W, kh, kw, ph, pw, sh, sw, &OH, &OW);
#pragma omp parallel for collapse(2) schedule(static)
  for (int n = 0; n < N; ++n) {
    for (int c = 0; c < C; ++c) {
      const float *xp = x + ((long)n * C + c) * H * W;
      float *yp = y + ((long)n * C + c) * OH * OW;
      for (int oh = 0; oh < OH; ++oh) {
        for (int ow = 0; ow < OW; ++ow) {
          int hs = oh * sh - ph, ws = ow * sw - pw;
          int he = hs + kh < H + ph ? hs + kh : H + ph;
          int we = ws + kw < W + pw ? ws + kw : W + pw;
          const int pool_size = (he - hs) * (we - ws); /* padded size :201 */
          if (hs < 0) hs = 0;
          if (ws < 0) ws = 0;
          if (he > H) he = H;
          if (we > W) we = W;
          double acc = 0.0;
          for (int h = hs; h < he; ++h)
            for (int w = ws; w < we; ++w) acc += xp[h * W + w];
          yp[oh * OW + ow] = (float)(acc / pool_size);
        }
      }
    }
  }
}

EXPORT void orc_pool_ave_bwd(const float *dy, int N, int C, int H, int W,
                             int kh, int kw, int ph, int pw, int sh, int sw,
                             float *dx) {
  int OH, OW;
  orc_pool_out_dim(H, W, kh, kw, ph, pw, sh, sw, &OH, &OW);
  memset(dx, 0, sizeof(float) * (long)N * C * H * W);
#pragma omp parallel for collapse(2) schedule(static)
  for (int n = 0; n < N; ++n) {
    for (int c = 0; c < C; ++c) {
      const float *dyp = dy + ((long)n * C + c) * OH * OW;
      float *dxp = dx + ((long)n * C + c) * H * W;
      for (int oh = 0; oh < OH; ++oh) {
        for (int ow = 0; ow < OW; ++ow) {
          int hs = oh * sh - ph, ws = ow * sw - pw;
          int he = hs + kh < H + ph ? hs + kh : H + ph;
          int we = ws + kw < W + pw ? ws + kw : W + pw;
          const int pool_size = (he - hs) * (we - ws);
          if (hs < 0) hs = 0;
          if (ws < 0) ws = 0;
          if (he > H) he = H;
          if (we > W) we = W;
          const float v = dyp[oh * OW + ow] / pool_size;
          for (int h = hs; h < he; ++h)
            for (int w = ws; w < we; ++w) dxp[h * W + w] += v;
        }
      }
    }
  }
}

/* ----------------------------------------------------------------------- */
/* BatchNorm — batch_norm_layer.cpp Forward_cpu/Backward_cpu.  S = H*W.    */
/* Variance is the biased E[(x-mean)^2]; inv_std = (var+eps)^-0.5; eps is  */
/* already clamped by the layer (max(eps,1e-5), :25).                      */

EXPORT void orc_bn_fwd_train(const float *x, int N, int C, int S, float eps,
                             const float *scale, const float *bias,
                             int scale_bias, float *mean, float *var,
                             float *inv_std, float *xnorm, float *y) {
#pragma omp parallel for schedule(static)
  for (int c = 0; c < C; ++c) {
    double acc = 0.0;
    for (int n = 0; n < N; ++n) {
      const float *xp = x + ((long)n * C + c) * S;
      for (int s = 0; s < S; ++s) acc += xp[s];
    }
    const double m = acc / ((double)N * S);
    mean[c] = (float)m;
    double vacc = 0.0;
    for (int n = 0; n < N; ++n) {
      const float *xp = x + ((long)n * C + c) * S;
      for (int s = 0; s < S; ++s) {
        /* reference computes y=x-mean in fp32, then squares (:180-188) */
        const float d = xp[s] - mean[c];
        vacc += (double)d * d;
      }
    }
    var[c] = (float)(vacc / ((double)N * S));
    inv_std[c] = (float)(1.0 / sqrt((double)var[c] + eps));
    for (int n = 0; n < N; ++n) {
      const float *xp = x + ((long)n * C + c) * S;
      float *xn = xnorm + ((long)n * C + c) * S;
      float *yp = y + ((long)n * C + c) * S;
      for (int s = 0; s < S; ++s) {
        const float v = (xp[s] - mean[c]) * inv_std[c];
        xn[s] = v;
        yp[s] = scale_bias ? v * scale[c] + bias[c] : v;
      }
    }
  }
}

EXPORT void orc_bn_fwd_test(const float *x, int N, int C, int S, float eps,
                            const float *gmean, const float *gvar,
                            const float *scale, const float *bias,
                            int scale_bias, float *y) {
#pragma omp parallel for schedule(static)
  for (int c = 0; c < C; ++c) {
    const float inv = (float)(1.0 / sqrt((double)gvar[c] + eps));
    for (int n = 0; n < N; ++n) {
      const float *xp = x + ((long)n * C + c) * S;
      float *yp = y + ((long)n * C + c) * S;
      for (int s = 0; s < S; ++s) {
        const float v = (xp[s] - gmean[c]) * inv;
        yp[s] = scale_bias ? v * scale[c] + bias[c] : v;
      }
    }
  }
}

/* dx = (dy' - mean(dy') - mean(dy'.*xn).*xn) * inv_std, dy' = dy*scale
 * (batch_norm_layer.cpp:311-293); dscale = sum(dy.*xn), dbias = sum(dy). */
EXPORT void orc_bn_bwd(const float *xnorm, const float *dy,
                       const float *inv_std, const float *scale,
                       int scale_bias, int N, int C, int S, float *dx,
                       float *dscale, float *dbias) {
#pragma omp parallel for schedule(static)
  for (int c = 0; c < C; ++c) {
    double dsc = 0.0, dbi = 0.0;
    for (int n = 0; n < N; ++n) {
      const float *dyp = dy + ((long)n * C + c) * S;
      const float *xn = xnorm + ((long)n * C + c) * S;
      for (int s = 0; s < S; ++s) {
        dsc += (double)dyp[s] * xn[s];
        dbi += (double)dyp[s];
      }
    }
    if (scale_bias) {
      dscale[c] = (float)dsc;
      dbias[c] = (float)dbi;
    }
    const float sc = scale_bias ? scale[c] : 1.f;
    /* means of dy' and dy'.*xn over N*S */
    double m_dy = 0.0, m_dyxn = 0.0;
    for (int n = 0; n < N; ++n) {
      const float *dyp = dy + ((long)n * C + c) * S;
      const float *xn = xnorm + ((long)n * C + c) * S;
      for (int s = 0; s < S; ++s) {
        const double d = (double)dyp[s] * sc;
        m_dy += d;
        m_dyxn += d * xn[s];
      }
    }
    m_dy /= (double)N * S;
    m_dyxn /= (double)N * S;
    for (int n = 0; n < N; ++n) {
      const float *dyp = dy + ((long)n * C + c) * S;
      const float *xn = xnorm + ((long)n * C + c) * S;
      float *dxp = dx + ((long)n * C + c) * S;
      for (int s = 0; s < S; ++s) {
        const double d = (double)dyp[s] * sc;
        dxp[s] = (float)((d - m_dy - m_dyxn * xn[s]) * inv_std[c]);
      }
    }
  }
}

/* ----------------------------------------------------------------------- */
EXPORT void orc_relu_fwd(const float *x, long n, float slope, float *y) {
#pragma omp parallel for schedule(static)
  for (long i = 0; i < n; ++i)
    y[i] = x[i] > 0 ? x[i] : slope * x[i];
}

EXPORT void orc_relu_bwd(const float *x, const float *dy, long n, float slope,
                         float *dx) {
#pragma omp parallel for schedule(static)
  for (long i = 0; i < n; ++i)
    dx[i] = dy[i] * ((x[i] > 0) + slope * (x[i] <= 0));
}

/* ----------------------------------------------------------------------- */
/* LRN ACROSS_CHANNELS — lrn_layer.cpp:108-233.                            */
/* scale = k + (alpha/size) * sliding channel sum of x^2; y = x*scale^-b.  */

EXPORT void orc_lrn_fwd(const float *x, int N, int C, int H, int W, int size,
                        float alpha, float beta, float k, float *scale,
                        float *y) {
  const long S = (long)H * W;
  const int pre = (size - 1) / 2;
  const float aos = alpha / size;
#pragma omp parallel for schedule(static)
  for (int n = 0; n < N; ++n) {
    const float *xn = x + (long)n * C * S;
    float *sn = scale + (long)n * C * S;
    for (long s = 0; s < S; ++s) {
      for (int c = 0; c < C; ++c) {
        double acc = 0.0;
        for (int cc = c - pre; cc <= c - pre + size - 1; ++cc)
          if (cc >= 0 && cc < C) {
            const double v = xn[cc * S + s];
            acc += v * v;
          }
        sn[c * S + s] = (float)(k + aos * acc);
      }
    }
  }
#pragma omp parallel for schedule(static)
  for (long i = 0; i < (long)N * C * S; ++i)
    y[i] = x[i] * powf(scale[i], -beta);
}

/* bottom_diff = dy*scale^-beta − (2αβ/size)·x·Σ_window(dy.*y./scale)      */
EXPORT void orc_lrn_bwd(const float *x, const float *y, const float *dy,
                        const float *scale, int N, int C, int H, int W,
                        int size, float alpha, float beta, float *dx) {
  const long S = (long)H * W;
  const int pre = (size - 1) / 2;
  const float cr = 2.f * alpha * beta / size;
#pragma omp parallel for schedule(static)
  for (int n = 0; n < N; ++n) {
    const float *xn = x + (long)n * C * S;
    const float *yn = y + (long)n * C * S;
    const float *dyn = dy + (long)n * C * S;
    const float *sn = scale + (long)n * C * S;
    float *dxn = dx + (long)n * C * S;
    for (long s = 0; s < S; ++s) {
      for (int c = 0; c < C; ++c) {
        double acc = 0.0;
        /* window of channels cc whose scale window includes c:
         * lrn_layer.cpp:200-231 accumulates ratio over [c-pre, c+size-1-pre]
         * mirrored — the window is cc in [c-(size-1-pre), c+pre] */
        for (int cc = c - (size - 1 - pre); cc <= c + pre; ++cc)
          if (cc >= 0 && cc < C) {
            const long i = cc * S + s;
            acc += (double)dyn[i] * yn[i] / sn[i];
          }
        const long i = c * S + s;
        dxn[i] =
            (float)(dyn[i] * powf(sn[i], -beta) - cr * xn[i] * acc);
      }
    }
  }
}

/* ----------------------------------------------------------------------- */
/* Softmax over channel axis — softmax_layer.cpp:26-60.                    */
EXPORT void orc_softmax_fwd(const float *x, int outer, int C, int inner,
                            float *y) {
#pragma omp parallel for collapse(2) schedule(static)
  for (int o = 0; o < outer; ++o) {
    for (int s = 0; s < inner; ++s) {
      const float *xp = x + (long)o * C * inner + s;
      float *yp = y + (long)o * C * inner + s;
      float mx = xp[0];
      for (int c = 1; c < C; ++c)
        if (xp[(long)c * inner] > mx) mx = xp[(long)c * inner];
      double sum = 0.0;
      for (int c = 0; c < C; ++c) {
        const float e = expf(xp[(long)c * inner] - mx);
        yp[(long)c * inner] = e;
        sum += e;
      }
      for (int c = 0; c < C; ++c)
        yp[(long)c * inner] = (float)(yp[(long)c * inner] / sum);
    }
  }
}

/* SoftmaxWithLoss forward: mean over VALID of -log(max(p,FLT_MIN))
 * (softmax_loss_layer.cpp:94-125; normalization VALID default).          */
EXPORT float orc_softmaxloss_fwd(const float *prob, const float *label,
                                 int outer, int C, int inner,
                                 int has_ignore, int ignore_label) {
  double loss = 0.0;
  long count = 0;
  for (int o = 0; o < outer; ++o)
    for (int s = 0; s < inner; ++s) {
      const int lv = (int)label[(long)o * inner + s];
      if (has_ignore && lv == ignore_label) continue;
      const float p = prob[((long)o * C + lv) * inner + s];
      loss -= log(p > FLT_MIN ? p : FLT_MIN);
      ++count;
    }
  const double norm = count > 0 ? (double)count : 1.0;
  return (float)(loss / norm);
}

/* Backward: dx = (prob − 1{label}) * loss_weight / normalizer             */
EXPORT void orc_softmaxloss_bwd(const float *prob, const float *label,
                                int outer, int C, int inner, int has_ignore,
                                int ignore_label, float loss_weight,
                                float *dx) {
  long count = 0;
  for (int o = 0; o < outer; ++o)
    for (int s = 0; s < inner; ++s) {
      const int lv = (int)label[(long)o * inner + s];
      if (!(has_ignore && lv == ignore_label)) ++count;
    }
  const float w = loss_weight / (count > 0 ? (float)count : 1.f);
  memcpy(dx, prob, sizeof(float) * (long)outer * C * inner);
  for (int o = 0; o < outer; ++o)
    for (int s = 0; s < inner; ++s) {
      const int lv = (int)label[(long)o * inner + s];
      if (has_ignore && lv == ignore_label) {
        for (int c = 0; c < C; ++c) dx[((long)o * C + c) * inner + s] = 0.f;
      } else {
        dx[((long)o * C + lv) * inner + s] -= 1.f;
      }
    }
  for (long i = 0; i < (long)outer * C * inner; ++i) dx[i] *= w;
}

/* ----------------------------------------------------------------------- */
/* InnerProduct: y[M][Nout] = x[M][K] · W[Nout][K]ᵀ (+ b)                  */
EXPORT void orc_ip_fwd(const float *x, const float *w, const float *b,
                       int has_bias, int M, int Nout, int K, float *y) {
  orc_gemm(0, 1, M, Nout, K, 1.f, x, w, 0.f, y);
  if (has_bias)
    for (int m = 0; m < M; ++m)
      for (int n = 0; n < Nout; ++n) y[(long)m * Nout + n] += b[n];
}

EXPORT void orc_ip_bwd(const float *x, const float *w, const float *dy,
                       int M, int Nout, int K, int want_dx, float *dx,
                       float *dw, float *db, int has_bias) {
  /* dW = dyᵀ·x ; db = dyᵀ·1 ; dx = dy·W  (inner_product_layer.cpp) */
  orc_gemm(1, 0, Nout, K, M, 1.f, dy, x, 0.f, dw);
  if (has_bias)
    for (int n = 0; n < Nout; ++n) {
      double acc = 0.0;
      for (int m = 0; m < M; ++m) acc += dy[(long)m * Nout + n];
      db[n] = (float)acc;
    }
  if (want_dx) orc_gemm(0, 0, M, K, Nout, 1.f, dy, w, 0.f, dx);
}

/* ----------------------------------------------------------------------- */
EXPORT void orc_sgd_update(long n, float *g, float *w, float *h,
                           float momentum, float lr, float decay,
                           float grad_scale) {
  /* fused SGDRegUpdateAllAndClear (sgd_solver.cu:10-20): L2 reg + momentum
   * + update + clear.  grad_scale = 1/nranks applied before reg
   * (net.cpp:910 scales the all-reduced bucket before ApplyUpdate). */
  for (long i = 0; i < n; ++i) {
    float gi = g[i] * grad_scale + decay * w[i];
    gi = h[i] = momentum * h[i] + lr * gi;
    w[i] -= gi;
    g[i] = 0.f;
  }
}

EXPORT void orc_dropout_fwd(const float *x, const unsigned int *mask, long n,
                            float scale, float *y) {
  for (long i = 0; i < n; ++i) y[i] = x[i] * mask[i] * scale;
}

EXPORT float orc_accuracy(const float *pred, const float *label, int outer,
                          int C, int inner, int top_k) {
  long correct = 0, total = 0;
  for (int o = 0; o < outer; ++o)
    for (int s = 0; s < inner; ++s) {
      const int lv = (int)label[(long)o * inner + s];
      const float pv = pred[((long)o * C + lv) * inner + s];
      int rank = 0;
      for (int c = 0; c < C; ++c)
        if (pred[((long)o * C + c) * inner + s] > pv) ++rank;
      if (rank < top_k) ++correct;
      ++total;
    }
  return total ? (float)correct / total : 0.f;
}
