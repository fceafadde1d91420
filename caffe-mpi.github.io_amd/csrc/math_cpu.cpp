#include <cmath>

#include "math.hpp"

namespace camd {

void pool_out_dim(int H, int W, int kh, int kw, int ph, int pw, int sh,
                  int sw, int* OH, int* OW) {
  int oh = (int)std::ceil((float)(H + 2 * ph - kh) / sh) + 1;
  int ow = (int)std::ceil((float)(W + 2 * pw - kw) / sw) + 1;
  if (ph || pw) {  // clip: last pool must start inside image+pad
    if ((oh - 1) * sh >= H + ph) --oh;
    if ((ow - 1) * sw >= W + pw) --ow;
  }
  *OH = oh;
  *OW = ow;
}

namespace cpu {

void gemm(bool transA, bool transB, long M, long N, long K, float alpha,
          const float* A, const float* B, float beta, float* C) {
  const long lda = transA ? M : K;
  const long ldb = transB ? K : N;
#pragma omp parallel for schedule(static)
  for (long m = 0; m < M; ++m) {
    float* c = C + m * N;
    if (beta == 0.f) {
      for (long n = 0; n < N; ++n) c[n] = 0.f;
    } else if (beta != 1.f) {
      for (long n = 0; n < N; ++n) c[n] *= beta;
    }
    if (!transB) {
      // accumulate k-by-k over rows of B: C[m,:] += a * B[k,:]
      for (long k = 0; k < K; ++k) {
        const float a =
            alpha * (transA ? A[k * lda + m] : A[m * lda + k]);
        if (a == 0.f) continue;
        const float* b = B + k * ldb;
        for (long n = 0; n < N; ++n) c[n] += a * b[n];
      }
    } else {
      for (long n = 0; n < N; ++n) {
        const float* b = B + n * ldb;
        float acc = 0.f;
        if (!transA) {
          const float* a = A + m * lda;
          for (long k = 0; k < K; ++k) acc += a[k] * b[k];
        } else {
          for (long k = 0; k < K; ++k) acc += A[k * lda + m] * b[k];
        }
        c[n] += alpha * acc;
      }
    }
  }
}

void axpy(long n, float a, const float* x, float* y) {
#pragma omp parallel for schedule(static)
  for (long i = 0; i < n; ++i) y[i] += a * x[i];
}

void axpby(long n, float a, const float* x, float b, float* y) {
#pragma omp parallel for schedule(static)
  for (long i = 0; i < n; ++i) y[i] = a * x[i] + b * y[i];
}

void scal(long n, float a, float* x) {
#pragma omp parallel for schedule(static)
  for (long i = 0; i < n; ++i) x[i] *= a;
}

void im2col(const float* im, int C, int H, int W, int kh, int kw, int ph,
            int pw, int sh, int sw, int dh, int dw, float* col) {
  const int OH = conv_out_dim(H, kh, ph, sh, dh);
  const int OW = conv_out_dim(W, kw, pw, sw, dw);
#pragma omp parallel for collapse(2) schedule(static)
  for (int c = 0; c < C; ++c) {
    for (int ki = 0; ki < kh * kw; ++ki) {
      const int i = ki / kw, j = ki % kw;
      float* dst = col + ((long)c * kh * kw + ki) * OH * OW;
      const float* src = im + (long)c * H * W;
      for (int oh = 0; oh < OH; ++oh) {
        const int h = oh * sh - ph + i * dh;
        for (int ow = 0; ow < OW; ++ow) {
          const int w = ow * sw - pw + j * dw;
          dst[oh * OW + ow] =
              (h >= 0 && h < H && w >= 0 && w < W) ? src[(long)h * W + w]
                                                   : 0.f;
        }
      }
    }
  }
}

void col2im(const float* col, int C, int H, int W, int kh, int kw, int ph,
            int pw, int sh, int sw, int dh, int dw, float* im) {
  const int OH = conv_out_dim(H, kh, ph, sh, dh);
  const int OW = conv_out_dim(W, kw, pw, sw, dw);
#pragma omp parallel for schedule(static)
  for (int c = 0; c < C; ++c) {
    for (int h = 0; h < H; ++h) {
      for (int w = 0; w < W; ++w) {
        float acc = 0.f;
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
        im[((long)c * H + h) * W + w] = acc;
      }
    }
  }
}

}  // namespace cpu
}  // namespace camd
