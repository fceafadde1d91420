// gemm_common.hpp — shared pieces of the MFMA GEMM family (fp32 + bf16):
// the kernel argument block and the 16-element operand reader that
// understands plain, channel-view and implicit-im2col-view addressing
// (strides included).  See gemm_f32.hip for the design notes.
#pragma once

#include <hip/hip_runtime.h>

#include "../layers.hpp"

namespace camd {
namespace gpu {

using f32x16 = __attribute__((ext_vector_type(16))) float;
using f32x4 = __attribute__((ext_vector_type(4))) float;

struct GemmArgs {
  const float* A;
  const float* B;
  float* C;
  long M, N, K;
  long lda, ldb, ldc;
  float alpha, beta;
  // epilogue
  long spad, S, n_stride;  // spad>0 => conv NCHW scatter
  const float* bias;
  int bias_per_col;
  int relu;
  long Srow;               // dest per-row stride (== S unless strided)
  int OWo, osh, osw, Wd;   // strided scatter (OWo>0)
  // operand views (spad==0 => plain)
  GemmView av, bv;
  // split-K
  float* slab;  // partials [SK][M][N] when SK>1
  int SK;
  // swizzle
  long tn, tiles;  // column-tile count, total tiles
};

// Read 16 consecutive elements along the contiguous axis.
//   plain:  base = P + r*ld + q,      valid j while q + j < qmax, r < rmax
//   view:   n = q / spad, sp = q % spad,
//           base = P + (n*chan + r)*S + sp, valid j while sp + j < S
// (q is 16-aligned and spad % 16 == 0, so a chunk never crosses images)
__device__ __forceinline__ void read16(const float* __restrict__ P, long r,
                                       long q, long ld, long rmax,
                                       long qmax, const GemmView& v,
                                       float (&out)[16]) {
  const float* p;
  long nvalid;  // elements valid from j=0
  if (v.spad && v.kh > 0) {
    // implicit im2col (d1, any stride): r = (c, ki, kj), sp = (oh, ow)
    const long n = q / v.spad;
    const long sp = q - n * v.spad;
    if (r >= rmax || sp >= v.S || q >= qmax) {
#pragma unroll
      for (int j = 0; j < 16; ++j) out[j] = 0.f;
      return;
    }
    const int c = (int)(r / (v.kh * v.kw));
    const int krem = (int)(r - (long)c * v.kh * v.kw);
    const int ki = krem / v.kw, kj = krem - (krem / v.kw) * v.kw;
    int oh = (int)(sp / v.OW);
    int ow = (int)(sp - (long)oh * v.OW);
    int h = oh * v.sh - v.ph + ki;
    int w = ow * v.sw - v.pw + kj;
    const float* xp = P + (n * v.chan + c) * (long)v.H * v.W;
    const long smax = v.S - sp;  // elements left in this image
    // fast path: chunk stays in one output row, fully interior
    if (smax >= 16 && ow + 16 <= v.OW && h >= 0 && h < v.H && w >= 0 &&
        w + 15 * v.sw < v.W) {
      if (v.sw == 1) {
        const float* p = xp + h * v.W + w;
        if ((((uintptr_t)p) & 15) == 0) {
          const f32x4* p4 = (const f32x4*)p;
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            const f32x4 t = p4[j];
            out[4 * j + 0] = t.x;
            out[4 * j + 1] = t.y;
            out[4 * j + 2] = t.z;
            out[4 * j + 3] = t.w;
          }
        } else {
#pragma unroll
          for (int j = 0; j < 16; ++j) out[j] = p[j];
        }
      } else {  // strided row gather: 16 independent loads
        const float* p = xp + h * v.W + w;
#pragma unroll
        for (int j = 0; j < 16; ++j) out[j] = p[j * v.sw];
      }
      return;
    }
    if (v.OW >= 16) {
      // at most ONE row wrap inside the 16-chunk: all 16 loads become
      // independent (the sequential walk would chain their addresses)
      const int jw = v.OW - ow;       // first j on the next output row
      const int w0 = w;               // input col at j=0
      const int wreset = -v.pw + kj;  // input col after the wrap
#pragma unroll
      for (int j = 0; j < 16; ++j) {
        const bool wrapped = j >= jw;
        const int hj = h + (wrapped ? v.sh : 0);
        const int wj = (wrapped ? wreset + (j - jw) * v.sw
                                : w0 + j * v.sw);
        const bool ok =
            j < smax && hj >= 0 && hj < v.H && wj >= 0 && wj < v.W;
        out[j] = ok ? xp[hj * v.W + wj] : 0.f;
      }
      return;
    }
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      const bool ok = j < smax && h >= 0 && h < v.H && w >= 0 && w < v.W;
      out[j] = ok ? xp[h * v.W + w] : 0.f;
      // next output pixel: ow+1 (input w += sw), wrapping to the next row
      if (++ow == v.OW) {
        ow = 0;
        ++oh;
        h += v.sh;
        w = -v.pw + kj;
      } else {
        w += v.sw;
      }
    }
    return;
  }
  if (v.spad) {
    const long n = q / v.spad;
    const long sp = q - n * v.spad;
    p = P + (n * v.chan + r) * v.S + sp;
    nvalid = (r < rmax && sp < v.S && q < qmax) ? v.S - sp : 0;
  } else {
    p = P + r * ld + q;
    nvalid = (r < rmax && q < qmax) ? qmax - q : 0;
  }
  if (nvalid >= 16 && (((uintptr_t)p) & 15) == 0) {
    const f32x4* p4 = (const f32x4*)p;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const f32x4 t = p4[j];
      out[4 * j + 0] = t.x;
      out[4 * j + 1] = t.y;
      out[4 * j + 2] = t.z;
      out[4 * j + 3] = t.w;
    }
  } else {
#pragma unroll
    for (int j = 0; j < 16; ++j) out[j] = j < nvalid ? p[j] : 0.f;
  }
}

}  // namespace gpu
}  // namespace camd
