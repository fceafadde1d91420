// elementwise.hip — gfx950 kernels for the HBM-bound hot-path ops:
// im2col/col2im (batched, [K][Nimg*Spad] col layout), pooling, fused
// BatchNorm (3 kernels/direction vs the reference's ~10-launch GEMV chain,
// batch_norm_layer.hpp:96-120), ReLU, LRN, dropout, softmax+NLL, fused SGD
// update (sgd_solver.cu:10-20 semantics), eltwise and synthetic-data fills.
//
// Design rules (cdna_hip_programming.md): grid-stride loops capped at
// 2048 blocks ×256 threads (G11), coalesced unit-stride innermost access,
// no atomics on the backward scatter paths (gather formulation instead —
// deterministic, matching the reference's no-atomics choice,
// conv_layer.cu:43-48).
#include <hip/hip_runtime.h>

#include "../math.hpp"

namespace camd {
namespace gpu {

constexpr int TPB = 256;
static inline int nblocks(long n, int per_thread = 1) {
  long b = (n + (long)TPB * per_thread - 1) / ((long)TPB * per_thread);
  return (int)std::min<long>(b, 2048);
}
#define GRID_STRIDE(i, n)                                        \
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < (n); \
       i += (long)gridDim.x * blockDim.x)

// ------------------------------------------------------------ im2col
// 3D grid: z = image, y = col row (c,ki,kj), x-threads sweep the output
// pixels — all inner indexing 32-bit, writes fully coalesced along sp.
__global__ void k_im2col_b(const float* __restrict__ x, int C, int H, int W,
                           int kh, int kw, int ph, int pw, int sh, int sw,
                           int dh, int dw, int OH, int OW, long Spad,
                           long cols, float* __restrict__ col) {
  const int n = blockIdx.z;
  const int row = blockIdx.y;  // c*kh*kw + ki*kw + kj
  const int c = row / (kh * kw);
  const int ki = (row / kw) % kh;
  const int kj = row % kw;
  const int S = OH * OW;
  const float* xp = x + ((long)n * C + c) * H * W;
  float* cp = col + (long)row * cols + (long)n * Spad;
  // 4-wide packs: the scalar form was ISSUE-bound (SQ: 51% instruction
  // stall — one div + bounds + load + store per element); an interior
  // s1 pack is one unaligned float4 read + one aligned float4 write
  using f4u = __attribute__((ext_vector_type(4), aligned(4))) float;
  const int Spad4 = (int)(Spad / 4);  // Spad % 16 == 0
  for (int p4 = blockIdx.x * blockDim.x + threadIdx.x; p4 < Spad4;
       p4 += gridDim.x * blockDim.x) {
    const int sp = 4 * p4;
    const int oh = sp / OW, ow = sp - oh * OW;
    const int h = oh * sh - ph + ki * dh;
    const int w = ow * sw - pw + kj * dw;
    if (sw == 1 && dw == 1 && sp + 3 < S && ow + 3 < OW && h >= 0 &&
        h < H && w >= 0 && w + 3 < W) {
      *(f4u*)(cp + sp) = *(const f4u*)(xp + h * W + w);
      continue;
    }
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float v = 0.f;
      const int spj = sp + j;
      if (spj < S) {
        const int ohj = spj / OW, owj = spj - ohj * OW;
        const int hj = ohj * sh - ph + ki * dh;
        const int wj = owj * sw - pw + kj * dw;
        if (hj >= 0 && hj < H && wj >= 0 && wj < W) v = xp[hj * W + wj];
      }
      cp[spj] = v;
    }
  }
}

void im2col_batched(hipStream_t s, const float* x, int Nimg, int C, int H,
                    int W, int kh, int kw, int ph, int pw, int sh, int sw,
                    int dh, int dw, int OH, int OW, long Spad, float* col) {
  const long total = (long)C * kh * kw * Nimg * Spad;
  PerfScope perf(PERF_CLASS("im2col"), s, 0,
                 8.0 * total);  // ~1 read + 1 write per element
  const int bx = (int)std::min<long>((Spad / 4 + TPB - 1) / TPB, 16);
  dim3 grid(bx, C * kh * kw, Nimg);
  hipLaunchKernelGGL(k_im2col_b, grid, dim3(TPB), 0, s, x, C, H, W, kh, kw,
                     ph, pw, sh, sw, dh, dw, OH, OW, Spad,
                     (long)Nimg * Spad, col);
}

// gather col2im (reference im2col.cu:256-295 pattern — no atomics)
// 3D grid: z = image, y = channel, x-threads sweep H*W (32-bit indices)
__global__ void k_col2im_b(const float* __restrict__ col, int C, int H,
                           int W, int kh, int kw, int ph, int pw, int sh,
                           int sw, int dh, int dw, int OH, int OW,
                           long Spad, long cols, float* __restrict__ dx) {
  const int n = blockIdx.z;
  const int c = blockIdx.y;
  const float* cp = col + (long)n * Spad;
  float* dp = dx + ((long)n * C + c) * H * W;
  for (int hw = blockIdx.x * blockDim.x + threadIdx.x; hw < H * W;
       hw += gridDim.x * blockDim.x) {
    const int h = hw / W, w = hw - h * W;
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
        const int row = (c * kh + i) * kw + j;
        acc += cp[(long)row * cols + hk * OW + wk];
      }
    }
    dp[hw] = acc;
  }
}

void col2im_batched(hipStream_t s, const float* dcol, int Nimg, int C, int H,
                    int W, int kh, int kw, int ph, int pw, int sh, int sw,
                    int dh, int dw, int OH, int OW, long Spad, float* dx) {
  const long total = (long)Nimg * C * H * W;
  PerfScope perf(PERF_CLASS("col2im"), s, 0, 8.0 * total * kh * kw / (sh * sw));
  const int bx = (int)std::min<long>(((long)H * W + TPB - 1) / TPB, 16);
  dim3 grid(bx, C, Nimg);
  hipLaunchKernelGGL(k_col2im_b, grid, dim3(TPB), 0, s, dcol, C, H, W, kh,
                     kw, ph, pw, sh, sw, dh, dw, OH, OW, Spad,
                     (long)Nimg * Spad, dx);
}

// ---- float4 elementwise helpers (tensors are 64B-padded, so the float4
// body covers n/4*4 and a scalar tail handles the rest)
using f4 = __attribute__((ext_vector_type(4))) float;
#define VEC_GRID(i, n4) \
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < (n4); \
       i += (long)gridDim.x * blockDim.x)
// ------------------------------------------------------------ relu
__global__ void k_relu_fwd(const f4* __restrict__ x, long n4, float slope,
                           f4* __restrict__ y) {
  VEC_GRID(i, n4) {
    f4 v = x[i];
#pragma unroll
    for (int j = 0; j < 4; ++j) v[j] = v[j] > 0.f ? v[j] : slope * v[j];
    y[i] = v;
  }
}
void relu_fwd(hipStream_t s, const float* x, long n, float slope, float* y) {
  PerfScope perf(PERF_CLASS("relu"), s, 0, 8.0 * n);
  const long n4 = (n + 3) / 4;  // blobs are 64B-padded
  hipLaunchKernelGGL(k_relu_fwd, dim3(nblocks(n4, 4)), dim3(TPB), 0, s,
                     (const f4*)x, n4, slope, (f4*)y);
}

__global__ void k_relu_bwd(const f4* __restrict__ x,
                           const f4* __restrict__ dy, long n4, float slope,
                           f4* __restrict__ dx) {
  VEC_GRID(i, n4) {
    const f4 v = x[i];
    f4 d = dy[i];
#pragma unroll
    for (int j = 0; j < 4; ++j) d[j] *= v[j] > 0.f ? 1.f : slope;
    dx[i] = d;
  }
}
void relu_bwd(hipStream_t s, const float* x, const float* dy, long n,
              float slope, float* dx) {
  PerfScope perf(PERF_CLASS("relu"), s, 0, 12.0 * n);
  const long n4 = (n + 3) / 4;
  hipLaunchKernelGGL(k_relu_bwd, dim3(nblocks(n4, 4)), dim3(TPB), 0, s,
                     (const f4*)x, (const f4*)dy, n4, slope, (f4*)dx);
}

__global__ void k_axpy(const f4* __restrict__ x, long n4, float a,
                       f4* __restrict__ y) {
  VEC_GRID(i, n4) {
    f4 v = y[i];
    const f4 xv = x[i];
#pragma unroll
    for (int j = 0; j < 4; ++j) v[j] += a * xv[j];
    y[i] = v;
  }
}
// y += a*x over the (64B-padded) gradient arena — iter_size accumulation
void axpy(hipStream_t s, long n, float a, const float* x, float* y) {
  PerfScope perf(PERF_CLASS("eltwise"), s, 0, 12.0 * n);
  const long n4 = (n + 3) / 4;
  hipLaunchKernelGGL(k_axpy, dim3(nblocks(n4, 4)), dim3(TPB), 0, s,
                     (const f4*)x, n4, a, (f4*)y);
}

// ------------------------------------------------------------ pooling
__global__ void k_pool_max_fwd(const float* __restrict__ x, int N, int C,
                               int H, int W, int kh, int kw, int ph, int pw,
                               int sh, int sw, int OH, int OW,
                               float* __restrict__ y, int* __restrict__ mask) {
  const long total = (long)N * C * OH * OW;
  GRID_STRIDE(idx, total) {
    const int ow = (int)(idx % OW);
    const int oh = (int)((idx / OW) % OH);
    const long nc = idx / ((long)OW * OH);
    const float* xp = x + nc * H * W;
    int hs = oh * sh - ph, ws = ow * sw - pw;
    const int he = min(hs + kh, H), we = min(ws + kw, W);
    hs = max(hs, 0);
    ws = max(ws, 0);
    float best = -3.402823466e38f;
    int bi = -1;
    for (int h = hs; h < he; ++h)
      for (int w = ws; w < we; ++w) {
        const float v = xp[h * W + w];
        if (v > best) {  // strict >, first max wins (pooling_layer.cpp:166)
          best = v;
          bi = h * W + w;
        }
      }
    y[idx] = best;
    mask[idx] = bi;
  }
}
// 3x3 specialization: the generic kernel's runtime-bounded window loop
// keeps only ~2 loads in flight per wave (SQ: 76-78% parked on the
// GoogLeNet pools); a fully-unrolled predicated 3x3 issues all 9 loads
// before any wait.  Same first-max-wins order (fixed (h,w) ascending).
__global__ void k_pool_max_fwd3(const float* __restrict__ x, int N, int C,
                                int H, int W, int ph, int pw, int sh,
                                int sw, int OH, int OW,
                                float* __restrict__ y,
                                int* __restrict__ mask) {
  const long total = (long)N * C * OH * OW;
  GRID_STRIDE(idx, total) {
    const int ow = (int)(idx % OW);
    const int oh = (int)((idx / OW) % OH);
    const long nc = idx / ((long)OW * OH);
    const float* xp = x + nc * H * W;
    const int hs = oh * sh - ph, ws = ow * sw - pw;
    float v[9];
#pragma unroll
    for (int a = 0; a < 3; ++a)
#pragma unroll
      for (int b = 0; b < 3; ++b) {
        const int h = hs + a, w = ws + b;
        const bool ok = h >= 0 && h < H && w >= 0 && w < W;
        v[a * 3 + b] = ok ? xp[h * W + w] : -3.402823466e38f;
      }
    float best = -3.402823466e38f;
    int bi = -1;
#pragma unroll
    for (int a = 0; a < 3; ++a)
#pragma unroll
      for (int b = 0; b < 3; ++b) {
        const int h = hs + a, w = ws + b;
        const bool ok = h >= 0 && h < H && w >= 0 && w < W;
        if (ok && v[a * 3 + b] > best) {  // strict >: first max wins
          best = v[a * 3 + b];
          bi = h * W + w;
        }
      }
    y[idx] = best;
    mask[idx] = bi;
  }
}

void pool_max_fwd(hipStream_t s, const float* x, int N, int C, int H, int W,
                  int kh, int kw, int ph, int pw, int sh, int sw, int OH,
                  int OW, float* y, int* mask) {
  const long total = (long)N * C * OH * OW;
  PerfScope perf(PERF_CLASS("pool"), s, 0, 4.0 * total * (kh * kw + 2));
  if (kh == 3 && kw == 3) {
    hipLaunchKernelGGL(k_pool_max_fwd3, dim3(nblocks(total, 2)), dim3(TPB),
                       0, s, x, N, C, H, W, ph, pw, sh, sw, OH, OW, y,
                       mask);
    return;
  }
  hipLaunchKernelGGL(k_pool_max_fwd, dim3(nblocks(total, 2)), dim3(TPB), 0,
                     s, x, N, C, H, W, kh, kw, ph, pw, sh, sw, OH, OW, y,
                     mask);
}

// deterministic gather backward: each input element scans the <= ceil(k/s)^2
// windows that can contain it and sums where mask points at it
__global__ void k_pool_max_bwd(const float* __restrict__ dy,
                               const int* __restrict__ mask, int N, int C,
                               int H, int W, int kh, int kw, int ph, int pw,
                               int sh, int sw, int OH, int OW,
                               float* __restrict__ dx) {
  const long total = (long)N * C * H * W;
  GRID_STRIDE(idx, total) {
    const int w = (int)(idx % W);
    const int h = (int)((idx / W) % H);
    const long nc = idx / ((long)W * H);
    const int me = h * W + w;
    const int ph0 = (h + ph < kh) ? 0 : (h + ph - kh) / sh + 1;
    const int ph1 = min((h + ph) / sh + 1, OH);
    const int pw0 = (w + pw < kw) ? 0 : (w + pw - kw) / sw + 1;
    const int pw1 = min((w + pw) / sw + 1, OW);
    float acc = 0.f;
    const float* dyp = dy + nc * OH * OW;
    const int* mp = mask + nc * OH * OW;
    for (int a = ph0; a < ph1; ++a)
      for (int b = pw0; b < pw1; ++b)
        if (mp[a * OW + b] == me) acc += dyp[a * OW + b];
    dx[idx] = acc;
  }
}
// 3x3 bwd specialization: for k=3 an input element sits in at most 3x3
// candidate windows — unroll them with predicates so all 18 mask/dy
// loads issue before any wait (the generic loop was latency-serialized,
// 78% parked).  Ascending (a, b) — fixed, deterministic order.
__global__ void k_pool_max_bwd3(const float* __restrict__ dy,
                                const int* __restrict__ mask, int N, int C,
                                int H, int W, int ph, int pw, int sh,
                                int sw, int OH, int OW,
                                float* __restrict__ dx) {
  const long total = (long)N * C * H * W;
  GRID_STRIDE(idx, total) {
    const int w = (int)(idx % W);
    const int h = (int)((idx / W) % H);
    const long nc = idx / ((long)W * H);
    const int me = h * W + w;
    const int ph0 = (h + ph < 3) ? 0 : (h + ph - 3) / sh + 1;
    const int ph1 = min((h + ph) / sh + 1, OH);
    const int pw0 = (w + pw < 3) ? 0 : (w + pw - 3) / sw + 1;
    const int pw1 = min((w + pw) / sw + 1, OW);
    const float* dyp = dy + nc * OH * OW;
    const int* mp = mask + nc * OH * OW;
    bool hit[9];
    float d[9];
#pragma unroll
    for (int da = 0; da < 3; ++da)
#pragma unroll
      for (int db = 0; db < 3; ++db) {
        const int a = ph0 + da, b = pw0 + db;
        const bool ok = a < ph1 && b < pw1;
        const int off = ok ? a * OW + b : 0;
        hit[da * 3 + db] = ok && mp[off] == me;
        d[da * 3 + db] = dyp[off];
      }
    float acc = 0.f;
#pragma unroll
    for (int i = 0; i < 9; ++i)
      if (hit[i]) acc += d[i];  // select, not multiply: a stray NaN in a
                                // NON-matching window must not leak in
    dx[idx] = acc;
  }
}

// 3x3 STRIDE-2 form (the ResNet/GoogLeNet/AlexNet downsampling pools):
// at stride 2 an input pixel overlaps at most 2x2 output windows, so the
// gather loop is 2x2 — 4 dy + 4 mask loads instead of the generic form's
// 9+9 — and the stride divisions become shifts
__global__ void k_pool_max_bwd3s2(const float* __restrict__ dy,
                                  const int* __restrict__ mask, int N,
                                  int C, int H, int W, int ph, int pw,
                                  int OH, int OW, float* __restrict__ dx) {
  const long total = (long)N * C * H * W;
  GRID_STRIDE(idx, total) {
    const int w = (int)(idx % W);
    const int h = (int)((idx / W) % H);
    const long nc = idx / ((long)W * H);
    const int me = h * W + w;
    const int th = h + ph, tw = w + pw;  // >= 0
    const int ph0 = (th < 3) ? 0 : ((th - 3) >> 1) + 1;
    const int ph1 = min((th >> 1) + 1, OH);
    const int pw0 = (tw < 3) ? 0 : ((tw - 3) >> 1) + 1;
    const int pw1 = min((tw >> 1) + 1, OW);
    const float* dyp = dy + nc * OH * OW;
    const int* mp = mask + nc * OH * OW;
    bool hit[4];
    float d[4];
#pragma unroll
    for (int da = 0; da < 2; ++da)
#pragma unroll
      for (int db = 0; db < 2; ++db) {
        const int a = ph0 + da, b = pw0 + db;
        const bool ok = a < ph1 && b < pw1;
        const int off = ok ? a * OW + b : 0;
        hit[da * 2 + db] = ok && mp[off] == me;
        d[da * 2 + db] = dyp[off];
      }
    float acc = 0.f;
#pragma unroll
    for (int i = 0; i < 4; ++i)
      if (hit[i]) acc += d[i];  // select, not multiply (NaN containment)
    dx[idx] = acc;
  }
}

// 3x3 STRIDE-1 form (the 13 GoogLeNet inception pools): window bounds
// become max/min — no per-element divisions at all beyond the index
// decompose (the generic form pays five)
__global__ void k_pool_max_bwd3s1(const float* __restrict__ dy,
                                  const int* __restrict__ mask, int N,
                                  int C, int H, int W, int ph, int pw,
                                  int OH, int OW, float* __restrict__ dx) {
  const long total = (long)N * C * H * W;
  GRID_STRIDE(idx, total) {
    const int w = (int)(idx % W);
    const int h = (int)((idx / W) % H);
    const long nc = idx / ((long)W * H);
    const int me = h * W + w;
    const int ph0 = max(h + ph - 2, 0);
    const int ph1 = min(h + ph + 1, OH);
    const int pw0 = max(w + pw - 2, 0);
    const int pw1 = min(w + pw + 1, OW);
    const float* dyp = dy + nc * OH * OW;
    const int* mp = mask + nc * OH * OW;
    bool hit[9];
    float d[9];
#pragma unroll
    for (int da = 0; da < 3; ++da)
#pragma unroll
      for (int db = 0; db < 3; ++db) {
        const int a = ph0 + da, b = pw0 + db;
        const bool ok = a < ph1 && b < pw1;
        const int off = ok ? a * OW + b : 0;
        hit[da * 3 + db] = ok && mp[off] == me;
        d[da * 3 + db] = dyp[off];
      }
    float acc = 0.f;
#pragma unroll
    for (int i = 0; i < 9; ++i)
      if (hit[i]) acc += d[i];
    dx[idx] = acc;
  }
}

void pool_max_bwd(hipStream_t s, const float* dy, const int* mask, int N,
                  int C, int H, int W, int kh, int kw, int ph, int pw,
                  int sh, int sw, int OH, int OW, float* dx) {
  const long total = (long)N * C * H * W;
  PerfScope perf(PERF_CLASS("pool"), s, 0, 12.0 * total);
  if (kh == 3 && kw == 3 && sh == 1 && sw == 1) {
    hipLaunchKernelGGL(k_pool_max_bwd3s1, dim3(nblocks(total, 2)),
                       dim3(TPB), 0, s, dy, mask, N, C, H, W, ph, pw, OH,
                       OW, dx);
    return;
  }
  if (kh == 3 && kw == 3 && sh == 2 && sw == 2) {
    hipLaunchKernelGGL(k_pool_max_bwd3s2, dim3(nblocks(total, 2)),
                       dim3(TPB), 0, s, dy, mask, N, C, H, W, ph, pw, OH,
                       OW, dx);
    return;
  }
  if (kh == 3 && kw == 3) {
    hipLaunchKernelGGL(k_pool_max_bwd3, dim3(nblocks(total, 2)), dim3(TPB),
                       0, s, dy, mask, N, C, H, W, ph, pw, sh, sw, OH, OW,
                       dx);
    return;
  }
  hipLaunchKernelGGL(k_pool_max_bwd, dim3(nblocks(total, 2)), dim3(TPB), 0,
                     s, dy, mask, N, C, H, W, kh, kw, ph, pw, sh, sw, OH,
                     OW, dx);
}

__global__ void k_pool_ave_fwd(const float* __restrict__ x, int N, int C,
                               int H, int W, int kh, int kw, int ph, int pw,
                               int sh, int sw, int OH, int OW,
                               float* __restrict__ y) {
  const long total = (long)N * C * OH * OW;
  GRID_STRIDE(idx, total) {
    const int ow = (int)(idx % OW);
    const int oh = (int)((idx / OW) % OH);
    const long nc = idx / ((long)OW * OH);
    const float* xp = x + nc * H * W;
    int hs = oh * sh - ph, ws = ow * sw - pw;
    int he = min(hs + kh, H + ph), we = min(ws + kw, W + pw);
    const int ps = (he - hs) * (we - ws);  // padded size (:201)
    hs = max(hs, 0);
    ws = max(ws, 0);
    he = min(he, H);
    we = min(we, W);
    float acc = 0.f;
    for (int h = hs; h < he; ++h)
      for (int w = ws; w < we; ++w) acc += xp[h * W + w];
    y[idx] = acc / ps;
  }
}
void pool_ave_fwd(hipStream_t s, const float* x, int N, int C, int H, int W,
                  int kh, int kw, int ph, int pw, int sh, int sw, int OH,
                  int OW, float* y) {
  const long total = (long)N * C * OH * OW;
  PerfScope perf(PERF_CLASS("pool"), s, 0, 4.0 * total * (kh * kw + 1));
  hipLaunchKernelGGL(k_pool_ave_fwd, dim3(nblocks(total, 2)), dim3(TPB), 0,
                     s, x, N, C, H, W, kh, kw, ph, pw, sh, sw, OH, OW, y);
}

__global__ void k_pool_ave_bwd(const float* __restrict__ dy, int N, int C,
                               int H, int W, int kh, int kw, int ph, int pw,
                               int sh, int sw, int OH, int OW,
                               float* __restrict__ dx) {
  const long total = (long)N * C * H * W;
  GRID_STRIDE(idx, total) {
    const int w = (int)(idx % W);
    const int h = (int)((idx / W) % H);
    const long nc = idx / ((long)W * H);
    const int ph0 = (h + ph < kh) ? 0 : (h + ph - kh) / sh + 1;
    const int ph1 = min((h + ph) / sh + 1, OH);
    const int pw0 = (w + pw < kw) ? 0 : (w + pw - kw) / sw + 1;
    const int pw1 = min((w + pw) / sw + 1, OW);
    float acc = 0.f;
    const float* dyp = dy + nc * OH * OW;
    for (int a = ph0; a < ph1; ++a)
      for (int b = pw0; b < pw1; ++b) {
        int hs = a * sh - ph, ws = b * sw - pw;
        const int he = min(hs + kh, H + ph), we = min(ws + kw, W + pw);
        const int ps = (he - hs) * (we - ws);
        acc += dyp[a * OW + b] / ps;
      }
    dx[idx] = acc;
  }
}
void pool_ave_bwd(hipStream_t s, const float* dy, int N, int C, int H, int W,
                  int kh, int kw, int ph, int pw, int sh, int sw, int OH,
                  int OW, float* dx) {
  const long total = (long)N * C * H * W;
  PerfScope perf(PERF_CLASS("pool"), s, 0, 8.0 * total);
  hipLaunchKernelGGL(k_pool_ave_bwd, dim3(nblocks(total, 2)), dim3(TPB), 0,
                     s, dy, N, C, H, W, kh, kw, ph, pw, sh, sw, OH, OW, dx);
}

// y = x * a[c] (+ b[c]) — Scale layer forward (b = bias or null) and its
// backward-data (a = scale, b = null), reference scale_layer.cpp
__global__ void k_chan_affine(const float* __restrict__ x,
                              const float* __restrict__ a,
                              const float* __restrict__ b, int C, long S,
                              long total, float* __restrict__ y) {
  GRID_STRIDE(i, total) {
    const int c = (int)((i / S) % C);
    y[i] = x[i] * a[c] + (b ? b[c] : 0.f);
  }
}
void chan_affine(hipStream_t s, const float* x, const float* a,
                 const float* b, int N, int C, long S, float* y) {
  const long total = (long)N * C * S;
  PerfScope perf(PERF_CLASS("eltwise"), s, 0, 8.0 * total);
  hipLaunchKernelGGL(k_chan_affine, dim3(nblocks(total, 8)), dim3(TPB), 0,
                     s, x, a, b, C, S, total, y);
}

// ------------------------------------------------------------ batchnorm
// partials layout: double2[C][nb] {sum, sumsq} (fwd) / {sum_dy, sum_dyxn}
// (bwd).  Deterministic: fixed block→slice mapping, in-block tree reduce.
int bn_blocks_per_channel(int N, int C) {
  // image-sliced reduction: nb batch slices per channel, ~2048 blocks total
  return std::max(1, std::min(N, 2048 / std::max(1, C)));
}

__global__ void k_bn_fwd_stats(const float* __restrict__ x, int N, int C,
                               long S, int nb, double2* __restrict__ out) {
  const int c = blockIdx.x % C;
  const int slice = blockIdx.x / C;
  const int n0 = (int)((long)N * slice / nb);
  const int n1 = (int)((long)N * (slice + 1) / nb);
  const int Si = (int)S;
  const int span = (n1 - n0) * Si;
  const int B = blockDim.x;
  auto addr = [&](int i) {
    const int n = n0 + i / Si;
    const int sp = i - (n - n0) * Si;
    return ((long)n * C + c) * S + sp;
  };
  double s1 = 0, s2 = 0, t1 = 0, t2 = 0;
  int i = threadIdx.x;
  for (; i + 3 * B < span; i += 4 * B) {  // 4 loads in flight (MLP)
    const float v0 = x[addr(i)], v1 = x[addr(i + B)];
    const float v2 = x[addr(i + 2 * B)], v3 = x[addr(i + 3 * B)];
    s1 += (double)v0 + v1;
    s2 += (double)v0 * v0 + (double)v1 * v1;
    t1 += (double)v2 + v3;
    t2 += (double)v2 * v2 + (double)v3 * v3;
  }
  for (; i < span; i += B) {
    const double v = x[addr(i)];
    s1 += v;
    s2 += v * v;
  }
  s1 += t1;
  s2 += t2;
  __shared__ double sh1[TPB], sh2[TPB];
  sh1[threadIdx.x] = s1;
  sh2[threadIdx.x] = s2;
  __syncthreads();
  for (int off = TPB / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) {
      sh1[threadIdx.x] += sh1[threadIdx.x + off];
      sh2[threadIdx.x] += sh2[threadIdx.x + off];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) out[(long)c * nb + slice] = {sh1[0], sh2[0]};
}

// f4 variant (S % 4 == 0: every ResNet spatial stage except 7x7=49):
// quarters the load count and the per-element index divisions.  4-way
// unrolled with independent partial sums: the rolled loop kept ONE load
// in flight per wave (the accumulate chained on it), measuring 2.85 TB/s
// with waves 59-81% parked — four independent loads per iteration is the
// memory-level-parallelism fix, not more occupancy.
__global__ void k_bn_fwd_stats_v4(const f4* __restrict__ x4, int N, int C,
                                  int S4, int nb, double2* __restrict__ out) {
  const int c = blockIdx.x % C;
  const int slice = blockIdx.x / C;
  const int n0 = (int)((long)N * slice / nb);
  const int n1 = (int)((long)N * (slice + 1) / nb);
  const int span = (n1 - n0) * S4;
  const int B = blockDim.x;
  double s1 = 0, s2 = 0;
  double t1 = 0, t2 = 0;  // second accumulator pair halves the add chain
  auto addr = [&](int i) {
    const int n = n0 + i / S4;
    const int sp = i - (n - n0) * S4;
    return ((long)n * C + c) * S4 + sp;
  };
  int i = threadIdx.x;
  for (; i + 3 * B < span; i += 4 * B) {
    const f4 v0 = x4[addr(i)];
    const f4 v1 = x4[addr(i + B)];
    const f4 v2 = x4[addr(i + 2 * B)];
    const f4 v3 = x4[addr(i + 3 * B)];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      s1 += (double)v0[j] + v1[j];
      s2 += (double)v0[j] * v0[j] + (double)v1[j] * v1[j];
      t1 += (double)v2[j] + v3[j];
      t2 += (double)v2[j] * v2[j] + (double)v3[j] * v3[j];
    }
  }
  for (; i < span; i += B) {
    const f4 v = x4[addr(i)];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      s1 += v[j];
      s2 += (double)v[j] * v[j];
    }
  }
  s1 += t1;
  s2 += t2;
  __shared__ double sh1[TPB], sh2[TPB];
  sh1[threadIdx.x] = s1;
  sh2[threadIdx.x] = s2;
  __syncthreads();
  for (int off = TPB / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) {
      sh1[threadIdx.x] += sh1[threadIdx.x + off];
      sh2[threadIdx.x] += sh2[threadIdx.x + off];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) out[(long)c * nb + slice] = {sh1[0], sh2[0]};
}

void bn_fwd_stats(hipStream_t s, const float* x, int N, int C, long S,
                  int nb, void* partials) {
  PerfScope perf(PERF_CLASS("bn"), s, 0, 4.0 * N * C * S);
  if (S % 4 == 0) {
    hipLaunchKernelGGL(k_bn_fwd_stats_v4, dim3(C * nb), dim3(TPB), 0, s,
                       (const f4*)x, N, C, (int)(S / 4), nb,
                       (double2*)partials);
    return;
  }
  hipLaunchKernelGGL(k_bn_fwd_stats, dim3(C * nb), dim3(TPB), 0, s, x, N, C,
                     S, nb, (double2*)partials);
}

__global__ void k_bn_fwd_finalize(const double2* __restrict__ partials,
                                  int nb, int C, long NS, float eps,
                                  float* __restrict__ mean,
                                  float* __restrict__ var,
                                  float* __restrict__ inv_std) {
  GRID_STRIDE(c, (long)C) {
    double s1 = 0, s2 = 0;
    for (int b = 0; b < nb; ++b) {
      s1 += partials[c * nb + b].x;
      s2 += partials[c * nb + b].y;
    }
    const double m = s1 / NS;
    const double v = s2 / NS - m * m;  // E[x^2]-m^2 in double ≈ E[(x-m)^2]
    mean[c] = (float)m;
    var[c] = (float)v;
    inv_std[c] = (float)(1.0 / sqrt(v + (double)eps));
  }
}
void bn_fwd_finalize(hipStream_t s, const void* partials, int nb, int C,
                     long NS, float eps, float* mean, float* var,
                     float* inv_std) {
  hipLaunchKernelGGL(k_bn_fwd_finalize, dim3(1), dim3(TPB), 0, s,
                     (const double2*)partials, nb, C, NS, eps, mean, var,
                     inv_std);
}

// flat float4 walk; per 4-pack ONE division recovers the channel (packs
// never straddle a channel boundary check: split handled elementwise)
__global__ void k_bn_fwd_norm(const f4* __restrict__ x,
                              const float* __restrict__ mean,
                              const float* __restrict__ inv_std,
                              const float* __restrict__ scale,
                              const float* __restrict__ bias, int sb, int C,
                              int S, int frelu, long n4,
                              const f4* __restrict__ add,
                              f4* __restrict__ y) {
  VEC_GRID(i, n4) {
    // 32-bit index math: the 64-bit division per 16B pack is ~3x the
    // instruction cost of the 32-bit one, and every count here fits
    // uint32 (largest tensor ~300M elements)
    const unsigned e0 = (unsigned)(i * 4);
    const int row = (int)(e0 / (unsigned)S);  // n*C + c
    const int c0 = row % C;
    const int rem = (int)(e0 - (unsigned)row * (unsigned)S);
    f4 v = x[i];
    if (rem + 4 <= S) {
      const float mu = mean[c0], inv = inv_std[c0];
      const float sc = sb ? scale[c0] : 1.f, bi = sb ? bias[c0] : 0.f;
#pragma unroll
      for (int j = 0; j < 4; ++j) v[j] = (v[j] - mu) * inv * sc + bi;
    } else {  // pack crosses a channel boundary
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const unsigned e = e0 + j;
        const int cc = (int)((e / (unsigned)S) % (unsigned)C);
        v[j] = (v[j] - mean[cc]) * inv_std[cc] * (sb ? scale[cc] : 1.f) +
               (sb ? bias[cc] : 0.f);
      }
    }
    if (add) {  // fused residual: bn(x) + other (ResNet block pattern)
      const f4 a = add[i];
#pragma unroll
      for (int j = 0; j < 4; ++j) v[j] += a[j];
    }
    if (frelu) {
#pragma unroll
      for (int j = 0; j < 4; ++j) v[j] = fmaxf(v[j], 0.f);
    }
    y[i] = v;
  }
}
void bn_fwd_norm(hipStream_t s, const float* x, const float* mean,
                 const float* inv_std, const float* scale, const float* bias,
                 int sb, int N, int C, long S, float* y, int fuse_relu,
                 const float* add) {
  const long total = (long)N * C * S;
  PerfScope perf(PERF_CLASS("bn"), s, 0, (add ? 12.0 : 8.0) * total);
  // blobs are 16-float padded: the final partial pack reads/writes pad
  // space with wrapped (but in-bounds) channel indices — harmless
  const long n4 = (total + 3) / 4;
  hipLaunchKernelGGL(k_bn_fwd_norm, dim3(nblocks(n4, 4)), dim3(TPB), 0, s,
                     (const f4*)x, mean, inv_std, scale, bias, sb, C,
                     (int)S, fuse_relu, n4, (const f4*)add, (f4*)y);
}

__global__ void k_bn_moving_avg(const float* __restrict__ mean,
                                const float* __restrict__ var, int C,
                                float maf, int copy_only,
                                float* __restrict__ gmean,
                                float* __restrict__ gvar) {
  GRID_STRIDE(c, (long)C) {
    if (copy_only) {
      gmean[c] = mean[c];
      gvar[c] = var[c];
    } else {  // batch_norm_layer.cpp:200-207
      gmean[c] = (1.f - maf) * mean[c] + maf * gmean[c];
      gvar[c] = (1.f - maf) * var[c] + maf * gvar[c];
    }
  }
}
void bn_moving_avg(hipStream_t s, const float* mean, const float* var, int C,
                   float maf, int copy_only, float* gmean, float* gvar) {
  hipLaunchKernelGGL(k_bn_moving_avg, dim3(1), dim3(TPB), 0, s, mean, var,
                     C, maf, copy_only, gmean, gvar);
}

__global__ void k_bn_fwd_test(const float* __restrict__ x,
                              const float* __restrict__ gmean,
                              const float* __restrict__ gvar,
                              const float* __restrict__ scale,
                              const float* __restrict__ bias, int sb, int N,
                              int C, long S, float eps,
                              float* __restrict__ y) {
  const long total = (long)N * C * S;
  GRID_STRIDE(i, total) {
    const int c = (int)((i / S) % C);
    const float inv = rsqrtf(gvar[c] + eps);
    const float v = (x[i] - gmean[c]) * inv;
    y[i] = sb ? v * scale[c] + bias[c] : v;
  }
}
void bn_fwd_test(hipStream_t s, const float* x, const float* gmean,
                 const float* gvar, const float* scale, const float* bias,
                 int sb, int N, int C, long S, float eps, float* y) {
  const long total = (long)N * C * S;
  PerfScope perf(PERF_CLASS("bn"), s, 0, 8.0 * total);
  hipLaunchKernelGGL(k_bn_fwd_test, dim3(nblocks(total, 8)), dim3(TPB), 0,
                     s, x, gmean, gvar, scale, bias, sb, N, C, S, eps, y);
}

__global__ void k_bn_bwd_stats(const float* __restrict__ x,
                               const float* __restrict__ dy,
                               const float* __restrict__ mean,
                               const float* __restrict__ inv_std, int N,
                               int C, long S, int nb,
                               const float* __restrict__ scale,
                               const float* __restrict__ bias, int frelu,
                               double2* __restrict__ out) {
  const int c = blockIdx.x % C;
  const int slice = blockIdx.x / C;
  const int n0 = (int)((long)N * slice / nb);
  const int n1 = (int)((long)N * (slice + 1) / nb);
  const int Si = (int)S;
  const int span = (n1 - n0) * Si;
  const int B = blockDim.x;
  const float m = mean[c], inv = inv_std[c];
  // fused ReLU backward: recompute the forward activation's sign exactly
  // (same fp32 op sequence as k_bn_fwd_norm) — no extra memory stream
  const float sc = scale ? scale[c] : 1.f;
  const float bi = bias ? bias[c] : 0.f;
  auto addr = [&](int i) {
    const int n = n0 + i / Si;
    return ((long)n * C + c) * S + (i - (n - n0) * Si);
  };
  double s_dy = 0, s_dyxn = 0, t_dy = 0, t_dyxn = 0;
  int i = threadIdx.x;
  for (; i + B < span; i += 2 * B) {  // 4 loads (2 pairs) in flight
    const long o0 = addr(i), o1 = addr(i + B);
    const float xn0 = (x[o0] - m) * inv, xn1 = (x[o1] - m) * inv;
    double d0 = dy[o0], d1 = dy[o1];
    if (frelu && xn0 * sc + bi <= 0.f) d0 = 0.0;
    if (frelu && xn1 * sc + bi <= 0.f) d1 = 0.0;
    s_dy += d0;
    s_dyxn += d0 * (double)xn0;
    t_dy += d1;
    t_dyxn += d1 * (double)xn1;
  }
  for (; i < span; i += B) {
    const long off = addr(i);
    const float xn = (x[off] - m) * inv;
    double d = dy[off];
    if (frelu && xn * sc + bi <= 0.f) d = 0.0;
    s_dy += d;
    s_dyxn += d * (double)xn;
  }
  s_dy += t_dy;
  s_dyxn += t_dyxn;
  __shared__ double sh1[TPB], sh2[TPB];
  sh1[threadIdx.x] = s_dy;
  sh2[threadIdx.x] = s_dyxn;
  __syncthreads();
  for (int off = TPB / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) {
      sh1[threadIdx.x] += sh1[threadIdx.x + off];
      sh2[threadIdx.x] += sh2[threadIdx.x + off];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) out[(long)c * nb + slice] = {sh1[0], sh2[0]};
}
__global__ void k_bn_bwd_stats_v4(const f4* __restrict__ x4,
                                  const f4* __restrict__ dy4,
                                  const float* __restrict__ mean,
                                  const float* __restrict__ inv_std, int N,
                                  int C, int S4, int nb,
                                  const float* __restrict__ scale,
                                  const float* __restrict__ bias, int frelu,
                                  double2* __restrict__ out) {
  const int c = blockIdx.x % C;
  const int slice = blockIdx.x / C;
  const int n0 = (int)((long)N * slice / nb);
  const int n1 = (int)((long)N * (slice + 1) / nb);
  const int span = (n1 - n0) * S4;
  const int B = blockDim.x;
  const float m = mean[c], inv = inv_std[c];
  const float sc = scale ? scale[c] : 1.f;
  const float bi = bias ? bias[c] : 0.f;
  auto addr = [&](int i) {
    const int n = n0 + i / S4;
    return ((long)n * C + c) * S4 + (i - (n - n0) * S4);
  };
  double s_dy = 0, s_dyxn = 0, t_dy = 0, t_dyxn = 0;
  int i = threadIdx.x;
  for (; i + B < span; i += 2 * B) {  // 4 loads (2 pairs) in flight
    const long o0 = addr(i), o1 = addr(i + B);
    const f4 xv0 = x4[o0], dv0 = dy4[o0];
    const f4 xv1 = x4[o1], dv1 = dy4[o1];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float xn0 = (xv0[j] - m) * inv;
      const float xn1 = (xv1[j] - m) * inv;
      double d0 = dv0[j], d1 = dv1[j];
      if (frelu && xn0 * sc + bi <= 0.f) d0 = 0.0;
      if (frelu && xn1 * sc + bi <= 0.f) d1 = 0.0;
      s_dy += d0;
      s_dyxn += d0 * (double)xn0;
      t_dy += d1;
      t_dyxn += d1 * (double)xn1;
    }
  }
  for (; i < span; i += B) {
    const long off = addr(i);
    const f4 xv = x4[off];
    const f4 dv = dy4[off];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float xn = (xv[j] - m) * inv;
      double d = dv[j];
      if (frelu && xn * sc + bi <= 0.f) d = 0.0;
      s_dy += d;
      s_dyxn += d * (double)xn;
    }
  }
  s_dy += t_dy;
  s_dyxn += t_dyxn;
  __shared__ double sh1[TPB], sh2[TPB];
  sh1[threadIdx.x] = s_dy;
  sh2[threadIdx.x] = s_dyxn;
  __syncthreads();
  for (int off = TPB / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) {
      sh1[threadIdx.x] += sh1[threadIdx.x + off];
      sh2[threadIdx.x] += sh2[threadIdx.x + off];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) out[(long)c * nb + slice] = {sh1[0], sh2[0]};
}

void bn_bwd_stats(hipStream_t s, const float* x, const float* dy,
                  const float* mean, const float* inv_std, int N, int C,
                  long S, int nb, const float* scale, const float* bias,
                  int frelu, void* partials) {
  PerfScope perf(PERF_CLASS("bn"), s, 0, 8.0 * N * C * S);
  if (S % 4 == 0) {
    hipLaunchKernelGGL(k_bn_bwd_stats_v4, dim3(C * nb), dim3(TPB), 0, s,
                       (const f4*)x, (const f4*)dy, mean, inv_std, N, C,
                       (int)(S / 4), nb, scale, bias, frelu,
                       (double2*)partials);
    return;
  }
  hipLaunchKernelGGL(k_bn_bwd_stats, dim3(C * nb), dim3(TPB), 0, s, x, dy,
                     mean, inv_std, N, C, S, nb, scale, bias, frelu,
                     (double2*)partials);
}

__global__ void k_bn_bwd_finalize(const double2* __restrict__ partials,
                                  int nb, int C, long NS,
                                  const float* __restrict__ scale, int sb,
                                  float* __restrict__ dscale,
                                  float* __restrict__ dbias,
                                  float* __restrict__ m_dy,
                                  float* __restrict__ m_dyxn) {
  GRID_STRIDE(c, (long)C) {
    double s_dy = 0, s_dyxn = 0;
    for (int b = 0; b < nb; ++b) {
      s_dy += partials[c * nb + b].x;
      s_dyxn += partials[c * nb + b].y;
    }
    if (sb) {
      dscale[c] = (float)s_dyxn;  // Σ dy·x̂ (batch_norm_layer.cpp:322-330)
      dbias[c] = (float)s_dy;     // Σ dy
    }
    const double sc = sb ? (double)scale[c] : 1.0;
    m_dy[c] = (float)(sc * s_dy / NS);
    m_dyxn[c] = (float)(sc * s_dyxn / NS);
  }
}
void bn_bwd_finalize(hipStream_t s, const void* partials, int nb, int C,
                     long NS, const float* scale, int sb, float* dscale,
                     float* dbias, float* m_dy, float* m_dyxn) {
  hipLaunchKernelGGL(k_bn_bwd_finalize, dim3(1), dim3(TPB), 0, s,
                     (const double2*)partials, nb, C, NS, scale, sb, dscale,
                     dbias, m_dy, m_dyxn);
}

__global__ void k_bn_bwd_apply(const f4* __restrict__ x,
                               const f4* __restrict__ dy,
                               const float* __restrict__ mean,
                               const float* __restrict__ inv_std,
                               const float* __restrict__ scale, int sb,
                               const float* __restrict__ m_dy,
                               const float* __restrict__ m_dyxn,
                               const float* __restrict__ bias, int frelu,
                               int C, int S, long n4, f4* __restrict__ dx) {
  VEC_GRID(i, n4) {
    const unsigned e0 = (unsigned)(i * 4);
    const int row = (int)(e0 / (unsigned)S);
    const int rem = (int)(e0 - (unsigned)row * (unsigned)S);
    const f4 xv = x[i];
    f4 d = dy[i];
    if (rem + 4 <= S) {
      const int c = row % C;
      const float mu = mean[c], inv = inv_std[c];
      const float sc = sb ? scale[c] : 1.f;
      const float bi = (frelu && bias) ? bias[c] : 0.f;
      const float mdy = m_dy[c], mdyxn = m_dyxn[c];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const float xn = (xv[j] - mu) * inv;
        float dj = d[j];
        if (frelu && xn * sc + bi <= 0.f) dj = 0.f;
        d[j] = (dj * sc - mdy - mdyxn * xn) * inv;
      }
    } else {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int cc = (int)(((e0 + j) / (unsigned)S) % (unsigned)C);
        const float xn = (xv[j] - mean[cc]) * inv_std[cc];
        const float scc = sb ? scale[cc] : 1.f;
        float dj = d[j];
        if (frelu && xn * scc + ((frelu && bias) ? bias[cc] : 0.f) <= 0.f)
          dj = 0.f;
        d[j] = (dj * scc - m_dy[cc] - m_dyxn[cc] * xn) * inv_std[cc];
      }
    }
    dx[i] = d;
  }
}
void bn_bwd_apply(hipStream_t s, const float* x, const float* dy,
                  const float* mean, const float* inv_std,
                  const float* scale, int sb, const float* m_dy,
                  const float* m_dyxn, int N, int C, long S,
                  const float* bias, int frelu, float* dx) {
  const long total = (long)N * C * S;
  PerfScope perf(PERF_CLASS("bn"), s, 0, 12.0 * total);
  const long n4 = (total + 3) / 4;
  hipLaunchKernelGGL(k_bn_bwd_apply, dim3(nblocks(n4, 4)), dim3(TPB), 0, s,
                     (const f4*)x, (const f4*)dy, mean, inv_std, scale, sb,
                     m_dy, m_dyxn, bias, frelu, C, (int)S, n4, (f4*)dx);
}

// ------------------------------------------------------------ LRN
// TWO (n, s) positions per thread, far apart so both streams stay
// coalesced: the single running cross-channel window was a serial
// load->accumulate chain (SQ: 66% parked) — interleaving two independent
// chains doubles the loads in flight (lrn_layer.cu:9-60 restated)
__global__ void k_lrn_fwd(const float* __restrict__ x, int N, int C, long S,
                          int size, float aos, float beta, float k,
                          float* __restrict__ scale, float* __restrict__ y) {
  const long total = (long)N * S;
  const long half = (total + 1) / 2;
  const int pre = (size - 1) / 2;
  GRID_STRIDE(i, half) {
    const long idx0 = i;
    const long idx1 = i + half;
    const bool two = idx1 < total;
    const int n0 = (int)(idx0 / S);
    const int n1 = two ? (int)(idx1 / S) : n0;
    const long b0 = (long)n0 * C * S + (idx0 - (long)n0 * S);
    const long b1 = two ? (long)n1 * C * S + (idx1 - (long)n1 * S) : b0;
    float a0 = 0.f, a1 = 0.f;
    for (int c = 0; c < size - pre && c < C; ++c) {
      const float v0 = x[b0 + (long)c * S];
      const float v1 = x[b1 + (long)c * S];
      a0 += v0 * v0;
      a1 += v1 * v1;
    }
    for (int c = 0; c < C; ++c) {
      if (c > 0) {
        const int head = c + size - 1 - pre;
        if (head < C) {
          const float v0 = x[b0 + (long)head * S];
          const float v1 = x[b1 + (long)head * S];
          a0 += v0 * v0;
          a1 += v1 * v1;
        }
        const int tail = c - 1 - pre;
        if (tail >= 0) {
          const float v0 = x[b0 + (long)tail * S];
          const float v1 = x[b1 + (long)tail * S];
          a0 -= v0 * v0;
          a1 -= v1 * v1;
        }
      }
      const long o0 = b0 + (long)c * S;
      const long o1 = b1 + (long)c * S;
      const float s0 = k + aos * a0;
      const float s1 = k + aos * a1;
      scale[o0] = s0;
      y[o0] = x[o0] * __powf(s0, -beta);
      if (two) {
        scale[o1] = s1;
        y[o1] = x[o1] * __powf(s1, -beta);
      }
    }
  }
}
// size-SPECIALIZED forward (local_size is 5 in every model-zoo LRN): a
// compile-time register ring of the last SZ window terms makes the
// running-window update ONE global load per channel — the tail term and
// the center value both come out of the ring instead of being re-read
// from HBM (3 loads/channel -> 1).  Same two-position interleave.
template <int SZ>
__global__ void k_lrn_fwd_t(const float* __restrict__ x, int N, int C,
                            long S, float aos, float beta, float k,
                            float* __restrict__ scale,
                            float* __restrict__ y) {
  constexpr int pre = (SZ - 1) / 2;       // window [c-pre, c+SZ-1-pre]
  const long total = (long)N * S;
  const long half = (total + 1) / 2;
  GRID_STRIDE(i, half) {
    const long idx0 = i;
    const long idx1 = i + half;
    const bool two = idx1 < total;
    const int n0 = (int)(idx0 / S);
    const int n1 = two ? (int)(idx1 / S) : n0;
    const long b0 = (long)n0 * C * S + (idx0 - (long)n0 * S);
    const long b1 = two ? (long)n1 * C * S + (idx1 - (long)n1 * S) : b0;
    // ring[j] at top of iter c holds channel c-1-pre+j's raw value /
    // square = window(c-1); out-of-range channels are zero
    float q0[SZ], q1[SZ], v0[SZ], v1[SZ];
    float a0 = 0.f, a1 = 0.f;
#pragma unroll
    for (int j = 0; j < SZ; ++j) q0[j] = q1[j] = v0[j] = v1[j] = 0.f;
#pragma unroll
    for (int j = 0; j + pre + 1 < SZ; ++j)  // channels 0..SZ-2-pre
      if (j < C) {
        const float u0 = x[b0 + (long)j * S];
        const float u1 = x[b1 + (long)j * S];
        v0[j + pre + 1] = u0;
        q0[j + pre + 1] = u0 * u0;
        a0 += u0 * u0;
        v1[j + pre + 1] = u1;
        q1[j + pre + 1] = u1 * u1;
        a1 += u1 * u1;
      }
    for (int c = 0; c < C; ++c) {
      const int head = c + SZ - 1 - pre;
      float h0 = 0.f, h1 = 0.f;
      if (head < C) {
        h0 = x[b0 + (long)head * S];
        h1 = x[b1 + (long)head * S];
      }
      a0 += h0 * h0 - q0[0];
      a1 += h1 * h1 - q1[0];
#pragma unroll
      for (int j = 0; j + 1 < SZ; ++j) {
        q0[j] = q0[j + 1]; q1[j] = q1[j + 1];
        v0[j] = v0[j + 1]; v1[j] = v1[j + 1];
      }
      q0[SZ - 1] = h0 * h0; q1[SZ - 1] = h1 * h1;
      v0[SZ - 1] = h0;      v1[SZ - 1] = h1;
      const long o0 = b0 + (long)c * S;
      const float s0 = k + aos * a0;
      scale[o0] = s0;
      y[o0] = v0[pre] * __powf(s0, -beta);  // ring slot pre == channel c
      if (two) {
        const long o1 = b1 + (long)c * S;
        const float s1 = k + aos * a1;
        scale[o1] = s1;
        y[o1] = v1[pre] * __powf(s1, -beta);
      }
    }
  }
}

void lrn_fwd(hipStream_t s, const float* x, int N, int C, int H, int W,
             int size, float alpha, float beta, float k, float* scale,
             float* y) {
  const long S = (long)H * W;
  PerfScope perf(PERF_CLASS("lrn"), s, 0, 12.0 * N * C * S);
  if (size == 5) {
    hipLaunchKernelGGL(k_lrn_fwd_t<5>, dim3(nblocks((N * S + 1) / 2)),
                       dim3(TPB), 0, s, x, N, C, S, alpha / size, beta, k,
                       scale, y);
    return;
  }
  hipLaunchKernelGGL(k_lrn_fwd, dim3(nblocks((N * S + 1) / 2)), dim3(TPB),
                     0, s, x, N, C, S, size, alpha / size, beta, k, scale,
                     y);
}

__global__ void k_lrn_bwd(const float* __restrict__ x,
                          const float* __restrict__ y,
                          const float* __restrict__ dy,
                          const float* __restrict__ scale, int N, int C,
                          long S, int size, float cr, float beta,
                          float* __restrict__ dx) {
  const long total = (long)N * S;
  const int pre = (size - 1) / 2;
  GRID_STRIDE(idx, total) {
    const int n = (int)(idx / S);
    const long sp = idx - (long)n * S;
    const long base = (long)n * C * S + sp;
    // ratio(c) = dy*y/scale; window for dx[c]: cc in [c-(size-1-pre), c+pre]
    float acc = 0.f;
    const int lo0 = -(size - 1 - pre);
    for (int cc = lo0; cc <= pre - 1; ++cc)
      if (cc >= 0 && cc < C) {
        const long i = base + (long)cc * S;
        acc += dy[i] * y[i] / scale[i];
      }
    for (int c = 0; c < C; ++c) {
      const int head = c + pre;
      if (head >= 0 && head < C) {
        const long i = base + (long)head * S;
        acc += dy[i] * y[i] / scale[i];
      }
      const long i = base + (long)c * S;
      dx[i] = dy[i] * __powf(scale[i], -beta) - cr * x[i] * acc;
      const int tail = c + lo0;
      if (tail >= 0 && tail < C) {
        const long j = base + (long)tail * S;
        acc -= dy[j] * y[j] / scale[j];
      }
    }
  }
}
// size-SPECIALIZED backward: register rings of the last SZ-1 window
// ratios (dy*y/scale) AND of the diagonal term dy*scale^-beta — each
// channel's (dy, y, scale) triple is loaded ONCE, as the window head,
// instead of three times (head ratio, tail ratio, diagonal): 9 loads +
// 2 divisions per channel -> 4 loads + 1 division.  Two (n, s) positions
// per thread as in the forward (the single running chain serializes).
template <int SZ>
__global__ void k_lrn_bwd_t(const float* __restrict__ x,
                            const float* __restrict__ y,
                            const float* __restrict__ dy,
                            const float* __restrict__ scale, int N, int C,
                            long S, float cr, float beta,
                            float* __restrict__ dx) {
  constexpr int pre = (SZ - 1) / 2;
  constexpr int ctr = SZ - 1 - pre;  // center channel's ring slot (-lo0)
  const long total = (long)N * S;
  const long half = (total + 1) / 2;
  GRID_STRIDE(i, half) {
    const long idx0 = i;
    const long idx1 = i + half;
    const bool two = idx1 < total;
    const int n0 = (int)(idx0 / S);
    const int n1 = two ? (int)(idx1 / S) : n0;
    const long b0 = (long)n0 * C * S + (idx0 - (long)n0 * S);
    const long b1 = two ? (long)n1 * C * S + (idx1 - (long)n1 * S) : b0;
    // rings hold channels [c-ctr, c+pre-1] at top of iter c
    float rr0[SZ - 1], rr1[SZ - 1], tt0[SZ - 1], tt1[SZ - 1];
    float a0 = 0.f, a1 = 0.f;
#pragma unroll
    for (int j = 0; j < SZ - 1; ++j) rr0[j] = rr1[j] = tt0[j] = tt1[j] = 0.f;
#pragma unroll
    for (int ch = 0; ch < pre; ++ch)    // prologue: channels 0..pre-1
      if (ch < C) {
        const long i0 = b0 + (long)ch * S, i1 = b1 + (long)ch * S;
        const float d0 = dy[i0], d1 = dy[i1];
        const float s0 = scale[i0], s1 = scale[i1];
        const float r0 = d0 * y[i0] / s0, r1 = d1 * y[i1] / s1;
        rr0[ch + ctr] = r0;
        tt0[ch + ctr] = d0 * __powf(s0, -beta);
        a0 += r0;
        rr1[ch + ctr] = r1;
        tt1[ch + ctr] = d1 * __powf(s1, -beta);
        a1 += r1;
      }
    for (int c = 0; c < C; ++c) {
      const int head = c + pre;
      float rh0 = 0.f, rh1 = 0.f, th0 = 0.f, th1 = 0.f;
      if (head < C) {
        const long i0 = b0 + (long)head * S, i1 = b1 + (long)head * S;
        const float d0 = dy[i0], d1 = dy[i1];
        const float s0 = scale[i0], s1 = scale[i1];
        rh0 = d0 * y[i0] / s0;
        th0 = d0 * __powf(s0, -beta);
        rh1 = d1 * y[i1] / s1;
        th1 = d1 * __powf(s1, -beta);
      }
      a0 += rh0;
      a1 += rh1;
      const long o0 = b0 + (long)c * S;
      dx[o0] = tt0[ctr] - cr * x[o0] * a0;
      if (two) {
        const long o1 = b1 + (long)c * S;
        dx[o1] = tt1[ctr] - cr * x[o1] * a1;
      }
      a0 -= rr0[0];
      a1 -= rr1[0];
#pragma unroll
      for (int j = 0; j + 1 < SZ - 1; ++j) {
        rr0[j] = rr0[j + 1]; rr1[j] = rr1[j + 1];
        tt0[j] = tt0[j + 1]; tt1[j] = tt1[j + 1];
      }
      rr0[SZ - 2] = rh0; rr1[SZ - 2] = rh1;
      tt0[SZ - 2] = th0; tt1[SZ - 2] = th1;
    }
  }
}

void lrn_bwd(hipStream_t s, const float* x, const float* y, const float* dy,
             const float* scale, int N, int C, int H, int W, int size,
             float alpha, float beta, float* dx) {
  const long S = (long)H * W;
  PerfScope perf(PERF_CLASS("lrn"), s, 0, 20.0 * N * C * S);
  if (size == 5) {
    hipLaunchKernelGGL(k_lrn_bwd_t<5>, dim3(nblocks((N * S + 1) / 2)),
                       dim3(TPB), 0, s, x, y, dy, scale, N, C, S,
                       2.f * alpha * beta / size, beta, dx);
    return;
  }
  hipLaunchKernelGGL(k_lrn_bwd, dim3(nblocks(N * S)), dim3(TPB), 0, s, x, y,
                     dy, scale, N, C, S, size, 2.f * alpha * beta / size,
                     beta, dx);
}

// ------------------------------------------------------------ softmax
// one wave per sample row (C up to 1000): wave-parallel max/sum reductions.
// Non-finite inputs PROPAGATE as NaN exactly like the CPU path: a NaN or
// +inf logit makes the whole row NaN (CPU: exp(NaN-mx)=NaN contaminates the
// sum), while -inf logits are benign (exp(-inf-mx)=0).  fmaxf silently
// drops NaN operands and __expf may launder NaN, so the contamination is
// detected explicitly — a divergence here once masked an all-NaN logits
// tensor as a clean uniform distribution.
__global__ void k_softmax_fwd(const float* __restrict__ x, int outer, int C,
                              int inner, float* __restrict__ y) {
  const long rows = (long)outer * inner;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wpb = blockDim.x >> 6;
  for (long r = (long)blockIdx.x * wpb + wave; r < rows;
       r += (long)gridDim.x * wpb) {
    const int o = (int)(r / inner);
    const int sp = (int)(r - (long)o * inner);
    const float* xp = x + (long)o * C * inner + sp;
    float* yp = y + (long)o * C * inner + sp;
    float mx = -3.402823466e38f;
    int bad = 0;  // any NaN / +inf in the row
    for (int c = lane; c < C; c += 64) {
      const float v = xp[(long)c * inner];
      mx = fmaxf(mx, v);
      bad |= (v != v) | (v > 3.402823466e38f);
    }
    for (int off = 32; off > 0; off >>= 1) {
      mx = fmaxf(mx, __shfl_xor(mx, off, 64));
      bad |= __shfl_xor(bad, off, 64);
    }
    float sum = 0.f;
    for (int c = lane; c < C; c += 64) {
      const float e = __expf(xp[(long)c * inner] - mx);
      yp[(long)c * inner] = e;
      sum += e;
    }
    for (int off = 32; off > 0; off >>= 1) sum += __shfl_xor(sum, off, 64);
    const float inv = 1.f / sum;
    const float fix = bad ? __builtin_nanf("") : 0.f;
    for (int c = lane; c < C; c += 64)
      yp[(long)c * inner] = yp[(long)c * inner] * inv + fix;
  }
}
void softmax_fwd(hipStream_t s, const float* x, int outer, int C, int inner,
                 float* prob) {
  const long rows = (long)outer * inner;
  PerfScope perf(PERF_CLASS("softmax"), s, 0, 12.0 * rows * C);
  const int blocks = (int)std::min<long>((rows + 3) / 4, 2048);
  hipLaunchKernelGGL(k_softmax_fwd, dim3(blocks), dim3(TPB), 0, s, x, outer,
                     C, inner, prob);
}

// two-stage deterministic loss sum: block partials then final add
__global__ void k_sm_loss(const float* __restrict__ prob,
                          const float* __restrict__ label, int outer, int C,
                          int inner, float norm, float* __restrict__ loss) {
  const long rows = (long)outer * inner;
  double acc = 0;
  for (long r = blockIdx.x * (long)blockDim.x + threadIdx.x; r < rows;
       r += (long)gridDim.x * blockDim.x) {
    const int o = (int)(r / inner);
    const int sp = (int)(r - (long)o * inner);
    const int lv = (int)label[(long)o * inner + sp];
    const float p = prob[((long)o * C + lv) * inner + sp];
    // NaN prob must surface as NaN loss (CPU parity): fmaxf(NaN, FLT_MIN)
    // would clamp it to a clean finite number and hide the divergence
    if (p != p)
      acc += (double)p;
    else
      acc -= (double)__logf(fmaxf(p, 1.175494351e-38f));
  }
  __shared__ double sh[TPB];
  sh[threadIdx.x] = acc;
  __syncthreads();
  for (int off = TPB / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) sh[threadIdx.x] += sh[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    // single block launch: write final
    loss[blockIdx.x] = (float)(sh[0] / norm);
  }
}
void softmaxloss_fwd(hipStream_t s, const float* prob, const float* label,
                     int outer, int C, int inner, float* loss_out) {
  PerfScope perf(PERF_CLASS("softmax"), s, 0, 8.0 * outer * inner);
  hipLaunchKernelGGL(k_sm_loss, dim3(1), dim3(TPB), 0, s, prob, label,
                     outer, C, inner, (float)((long)outer * inner),
                     loss_out);
}

__global__ void k_sm_loss_bwd(const float* __restrict__ prob,
                              const float* __restrict__ label, int outer,
                              int C, int inner, float scale,
                              float* __restrict__ dx) {
  const long total = (long)outer * C * inner;
  GRID_STRIDE(i, total) {
    const long o = i / ((long)C * inner);
    const long rem = i - o * C * inner;
    const int c = (int)(rem / inner);
    const int sp = (int)(rem - (long)c * inner);
    const int lv = (int)label[o * inner + sp];
    dx[i] = (prob[i] - (c == lv ? 1.f : 0.f)) * scale;
  }
}
void softmaxloss_bwd(hipStream_t s, const float* prob, const float* label,
                     int outer, int C, int inner, float scale, float* dx) {
  const long total = (long)outer * C * inner;
  PerfScope perf(PERF_CLASS("softmax"), s, 0, 8.0 * total);
  hipLaunchKernelGGL(k_sm_loss_bwd, dim3(nblocks(total, 4)), dim3(TPB), 0,
                     s, prob, label, outer, C, inner, scale, dx);
}

// ------------------------------------------------------- row/col sums
__global__ void k_colsum(const float* __restrict__ A, long M, long N,
                         float* __restrict__ out) {
  GRID_STRIDE(n, N) {
    double acc = 0;
    for (long m = 0; m < M; ++m) acc += A[m * N + n];
    out[n] = (float)acc;
  }
}
void colsum(hipStream_t s, const float* A, long M, long N, float* out) {
  PerfScope perf(PERF_CLASS("reduce"), s, 0, 4.0 * M * N);
  hipLaunchKernelGGL(k_colsum, dim3(nblocks(N)), dim3(TPB), 0, s, A, M, N,
                     out);
}

// ------------------------------------------------------------ eltwise etc.
__global__ void k_axpby(long n, float a, const float* __restrict__ x,
                        float b, float* __restrict__ y) {
  GRID_STRIDE(i, n) y[i] = a * x[i] + b * y[i];
}
void axpby(hipStream_t s, long n, float a, const float* x, float b,
           float* y) {
  PerfScope perf(PERF_CLASS("eltwise"), s, 0, 12.0 * n);
  hipLaunchKernelGGL(k_axpby, dim3(nblocks(n, 8)), dim3(TPB), 0, s, n, a, x,
                     b, y);
}

__global__ void k_copy(long n4, const f4* __restrict__ x,
                       f4* __restrict__ y) {
  VEC_GRID(i, n4) y[i] = x[i];
}
void copy(hipStream_t s, long n, const float* x, float* y) {
  if (x == y) return;
  PerfScope perf(PERF_CLASS("eltwise"), s, 0, 8.0 * n);
  const long n4 = (n + 3) / 4;
  hipLaunchKernelGGL(k_copy, dim3(nblocks(n4, 4)), dim3(TPB), 0, s, n4,
                     (const f4*)x, (f4*)y);
}

__global__ void k_set(long n, float v, float* __restrict__ y) {
  GRID_STRIDE(i, n) y[i] = v;
}
void set_const(hipStream_t s, long n, float v, float* y) {
  hipLaunchKernelGGL(k_set, dim3(nblocks(n, 8)), dim3(TPB), 0, s, n, v, y);
}

__global__ void k_add3(long n4, const f4* __restrict__ a,
                       const f4* __restrict__ b, int frelu,
                       f4* __restrict__ y) {
  VEC_GRID(i, n4) {
    f4 v = a[i] + b[i];
    if (frelu) {
#pragma unroll
      for (int j = 0; j < 4; ++j) v[j] = fmaxf(v[j], 0.f);
    }
    y[i] = v;
  }
}
void add3(hipStream_t s, long n, const float* a, const float* b, float* y,
          int fuse_relu) {
  PerfScope perf(PERF_CLASS("eltwise"), s, 0, 12.0 * n);
  const long n4 = (n + 3) / 4;
  hipLaunchKernelGGL(k_add3, dim3(nblocks(n4, 4)), dim3(TPB), 0, s, n4,
                     (const f4*)a, (const f4*)b, fuse_relu, (f4*)y);
}

__global__ void k_acc(long n4, const f4* __restrict__ x,
                      f4* __restrict__ y) {
  VEC_GRID(i, n4) y[i] += x[i];
}
void acc(hipStream_t s, long n, const float* x, float* y) {
  PerfScope perf(PERF_CLASS("eltwise"), s, 0, 12.0 * n);
  const long n4 = (n + 3) / 4;
  hipLaunchKernelGGL(k_acc, dim3(nblocks(n4, 4)), dim3(TPB), 0, s, n4,
                     (const f4*)x, (f4*)y);
}

__global__ void k_concat_fwd(const float* __restrict__ x, int N, int Cs,
                             long S, int Cd, int c_off,
                             float* __restrict__ y) {
  const long total = (long)N * Cs * S;
  GRID_STRIDE(i, total) {
    const int n = (int)(i / ((long)Cs * S));
    const long rem = i - (long)n * Cs * S;
    y[((long)n * Cd + c_off) * S + rem] = x[i];
  }
}
void concat_fwd(hipStream_t s, const float* x, int N, int Cs, long S, int Cd,
                int c_off, float* y) {
  const long total = (long)N * Cs * S;
  PerfScope perf(PERF_CLASS("concat"), s, 0, 8.0 * total);
  hipLaunchKernelGGL(k_concat_fwd, dim3(nblocks(total, 4)), dim3(TPB), 0, s,
                     x, N, Cs, S, Cd, c_off, y);
}

__global__ void k_concat_bwd(const float* __restrict__ dy, int N, int Cs,
                             long S, int Cd, int c_off,
                             float* __restrict__ dx) {
  const long total = (long)N * Cs * S;
  GRID_STRIDE(i, total) {
    const int n = (int)(i / ((long)Cs * S));
    const long rem = i - (long)n * Cs * S;
    dx[i] = dy[((long)n * Cd + c_off) * S + rem];
  }
}
void concat_bwd(hipStream_t s, const float* dy, int N, int Cs, long S,
                int Cd, int c_off, float* dx) {
  const long total = (long)N * Cs * S;
  PerfScope perf(PERF_CLASS("concat"), s, 0, 8.0 * total);
  hipLaunchKernelGGL(k_concat_bwd, dim3(nblocks(total, 4)), dim3(TPB), 0, s,
                     dy, N, Cs, S, Cd, c_off, dx);
}

// ------------------------------------------------------------ dropout/rng
__device__ __forceinline__ uint64_t d_splitmix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}
__device__ __forceinline__ float d_u01(uint64_t h) {
  return (float)((h >> 11) * (1.0 / 9007199254740992.0));
}

__global__ void k_dropout_fwd(const float* __restrict__ x, long n,
                              uint64_t key, float threshold, float scale,
                              float* __restrict__ y,
                              uint8_t* __restrict__ mask) {
  GRID_STRIDE(i, n) {
    const uint8_t m = d_u01(d_splitmix64(key ^ (uint64_t)i)) >= threshold;
    mask[i] = m;
    y[i] = x[i] * m * scale;
  }
}
void dropout_fwd(hipStream_t s, const float* x, long n, uint64_t seed,
                 uint64_t counter, float threshold, float scale, float* y,
                 uint8_t* mask) {
  PerfScope perf(PERF_CLASS("dropout"), s, 0, 9.0 * n);
  (void)counter;  // caller passes the pre-mixed key via `seed`
  hipLaunchKernelGGL(k_dropout_fwd, dim3(nblocks(n, 4)), dim3(TPB), 0, s, x,
                     n, seed, threshold, scale, y, mask);
}

__global__ void k_dropout_bwd(const float* __restrict__ dy,
                              const uint8_t* __restrict__ mask, long n,
                              float scale, float* __restrict__ dx) {
  GRID_STRIDE(i, n) dx[i] = dy[i] * mask[i] * scale;
}
void dropout_bwd(hipStream_t s, const float* dy, const uint8_t* mask, long n,
                 float scale, float* dx) {
  PerfScope perf(PERF_CLASS("dropout"), s, 0, 9.0 * n);
  hipLaunchKernelGGL(k_dropout_bwd, dim3(nblocks(n, 4)), dim3(TPB), 0, s,
                     dy, mask, n, scale, dx);
}

// ------------------------------------------------------------ SGD
__global__ void k_sgd(long n4, f4* __restrict__ g, f4* __restrict__ w,
                      f4* __restrict__ h, float mom, float lr, float decay,
                      float gscale) {
  VEC_GRID(i, n4) {
    f4 gv = g[i], wv = w[i], hv = h[i];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float gi = gv[j] * gscale + decay * wv[j];
      gi = hv[j] = mom * hv[j] + lr * gi;
      wv[j] -= gi;
    }
    w[i] = wv;
    h[i] = hv;
    g[i] = f4{0.f, 0.f, 0.f, 0.f};
  }
}
void sgd_update(hipStream_t s, long n, float* g, float* w, float* h,
                float mom, float lr, float decay, float gscale) {
  // arena offsets are 16-float aligned and blob memory is 64B-padded, so
  // the float4 body may run over the pad (pad floats update to garbage in
  // h; g pad stays 0; w pad unused) — correct for all live elements
  PerfScope perf(PERF_CLASS("sgd"), s, 0, 20.0 * n);
  const long n4 = (n + 3) / 4;
  hipLaunchKernelGGL(k_sgd, dim3(nblocks(n4, 4)), dim3(TPB), 0, s, n4,
                     (f4*)g, (f4*)w, (f4*)h, mom, lr, decay, gscale);
}

// segmented fused SGD: one launch covers a whole bucket's params.
// segs: sorted arena offsets; per-seg weight base pointer and lr/decay
// MULTIPLIERS (uploaded once — the per-iteration lr/decay ride in as
// kernel args, so no H2D traffic in the iteration loop).
// Elements in inter-param padding update harmlessly (their g is 0).
__global__ void k_sgd_seg(long lo, long hi, float* __restrict__ g_arena,
                          float* __restrict__ h_arena,
                          const long* __restrict__ seg_off,
                          float* const* __restrict__ w_ptrs,
                          const float* __restrict__ lr_mults,
                          const float* __restrict__ decay_mults, int nseg,
                          float mom, float lr, float decay, float gscale) {
  for (long i = lo + blockIdx.x * (long)blockDim.x + threadIdx.x; i < hi;
       i += (long)gridDim.x * blockDim.x) {
    // binary search: largest k with seg_off[k] <= i
    int a = 0, b = nseg - 1;
    while (a < b) {
      const int m = (a + b + 1) >> 1;
      if (seg_off[m] <= i)
        a = m;
      else
        b = m - 1;
    }
    float* w = w_ptrs[a] + (i - seg_off[a]);
    float gi = g_arena[i] * gscale + (decay * decay_mults[a]) * *w;
    gi = h_arena[i] = mom * h_arena[i] + (lr * lr_mults[a]) * gi;
    *w -= gi;
    g_arena[i] = 0.f;
  }
}
void sgd_update_segmented(hipStream_t s, long lo, long hi, float* g_arena,
                          float* h_arena, const long* seg_off,
                          float* const* w_ptrs, const float* lr_mults,
                          const float* decay_mults, int nseg, float mom,
                          float lr, float decay, float gscale) {
  PerfScope perf(PERF_CLASS("sgd"), s, 0, 20.0 * (hi - lo));
  hipLaunchKernelGGL(k_sgd_seg, dim3(nblocks(hi - lo, 4)), dim3(TPB), 0, s,
                     lo, hi, g_arena, h_arena, seg_off, w_ptrs, lr_mults,
                     decay_mults, nseg, mom, lr, decay, gscale);
}

// ---------------------------------------------------- LMDB transform
// uint8 -> fp32 crop/mirror/mean/scale (reference data_transformer.cu:
// 14-100): one thread per output element; geo = per-image (ho, wo,
// mirror); mean subtracted in SOURCE coordinates (mean_file semantics).
__global__ void k_transform_u8(const uint8_t* __restrict__ in, int N,
                               int C, int inH, int inW, int outH, int outW,
                               const int* __restrict__ geo,
                               const float* __restrict__ mean,
                               int mean_mode, float scale,
                               float* __restrict__ out) {
  const long total = (long)N * C * outH * outW;
  GRID_STRIDE(i, total) {
    const int x = (int)(i % outW);
    const int y = (int)((i / outW) % outH);
    const int c = (int)((i / ((long)outW * outH)) % C);
    const int n = (int)(i / ((long)outW * outH * C));
    const int ho = geo[3 * n], wo = geo[3 * n + 1], mir = geo[3 * n + 2];
    const int sx = wo + (mir ? outW - 1 - x : x);
    const long src = (((long)n * C + c) * inH + ho + y) * inW + sx;
    float m = 0.f;
    if (mean_mode == 1)
      m = mean[c];
    else if (mean_mode == 2)
      m = mean[((long)c * inH + ho + y) * inW + sx];
    out[i] = ((float)in[src] - m) * scale;
  }
}
void transform_u8(hipStream_t s, const uint8_t* in, int N, int C, int inH,
                  int inW, int outH, int outW, const int* geo,
                  const float* mean, int mean_mode, float scale,
                  float* out) {
  const long total = (long)N * C * outH * outW;
  PerfScope perf(PERF_CLASS("data"), s, 0, 5.0 * total);
  hipLaunchKernelGGL(k_transform_u8, dim3(nblocks(total, 4)), dim3(TPB), 0,
                     s, in, N, C, inH, inW, outH, outW, geo, mean,
                     mean_mode, scale, out);
}

// ------------------------------------------------------------ synthetic
__global__ void k_fill_uniform(long n, uint64_t key, float lo, float hi,
                               float* __restrict__ y) {
  GRID_STRIDE(i, n)
  y[i] = lo + (hi - lo) * d_u01(d_splitmix64(key ^ (uint64_t)i));
}
void fill_uniform(hipStream_t s, long n, uint64_t seed, uint64_t counter,
                  float lo, float hi, float* y) {
  PerfScope perf(PERF_CLASS("data"), s, 0, 4.0 * n);
  (void)counter;
  hipLaunchKernelGGL(k_fill_uniform, dim3(nblocks(n, 4)), dim3(TPB), 0, s,
                     n, seed, lo, hi, y);
}

__global__ void k_fill_labels(long n, uint64_t key, int classes,
                              float* __restrict__ y) {
  GRID_STRIDE(i, n)
  y[i] = (float)(d_splitmix64(key ^ 0xABCDull ^ (uint64_t)i) %
                 (uint64_t)classes);
}
void fill_labels(hipStream_t s, long n, uint64_t seed, uint64_t counter,
                 int classes, float* y) {
  (void)counter;
  hipLaunchKernelGGL(k_fill_labels, dim3(nblocks(n)), dim3(TPB), 0, s, n,
                     seed, classes, y);
}

}  // namespace gpu
}  // namespace camd
