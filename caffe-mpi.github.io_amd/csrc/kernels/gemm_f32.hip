// gemm_f32.hip — hand-written fp32 MFMA GEMM for gfx950 (CDNA4).
//
// The one kernel family that carries the conv/IP contractions (SURVEY.md
// §8a a1-a4).  Uses the f32-input MFMA `v_mfma_f32_32x32x2_f32`
// (exact f32, 157 TF chip peak = the f32 vector peak; no xf32 on gfx950 —
// cdna_hip_programming.md §3).  Structure:
//   128×128 block tile, 256 threads = 4 waves in a 2×2 grid, each wave owns
//   a 64×64 sub-tile as 2×2 MFMA 32×32 accumulators (4 independent
//   accumulators per wave reach the 64-cyc issue rate with one wave/SIMD).
//   BK=32 K-steps staged via registers into double-buffered LDS
//   [BK][128+1] (pad +1 → conflict-free b32 reads/writes), T14-style
//   issue-early/write-late (cdna_hip_programming.md §6 G15).
// All four transpose combos; α/β; fused epilogue: per-row or per-col bias,
// optional NCHW scatter (conv output goes straight to N-major layout, no
// separate bias/copy pass).  Deterministic split-K for the
// reduction-over-batch wgrad shapes (few output tiles, huge K): partial
// slabs + fixed-order reduce — no atomics (SURVEY.md §7 hard part (b)).
#include <hip/hip_runtime.h>

#include "../layers.hpp"

namespace camd {
namespace gpu {

using f32x16 = __attribute__((ext_vector_type(16))) float;

constexpr int BM = 128, BN = 128, BK = 32;
constexpr int LDA_S = BM + 1;  // LDS row stride (floats)
constexpr int LDB_S = BN + 1;

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
  // split-K
  float* slab;  // partials [SK][M][N] when SK>1
  int SK;
};

// stage op(A) tile rows [m0,m0+BM) x [k0,k0+BK) into regs (16 floats)
template <bool TRANS>
__device__ __forceinline__ void stage_a_load(const GemmArgs& g, long m0,
                                             long k0, float (&r)[16]) {
  const int t = threadIdx.x;
  if (!TRANS) {
    // A[M][K]: thread reads A[m0 + (t&127)][k0 + (t>>7)*16 + j]
    const long m = m0 + (t & 127);
    const long kb = k0 + (t >> 7) * 16;
    const float* p = g.A + m * g.lda + kb;
    const bool mok = m < g.M;
#pragma unroll
    for (int j = 0; j < 16; ++j)
      r[j] = (mok && kb + j < g.K) ? p[j] : 0.f;
  } else {
    // A stored [K][M]; op(A)(m,k)=A[k][m]: read A[k0 + (t&31)][m0+(t>>5)*16+j]
    const long k = k0 + (t & 31);
    const long mb = m0 + (t >> 5) * 16;
    const float* p = g.A + k * g.lda + mb;
    const bool kok = k < g.K;
#pragma unroll
    for (int j = 0; j < 16; ++j)
      r[j] = (kok && mb + j < g.M) ? p[j] : 0.f;
  }
}

template <bool TRANS>
__device__ __forceinline__ void stage_a_write(float* As,
                                              const float (&r)[16]) {
  const int t = threadIdx.x;
  if (!TRANS) {
    const int m = t & 127;
    const int kb = (t >> 7) * 16;
#pragma unroll
    for (int j = 0; j < 16; ++j) As[(kb + j) * LDA_S + m] = r[j];
  } else {
    const int k = t & 31;
    const int mb = (t >> 5) * 16;
#pragma unroll
    for (int j = 0; j < 16; ++j) As[k * LDA_S + mb + j] = r[j];
  }
}

template <bool TRANS>
__device__ __forceinline__ void stage_b_load(const GemmArgs& g, long n0,
                                             long k0, float (&r)[16]) {
  const int t = threadIdx.x;
  if (!TRANS) {
    // B[K][N]: read B[k0 + (t&31)][n0 + (t>>5)*16 + j]
    const long k = k0 + (t & 31);
    const long nb = n0 + (t >> 5) * 16;
    const float* p = g.B + k * g.ldb + nb;
    const bool kok = k < g.K;
#pragma unroll
    for (int j = 0; j < 16; ++j)
      r[j] = (kok && nb + j < g.N) ? p[j] : 0.f;
  } else {
    // B stored [N][K]; op(B)(k,n)=B[n][k]: read B[n0+(t&127)][k0+(t>>7)*16+j]
    const long n = n0 + (t & 127);
    const long kb = k0 + (t >> 7) * 16;
    const float* p = g.B + n * g.ldb + kb;
    const bool nok = n < g.N;
#pragma unroll
    for (int j = 0; j < 16; ++j)
      r[j] = (nok && kb + j < g.K) ? p[j] : 0.f;
  }
}

template <bool TRANS>
__device__ __forceinline__ void stage_b_write(float* Bs,
                                              const float (&r)[16]) {
  const int t = threadIdx.x;
  if (!TRANS) {
    const int k = t & 31;
    const int nb = (t >> 5) * 16;
#pragma unroll
    for (int j = 0; j < 16; ++j) Bs[k * LDB_S + nb + j] = r[j];
  } else {
    const int n = t & 127;
    const int kb = (t >> 7) * 16;
#pragma unroll
    for (int j = 0; j < 16; ++j) Bs[(kb + j) * LDB_S + n] = r[j];
  }
}

template <bool TA, bool TB, bool SPLITK>
__launch_bounds__(256, 2) __global__ void k_gemm_f32(GemmArgs g) {
  __shared__ float As[2][BK * LDA_S];
  __shared__ float Bs[2][BK * LDB_S];

  const long tile_n = blockIdx.x;
  const long tile_m = blockIdx.y;
  const long m0 = tile_m * BM, n0 = tile_n * BN;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wr = wave >> 1, wc = wave & 1;  // 2x2 wave grid
  const int row_in = lane & 31;             // MFMA row/col index
  const int ksel = lane >> 5;               // which of the 2 K elems

  // K range for this block (split-K slice)
  long k_lo = 0, k_hi = g.K;
  if (SPLITK) {
    const int sk = blockIdx.z;
    k_lo = g.K * sk / g.SK;
    k_hi = g.K * (sk + 1) / g.SK;
    // align slice starts to BK so staging tiles stay aligned
    k_lo = k_lo / BK * BK;
    k_hi = (blockIdx.z == g.SK - 1) ? g.K : k_hi / BK * BK;
    if (k_lo >= k_hi) return;
  }
  const long ntiles = (k_hi - k_lo + BK - 1) / BK;

  f32x16 acc00 = {}, acc01 = {}, acc10 = {}, acc11 = {};
  float ra[16], rb[16];

  stage_a_load<TA>(g, m0, k_lo, ra);
  stage_b_load<TB>(g, n0, k_lo, rb);
  stage_a_write<TA>(As[0], ra);
  stage_b_write<TB>(Bs[0], rb);
  __syncthreads();

  int cur = 0;
  for (long t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles) {  // issue next tile's global loads early (T14)
      stage_a_load<TA>(g, m0, k_lo + (t + 1) * BK, ra);
      stage_b_load<TB>(g, n0, k_lo + (t + 1) * BK, rb);
    }
    const float* Ab = As[cur];
    const float* Bb = Bs[cur];
#pragma unroll
    for (int kk = 0; kk < BK; kk += 2) {
      const int krow = kk + ksel;
      const float a0 = Ab[krow * LDA_S + wr * 64 + row_in];
      const float a1 = Ab[krow * LDA_S + wr * 64 + 32 + row_in];
      const float b0 = Bb[krow * LDB_S + wc * 64 + row_in];
      const float b1 = Bb[krow * LDB_S + wc * 64 + 32 + row_in];
      acc00 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc00, 0, 0, 0);
      acc01 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc01, 0, 0, 0);
      acc10 = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc10, 0, 0, 0);
      acc11 = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc11, 0, 0, 0);
    }
    __syncthreads();
    if (t + 1 < ntiles) {
      stage_a_write<TA>(As[cur ^ 1], ra);
      stage_b_write<TB>(Bs[cur ^ 1], rb);
    }
    __syncthreads();
    cur ^= 1;
  }

  // ---- epilogue: acc reg r -> (row, col)
  // C/D map for 32x32 MFMA: col = lane&31, row = (r&3) + 8*(r>>2) + 4*(lane>>5)
  // (rule 20: no runtime-indexed vector arrays — four explicit calls)
  auto epi_tile = [&](const f32x16& a, int ti, int tj) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const long row =
          m0 + wr * 64 + ti * 32 + ((r & 3) + 8 * (r >> 2) + 4 * ksel);
      const long col = n0 + wc * 64 + tj * 32 + row_in;
      if (row >= g.M || col >= g.N) continue;
      float v = g.alpha * a[r];
      if (SPLITK) {
        g.slab[((long)blockIdx.z * g.M + row) * g.N + col] = v;
        continue;
      }
      if (g.bias) v += g.bias[g.bias_per_col ? col : row];
      if (g.relu) v = fmaxf(v, 0.f);
      long off;
      if (g.spad > 0) {
        const long n = col / g.spad;
        const long sp = col - n * g.spad;
        if (sp >= g.S) continue;  // padding column
        off = n * g.n_stride + row * g.S + sp;
      } else {
        off = row * g.ldc + col;
      }
      if (g.beta != 0.f) v += g.beta * g.C[off];
      g.C[off] = v;
    }
  };
  epi_tile(acc00, 0, 0);
  epi_tile(acc01, 0, 1);
  epi_tile(acc10, 1, 0);
  epi_tile(acc11, 1, 1);
}

// fixed-order split-K reduce: C[i] = Σ_sk slab[sk][i]
__global__ void k_splitk_reduce(const float* __restrict__ slab, long MN,
                                int SK, float* __restrict__ C) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < MN;
       i += (long)gridDim.x * blockDim.x) {
    float acc = 0.f;
    for (int s = 0; s < SK; ++s) acc += slab[(long)s * MN + i];
    C[i] = acc;
  }
}

void gemm(hipStream_t s, bool transA, bool transB, long M, long N, long K,
          float alpha, const float* A, long lda, const float* B, long ldb,
          float beta, float* C, long ldc, const GemmEpi* epi) {
  CHECK_GT_(M, 0);
  CHECK_GT_(N, 0);
  CHECK_GT_(K, 0);
  GemmArgs g{};
  g.A = A;
  g.B = B;
  g.C = C;
  g.M = M;
  g.N = N;
  g.K = K;
  g.lda = lda;
  g.ldb = ldb;
  g.ldc = ldc;
  g.alpha = alpha;
  g.beta = beta;
  if (epi) {
    g.spad = epi->spad;
    g.S = epi->S;
    g.n_stride = epi->n_stride;
    g.bias = epi->bias;
    g.bias_per_col = epi->bias_per_col;
    g.relu = epi->relu;
  }
  const long tm = (M + BM - 1) / BM, tn = (N + BN - 1) / BN;
  // split-K when the output grid cannot fill the chip (wgrad shapes);
  // restricted to plain-C calls (no epilogue, beta=0) — the fused-epilogue
  // GEMMs (conv/IP forward) have huge N and never need it
  int SK = 1;
  if (!epi && beta == 0.f && tm * tn < 512 && K > 4 * BK) {
    SK = (int)std::min<long>(
        {512 / (tm * tn) + 1, (K + 4 * BK - 1) / (4 * BK), 64});
  }
  PerfScope perf("gemm", s, 2.0 * M * N * K,
                 4.0 * (M * K + N * K + M * N));
  dim3 grid((unsigned)tn, (unsigned)tm, (unsigned)SK);
  dim3 block(256);
  if (SK > 1) {
    CHECK_(!epi && beta == 0.f) << "split-K path supports plain C only";
    float* slab = (float*)Workspace::get_global().get(
        10, sizeof(float) * (size_t)SK * M * N);
    g.slab = slab;
    g.SK = SK;
    if (!transA && !transB)
      hipLaunchKernelGGL((k_gemm_f32<false, false, true>), grid, block, 0,
                         s, g);
    else if (!transA && transB)
      hipLaunchKernelGGL((k_gemm_f32<false, true, true>), grid, block, 0, s,
                         g);
    else if (transA && !transB)
      hipLaunchKernelGGL((k_gemm_f32<true, false, true>), grid, block, 0, s,
                         g);
    else
      hipLaunchKernelGGL((k_gemm_f32<true, true, true>), grid, block, 0, s,
                         g);
    const long MN = M * N;
    const int blocks = (int)std::min<long>((MN + 255) / 256, 2048);
    hipLaunchKernelGGL(k_splitk_reduce, dim3(blocks), dim3(256), 0, s, slab,
                       MN, SK, C);
    return;
  }
  if (!transA && !transB)
    hipLaunchKernelGGL((k_gemm_f32<false, false, false>), grid, block, 0, s,
                       g);
  else if (!transA && transB)
    hipLaunchKernelGGL((k_gemm_f32<false, true, false>), grid, block, 0, s,
                       g);
  else if (transA && !transB)
    hipLaunchKernelGGL((k_gemm_f32<true, false, false>), grid, block, 0, s,
                       g);
  else
    hipLaunchKernelGGL((k_gemm_f32<true, true, false>), grid, block, 0, s,
                       g);
}

}  // namespace gpu
}  // namespace camd
