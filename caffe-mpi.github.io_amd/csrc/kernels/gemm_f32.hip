// gemm_f32.hip — hand-written fp32 MFMA GEMM for gfx950 (CDNA4).
//
// The kernel family carrying the conv/IP contractions (SURVEY.md §8a
// a1-a4), built on the f32-input MFMA `v_mfma_f32_32x32x2_f32` (exact f32,
// 157.3 TF chip peak; no xf32 on gfx950 — cdna_hip_programming.md §3).
//
// Structure: 128×128 block tile, 256 threads = 4 waves (2×2), each wave a
// 64×64 sub-tile as 2×2 MFMA 32×32 accumulators; BK=32 K-steps staged
// through double-buffered LDS [BK][128+1] (+1 pad → conflict-free b32
// banking), T14 issue-early/write-late, vectorized float4 staging loads on
// the aligned in-bounds fast path, bijective XCD-aware block swizzle (T1).
//
// Operands can be NCHW *views* (GemmView): conv GEMMs read x / dY straight
// from NCHW layout — no transpose pass, and 1×1/s1 convolutions run with no
// im2col/col2im at all.  Fused epilogue: bias (per row/col), optional
// ReLU, NCHW scatter.  Deterministic split-K (partial slabs + fixed-order
// reduce, no atomics — SURVEY.md §7 hard part (b)) for the few-tile/huge-K
// wgrad shapes.
#include "gemm_common.hpp"

namespace camd {
namespace gpu {

constexpr int BM = 128, BN = 128, BK = 32;
constexpr int LDS_S = BM + 1;  // LDS row stride (floats), BM == BN

// bf16 tensor-core variant (gemm_bf16.hip): same call surface, selected
// by Engine::gemm_bf16 (mixed precision — fp32 storage, bf16 MFMA)
void gemm_launch_bf16(hipStream_t s, bool transA, bool transB, dim3 grid,
                      dim3 block, const GemmArgs& g, bool splitk,
                      bool wide);

// ---- staging: global -> regs (load) and regs -> LDS (write)
// A tile is held in LDS as As[k][m] (k-major); B as Bs[k][n].
template <bool TRANS>
__device__ __forceinline__ void stage_a_load(const GemmArgs& g, long m0,
                                             long k0, float (&r)[16]) {
  const int t = threadIdx.x;
  if (!TRANS) {
    // A[M][Kd] — contiguous along Kd (view axis allowed)
    read16(g.A, m0 + (t & 127), k0 + (t >> 7) * 16, g.lda, g.M, g.K, g.av,
           r);
  } else {
    // A stored [Kd][M] — contiguous along M (no view on this case)
    read16(g.A, k0 + (t & 31), m0 + (t >> 5) * 16, g.lda, g.K, g.M, g.av,
           r);
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
    for (int j = 0; j < 16; ++j) As[(kb + j) * LDS_S + m] = r[j];
  } else {
    const int k = t & 31;
    const int mb = (t >> 5) * 16;
#pragma unroll
    for (int j = 0; j < 16; ++j) As[k * LDS_S + mb + j] = r[j];
  }
}

template <bool TRANS>
__device__ __forceinline__ void stage_b_load(const GemmArgs& g, long n0,
                                             long k0, float (&r)[16]) {
  const int t = threadIdx.x;
  if (!TRANS) {
    // B[Kd][N] — contiguous along N (dY view in fwd/dgrad)
    read16(g.B, k0 + (t & 31), n0 + (t >> 5) * 16, g.ldb, g.K, g.N, g.bv,
           r);
  } else {
    // B stored [N][Kd] — contiguous along Kd (x/col view in wgrad)
    read16(g.B, n0 + (t & 127), k0 + (t >> 7) * 16, g.ldb, g.N, g.K, g.bv,
           r);
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
    for (int j = 0; j < 16; ++j) Bs[k * LDS_S + nb + j] = r[j];
  } else {
    const int n = t & 127;
    const int kb = (t >> 7) * 16;
#pragma unroll
    for (int j = 0; j < 16; ++j) Bs[(kb + j) * LDS_S + n] = r[j];
  }
}

template <bool TA, bool TB, bool SPLITK>
__launch_bounds__(256, 2) __global__ void k_gemm_f32(GemmArgs g) {
  // one LDS arena carved into As[2] | Bs[2]; the tail-wave reduction below
  // reuses it after the K-loop
  __shared__ float smem[4 * BK * LDS_S];
  auto As = [&](int buf) -> float* { return smem + buf * (BK * LDS_S); };
  auto Bs = [&](int buf) -> float* {
    return smem + (2 + buf) * (BK * LDS_S);
  };

  // bijective XCD-aware swizzle (cdna_hip_programming.md T1): contiguous
  // tile chunks per XCD so neighbouring tiles share L2-resident panels
  long flat = blockIdx.x;
  {
    const long nwg = g.tiles;
    const long q = nwg / 8, rr = nwg % 8;
    const long xcd = flat % 8, idx = flat / 8;
    flat = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  const long tile_m = flat / g.tn;
  const long tile_n = flat - tile_m * g.tn;
  const long m0 = tile_m * BM, n0 = tile_n * BN;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  // wave decomposition: 2x2 over the 128x128 tile normally; when M or N
  // fits in 64 the spare waves take K-ranges instead (intra-block split-K,
  // combined deterministically through LDS) so small-Cout layers keep all
  // 4 waves busy
  const int mwaves = g.M > 64 ? 2 : 1;
  const int nwaves = g.N > 64 ? 2 : 1;
  const int mn = mwaves * nwaves;
  const int kwaves = 4 / mn;
  const int wr = wave % mwaves;
  const int wc = (wave / mwaves) % nwaves;
  const int wk = wave / mn;
  const int row_in = lane & 31;             // MFMA row/col index
  const int ksel = lane >> 5;               // which of the 2 K elems

  // K range (split-K slice; starts BK-aligned so chunks stay 16-aligned)
  long k_lo = 0, k_hi = g.K;
  if (SPLITK) {
    const int sk = blockIdx.z;
    k_lo = g.K * sk / g.SK / BK * BK;
    k_hi = (sk == g.SK - 1) ? g.K : g.K * (sk + 1) / g.SK / BK * BK;
    if (k_lo >= k_hi) return;  // cannot happen for SK <= ceil(K/4BK)
  }
  const long ntiles = (k_hi - k_lo + BK - 1) / BK;

  f32x16 acc00 = {}, acc01 = {}, acc10 = {}, acc11 = {};
  float ra[16], rb[16];

  stage_a_load<TA>(g, m0, k_lo, ra);
  stage_b_load<TB>(g, n0, k_lo, rb);
  stage_a_write<TA>(As(0), ra);
  stage_b_write<TB>(Bs(0), rb);
  __syncthreads();

  int cur = 0;
  if (kwaves == 1) {
    for (long t = 0; t < ntiles; ++t) {
      if (t + 1 < ntiles) {  // issue next tile's global loads early (T14)
        stage_a_load<TA>(g, m0, k_lo + (t + 1) * BK, ra);
        stage_b_load<TB>(g, n0, k_lo + (t + 1) * BK, rb);
      }
      {
        const float* Ab = As(cur);
        const float* Bb = Bs(cur);
#pragma unroll
        for (int kk = 0; kk < BK; kk += 2) {
          const int krow = kk + ksel;
          const float a0 = Ab[krow * LDS_S + wr * 64 + row_in];
          const float a1 = Ab[krow * LDS_S + wr * 64 + 32 + row_in];
          const float b0 = Bb[krow * LDS_S + wc * 64 + row_in];
          const float b1 = Bb[krow * LDS_S + wc * 64 + 32 + row_in];
          acc00 =
              __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc00, 0, 0, 0);
          acc01 =
              __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc01, 0, 0, 0);
          acc10 =
              __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc10, 0, 0, 0);
          acc11 =
              __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc11, 0, 0, 0);
        }
      }
      // single barrier per K-tile: writing buf[cur^1] here is safe — its
      // last readers finished before the previous iteration's barrier; the
      // barrier below publishes these writes for the next iteration's reads
      if (t + 1 < ntiles) {
        stage_a_write<TA>(As(cur ^ 1), ra);
        stage_b_write<TB>(Bs(cur ^ 1), rb);
      }
      __syncthreads();
      cur ^= 1;
    }
  } else {
    // kwaves path: BOTH LDS buffers are live per iteration — the wave group
    // with (wk & 1) == p computes buffer p, so all 4 waves issue MFMAs
    // concurrently even when the output tile only needs 1 or 2 of them.
    // For kwaves == 4 the two waves sharing a buffer split the K-step in
    // half; the partials are summed in the LDS reduction below.
    const int grp = wk & 1;          // which buffer this wave computes
    const int khalf = wk >> 1;       // kwaves==4: kk half within the step
    const int kk0 = (kwaves == 4) ? khalf * (BK / 2) : 0;
    const int kk1 = (kwaves == 4) ? kk0 + BK / 2 : BK;
    // prologue already staged tile 0 into buf0; stage tile 1 into buf1
    if (1 < ntiles) {
      stage_a_load<TA>(g, m0, k_lo + 1 * BK, ra);
      stage_b_load<TB>(g, n0, k_lo + 1 * BK, rb);
      stage_a_write<TA>(As(1), ra);
      stage_b_write<TB>(Bs(1), rb);
    }
    __syncthreads();
    for (long p = 0; p < ntiles; p += 2) {
      const long t_next0 = p + 2, t_next1 = p + 3;
      if (t_next0 < ntiles) {
        stage_a_load<TA>(g, m0, k_lo + t_next0 * BK, ra);
        stage_b_load<TB>(g, n0, k_lo + t_next0 * BK, rb);
      }
      float ra2[16], rb2[16];
      if (t_next1 < ntiles) {
        stage_a_load<TA>(g, m0, k_lo + t_next1 * BK, ra2);
        stage_b_load<TB>(g, n0, k_lo + t_next1 * BK, rb2);
      }
      const long t_mine = p + grp;
      if (t_mine < ntiles) {
        const float* Ab = As(grp);
        const float* Bb = Bs(grp);
        for (int kk = kk0; kk < kk1; kk += 2) {
          const int krow = kk + ksel;
          const float a0 = Ab[krow * LDS_S + wr * 64 + row_in];
          const float a1 = Ab[krow * LDS_S + wr * 64 + 32 + row_in];
          const float b0 = Bb[krow * LDS_S + wc * 64 + row_in];
          const float b1 = Bb[krow * LDS_S + wc * 64 + 32 + row_in];
          acc00 =
              __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc00, 0, 0, 0);
          acc01 =
              __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc01, 0, 0, 0);
          acc10 =
              __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc10, 0, 0, 0);
          acc11 =
              __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc11, 0, 0, 0);
        }
      }
      __syncthreads();  // everyone done computing before rewriting buffers
      if (t_next0 < ntiles) {
        stage_a_write<TA>(As(0), ra);
        stage_b_write<TB>(Bs(0), rb);
      }
      if (t_next1 < ntiles) {
        stage_a_write<TA>(As(1), ra2);
        stage_b_write<TB>(Bs(1), rb2);
      }
      __syncthreads();
    }
  }

  if (kwaves > 1) {
    // combine the K-range partials: spare waves park their accumulators in
    // the (now idle) staging LDS; the wk==0 wave adds them in ascending wk
    // order — deterministic
    __syncthreads();  // all reads of the staging buffers are done
    if (wk > 0) {
      float* slot = smem + (long)(wave - mn) * 4096 + lane * 64;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        slot[r] = acc00[r];
        slot[16 + r] = acc01[r];
        slot[32 + r] = acc10[r];
        slot[48 + r] = acc11[r];
      }
    }
    __syncthreads();
    if (wk > 0) return;  // primaries finish the epilogue
    for (int k2 = 1; k2 < kwaves; ++k2) {
      const int src_wave = k2 * mn + (wc * mwaves + wr);
      const float* slot = smem + (long)(src_wave - mn) * 4096 + lane * 64;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        acc00[r] += slot[r];
        acc01[r] += slot[16 + r];
        acc10[r] += slot[32 + r];
        acc11[r] += slot[48 + r];
      }
    }
  }

  // ---- epilogue: acc reg r -> (row, col)
  // C/D map (32x32 MFMA): col = lane&31, row = (r&3)+8*(r>>2)+4*(lane>>5)
  auto epi_tile = [&](const f32x16& a, int ti, int tj) {
    // col is constant across the 16 regs — hoist the scatter division
    const long col = n0 + wc * 64 + tj * 32 + row_in;
    if (col >= g.N) return;
    long col_base = 0;  // n*n_stride + pix for scatter, col for plain
    bool col_ok = true;
    if (!SPLITK && g.spad > 0) {
      const long n = col / g.spad;
      const long sp = col - n * g.spad;
      col_ok = sp < g.S;
      long pix = sp;
      if (g.OWo > 0) {  // strided scatter (1x1/s2 dgrad)
        const int oh = (int)(sp / g.OWo);
        const int ow = (int)(sp - (long)oh * g.OWo);
        pix = ((long)oh * g.osh) * g.Wd + (long)ow * g.osw;
      }
      col_base = n * g.n_stride + pix;
    }
    if (!col_ok) return;
    const float cbias =
        (!SPLITK && g.bias && g.bias_per_col) ? g.bias[col] : 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const long row =
          m0 + wr * 64 + ti * 32 + ((r & 3) + 8 * (r >> 2) + 4 * ksel);
      if (row >= g.M) continue;
      float v = g.alpha * a[r];
      if (SPLITK) {
        g.slab[((long)blockIdx.z * g.M + row) * g.N + col] = v;
        continue;
      }
      if (g.bias) v += g.bias_per_col ? cbias : g.bias[row];
      if (g.relu) v = fmaxf(v, 0.f);
      const long off = g.spad > 0 ? col_base + row * g.Srow
                                  : row * g.ldc + col;
      if (g.beta != 0.f) v += g.beta * g.C[off];
      g.C[off] = v;
    }
  };
  epi_tile(acc00, 0, 0);
  epi_tile(acc01, 0, 1);
  epi_tile(acc10, 1, 0);
  epi_tile(acc11, 1, 1);
}

// ==================================================================
// k_gemm_slim — small-M tile variant (round 2): TM=32 with
// TN=256, or TM=64 with TN=128.  The 128x128 tile leaves most MFMA rows
// empty for the inception-branch and stage-1 shapes (Cout 16..64:
// GoogLeNet conv GEMMs measured 25% of peak, ResNet-50's
// 64x401408x576 41 TF); here each wave owns a 32(M) x 64(N) sub-tile
// with TWO 32x32 accumulators, waves tile (TM/32) x (4/(TM/32)) — every
// MFMA row is a real output row.  Structure otherwise mirrors k_gemm_f32
// (register T14 pipeline, one barrier per K-tile, same epilogue/split-K
// slabs); at TM=64 the LDS footprint (50 KB) admits 3 blocks/CU.
template <bool TA, bool TB, int TM, bool SPLITK>
__launch_bounds__(256, TM == 64 ? 3 : 2) __global__
    void k_gemm_slim(GemmArgs g) {
  constexpr int TN = TM == 32 ? 256 : 128;
  constexpr int AS_S = TM + 1;  // +1 pad: conflict-free b32 banking
  constexpr int BS_S = TN + 1;
  __shared__ float smem[2 * BK * AS_S + 2 * BK * BS_S];
  auto As = [&](int buf) -> float* { return smem + buf * (BK * AS_S); };
  auto Bs = [&](int buf) -> float* {
    return smem + 2 * BK * AS_S + buf * (BK * BS_S);
  };

  long flat = blockIdx.x;
  {
    const long nwg = g.tiles;
    const long q = nwg / 8, rr = nwg % 8;
    const long xcd = flat % 8, idx = flat / 8;
    flat = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  const long tile_m = flat / g.tn;
  const long tile_n = flat - tile_m * g.tn;
  const long m0 = tile_m * TM, n0 = tile_n * TN;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  constexpr int mw = TM / 32;
  const int wr = wave % mw;
  const int wc = wave / mw;
  const int row_in = lane & 31;
  const int ksel = lane >> 5;

  long k_lo = 0, k_hi = g.K;
  if (SPLITK) {
    const int sk = blockIdx.z;
    k_lo = g.K * sk / g.SK / BK * BK;
    k_hi = (sk == g.SK - 1) ? g.K : g.K * (sk + 1) / g.SK / BK * BK;
    if (k_lo >= k_hi) return;
  }
  const long ntiles = (k_hi - k_lo + BK - 1) / BK;

  f32x16 acc0 = {}, acc1 = {};
  float ra[16], rb[16], rb2[16];

  // staging roles (per K-tile):
  //  A !TA ([M][K] contig K): TM*BK floats — TM=32: threads 0..63 cover
  //    (m = t&31, kb = ((t>>5)&1)*16); TM=64: threads 0..127,
  //    (m = t&63, kb = ((t>>6)&1)*16)
  //  A TA ([K][M] contig M): thread (k = t&31, mb = (t>>5)*16 while
  //    mb < TM) — read16 zero-fills past M
  //  B !TB ([K][N] contig N): TN=256: (k = t&31, nb = (t>>5)*32), two
  //    chunks; TN=128: (k = t&31, nb = (t>>5)*16), one chunk
  //  B TB ([N][K] contig K): TN=256: (n = t, kb 0/16), two chunks over
  //    k; TN=128: (n = t&127, kb = (t>>7)*16), one chunk
  auto a_load = [&](long k0) {
    if (!TA) {
      const int m = TM == 32 ? (t & 31) : (t & 63);
      const int kb = TM == 32 ? ((t >> 5) & 1) * 16 : ((t >> 6) & 1) * 16;
      const bool mine = TM == 32 ? t < 64 : t < 128;
      if (mine) read16(g.A, m0 + m, k0 + kb, g.lda, g.M, g.K, g.av, ra);
    } else {
      const int k = t & 31;
      const int mb = (t >> 5) * 16;
      if (mb < TM)
        read16(g.A, k0 + k, m0 + mb, g.lda, g.K, g.M, g.av, ra);
    }
  };
  auto a_write = [&](float* A_) {
    if (!TA) {
      const int m = TM == 32 ? (t & 31) : (t & 63);
      const int kb = TM == 32 ? ((t >> 5) & 1) * 16 : ((t >> 6) & 1) * 16;
      const bool mine = TM == 32 ? t < 64 : t < 128;
      if (mine) {
#pragma unroll
        for (int j = 0; j < 16; ++j) A_[(kb + j) * AS_S + m] = ra[j];
      }
    } else {
      const int k = t & 31;
      const int mb = (t >> 5) * 16;
      if (mb < TM) {
#pragma unroll
        for (int j = 0; j < 16; ++j) A_[k * AS_S + mb + j] = ra[j];
      }
    }
  };
  auto b_load = [&](long k0) {
    if (!TB) {
      const int k = t & 31;
      if (TN == 256) {
        const int nb = (t >> 5) * 32;
        read16(g.B, k0 + k, n0 + nb, g.ldb, g.K, g.N, g.bv, rb);
        read16(g.B, k0 + k, n0 + nb + 16, g.ldb, g.K, g.N, g.bv, rb2);
      } else {
        const int nb = (t >> 5) * 16;
        read16(g.B, k0 + k, n0 + nb, g.ldb, g.K, g.N, g.bv, rb);
      }
    } else {
      if (TN == 256) {
        read16(g.B, n0 + t, k0, g.ldb, g.N, g.K, g.bv, rb);
        read16(g.B, n0 + t, k0 + 16, g.ldb, g.N, g.K, g.bv, rb2);
      } else {
        const int n = t & 127;
        const int kb = (t >> 7) * 16;
        read16(g.B, n0 + n, k0 + kb, g.ldb, g.N, g.K, g.bv, rb);
      }
    }
  };
  auto b_write = [&](float* B_) {
    if (!TB) {
      const int k = t & 31;
      if (TN == 256) {
        const int nb = (t >> 5) * 32;
#pragma unroll
        for (int j = 0; j < 16; ++j) {
          B_[k * BS_S + nb + j] = rb[j];
          B_[k * BS_S + nb + 16 + j] = rb2[j];
        }
      } else {
        const int nb = (t >> 5) * 16;
#pragma unroll
        for (int j = 0; j < 16; ++j) B_[k * BS_S + nb + j] = rb[j];
      }
    } else {
      if (TN == 256) {
#pragma unroll
        for (int j = 0; j < 16; ++j) {
          B_[j * BS_S + t] = rb[j];
          B_[(16 + j) * BS_S + t] = rb2[j];
        }
      } else {
        const int n = t & 127;
        const int kb = (t >> 7) * 16;
#pragma unroll
        for (int j = 0; j < 16; ++j) B_[(kb + j) * BS_S + n] = rb[j];
      }
    }
  };

  a_load(k_lo);
  b_load(k_lo);
  a_write(As(0));
  b_write(Bs(0));
  __syncthreads();

  int cur = 0;
  for (long tt = 0; tt < ntiles; ++tt) {
    if (tt + 1 < ntiles) {
      a_load(k_lo + (tt + 1) * BK);
      b_load(k_lo + (tt + 1) * BK);
    }
    {
      const float* Ab = As(cur);
      const float* Bb = Bs(cur);
#pragma unroll
      for (int kk = 0; kk < BK; kk += 2) {
        const int krow = kk + ksel;
        const float a0 = Ab[krow * AS_S + wr * 32 + row_in];
        const float b0 = Bb[krow * BS_S + wc * 64 + row_in];
        const float b1 = Bb[krow * BS_S + wc * 64 + 32 + row_in];
        acc0 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc0, 0, 0, 0);
        acc1 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc1, 0, 0, 0);
      }
    }
    if (tt + 1 < ntiles) {
      a_write(As(cur ^ 1));
      b_write(Bs(cur ^ 1));
    }
    __syncthreads();
    cur ^= 1;
  }

  auto epi_tile = [&](const f32x16& a, int tj) {
    const long col = n0 + wc * 64 + tj * 32 + row_in;
    if (col >= g.N) return;
    long col_base = 0;
    bool col_ok = true;
    if (!SPLITK && g.spad > 0) {
      const long n = col / g.spad;
      const long sp = col - n * g.spad;
      col_ok = sp < g.S;
      long pix = sp;
      if (g.OWo > 0) {
        const int oh = (int)(sp / g.OWo);
        const int ow = (int)(sp - (long)oh * g.OWo);
        pix = ((long)oh * g.osh) * g.Wd + (long)ow * g.osw;
      }
      col_base = n * g.n_stride + pix;
    }
    if (!col_ok) return;
    const float cbias =
        (!SPLITK && g.bias && g.bias_per_col) ? g.bias[col] : 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const long row =
          m0 + wr * 32 + ((r & 3) + 8 * (r >> 2) + 4 * ksel);
      if (row >= g.M) continue;
      float v = g.alpha * a[r];
      if (SPLITK) {
        g.slab[((long)blockIdx.z * g.M + row) * g.N + col] = v;
        continue;
      }
      if (g.bias) v += g.bias_per_col ? cbias : g.bias[row];
      if (g.relu) v = fmaxf(v, 0.f);
      const long off =
          g.spad > 0 ? col_base + row * g.Srow : row * g.ldc + col;
      if (g.beta != 0.f) v += g.beta * g.C[off];
      g.C[off] = v;
    }
  };
  epi_tile(acc0, 0);
  epi_tile(acc1, 1);
}

// ==================================================================
// k_gemm2 — glds (global_load_lds) staged fp32 MFMA GEMM, BK=64.
//
// Round-2 load-path redesign (cdna_hip_programming.md "glds vs register
// staging"): the v1 kernel is register-staged at 223 VGPRs / 2 waves per
// SIMD with ~50% issue stall; glds removes the ~32 staging VGPRs and 3/4
// of the LDS instructions, and BK=64 halves the barrier count.  LDS
// images are LINEAR in the exact lane order glds writes (wave-uniform
// base + lane*16B — the swizzle lives in the per-lane GLOBAL source
// address, guide rule 21):
//   A (TA=false, A[M][K] row-major): m-major [128][64], element (m,k) at
//     word m*64 + ((k/4 ^ (m&15))*4 + k%4) — the XOR spread makes the
//     MFMA column reads (consecutive m at fixed k) conflict-free;
//   A (TA=true, A[K][M]): k-major [64][128] linear (source contiguous
//     along m — no swizzle needed, rows of consecutive words);
//   B (B[K][N] or channel view): k-major [64][128] linear.
// 2 LDS buffers (2*(32+32) KB = 128 KB, 1 block/CU), glds for tile t+1
// issued before the MFMA loop on tile t, plain __syncthreads() per tile
// (its implicit vmcnt(0) drains the DMA).  The K-remainder tile is
// register-staged through read16 (zero-filled — glds cannot zero).
// Operand scope: plain or channel-view (kh==0) only; implicit-im2col
// views (kh>0) and TB=true stay on k_gemm_f32 (their contracted axis
// carries per-image zero padding glds cannot inject).
// Out-of-range lanes are exec-masked off: their LDS slots keep garbage,
// which only ever lands in output rows/cols the epilogue drops.
constexpr int BK2 = 64;

template <bool TA>
__launch_bounds__(256, 1) __global__ void k_gemm2(GemmArgs g) {
  extern __shared__ float smem2[];  // [2][A 128*64] [2][B 64*128]
  auto Aimg = [&](int b) -> float* { return smem2 + b * (128 * BK2); };
  auto Bimg = [&](int b) -> float* {
    return smem2 + (2 + b) * (128 * BK2);
  };

  long flat = blockIdx.x;
  {
    const long nwg = g.tiles;
    const long q = nwg / 8, rr = nwg % 8;
    const long xcd = flat % 8, idx = flat / 8;
    flat = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  const long tile_m = flat / g.tn;
  const long tile_n = flat - tile_m * g.tn;
  const long m0 = tile_m * BM, n0 = tile_n * BN;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave & 1;
  const int wc = (wave >> 1) & 1;
  const int row_in = lane & 31;
  const int ksel = lane >> 5;

  const long ntiles = (g.K + BK2 - 1) / BK2;
  const long nfull = g.K / BK2;  // glds-able full tiles

  // ---- per-lane staging geometry (fixed across K-tiles)
  // B: wave w fills k-rows [16w,16w+16), 8 instrs x 2 rows; lane: k-row
  // 16w+2i+lane/32, 4-float chunk at q = n0 + 4*(lane%32)
  const long bq = n0 + 4 * (lane & 31);
  const float* bbase;
  long bstride;
  if (g.bv.spad) {  // channel view: q -> (n_img, sp), row stride S
    const long n_img = bq / g.bv.spad;
    const long sp = bq - n_img * g.bv.spad;
    bbase = g.B + n_img * g.bv.chan * g.bv.S + sp;
    bstride = g.bv.S;
  } else {
    bbase = g.B + bq;
    bstride = g.ldb;
  }
  // A (TA=false): wave w fills m-rows [32w,32w+32), 8 instrs x 4 rows;
  // lane: m-row 32w+4i+lane/16, k-chunk kc = (lane%16) ^ (m&15)
  // A (TA=true): like B with lda stride: k-row 16w+2i+lane/32, m-chunk
  // m0 + 4*(lane%32)
  const long am_chunk = TA ? m0 + 4 * (lane & 31) : 0;

  auto stage_glds = [&](int buf, long kt) {
    const long k0 = kt * BK2;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      {  // B instr i
        const int krow = 16 * wave + 2 * i + (lane >> 5);
        if (bq < g.N) {
          const float* src = bbase + (k0 + krow) * bstride;
          __builtin_amdgcn_global_load_lds(
              (const __attribute__((address_space(1))) uint32_t*)(const
                                                                  void*)src,
              (__attribute__((address_space(3))) uint32_t*)
                  &Bimg(buf)[(16 * wave + 2 * i) * 128],
              16, 0, 0);
        }
      }
      if (!TA) {  // A instr i: 4 m-rows, swizzled k source
        const int mrow = 32 * wave + 4 * i + (lane >> 4);
        const int kc = (lane & 15) ^ (mrow & 15);
        if (m0 + mrow < g.M) {
          const float* src = g.A + (m0 + mrow) * g.lda + k0 + 4 * kc;
          __builtin_amdgcn_global_load_lds(
              (const __attribute__((address_space(1))) uint32_t*)(const
                                                                  void*)src,
              (__attribute__((address_space(3))) uint32_t*)
                  &Aimg(buf)[(32 * wave + 4 * i) * BK2],
              16, 0, 0);
        }
      } else {  // A instr i: 2 k-rows, contiguous m source
        const int krow = 16 * wave + 2 * i + (lane >> 5);
        if (am_chunk < g.M) {
          const float* src = g.A + (k0 + krow) * g.lda + am_chunk;
          __builtin_amdgcn_global_load_lds(
              (const __attribute__((address_space(1))) uint32_t*)(const
                                                                  void*)src,
              (__attribute__((address_space(3))) uint32_t*)
                  &Aimg(buf)[(16 * wave + 2 * i) * 128],
              16, 0, 0);
        }
      }
    }
  };

  // K-remainder tile: register-staged with zero fill (read16 masks).
  auto stage_tail = [&](int buf, long kt) {
    const long k0 = kt * BK2;
    float r[16];
    // B: thread t covers k-row (t&63), n-chunks (t>>6)*32 + {0,16}
    const int bk = t & 63;
    const int nb = (t >> 6) * 32;
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      read16(g.B, k0 + bk, n0 + nb + 16 * h, g.ldb, g.K, g.N, g.bv, r);
#pragma unroll
      for (int j = 0; j < 16; ++j)
        Bimg(buf)[bk * 128 + nb + 16 * h + j] = r[j];
    }
    if (!TA) {
      // A: thread t covers m-row (t&127), k-chunks (t>>7)*32 + {0,16},
      // written through the m-swizzle
      const int m = t & 127;
#pragma unroll
      for (int h = 0; h < 2; ++h) {
        const int kb = (t >> 7) * 32 + 16 * h;
        read16(g.A, m, k0 + kb, g.lda, g.M, g.K, g.av, r);
#pragma unroll
        for (int j = 0; j < 16; ++j) {
          const int k = kb + j;
          Aimg(buf)[m * BK2 + (((k >> 2) ^ (m & 15)) << 2) + (k & 3)] =
              r[j];
        }
      }
    } else {
      const int ak = t & 63;
      const int mb = (t >> 6) * 32;
#pragma unroll
      for (int h = 0; h < 2; ++h) {
        read16(g.A, k0 + ak, m0 + mb + 16 * h, g.lda, g.K, g.M, g.av, r);
#pragma unroll
        for (int j = 0; j < 16; ++j)
          Aimg(buf)[ak * 128 + mb + 16 * h + j] = r[j];
      }
    }
  };

  auto stage = [&](int buf, long kt) {
    if (kt < nfull)
      stage_glds(buf, kt);
    else
      stage_tail(buf, kt);
  };

  f32x16 acc00 = {}, acc01 = {}, acc10 = {}, acc11 = {};

  stage(0, 0);
  __syncthreads();

  // per-lane read bases
  const int a0base = TA ? (wr * 64 + row_in) : (wr * 64 + row_in) * BK2;
  const int bbase_r = wc * 64 + row_in;
  const int aswz = row_in & 15;

  for (long tt = 0; tt < ntiles; ++tt) {
    if (tt + 1 < ntiles) stage((tt + 1) & 1, tt + 1);
    {
      const float* Ab = Aimg(tt & 1);
      const float* Bb = Bimg(tt & 1);
#pragma unroll 8
      for (int kk = 0; kk < BK2; kk += 2) {
        const int krow = kk + ksel;
        float a0, a1;
        if (!TA) {
          const int s = (((krow >> 2) ^ aswz) << 2) + (krow & 3);
          a0 = Ab[a0base + s];
          a1 = Ab[a0base + 32 * BK2 + s];
        } else {
          a0 = Ab[krow * 128 + a0base];
          a1 = Ab[krow * 128 + a0base + 32];
        }
        const float b0 = Bb[krow * 128 + bbase_r];
        const float b1 = Bb[krow * 128 + bbase_r + 32];
        acc00 =
            __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc00, 0, 0, 0);
        acc01 =
            __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc01, 0, 0, 0);
        acc10 =
            __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc10, 0, 0, 0);
        acc11 =
            __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc11, 0, 0, 0);
      }
    }
    __syncthreads();  // publishes stage writes; implicit vmcnt(0) drains
                      // the outstanding glds
  }

  // ---- epilogue (same acc->(row,col) map as k_gemm_f32, no split-K)
  auto epi_tile = [&](const f32x16& a, int ti, int tj) {
    const long col = n0 + wc * 64 + tj * 32 + row_in;
    if (col >= g.N) return;
    long col_base = 0;
    bool col_ok = true;
    if (g.spad > 0) {
      const long n = col / g.spad;
      const long sp = col - n * g.spad;
      col_ok = sp < g.S;
      long pix = sp;
      if (g.OWo > 0) {
        const int oh = (int)(sp / g.OWo);
        const int ow = (int)(sp - (long)oh * g.OWo);
        pix = ((long)oh * g.osh) * g.Wd + (long)ow * g.osw;
      }
      col_base = n * g.n_stride + pix;
    }
    if (!col_ok) return;
    const float cbias = (g.bias && g.bias_per_col) ? g.bias[col] : 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const long row =
          m0 + wr * 64 + ti * 32 + ((r & 3) + 8 * (r >> 2) + 4 * ksel);
      if (row >= g.M) continue;
      float v = g.alpha * a[r];
      if (g.bias) v += g.bias_per_col ? cbias : g.bias[row];
      if (g.relu) v = fmaxf(v, 0.f);
      const long off =
          g.spad > 0 ? col_base + row * g.Srow : row * g.ldc + col;
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

// MN % 4 == 0 fast path (wgrad MN = Cout x C*kh*kw — always a multiple of
// 4 in practice): float4 rows + 4-way slab unroll keeps 16 loads in
// flight per thread instead of 1 — the scalar loop is latency-bound, not
// bandwidth-bound.  Fixed-shape summation (deterministic).
__global__ void k_splitk_reduce_v4(const float* __restrict__ slab, long MN4,
                                   long MN, int SK, float4* __restrict__ C) {
  const float4* s4 = (const float4*)slab;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < MN4;
       i += (long)gridDim.x * blockDim.x) {
    const long mn4 = MN >> 2;  // f4 elements per slab
    float4 a0 = {0, 0, 0, 0}, a1 = {0, 0, 0, 0};
    float4 a2 = {0, 0, 0, 0}, a3 = {0, 0, 0, 0};
    int s = 0;
    for (; s + 4 <= SK; s += 4) {
      const float4 v0 = s4[(long)s * mn4 + i];
      const float4 v1 = s4[(long)(s + 1) * mn4 + i];
      const float4 v2 = s4[(long)(s + 2) * mn4 + i];
      const float4 v3 = s4[(long)(s + 3) * mn4 + i];
      a0.x += v0.x; a0.y += v0.y; a0.z += v0.z; a0.w += v0.w;
      a1.x += v1.x; a1.y += v1.y; a1.z += v1.z; a1.w += v1.w;
      a2.x += v2.x; a2.y += v2.y; a2.z += v2.z; a2.w += v2.w;
      a3.x += v3.x; a3.y += v3.y; a3.z += v3.z; a3.w += v3.w;
    }
    for (; s < SK; ++s) {
      const float4 v = s4[(long)s * mn4 + i];
      a0.x += v.x; a0.y += v.y; a0.z += v.z; a0.w += v.w;
    }
    float4 r;
    r.x = (a0.x + a1.x) + (a2.x + a3.x);
    r.y = (a0.y + a1.y) + (a2.y + a3.y);
    r.z = (a0.z + a1.z) + (a2.z + a3.z);
    r.w = (a0.w + a1.w) + (a2.w + a3.w);
    C[i] = r;
  }
}

template <int TM, bool SPLITK>
static void launch_slim(bool transA, bool transB, dim3 grid, dim3 block,
                        hipStream_t s, const GemmArgs& g) {
  if (!transA && !transB)
    hipLaunchKernelGGL((k_gemm_slim<false, false, TM, SPLITK>), grid, block,
                       0, s, g);
  else if (!transA && transB)
    hipLaunchKernelGGL((k_gemm_slim<false, true, TM, SPLITK>), grid, block,
                       0, s, g);
  else if (transA && !transB)
    hipLaunchKernelGGL((k_gemm_slim<true, false, TM, SPLITK>), grid, block,
                       0, s, g);
  else
    hipLaunchKernelGGL((k_gemm_slim<true, true, TM, SPLITK>), grid, block,
                       0, s, g);
}

void gemm(hipStream_t s, bool transA, bool transB, long M, long N, long K,
          float alpha, const float* A, long lda, const float* B, long ldb,
          float beta, float* C, long ldc, const GemmEpi* epi,
          const GemmView* aview, const GemmView* bview) {
  CHECK_GT_(M, 0);
  CHECK_GT_(N, 0);
  CHECK_GT_(K, 0);
  CHECK_(!(aview && transA)) << "A view requires no transpose";
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
    g.Srow = epi->Srow > 0 ? epi->Srow : epi->S;
    g.OWo = epi->OWo;
    g.osh = epi->osh;
    g.osw = epi->osw;
    g.Wd = epi->Wd;
  }
  if (aview) g.av = *aview;
  if (bview) g.bv = *bview;
  // small-M slim tile (32x256 / 64x128): every MFMA row a real output row
  // — the 128^2 tile idles most rows for Cout<=64 layers (inception
  // branches, stage-1 convs).  fp32 only (the bf16 kernel keeps its own
  // kwaves path).  CAFFE_GEMM_SLIM=0 reverts.
  static const int slim_maxm = [] {
    const char* e = getenv("CAFFE_GEMM_SLIM");  // 0 = off, else max M
    return e ? atoi(e) : 64;
  }();
  // the slim tile also serves bf16 mode: M<=64 contractions run the fp32
  // slim kernel (better utilization than the bf16 kwaves path AND exact
  // arithmetic — AMP-style per-shape heterogeneity)
  int TMv = 128;
  if (M <= slim_maxm && N > 64) {
    TMv = M <= 32 ? 32 : 64;
  } else if (!Engine::get().gemm_bf16 && N > 64 && M > 128 && M <= 448 &&
             M % 128 >= 1 && M % 128 <= 64) {
    // remainder rule: a 128-tile grid wastes (128 - M%128) rows of its
    // last tile row (e.g. GoogLeNet conv2 M=192: 25% idle MFMA rows);
    // 64-row tiles fit these exactly at the cost of one extra B pass
    TMv = 64;
  }
  // bf16 wide tile (256x256): quadruple MACs per staged byte — the bf16
  // kernel is staging-bound (see gemm_bf16.hip)
  bool bf16_wide = false;
  if (Engine::get().gemm_bf16 && TMv == 128 && M > 128 && N > 128) {
    bf16_wide = true;
    TMv = 256;
  }
  const int TNv = TMv == 256 ? 256
                 : TMv == 128 ? 128
                 : (TMv == 32 ? 256 : 128);
  const long tm = (M + TMv - 1) / TMv, tn = (N + TNv - 1) / TNv;
  g.tn = tn;
  g.tiles = tm * tn;
  // split-K when the output grid cannot fill the chip (wgrad shapes);
  // plain-C calls only (no epilogue) — fused-epilogue GEMMs have huge N
  int SK = 1;
  // split-K block target: enough tiles*SK to oversubscribe the 256 CUs
  // (A/B-able via CAFFE_SK_TARGET).  Round 2: 2048 re-measured BEST after
  // the slim tiles landed (RN50 1996 -> 2013, GoogLeNet 3223 -> 3248;
  // 4096 regresses) — the slim wgrad grids have more, smaller tiles, so
  // deeper K-slicing now pays where it was neutral in round 1.
  static const long sk_target = [] {
    const char* e = getenv("CAFFE_SK_TARGET");
    return e ? atol(e) : 2048L;
  }();
  if (!epi && beta == 0.f && g.tiles < sk_target && K > 4 * BK) {
    SK = (int)std::min<long>(
        {sk_target / g.tiles + 1, (K + 4 * BK - 1) / (4 * BK), 256});
  }
  const char* pcls = !transA ? (!transB ? "gemm_nn" : "gemm_nt")
                            : (!transB ? "gemm_tn" : "gemm_tt");
  static const bool by_shape = getenv("CAFFE_GEMM_BY_SHAPE") != nullptr;
  PerfClass* pc;
  if (by_shape) {
    pc = &Engine::get().perf(std::string(pcls) + "_" + std::to_string(M) +
                             "x" + std::to_string(N) + "x" +
                             std::to_string(K));
  } else {
    static PerfClass* nn = &Engine::get().perf("gemm_nn");
    static PerfClass* nt = &Engine::get().perf("gemm_nt");
    static PerfClass* tn = &Engine::get().perf("gemm_tn");
    static PerfClass* tt = &Engine::get().perf("gemm_tt");
    pc = !transA ? (!transB ? nn : nt) : (!transB ? tn : tt);
  }
  PerfScope perf(pc, s, 2.0 * M * N * K,
                 4.0 * (M * K + N * K + M * N));
  dim3 grid((unsigned)g.tiles, 1, (unsigned)SK);
  dim3 block(256);
  if (SK > 1) {
    float* slab = (float*)Workspace::get_global().get(
        10, sizeof(float) * (size_t)SK * M * N);
    g.slab = slab;
    g.SK = SK;
    if (TMv == 32) {
      launch_slim<32, true>(transA, transB, grid, block, s, g);
    } else if (TMv == 64) {
      launch_slim<64, true>(transA, transB, grid, block, s, g);
    } else if (Engine::get().gemm_bf16) {
      gemm_launch_bf16(s, transA, transB, grid, block, g, true, bf16_wide);
    } else if (!transA && !transB)
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
    if (MN % 4 == 0) {
      const long MN4 = MN / 4;
      const int blocks = (int)std::min<long>((MN4 + 255) / 256, 2048);
      hipLaunchKernelGGL(k_splitk_reduce_v4, dim3(blocks), dim3(256), 0, s,
                         slab, MN4, MN, SK, (float4*)C);
      return;
    }
    const int blocks = (int)std::min<long>((MN + 255) / 256, 2048);
    hipLaunchKernelGGL(k_splitk_reduce, dim3(blocks), dim3(256), 0, s, slab,
                       MN, SK, C);
    return;
  }
  if (TMv == 32) {
    launch_slim<32, false>(transA, transB, grid, block, s, g);
    return;
  }
  if (TMv == 64) {
    launch_slim<64, false>(transA, transB, grid, block, s, g);
    return;
  }
  if (Engine::get().gemm_bf16) {
    gemm_launch_bf16(s, transA, transB, grid, block, g, false, bf16_wide);
    return;
  }
  // glds v2 path (BK=64, 1 block/CU): NN/TN with plain or channel-view
  // operands; implicit-im2col views (kh>0) and NT keep the register
  // kernel (their contracted axis has per-image zero padding a DMA
  // cannot inject).  MEASURED NEGATIVE on the model shapes:
  // 4096^3 98 vs 115 TF, ResNet shape mix ~-10% per class — at fp32's
  // arithmetic density the 128KB/2-buffer variant's 1 block/CU loses the
  // cross-block overlap v1 gets at 2 blocks/CU, and 3 buffers don't fit
  // LDS at BK=64.  Kept behind CAFFE_GEMM_V2=1 as the measured record.
  static const int v2 = [] {
    const char* e = getenv("CAFFE_GEMM_V2");
    return e ? atoi(e) : 0;
  }();
  if (v2 && !transB && M > 64 && N > 64 && !aview &&
      (!bview || bview->kh == 0)) {
    static const bool lds_ok = [] {
      return hipFuncSetAttribute(
                 (const void*)&k_gemm2<false>,
                 hipFuncAttributeMaxDynamicSharedMemorySize,
                 4 * 128 * BK2 * (int)sizeof(float)) == hipSuccess &&
             hipFuncSetAttribute(
                 (const void*)&k_gemm2<true>,
                 hipFuncAttributeMaxDynamicSharedMemorySize,
                 4 * 128 * BK2 * (int)sizeof(float)) == hipSuccess;
    }();
    if (lds_ok) {
      const size_t shmem = 4 * 128 * BK2 * sizeof(float);
      if (!transA)
        hipLaunchKernelGGL((k_gemm2<false>), grid, block, shmem, s, g);
      else
        hipLaunchKernelGGL((k_gemm2<true>), grid, block, shmem, s, g);
      return;
    }
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

// Wt[g*Cin_g + ci][co_local*kh*kw + ki*kw + kj] =
//   W[g*Coutg + co_local][ci][kh-1-ki][kw-1-kj]
__global__ void k_weight_flip(const float* __restrict__ w, int Cout,
                              int Cin_g, int kh, int kw, int groups,
                              float* __restrict__ wt) {
  const int Coutg = Cout / groups;
  const long total = (long)Cout * Cin_g * kh * kw;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    // destination indexing: [g][ci][co_local][ki][kj]
    long rem = i;
    const int kj = (int)(rem % kw);
    rem /= kw;
    const int ki = (int)(rem % kh);
    rem /= kh;
    const int co_l = (int)(rem % Coutg);
    rem /= Coutg;
    const int ci = (int)(rem % Cin_g);
    const int g = (int)(rem / Cin_g);
    wt[i] = w[((((long)(g * Coutg + co_l) * Cin_g + ci) * kh +
               (kh - 1 - ki)) * kw) + (kw - 1 - kj)];
  }
}
void weight_flip(hipStream_t s, const float* w, int Cout, int Cin_g, int kh,
                 int kw, float* wt) {
  // exported with groups folded in by the caller convention below
  const long total = (long)Cout * Cin_g * kh * kw;
  hipLaunchKernelGGL(k_weight_flip, dim3((int)std::min<long>((total + 255) / 256, 2048)),
                     dim3(256), 0, s, w, Cout, Cin_g, kh, kw, 1, wt);
}
void weight_flip_grouped(hipStream_t s, const float* w, int Cout, int Cin_g,
                         int kh, int kw, int groups, float* wt) {
  const long total = (long)Cout * Cin_g * kh * kw;
  hipLaunchKernelGGL(k_weight_flip, dim3((int)std::min<long>((total + 255) / 256, 2048)),
                     dim3(256), 0, s, w, Cout, Cin_g, kh, kw, groups, wt);
}

// db[c] = Σ_n Σ_s dy[n][c][s] — image-sliced partials + fixed-order sum
__global__ void k_bias_grad_part(const float* __restrict__ dy, int N, int C,
                                 int S, int nb, double* __restrict__ part) {
  const int c = blockIdx.x % C;
  const int slice = blockIdx.x / C;
  const int n0 = (int)((long)N * slice / nb);
  const int n1 = (int)((long)N * (slice + 1) / nb);
  const int span = (n1 - n0) * S;
  const int B = blockDim.x;
  auto addr = [&](int i) {
    const int n = n0 + i / S;
    return ((long)n * C + c) * S + (i - (n - n0) * S);
  };
  double acc = 0, acc2 = 0;
  int i = threadIdx.x;
  for (; i + 3 * B < span; i += 4 * B) {  // 4 loads in flight (MLP)
    const float v0 = dy[addr(i)], v1 = dy[addr(i + B)];
    const float v2 = dy[addr(i + 2 * B)], v3 = dy[addr(i + 3 * B)];
    acc += (double)v0 + v1;
    acc2 += (double)v2 + v3;
  }
  for (; i < span; i += B) acc += dy[addr(i)];
  acc += acc2;
  __shared__ double sh[256];
  sh[threadIdx.x] = acc;
  __syncthreads();
  for (int off = 128; off > 0; off >>= 1) {
    if (threadIdx.x < off) sh[threadIdx.x] += sh[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) part[(long)c * nb + slice] = sh[0];
}
__global__ void k_bias_grad_fin(const double* __restrict__ part, int nb,
                                int C, float* __restrict__ db) {
  for (int c = blockIdx.x * blockDim.x + threadIdx.x; c < C;
       c += gridDim.x * blockDim.x) {
    double acc = 0;
    for (int b = 0; b < nb; ++b) acc += part[(long)c * nb + b];
    db[c] = (float)acc;
  }
}
void bias_grad(hipStream_t s, const float* dy, int N, int C, long S,
               float* db) {
  PerfScope perf(PERF_CLASS("reduce"), s, 0, 4.0 * N * C * S);
  const int nb = std::max(1, std::min(N, 2048 / std::max(1, C)));
  double* part = (double*)Workspace::get_global().get(
      12, sizeof(double) * (size_t)C * nb);
  hipLaunchKernelGGL(k_bias_grad_part, dim3(C * nb), dim3(256), 0, s, dy, N,
                     C, (int)S, nb, part);
  hipLaunchKernelGGL(k_bias_grad_fin, dim3(1), dim3(256), 0, s, part, nb, C,
                     db);
}

}  // namespace gpu
}  // namespace camd
