// gemm_bf16.hip — bf16 tensor-core GEMM for gfx950 (CDNA4):
// `v_mfma_f32_32x32x16_bf16` (≈2.52 PF dense chip peak, 16x the fp32 MFMA
// rate) with fp32 accumulation and FP32 STORAGE everywhere — the staging
// pass reads the same fp32 tensors as the fp32 kernel (plain, channel-view
// or implicit-im2col-view operands, strides included) and converts to
// bf16 in-register on the way into LDS.  This is the MI355X-native
// mixed-precision contract (AMP semantics): conv/IP contractions compute
// in bf16, every other op, the master weights, the gradients'
// accumulation and the solver stay fp32.  The reference's analog is the
// NVCaffe Ftype/Btype fp16 machinery (net.cpp:100-156, type.hpp:13-47)
// whose headline was pseudo-fp16 (fp16 storage, fp32 math,
// models/bvlc_alexnet/logs/alexnet_pfp16.log); on CDNA4 the win is in the
// matrix unit, not storage, so the design inverts: storage fp32,
// multiply bf16.
//
// Structure mirrors k_gemm_f32 (the measured-best register pipeline, 2
// blocks/CU): 128x128 tile, 4 waves, 2x2 accs of 32x32, BK=64 (4 MFMA
// k-steps of 16), double-buffered LDS, T14 issue-early/write-late, one
// barrier per K-tile, kwaves intra-block split for M<=64 / N<=64 tiles,
// deterministic split-K with fp32 slabs.
//
// LDS layout (paired-k): element (row r, k) of a tile lives at bf16 slot
//   (k>>1)*PSTR + 2*r + (k&1),  PSTR = 2*128 + 2
// so the MFMA fragment (r fixed, 8 consecutive k) is 4 conflict-free b32
// reads (consecutive lanes -> consecutive words; the +1-word pad per
// pair-row staggers banks for the k-contiguous staging writes).
#include "gemm_common.hpp"

namespace camd {
namespace gpu {

namespace bf16gemm {

using bf16 = __bf16;
using bf16x2 = __attribute__((ext_vector_type(2))) __bf16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int PSTR = 2 * BM + 2;          // bf16 per pair-row (+1 word pad)
constexpr int OPSZ = (BK / 2) * PSTR;     // bf16 per operand image

// ---- staging: fp32 read16 -> bf16 convert -> LDS scatter
// (templated on the image's pair-row stride so the 256-wide tile reuses
// the same layout algebra)
// row-contiguous source (thread holds (r, k0..k0+15)): paired b32 writes
template <int PS>
__device__ __forceinline__ void write_rowk(bf16* img, int r, int k0,
                                           const float (&v)[16]) {
#pragma unroll
  for (int j = 0; j < 16; j += 2) {
    bf16x2 p;
    p[0] = (bf16)v[j];
    p[1] = (bf16)v[j + 1];
    *(bf16x2*)&img[((k0 + j) >> 1) * PS + 2 * r] = p;
  }
}
// k-contiguous source (thread holds (k, n0..n0+15)): u16 scatter (2-way
// word sharing across the k-pair lanes — bank-staggered by the pad)
template <int PS>
__device__ __forceinline__ void write_kn(bf16* img, int k, int n0,
                                         const float (&v)[16]) {
  bf16* base = img + (k >> 1) * PS + (k & 1);
#pragma unroll
  for (int j = 0; j < 16; ++j) base[2 * (n0 + j)] = (bf16)v[j];
}

// A tile: rows are the M axis.
template <bool TRANS>
__device__ __forceinline__ void stage_a_load(const GemmArgs& g, long m0,
                                             long k0, float (&r)[16],
                                             float (&r2)[16]) {
  const int t = threadIdx.x;
  if (!TRANS) {  // A[M][Kd] contiguous along Kd: (m, 32-k chunk)
    read16(g.A, m0 + (t & 127), k0 + (t >> 7) * 32, g.lda, g.M, g.K, g.av,
           r);
    read16(g.A, m0 + (t & 127), k0 + (t >> 7) * 32 + 16, g.lda, g.M, g.K,
           g.av, r2);
  } else {  // A[Kd][M] contiguous along M: (k, 32-m chunk)
    read16(g.A, k0 + (t & 63), m0 + (t >> 6) * 32, g.lda, g.K, g.M, g.av,
           r);
    read16(g.A, k0 + (t & 63), m0 + (t >> 6) * 32 + 16, g.lda, g.K, g.M,
           g.av, r2);
  }
}
template <bool TRANS>
__device__ __forceinline__ void stage_a_write(bf16* As,
                                              const float (&r)[16],
                                              const float (&r2)[16]) {
  const int t = threadIdx.x;
  if (!TRANS) {
    write_rowk<PSTR>(As, t & 127, (t >> 7) * 32, r);
    write_rowk<PSTR>(As, t & 127, (t >> 7) * 32 + 16, r2);
  } else {
    write_kn<PSTR>(As, t & 63, (t >> 6) * 32, r);
    write_kn<PSTR>(As, t & 63, (t >> 6) * 32 + 16, r2);
  }
}
// B tile: rows are the N axis.
template <bool TRANS>
__device__ __forceinline__ void stage_b_load(const GemmArgs& g, long n0,
                                             long k0, float (&r)[16],
                                             float (&r2)[16]) {
  const int t = threadIdx.x;
  if (!TRANS) {  // B[Kd][N] contiguous along N
    read16(g.B, k0 + (t & 63), n0 + (t >> 6) * 32, g.ldb, g.K, g.N, g.bv,
           r);
    read16(g.B, k0 + (t & 63), n0 + (t >> 6) * 32 + 16, g.ldb, g.K, g.N,
           g.bv, r2);
  } else {  // B[N][Kd] contiguous along Kd
    read16(g.B, n0 + (t & 127), k0 + (t >> 7) * 32, g.ldb, g.N, g.K, g.bv,
           r);
    read16(g.B, n0 + (t & 127), k0 + (t >> 7) * 32 + 16, g.ldb, g.N, g.K,
           g.bv, r2);
  }
}
template <bool TRANS>
__device__ __forceinline__ void stage_b_write(bf16* Bs,
                                              const float (&r)[16],
                                              const float (&r2)[16]) {
  const int t = threadIdx.x;
  if (!TRANS) {
    write_kn<PSTR>(Bs, t & 63, (t >> 6) * 32, r);
    write_kn<PSTR>(Bs, t & 63, (t >> 6) * 32 + 16, r2);
  } else {
    write_rowk<PSTR>(Bs, t & 127, (t >> 7) * 32, r);
    write_rowk<PSTR>(Bs, t & 127, (t >> 7) * 32 + 16, r2);
  }
}

// fragment: (r, ksel*8 + 0..7) as 4 b32 reads
template <int PS>
__device__ __forceinline__ bf16x8 frag(const bf16* img, int r, int kbase) {
  bf16x8 a;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const bf16x2 p =
        *(const bf16x2*)&img[((kbase + 2 * i) >> 1) * PS + 2 * r];
    a[2 * i] = p[0];
    a[2 * i + 1] = p[1];
  }
  return a;
}

template <bool TA, bool TB, bool SPLITK>
__launch_bounds__(256, 2) __global__ void k_gemm_bf16(GemmArgs g) {
  __shared__ bf16 smem[4 * OPSZ];
  auto As = [&](int buf) -> bf16* { return smem + buf * OPSZ; };
  auto Bs = [&](int buf) -> bf16* { return smem + (2 + buf) * OPSZ; };

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
  const int mwaves = g.M > 64 ? 2 : 1;
  const int nwaves = g.N > 64 ? 2 : 1;
  const int mn = mwaves * nwaves;
  const int kwaves = 4 / mn;
  const int wr = wave % mwaves;
  const int wc = (wave / mwaves) % nwaves;
  const int wk = wave / mn;
  const int row_in = lane & 31;
  const int ksel = lane >> 5;  // which 8-deep half of the 16-k step

  long k_lo = 0, k_hi = g.K;
  if (SPLITK) {
    const int sk = blockIdx.z;
    k_lo = g.K * sk / g.SK / BK * BK;
    k_hi = (sk == g.SK - 1) ? g.K : g.K * (sk + 1) / g.SK / BK * BK;
    if (k_lo >= k_hi) return;
  }
  const long ntiles = (k_hi - k_lo + BK - 1) / BK;

  f32x16 acc00 = {}, acc01 = {}, acc10 = {}, acc11 = {};
  float ra[16], ra2[16], rb[16], rb2[16];

  stage_a_load<TA>(g, m0, k_lo, ra, ra2);
  stage_b_load<TB>(g, n0, k_lo, rb, rb2);
  stage_a_write<TA>(As(0), ra, ra2);
  stage_b_write<TB>(Bs(0), rb, rb2);
  __syncthreads();

  int cur = 0;
  if (kwaves == 1) {
    for (long t = 0; t < ntiles; ++t) {
      if (t + 1 < ntiles) {  // issue next tile's global loads early (T14)
        stage_a_load<TA>(g, m0, k_lo + (t + 1) * BK, ra, ra2);
        stage_b_load<TB>(g, n0, k_lo + (t + 1) * BK, rb, rb2);
      }
      {
        const bf16* Ab = As(cur);
        const bf16* Bb = Bs(cur);
#pragma unroll
        for (int kk = 0; kk < BK; kk += 16) {
          const int kbase = kk + ksel * 8;
          const bf16x8 a0 = frag<PSTR>(Ab, wr * 64 + row_in, kbase);
          const bf16x8 a1 = frag<PSTR>(Ab, wr * 64 + 32 + row_in, kbase);
          const bf16x8 b0 = frag<PSTR>(Bb, wc * 64 + row_in, kbase);
          const bf16x8 b1 = frag<PSTR>(Bb, wc * 64 + 32 + row_in, kbase);
          acc00 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b0, acc00, 0,
                                                          0, 0);
          acc01 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b1, acc01, 0,
                                                          0, 0);
          acc10 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b0, acc10, 0,
                                                          0, 0);
          acc11 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b1, acc11, 0,
                                                          0, 0);
        }
      }
      if (t + 1 < ntiles) {
        stage_a_write<TA>(As(cur ^ 1), ra, ra2);
        stage_b_write<TB>(Bs(cur ^ 1), rb, rb2);
      }
      __syncthreads();
      cur ^= 1;
    }
  } else {
    // kwaves path: both buffers live per iteration (small-M/N tiles keep
    // all 4 waves computing); kwaves==4 splits the 64-deep step in half
    const int grp = wk & 1;
    const int khalf = wk >> 1;
    const int kk0 = (kwaves == 4) ? khalf * (BK / 2) : 0;
    const int kk1 = (kwaves == 4) ? kk0 + BK / 2 : BK;
    if (1 < ntiles) {
      stage_a_load<TA>(g, m0, k_lo + 1 * BK, ra, ra2);
      stage_b_load<TB>(g, n0, k_lo + 1 * BK, rb, rb2);
      stage_a_write<TA>(As(1), ra, ra2);
      stage_b_write<TB>(Bs(1), rb, rb2);
    }
    __syncthreads();
    for (long p = 0; p < ntiles; p += 2) {
      const long t_next0 = p + 2, t_next1 = p + 3;
      if (t_next0 < ntiles) {
        stage_a_load<TA>(g, m0, k_lo + t_next0 * BK, ra, ra2);
        stage_b_load<TB>(g, n0, k_lo + t_next0 * BK, rb, rb2);
      }
      float sa[16], sa2[16], sb[16], sb2[16];
      if (t_next1 < ntiles) {
        stage_a_load<TA>(g, m0, k_lo + t_next1 * BK, sa, sa2);
        stage_b_load<TB>(g, n0, k_lo + t_next1 * BK, sb, sb2);
      }
      const long t_mine = p + grp;
      if (t_mine < ntiles) {
        const bf16* Ab = As(grp);
        const bf16* Bb = Bs(grp);
        for (int kk = kk0; kk < kk1; kk += 16) {
          const int kbase = kk + ksel * 8;
          const bf16x8 a0 = frag<PSTR>(Ab, wr * 64 + row_in, kbase);
          const bf16x8 a1 = frag<PSTR>(Ab, wr * 64 + 32 + row_in, kbase);
          const bf16x8 b0 = frag<PSTR>(Bb, wc * 64 + row_in, kbase);
          const bf16x8 b1 = frag<PSTR>(Bb, wc * 64 + 32 + row_in, kbase);
          acc00 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b0, acc00, 0,
                                                          0, 0);
          acc01 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b1, acc01, 0,
                                                          0, 0);
          acc10 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b0, acc10, 0,
                                                          0, 0);
          acc11 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b1, acc11, 0,
                                                          0, 0);
        }
      }
      __syncthreads();
      if (t_next0 < ntiles) {
        stage_a_write<TA>(As(0), ra, ra2);
        stage_b_write<TB>(Bs(0), rb, rb2);
      }
      if (t_next1 < ntiles) {
        stage_a_write<TA>(As(1), sa, sa2);
        stage_b_write<TB>(Bs(1), sb, sb2);
      }
      __syncthreads();
    }
  }

  if (kwaves > 1) {
    // combine K-range partials through the (idle) staging LDS in fixed
    // wk order — deterministic (same scheme as the fp32 kernel; the bf16
    // arena is byte-compatible scratch here)
    float* fsm = (float*)smem;
    __syncthreads();
    if (wk > 0) {
      float* slot = fsm + (long)(wave - mn) * 4096 + lane * 64;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        slot[r] = acc00[r];
        slot[16 + r] = acc01[r];
        slot[32 + r] = acc10[r];
        slot[48 + r] = acc11[r];
      }
    }
    __syncthreads();
    if (wk > 0) return;
    for (int k2 = 1; k2 < kwaves; ++k2) {
      const int src_wave = k2 * mn + (wc * mwaves + wr);
      const float* slot = fsm + (long)(src_wave - mn) * 4096 + lane * 64;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        acc00[r] += slot[r];
        acc01[r] += slot[16 + r];
        acc10[r] += slot[32 + r];
        acc11[r] += slot[48 + r];
      }
    }
  }

  // ---- epilogue: identical acc->(row,col) map to the fp32 kernel
  auto epi_tile = [&](const f32x16& a, int ti, int tj) {
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
          m0 + wr * 64 + ti * 32 + ((r & 3) + 8 * (r >> 2) + 4 * ksel);
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
  epi_tile(acc00, 0, 0);
  epi_tile(acc01, 0, 1);
  epi_tile(acc10, 1, 0);
  epi_tile(acc11, 1, 1);
}


// ---------------------------------------------------------------
// Wide 256x256 tile (round 2): the 128^2 bf16 kernel is STAGING-bound —
// the 16x MFMA rate moved GEMM time only ~13% — so quadruple the MACs
// per staged byte.  512 threads = 8 waves as 4(M) x 2(N), each wave a
// 64(M) x 128(N) sub-tile of 2x4 32x32 accumulators (128 acc VGPRs);
// LDS 2 x (A 256 + B 256 rows) paired-k images = 131.6 KB -> 1 block/CU
// (2 waves/SIMD, same as the 128^2 variant, at half the staging per MAC).
// Dispatched for M > 128 && N > 128; edges masked as usual.
constexpr int WPS = 2 * 256 + 2;           // pair-row stride, 256-row image
constexpr int WOPSZ = (BK / 2) * WPS;

template <bool TA>
__device__ __forceinline__ void wstage_a_load(const GemmArgs& g, long m0,
                                              long k0, float (&r)[16],
                                              float (&r2)[16]) {
  const int t = threadIdx.x;
  if (!TA) {  // [M][K] contig K: r = t&255, 32-k chunk
    read16(g.A, m0 + (t & 255), k0 + (t >> 8) * 32, g.lda, g.M, g.K, g.av,
           r);
    read16(g.A, m0 + (t & 255), k0 + (t >> 8) * 32 + 16, g.lda, g.M, g.K,
           g.av, r2);
  } else {  // [K][M] contig M: k = t&63, 32-m chunk
    read16(g.A, k0 + (t & 63), m0 + (t >> 6) * 32, g.lda, g.K, g.M, g.av,
           r);
    read16(g.A, k0 + (t & 63), m0 + (t >> 6) * 32 + 16, g.lda, g.K, g.M,
           g.av, r2);
  }
}
template <bool TA>
__device__ __forceinline__ void wstage_a_write(bf16* As,
                                               const float (&r)[16],
                                               const float (&r2)[16]) {
  const int t = threadIdx.x;
  if (!TA) {
    write_rowk<WPS>(As, t & 255, (t >> 8) * 32, r);
    write_rowk<WPS>(As, t & 255, (t >> 8) * 32 + 16, r2);
  } else {
    write_kn<WPS>(As, t & 63, (t >> 6) * 32, r);
    write_kn<WPS>(As, t & 63, (t >> 6) * 32 + 16, r2);
  }
}
template <bool TB>
__device__ __forceinline__ void wstage_b_load(const GemmArgs& g, long n0,
                                              long k0, float (&r)[16],
                                              float (&r2)[16]) {
  const int t = threadIdx.x;
  if (!TB) {  // [K][N] contig N
    read16(g.B, k0 + (t & 63), n0 + (t >> 6) * 32, g.ldb, g.K, g.N, g.bv,
           r);
    read16(g.B, k0 + (t & 63), n0 + (t >> 6) * 32 + 16, g.ldb, g.K, g.N,
           g.bv, r2);
  } else {  // [N][K] contig K
    read16(g.B, n0 + (t & 255), k0 + (t >> 8) * 32, g.ldb, g.N, g.K, g.bv,
           r);
    read16(g.B, n0 + (t & 255), k0 + (t >> 8) * 32 + 16, g.ldb, g.N, g.K,
           g.bv, r2);
  }
}
template <bool TB>
__device__ __forceinline__ void wstage_b_write(bf16* Bs,
                                               const float (&r)[16],
                                               const float (&r2)[16]) {
  const int t = threadIdx.x;
  if (!TB) {
    write_kn<WPS>(Bs, t & 63, (t >> 6) * 32, r);
    write_kn<WPS>(Bs, t & 63, (t >> 6) * 32 + 16, r2);
  } else {
    write_rowk<WPS>(Bs, t & 255, (t >> 8) * 32, r);
    write_rowk<WPS>(Bs, t & 255, (t >> 8) * 32 + 16, r2);
  }
}

template <bool TA, bool TB, bool SPLITK>
__launch_bounds__(512, 1) __global__ void k_gemm_bf16_wide(GemmArgs g) {
  __shared__ bf16 smem[4 * WOPSZ];
  auto As = [&](int buf) -> bf16* { return smem + buf * WOPSZ; };
  auto Bs = [&](int buf) -> bf16* { return smem + (2 + buf) * WOPSZ; };

  long flat = blockIdx.x;
  {
    const long nwg = g.tiles;
    const long q = nwg / 8, rr = nwg % 8;
    const long xcd = flat % 8, idx = flat / 8;
    flat = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  const long tile_m = flat / g.tn;
  const long tile_n = flat - tile_m * g.tn;
  const long m0 = tile_m * 256, n0 = tile_n * 256;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wr = wave & 3;        // 4 row blocks of 64
  const int wc = wave >> 2;       // 2 col blocks of 128
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

  f32x16 acc[2][4] = {};
  float ra[16], ra2[16], rb[16], rb2[16];

  wstage_a_load<TA>(g, m0, k_lo, ra, ra2);
  wstage_b_load<TB>(g, n0, k_lo, rb, rb2);
  wstage_a_write<TA>(As(0), ra, ra2);
  wstage_b_write<TB>(Bs(0), rb, rb2);
  __syncthreads();

  int cur = 0;
  for (long tt = 0; tt < ntiles; ++tt) {
    if (tt + 1 < ntiles) {
      wstage_a_load<TA>(g, m0, k_lo + (tt + 1) * BK, ra, ra2);
      wstage_b_load<TB>(g, n0, k_lo + (tt + 1) * BK, rb, rb2);
    }
    {
      const bf16* Ab = As(cur);
      const bf16* Bb = Bs(cur);
#pragma unroll
      for (int kk = 0; kk < BK; kk += 16) {
        const int kbase = kk + ksel * 8;
        const bf16x8 a0 = frag<WPS>(Ab, wr * 64 + row_in, kbase);
        const bf16x8 a1 = frag<WPS>(Ab, wr * 64 + 32 + row_in, kbase);
        bf16x8 b[4];
#pragma unroll
        for (int j = 0; j < 4; ++j)
          b[j] = frag<WPS>(Bb, wc * 128 + 32 * j + row_in, kbase);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          acc[0][j] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              a0, b[j], acc[0][j], 0, 0, 0);
          acc[1][j] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              a1, b[j], acc[1][j], 0, 0, 0);
        }
      }
    }
    if (tt + 1 < ntiles) {
      wstage_a_write<TA>(As(cur ^ 1), ra, ra2);
      wstage_b_write<TB>(Bs(cur ^ 1), rb, rb2);
    }
    __syncthreads();
    cur ^= 1;
  }

  auto epi_tile = [&](const f32x16& a, int ti, int tj) {
    const long col = n0 + wc * 128 + tj * 32 + row_in;
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
          m0 + wr * 64 + ti * 32 + ((r & 3) + 8 * (r >> 2) + 4 * ksel);
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
#pragma unroll
  for (int ti = 0; ti < 2; ++ti)
#pragma unroll
    for (int tj = 0; tj < 4; ++tj) epi_tile(acc[ti][tj], ti, tj);
}

}  // namespace bf16gemm

// launcher used by gemm() in gemm_f32.hip when bf16 compute is enabled;
// `wide` selects the 256x256-tile kernel (grid must be 256-tile-based)
void gemm_launch_bf16(hipStream_t s, bool transA, bool transB, dim3 grid,
                      dim3 block, const GemmArgs& g, bool splitk,
                      bool wide) {
  using namespace bf16gemm;
  if (wide) {
    dim3 wblock(512);
    if (splitk) {
      if (!transA && !transB)
        hipLaunchKernelGGL((k_gemm_bf16_wide<false, false, true>), grid,
                           wblock, 0, s, g);
      else if (!transA && transB)
        hipLaunchKernelGGL((k_gemm_bf16_wide<false, true, true>), grid,
                           wblock, 0, s, g);
      else if (transA && !transB)
        hipLaunchKernelGGL((k_gemm_bf16_wide<true, false, true>), grid,
                           wblock, 0, s, g);
      else
        hipLaunchKernelGGL((k_gemm_bf16_wide<true, true, true>), grid,
                           wblock, 0, s, g);
      return;
    }
    if (!transA && !transB)
      hipLaunchKernelGGL((k_gemm_bf16_wide<false, false, false>), grid,
                         wblock, 0, s, g);
    else if (!transA && transB)
      hipLaunchKernelGGL((k_gemm_bf16_wide<false, true, false>), grid,
                         wblock, 0, s, g);
    else if (transA && !transB)
      hipLaunchKernelGGL((k_gemm_bf16_wide<true, false, false>), grid,
                         wblock, 0, s, g);
    else
      hipLaunchKernelGGL((k_gemm_bf16_wide<true, true, false>), grid,
                         wblock, 0, s, g);
    return;
  }
  if (splitk) {
    if (!transA && !transB)
      hipLaunchKernelGGL((k_gemm_bf16<false, false, true>), grid, block, 0,
                         s, g);
    else if (!transA && transB)
      hipLaunchKernelGGL((k_gemm_bf16<false, true, true>), grid, block, 0,
                         s, g);
    else if (transA && !transB)
      hipLaunchKernelGGL((k_gemm_bf16<true, false, true>), grid, block, 0,
                         s, g);
    else
      hipLaunchKernelGGL((k_gemm_bf16<true, true, true>), grid, block, 0,
                         s, g);
    return;
  }
  if (!transA && !transB)
    hipLaunchKernelGGL((k_gemm_bf16<false, false, false>), grid, block, 0,
                       s, g);
  else if (!transA && transB)
    hipLaunchKernelGGL((k_gemm_bf16<false, true, false>), grid, block, 0, s,
                       g);
  else if (transA && !transB)
    hipLaunchKernelGGL((k_gemm_bf16<true, false, false>), grid, block, 0, s,
                       g);
  else
    hipLaunchKernelGGL((k_gemm_bf16<true, true, false>), grid, block, 0, s,
                       g);
}

}  // namespace gpu
}  // namespace camd
