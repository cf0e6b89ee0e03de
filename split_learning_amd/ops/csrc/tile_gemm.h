// MFMA f32 tile-GEMM framework for gfx950 (CDNA4).
//
// One templated kernel computes C = A @ B for fp32 operands on the exact-f32
// MFMA path (v_mfma_f32_16x16x4_f32 — bitwise an fmaf chain, §3 of the CDNA
// guide), with pluggable gather functors so the same MFMA/LDS core serves:
//   * plain strided (batched) GEMM               (gemm_f32.hip)
//   * implicit-GEMM conv2d fwd / bwd-data / bwd-weight (conv2d.hip)
//
// Geometry: 256 threads = 4 waves; block tile 64(M) x 64(N); BK=16 K-step;
// each wave owns a 32x32 sub-tile = 2x2 MFMA fragments with 4 f32x4
// accumulators.  Staging is register-prefetch double-buffered (next K-tile's
// gathers issue under the MFMA phase).
//
// Gather contract (PMC-driven redesign — the first version's per-element
// bounds branches compiled to 56 s_and_saveexec chains per loop and put waves
// 53% issue-stalled, profiles/):
//   * prepA(batch, m_clamped, valid, dk) / prepB(batch, n_clamped, valid, dk)
//     run ONCE per thread (m/n are staging-loop invariants): they hoist the
//     row/column address decomposition into a small ctx.  `dk` is the
//     thread's CONSTANT intra-tile k offset (A staging is k-fast: every one
//     of a thread's RA elements shares dk = tid & 15; B staging is n-fast:
//     ctx i loads dk = (tid >> 6) + i * waves) — functors may fold dk into
//     hoisted addresses;
//   * prepK(k0) runs once per K-tile with a wave-uniform k0 and returns a
//     small KCtx of SALU-computed per-tile scalars (e.g. the hoisted
//     (batch, row) window decomposition of the round-2 conv gathers);
//   * loadA(ctx, kctx, k)/loadB(ctx, kctx, k) must be BRANCHLESS: always load
//     from a clamped in-bounds address and select 0 via the valid flags
//     (cndmask, not exec-mask branches).  k arrives already clamped to
//     [0, K-1]; k_valid covers the K tail.
#pragma once

#include "common.h"

constexpr int SLK_BM = 64;
constexpr int SLK_BN = 64;
constexpr int SLK_BK = 16;

// LDS layout for the staged A/B tiles: PMC analysis (profiles/SUMMARY.md "LDS
// conflict fix") showed the per-CU LDS array saturated — occupancy is maxed
// (36-40 VGPR -> 8 waves/SIMD) yet MFMA sits at 28% busy with
// SQ_LDS_BANK_CONFLICT ~400/wave: the old [k][m] rows at stride BM+4 (≡ 4 mod
// 32 banks) put the two 16-lane row-groups of every ds_read_b32 on 12
// overlapping banks (2-way), and operand pairs (m, m+16) were not adjacent so
// each needed its own b32 read.
//
// phys(k, m) = k*96 + 32*(m>>5) + ((2*(m&15) + ((m>>4)&1)) ^ (2*k)):
//   * pair-adjacency: cols m and m+16 differ only in (m>>4)&1 -> adjacent
//     dwords -> ONE ds_read_b64 per operand pair (8 wide reads per tile
//     instead of 16 narrow: LDS is 64 dwords/clk for b64, 32 for b32);
//   * row separation: stride 96 ≡ 32 (mod 64 b64-banks) + the 32*(m>>5) half
//     split puts the two rows of a read's lane groups in disjoint bank
//     halves -> conflict-free reads;
//   * the ^(2*k) swizzle spreads the k-fast staging writes (16 rows x 2 cols
//     per 32-lane group) over 16 banks -> writes stay 2-way (same as the old
//     layout), and it preserves pair adjacency (even, low-5-bit operand).
constexpr int SLK_LDS_ROW = 96;

__device__ __forceinline__ int slk_lds_phys(int k, int m) {
  return k * SLK_LDS_ROW + 32 * (m >> 5) +
         (((2 * (m & 15)) | ((m >> 4) & 1)) ^ (2 * k));
}

// PP (ping-pong) axis of the round-2 core experiment (PMC: 53% issue-stall +
// 32% barrier-parked, MFMA pipe only ~37% busy):
//   PP=true : LDS double-buffered, ONE barrier per BK tile (stage into buf
//             p^1 while reading buf p) — but 24.6 KB LDS caps the CU at 6
//             blocks (6 waves/SIMD);
//   PP=false: single-buffer two-barrier loop, 12.3 KB LDS, register-limited
//             7 waves/SIMD.
// Both compile; SLK_PP picks at launch so one GPU session can A/B them.
template <typename Gather, typename Store, bool PP>
__global__ __launch_bounds__(256) void slk_mfma_gemm_kernel(
    Gather g, Store st, int M, int N, int K, int split_k, int k_per_split,
    int xcd_remap) {
  __shared__ __align__(16) float ldsA[PP ? 2 : 1][SLK_BK * SLK_LDS_ROW];
  __shared__ __align__(16) float ldsB[PP ? 2 : 1][SLK_BK * SLK_LDS_ROW];

  // XCD-aware n-tile placement: consecutive blockIdx.x land on XCD b % 8
  // with disjoint per-XCD L2s, so the linear map sprays one image's B-side
  // gathers across all 8 L2s.  When the n-grid is a multiple of 8, remap so
  // XCD k owns the CONTIGUOUS n-range [k*T/8, (k+1)*T/8): bijective, and the
  // per-XCD working set of x drops 8x (one L2 can then hold its slice).
  int tile_n = blockIdx.x;
  if (xcd_remap && (gridDim.x & 7) == 0) {
    tile_n = (blockIdx.x & 7) * (gridDim.x >> 3) + (blockIdx.x >> 3);
  }
  const int tile_m = blockIdx.y;
  const int batch = blockIdx.z / split_k;
  const int ks = blockIdx.z % split_k;

  const int m0 = tile_m * SLK_BM;
  const int n0 = tile_n * SLK_BN;
  const int k_begin = ks * k_per_split;
  const int k_end = min(K, k_begin + k_per_split);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = (wid >> 1) * 32;  // wave row offset in tile
  const int wn = (wid & 1) * 32;   // wave col offset in tile

  const int frag_r = lane >> 4;    // 0..3: k sub-index for A/B operands
  const int frag_c = lane & 15;    // 0..15

#ifdef SLK_MFMA32
  // 32x32x2 variant: one accumulator chain per wave (issue interval ==
  // dependent latency == 64 cyc, so a single chain back-to-backs), 8 MFMA
  // instructions per BK=16 tile instead of 16 — fewer issue slots in an
  // issue-bound kernel; k-order (and thus numerics) bit-identical.
  f32x16 acc32 = {};
  (void)frag_r; (void)frag_c;
#else
  f32x4 acc[2][2] = {};
#endif

  constexpr int RA = (SLK_BM * SLK_BK) / 256;  // per-thread A elements
  constexpr int RB = (SLK_BN * SLK_BK) / 256;  // per-thread B elements
  float ra[RA], rb[RB];

  // hoisted per-thread staging contexts (m/n fixed across the K loop).
  // A staging is k-fast: (tid + i*256) & 15 == tid & 15 for every i, so ALL
  // of a thread's A elements share one intra-tile k offset a_dk.  B staging
  // is n-fast with (tid + i*256) & 63 == tid & 63: every B element of a
  // thread shares ONE column context (a single BCtx, not RB copies — the
  // duplicate contexts cost ~10 VGPR and a wave/SIMD of occupancy).
  const int a_dk = tid & (SLK_BK - 1);
  typename Gather::ACtx actx[RA];
  #pragma unroll
  for (int i = 0; i < RA; ++i) {
    const int idx = tid + i * 256;
    const int m = m0 + (idx >> 4);      // k-fast: contiguous global rows
    actx[i] = g.prepA(batch, min(m, M - 1), m < M, a_dk);
  }
  const int nb = n0 + (tid & 63);       // n-fast: coalesced for row-major B
  typename Gather::BCtx bctx = g.prepB(batch, min(nb, N - 1), nb < N, tid >> 6);

  auto load_tile = [&](int k0) {
    // prepK: wave-uniform per-tile scalars (SALU) shared by every load
    const typename Gather::KCtx kc =
        g.prepK(__builtin_amdgcn_readfirstlane(k0));
    #pragma unroll
    for (int i = 0; i < RA; ++i) {
      const int k = k0 + a_dk;
      ra[i] = g.loadA(actx[i], kc, min(k, K - 1), k < k_end);
    }
    #pragma unroll
    for (int i = 0; i < RB; ++i) {
      // (tid + i*256) >> 6 is WAVE-UNIFORM by construction (the >>6
      // collapses the 64 lanes of a wave to its wave id); readfirstlane makes
      // that provable so the functor's whole k-decomposition (divisions,
      // FastDiv multiplies) compiles to SALU ops instead of per-lane VALU —
      // the gather address math was 139 VALU per 16 MFMA in the loop body.
      const int k = __builtin_amdgcn_readfirstlane(k0 + (tid >> 6) + i * 4);
      rb[i] = g.loadB(bctx, kc, min(k, K - 1), k < k_end);
    }
  };

  auto stage_to = [&](int p) {
    #pragma unroll
    for (int i = 0; i < RA; ++i) {
      const int idx = tid + i * 256;
      ldsA[p][slk_lds_phys(idx & 15, idx >> 4)] = ra[i];
    }
    #pragma unroll
    for (int i = 0; i < RB; ++i) {
      const int idx = tid + i * 256;
      ldsB[p][slk_lds_phys(idx >> 6, idx & 63)] = rb[i];
    }
  };

  auto do_mfma = [&](int p) {
#ifdef SLK_MFMA32
    // operand map (ISA): lane l supplies A[i=l&31][k=l>>5], B[k=l>>5][j=l&31]
    const int kh = lane >> 5;
    const int l31 = lane & 31;
    #pragma unroll
    for (int kk = 0; kk < SLK_BK / 2; ++kk) {
      float a = ldsA[p][slk_lds_phys(kk * 2 + kh, wm + l31)];
      float b = ldsB[p][slk_lds_phys(kk * 2 + kh, wn + l31)];
      acc32 = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b, acc32, 0, 0, 0);
    }
#else
    // operand pairs (frag_c, frag_c+16) are adjacent dwords in the swizzled
    // layout -> each f32x2 load is one conflict-free ds_read_b64.
    // (Batching all 8 reads up front measured SLOWER: +16 VGPR pushed the
    // allocation over the 64-register step, 8 -> 6 waves/SIMD.)
    #pragma unroll
    for (int kk = 0; kk < SLK_BK / 4; ++kk) {
      const int kr = kk * 4 + frag_r;
      f32x2 a01 = *reinterpret_cast<const f32x2*>(
          &ldsA[p][slk_lds_phys(kr, wm + frag_c)]);
      f32x2 b01 = *reinterpret_cast<const f32x2*>(
          &ldsB[p][slk_lds_phys(kr, wn + frag_c)]);
      acc[0][0] = __builtin_amdgcn_mfma_f32_16x16x4f32(a01.x, b01.x, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_16x16x4f32(a01.x, b01.y, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_16x16x4f32(a01.y, b01.x, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_16x16x4f32(a01.y, b01.y, acc[1][1], 0, 0, 0);
    }
#endif
  };

  load_tile(k_begin);
  if (PP) {
    stage_to(0);
    __syncthreads();
    int p = 0;
    for (int k0 = k_begin; k0 < k_end; k0 += SLK_BK) {
      const bool has_next = k0 + SLK_BK < k_end;
      if (has_next) load_tile(k0 + SLK_BK);  // global loads overlap the MFMAs
      do_mfma(p);
      if (has_next) stage_to(p ^ 1);  // write the OTHER buffer: no hazard
      // one barrier per tile: publishes buf p^1's writes AND closes buf p's
      // reads before it is overwritten next iteration
      __syncthreads();
      p ^= 1;
    }
  } else {
    for (int k0 = k_begin; k0 < k_end; k0 += SLK_BK) {
      stage_to(0);
      __syncthreads();
      if (k0 + SLK_BK < k_end) load_tile(k0 + SLK_BK);
      do_mfma(0);
      __syncthreads();
    }
  }

#ifdef SLK_MFMA32
  // epilogue: C/D map for 32x32 shapes: col = lane&31,
  // row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
  #pragma unroll
  for (int r = 0; r < 16; ++r) {
    int row = m0 + wm + (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    int col = n0 + wn + (lane & 31);
    if (row < M && col < N) st.store(batch, row, col, acc32[r], ks);
  }
#else
  // epilogue: C/D fragment mapping for 16x16x4: col = lane&15, row = (lane>>4)*4 + i
  #pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
    #pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        int row = m0 + wm + mi * 16 + frag_r * 4 + i;
        int col = n0 + wn + ni * 16 + frag_c;
        if (row < M && col < N) st.store(batch, row, col, acc[mi][ni][i], ks);
      }
    }
  }
#endif
}

// ---------------------------------------------------------------------------
// Big-tile variant: 128x128 block, 64x64 per wave as a 2x2 grid of
// v_mfma_f32_32x32x2_f32 chains (16 accumulator VGPRs each, issue interval ==
// dependent latency == 64 cyc so the 4 interleaved chains saturate the MAI
// pipe).  4x the MFMA work per staged element vs the 64x64 kernel — built to
// close the large plain-GEMM gap vs rocBLAS (ROADMAP.md item 2; the 64x64
// kernel measures 91 TF at 4096^3 vs rocBLAS 150).  Experimental: selected by
// SLK_GEMM_BIG=1 for large strided GEMMs only (gemm_f32.hip), never conv.
//
// LDS layout: columns are stored pair-interleaved so a wave's two mfma32
// operands (cols c and c+32 for its 64-wide tile half) are ADJACENT dwords —
// one conflict-free ds_read_b64 per operand pair instead of two b32 reads
// (lane l31 hits b64 banks {2*l31, 2*l31+1}: the full 64-bank row).  Staging
// writes become 2-way conflicted (same as the 64x64 kernel's writes) — a
// measured-good trade, reads outnumber writes 2:1.
constexpr int SLK_BM2 = 128;
constexpr int SLK_BN2 = 128;
constexpr int SLK_LDS_ROW2 = 130;

__device__ __forceinline__ int slk_big_perm(int c) {
  // (c, c+32) -> adjacent dwords within each 64-column half
  return 2 * (c & 31) + ((c >> 5) & 1) + 64 * (c >> 6);
}

// 2nd launch-bounds arg = min waves per SIMD: forces the allocator to 128
// total registers (64 VGPR + the 64 MFMA accumulator AGPRs) so FOUR waves
// fit per SIMD instead of three — the allocator does not get there on its
// own (74 VGPR unforced)
template <typename Gather, typename Store>
__global__ __launch_bounds__(256, 4) void slk_mfma_gemm_kernel_big(
    Gather g, Store st, int M, int N, int K, int split_k, int k_per_split,
    int xcd_remap) {
  // (LDS ping-pong was tried here too — 2 x 16.6 KB keeps 4 blocks/CU so it
  // is occupancy-free — and still measured ~3% SLOWER at 4096^3, likely the
  // +3 VGPR spills in the forced-128-register budget.  Two-barrier loop
  // retained; profiles/SUMMARY.md round 2.)
  __shared__ __align__(16) float ldsA[1][SLK_BK * SLK_LDS_ROW2];
  __shared__ __align__(16) float ldsB[1][SLK_BK * SLK_LDS_ROW2];

  // XCD-aware contiguous n-tile ownership (see the 64x64 kernel's note)
  int tile_n = blockIdx.x;
  if (xcd_remap && (gridDim.x & 7) == 0) {
    tile_n = (blockIdx.x & 7) * (gridDim.x >> 3) + (blockIdx.x >> 3);
  }
  const int tile_m = blockIdx.y;
  const int batch = blockIdx.z / split_k;
  const int ks = blockIdx.z % split_k;

  const int m0 = tile_m * SLK_BM2;
  const int n0 = tile_n * SLK_BN2;
  const int k_begin = ks * k_per_split;
  const int k_end = min(K, k_begin + k_per_split);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = (wid >> 1) * 64;
  const int wn = (wid & 1) * 64;
  const int kh = lane >> 5;      // mfma32 k half: A[i=l&31][k=l>>5]
  const int l31 = lane & 31;

  f32x16 acc[2][2] = {};         // (mi, ni) 32x32 chains

  constexpr int RA = (SLK_BM2 * SLK_BK) / 256;  // 8
  constexpr int RB = (SLK_BN2 * SLK_BK) / 256;  // 8
  float ra[RA], rb[RB];

  const int a_dk = tid & (SLK_BK - 1);
  typename Gather::ACtx actx[RA];
  #pragma unroll
  for (int i = 0; i < RA; ++i) {
    const int idx = tid + i * 256;
    const int m = m0 + (idx >> 4);      // k-fast staging, 128 rows
    actx[i] = g.prepA(batch, min(m, M - 1), m < M, a_dk);
  }
  const int nb = n0 + (tid & 127);      // n-fast staging, 128 cols
  typename Gather::BCtx bctx = g.prepB(batch, min(nb, N - 1), nb < N, tid >> 7);

  auto load_tile = [&](int k0) {
    const typename Gather::KCtx kc =
        g.prepK(__builtin_amdgcn_readfirstlane(k0));
    #pragma unroll
    for (int i = 0; i < RA; ++i) {
      const int k = k0 + a_dk;
      ra[i] = g.loadA(actx[i], kc, min(k, K - 1), k < k_end);
    }
    #pragma unroll
    for (int i = 0; i < RB; ++i) {
      // (tid + i*256) >> 7 is wave-uniform (see the 64x64 kernel's note)
      const int k = __builtin_amdgcn_readfirstlane(k0 + (tid >> 7) + i * 2);
      rb[i] = g.loadB(bctx, kc, min(k, K - 1), k < k_end);
    }
  };

  auto stage_to = [&](int p) {
    #pragma unroll
    for (int i = 0; i < RA; ++i) {
      const int idx = tid + i * 256;
      ldsA[p][(idx & 15) * SLK_LDS_ROW2 + slk_big_perm(idx >> 4)] = ra[i];
    }
    #pragma unroll
    for (int i = 0; i < RB; ++i) {
      const int idx = tid + i * 256;
      ldsB[p][(idx >> 7) * SLK_LDS_ROW2 + slk_big_perm(idx & 127)] = rb[i];
    }
  };

  load_tile(k_begin);
  for (int k0 = k_begin; k0 < k_end; k0 += SLK_BK) {
    stage_to(0);
    __syncthreads();
    if (k0 + SLK_BK < k_end) load_tile(k0 + SLK_BK);

    #pragma unroll
    for (int kk = 0; kk < SLK_BK / 2; ++kk) {
      const int kr = kk * 2 + kh;
      // pairs (wm+l31, wm+32+l31) / (wn+l31, wn+32+l31) are adjacent dwords
      const f32x2 a01 = *reinterpret_cast<const f32x2*>(
          &ldsA[0][kr * SLK_LDS_ROW2 + 64 * (wm >> 6) + 2 * l31]);
      const f32x2 b01 = *reinterpret_cast<const f32x2*>(
          &ldsB[0][kr * SLK_LDS_ROW2 + 64 * (wn >> 6) + 2 * l31]);
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a01.x, b01.x, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a01.x, b01.y, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a01.y, b01.x, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a01.y, b01.y, acc[1][1], 0, 0, 0);
    }
    __syncthreads();
  }

  // C/D map for 32x32 shapes: col = lane&31, row = (r&3) + 8*(r>>2) + 4*(lane>>5)
  #pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
    #pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        int row = m0 + wm + mi * 32 + (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
        int col = n0 + wn + ni * 32 + (lane & 31);
        if (row < M && col < N) st.store(batch, row, col, acc[mi][ni][r], ks);
      }
    }
  }
}

// SLK_GEMM_BIG=1 enables the 128x128 kernel for large plain GEMMs
inline bool slk_use_big(int M, int N) {
  static int v = [] {
    const char* e = std::getenv("SLK_GEMM_BIG");
    return e ? atoi(e) : 0;
  }();
  return v != 0 && M >= 96 && N >= 96;
}

// BK-alignment can REDUCE the effective split count (K=4608 at split 64 ->
// k_per_split 80 -> 58 splits).  Callers sizing per-split slabs must use this
// exact value or trailing slab slices stay uninitialized.
inline int slk_effective_split(int K, int split_k) {
  if (split_k < 1) return 1;
  int k_per_split = ceil_div(K, split_k);
  k_per_split = ((k_per_split + SLK_BK - 1) / SLK_BK) * SLK_BK;
  return ceil_div(K, k_per_split);
}

// SLK_PP=1 selects the ping-pong single-barrier loop.  Measured on MI355X
// (profiles/SUMMARY.md round 2): PP loses everywhere — its 24.6 KB LDS caps
// the CU at 6 blocks vs the two-barrier loop's register-limited 7 waves/SIMD,
// and occupancy beats barrier elision (bench 10297 vs 9448 img/s).
inline bool slk_pp_mode() {
  static int v = [] {
    const char* e = std::getenv("SLK_PP");
    return e ? atoi(e) : 0;
  }();
  return v != 0;
}

// Host-side launch helper.
template <typename Gather, typename Store>
inline void slk_launch_gemm(const Gather& g, const Store& st, int M, int N, int K,
                            int n_batch, int split_k, hipStream_t stream,
                            bool big = false) {
  if (split_k < 1) split_k = 1;
  int k_per_split = ceil_div(K, split_k);
  // round k_per_split up to a BK multiple so every split starts aligned
  k_per_split = ((k_per_split + SLK_BK - 1) / SLK_BK) * SLK_BK;
  split_k = ceil_div(K, k_per_split);
  static const int xcd = [] {
    const char* e = std::getenv("SLK_XCD");
    return e ? atoi(e) : 1;
  }();
  if (big) {
    dim3 grid(ceil_div(N, SLK_BN2), ceil_div(M, SLK_BM2), n_batch * split_k);
    hipLaunchKernelGGL((slk_mfma_gemm_kernel_big<Gather, Store>), grid, dim3(256),
                       0, stream, g, st, M, N, K, split_k, k_per_split, xcd);
    return;
  }
  dim3 grid(ceil_div(N, SLK_BN), ceil_div(M, SLK_BM), n_batch * split_k);
  if (slk_pp_mode()) {
    hipLaunchKernelGGL((slk_mfma_gemm_kernel<Gather, Store, true>), grid,
                       dim3(256), 0, stream, g, st, M, N, K, split_k,
                       k_per_split, xcd);
  } else {
    hipLaunchKernelGGL((slk_mfma_gemm_kernel<Gather, Store, false>), grid,
                       dim3(256), 0, stream, g, st, M, N, K, split_k,
                       k_per_split, xcd);
  }
}

// Heuristic: pick split_k so the grid oversubscribes the 256 CUs (~5 blocks/CU:
// the GPU split-K sweep in profiles/ shows throughput rising monotonically to a
// ~1280-block target — the gather path needs many resident waves, and 48-way
// fp32 atomic accumulation costs less than the idle CUs it fills).
inline int slk_pick_split_k(int M, int N, int K, int n_batch,
                            int bm = SLK_BM, int bn = SLK_BN) {
  // env-tunable (host-side, read per call so in-process sweeps work):
  // SLK_SPLIT_TARGET = block-count target, SLK_SPLIT_CAP = max split factor
  static auto readenv = [](const char* n, long d) {
    const char* v = std::getenv(n);
    return v ? atol(v) : d;
  };
  const long target = readenv("SLK_SPLIT_TARGET", 1280);
  const long cap = readenv("SLK_SPLIT_CAP", 64);
  long tiles = (long)ceil_div(M, bm) * ceil_div(N, bn) * (n_batch > 0 ? n_batch : 1);
  if (tiles >= target || K <= SLK_BK * 2) return 1;
  long want = (target + tiles - 1) / tiles;
  if (want > cap) want = cap;
  long maxk = (K + 2 * SLK_BK - 1) / (2 * SLK_BK);  // keep >=2 BK steps per split
  long sk = want < maxk ? want : maxk;
  return (int)(sk < 1 ? 1 : sk);
}
