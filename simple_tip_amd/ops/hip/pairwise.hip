// MFMA fp32 pairwise-distance kernel family for gfx950 (CDNA4).
//
// The workload's hot core: squared L2 distances
// between a test-AT matrix A [M,K] and a train-AT matrix B [N,K], computed
// as ||a||^2 + ||b||^2 - 2 a.b with the Gram matrix on the f32-input MFMA
// (v_mfma_f32_32x32x2_f32: exact fp32 fmaf-chain numerics at the 157 TF f32
// peak), staged through LDS in k-major tiles, with three fused epilogues:
//   EPI_FULL   - write the D tile            (silhouette, debugging)
//   EPI_ROWMIN - per-row min + argmin        (DSA, kmeans assign)
//   EPI_KDE    - per-row online logsumexp(-d/2) partials (LSA KDE)
// Per-(block-column) partials are combined by small deterministic
// second-pass kernels so 1-GPU and sharded runs are bitwise reproducible.
//
// Geometry: 256 threads = 4 waves as 2x2; block tile 128x128, K-step 32;
// each wave owns a 64x64 sub-tile = 2x2 MFMA 32x32 accumulators.

#include "tip_common.h"

#include <cfloat>
#include <cmath>

using f32x16 = __attribute__((__vector_size__(16 * sizeof(float)))) float;

constexpr int BM = 128;
constexpr int BN = 128;
constexpr int BK = 32;
constexpr int LDS_PAD = 4;  // pad k-major rows to de-conflict staging writes

enum Epilogue { EPI_FULL = 0, EPI_ROWMIN = 1, EPI_KDE = 2 };

// Staging is split T14-style (issue-early / write-late): the global loads
// for tile t+1 are issued into registers BEFORE tile t's MFMA loop (HBM
// latency hides under the ~4k-cycle f32-MFMA compute phase) and the LDS
// write pass runs after the barrier. LDS image is k-major
// lds[BK][BM+pad]: lds[k][r] = src[row0+r][k0+k]; 256 threads, 16 floats
// each (4x float4).

struct StageRegs {
  float4 v[4];
};

TIP_DEV StageRegs stage_load(
    const float* __restrict__ src, int rows, int K, int row0, int k0) {
  StageRegs sr;
  const int t = threadIdx.x;
  const int r = t >> 1;                 // 0..127
  const int kq = (t & 1) * (BK / 2);    // 0 or 16
  const int grow = row0 + r;
  const bool row_ok = grow < rows;
#pragma unroll
  for (int q4 = 0; q4 < 4; ++q4) {
    float4 v = make_float4(0.f, 0.f, 0.f, 0.f);
    const int gk = k0 + kq + q4 * 4;
    // float4 path only when rows are 16B-aligned (K % 4 == 0)
    if (row_ok && gk + 3 < K && (K & 3) == 0) {
      v = *reinterpret_cast<const float4*>(&src[(int64_t)grow * K + gk]);
    } else if (row_ok) {
      // K tail: scalar guarded loads (zeros contribute nothing to the dot)
      float tmp[4] = {0.f, 0.f, 0.f, 0.f};
      for (int e = 0; e < 4; ++e)
        if (gk + e < K) tmp[e] = src[(int64_t)grow * K + gk + e];
      v = make_float4(tmp[0], tmp[1], tmp[2], tmp[3]);
    }
    sr.v[q4] = v;
  }
  return sr;
}

TIP_DEV void stage_write(float* lds, const StageRegs& sr) {
  const int t = threadIdx.x;
  const int r = t >> 1;
  const int kq = (t & 1) * (BK / 2);
#pragma unroll
  for (int q4 = 0; q4 < 4; ++q4) {
    lds[(kq + q4 * 4 + 0) * (BM + LDS_PAD) + r] = sr.v[q4].x;
    lds[(kq + q4 * 4 + 1) * (BM + LDS_PAD) + r] = sr.v[q4].y;
    lds[(kq + q4 * 4 + 2) * (BM + LDS_PAD) + r] = sr.v[q4].z;
    lds[(kq + q4 * 4 + 3) * (BM + LDS_PAD) + r] = sr.v[q4].w;
  }
}

template <int EPI>
__launch_bounds__(256, 2) __global__ void pairwise_kernel(
    const float* __restrict__ A,      // [M, K]
    const float* __restrict__ B,      // [N, K]
    const float* __restrict__ anorm,  // [M] row squared norms
    const float* __restrict__ bnorm,  // [N]
    int M, int N, int K,
    float* __restrict__ out_full,     // EPI_FULL: [M, N]
    float* __restrict__ pmin_val,     // EPI_ROWMIN: [jblocks, M]
    int* __restrict__ pmin_idx,       //             [jblocks, M]
    float2* __restrict__ pkde) {      // EPI_KDE:   [jblocks, M] (max, sum)
  __shared__ float lds[2 * BK * (BM + LDS_PAD)];
  float* As = lds;
  float* Bs = lds + BK * (BM + LDS_PAD);
  // reduction scratch: per block row x 2 column-halves
  __shared__ float red_v[BM][2];
  __shared__ int red_i[BM][2];
  __shared__ float red_s[BM][2];

  // XCD-aware block swizzle (T1, bijective form): consecutive remapped ids
  // land on one XCD and share the A panel in that XCD's L2.
  const int jblocks = gridDim.x;
  int bi, bj;
  {
    const int nwg = gridDim.x * gridDim.y;
    const int id = blockIdx.y * gridDim.x + blockIdx.x;
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = id % 8, pos = id / 8;
    const int newid =
        (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
    bi = newid / gridDim.x;
    bj = newid % gridDim.x;
  }
  const int row0 = bi * BM;
  const int col0 = bj * BN;

  const int lane = lane_id();
  const int wid = wave_id();
  const int wr = wid >> 1;  // wave row (0..1) -> 64 rows
  const int wc = wid & 1;   // wave col (0..1) -> 64 cols

  f32x16 acc[2][2] = {};

  // T14 pipeline: prologue stages tile 0; each iteration issues tile t+1's
  // global loads before tile t's MFMAs and writes them to LDS afterwards.
  StageRegs ra = stage_load(A, M, K, row0, 0);
  StageRegs rb = stage_load(B, N, K, col0, 0);
  stage_write(As, ra);
  stage_write(Bs, rb);
  __syncthreads();

  for (int k0 = 0; k0 < K; k0 += BK) {
    const bool has_next = (k0 + BK) < K;
    if (has_next) {
      ra = stage_load(A, M, K, row0, k0 + BK);
      rb = stage_load(B, N, K, col0, k0 + BK);
    }
#pragma unroll
    for (int kk = 0; kk < BK; kk += 2) {
      const int k = kk + (lane >> 5);
      const float* as = &As[k * (BM + LDS_PAD) + wr * 64 + (lane & 31)];
      const float* bs = &Bs[k * (BM + LDS_PAD) + wc * 64 + (lane & 31)];
      const float a0 = as[0], a1 = as[32];
      const float b0 = bs[0], b1 = bs[32];
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
    }
    __syncthreads();  // all reads of tile t done
    if (has_next) {
      stage_write(As, ra);
      stage_write(Bs, rb);
    }
    __syncthreads();  // tile t+1 visible
  }

  // ---- epilogue ----
  // C/D layout of v_mfma_f32_32x32x2_f32 (standard 32x32 map):
  //   col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
  const int jl0 = col0 + wc * 64 + (lane & 31);  // n=0 column
  const float bn0 = (jl0 < N) ? bnorm[jl0] : 0.f;
  const float bn1 = (jl0 + 32 < N) ? bnorm[jl0 + 32] : 0.f;

#pragma unroll
  for (int m = 0; m < 2; ++m) {
    // process the 16 regs; rows repeat across n, so handle n jointly
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      const int row_local = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
      const int block_row = wr * 64 + m * 32 + row_local;
      const int i = row0 + block_row;
      const float an = (i < M) ? anorm[i] : 0.f;
      float d0 = an + bn0 - 2.f * acc[m][0][reg];
      float d1 = an + bn1 - 2.f * acc[m][1][reg];
      d0 = fmaxf(d0, 0.f);
      d1 = fmaxf(d1, 0.f);
      if (jl0 >= N) d0 = FLT_MAX;
      if (jl0 + 32 >= N) d1 = FLT_MAX;

      if (EPI == EPI_FULL) {
        if (i < M) {
          if (jl0 < N) out_full[(int64_t)i * N + jl0] = d0;
          if (jl0 + 32 < N) out_full[(int64_t)i * N + jl0 + 32] = d1;
        }
      } else if (EPI == EPI_ROWMIN) {
        MinIdx mi{d0, jl0};
        mi = min_idx_combine(mi, MinIdx{d1, jl0 + 32});
        mi = half_reduce_min(mi);
        if ((lane & 31) == 0) {
          red_v[block_row][wc] = mi.v;
          red_i[block_row][wc] = mi.i;
        }
      } else {  // EPI_KDE: per-row max of t=-d/2 and sum exp(t - max)
        MinIdx mi{d0, 0};
        mi = min_idx_combine(mi, MinIdx{d1, 0});
        mi = half_reduce_min(mi);
        const float tmax = -0.5f * mi.v;  // = max_j(-d/2) over these 64 cols
        float s = 0.f;
        if (d0 != FLT_MAX) s += __expf(-0.5f * d0 - tmax);
        if (d1 != FLT_MAX) s += __expf(-0.5f * d1 - tmax);
        s = half_reduce_sum(s);
        if ((lane & 31) == 0) {
          red_v[block_row][wc] = tmax;
          red_s[block_row][wc] = s;
        }
      }
    }
  }

  if (EPI == EPI_FULL) return;
  __syncthreads();
  // combine the two column-halves (wc=0 covers lower j: ties prefer it)
  for (int r = threadIdx.x; r < BM; r += blockDim.x) {
    const int i = row0 + r;
    if (i >= M) continue;
    if (EPI == EPI_ROWMIN) {
      MinIdx a{red_v[r][0], red_i[r][0]};
      MinIdx b{red_v[r][1], red_i[r][1]};
      MinIdx best = min_idx_combine(a, b);
      pmin_val[(int64_t)bj * M + i] = best.v;
      pmin_idx[(int64_t)bj * M + i] = best.i;
    } else if (EPI == EPI_KDE) {
      float m0 = red_v[r][0], s0 = red_s[r][0];
      float m1 = red_v[r][1], s1 = red_s[r][1];
      float mm, ss;
      if (m0 >= m1) {
        mm = m0;
        ss = s0 + ((s1 > 0.f) ? s1 * __expf(m1 - m0) : 0.f);
      } else {
        mm = m1;
        ss = s1 + ((s0 > 0.f) ? s0 * __expf(m0 - m1) : 0.f);
      }
      pkde[(int64_t)bj * M + i] = make_float2(mm, ss);
    }
  }
}

// ---------------------------------------------------------------------------
// Grouped (segmented) variant: test rows are sorted by class and padded to
// 128-row segments (tseg offsets, C+1 entries); train rows are the
// class-concatenated set with raw offsets (nseg). One launch covers every
// class — the per-class python loop (10+ launches, ~200 host ops per step)
// becomes one kernel + one combine. Each block derives its class from its
// row-block (segments are 128-aligned), and only columns inside the class's
// train range participate. Argmin indices are GLOBAL train rows, so the
// precomputed DSA b-table gathers directly.
// ---------------------------------------------------------------------------

template <int EPI>
__launch_bounds__(256, 2) __global__ void grouped_pairwise_kernel(
    const float* __restrict__ A,      // [Bp, K] class-sorted, 128-padded
    const float* __restrict__ B,      // [Ntot, K] class-concatenated
    const float* __restrict__ anorm,  // [Bp]
    const float* __restrict__ bnorm,  // [Ntot]
    const int* __restrict__ tseg,     // [C+1] padded test offsets (x128)
    const int* __restrict__ nseg,     // [C+1] train offsets
    int nclasses, int Bp, int K, int jb_max_g,
    float* __restrict__ pmin_val,     // [jb_max, Bp]
    int* __restrict__ pmin_idx,
    float2* __restrict__ pkde) {
  __shared__ float lds[2 * BK * (BM + LDS_PAD)];
  float* As = lds;
  float* Bs = lds + BK * (BM + LDS_PAD);
  __shared__ float red_v[BM][2];
  __shared__ int red_i[BM][2];
  __shared__ float red_s[BM][2];

  // Block->tile mapping. TIP_GXY selects the variant (hardware-swept):
  //  0: bi = blockIdx.y, bj = blockIdx.x (column block fast — baseline)
  //  1: grid swapped, row block fast: consecutive ids share a column tile
  //  2: XCD-aware (1-D grid padded to 8*per): consecutive block ids land on
  //     different XCDs (round-robin dispatch), so give each XCD a
  //     CONTIGUOUS tile range decomposed row-block-fast — every XCD sweeps
  //     the row blocks of one (or few) column tiles and its 2-MB B tile
  //     stays L2-resident while the A tiles stream.
#ifndef TIP_GXY
#define TIP_GXY 0
#endif
  int bi, bj;
#if TIP_GXY == 0
  bi = blockIdx.y;
  bj = blockIdx.x;
#elif TIP_GXY == 1
  bi = blockIdx.x;
  bj = blockIdx.y;
#else
  {
    // 1-D grid of 8*per blocks (per = ceil(nrows*jb_max/8))
    const int nrows = (Bp + BM - 1) / BM;
    const int total = nrows * jb_max_g;
    constexpr int NXCD = 8;
    const int per = gridDim.x / NXCD;
    const int t0 = blockIdx.x;
    const int t = (t0 % NXCD) * per + t0 / NXCD;
    if (t >= total) return;
    bi = t % nrows;
    bj = t / nrows;
  }
#endif
  const int row0 = bi * BM;
  if (row0 >= Bp) return;
  // class of this 128-aligned row block
  int cls = 0;
  for (int c = 0; c < nclasses; ++c)
    if (tseg[c] <= row0) cls = c;
  const int ncol0 = nseg[cls], ncol1 = nseg[cls + 1];
  const int col0 = ncol0 + bj * BN;
  if (col0 >= ncol1) return;  // this class has fewer column blocks

  const int lane = lane_id();
  const int wid = wave_id();
  const int wr = wid >> 1;
  const int wc = wid & 1;

  f32x16 acc[2][2] = {};
  StageRegs ra = stage_load(A, Bp, K, row0, 0);
  StageRegs rb = stage_load(B, ncol1, K, col0, 0);
  stage_write(As, ra);
  stage_write(Bs, rb);
  __syncthreads();
  for (int k0 = 0; k0 < K; k0 += BK) {
    const bool has_next = (k0 + BK) < K;
    if (has_next) {
      ra = stage_load(A, Bp, K, row0, k0 + BK);
      rb = stage_load(B, ncol1, K, col0, k0 + BK);
    }
#pragma unroll
    for (int kk = 0; kk < BK; kk += 2) {
      const int k = kk + (lane >> 5);
      const float* as = &As[k * (BM + LDS_PAD) + wr * 64 + (lane & 31)];
      const float* bs = &Bs[k * (BM + LDS_PAD) + wc * 64 + (lane & 31)];
      const float a0 = as[0], a1 = as[32];
      const float b0 = bs[0], b1 = bs[32];
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
    }
    __syncthreads();
    if (has_next) {
      stage_write(As, ra);
      stage_write(Bs, rb);
    }
    __syncthreads();
  }

  const int jl0 = col0 + wc * 64 + (lane & 31);
  const float bn0 = (jl0 < ncol1) ? bnorm[jl0] : 0.f;
  const float bn1 = (jl0 + 32 < ncol1) ? bnorm[jl0 + 32] : 0.f;
#pragma unroll
  for (int m = 0; m < 2; ++m) {
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      const int row_local = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
      const int block_row = wr * 64 + m * 32 + row_local;
      const int i = row0 + block_row;
      const float an = (i < Bp) ? anorm[i] : 0.f;
      float d0 = fmaxf(an + bn0 - 2.f * acc[m][0][reg], 0.f);
      float d1 = fmaxf(an + bn1 - 2.f * acc[m][1][reg], 0.f);
      if (jl0 >= ncol1) d0 = FLT_MAX;
      if (jl0 + 32 >= ncol1) d1 = FLT_MAX;
      if (EPI == EPI_ROWMIN) {
        MinIdx mi{d0, jl0};
        mi = min_idx_combine(mi, MinIdx{d1, jl0 + 32});
        mi = half_reduce_min(mi);
        if ((lane & 31) == 0) {
          red_v[block_row][wc] = mi.v;
          red_i[block_row][wc] = mi.i;
        }
      } else {
        MinIdx mi{d0, 0};
        mi = min_idx_combine(mi, MinIdx{d1, 0});
        mi = half_reduce_min(mi);
        const float tmax = -0.5f * mi.v;
        float s = 0.f;
        if (d0 != FLT_MAX) s += __expf(-0.5f * d0 - tmax);
        if (d1 != FLT_MAX) s += __expf(-0.5f * d1 - tmax);
        s = half_reduce_sum(s);
        if ((lane & 31) == 0) {
          red_v[block_row][wc] = tmax;
          red_s[block_row][wc] = s;
        }
      }
    }
  }
  __syncthreads();
  for (int r = threadIdx.x; r < BM; r += blockDim.x) {
    const int i = row0 + r;
    if (i >= Bp) continue;
    if (EPI == EPI_ROWMIN) {
      MinIdx best = min_idx_combine(
          MinIdx{red_v[r][0], red_i[r][0]}, MinIdx{red_v[r][1], red_i[r][1]});
      pmin_val[(int64_t)bj * Bp + i] = best.v;
      pmin_idx[(int64_t)bj * Bp + i] = best.i;
    } else {
      float m0 = red_v[r][0], s0 = red_s[r][0];
      float m1 = red_v[r][1], s1 = red_s[r][1];
      float mm, ss;
      if (m0 >= m1) {
        mm = m0;
        ss = s0 + ((s1 > 0.f) ? s1 * __expf(m1 - m0) : 0.f);
      } else {
        mm = m1;
        ss = s1 + ((s0 > 0.f) ? s0 * __expf(m0 - m1) : 0.f);
      }
      pkde[(int64_t)bj * Bp + i] = make_float2(mm, ss);
    }
  }
}

__global__ void grouped_rowmin_combine_kernel(
    const float* __restrict__ pval, const int* __restrict__ pidx,
    const int* __restrict__ tseg, const int* __restrict__ nseg, int nclasses,
    int Bp, float* __restrict__ out_dist, int64_t* __restrict__ out_idx) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= Bp) return;
  int cls = 0;
  for (int c = 0; c < nclasses; ++c)
    if (tseg[c] <= i) cls = c;
  const int jcount = (nseg[cls + 1] - nseg[cls] + BN - 1) / BN;
  MinIdx best{FLT_MAX, 0x7fffffff};
  for (int b = 0; b < jcount; ++b)
    best = min_idx_combine(
        best, MinIdx{pval[(int64_t)b * Bp + i], pidx[(int64_t)b * Bp + i]});
  if (jcount == 0) {
    out_dist[i] = FLT_MAX;
    out_idx[i] = -1;
  } else {
    out_dist[i] = sqrtf(best.v);
    out_idx[i] = best.i;
  }
}

__global__ void grouped_kde_combine_kernel(
    const float2* __restrict__ pkde, const int* __restrict__ tseg,
    const int* __restrict__ nseg, int nclasses, int Bp,
    float* __restrict__ out_lse) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= Bp) return;
  int cls = 0;
  for (int c = 0; c < nclasses; ++c)
    if (tseg[c] <= i) cls = c;
  const int jcount = (nseg[cls + 1] - nseg[cls] + BN - 1) / BN;
  float mm = -FLT_MAX, ss = 0.f;
  for (int b = 0; b < jcount; ++b) {
    const float2 p = pkde[(int64_t)b * Bp + i];
    if (p.y <= 0.f) continue;
    if (p.x > mm) {
      ss = p.y + ((ss > 0.f) ? ss * __expf(mm - p.x) : 0.f);
      mm = p.x;
    } else {
      ss += p.y * __expf(p.x - mm);
    }
  }
  out_lse[i] = (ss > 0.f) ? (mm + __logf(ss)) : -FLT_MAX;
}

// ---------------------------------------------------------------------------
// bf16 grouped variant (fp32 accumulate). The activation traces come out of
// a bf16 forward, so bf16 operands lose almost nothing while the MFMA
// throughput ceiling rises ~16x over v_mfma_f32_32x32x2_f32 (the fp32 path
// measured ~95-119 TF; dense bf16 peak is ~2.5 PF). Fragment layout is the
// one hardware-verified by mfma_probe (resnet_fused.hip): A lane l holds
// A[i=l&15][k=(l>>4)*8+e]; B lane l holds B-col[j=l&15][same k]; D row =
// (l>>4)*4+reg, col = l&15. LDS tiles are [128][BBK+8] halves — the +8 pad
// makes the 16-lane fragment gathers stride 5 units (coprime to the bank
// period). Partial outputs are identical in format to the fp32 grouped
// kernel, so the same combine kernels run afterwards.
// ---------------------------------------------------------------------------

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4b = __attribute__((ext_vector_type(4))) float;

constexpr int BBK = 32;             // k per MFMA / staging step
// LDS halves per tile row. Must be a multiple of 8 (16-B-aligned b128
// reads); the pad past BBK sets the bank-group stride of the fragment
// gathers. TIP_BROW is sweepable on hardware (40 = 5-unit stride default).
#ifndef TIP_BROW
#define TIP_BROW 40
#endif
constexpr int BROW = TIP_BROW;

// Stage a [128, 32] bf16 tile: thread t loads row r = t>>1, halves
// kq = (t&1)*16 .. +15 (two 16-B units); OOB rows/ks stage zeros.
struct BStage {
  bf16x8 v[2];
};

TIP_DEV BStage bstage_load(const short* __restrict__ src, int rows, int K,
                           int row0, int k0) {
  BStage s;
  const int t = threadIdx.x;
  const int r = t >> 1;
  const int kq = (t & 1) * 16;
  const int grow = row0 + r;
  const bf16x8 zero = {0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
  for (int h = 0; h < 2; ++h) {
    const int gk = k0 + kq + h * 8;
    if (grow < rows && gk + 7 < K) {
      s.v[h] = *reinterpret_cast<const bf16x8*>(&src[(int64_t)grow * K + gk]);
    } else if (grow < rows) {
      bf16x8 tmp = zero;
      for (int e = 0; e < 8; ++e)
        if (gk + e < K) tmp[e] = src[(int64_t)grow * K + gk + e];
      s.v[h] = tmp;
    } else {
      s.v[h] = zero;
    }
  }
  return s;
}

TIP_DEV void bstage_write(short* lds, const BStage& s) {
  const int t = threadIdx.x;
  const int r = t >> 1;
  const int kq = (t & 1) * 16;
  *reinterpret_cast<bf16x8*>(&lds[r * BROW + kq]) = s.v[0];
  *reinterpret_cast<bf16x8*>(&lds[r * BROW + kq + 8]) = s.v[1];
}

template <int EPI>
__launch_bounds__(256, 2) __global__ void grouped_pairwise_bf16_kernel(
    const short* __restrict__ A,      // [Bp, K] bf16, class-sorted padded
    const short* __restrict__ B,      // [Ntot, K] bf16 class-concatenated
    const float* __restrict__ anorm,  // [Bp] norms of the bf16 values
    const float* __restrict__ bnorm,  // [Ntot]
    const int* __restrict__ tseg, const int* __restrict__ nseg,
    int nclasses, int Bp, int K,
    float* __restrict__ pmin_val, int* __restrict__ pmin_idx,
    float2* __restrict__ pkde) {
  __shared__ short As[128 * BROW];
  __shared__ short Bs[128 * BROW];
  __shared__ float red_v[BM][2];
  __shared__ int red_i[BM][2];
  __shared__ float red_s[BM][2];

  const int bi = blockIdx.y;
  const int bj = blockIdx.x;
  const int row0 = bi * BM;
  int cls = 0;
  for (int c = 0; c < nclasses; ++c)
    if (tseg[c] <= row0) cls = c;
  const int ncol0 = nseg[cls], ncol1 = nseg[cls + 1];
  const int col0 = ncol0 + bj * BN;
  if (col0 >= ncol1) return;

  const int lane = lane_id();
  const int wid = wave_id();
  const int wr = wid >> 1;  // row half (64 rows)
  const int wc = wid & 1;   // col half (64 cols)
  const int fj = lane & 15;
  const int fg = lane >> 4;

  f32x4b acc[4][4] = {};
  BStage ra = bstage_load(A, Bp, K, row0, 0);
  BStage rb = bstage_load(B, ncol1, K, col0, 0);
  bstage_write(As, ra);
  bstage_write(Bs, rb);
  __syncthreads();
  for (int k0 = 0; k0 < K; k0 += BBK) {
    const bool has_next = (k0 + BBK) < K;
    if (has_next) {
      ra = bstage_load(A, Bp, K, row0, k0 + BBK);
      rb = bstage_load(B, ncol1, K, col0, k0 + BBK);
    }
    bf16x8 af[4], bf[4];
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      const int i = wr * 64 + t * 16 + fj;
      const int j = wc * 64 + t * 16 + fj;
      af[t] = *reinterpret_cast<const bf16x8*>(&As[i * BROW + fg * 8]);
      bf[t] = *reinterpret_cast<const bf16x8*>(&Bs[j * BROW + fg * 8]);
    }
#pragma unroll
    for (int tr = 0; tr < 4; ++tr)
#pragma unroll
      for (int tc = 0; tc < 4; ++tc)
        acc[tr][tc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[tr], bf[tc], acc[tr][tc], 0, 0, 0);
    __syncthreads();
    if (has_next) {
      bstage_write(As, ra);
      bstage_write(Bs, rb);
    }
    __syncthreads();
  }

  // Epilogue. D tile (tr, tc): row i = wr*64 + tr*16 + fg*4 + reg,
  // col j = wc*64 + tc*16 + fj. Per row: reduce over this wave's 64 cols
  // (4 tc in registers + 16 fj lanes via xor shuffles), stash per col-half.
#pragma unroll
  for (int tr = 0; tr < 4; ++tr) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int row_local = wr * 64 + tr * 16 + fg * 4 + reg;
      const int gi = row0 + row_local;
      const float an = anorm[gi];
      MinIdx mi{FLT_MAX, 0x7fffffff};
      float dvals[4];
#pragma unroll
      for (int tc = 0; tc < 4; ++tc) {
        const int gj = col0 + wc * 64 + tc * 16 + fj;
        float d;
        if (gj < ncol1) {
          d = fmaxf(an + bnorm[gj] - 2.f * acc[tr][tc][reg], 0.f);
        } else {
          d = FLT_MAX;
        }
        dvals[tc] = d;
        mi = min_idx_combine(mi, MinIdx{d, gj});
      }
      // 16-lane (same fg) xor reduce; offsets < 16 stay in the group
      for (int off = 8; off >= 1; off >>= 1) {
        MinIdx o;
        o.v = __shfl_xor(mi.v, off);
        o.i = __shfl_xor(mi.i, off);
        mi = min_idx_combine(mi, o);
      }
      if (EPI == EPI_ROWMIN) {
        if (fj == 0) {
          red_v[row_local][wc] = mi.v;
          red_i[row_local][wc] = mi.i;
        }
      } else {
        const float tmax = (mi.v == FLT_MAX) ? -FLT_MAX : -0.5f * mi.v;
        float s = 0.f;
#pragma unroll
        for (int tc = 0; tc < 4; ++tc)
          if (dvals[tc] != FLT_MAX) s += __expf(-0.5f * dvals[tc] - tmax);
        for (int off = 8; off >= 1; off >>= 1) s += __shfl_xor(s, off);
        if (fj == 0) {
          red_v[row_local][wc] = tmax;
          red_s[row_local][wc] = s;
        }
      }
    }
  }
  __syncthreads();
  for (int r = threadIdx.x; r < BM; r += blockDim.x) {
    const int i = row0 + r;
    if (i >= Bp) continue;
    if (EPI == EPI_ROWMIN) {
      MinIdx best = min_idx_combine(
          MinIdx{red_v[r][0], red_i[r][0]}, MinIdx{red_v[r][1], red_i[r][1]});
      pmin_val[(int64_t)bj * Bp + i] = best.v;
      pmin_idx[(int64_t)bj * Bp + i] = best.i;
    } else {
      float m0 = red_v[r][0], s0 = red_s[r][0];
      float m1 = red_v[r][1], s1 = red_s[r][1];
      float mm, ss;
      if (m0 >= m1) {
        mm = m0;
        ss = s0 + ((s1 > 0.f) ? s1 * __expf(m1 - m0) : 0.f);
      } else {
        mm = m1;
        ss = s1 + ((s0 > 0.f) ? s0 * __expf(m0 - m1) : 0.f);
      }
      pkde[(int64_t)bj * Bp + i] = make_float2(mm, ss);
    }
  }
}

// Deterministic cross-block-column combines (ascending bj keeps np.argmin
// lowest-index tie semantics; fixed order keeps results bitwise stable).
__global__ void rowmin_combine_kernel(
    const float* __restrict__ pval, const int* __restrict__ pidx, int jblocks,
    int M, float* __restrict__ out_dist, int64_t* __restrict__ out_idx) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= M) return;
  MinIdx best{FLT_MAX, 0x7fffffff};
  for (int b = 0; b < jblocks; ++b) {
    MinIdx cand{pval[(int64_t)b * M + i], pidx[(int64_t)b * M + i]};
    best = min_idx_combine(best, cand);
  }
  out_dist[i] = sqrtf(best.v);
  out_idx[i] = best.i;
}

__global__ void kde_combine_kernel(
    const float2* __restrict__ pkde, int jblocks, int M,
    float* __restrict__ out_lse) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= M) return;
  float mm = -FLT_MAX, ss = 0.f;
  for (int b = 0; b < jblocks; ++b) {
    const float2 p = pkde[(int64_t)b * M + i];
    if (p.y <= 0.f) continue;
    if (p.x > mm) {
      ss = p.y + ((ss > 0.f) ? ss * __expf(mm - p.x) : 0.f);
      mm = p.x;
    } else {
      ss += p.y * __expf(p.x - mm);
    }
  }
  out_lse[i] = (ss > 0.f) ? (mm + __logf(ss)) : -FLT_MAX;
}

// Row squared norms: one wave per row.
__global__ void rownorm_kernel(
    const float* __restrict__ X, int rows, int K, float* __restrict__ out) {
  const int row = blockIdx.x * (blockDim.x / WAVE) + wave_id();
  if (row >= rows) return;
  const float* p = X + (int64_t)row * K;
  float s = 0.f;
  for (int k = lane_id(); k < K; k += WAVE) {
    const float v = p[k];
    s += v * v;
  }
  for (int off = 32; off >= 1; off >>= 1) s += __shfl_xor(s, off);
  if (lane_id() == 0) out[row] = s;
}

// ---- host-side launchers (called from bindings.cpp) ----

void launch_rownorm(const float* x, int rows, int k, float* out, hipStream_t s) {
  const int wpb = 4;
  rownorm_kernel<<<ceil_div(rows, wpb), wpb * WAVE, 0, s>>>(x, rows, k, out);
}

void launch_pairwise_full(const float* a, const float* b, const float* an,
                          const float* bn, int m, int n, int k, float* out,
                          hipStream_t s) {
  dim3 grid(ceil_div(n, BN), ceil_div(m, BM));
  pairwise_kernel<EPI_FULL><<<grid, 256, 0, s>>>(
      a, b, an, bn, m, n, k, out, nullptr, nullptr, nullptr);
}

void launch_pairwise_rowmin(const float* a, const float* b, const float* an,
                            const float* bn, int m, int n, int k,
                            float* pval, int* pidx, float* out_dist,
                            int64_t* out_idx, hipStream_t s) {
  const int jb = ceil_div(n, BN);
  dim3 grid(jb, ceil_div(m, BM));
  pairwise_kernel<EPI_ROWMIN><<<grid, 256, 0, s>>>(
      a, b, an, bn, m, n, k, nullptr, pval, pidx, nullptr);
  rowmin_combine_kernel<<<ceil_div(m, 256), 256, 0, s>>>(
      pval, pidx, jb, m, out_dist, out_idx);
}

void launch_pairwise_kde(const float* a, const float* b, const float* an,
                         const float* bn, int m, int n, int k, float2* pkde,
                         float* out_lse, hipStream_t s) {
  const int jb = ceil_div(n, BN);
  dim3 grid(jb, ceil_div(m, BM));
  pairwise_kernel<EPI_KDE><<<grid, 256, 0, s>>>(
      a, b, an, bn, m, n, k, nullptr, nullptr, nullptr, pkde);
  kde_combine_kernel<<<ceil_div(m, 256), 256, 0, s>>>(pkde, jb, m, out_lse);
}

// Grid geometry matching the kernel's TIP_GXY block->tile mapping.
static dim3 grouped_grid(int bp, int jb_max) {
#if TIP_GXY == 0
  return dim3(jb_max, ceil_div(bp, BM));
#elif TIP_GXY == 1
  return dim3(ceil_div(bp, BM), jb_max);
#else
  const int total = ceil_div(bp, BM) * jb_max;
  return dim3(8 * ceil_div(total, 8), 1);
#endif
}

void launch_grouped_rowmin(const float* a, const float* b, const float* an,
                           const float* bn, const int* tseg, const int* nseg,
                           int nclasses, int bp, int k, int jb_max,
                           float* pval, int* pidx, float* out_dist,
                           int64_t* out_idx, hipStream_t s) {
  grouped_pairwise_kernel<EPI_ROWMIN><<<grouped_grid(bp, jb_max), 256, 0, s>>>(
      a, b, an, bn, tseg, nseg, nclasses, bp, k, jb_max, pval, pidx, nullptr);
  grouped_rowmin_combine_kernel<<<ceil_div(bp, 256), 256, 0, s>>>(
      pval, pidx, tseg, nseg, nclasses, bp, out_dist, out_idx);
}

void launch_grouped_kde(const float* a, const float* b, const float* an,
                        const float* bn, const int* tseg, const int* nseg,
                        int nclasses, int bp, int k, int jb_max, float2* pkde,
                        float* out_lse, hipStream_t s) {
  grouped_pairwise_kernel<EPI_KDE><<<grouped_grid(bp, jb_max), 256, 0, s>>>(
      a, b, an, bn, tseg, nseg, nclasses, bp, k, jb_max, nullptr, nullptr,
      pkde);
  grouped_kde_combine_kernel<<<ceil_div(bp, 256), 256, 0, s>>>(
      pkde, tseg, nseg, nclasses, bp, out_lse);
}

void launch_grouped_rowmin_bf16(const short* a, const short* b,
                                const float* an, const float* bn,
                                const int* tseg, const int* nseg,
                                int nclasses, int bp, int k, int jb_max,
                                float* pval, int* pidx, float* out_dist,
                                int64_t* out_idx, hipStream_t s) {
  dim3 grid(jb_max, ceil_div(bp, BM));
  grouped_pairwise_bf16_kernel<EPI_ROWMIN><<<grid, 256, 0, s>>>(
      a, b, an, bn, tseg, nseg, nclasses, bp, k, pval, pidx, nullptr);
  grouped_rowmin_combine_kernel<<<ceil_div(bp, 256), 256, 0, s>>>(
      pval, pidx, tseg, nseg, nclasses, bp, out_dist, out_idx);
}

void launch_grouped_kde_bf16(const short* a, const short* b, const float* an,
                             const float* bn, const int* tseg,
                             const int* nseg, int nclasses, int bp, int k,
                             int jb_max, float2* pkde, float* out_lse,
                             hipStream_t s) {
  dim3 grid(jb_max, ceil_div(bp, BM));
  grouped_pairwise_bf16_kernel<EPI_KDE><<<grid, 256, 0, s>>>(
      a, b, an, bn, tseg, nseg, nclasses, bp, k, nullptr, nullptr, pkde);
  grouped_kde_combine_kernel<<<ceil_div(bp, 256), 256, 0, s>>>(
      pkde, tseg, nseg, nclasses, bp, out_lse);
}
