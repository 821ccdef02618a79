// Coverage-profile bitmap kernels for gfx950.
//
// Profiles are packed 64-bit words (bit j of word w = profile column
// w*64 + j, matching ops/fallback.py). The wave64 __ballot primitive packs
// one word per wave step — each lane evaluates one profile BIT, the ballot
// IS the bitmap word. Scores (popcount) are fused into the same pass.

#include "tip_common.h"

#include <cfloat>

enum ProfMode {
  PROF_NAC = 0,   // acts[col] > thr                      (S=1)
  PROF_SNAC = 1,  // acts[col] >= hi[col]                 (S=1)
  PROF_NBC = 2,   // s=0: acts <= lo; s=1: acts >= hi     (S=2)
  PROF_KMNC = 3,  // lo+jump*s <= a < lo+jump*(s+1)       (S=sections)
  PROF_PACK = 4,  // bool input != 0                      (S=1)
};

// One wave per row; lanes sweep the row's profile bits 64 at a time.
// words: [rows, W]; scores: [rows] popcount (may be null).
template <int MODE>
__global__ void profile_kernel(
    const float* __restrict__ acts,  // [rows, K] (or bool bytes for PACK)
    const unsigned char* __restrict__ bools,
    const float* __restrict__ lo,    // [K] per-neuron lower bound / mins
    const float* __restrict__ hi,    // [K] per-neuron upper bound / maxs
    float thr, int sections,
    int rows, int K, int W,
    unsigned long long* __restrict__ words,
    long long* __restrict__ scores) {
  const int row = blockIdx.x * (blockDim.x / WAVE) + wave_id();
  if (row >= rows) return;
  const int lane = lane_id();
  const float* arow = acts ? acts + (int64_t)row * K : nullptr;
  long long count = 0;
  for (int w = 0; w < W; ++w) {
    const int bit = w * WAVE + lane;
    bool pred = false;
    if (MODE == PROF_PACK) {
      if (bit < K) pred = bools[(int64_t)row * K + bit] != 0;
    } else {
      const int col = (MODE == PROF_NBC)    ? bit >> 1
                      : (MODE == PROF_KMNC) ? bit / sections
                                            : bit;
      if (col < K) {
        const float a = arow[col];
        if (MODE == PROF_NAC) {
          pred = a > thr;
        } else if (MODE == PROF_SNAC) {
          pred = a >= hi[col];
        } else if (MODE == PROF_NBC) {
          pred = (bit & 1) ? (a >= hi[col]) : (a <= lo[col]);
        } else {  // KMNC: half-open k-section membership
          const int s = bit - col * sections;
          const float jump = (hi[col] - lo[col]) / sections;
          const float b0 = lo[col] + jump * s;
          const float b1 = lo[col] + jump * (s + 1);
          pred = (b0 <= a) && (a < b1);
        }
      }
    }
    const unsigned long long word = __ballot(pred);
    if (lane == 0) words[(int64_t)row * W + w] = word;
    count += __popcll(word);  // wave-uniform after ballot
  }
  if (lane == 0 && scores) scores[row] = count;
}

// Row-wise popcount of an existing packed profile.
__global__ void popcount_kernel(
    const unsigned long long* __restrict__ words, int rows, int W,
    long long* __restrict__ out) {
  const int row = blockIdx.x * (blockDim.x / WAVE) + wave_id();
  if (row >= rows) return;
  long long c = 0;
  for (int w = lane_id(); w < W; w += WAVE) c += __popcll(words[(int64_t)row * W + w]);
  for (int off = 32; off >= 1; off >>= 1) c += __shfl_xor(c, off);
  if (lane_id() == 0) out[row] = c;
}

// TKNC: top-k (k<=4) neurons per layer, bits scattered at bit_offset+col.
// One wave per row; each lane keeps a local sorted top-k over its strided
// columns, then a wave tree-merge; ties prefer the lower column index
// (torch.topk semantics).
__global__ void tknc_kernel(
    const float* __restrict__ layer,  // [rows, K]
    int rows, int K, int k, int bit_offset, int W,
    unsigned long long* __restrict__ words) {
  const int row = blockIdx.x * (blockDim.x / WAVE) + wave_id();
  if (row >= rows) return;
  const int lane = lane_id();
  const float* arow = layer + (int64_t)row * K;

  float tv[4];
  int ti[4];
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    tv[q] = -FLT_MAX;
    ti[q] = 0x7fffffff;
  }
  for (int c = lane; c < K; c += WAVE) {
    const float v = arow[c];
    // insertion into the sorted (desc by v, asc by idx) quad
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      if (v > tv[q] || (v == tv[q] && c < ti[q])) {
        for (int r = 3; r > q; --r) {
          tv[r] = tv[r - 1];
          ti[r] = ti[r - 1];
        }
        tv[q] = v;
        ti[q] = c;
        break;
      }
    }
  }
  // wave merge: fold halves together
  for (int off = 32; off >= 1; off >>= 1) {
    float ov[4];
    int oi[4];
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      ov[q] = __shfl_xor(tv[q], off);
      oi[q] = __shfl_xor(ti[q], off);
    }
    // merge two sorted quads -> keep top 4
    float mv[4];
    int mi[4];
    int a = 0, b = 0;
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      const bool take_a =
          (tv[a] > ov[b]) || (tv[a] == ov[b] && ti[a] < oi[b]);
      if (take_a) {
        mv[q] = tv[a];
        mi[q] = ti[a];
        ++a;
      } else {
        mv[q] = ov[b];
        mi[q] = oi[b];
        ++b;
      }
    }
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      tv[q] = mv[q];
      ti[q] = mi[q];
    }
  }
  if (lane == 0) {
    for (int q = 0; q < k && q < K; ++q) {
      const int bit = bit_offset + ti[q];
      atomicOr(&words[(int64_t)row * W + (bit >> 6)],
               1ull << (bit & 63));
    }
  }
}

// Surprise-coverage bucketize: one thread per value; thresholds are fp64
// (reference uses a float64 linspace; bucket membership must match the CPU
// path bit-for-bit). Only bucket s with thr[s] <= v < thr[s+1] is set.
__global__ void bucketize_kernel(
    const double* __restrict__ values, const double* __restrict__ thr,
    int n, int sections, int W, unsigned long long* __restrict__ words) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const double v = values[i];
  // binary search: rightmost s with thr[s] <= v
  int lo = 0, hi = sections + 1;  // thresholds has sections+1 entries
  while (lo < hi) {
    const int mid = (lo + hi) >> 1;
    if (thr[mid] <= v)
      lo = mid + 1;
    else
      hi = mid;
  }
  const int s = lo - 1;
  for (int w = 0; w < W; ++w) words[(int64_t)i * W + w] = 0ull;
  if (s >= 0 && s < sections && v < thr[s + 1])
    words[(int64_t)i * W + (s >> 6)] = 1ull << (s & 63);
}

// ---- launchers ----

static constexpr int WPB = 4;  // waves per block for row-per-wave kernels

void launch_profile(int mode, const float* acts, const unsigned char* bools,
                    const float* lo, const float* hi, float thr, int sections,
                    int rows, int K, int W, unsigned long long* words,
                    long long* scores, hipStream_t s) {
  dim3 grid(ceil_div(rows, WPB));
  dim3 block(WPB * WAVE);
  switch (mode) {
    case PROF_NAC:
      profile_kernel<PROF_NAC><<<grid, block, 0, s>>>(
          acts, bools, lo, hi, thr, sections, rows, K, W, words, scores);
      break;
    case PROF_SNAC:
      profile_kernel<PROF_SNAC><<<grid, block, 0, s>>>(
          acts, bools, lo, hi, thr, sections, rows, K, W, words, scores);
      break;
    case PROF_NBC:
      profile_kernel<PROF_NBC><<<grid, block, 0, s>>>(
          acts, bools, lo, hi, thr, sections, rows, K, W, words, scores);
      break;
    case PROF_KMNC:
      profile_kernel<PROF_KMNC><<<grid, block, 0, s>>>(
          acts, bools, lo, hi, thr, sections, rows, K, W, words, scores);
      break;
    case PROF_PACK:
      profile_kernel<PROF_PACK><<<grid, block, 0, s>>>(
          acts, bools, lo, hi, thr, sections, rows, K, W, words, scores);
      break;
  }
}

void launch_popcount(const unsigned long long* words, int rows, int W,
                     long long* out, hipStream_t s) {
  popcount_kernel<<<ceil_div(rows, WPB), WPB * WAVE, 0, s>>>(words, rows, W, out);
}

void launch_tknc(const float* layer, int rows, int K, int k, int bit_offset,
                 int W, unsigned long long* words, hipStream_t s) {
  tknc_kernel<<<ceil_div(rows, WPB), WPB * WAVE, 0, s>>>(
      layer, rows, K, k, bit_offset, W, words);
}

void launch_bucketize(const double* values, const double* thr, int n,
                      int sections, int W, unsigned long long* words,
                      hipStream_t s) {
  bucketize_kernel<<<ceil_div(n, 256), 256, 0, s>>>(
      values, thr, n, sections, W, words);
}
