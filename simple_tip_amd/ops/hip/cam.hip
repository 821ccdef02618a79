// Device-resident CAM greedy set cover.
//
// Each iteration: score[i] = popcount(profile[i] & uncovered) with a fused
// block argmax (ties -> lowest row, np.argmax semantics), a tiny combine
// kernel picks the winner and updates the uncovered mask in place. The host
// loop only reads back the (value, row) pair per iteration; the O(N*W)
// popcount sweep — the reference's per-iteration numpy hot loop
// (prioritizers.py:16-59) — never leaves the device.

#include "tip_common.h"

struct MaxIdxLL {
  long long v;
  int i;
};

TIP_DEV MaxIdxLL max_combine(MaxIdxLL a, MaxIdxLL b) {
  if (b.v > a.v || (b.v == a.v && b.i < a.i)) return b;
  return a;
}

// Rows per wave: coverage profiles are narrow (W <= ~100 words), so one
// row per wave under-uses lanes AND inflates the partial count the pick
// kernel must reduce. Each wave scans ROWS_PER_WAVE rows sequentially.
constexpr int CAM_ROWS_PER_WAVE = 4;
constexpr int CAM_WPB = 8;

// Each wave reduces its rows' (count, row) maxima; block writes one partial.
__global__ void cam_score_kernel(
    const unsigned long long* __restrict__ words, int rows, int W,
    const unsigned long long* __restrict__ uncovered,
    const unsigned char* __restrict__ used,
    long long* __restrict__ part_val, int* __restrict__ part_idx) {
  __shared__ long long sv[CAM_WPB];
  __shared__ int si[CAM_WPB];
  const int wid = wave_id();
  const int lane = lane_id();
  const int row0 =
      (blockIdx.x * CAM_WPB + wid) * CAM_ROWS_PER_WAVE;

  MaxIdxLL best{-1, 0x7fffffff};
  for (int q = 0; q < CAM_ROWS_PER_WAVE; ++q) {
    const int row = row0 + q;
    if (row >= rows || used[row]) continue;
    long long c = 0;
    const unsigned long long* r = words + (int64_t)row * W;
    for (int w = lane; w < W; w += WAVE) c += __popcll(r[w] & uncovered[w]);
    for (int off = 32; off >= 1; off >>= 1) c += __shfl_xor(c, off);
    best = max_combine(best, MaxIdxLL{c, row});
  }
  if (lane == 0) {
    sv[wid] = best.v;
    si[wid] = best.i;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    MaxIdxLL b{-1, 0x7fffffff};
    for (int w = 0; w < CAM_WPB; ++w)
      b = max_combine(b, MaxIdxLL{sv[w], si[w]});
    part_val[blockIdx.x] = b.v;
    part_idx[blockIdx.x] = b.i;
  }
}

// Single block: pick the global winner, mark it used, clear its newly
// covered columns. The partial scan is a 256-thread strided + tree
// reduction — the (max, min-index) combine is associative/commutative, so
// any reduction order preserves np.argmax's lowest-index tie rule. (A
// single-thread scan here made the whole greedy loop partial-count-bound:
// 454 ms for 20k rows x 1000 picks.)
__global__ void cam_pick_kernel(
    const long long* __restrict__ part_val, const int* __restrict__ part_idx,
    int nparts, const unsigned long long* __restrict__ words, int W,
    unsigned long long* __restrict__ uncovered,
    unsigned char* __restrict__ used,
    long long* __restrict__ result) {  // result = {picked_row, newly_covered}
  __shared__ long long sv[256];
  __shared__ int si[256];
  MaxIdxLL b{-1, 0x7fffffff};
  for (int p = threadIdx.x; p < nparts; p += blockDim.x)
    b = max_combine(b, MaxIdxLL{part_val[p], part_idx[p]});
  sv[threadIdx.x] = b.v;
  si[threadIdx.x] = b.i;
  __syncthreads();
  for (int off = blockDim.x >> 1; off; off >>= 1) {
    if (threadIdx.x < off) {
      MaxIdxLL m = max_combine(
          MaxIdxLL{sv[threadIdx.x], si[threadIdx.x]},
          MaxIdxLL{sv[threadIdx.x + off], si[threadIdx.x + off]});
      sv[threadIdx.x] = m.v;
      si[threadIdx.x] = m.i;
    }
    __syncthreads();
  }
  const long long win_v = sv[0];
  const int win_i = si[0];
  if (threadIdx.x == 0) {
    result[0] = (win_v > 0) ? win_i : -1;
    result[1] = win_v;
    if (win_v > 0) used[win_i] = 1;
  }
  if (win_v > 0) {
    const unsigned long long* r = words + (int64_t)win_i * W;
    for (int w = threadIdx.x; w < W; w += blockDim.x) uncovered[w] &= ~r[w];
  }
}

void launch_cam_iteration(const unsigned long long* words, int rows, int W,
                          unsigned long long* uncovered, unsigned char* used,
                          long long* part_val, int* part_idx,
                          long long* result, hipStream_t s) {
  const int nblocks = ceil_div(rows, CAM_WPB * CAM_ROWS_PER_WAVE);
  cam_score_kernel<<<nblocks, CAM_WPB * WAVE, 0, s>>>(
      words, rows, W, uncovered, used, part_val, part_idx);
  cam_pick_kernel<<<1, 256, 0, s>>>(
      part_val, part_idx, nblocks, words, W, uncovered, used, result);
}

// Persistent multi-block CAM — the production path. Alternatives measured
// and rejected (profiles/r02_cam_ladder.md): the per-iteration two-kernel
// form above (kept as the wide-mask fallback) is dispatch-bound at
// ~200 us/pick, and a single-block whole-loop kernel was sweep-bound on
// one CU (~340 us/pick). This kernel keeps CAM_NB blocks RESIDENT (1 block
// per CU is guaranteed co-residency on 256 CUs, so the software grid
// barrier cannot deadlock) and runs the whole greedy loop with two
// barriers per iteration: parallel sweep -> block partials -> barrier ->
// block 0 picks + updates the global uncovered mask -> barrier.
// Rows whose remaining coverage hits 0 are marked dead (uncovered only
// shrinks, so c==0 is permanent) and skip future sweeps.
constexpr int CAM_NB = 64;

TIP_DEV void cam_grid_barrier(int* ctr, int* round_) {
  __syncthreads();
  if (threadIdx.x == 0) {
    __threadfence();
    __hip_atomic_fetch_add(ctr, 1, __ATOMIC_RELEASE,
                           __HIP_MEMORY_SCOPE_AGENT);
    const int target = (++(*round_)) * CAM_NB;
    // spin on relaxed LOADS: an RMW spin serializes all waiters on one
    // cacheline and cost ~100 us per barrier at 64 blocks
    while (__hip_atomic_load(ctr, __ATOMIC_ACQUIRE,
                             __HIP_MEMORY_SCOPE_AGENT) < target)
      __builtin_amdgcn_s_sleep(2);
  }
  __syncthreads();
}

__launch_bounds__(256) __global__ void cam_greedy_coop_kernel(
    const unsigned long long* __restrict__ words, int rows, int W,
    unsigned long long init_tail_mask, unsigned char* __restrict__ used,
    unsigned long long* __restrict__ uncovered,  // [W] global
    long long* __restrict__ part_val, int* __restrict__ part_idx,
    long long* __restrict__ order_out, int* __restrict__ n_out,
    int* __restrict__ barrier_ctr, int* __restrict__ win_slot) {
  __shared__ long long sv[4];
  __shared__ int si[4];
  constexpr int LDS_W = 2048;  // up to 128K profile bits staged in LDS
  __shared__ unsigned long long sunc[LDS_W];
  const bool use_lds = W <= LDS_W;
  const int tid = threadIdx.x;
  const int lane = lane_id();
  const int wid = wave_id();
  const int gthread = blockIdx.x * blockDim.x + tid;
  const int nthreads = CAM_NB * blockDim.x;
  int round_ = 0;
  if (gthread < W) uncovered[gthread] = (gthread == W - 1) ? init_tail_mask : ~0ull;
  if (gthread == 0) {
    *n_out = 0;
    *win_slot = 0;
  }
  cam_grid_barrier(barrier_ctr, &round_);

  for (;;) {
    if (use_lds) {  // per-block copy: sweep reads hit LDS, not L2
      for (int w = tid; w < W; w += blockDim.x) sunc[w] = uncovered[w];
      __syncthreads();
    }
    const unsigned long long* unc = use_lds ? sunc : uncovered;
    MaxIdxLL best{-1, 0x7fffffff};
    for (int row = gthread; row < rows; row += nthreads) {
      if (used[row]) continue;
      long long c = 0;
      const unsigned long long* r = words + (int64_t)row * W;
      for (int w = 0; w < W; ++w) c += __popcll(r[w] & unc[w]);
      if (c == 0) {
        used[row] = 1;  // permanently dead: uncovered only shrinks
        continue;
      }
      best = max_combine(best, MaxIdxLL{c, row});
    }
    for (int off = 32; off >= 1; off >>= 1) {
      MaxIdxLL o{__shfl_xor(best.v, off), __shfl_xor(best.i, off)};
      best = max_combine(best, o);
    }
    if (lane == 0) {
      sv[wid] = best.v;
      si[wid] = best.i;
    }
    __syncthreads();
    if (tid == 0) {
      MaxIdxLL b{-1, 0x7fffffff};
      for (int w = 0; w < (int)(blockDim.x / WAVE); ++w)
        b = max_combine(b, MaxIdxLL{sv[w], si[w]});
      part_val[blockIdx.x] = b.v;
      part_idx[blockIdx.x] = b.i;
    }
    cam_grid_barrier(barrier_ctr, &round_);
    if (blockIdx.x == 0) {
      if (tid == 0) {
        MaxIdxLL b{-1, 0x7fffffff};
        for (int p = 0; p < CAM_NB; ++p)
          b = max_combine(b, MaxIdxLL{part_val[p], part_idx[p]});
        if (b.v > 0) {
          order_out[(*n_out)++] = b.i;
          used[b.i] = 1;
          *win_slot = b.i;
        } else {
          *win_slot = -1;
        }
      }
      __syncthreads();
      const int win = *win_slot;
      if (win >= 0) {
        const unsigned long long* r = words + (int64_t)win * W;
        for (int w = tid; w < W; w += blockDim.x) uncovered[w] &= ~r[w];
      }
    }
    cam_grid_barrier(barrier_ctr, &round_);
    if (*win_slot < 0 || *n_out >= rows) break;
  }
}

void launch_cam_greedy_coop(const unsigned long long* words, int rows, int W,
                            unsigned long long init_tail_mask,
                            unsigned char* used,
                            unsigned long long* uncovered, long long* part_val,
                            int* part_idx, long long* order_out, int* n_out,
                            int* barrier_ctr, int* win_slot, hipStream_t s) {
  cam_greedy_coop_kernel<<<CAM_NB, 256, 0, s>>>(
      words, rows, W, init_tail_mask, used, uncovered, part_val, part_idx,
      order_out, n_out, barrier_ctr, win_slot);
}
