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

// One wave per row-group step; block reduces its rows' (count, row) maxima.
__global__ void cam_score_kernel(
    const unsigned long long* __restrict__ words, int rows, int W,
    const unsigned long long* __restrict__ uncovered,
    const unsigned char* __restrict__ used,
    long long* __restrict__ part_val, int* __restrict__ part_idx) {
  __shared__ long long sv[8];
  __shared__ int si[8];
  const int wid = wave_id();
  const int lane = lane_id();
  const int waves_per_block = blockDim.x / WAVE;
  const int row = blockIdx.x * waves_per_block + wid;

  MaxIdxLL best{-1, 0x7fffffff};
  if (row < rows && !used[row]) {
    long long c = 0;
    const unsigned long long* r = words + (int64_t)row * W;
    for (int w = lane; w < W; w += WAVE) c += __popcll(r[w] & uncovered[w]);
    for (int off = 32; off >= 1; off >>= 1) c += __shfl_xor(c, off);
    best = MaxIdxLL{c, row};
  }
  if (lane == 0) {
    sv[wid] = best.v;
    si[wid] = best.i;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    MaxIdxLL b{-1, 0x7fffffff};
    for (int w = 0; w < waves_per_block; ++w)
      b = max_combine(b, MaxIdxLL{sv[w], si[w]});
    part_val[blockIdx.x] = b.v;
    part_idx[blockIdx.x] = b.i;
  }
}

// Single block: pick the global winner (ascending block order keeps the
// lowest-index tie rule), mark it used, clear its newly covered columns.
__global__ void cam_pick_kernel(
    const long long* __restrict__ part_val, const int* __restrict__ part_idx,
    int nparts, const unsigned long long* __restrict__ words, int W,
    unsigned long long* __restrict__ uncovered,
    unsigned char* __restrict__ used,
    long long* __restrict__ result) {  // result = {picked_row, newly_covered}
  __shared__ int s_row;
  __shared__ long long s_val;
  if (threadIdx.x == 0) {
    MaxIdxLL b{-1, 0x7fffffff};
    for (int p = 0; p < nparts; ++p)
      b = max_combine(b, MaxIdxLL{part_val[p], part_idx[p]});
    s_row = b.i;
    s_val = b.v;
    result[0] = (b.v > 0) ? b.i : -1;
    result[1] = b.v;
    if (b.v > 0) used[b.i] = 1;
  }
  __syncthreads();
  if (s_val > 0) {
    const unsigned long long* r = words + (int64_t)s_row * W;
    for (int w = threadIdx.x; w < W; w += blockDim.x) uncovered[w] &= ~r[w];
  }
}

void launch_cam_iteration(const unsigned long long* words, int rows, int W,
                          unsigned long long* uncovered, unsigned char* used,
                          long long* part_val, int* part_idx,
                          long long* result, hipStream_t s) {
  const int wpb = 8;
  const int nblocks = ceil_div(rows, wpb);
  cam_score_kernel<<<nblocks, wpb * WAVE, 0, s>>>(
      words, rows, W, uncovered, used, part_val, part_idx);
  cam_pick_kernel<<<1, 256, 0, s>>>(
      part_val, part_idx, nblocks, words, W, uncovered, used, result);
}
