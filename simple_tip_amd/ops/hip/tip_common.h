// Shared helpers for the simple_tip_amd HIP/CDNA4 (gfx950) kernels.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

#define TIP_DEV __device__ __forceinline__

constexpr int WAVE = 64;  // CDNA wavefront width

TIP_DEV int lane_id() { return threadIdx.x & (WAVE - 1); }
TIP_DEV int wave_id() { return threadIdx.x / WAVE; }

static inline int ceil_div(int64_t a, int64_t b) { return (int)((a + b - 1) / b); }

// (value, index) minimum with np.argmin tie-breaking (lowest index wins).
struct MinIdx {
  float v;
  int i;
};

TIP_DEV MinIdx min_idx_combine(MinIdx a, MinIdx b) {
  if (b.v < a.v || (b.v == a.v && b.i < a.i)) return b;
  return a;
}

// Wave-half (32-lane) xor-reduction of (value, index) minima. After the loop
// every lane of the 32-lane group holds the group's minimum.
TIP_DEV MinIdx half_reduce_min(MinIdx m) {
  for (int off = 16; off >= 1; off >>= 1) {
    MinIdx o;
    o.v = __shfl_xor(m.v, off);
    o.i = __shfl_xor(m.i, off);
    m = min_idx_combine(m, o);
  }
  return m;
}

TIP_DEV float half_reduce_sum(float v) {
  for (int off = 16; off >= 1; off >>= 1) v += __shfl_xor(v, off);
  return v;
}
