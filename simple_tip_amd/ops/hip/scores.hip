// Fused softmax-family uncertainty scores.
//
// One pass over the softmax outputs computes all four point-prediction
// scores in registers: negated max-softmax, negated PCS (top1-top2),
// softmax entropy (natural log) and DeepGini (1 - sum p^2). C is small
// (<= 64 for all case studies), so one thread per input row.

#include "tip_common.h"

#include <cfloat>

__global__ void softmax_scores_kernel(
    const float* __restrict__ probs, int n, int c,
    float* __restrict__ neg_max, float* __restrict__ neg_pcs,
    float* __restrict__ entropy, float* __restrict__ gini) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const float* p = probs + (int64_t)i * c;
  float p1 = -FLT_MAX, p2 = -FLT_MAX, ent = 0.f, sq = 0.f;
  for (int j = 0; j < c; ++j) {
    const float v = p[j];
    if (v > p1) {
      p2 = p1;
      p1 = v;
    } else if (v > p2) {
      p2 = v;
    }
    if (v > 0.f) ent -= v * __logf(v);
    sq += v * v;
  }
  if (c == 1) p2 = 0.f;
  neg_max[i] = -p1;
  neg_pcs[i] = -(p1 - p2);
  entropy[i] = ent;
  gini[i] = 1.f - sq;
}

void launch_softmax_scores(const float* probs, int n, int c, float* neg_max,
                           float* neg_pcs, float* entropy, float* gini,
                           hipStream_t s) {
  softmax_scores_kernel<<<ceil_div(n, 256), 256, 0, s>>>(
      probs, n, c, neg_max, neg_pcs, entropy, gini);
}
