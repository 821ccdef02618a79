// Standalone (torch-free) kernel probe for rocprofv3 PMC runs.
//
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 -I simple_tip_amd/ops/hip \
//          scripts/kernel_probe.hip -o gpurun_out/kernel_probe
// Run:   ./kernel_probe pairwise|resblock [iters]
//
// Includes the production kernel sources directly so the profiled code is
// byte-identical to what the extension ships.

#include "../simple_tip_amd/ops/hip/pairwise.hip"
#include "../simple_tip_amd/ops/hip/resnet_fused.hip"

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

#define CHECK(x)                                                    \
  do {                                                              \
    hipError_t e = (x);                                             \
    if (e != hipSuccess) {                                          \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(e), \
              __FILE__, __LINE__);                                  \
      exit(1);                                                      \
    }                                                               \
  } while (0)

static float frand() { return 2.f * rand() / RAND_MAX - 1.f; }

// process-independent deterministic fill (rand() is per-process anyway,
// but this removes every doubt when chasing cross-process determinism)
static unsigned lcg_state = 12345u;
static float frand_det() {
  lcg_state = lcg_state * 1664525u + 1013904223u;
  return 2.f * (lcg_state >> 8) / 16777216.f - 1.f;
}

static void run_pairwise(int iters) {
  const int m = 4096, n = 8192, k = 2048;
  float *a, *b, *an, *bn, *pval, *dist;
  int* pidx;
  int64_t* idx;
  const int jb = (n + 127) / 128;
  CHECK(hipMalloc(&a, (size_t)m * k * 4));
  CHECK(hipMalloc(&b, (size_t)n * k * 4));
  CHECK(hipMalloc(&an, m * 4));
  CHECK(hipMalloc(&bn, n * 4));
  CHECK(hipMalloc(&pval, (size_t)jb * m * 4));
  CHECK(hipMalloc(&pidx, (size_t)jb * m * 4));
  CHECK(hipMalloc(&dist, m * 4));
  CHECK(hipMalloc(&idx, m * 8));
  std::vector<float> host((size_t)n * k);
  for (auto& v : host) v = frand();
  CHECK(hipMemcpy(a, host.data(), (size_t)m * k * 4, hipMemcpyHostToDevice));
  CHECK(hipMemcpy(b, host.data(), (size_t)n * k * 4, hipMemcpyHostToDevice));
  launch_rownorm(a, m, k, an, 0);
  launch_rownorm(b, n, k, bn, 0);
  CHECK(hipDeviceSynchronize());
  hipEvent_t t0, t1;
  hipEventCreate(&t0);
  hipEventCreate(&t1);
  hipEventRecord(t0);
  for (int i = 0; i < iters; ++i)
    launch_pairwise_rowmin(a, b, an, bn, m, n, k, pval, pidx, dist, idx, 0);
  hipEventRecord(t1);
  CHECK(hipDeviceSynchronize());
  float ms;
  hipEventElapsedTime(&ms, t0, t1);
  const double tf = 2.0 * m * n * k * iters / (ms / 1e3) / 1e12;
  printf("pairwise_rowmin %dx%dx%d: %.2f ms/iter, %.1f TF\n", m, n, k,
         ms / iters, tf);
}

static void run_resblock(int iters, int variant) {
  const int batch = 4096;
  const int HH[3] = {32, 16, 8}, CC[3] = {16, 32, 64};
  const int H = HH[variant], W = HH[variant], C = CC[variant];
  const size_t plane = (size_t)H * W * C;
  short *gin, *gout, *w1, *w2;
  float *b1, *b2;
  const int KSTEPS = (9 * C + 31) / 32;
  CHECK(hipMalloc(&gin, batch * plane * 2));
  CHECK(hipMalloc(&gout, batch * plane * 2));
  CHECK(hipMalloc(&w1, (size_t)KSTEPS * 64 * 8 * 2));
  CHECK(hipMalloc(&w2, (size_t)KSTEPS * 64 * 8 * 2));
  CHECK(hipMalloc(&b1, C * 4));
  CHECK(hipMalloc(&b2, C * 4));
  // bf16 random fill via float->bf16 truncation on host
  std::vector<short> host(batch * plane);
  for (auto& v : host) {
    float f = frand();
    unsigned u;
    memcpy(&u, &f, 4);
    v = (short)(u >> 16);
  }
  CHECK(hipMemcpy(gin, host.data(), batch * plane * 2, hipMemcpyHostToDevice));
  CHECK(hipMemcpy(w1, host.data(), (size_t)KSTEPS * 64 * 8 * 2, hipMemcpyHostToDevice));
  CHECK(hipMemcpy(w2, host.data(), (size_t)KSTEPS * 64 * 8 * 2, hipMemcpyHostToDevice));
  CHECK(hipMemset(b1, 0, C * 4));
  CHECK(hipMemset(b2, 0, C * 4));
  launch_resblock(variant, batch, gin, gout, w1, b1, w2, b2, 0);
  CHECK(hipDeviceSynchronize());
  hipEvent_t t0, t1;
  hipEventCreate(&t0);
  hipEventCreate(&t1);
  hipEventRecord(t0);
  for (int i = 0; i < iters; ++i)
    launch_resblock(variant, batch, gin, gout, w1, b1, w2, b2, 0);
  hipEventRecord(t1);
  CHECK(hipDeviceSynchronize());
  float ms;
  hipEventElapsedTime(&ms, t0, t1);
  const double flops = 2.0 * batch * 2 * (double)(H * W) * C * (9 * C);
  printf("resblock<%d,%d,%d> b=%d: %.3f ms/iter, %.1f TF\n", H, W, C, batch,
         ms / iters, flops * iters / (ms / 1e3) / 1e12);
}

// Grouped (segmented) rowmin at the bench's serving geometry: nclasses
// class-sorted 128-padded test segments scored against per-class train
// segments. Defaults mirror bench.py (10 classes, ~1024 test rows and
// ~1500 train rows per class, K = 4096).
static void run_grouped(int iters, int perclass_test, int perclass_train,
                        int k) {
  const int C = 10;
  const int pc_pad = (perclass_test + 127) / 128 * 128;
  const int bp = C * pc_pad;
  const int ntot = C * perclass_train;
  const int jb_max = (perclass_train + BN - 1) / BN;
  float *a, *b, *an, *bn, *pval, *dist;
  int* pidx;
  int64_t* idx;
  int *tseg, *nseg;
  CHECK(hipMalloc(&a, (size_t)bp * k * 4));
  CHECK(hipMalloc(&b, (size_t)ntot * k * 4));
  CHECK(hipMalloc(&an, bp * 4));
  CHECK(hipMalloc(&bn, ntot * 4));
  CHECK(hipMalloc(&pval, (size_t)jb_max * bp * 4));
  CHECK(hipMalloc(&pidx, (size_t)jb_max * bp * 4));
  CHECK(hipMalloc(&dist, bp * 4));
  CHECK(hipMalloc(&idx, bp * 8));
  CHECK(hipMalloc(&tseg, (C + 1) * 4));
  CHECK(hipMalloc(&nseg, (C + 1) * 4));
  std::vector<int> th(C + 1), nh(C + 1);
  for (int c = 0; c <= C; ++c) {
    th[c] = c * pc_pad;
    nh[c] = c * perclass_train;
  }
  CHECK(hipMemcpy(tseg, th.data(), (C + 1) * 4, hipMemcpyHostToDevice));
  CHECK(hipMemcpy(nseg, nh.data(), (C + 1) * 4, hipMemcpyHostToDevice));
  std::vector<float> host((size_t)(bp > ntot ? bp : ntot) * k);
  for (auto& v : host) v = frand_det();
  {
    double hsum = 0;
    for (size_t i = 0; i < host.size(); i += 97) hsum += host[i];
    printf("  host data checksum %.9e (n=%zu)\n", hsum, host.size());
  }
  CHECK(hipMemcpy(a, host.data(), (size_t)bp * k * 4, hipMemcpyHostToDevice));
  CHECK(hipMemcpy(b, host.data(), (size_t)ntot * k * 4, hipMemcpyHostToDevice));
  launch_rownorm(a, bp, k, an, 0);
  launch_rownorm(b, ntot, k, bn, 0);
  CHECK(hipDeviceSynchronize());
  {
    std::vector<float> anh(bp);
    CHECK(hipMemcpy(anh.data(), an, bp * 4, hipMemcpyDeviceToHost));
    double s = 0;
    for (int i = 0; i < bp; ++i) s += anh[i];
    printf("  anorm checksum %.9e\n", s);
  }
  hipEvent_t t0, t1;
  hipEventCreate(&t0);
  hipEventCreate(&t1);
  launch_grouped_rowmin(a, b, an, bn, tseg, nseg, C, bp, k, jb_max, pval,
                        pidx, dist, idx, 0);
  CHECK(hipDeviceSynchronize());
  {
    std::vector<float> dh(bp);
    std::vector<int64_t> ih(bp);
    CHECK(hipMemcpy(dh.data(), dist, bp * 4, hipMemcpyDeviceToHost));
    CHECK(hipMemcpy(ih.data(), idx, bp * 8, hipMemcpyDeviceToHost));
    double dsum = 0;
    long long isum = 0;
    for (int i = 0; i < bp; ++i) {
      dsum += dh[i];
      isum += ih[i];
    }
    printf("  warmup checksum d=%.6e i=%lld\n", dsum, isum);
    // dense per-class ground truth; report mismatch structure
    float* dref;
    int64_t* iref;
    CHECK(hipMalloc(&dref, bp * 4));
    CHECK(hipMalloc(&iref, bp * 8));
    for (int c = 0; c < C; ++c) {
      launch_pairwise_rowmin(a + (size_t)th[c] * k, b + (size_t)nh[c] * k,
                             an + th[c], bn + nh[c], pc_pad, perclass_train,
                             k, pval, pidx, dref + th[c], iref + th[c], 0);
      CHECK(hipDeviceSynchronize());
    }
    std::vector<float> dr(bp);
    std::vector<int64_t> ir(bp);
    CHECK(hipMemcpy(dr.data(), dref, bp * 4, hipMemcpyDeviceToHost));
    CHECK(hipMemcpy(ir.data(), iref, bp * 8, hipMemcpyDeviceToHost));
    int mm = 0, firstbad = -1, lastbad = -1;
    for (int i = 0; i < bp; ++i) {
      const int c = i / pc_pad;
      if (dh[i] != dr[i] || ih[i] != ir[i] + nh[c]) {
        ++mm;
        if (firstbad < 0) firstbad = i;
        lastbad = i;
      }
    }
    printf("  vs-dense mismatches=%d first=%d last=%d\n", mm, firstbad,
           lastbad);
    if (firstbad >= 0)
      printf("  first: d=%.9g ref=%.9g i=%lld iref=%lld\n", dh[firstbad],
             dr[firstbad], (long long)ih[firstbad],
             (long long)(ir[firstbad] + nh[firstbad / pc_pad]));
    hipFree(dref);
    hipFree(iref);
    // re-run the grouped warmup so the timing loop below is unaffected
    launch_grouped_rowmin(a, b, an, bn, tseg, nseg, C, bp, k, jb_max, pval,
                          pidx, dist, idx, 0);
    CHECK(hipDeviceSynchronize());
  }
  hipEventRecord(t0);
  for (int i = 0; i < iters; ++i)
    launch_grouped_rowmin(a, b, an, bn, tseg, nseg, C, bp, k, jb_max, pval,
                          pidx, dist, idx, 0);
  hipEventRecord(t1);
  CHECK(hipDeviceSynchronize());
  float ms;
  hipEventElapsedTime(&ms, t0, t1);
  const double tf =
      2.0 * bp * perclass_train * k * iters / (ms / 1e3) / 1e12;
  // cross-variant correctness checksum (mapping changes must not change
  // results): sum of distances + sum of argmin indices
  std::vector<float> dh(bp);
  std::vector<int64_t> ih(bp);
  CHECK(hipMemcpy(dh.data(), dist, bp * 4, hipMemcpyDeviceToHost));
  CHECK(hipMemcpy(ih.data(), idx, bp * 8, hipMemcpyDeviceToHost));
  double dsum = 0;
  long long isum = 0;
  for (int i = 0; i < bp; ++i) {
    dsum += dh[i];
    isum += ih[i];
  }
  printf("grouped_rowmin C=%d bp=%d ntr/cls=%d k=%d (GXY=%d): %.3f ms/iter, "
         "%.1f TF  checksum d=%.6e i=%lld\n",
         C, bp, perclass_train, k, TIP_GXY, ms / iters, tf, dsum, isum);
}

// Determinism + correctness check: launch the grouped rowmin twice into
// separate outputs, diff them, and diff run 1 against the dense per-class
// rowmin reference.
static void run_verify(int perclass_test, int perclass_train, int k) {
  const int C = 10;
  const int pc_pad = (perclass_test + 127) / 128 * 128;
  const int bp = C * pc_pad;
  const int ntot = C * perclass_train;
  const int jb_max = (perclass_train + BN - 1) / BN;
  float *a, *b, *an, *bn, *pval, *dist1, *dist2, *dist_ref;
  int* pidx;
  int64_t *idx1, *idx2, *idx_ref;
  int *tseg, *nseg;
  CHECK(hipMalloc(&a, (size_t)bp * k * 4));
  CHECK(hipMalloc(&b, (size_t)ntot * k * 4));
  CHECK(hipMalloc(&an, bp * 4));
  CHECK(hipMalloc(&bn, ntot * 4));
  CHECK(hipMalloc(&pval, (size_t)jb_max * bp * 4));
  CHECK(hipMalloc(&pidx, (size_t)jb_max * bp * 4));
  CHECK(hipMalloc(&dist1, bp * 4));
  CHECK(hipMalloc(&dist2, bp * 4));
  CHECK(hipMalloc(&dist_ref, bp * 4));
  CHECK(hipMalloc(&idx1, bp * 8));
  CHECK(hipMalloc(&idx2, bp * 8));
  CHECK(hipMalloc(&idx_ref, bp * 8));
  CHECK(hipMalloc(&tseg, (C + 1) * 4));
  CHECK(hipMalloc(&nseg, (C + 1) * 4));
  std::vector<int> th(C + 1), nh(C + 1);
  for (int c = 0; c <= C; ++c) {
    th[c] = c * pc_pad;
    nh[c] = c * perclass_train;
  }
  CHECK(hipMemcpy(tseg, th.data(), (C + 1) * 4, hipMemcpyHostToDevice));
  CHECK(hipMemcpy(nseg, nh.data(), (C + 1) * 4, hipMemcpyHostToDevice));
  std::vector<float> host((size_t)(bp > ntot ? bp : ntot) * k);
  for (auto& v : host) v = frand();
  CHECK(hipMemcpy(a, host.data(), (size_t)bp * k * 4, hipMemcpyHostToDevice));
  CHECK(hipMemcpy(b, host.data(), (size_t)ntot * k * 4, hipMemcpyHostToDevice));
  launch_rownorm(a, bp, k, an, 0);
  launch_rownorm(b, ntot, k, bn, 0);
  CHECK(hipDeviceSynchronize());
  launch_grouped_rowmin(a, b, an, bn, tseg, nseg, C, bp, k, jb_max, pval,
                        pidx, dist1, idx1, 0);
  CHECK(hipDeviceSynchronize());
  CHECK(hipMemset(pval, 0xFF, (size_t)jb_max * bp * 4));
  CHECK(hipMemset(pidx, 0xFF, (size_t)jb_max * bp * 4));
  launch_grouped_rowmin(a, b, an, bn, tseg, nseg, C, bp, k, jb_max, pval,
                        pidx, dist2, idx2, 0);
  CHECK(hipDeviceSynchronize());
  // dense per-class reference: rowmin of each class block vs its segment
  for (int c = 0; c < C; ++c) {
    launch_pairwise_rowmin(
        a + (size_t)th[c] * k, b + (size_t)nh[c] * k, an + th[c], bn + nh[c],
        pc_pad, perclass_train, k, pval, pidx, dist_ref + th[c],
        idx_ref + th[c], 0);
    CHECK(hipDeviceSynchronize());
  }
  std::vector<float> d1(bp), d2(bp), dr(bp);
  std::vector<int64_t> i1(bp), i2(bp), ir(bp);
  CHECK(hipMemcpy(d1.data(), dist1, bp * 4, hipMemcpyDeviceToHost));
  CHECK(hipMemcpy(d2.data(), dist2, bp * 4, hipMemcpyDeviceToHost));
  CHECK(hipMemcpy(dr.data(), dist_ref, bp * 4, hipMemcpyDeviceToHost));
  CHECK(hipMemcpy(i1.data(), idx1, bp * 8, hipMemcpyDeviceToHost));
  CHECK(hipMemcpy(i2.data(), idx2, bp * 8, hipMemcpyDeviceToHost));
  CHECK(hipMemcpy(ir.data(), idx_ref, bp * 8, hipMemcpyDeviceToHost));
  int dmm = 0, imm = 0, refmm = 0;
  float maxd = 0;
  int first = -1;
  for (int i = 0; i < bp; ++i) {
    if (d1[i] != d2[i] || i1[i] != i2[i]) {
      ++dmm;
      if (first < 0) first = i;
      float dd = fabsf(d1[i] - d2[i]);
      if (dd > maxd) maxd = dd;
    }
    if (i1[i] != i2[i]) ++imm;
    const int cls = i / pc_pad;
    if (ir[i] + nh[cls] != i1[i] && dr[i] != d1[i]) ++refmm;
  }
  double dsum = 0;
  long long isum = 0;
  for (int i = 0; i < bp; ++i) {
    dsum += d1[i];
    isum += i1[i];
  }
  printf("verify C=%d bp=%d ntr=%d k=%d: run1-vs-run2 mismatches=%d "
         "(idx=%d, max|dd|=%g, first=%d) run1-vs-dense=%d checksum "
         "d=%.6e i=%lld\n",
         C, bp, perclass_train, k, dmm, imm, maxd, first, refmm, dsum, isum);
  if (first >= 0) {
    printf("  first row %d: d1=%.9g d2=%.9g i1=%lld i2=%lld\n", first,
           d1[first], d2[first], (long long)i1[first], (long long)i2[first]);
  }
}

// bf16 grouped rowmin at the same geometry (cross-checks vs fp32 run).
static void run_grouped_bf16(int iters, int perclass_test, int perclass_train,
                             int k) {
  const int C = 10;
  const int pc_pad = (perclass_test + 127) / 128 * 128;
  const int bp = C * pc_pad;
  const int ntot = C * perclass_train;
  const int jb_max = (perclass_train + BN - 1) / BN;
  short *a16, *b16;
  float *an, *bn, *pval, *dist;
  int* pidx;
  int64_t* idx;
  int *tseg, *nseg;
  CHECK(hipMalloc(&a16, (size_t)bp * k * 2));
  CHECK(hipMalloc(&b16, (size_t)ntot * k * 2));
  CHECK(hipMalloc(&an, bp * 4));
  CHECK(hipMalloc(&bn, ntot * 4));
  CHECK(hipMalloc(&pval, (size_t)jb_max * bp * 4));
  CHECK(hipMalloc(&pidx, (size_t)jb_max * bp * 4));
  CHECK(hipMalloc(&dist, bp * 4));
  CHECK(hipMalloc(&idx, bp * 8));
  CHECK(hipMalloc(&tseg, (C + 1) * 4));
  CHECK(hipMalloc(&nseg, (C + 1) * 4));
  std::vector<int> th(C + 1), nh(C + 1);
  for (int c = 0; c <= C; ++c) {
    th[c] = c * pc_pad;
    nh[c] = c * perclass_train;
  }
  CHECK(hipMemcpy(tseg, th.data(), (C + 1) * 4, hipMemcpyHostToDevice));
  CHECK(hipMemcpy(nseg, nh.data(), (C + 1) * 4, hipMemcpyHostToDevice));
  const size_t nmax = (size_t)(bp > ntot ? bp : ntot) * k;
  std::vector<short> h16(nmax);
  std::vector<float> norms(bp > ntot ? bp : ntot, 0.f);
  lcg_state = 12345u;
  for (size_t i = 0; i < nmax; ++i) {
    const float f = frand_det();
    unsigned u;
    memcpy(&u, &f, 4);
    h16[i] = (short)(u >> 16);
    float bf;
    u &= 0xffff0000u;
    memcpy(&bf, &u, 4);
    norms[i / k] += bf * bf;
  }
  CHECK(hipMemcpy(a16, h16.data(), (size_t)bp * k * 2, hipMemcpyHostToDevice));
  CHECK(hipMemcpy(b16, h16.data(), (size_t)ntot * k * 2, hipMemcpyHostToDevice));
  CHECK(hipMemcpy(an, norms.data(), bp * 4, hipMemcpyHostToDevice));
  CHECK(hipMemcpy(bn, norms.data(), ntot * 4, hipMemcpyHostToDevice));
  launch_grouped_rowmin_bf16(a16, b16, an, bn, tseg, nseg, C, bp, k, jb_max,
                             pval, pidx, dist, idx, 0);
  CHECK(hipDeviceSynchronize());
  hipEvent_t t0, t1;
  hipEventCreate(&t0);
  hipEventCreate(&t1);
  hipEventRecord(t0);
  for (int i = 0; i < iters; ++i)
    launch_grouped_rowmin_bf16(a16, b16, an, bn, tseg, nseg, C, bp, k,
                               jb_max, pval, pidx, dist, idx, 0);
  hipEventRecord(t1);
  CHECK(hipDeviceSynchronize());
  float ms;
  hipEventElapsedTime(&ms, t0, t1);
  const double tf = 2.0 * bp * perclass_train * k * iters / (ms / 1e3) / 1e12;
  std::vector<float> dh(bp);
  std::vector<int64_t> ih(bp);
  CHECK(hipMemcpy(dh.data(), dist, bp * 4, hipMemcpyDeviceToHost));
  CHECK(hipMemcpy(ih.data(), idx, bp * 8, hipMemcpyDeviceToHost));
  double dsum = 0;
  long long isum = 0;
  for (int i = 0; i < bp; ++i) {
    dsum += dh[i];
    isum += ih[i];
  }
  printf("grouped_rowmin_bf16 C=%d bp=%d ntr/cls=%d k=%d: %.3f ms/iter, "
         "%.1f TF  checksum d=%.6e i=%lld\n",
         C, bp, perclass_train, k, ms / iters, tf, dsum, isum);
}

static void run_downblock(int iters, int variant) {
  const int batch = 4096;
  const int HH[2] = {32, 16}, CC[2] = {16, 32};
  const int H = HH[variant], W = HH[variant], C = CC[variant];
  const int OH = H / 2, OW = W / 2, C2 = 2 * C;
  short *gin, *gout, *w1, *w2, *wsc;
  float *b1, *b2, *bsc;
  const int KS1 = (9 * C + 31) / 32, KS2 = (9 * C2 + 31) / 32,
            KSC = (C + 31) / 32;
  CHECK(hipMalloc(&gin, (size_t)batch * H * W * C * 2));
  CHECK(hipMalloc(&gout, (size_t)batch * OH * OW * C2 * 2));
  CHECK(hipMalloc(&w1, (size_t)(C2 / 16) * KS1 * 64 * 8 * 2));
  CHECK(hipMalloc(&w2, (size_t)(C2 / 16) * KS2 * 64 * 8 * 2));
  CHECK(hipMalloc(&wsc, (size_t)(C2 / 16) * KSC * 64 * 8 * 2));
  CHECK(hipMalloc(&b1, C2 * 4));
  CHECK(hipMalloc(&b2, C2 * 4));
  CHECK(hipMalloc(&bsc, C2 * 4));
  CHECK(hipMemset(b1, 0, C2 * 4));
  CHECK(hipMemset(b2, 0, C2 * 4));
  CHECK(hipMemset(bsc, 0, C2 * 4));
  launch_downblock(variant, batch, gin, gout, w1, b1, w2, b2, wsc, bsc, 0);
  CHECK(hipDeviceSynchronize());
  hipEvent_t t0, t1;
  hipEventCreate(&t0);
  hipEventCreate(&t1);
  hipEventRecord(t0);
  for (int i = 0; i < iters; ++i)
    launch_downblock(variant, batch, gin, gout, w1, b1, w2, b2, wsc, bsc, 0);
  hipEventRecord(t1);
  CHECK(hipDeviceSynchronize());
  float ms;
  hipEventElapsedTime(&ms, t0, t1);
  // conv1 (stride2: OHxOW out, K=9C) + conv2 (K=9*C2) + 1x1 shortcut
  const double flops =
      2.0 * batch * (double)(OH * OW) * C2 * (9.0 * C + 9.0 * C2 + C);
  printf("downblock<%d,%d,%d> b=%d: %.3f ms/iter, %.1f TF\n", H, W, C,
         batch, ms / iters, flops * iters / (ms / 1e3) / 1e12);
}

int main(int argc, char** argv) {
  const char* which = argc > 1 ? argv[1] : "pairwise";
  const int iters = argc > 2 ? atoi(argv[2]) : 10;
  srand(0);
  if (!strcmp(which, "pairwise"))
    run_pairwise(iters);
  else if (!strcmp(which, "grouped"))
    run_grouped(iters, argc > 3 ? atoi(argv[3]) : 1024,
                argc > 4 ? atoi(argv[4]) : 1500, argc > 5 ? atoi(argv[5]) : 4096);
  else if (!strcmp(which, "grouped16"))
    run_grouped_bf16(iters, argc > 3 ? atoi(argv[3]) : 1024,
                     argc > 4 ? atoi(argv[4]) : 1500,
                     argc > 5 ? atoi(argv[5]) : 4096);
  else if (!strcmp(which, "verify"))
    run_verify(argc > 2 ? atoi(argv[2]) : 1024, argc > 3 ? atoi(argv[3]) : 1500,
               argc > 4 ? atoi(argv[4]) : 4096);
  else if (!strcmp(which, "downblock"))
    run_downblock(iters, argc > 3 ? atoi(argv[3]) : 0);
  else
    run_resblock(iters, argc > 3 ? atoi(argv[3]) : 0);
  return 0;
}
