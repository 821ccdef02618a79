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

static void run_resblock(int iters) {
  const int batch = 4096;
  constexpr int H = 32, W = 32, C = 16;
  const size_t plane = (size_t)H * W * C;
  short *gin, *gout, *w1, *w2;
  float *b1, *b2;
  constexpr int KSTEPS = (9 * C + 31) / 32;
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
  launch_resblock(0, batch, gin, gout, w1, b1, w2, b2, 0);
  CHECK(hipDeviceSynchronize());
  hipEvent_t t0, t1;
  hipEventCreate(&t0);
  hipEventCreate(&t1);
  hipEventRecord(t0);
  for (int i = 0; i < iters; ++i)
    launch_resblock(0, batch, gin, gout, w1, b1, w2, b2, 0);
  hipEventRecord(t1);
  CHECK(hipDeviceSynchronize());
  float ms;
  hipEventElapsedTime(&ms, t0, t1);
  const double flops = 2.0 * batch * 2 * 1024 * 16 * 144;
  printf("resblock<32,32,16> b=%d: %.3f ms/iter, %.1f TF\n", batch,
         ms / iters, flops * iters / (ms / 1e3) / 1e12);
}

int main(int argc, char** argv) {
  const char* which = argc > 1 ? argv[1] : "pairwise";
  const int iters = argc > 2 ? atoi(argv[2]) : 10;
  srand(0);
  if (!strcmp(which, "pairwise"))
    run_pairwise(iters);
  else
    run_resblock(iters);
  return 0;
}
