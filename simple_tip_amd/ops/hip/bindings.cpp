// Python bindings for the simple_tip_amd HIP/CDNA4 kernels (gfx950).
//
// Native HIP throughout — no CUDA-compat paths; streams come from torch's
// HIP stream pool so kernels serialize correctly with torch ops.

#include <torch/extension.h>

#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <vector>

// launchers defined in the .hip translation units
void launch_rownorm(const float*, int, int, float*, hipStream_t);
void launch_pairwise_full(const float*, const float*, const float*,
                          const float*, int, int, int, float*, hipStream_t);
void launch_pairwise_rowmin(const float*, const float*, const float*,
                            const float*, int, int, int, float*, int*, float*,
                            int64_t*, hipStream_t);
void launch_pairwise_kde(const float*, const float*, const float*,
                         const float*, int, int, int, float2*, float*,
                         hipStream_t);
void launch_grouped_rowmin(const float*, const float*, const float*,
                           const float*, const int*, const int*, int, int,
                           int, int, float*, int*, float*, int64_t*,
                           hipStream_t);
void launch_grouped_kde(const float*, const float*, const float*,
                        const float*, const int*, const int*, int, int, int,
                        int, float2*, float*, hipStream_t);
void launch_grouped_rowmin_bf16(const short*, const short*, const float*,
                                const float*, const int*, const int*, int,
                                int, int, int, float*, int*, float*,
                                int64_t*, hipStream_t);
void launch_grouped_kde_bf16(const short*, const short*, const float*,
                             const float*, const int*, const int*, int, int,
                             int, int, float2*, float*, hipStream_t);
void launch_profile(int, const float*, const unsigned char*, const float*,
                    const float*, float, int, int, int, int,
                    unsigned long long*, long long*, hipStream_t);
void launch_popcount(const unsigned long long*, int, int, long long*,
                     hipStream_t);
void launch_tknc(const float*, int, int, int, int, int, unsigned long long*,
                 hipStream_t);
void launch_bucketize(const double*, const double*, int, int, int,
                      unsigned long long*, hipStream_t);
void launch_cam_iteration(const unsigned long long*, int, int,
                          unsigned long long*, unsigned char*, long long*,
                          int*, long long*, hipStream_t);
void launch_cam_greedy_coop(const unsigned long long*, int, int,
                            unsigned long long, unsigned char*,
                            unsigned long long*, long long*, int*, long long*,
                            int*, int*, int*, hipStream_t);
void launch_softmax_scores(const float*, int, int, float*, float*, float*,
                           float*, hipStream_t);
void launch_mfma_probe(const short*, const short*, float*, hipStream_t);
void launch_resblock(int, int, const short*, short*, const short*,
                     const float*, const short*, const float*, hipStream_t);
void launch_downblock(int, int, const short*, short*, const short*,
                      const float*, const short*, const float*, const short*,
                      const float*, hipStream_t);
void launch_stem(int, const short*, short*, const short*, const float*,
                 hipStream_t);

namespace {

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

void check_f32_2d(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.dtype() == torch::kFloat32, name, " must be fp32");
  TORCH_CHECK(t.dim() == 2 && t.is_contiguous(), name,
              " must be 2D contiguous");
}

constexpr int kBN = 128;  // pairwise kernel column-block (matches pairwise.hip)

torch::Tensor rownorm(torch::Tensor x) {
  check_f32_2d(x, "x");
  auto out = torch::empty({x.size(0)}, x.options());
  launch_rownorm(x.data_ptr<float>(), x.size(0), x.size(1),
                 out.data_ptr<float>(), cur_stream());
  return out;
}

torch::Tensor pairwise_sqdist(torch::Tensor a, torch::Tensor b) {
  check_f32_2d(a, "a");
  check_f32_2d(b, "b");
  TORCH_CHECK(a.size(1) == b.size(1), "feature dims differ");
  const int m = a.size(0), n = b.size(0), k = a.size(1);
  auto an = rownorm(a);
  auto bn = rownorm(b);
  auto out = torch::empty({m, n}, a.options());
  launch_pairwise_full(a.data_ptr<float>(), b.data_ptr<float>(),
                       an.data_ptr<float>(), bn.data_ptr<float>(), m, n, k,
                       out.data_ptr<float>(), cur_stream());
  return out;
}

std::vector<torch::Tensor> rowmin_l2(torch::Tensor a, torch::Tensor b,
                                     c10::optional<torch::Tensor> bnorm) {
  check_f32_2d(a, "a");
  check_f32_2d(b, "b");
  TORCH_CHECK(a.size(1) == b.size(1), "feature dims differ");
  TORCH_CHECK(b.size(0) > 0, "rowmin_l2 against empty set");
  const int m = a.size(0), n = b.size(0), k = a.size(1);
  const int jb = (n + kBN - 1) / kBN;
  auto an = rownorm(a);
  auto bn = bnorm.has_value() ? bnorm.value() : rownorm(b);
  auto pval = torch::empty({jb, m}, a.options());
  auto pidx = torch::empty({jb, m}, a.options().dtype(torch::kInt32));
  auto dist = torch::empty({m}, a.options());
  auto idx = torch::empty({m}, a.options().dtype(torch::kInt64));
  launch_pairwise_rowmin(a.data_ptr<float>(), b.data_ptr<float>(),
                         an.data_ptr<float>(), bn.data_ptr<float>(), m, n, k,
                         pval.data_ptr<float>(), pidx.data_ptr<int>(),
                         dist.data_ptr<float>(), idx.data_ptr<int64_t>(),
                         cur_stream());
  return {dist, idx};
}

std::vector<torch::Tensor> grouped_rowmin(
    torch::Tensor testS, torch::Tensor trainS, torch::Tensor tseg,
    torch::Tensor nseg, torch::Tensor bnorm, int64_t jb_max) {
  check_f32_2d(testS, "testS");
  check_f32_2d(trainS, "trainS");
  TORCH_CHECK(tseg.is_cuda() && tseg.dtype() == torch::kInt32);
  TORCH_CHECK(nseg.is_cuda() && nseg.dtype() == torch::kInt32);
  const int bp = testS.size(0), k = testS.size(1);
  const int nclasses = tseg.size(0) - 1;
  auto an = rownorm(testS);
  auto pval = torch::empty({jb_max, bp}, testS.options());
  auto pidx = torch::empty({jb_max, bp}, testS.options().dtype(torch::kInt32));
  auto dist = torch::empty({bp}, testS.options());
  auto idx = torch::empty({bp}, testS.options().dtype(torch::kInt64));
  launch_grouped_rowmin(
      testS.data_ptr<float>(), trainS.data_ptr<float>(), an.data_ptr<float>(),
      bnorm.data_ptr<float>(), tseg.data_ptr<int>(), nseg.data_ptr<int>(),
      nclasses, bp, k, jb_max, pval.data_ptr<float>(), pidx.data_ptr<int>(),
      dist.data_ptr<float>(), idx.data_ptr<int64_t>(), cur_stream());
  return {dist, idx};
}

torch::Tensor grouped_kde(torch::Tensor testWS, torch::Tensor trainWS,
                          torch::Tensor tseg, torch::Tensor nseg,
                          torch::Tensor bnorm, int64_t jb_max) {
  check_f32_2d(testWS, "testWS");
  check_f32_2d(trainWS, "trainWS");
  const int bp = testWS.size(0), k = testWS.size(1);
  const int nclasses = tseg.size(0) - 1;
  auto an = rownorm(testWS);
  auto pkde = torch::empty({jb_max, bp, 2}, testWS.options());
  auto out = torch::empty({bp}, testWS.options());
  launch_grouped_kde(
      testWS.data_ptr<float>(), trainWS.data_ptr<float>(),
      an.data_ptr<float>(), bnorm.data_ptr<float>(), tseg.data_ptr<int>(),
      nseg.data_ptr<int>(), nclasses, bp, k, jb_max,
      reinterpret_cast<float2*>(pkde.data_ptr<float>()),
      out.data_ptr<float>(), cur_stream());
  return out;
}

std::vector<torch::Tensor> grouped_rowmin_bf16(
    torch::Tensor testS, torch::Tensor trainS, torch::Tensor tseg,
    torch::Tensor nseg, torch::Tensor anorm, torch::Tensor bnorm,
    int64_t jb_max) {
  TORCH_CHECK(testS.is_cuda() && testS.dtype() == torch::kBFloat16 &&
              testS.is_contiguous());
  TORCH_CHECK(trainS.dtype() == torch::kBFloat16 && trainS.is_contiguous());
  const int bp = testS.size(0), k = testS.size(1);
  const int nclasses = tseg.size(0) - 1;
  auto fopts = anorm.options();
  auto pval = torch::empty({jb_max, bp}, fopts);
  auto pidx = torch::empty({jb_max, bp}, fopts.dtype(torch::kInt32));
  auto dist = torch::empty({bp}, fopts);
  auto idx = torch::empty({bp}, fopts.dtype(torch::kInt64));
  launch_grouped_rowmin_bf16(
      reinterpret_cast<const short*>(testS.data_ptr()),
      reinterpret_cast<const short*>(trainS.data_ptr()),
      anorm.data_ptr<float>(), bnorm.data_ptr<float>(), tseg.data_ptr<int>(),
      nseg.data_ptr<int>(), nclasses, bp, k, jb_max, pval.data_ptr<float>(),
      pidx.data_ptr<int>(), dist.data_ptr<float>(), idx.data_ptr<int64_t>(),
      cur_stream());
  return {dist, idx};
}

torch::Tensor grouped_kde_bf16(torch::Tensor testWS, torch::Tensor trainWS,
                               torch::Tensor tseg, torch::Tensor nseg,
                               torch::Tensor anorm, torch::Tensor bnorm,
                               int64_t jb_max) {
  TORCH_CHECK(testWS.is_cuda() && testWS.dtype() == torch::kBFloat16 &&
              testWS.is_contiguous());
  TORCH_CHECK(trainWS.dtype() == torch::kBFloat16 && trainWS.is_contiguous());
  const int bp = testWS.size(0), k = testWS.size(1);
  const int nclasses = tseg.size(0) - 1;
  auto fopts = anorm.options();
  auto pkde = torch::empty({jb_max, bp, 2}, fopts);
  auto out = torch::empty({bp}, fopts);
  launch_grouped_kde_bf16(
      reinterpret_cast<const short*>(testWS.data_ptr()),
      reinterpret_cast<const short*>(trainWS.data_ptr()),
      anorm.data_ptr<float>(), bnorm.data_ptr<float>(), tseg.data_ptr<int>(),
      nseg.data_ptr<int>(), nclasses, bp, k, jb_max,
      reinterpret_cast<float2*>(pkde.data_ptr<float>()),
      out.data_ptr<float>(), cur_stream());
  return out;
}

torch::Tensor kde_logsumexp(torch::Tensor test, torch::Tensor train) {
  check_f32_2d(test, "test");
  check_f32_2d(train, "train");
  TORCH_CHECK(test.size(1) == train.size(1), "feature dims differ");
  TORCH_CHECK(train.size(0) > 0, "kde over empty train set");
  const int m = test.size(0), n = train.size(0), k = test.size(1);
  const int jb = (n + kBN - 1) / kBN;
  auto an = rownorm(test);
  auto bn = rownorm(train);
  auto pkde = torch::empty({jb, m, 2}, test.options());
  auto out = torch::empty({m}, test.options());
  launch_pairwise_kde(test.data_ptr<float>(), train.data_ptr<float>(),
                      an.data_ptr<float>(), bn.data_ptr<float>(), m, n, k,
                      reinterpret_cast<float2*>(pkde.data_ptr<float>()),
                      out.data_ptr<float>(), cur_stream());
  return out;
}

std::vector<torch::Tensor> profile(int mode, torch::Tensor acts,
                                   torch::Tensor lo, torch::Tensor hi,
                                   double thr, int64_t sections,
                                   int64_t nbits) {
  check_f32_2d(acts, "acts");
  const int rows = acts.size(0), K = acts.size(1);
  const int W = (nbits + 63) / 64;
  auto words = torch::empty({rows, W}, acts.options().dtype(torch::kInt64));
  auto scores = torch::empty({rows}, acts.options().dtype(torch::kInt64));
  const float* lop = lo.defined() && lo.numel() ? lo.data_ptr<float>() : nullptr;
  const float* hip = hi.defined() && hi.numel() ? hi.data_ptr<float>() : nullptr;
  launch_profile(mode, acts.data_ptr<float>(), nullptr, lop, hip,
                 static_cast<float>(thr), sections, rows, K, W,
                 reinterpret_cast<unsigned long long*>(words.data_ptr<int64_t>()),
                 reinterpret_cast<long long*>(scores.data_ptr<int64_t>()), cur_stream());
  return {words, scores};
}

torch::Tensor pack_bits(torch::Tensor boolmat) {
  TORCH_CHECK(boolmat.is_cuda() && boolmat.dtype() == torch::kBool &&
              boolmat.dim() == 2 && boolmat.is_contiguous());
  const int rows = boolmat.size(0), K = boolmat.size(1);
  const int W = (K + 63) / 64;
  auto words = torch::empty({rows, W},
                            boolmat.options().dtype(torch::kInt64));
  launch_profile(4 /*PROF_PACK*/, nullptr,
                 boolmat.data_ptr<bool>()
                     ? reinterpret_cast<const unsigned char*>(
                           boolmat.data_ptr<bool>())
                     : nullptr,
                 nullptr, nullptr, 0.f, 1, rows, K, W,
                 reinterpret_cast<unsigned long long*>(words.data_ptr<int64_t>()),
                 nullptr, cur_stream());
  return words;
}

torch::Tensor popcount_rows(torch::Tensor words) {
  TORCH_CHECK(words.is_cuda() && words.dtype() == torch::kInt64 &&
              words.dim() == 2 && words.is_contiguous());
  auto out = torch::empty({words.size(0)}, words.options());
  launch_popcount(
      reinterpret_cast<const unsigned long long*>(words.data_ptr<int64_t>()),
      words.size(0), words.size(1),
      reinterpret_cast<long long*>(out.data_ptr<int64_t>()), cur_stream());
  return out;
}

void tknc_layer(torch::Tensor layer, int64_t k, int64_t bit_offset,
                torch::Tensor words) {
  check_f32_2d(layer, "layer");
  TORCH_CHECK(k >= 1 && k <= 4, "tknc supports k in [1,4]");
  launch_tknc(layer.data_ptr<float>(), layer.size(0), layer.size(1), k,
              bit_offset, words.size(1),
              reinterpret_cast<unsigned long long*>(words.data_ptr<int64_t>()),
              cur_stream());
}

torch::Tensor bucketize(torch::Tensor values, torch::Tensor thresholds,
                        int64_t sections) {
  TORCH_CHECK(values.is_cuda() && values.dtype() == torch::kFloat64);
  TORCH_CHECK(thresholds.is_cuda() && thresholds.dtype() == torch::kFloat64);
  const int n = values.size(0);
  const int W = (sections + 63) / 64;
  auto words = torch::empty({n, W}, values.options().dtype(torch::kInt64));
  launch_bucketize(values.data_ptr<double>(), thresholds.data_ptr<double>(), n,
                   sections, W,
                   reinterpret_cast<unsigned long long*>(words.data_ptr<int64_t>()),
                   cur_stream());
  return words;
}

// Greedy CAM loop: device does all O(N*W) work, host only reads the picked
// row per iteration. Returns the picked prefix (rows that added coverage).
torch::Tensor cam_greedy(torch::Tensor words, int64_t nbits) {
  TORCH_CHECK(words.is_cuda() && words.dtype() == torch::kInt64 &&
              words.dim() == 2 && words.is_contiguous());
  const int rows = words.size(0), W = words.size(1);
  auto opts = words.options();
  const int tail = nbits % 64;
  const unsigned long long tail_mask =
      tail ? ((1ull << tail) - 1) : ~0ull;
  auto used = torch::zeros({rows}, opts.dtype(torch::kUInt8));
  auto stream = cur_stream();

  // Fast path: the WHOLE greedy loop in one persistent cooperative kernel
  // (64 resident blocks + software grid barriers). The per-iteration
  // two-kernel form paid ~200 us of dependent-launch latency per pick and
  // a single-block loop is sweep-bound on one CU (~340 us/pick); this form
  // has zero dispatch and a grid-parallel sweep (profiles/r02).
  {
    auto order_dev = torch::empty({rows}, opts);
    auto n_dev = torch::zeros({1}, opts.dtype(torch::kInt32));
    auto uncovered_dev = torch::empty({W}, opts);
    auto part_val = torch::empty({64}, opts);
    auto part_idx = torch::empty({64}, opts.dtype(torch::kInt32));
    auto aux = torch::zeros({2}, opts.dtype(torch::kInt32));  // barrier, win
    launch_cam_greedy_coop(
        reinterpret_cast<const unsigned long long*>(
            words.data_ptr<int64_t>()),
        rows, W, tail_mask, used.data_ptr<uint8_t>(),
        reinterpret_cast<unsigned long long*>(
            uncovered_dev.data_ptr<int64_t>()),
        reinterpret_cast<long long*>(part_val.data_ptr<int64_t>()),
        part_idx.data_ptr<int>(),
        reinterpret_cast<long long*>(order_dev.data_ptr<int64_t>()),
        n_dev.data_ptr<int>(), aux.data_ptr<int>(),
        aux.data_ptr<int>() + 1, stream);
    const int n = n_dev.cpu().item<int>();
    return order_dev.narrow(0, 0, n).cpu();
  }

  // Fallback (coverage mask too wide for LDS): per-iteration kernels with
  // CHAIN iterations enqueued per host sync (iterations are idempotent
  // once coverage is exhausted, so over-running is safe).
  auto uncovered = torch::full({W}, -1, opts);  // all ones
  if (tail)
    uncovered.index_put_({W - 1}, (int64_t)tail_mask);
  const int wpb = 8;
  const int nblocks = (rows + wpb - 1) / wpb;
  auto part_val = torch::empty({nblocks}, opts);
  auto part_idx = torch::empty({nblocks}, opts.dtype(torch::kInt32));
  constexpr int CHAIN = 256;
  auto result = torch::empty({2 * CHAIN}, opts);
  std::vector<int64_t> order;
  order.reserve(std::min<int64_t>(rows, nbits));
  std::vector<long long> host_result(2 * CHAIN);
  bool done = false;
  while (!done) {
    for (int j = 0; j < CHAIN; ++j) {
      launch_cam_iteration(
          reinterpret_cast<const unsigned long long*>(
              words.data_ptr<int64_t>()),
          rows, W,
          reinterpret_cast<unsigned long long*>(
              uncovered.data_ptr<int64_t>()),
          used.data_ptr<uint8_t>(),
          reinterpret_cast<long long*>(part_val.data_ptr<int64_t>()),
          part_idx.data_ptr<int>(),
          reinterpret_cast<long long*>(result.data_ptr<int64_t>()) + 2 * j,
          stream);
    }
    C10_HIP_CHECK(hipMemcpyAsync(host_result.data(),
                                 result.data_ptr<int64_t>(),
                                 2 * CHAIN * sizeof(long long),
                                 hipMemcpyDeviceToHost, stream));
    C10_HIP_CHECK(hipStreamSynchronize(stream));
    for (int j = 0; j < CHAIN; ++j) {
      if (host_result[2 * j] < 0) {
        done = true;
        break;
      }
      order.push_back(host_result[2 * j]);
      if ((int64_t)order.size() >= rows) {
        done = true;
        break;
      }
    }
  }
  return torch::tensor(order, torch::dtype(torch::kInt64));
}

// All-pairs Levenshtein distance matrix (CPU, threaded) for the text
// corruptor's autocorrect dictionary (the reference uses the
// polyleven pip package on a thread pool, text_corruptor.py:282-309).
torch::Tensor levenshtein_matrix(std::vector<std::string> words) {
  const int64_t n = (int64_t)words.size();
  auto out = torch::zeros({n, n}, torch::dtype(torch::kUInt8));
  auto acc = out.accessor<uint8_t, 2>();
  at::parallel_for(0, n, 1, [&](int64_t begin, int64_t end) {
    std::vector<int> dp0(64), dp1(64);
    for (int64_t i = begin; i < end; ++i) {
      const std::string& a = words[i];
      const int la = (int)a.size();
      for (int64_t j = i + 1; j < n; ++j) {
        const std::string& b = words[j];
        const int lb = (int)b.size();
        if (lb + 1 > (int)dp0.size()) {
          dp0.resize(lb + 1);
          dp1.resize(lb + 1);
        }
        for (int c = 0; c <= lb; ++c) dp0[c] = c;
        for (int r = 1; r <= la; ++r) {
          dp1[0] = r;
          const char ca = a[r - 1];
          for (int c = 1; c <= lb; ++c) {
            const int sub = dp0[c - 1] + (ca != b[c - 1] ? 1 : 0);
            const int del = dp0[c] + 1;
            const int ins = dp1[c - 1] + 1;
            dp1[c] = std::min(sub, std::min(del, ins));
          }
          std::swap(dp0, dp1);
        }
        const int d = std::min(dp0[lb], 255);
        acc[i][j] = (uint8_t)d;
        acc[j][i] = (uint8_t)d;
      }
    }
  });
  return out;
}

std::vector<torch::Tensor> softmax_scores(torch::Tensor probs) {
  check_f32_2d(probs, "probs");
  const int n = probs.size(0), c = probs.size(1);
  auto neg_max = torch::empty({n}, probs.options());
  auto neg_pcs = torch::empty({n}, probs.options());
  auto entropy = torch::empty({n}, probs.options());
  auto gini = torch::empty({n}, probs.options());
  launch_softmax_scores(probs.data_ptr<float>(), n, c,
                        neg_max.data_ptr<float>(), neg_pcs.data_ptr<float>(),
                        entropy.data_ptr<float>(), gini.data_ptr<float>(),
                        cur_stream());
  return {neg_max, neg_pcs, entropy, gini};
}

// ---- fused ResNet-20 inference (bf16 NHWC) ----

const short* bf16_ptr(const torch::Tensor& t) {
  TORCH_CHECK(t.is_cuda() && t.dtype() == torch::kBFloat16 && t.is_contiguous());
  return reinterpret_cast<const short*>(t.data_ptr());
}

torch::Tensor mfma_probe(torch::Tensor a, torch::Tensor b) {
  auto d = torch::empty({16, 16}, a.options().dtype(torch::kFloat32));
  launch_mfma_probe(bf16_ptr(a), bf16_ptr(b),
                    d.data_ptr<float>(), cur_stream());
  return d;
}

torch::Tensor resnet_block(int64_t variant, torch::Tensor x, torch::Tensor w1,
                           torch::Tensor b1, torch::Tensor w2, torch::Tensor b2) {
  const int batch = x.size(0);
  auto out = torch::empty_like(x);
  launch_resblock(variant, batch, bf16_ptr(x),
                  const_cast<short*>(bf16_ptr(out)), bf16_ptr(w1),
                  b1.data_ptr<float>(), bf16_ptr(w2), b2.data_ptr<float>(),
                  cur_stream());
  return out;
}

torch::Tensor resnet_down(int64_t variant, torch::Tensor x, torch::Tensor w1,
                          torch::Tensor b1, torch::Tensor w2, torch::Tensor b2,
                          torch::Tensor wsc, torch::Tensor bsc) {
  const int batch = x.size(0);
  const int64_t out_elems = x.size(1) / 2;  // H*W*C -> (H/2)(W/2)(2C)
  auto out = torch::empty({batch, out_elems}, x.options());
  launch_downblock(variant, batch, bf16_ptr(x),
                   const_cast<short*>(bf16_ptr(out)), bf16_ptr(w1),
                   b1.data_ptr<float>(), bf16_ptr(w2), b2.data_ptr<float>(),
                   bf16_ptr(wsc), bsc.data_ptr<float>(), cur_stream());
  return out;
}

torch::Tensor resnet_stem(torch::Tensor x, torch::Tensor w, torch::Tensor b) {
  const int batch = x.size(0);
  const int64_t out_elems = 32 * 32 * 16;
  auto out = torch::empty({batch, out_elems}, x.options());
  launch_stem(batch, bf16_ptr(x), const_cast<short*>(bf16_ptr(out)),
              bf16_ptr(w), b.data_ptr<float>(), cur_stream());
  return out;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "simple_tip_amd HIP/CDNA4 kernels (gfx950)";
  m.def("rownorm", &rownorm);
  m.def("pairwise_sqdist", &pairwise_sqdist);
  m.def("rowmin_l2", &rowmin_l2, py::arg("a"), py::arg("b"),
        py::arg("bnorm") = py::none());
  m.def("kde_logsumexp", &kde_logsumexp);
  m.def("grouped_rowmin", &grouped_rowmin);
  m.def("grouped_kde", &grouped_kde);
  m.def("grouped_rowmin_bf16", &grouped_rowmin_bf16);
  m.def("grouped_kde_bf16", &grouped_kde_bf16);
  m.def("profile", &profile);
  m.def("pack_bits", &pack_bits);
  m.def("popcount_rows", &popcount_rows);
  m.def("tknc_layer", &tknc_layer);
  m.def("bucketize", &bucketize);
  m.def("cam_greedy", &cam_greedy);
  m.def("softmax_scores", &softmax_scores);
  m.def("levenshtein_matrix", &levenshtein_matrix);
  m.def("mfma_probe", &mfma_probe);
  m.def("resnet_block", &resnet_block);
  m.def("resnet_down", &resnet_down);
  m.def("resnet_stem", &resnet_stem);
}
