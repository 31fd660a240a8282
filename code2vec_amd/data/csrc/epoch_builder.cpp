// Native (C++/OpenMP) per-epoch dataset construction — the host-side hot
// loop of training (reference model/dataset_builder.py:112-150 semantics:
// per item, resample up to C of its path-contexts without replacement,
// replace the @method_0 terminal with @question, zero-pad).
//
// Items are stored flat: contexts [total, 3] int32 with per-item offsets.
// Per-item counter RNG (splitmix64 on (seed, item)) keeps epochs
// deterministic, rank-aware and order-independent under OpenMP.

#include <torch/extension.h>

#include <atomic>
#include <cstdint>
#include <cstring>
#include <vector>

#ifdef _OPENMP
#include <omp.h>
#endif

namespace {

struct Rng {
  uint64_t s;
  explicit Rng(uint64_t seed) : s(seed) {}
  uint64_t next() {
    s += 0x9E3779B97F4A7C15ull;
    uint64_t z = s;
    z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
    z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
    return z ^ (z >> 31);
  }
  // uniform integer in [0, n)
  uint32_t below(uint32_t n) { return (uint32_t)(next() % n); }
};

}  // namespace

// offsets: [N+1] int64 (contexts of item i at [offsets[i], offsets[i+1]))
// contexts: [total, 3] int32 (start, path, end; question offset applied)
// out_*: [N, C] int32, pre-allocated (overwritten fully)
void build_method_epoch(torch::Tensor offsets, torch::Tensor contexts,
                        torch::Tensor item_idx, torch::Tensor out_starts,
                        torch::Tensor out_paths, torch::Tensor out_ends,
                        int64_t method_token_index, int64_t question_index,
                        int64_t seed) {
  TORCH_CHECK(offsets.dtype() == torch::kInt64 && offsets.is_contiguous());
  TORCH_CHECK(contexts.dtype() == torch::kInt32 && contexts.is_contiguous());
  TORCH_CHECK(item_idx.dtype() == torch::kInt64 && item_idx.is_contiguous());
  TORCH_CHECK(out_starts.dtype() == torch::kInt32 && out_starts.is_contiguous());
  const int64_t N = item_idx.numel();  // output rows (this rank's shard)
  const int64_t C = out_starts.size(1);
  TORCH_CHECK(out_starts.size(0) == N);
  const int64_t* items = item_idx.data_ptr<int64_t>();
  const int64_t* off = offsets.data_ptr<int64_t>();
  const int32_t* ctx = contexts.data_ptr<int32_t>();
  int32_t* os = out_starts.data_ptr<int32_t>();
  int32_t* op = out_paths.data_ptr<int32_t>();
  int32_t* oe = out_ends.data_ptr<int32_t>();
  const int32_t mtok = (int32_t)method_token_index;
  const int32_t qtok = (int32_t)question_index;

  // release the GIL: the trainer builds epoch N+1 on a worker thread while
  // epoch N trains on the GPU
  pybind11::gil_scoped_release release;

#pragma omp parallel
  {
    std::vector<int32_t> idx;
#pragma omp for schedule(dynamic, 64)
    for (int64_t t_row = 0; t_row < N; ++t_row) {
      const int64_t i = items[t_row];
      const int64_t lo = off[i];
      const int64_t n = off[i + 1] - lo;
      const int64_t k = n < C ? n : C;
      Rng rng((uint64_t)seed * 0x100000001B3ull + (uint64_t)i);
      idx.resize(n);
      for (int64_t t = 0; t < n; ++t) idx[t] = (int32_t)t;
      // partial Fisher-Yates: first k positions are a uniform sample
      // without replacement, in random order
      for (int64_t t = 0; t < k; ++t) {
        const int64_t j = t + rng.below((uint32_t)(n - t));
        std::swap(idx[t], idx[j]);
      }
      int32_t* rs = os + t_row * C;
      int32_t* rp = op + t_row * C;
      int32_t* re = oe + t_row * C;
      for (int64_t t = 0; t < k; ++t) {
        const int32_t* triple = ctx + (lo + idx[t]) * 3;
        int32_t s = triple[0];
        int32_t e = triple[2];
        rs[t] = s == mtok ? qtok : s;
        rp[t] = triple[1];
        re[t] = e == mtok ? qtok : e;
      }
      for (int64_t t = k; t < C; ++t) {
        rs[t] = 0;
        rp[t] = 0;
        re[t] = 0;
      }
    }
  }
}

// Corpus line-oriented fast path parse of the "paths:" triples section is in
// Python (reader.py); numeric-heavy per-epoch work lives here.

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("build_method_epoch", &build_method_epoch,
        "per-epoch resample+pad (OpenMP)");
}
