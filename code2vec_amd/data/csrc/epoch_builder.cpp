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

// ---------------------------------------------------------------------------
// Native corpus parse (reference model/dataset_reader.py:72-128 semantics).
// Single C++ pass over the file (GIL released); label normalization and
// vocab building stay in Python (regex parity).  Start/end terminal
// indexes are returned +1-shifted (the @question offset); paths unshifted.

#include <fstream>
#include <string>

namespace {
struct Span { size_t off; size_t len; };

inline const char* strip(const char* b, const char* e, size_t* len) {
  while (b < e && (*b == ' ' || *b == '\r' || *b == '\t')) ++b;
  while (e > b && (e[-1] == ' ' || e[-1] == '\r' || e[-1] == '\t'))
    --e;
  *len = (size_t)(e - b);
  return b;
}

inline long parse_int(const char* p, const char* e, const char** out) {
  long v = 0;
  bool neg = false;
  if (p < e && *p == '-') { neg = true; ++p; }
  while (p < e && *p >= '0' && *p <= '9') v = v * 10 + (*p++ - '0');
  *out = p;
  return neg ? -v : v;
}

// strict: must consume at least one digit; advances *pp past the number
inline bool parse_int_strict(const char** pp, const char* e, long* out) {
  const char* p = *pp;
  bool neg = false;
  if (p < e && *p == '-') { neg = true; ++p; }
  const char* d0 = p;
  long v = 0;
  while (p < e && *p >= '0' && *p <= '9') v = v * 10 + (*p++ - '0');
  if (p == d0) return false;
  *pp = p;
  *out = neg ? -v : v;
  return true;
}
}  // namespace

pybind11::dict parse_corpus(const std::string& path) {
  std::string buf;
  std::vector<int64_t> ids;
  std::vector<Span> labels, sources;
  std::vector<int32_t> triples;
  std::vector<int64_t> rec_off;          // per-record triple offsets
  std::vector<int64_t> var_rec_off;      // per-record var offsets
  std::vector<Span> var_orig, var_alias;
  {
    pybind11::gil_scoped_release release;
    std::ifstream f(path, std::ios::binary);
    if (!f) throw std::runtime_error("cannot open corpus: " + path);
    f.seekg(0, std::ios::end);
    buf.resize((size_t)f.tellg());
    f.seekg(0);
    f.read(&buf[0], (std::streamsize)buf.size());

    bool in_record = false;
    int mode = 0;
    size_t lineno = 0;
    auto flush = [&]() {
      if (in_record) {
        rec_off.push_back((int64_t)(triples.size() / 3));
        var_rec_off.push_back((int64_t)var_orig.size());
        in_record = false;
      }
    };
    const char* base = buf.data();
    const char* end = base + buf.size();
    const char* line = base;
    while (line <= end) {
      ++lineno;
      const char* nl = (const char*)memchr(line, '\n', (size_t)(end - line));
      const char* le = nl ? nl : end;
      size_t len;
      const char* lb = strip(line, le, &len);
      if (len == 0) {
        flush();
      } else {
        if (!in_record) {
          in_record = true;
          mode = 0;
          ids.push_back(-1);
          labels.push_back({0, 0});
          sources.push_back({0, 0});
        }
        if (lb[0] == '#') {
          const char* q;
          ids.back() = parse_int(lb + 1, lb + len, &q);
        } else if (len >= 6 && memcmp(lb, "label:", 6) == 0) {
          labels.back() = {(size_t)(lb + 6 - base), len - 6};
        } else if (len >= 6 && memcmp(lb, "class:", 6) == 0) {
          sources.back() = {(size_t)(lb + 6 - base), len - 6};
        } else if (len >= 6 && memcmp(lb, "paths:", 6) == 0) {
          mode = 1;
        } else if (len >= 5 && memcmp(lb, "vars:", 5) == 0) {
          mode = 2;
        } else if (len >= 4 && memcmp(lb, "doc:", 4) == 0) {
          // parsed and discarded
        } else if (mode == 1) {
          // STRICT start\tpath\tend triple — a malformed line must fail
          // loudly (the reference's int()/unpack would crash too); silent
          // tolerance here fabricated PAD-ish triples from corrupt input
          const char* q = lb;
          const char* le2 = lb + len;
          long s, pth, e2;
          bool ok = parse_int_strict(&q, le2, &s);
          ok = ok && q < le2 && *q == '\t';
          if (ok) ++q;
          ok = ok && parse_int_strict(&q, le2, &pth);
          ok = ok && q < le2 && *q == '\t';
          if (ok) ++q;
          ok = ok && parse_int_strict(&q, le2, &e2);
          ok = ok && q == le2;  // no trailing fields (split() parity)
          if (!ok)
            throw std::runtime_error(
                "malformed path-context line " + std::to_string(lineno) +
                " in " + path);
          triples.push_back((int32_t)(s + 1));  // +@question shift
          triples.push_back((int32_t)pth);
          triples.push_back((int32_t)(e2 + 1));
        } else if (mode == 2) {
          const char* tab =
              (const char*)memchr(lb, '\t', len);
          if (!tab)
            throw std::runtime_error(
                "malformed vars line " + std::to_string(lineno) + " in " +
                path);
          var_orig.push_back({(size_t)(lb - base), (size_t)(tab - lb)});
          var_alias.push_back({(size_t)(tab + 1 - base),
                               len - (size_t)(tab + 1 - lb)});
        }
      }
      if (!nl) break;
      line = nl + 1;
    }
    flush();
  }

  // with the GIL: assemble python objects
  namespace py = pybind11;
  const int64_t n = (int64_t)ids.size();
  auto ids_t = torch::from_blob(ids.data(), {n}, torch::kInt64).clone();
  std::vector<int64_t> off(n + 1, 0);
  for (int64_t i = 0; i < n; ++i) off[i + 1] = rec_off[(size_t)i];
  auto off_t = torch::from_blob(off.data(), {n + 1}, torch::kInt64).clone();
  std::vector<int64_t> voff(n + 1, 0);
  for (int64_t i = 0; i < n; ++i) voff[i + 1] = var_rec_off[(size_t)i];
  auto voff_t = torch::from_blob(voff.data(), {n + 1}, torch::kInt64).clone();
  auto tri_t = torch::from_blob(triples.data(),
                                {(int64_t)(triples.size() / 3), 3},
                                torch::kInt32).clone();
  py::list label_l, source_l, vorig_l, valias_l;
  for (auto& sp : labels)
    label_l.append(py::str(buf.data() + sp.off, sp.len));
  for (auto& sp : sources)
    source_l.append(py::str(buf.data() + sp.off, sp.len));
  for (auto& sp : var_orig)
    vorig_l.append(py::str(buf.data() + sp.off, sp.len));
  for (auto& sp : var_alias)
    valias_l.append(py::str(buf.data() + sp.off, sp.len));

  py::dict out;
  out["ids"] = ids_t;
  out["labels"] = label_l;
  out["sources"] = source_l;
  out["offsets"] = off_t;
  out["contexts"] = tri_t;
  out["var_offsets"] = voff_t;
  out["var_originals"] = vorig_l;
  out["var_aliases"] = valias_l;
  return out;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("build_method_epoch", &build_method_epoch,
        "per-epoch resample+pad (OpenMP)");
  m.def("parse_corpus", &parse_corpus, "native corpus parse");
}
