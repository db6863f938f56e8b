// CPU-native SGNS trainer: the fast host-side training path (BASELINE.json
// config 1, "text8 ... on CPU single process") and the exact-semantics twin
// of the fused HIP kernel.
//
// Semantics are normative per glint_word2vec_amd/ops/cpu_ref.py (the Python
// oracle): identical RNG (splitmix64-seeded xorshift64*, see rng.py for the
// draw-order contract), identical update math (center row cached per
// position, syn1 updated with the cached row, syn0 updated after the
// window).  Reference call-path being replaced: the Glint server-side
// dotprod/adjust ops driven by the mini-batch loop at
// mllib ServerSideGlintWord2Vec.scala:419-429.
//
// Threads: optional hogwild parallelism over sentences (the analog of the
// reference's numPartitions concurrent workers, mllib:120-127) using plain
// unsynchronised updates — races embraced exactly as the reference does.

#include <pybind11/pybind11.h>
#include <pybind11/numpy.h>
#include <atomic>
#include <cmath>
#include <cstdint>
#include <cstring>
#include <thread>
#include <vector>

namespace py = pybind11;

static constexpr float kMaxExp = 6.0f;

// --- counter-based RNG (must match glint_word2vec_amd/rng.py bit-for-bit) -
static constexpr uint64_t kGolden = 0x9E3779B97F4A7C15ULL;
static constexpr uint64_t kWinBase = 1ULL << 20;
static constexpr uint64_t kNegBase = 1ULL << 21;

static inline uint64_t splitmix64(uint64_t x) {
  uint64_t z = x + kGolden;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
  return z ^ (z >> 31);
}

static inline uint64_t sentence_base(uint64_t seed, uint64_t sentence_id) {
  return splitmix64(seed ^ (sentence_id * kGolden));
}

static inline uint32_t draw_u32(uint64_t base, uint64_t k) {
  return (uint32_t)(splitmix64(base + k * kGolden) >> 32);
}

static inline uint32_t keep_thr(float kp) {
  double t = (double)kp * 4294967296.0;
  return t >= 4294967295.0 ? 0xFFFFFFFFu : (uint32_t)t;
}

static inline float sigmoid_clipped(float f) {
  if (f > kMaxExp) return 1.0f;
  if (f < -kMaxExp) return 0.0f;
  return 1.0f / (1.0f + std::exp(-f));
}

struct Stats {
  int64_t pairs = 0;
  int64_t positives = 0;
  double sum_fplus = 0.0;
  int64_t words_trained = 0;
};

// Train the sentences [s_begin, s_end) of one batch.
static void train_sentences(float* syn0, float* syn1, int64_t dim,
                            const int32_t* tokens, const int32_t* offsets,
                            int64_t s_begin, int64_t s_end,
                            const float* keep_prob,  // nullptr = off
                            const int32_t* table, int64_t table_size,
                            float alpha, int window, int n_neg,
                            uint64_t seed, int64_t sent_id_base,
                            bool reference_window, Stats* stats,
                            std::vector<int32_t>& kept,
                            std::vector<float>& c_row,
                            std::vector<float>& grad) {
  for (int64_t s = s_begin; s < s_end; ++s) {
    const int32_t* sent = tokens + offsets[s];
    int64_t len = offsets[s + 1] - offsets[s];
    uint64_t base = sentence_base(seed, (uint64_t)(sent_id_base + s));
    kept.clear();
    if (keep_prob) {
      for (int64_t p = 0; p < len; ++p) {
        uint32_t u = draw_u32(base, (uint64_t)p);
        int32_t w = sent[p];
        if (u <= keep_thr(keep_prob[w])) kept.push_back(w);
      }
    } else {
      kept.assign(sent, sent + len);
    }
    int64_t L = (int64_t)kept.size();
    for (int64_t i = 0; i < L; ++i) {
      int32_t c = kept[i];
      uint32_t u = draw_u32(base, kWinBase + (uint64_t)i);
      int64_t lo, hi;
      if (!reference_window) {
        int64_t b = 1 + (int64_t)(u % (uint32_t)window);
        lo = i - b < 0 ? 0 : i - b;
        hi = i + b >= L ? L - 1 : i + b;
      } else {  // B2 semantics (mllib:385-387): left b, right b-1, may be empty
        int64_t b = (int64_t)(u % (uint32_t)window);
        if (b == 0) { lo = i; hi = i; }
        else {
          lo = i - b < 0 ? 0 : i - b;
          hi = i + b - 1 >= L ? L - 1 : i + b - 1;
        }
      }
      bool any = false;
      for (int64_t j = lo; j <= hi; ++j) if (j != i) { any = true; break; }
      if (!any) continue;
      float* c0 = syn0 + (int64_t)c * dim;
      std::memcpy(c_row.data(), c0, dim * sizeof(float));
      std::memset(grad.data(), 0, dim * sizeof(float));
      for (int64_t j = lo; j <= hi; ++j) {
        if (j == i) continue;
        int32_t t = kept[j];
        // positive pair
        {
          float* t1 = syn1 + (int64_t)t * dim;
          float f = 0.0f;
          for (int64_t d = 0; d < dim; ++d) f += c_row[d] * t1[d];
          float g = (1.0f - sigmoid_clipped(f)) * alpha;
          for (int64_t d = 0; d < dim; ++d) {
            grad[d] += g * t1[d];
            t1[d] += g * c_row[d];
          }
          stats->pairs++; stats->positives++; stats->sum_fplus += f;
        }
        uint64_t kbase = kNegBase +
            (uint64_t)(i * (2 * window + 1) + (j - i + window)) * (uint64_t)n_neg;
        for (int k = 0; k < n_neg; ++k) {
          uint32_t un = draw_u32(base, kbase + (uint64_t)k);
          int32_t neg = table[un % (uint64_t)table_size];
          if (neg == t) continue;
          float* t1 = syn1 + (int64_t)neg * dim;
          float f = 0.0f;
          for (int64_t d = 0; d < dim; ++d) f += c_row[d] * t1[d];
          float g = (0.0f - sigmoid_clipped(f)) * alpha;
          for (int64_t d = 0; d < dim; ++d) {
            grad[d] += g * t1[d];
            t1[d] += g * c_row[d];
          }
          stats->pairs++;
        }
      }
      for (int64_t d = 0; d < dim; ++d) c0[d] += grad[d];
      stats->words_trained++;
    }
  }
}

static py::dict train_batch(
    py::array_t<float, py::array::c_style> syn0,
    py::array_t<float, py::array::c_style> syn1,
    py::array_t<int32_t, py::array::c_style> tokens,
    py::array_t<int32_t, py::array::c_style> offsets,
    py::object keep_prob_obj,
    py::array_t<int32_t, py::array::c_style> table,
    float alpha, int window, int n_neg,
    uint64_t seed, int64_t sent_id_base,
    std::string window_mode, int num_threads) {
  if (syn0.ndim() != 2 || syn1.ndim() != 2)
    throw std::runtime_error("syn0/syn1 must be 2-D float32");
  int64_t dim = syn0.shape(1);
  if (syn1.shape(1) != dim) throw std::runtime_error("dim mismatch");
  int64_t num_sent = offsets.shape(0) - 1;
  const float* keep_prob = nullptr;
  py::array_t<float, py::array::c_style> kp_arr;
  if (!keep_prob_obj.is_none()) {
    kp_arr = keep_prob_obj.cast<py::array_t<float, py::array::c_style>>();
    keep_prob = kp_arr.data();
  }
  bool ref_window = (window_mode == "reference");
  if (window <= 0 || n_neg < 0) throw std::runtime_error("bad window/n");

  float* s0 = syn0.mutable_data();
  float* s1 = syn1.mutable_data();
  const int32_t* tok = tokens.data();
  const int32_t* off = offsets.data();
  const int32_t* tab = table.data();
  int64_t tab_size = table.shape(0);
  if (tab_size <= 0) throw std::runtime_error("empty unigram table");

  Stats total;
  {
    py::gil_scoped_release nogil;
    if (num_threads <= 1 || num_sent < 2 * num_threads) {
      std::vector<int32_t> kept; std::vector<float> cr(dim), gr(dim);
      train_sentences(s0, s1, dim, tok, off, 0, num_sent, keep_prob, tab,
                      tab_size, alpha, window, n_neg, seed, sent_id_base,
                      ref_window, &total, kept, cr, gr);
    } else {
      std::vector<std::thread> threads;
      std::vector<Stats> st(num_threads);
      int64_t per = (num_sent + num_threads - 1) / num_threads;
      for (int t = 0; t < num_threads; ++t) {
        int64_t b = t * per, e = std::min<int64_t>(num_sent, b + per);
        if (b >= e) break;
        threads.emplace_back([=, &st]() {
          std::vector<int32_t> kept; std::vector<float> cr(dim), gr(dim);
          train_sentences(s0, s1, dim, tok, off, b, e, keep_prob, tab,
                          tab_size, alpha, window, n_neg, seed, sent_id_base,
                          ref_window, &st[t], kept, cr, gr);
        });
      }
      for (auto& th : threads) th.join();
      for (auto& s : st) {
        total.pairs += s.pairs; total.positives += s.positives;
        total.sum_fplus += s.sum_fplus; total.words_trained += s.words_trained;
      }
    }
  }
  py::dict d;
  d["pairs"] = total.pairs;
  d["positives"] = total.positives;
  d["sum_fplus"] = total.sum_fplus;
  d["words_trained"] = total.words_trained;
  return d;
}

PYBIND11_MODULE(_cpu_native, m) {
  m.doc() = "CPU-native fused SGNS trainer (exact twin of the HIP kernel)";
  m.def("train_batch", &train_batch,
        py::arg("syn0"), py::arg("syn1"), py::arg("tokens"),
        py::arg("offsets"), py::arg("keep_prob"), py::arg("table"),
        py::arg("alpha"), py::arg("window"), py::arg("n_neg"),
        py::arg("seed"), py::arg("sent_id_base") = 0,
        py::arg("window_mode") = "canonical", py::arg("num_threads") = 1);
}
