// Native dataset index-map builders (CPU, pybind11).
//
// Re-implements the semantics of the reference's C++ helpers
// (data_tools/cpp/fast_index_map_helpers.cpp:32 build_blending_indices,
// :92 build_sample_idx) for the MI355X rebuild's data layer.
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>

#include <cstdint>
#include <stdexcept>

namespace py = pybind11;

// Pack shuffled documents into (seq_len+1)-token samples; sample i spans
// sample_idx[i]..sample_idx[i+1], the boundary token shared.
static py::array_t<int64_t> build_sample_idx(
    py::array_t<int64_t, py::array::c_style> doc_lens,
    py::array_t<int32_t, py::array::c_style> doc_idx, int seq_len,
    int num_epochs, int64_t tokens_per_epoch) {
  auto lens = doc_lens.unchecked<1>();
  auto didx = doc_idx.unchecked<1>();
  const int64_t n_docs = didx.shape(0);
  const int64_t num_samples = (num_epochs * tokens_per_epoch - 1) / seq_len;

  auto out = py::array_t<int64_t>({num_samples + 1, (int64_t)2});
  auto o = out.mutable_unchecked<2>();

  int64_t si = 0, di = 0, off = 0;
  o(0, 0) = 0;
  o(0, 1) = 0;
  ++si;
  while (si <= num_samples) {
    int64_t remaining = seq_len + 1;
    while (remaining != 0) {
      if (di >= n_docs) {
        // ran out of documents: truncate
        auto trunc = py::array_t<int64_t>({si, (int64_t)2});
        auto t = trunc.mutable_unchecked<2>();
        for (int64_t i = 0; i < si; ++i) {
          t(i, 0) = o(i, 0);
          t(i, 1) = o(i, 1);
        }
        return trunc;
      }
      const int64_t doc_len = lens(didx(di)) - off;
      remaining -= doc_len;
      if (remaining <= 0) {
        off += remaining + doc_len - 1;
        remaining = 0;
      } else {
        ++di;
        off = 0;
      }
    }
    o(si, 0) = di;
    o(si, 1) = off;
    ++si;
  }
  return out;
}

// Weighted multi-dataset blending: greedy error-minimizing assignment.
static py::tuple build_blending_indices(
    py::array_t<double, py::array::c_style> weights, int64_t num_samples) {
  auto w = weights.unchecked<1>();
  const int64_t n = w.shape(0);
  auto dataset_index = py::array_t<int8_t>(num_samples);
  auto dataset_sample_index = py::array_t<int64_t>(num_samples);
  auto di = dataset_index.mutable_unchecked<1>();
  auto ds = dataset_sample_index.mutable_unchecked<1>();
  std::vector<int64_t> current(n, 0);
  for (int64_t i = 0; i < num_samples; ++i) {
    double best_err = -1e300;
    int64_t best = 0;
    for (int64_t d = 0; d < n; ++d) {
      const double err = w(d) * (double)(i + 1) - (double)current[d];
      if (err > best_err) {
        best_err = err;
        best = d;
      }
    }
    di(i) = (int8_t)best;
    ds(i) = current[best];
    ++current[best];
  }
  return py::make_tuple(dataset_index, dataset_sample_index);
}

// xorshift64 — specified exactly so the Python fallback is bit-identical
static inline uint64_t xs64(uint64_t &x) {
  x ^= x << 13;
  x ^= x >> 7;
  x ^= x << 17;
  return x;
}

// BERT/ERNIE-style sentence-pair sample mapping (semantics of the
// reference's build_mapping_impl, fast_index_map_helpers.cpp:195-430):
// docs = sentence-index boundaries per document [n_docs+1]; sizes =
// token count per sentence. Emits (sent_start, sent_end_exclusive,
// target_seq_len): sentences packed greedily up to a target length that
// is max_seq_length, or short (uniform in [4, max_seq_length]) with
// probability short_seq_prob. Deterministic in `seed`.
static py::array_t<int64_t> build_pair_mapping(
    py::array_t<int64_t, py::array::c_style> docs,
    py::array_t<int32_t, py::array::c_style> sizes, int num_epochs,
    int64_t max_num_samples, int max_seq_length, double short_seq_prob,
    uint64_t seed, int min_num_sent) {
  auto d = docs.unchecked<1>();
  auto sz = sizes.unchecked<1>();
  const int64_t n_docs = d.shape(0) - 1;
  std::vector<int64_t> flat;
  uint64_t rng = seed ? seed : 1;
  const uint64_t short_cut =
      (uint64_t)(short_seq_prob * 18446744073709551615.0);
  for (int e = 0; e < num_epochs; ++e) {
    for (int64_t doc = 0; doc < n_docs; ++doc) {
      if ((int64_t)(flat.size() / 3) >= max_num_samples) break;
      const int64_t s0 = d(doc), s1 = d(doc + 1);
      if (s1 - s0 < min_num_sent) continue;  // need a pairable doc
      int64_t start = s0;
      int64_t tok = 0;
      int nsent = 0;
      int target = max_seq_length;
      if (short_seq_prob > 0 && xs64(rng) < short_cut)
        target = 4 + (int)(xs64(rng) % (uint64_t)(max_seq_length - 3));
      for (int64_t s = s0; s < s1; ++s) {
        tok += sz(s);
        ++nsent;
        const bool last = (s + 1 == s1);
        if ((tok >= target && nsent >= min_num_sent) || last) {
          if (nsent >= min_num_sent) {
            flat.push_back(start);
            flat.push_back(s + 1);
            flat.push_back(target);
          }
          start = s + 1;
          tok = 0;
          nsent = 0;
          target = max_seq_length;
          if (short_seq_prob > 0 && xs64(rng) < short_cut)
            target = 4 + (int)(xs64(rng) % (uint64_t)(max_seq_length - 3));
          if ((int64_t)(flat.size() / 3) >= max_num_samples) break;
        }
      }
    }
    if ((int64_t)(flat.size() / 3) >= max_num_samples) break;
  }
  const int64_t n = (int64_t)(flat.size() / 3);
  auto out = py::array_t<int64_t>({n, (int64_t)3});
  auto o = out.mutable_unchecked<2>();
  for (int64_t i = 0; i < n; ++i) {
    o(i, 0) = flat[3 * i];
    o(i, 1) = flat[3 * i + 1];
    o(i, 2) = flat[3 * i + 2];
  }
  return out;
}

PYBIND11_MODULE(_index_map, m) {
  m.def("build_sample_idx", &build_sample_idx, "pack docs into samples",
        py::arg("doc_lens"), py::arg("doc_idx"), py::arg("seq_len"),
        py::arg("num_epochs"), py::arg("tokens_per_epoch"));
  m.def("build_blending_indices", &build_blending_indices,
        "weighted dataset blending", py::arg("weights"),
        py::arg("num_samples"));
  m.def("build_pair_mapping", &build_pair_mapping,
        "sentence-pair sample mapping", py::arg("docs"), py::arg("sizes"),
        py::arg("num_epochs"), py::arg("max_num_samples"),
        py::arg("max_seq_length"), py::arg("short_seq_prob"),
        py::arg("seed"), py::arg("min_num_sent") = 2);
}
