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

PYBIND11_MODULE(_index_map, m) {
  m.def("build_sample_idx", &build_sample_idx, "pack docs into samples",
        py::arg("doc_lens"), py::arg("doc_idx"), py::arg("seq_len"),
        py::arg("num_epochs"), py::arg("tokens_per_epoch"));
  m.def("build_blending_indices", &build_blending_indices,
        "weighted dataset blending", py::arg("weights"),
        py::arg("num_samples"));
}
