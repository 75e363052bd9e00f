// Native dataset index builders (reference core/datasets/helpers.cpp:838-846
// exports build_sample_idx / build_blending_indices; same contracts here,
// fresh implementation).  CPU-only — no HIP required; compiled with g++ by
// megatronapp_amd/core/datasets/build_helpers.py and loaded via pybind11.

#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>

#include <cstdint>
#include <stdexcept>

namespace py = pybind11;

// sample_idx[i] = (position in doc_idx, offset inside that document) of the
// first token of sample i; samples are seq_length+1 tokens and may span
// document boundaries.  Walks `num_epochs` epochs of the shuffled doc_idx.
static py::array_t<int64_t> build_sample_idx(
    py::array_t<int32_t, py::array::c_style | py::array::forcecast> sizes,
    py::array_t<int32_t, py::array::c_style | py::array::forcecast> doc_idx,
    int seq_length, int num_epochs, int64_t tokens_per_epoch) {
  const auto sizes_r = sizes.unchecked<1>();
  const auto doc_r = doc_idx.unchecked<1>();
  const int64_t num_samples =
      (num_epochs * tokens_per_epoch - 1) / seq_length;

  py::array_t<int64_t> out({num_samples + 1, (int64_t)2});
  auto o = out.mutable_unchecked<2>();

  int64_t doc_pos = 0;   // index into doc_idx
  int64_t doc_off = 0;   // token offset within current doc
  o(0, 0) = 0;
  o(0, 1) = 0;
  for (int64_t s = 1; s <= num_samples; ++s) {
    int64_t remaining = seq_length;  // +1 token overlaps the next sample
    while (remaining > 0) {
      if (doc_pos >= doc_r.shape(0))
        throw std::runtime_error("sample_idx: ran out of documents");
      const int64_t doc_len = sizes_r(doc_r(doc_pos)) - doc_off;
      if (doc_len > remaining) {
        doc_off += remaining;
        remaining = 0;
      } else {
        remaining -= doc_len;
        ++doc_pos;
        doc_off = 0;
      }
    }
    o(s, 0) = doc_pos;
    o(s, 1) = doc_off;
  }
  return out;
}

// For blended datasets: assign each of `size` samples to a source dataset
// so realized proportions track `weights` greedily (reference
// build_blending_indices contract).
static py::tuple build_blending_indices(
    py::array_t<double, py::array::c_style | py::array::forcecast> weights,
    int64_t size) {
  const auto w = weights.unchecked<1>();
  const int n = (int)w.shape(0);
  py::array_t<int16_t> dataset_index(size);
  py::array_t<int64_t> dataset_sample_index(size);
  auto di = dataset_index.mutable_unchecked<1>();
  auto dsi = dataset_sample_index.mutable_unchecked<1>();
  std::vector<int64_t> counts(n, 0);
  for (int64_t i = 0; i < size; ++i) {
    // pick the dataset whose realized share lags its weight the most
    int best = 0;
    double best_err = -1e300;
    for (int d = 0; d < n; ++d) {
      const double err = w(d) * (double)(i + 1) - (double)counts[d];
      if (err > best_err) {
        best_err = err;
        best = d;
      }
    }
    di(i) = (int16_t)best;
    dsi(i) = counts[best];
    ++counts[best];
  }
  return py::make_tuple(dataset_index, dataset_sample_index);
}

PYBIND11_MODULE(_helpers, m) {
  m.def("build_sample_idx", &build_sample_idx, py::arg("sizes"),
        py::arg("doc_idx"), py::arg("seq_length"), py::arg("num_epochs"),
        py::arg("tokens_per_epoch"));
  m.def("build_blending_indices", &build_blending_indices, py::arg("weights"),
        py::arg("size"));
}
