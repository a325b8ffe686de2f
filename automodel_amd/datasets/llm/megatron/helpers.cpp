// Native index builders for the Megatron-style pretraining data pipeline.
//
// MI355X-native counterpart of the reference's in-tree C++ component
// (nemo_automodel/components/datasets/llm/megatron/helpers.cpp:75,143,266 —
// dataset blending index builders and sample-index builders). Re-implemented
// from the algorithm semantics, CPU-only, pybind11.
//
// Build: automodel_amd/datasets/llm/megatron/build.py (g++ -O3 -shared).

#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>

#include <cstdint>
#include <random>
#include <stdexcept>
#include <vector>

namespace py = pybind11;

// Pick, for each output sample i, the dataset whose running count is furthest
// below its target weight*i; record the within-dataset running sample index.
static void build_blending_indices(py::array_t<int16_t>& dataset_index,
                                   py::array_t<int64_t>& dataset_sample_index,
                                   const py::array_t<double>& weights,
                                   int32_t n_datasets, int64_t size) {
  auto di = dataset_index.mutable_unchecked<1>();
  auto dsi = dataset_sample_index.mutable_unchecked<1>();
  auto w = weights.unchecked<1>();
  std::vector<int64_t> counts(n_datasets, 0);
  for (int64_t i = 0; i < size; ++i) {
    double max_err = -1e9;
    int best = 0;
    for (int32_t d = 0; d < n_datasets; ++d) {
      double err = w(d) * (double)(i + 1) - (double)counts[d];
      if (err > max_err) {
        max_err = err;
        best = d;
      }
    }
    di(i) = (int16_t)best;
    dsi(i) = counts[best];
    counts[best] += 1;
  }
}

// Map each training sample to (document index, token offset) pairs so that
// consecutive samples tile the token stream in windows of seq_length.
// Returns int64 array [num_samples + 1][2].
static py::array_t<int64_t> build_sample_idx(const py::array_t<int32_t>& sizes,
                                             const py::array_t<int32_t>& doc_idx,
                                             int32_t seq_length, int32_t num_epochs,
                                             int64_t tokens_per_epoch) {
  auto sz = sizes.unchecked<1>();
  auto di = doc_idx.unchecked<1>();
  const int64_t num_samples = (num_epochs * tokens_per_epoch - 1) / seq_length;
  auto out = py::array_t<int64_t>({num_samples + 1, (int64_t)2});
  auto o = out.mutable_unchecked<2>();

  int64_t doc_i = 0;      // index into doc_idx
  int32_t offset = 0;     // token offset within current document
  o(0, 0) = doc_i;
  o(0, 1) = offset;
  for (int64_t s = 1; s <= num_samples; ++s) {
    int64_t remaining = seq_length;
    while (remaining > 0) {
      const int64_t doc_len = sz(di(doc_i)) - offset;
      if (doc_len > remaining) {
        offset += (int32_t)remaining;
        remaining = 0;
      } else {
        remaining -= doc_len;
        doc_i += 1;
        offset = 0;
        if (doc_i >= doc_idx.shape(0)) {
          doc_i = doc_idx.shape(0) - 1;  // clamp at stream end
          remaining = 0;
        }
      }
    }
    o(s, 0) = doc_i;
    o(s, 1) = offset;
  }
  return out;
}

// Fisher-Yates shuffle of [0, size) with a fixed seed.
static py::array_t<int32_t> build_shuffle_idx(int64_t size, uint32_t seed) {
  auto out = py::array_t<int32_t>(size);
  auto o = out.mutable_unchecked<1>();
  for (int64_t i = 0; i < size; ++i) o(i) = (int32_t)i;
  std::mt19937 rng(seed);
  for (int64_t i = size - 1; i > 0; --i) {
    std::uniform_int_distribution<int64_t> dist(0, i);
    std::swap(o(i), o(dist(rng)));
  }
  return out;
}

PYBIND11_MODULE(helpers_cpp, m) {
  m.doc() = "automodel_amd megatron data index builders (C++)";
  m.def("build_blending_indices", &build_blending_indices);
  m.def("build_sample_idx", &build_sample_idx);
  m.def("build_shuffle_idx", &build_shuffle_idx);
}
