// Dataset index builders (pybind11 module _galvatron_dataset_helpers).
//
// Reference role: galvatron/core/runtime/datasets/megatron/helpers.cpp
// (846 LoC: build_sample_idx, build_blending_indices, shuffle builders) —
// the load-bearing O(tokens) loops behind GPT-style pretraining datasets,
// re-implemented for this framework's simpler indexed format.
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>

#include <cstdint>
#include <random>
#include <stdexcept>

namespace py = pybind11;

// sample_idx[i] = (doc_index, doc_offset) of sample i's first token; each
// sample spans seq_length+1 tokens across consecutive docs (epochs wrap).
py::array_t<int64_t> build_sample_idx(py::array_t<int64_t> doc_lens_a,
                                      int64_t seq_length,
                                      int64_t num_samples) {
  const auto lens = doc_lens_a.unchecked<1>();
  const int64_t n_docs = lens.shape(0);
  int64_t total = 0;
  for (int64_t i = 0; i < n_docs; ++i) total += lens(i);
  if (total <= seq_length)
    throw std::runtime_error("corpus shorter than one sample");

  py::array_t<int64_t> out({num_samples, (int64_t)2});
  auto o = out.mutable_unchecked<2>();
  int64_t doc = 0, off = 0;
  for (int64_t s = 0; s < num_samples; ++s) {
    o(s, 0) = doc;
    o(s, 1) = off;
    int64_t remaining = seq_length;  // advance seq_length tokens (samples
                                     // overlap by 1 for the shifted labels)
    while (remaining > 0) {
      const int64_t in_doc = lens(doc) - off;
      if (in_doc > remaining) {
        off += remaining;
        remaining = 0;
      } else {
        remaining -= in_doc;
        doc = (doc + 1) % n_docs;
        off = 0;
      }
    }
  }
  return out;
}

// Weighted round-robin dataset blending (reference build_blending_indices):
// for each global sample, pick the dataset whose current ratio lags its
// weight most; record (dataset_index, within-dataset sample index).
void build_blending_indices(py::array_t<int16_t> dataset_index_a,
                            py::array_t<int64_t> dataset_sample_index_a,
                            py::array_t<double> weights_a,
                            int32_t n_datasets, int64_t size) {
  auto didx = dataset_index_a.mutable_unchecked<1>();
  auto dsidx = dataset_sample_index_a.mutable_unchecked<1>();
  const auto w = weights_a.unchecked<1>();
  std::vector<int64_t> counts(n_datasets, 0);
  for (int64_t i = 0; i < size; ++i) {
    double best_err = -1e30;
    int32_t best = 0;
    const double denom = (double)(i + 1);
    for (int32_t d = 0; d < n_datasets; ++d) {
      const double err = w(d) * denom - (double)counts[d];
      if (err > best_err) {
        best_err = err;
        best = d;
      }
    }
    didx(i) = (int16_t)best;
    dsidx(i) = counts[best];
    counts[best] += 1;
  }
}

py::array_t<int64_t> build_shuffle_idx(int64_t size, uint64_t seed) {
  py::array_t<int64_t> out(size);
  auto o = out.mutable_unchecked<1>();
  for (int64_t i = 0; i < size; ++i) o(i) = i;
  std::mt19937_64 rng(seed);
  for (int64_t i = size - 1; i > 0; --i) {
    std::uniform_int_distribution<int64_t> d(0, i);
    std::swap(o(i), o(d(rng)));
  }
  return out;
}

PYBIND11_MODULE(_galvatron_dataset_helpers, m) {
  m.doc() = "GPT pretraining dataset index builders";
  m.def("build_sample_idx", &build_sample_idx, py::arg("doc_lens"),
        py::arg("seq_length"), py::arg("num_samples"));
  m.def("build_blending_indices", &build_blending_indices);
  m.def("build_shuffle_idx", &build_shuffle_idx, py::arg("size"),
        py::arg("seed"));
}
