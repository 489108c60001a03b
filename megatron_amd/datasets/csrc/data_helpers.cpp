// CPU index builders for the data pipeline (capability analog of reference
// megatron/core/datasets/helpers.cpp:145 build_sample_idx and :77
// build_blending_indices). Re-implemented for this framework: O(tokens)
// single pass building the sample index, greedy error-minimizing blend.
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>

#include <cstdint>
#include <stdexcept>
#include <vector>

namespace py = pybind11;

// Build the [num_samples + 1, 2] map from training sample -> (position in the
// shuffled document order, token offset inside that document). Each sample
// consumes seq_length tokens and shares one boundary token with its successor
// (targets are inputs shifted by one).
static py::array build_sample_idx(py::array_t<int32_t, py::array::c_style | py::array::forcecast> sizes,
                                  py::array_t<int32_t, py::array::c_style | py::array::forcecast> doc_idx,
                                  int32_t seq_length, int32_t num_epochs, int64_t tokens_per_epoch) {
  const int32_t* sizes_p = sizes.data();
  const int32_t* doc_p = doc_idx.data();
  const int64_t num_samples = (static_cast<int64_t>(num_epochs) * tokens_per_epoch - 1) / seq_length;

  auto out = py::array_t<int64_t>({num_samples + 1, static_cast<int64_t>(2)});
  int64_t* s = out.mutable_data();

  int64_t doc_cursor = 0;  // index into doc_idx
  int64_t offset = 0;      // token offset inside doc_idx[doc_cursor]
  s[0] = doc_cursor;
  s[1] = offset;
  for (int64_t i = 1; i <= num_samples; ++i) {
    int64_t remaining = seq_length + 1;  // +1: label shift overlap
    while (remaining > 0) {
      const int64_t doc_len = sizes_p[doc_p[doc_cursor]] - offset;
      if (doc_len > remaining) {
        offset += remaining - 1;  // next sample re-reads the boundary token
        remaining = 0;
      } else if (doc_len == remaining) {
        offset += remaining - 1;
        remaining = 0;
      } else {
        remaining -= doc_len;
        ++doc_cursor;
        offset = 0;
      }
    }
    s[2 * i] = doc_cursor;
    s[2 * i + 1] = offset;
  }
  return out;
}

// Greedy proportional interleave of `num_datasets` weighted datasets over
// `size` samples: at each step pick the dataset with the largest sampling
// deficit (weight * step - samples_taken).
static py::tuple build_blending_indices(py::array_t<double, py::array::c_style | py::array::forcecast> weights,
                                        int64_t size) {
  const double* w = weights.data();
  const int64_t n = weights.size();
  if (n <= 0 || n > 32767) throw std::runtime_error("bad dataset count");

  auto dataset_index = py::array_t<int16_t>(size);
  auto dataset_sample_index = py::array_t<int64_t>(size);
  int16_t* di = dataset_index.mutable_data();
  int64_t* dsi = dataset_sample_index.mutable_data();

  std::vector<int64_t> taken(n, 0);
  for (int64_t i = 0; i < size; ++i) {
    const double step = static_cast<double>(i > 0 ? i : 1);
    int64_t best = 0;
    double best_err = w[0] * step - static_cast<double>(taken[0]);
    for (int64_t d = 1; d < n; ++d) {
      const double err = w[d] * step - static_cast<double>(taken[d]);
      if (err > best_err) {
        best_err = err;
        best = d;
      }
    }
    di[i] = static_cast<int16_t>(best);
    dsi[i] = taken[best];
    ++taken[best];
  }
  return py::make_tuple(dataset_index, dataset_sample_index);
}

PYBIND11_MODULE(_data_helpers, m) {
  m.def("build_sample_idx", &build_sample_idx, py::arg("sizes"), py::arg("doc_idx"),
        py::arg("seq_length"), py::arg("num_epochs"), py::arg("tokens_per_epoch"));
  m.def("build_blending_indices", &build_blending_indices, py::arg("weights"), py::arg("size"));
}
