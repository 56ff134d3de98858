// Fast sample-index construction over mmap token bins (CPU, pybind11).
// MI355X-native equivalent of the compiled Megatron-LM dataset helpers the
// reference builds in install_setup.sh:8-12 (used inside GPTDataset index
// building, gpt_dataset_patch.py:418+). Fresh implementation of the
// standard sample→(doc, offset) mapping.
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>

#include <cstdint>
#include <stdexcept>

namespace py = pybind11;

// Map each training sample s to the (doc_idx position, token offset) where
// its seq_length+1-token window starts. Documents are concatenated in
// doc_idx order; windows advance seq_length tokens per sample.
static py::array_t<int64_t> build_sample_idx(
    py::array_t<int32_t, py::array::c_style | py::array::forcecast> sizes_,
    py::array_t<int64_t, py::array::c_style | py::array::forcecast> doc_idx_,
    int64_t seq_length, int64_t num_epochs, int64_t tokens_per_epoch) {
  const int32_t* sizes = sizes_.data();
  const int64_t* doc_idx = doc_idx_.data();
  const int64_t n_docs = doc_idx_.shape(0);

  int64_t num_samples = (num_epochs * tokens_per_epoch - 1) / seq_length;
  py::array_t<int64_t> out({num_samples + 1, (int64_t)2});
  auto o = out.mutable_unchecked<2>();

  int64_t sample = 0, di = 0, offset = 0;
  o(0, 0) = 0;
  o(0, 1) = 0;
  while (sample < num_samples) {
    int64_t remaining = seq_length + 1;  // +1: labels are inputs shifted
    while (remaining > 0) {
      if (di >= n_docs) throw std::runtime_error("doc_idx exhausted");
      int64_t doc_len = (int64_t)sizes[doc_idx[di]] - offset;
      if (doc_len >= remaining) {
        // window ends inside (or exactly at the end of) this doc; the next
        // sample re-reads the window's last token (label overlap)
        offset += remaining - 1;
        remaining = 0;
      } else {
        remaining -= doc_len;
        ++di;
        offset = 0;
      }
    }
    ++sample;
    o(sample, 0) = di;
    o(sample, 1) = offset;
  }
  return out;
}

PYBIND11_MODULE(_helpers_cpp, m) {
  m.doc() = "sample-index builders for the megatron-style data pipeline";
  m.def("build_sample_idx", &build_sample_idx, py::arg("sizes"),
        py::arg("doc_idx"), py::arg("seq_length"), py::arg("num_epochs"),
        py::arg("tokens_per_epoch"));
}
