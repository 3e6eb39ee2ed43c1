// Fast dataset index-map builders (pybind11, CPU).
//
// Re-implementation of the reference's 4 runtime-compiled helpers
// (reference: libai/data/data_utils/helpers.cpp:34-606): GPT flat-token ->
// sample maps, weighted multi-corpus blending, and BERT/T5 sentence-pair
// block maps.  Built as libai_amd/_data_helpers.so (no torch dependency).
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>

#include <algorithm>
#include <cstdint>
#include <random>
#include <stdexcept>
#include <vector>

namespace py = pybind11;

// Weighted blending of datasets: greedily assign each global sample to the
// dataset whose current fraction lags its weight most (reference
// helpers.cpp:34-84 semantics).
static py::tuple build_blending_indices(py::array_t<double> weights,
                                        int64_t num_samples) {
  auto w = weights.unchecked<1>();
  const int n = (int)w.shape(0);
  py::array_t<uint8_t> dataset_index(num_samples);
  py::array_t<int64_t> dataset_sample_index(num_samples);
  auto di = dataset_index.mutable_unchecked<1>();
  auto dsi = dataset_sample_index.mutable_unchecked<1>();
  std::vector<int64_t> counts(n, 0);
  for (int64_t i = 0; i < num_samples; ++i) {
    double best_err = -1e300;
    int best = 0;
    for (int d = 0; d < n; ++d) {
      double err = w(d) * (double)(i + 1) - (double)counts[d];
      if (err > best_err) {
        best_err = err;
        best = d;
      }
    }
    di(i) = (uint8_t)best;
    dsi(i) = counts[best];
    counts[best] += 1;
  }
  return py::make_tuple(dataset_index, dataset_sample_index);
}

// GPT sample index: [num_samples+1][2] of (doc_idx position, token offset)
// walking seq_length-token windows across the epoch-replicated doc stream
// (reference helpers.cpp:86-180 semantics).
static py::array_t<int32_t> build_sample_idx(py::array_t<int32_t> sizes,
                                             py::array_t<int32_t> doc_idx,
                                             int32_t seq_length, int32_t num_epochs,
                                             int64_t tokens_per_epoch) {
  auto sz = sizes.unchecked<1>();
  auto di = doc_idx.unchecked<1>();
  const int64_t num_samples = (num_epochs * tokens_per_epoch - 1) / seq_length;
  py::array_t<int32_t> sample_idx({num_samples + 1, (int64_t)2});
  auto out = sample_idx.mutable_unchecked<2>();

  int64_t sample = 0;
  int64_t doc_pos = 0;   // position in doc_idx
  int32_t doc_off = 0;   // token offset inside current doc
  out(0, 0) = (int32_t)doc_pos;
  out(0, 1) = doc_off;
  while (sample < num_samples) {
    int32_t remaining = seq_length + 1;  // +1 for the shifted label
    while (remaining > 0) {
      int32_t doc_len = sz(di(doc_pos)) - doc_off;
      if (doc_len > remaining) {
        doc_off += remaining - 1;  // last token reused as next sample's first
        remaining = 0;
      } else {
        remaining -= doc_len;
        if (remaining == 0) {
          // consumed exactly; keep final token as next first
          doc_off = sz(di(doc_pos)) - 1;
        } else {
          ++doc_pos;
          doc_off = 0;
        }
      }
    }
    ++sample;
    out(sample, 0) = (int32_t)doc_pos;
    out(sample, 1) = doc_off;
  }
  return sample_idx;
}

// BERT-style mapping: samples of sentence ranges [start, end) with target
// length, epoch-replicated (compact equivalent of helpers.cpp
// build_mapping).  Returns [N][3]: (start_sentence, end_sentence, target_len).
static py::array_t<int64_t> build_mapping(py::array_t<int64_t> docs,
                                          py::array_t<int32_t> sizes,
                                          int32_t num_epochs, int64_t max_num_samples,
                                          int32_t max_seq_length, double short_seq_prob,
                                          int32_t seed, bool verbose,
                                          int32_t min_num_sent) {
  auto doc = docs.unchecked<1>();
  auto sz = sizes.unchecked<1>();
  const int64_t n_docs = doc.shape(0) - 1;
  std::mt19937_64 rng(seed);
  std::uniform_real_distribution<double> uni(0.0, 1.0);

  std::vector<int64_t> rows;
  rows.reserve(std::min<int64_t>(max_num_samples * 3, int64_t(1) << 24));
  int64_t count = 0;
  for (int32_t ep = 0; ep < num_epochs && count < max_num_samples; ++ep) {
    for (int64_t d = 0; d < n_docs && count < max_num_samples; ++d) {
      const int64_t s0 = doc(d), s1 = doc(d + 1);
      int64_t sent = s0;
      while (sent < s1 && count < max_num_samples) {
        int32_t target = max_seq_length;
        if (uni(rng) < short_seq_prob)
          target = 2 + (int32_t)(uni(rng) * (max_seq_length - 2));
        int64_t end = sent;
        int64_t tok = 0;
        while (end < s1 && tok + sz(end) <= target) {
          tok += sz(end);
          ++end;
        }
        if (end == sent) end = sent + 1;  // oversized sentence: take it alone
        if (end - sent >= min_num_sent || end >= s1) {
          rows.push_back(sent);
          rows.push_back(end);
          rows.push_back(target);
          ++count;
        }
        sent = end;
      }
    }
  }
  const int64_t n = (int64_t)rows.size() / 3;
  py::array_t<int64_t> out({n, (int64_t)3});
  auto o = out.mutable_unchecked<2>();
  for (int64_t i = 0; i < n; ++i) {
    o(i, 0) = rows[i * 3];
    o(i, 1) = rows[i * 3 + 1];
    o(i, 2) = rows[i * 3 + 2];
  }
  return out;
}

// T5-style block mapping: like build_mapping but each row also carries the
// document index (reference build_blocks_mapping).  [N][4].
static py::array_t<int64_t> build_blocks_mapping(
    py::array_t<int64_t> docs, py::array_t<int32_t> sizes, py::array_t<int32_t> titles,
    int32_t num_epochs, int64_t max_num_samples, int32_t max_seq_length,
    int32_t seed, bool verbose, bool use_one_sent_blocks) {
  auto doc = docs.unchecked<1>();
  auto sz = sizes.unchecked<1>();
  const int64_t n_docs = doc.shape(0) - 1;
  std::mt19937_64 rng(seed);

  std::vector<int64_t> rows;
  int64_t count = 0;
  for (int32_t ep = 0; ep < num_epochs && count < max_num_samples; ++ep) {
    for (int64_t d = 0; d < n_docs && count < max_num_samples; ++d) {
      const int64_t s0 = doc(d), s1 = doc(d + 1);
      int64_t sent = s0;
      while (sent < s1 && count < max_num_samples) {
        int64_t end = sent;
        int64_t tok = 0;
        while (end < s1 && tok + sz(end) <= max_seq_length) {
          tok += sz(end);
          ++end;
          if (use_one_sent_blocks) break;
        }
        if (end == sent) end = sent + 1;
        rows.push_back(sent);
        rows.push_back(end);
        rows.push_back(d);
        rows.push_back(max_seq_length);
        ++count;
        sent = end;
      }
    }
  }
  const int64_t n = (int64_t)rows.size() / 4;
  py::array_t<int64_t> out({n, (int64_t)4});
  auto o = out.mutable_unchecked<2>();
  for (int64_t i = 0; i < n; ++i)
    for (int j = 0; j < 4; ++j) o(i, j) = rows[i * 4 + j];
  return out;
}

PYBIND11_MODULE(_data_helpers, m) {
  m.def("build_blending_indices", &build_blending_indices);
  m.def("build_sample_idx", &build_sample_idx);
  m.def("build_mapping", &build_mapping);
  m.def("build_blocks_mapping", &build_blocks_mapping);
}
