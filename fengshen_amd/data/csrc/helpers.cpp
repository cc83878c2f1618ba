// C++ dataset index builders (CPU, pybind11).
//
// Behavioral parity: reference data/megatron_dataloader/helpers.cpp —
// build_sample_idx (:101, GPT contiguous-stream [num_samples+1, 2] of
// (doc idx, offset)), build_mapping (:214/:475, BERT-style [num_samples, 3]
// (start sentence, end sentence, target len) with short-seq prob),
// build_blending_indices (:34, weighted multi-corpus mix).
// Re-implemented from the behavioral spec; Python oracles live in
// fengshen_amd/data/helpers_py.py and the test suite checks equivalence.
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>

#include <cstdint>
#include <random>
#include <stdexcept>
#include <vector>

namespace py = pybind11;

// GPT-style: map sample i -> (document index, offset) so that samples are
// consecutive seq_length+1 token windows over the concatenated corpus.
static py::array build_sample_idx(py::array_t<int32_t> sizes_,
                                  py::array_t<int32_t> doc_idx_,
                                  int32_t seq_length, int32_t num_epochs,
                                  int64_t tokens_per_epoch) {
  auto sizes = sizes_.unchecked<1>();
  auto docs = doc_idx_.unchecked<1>();
  int64_t num_samples = (num_epochs * tokens_per_epoch - 1) / seq_length;
  int32_t* sample_idx = new int32_t[2 * (num_samples + 1)];

  int64_t sample_index = 0;
  int64_t doc_idx_index = 0;
  int32_t doc_offset = 0;
  sample_idx[0] = (int32_t)doc_idx_index;
  sample_idx[1] = doc_offset;
  ++sample_index;

  while (sample_index <= num_samples) {
    int64_t remaining_seq_length = seq_length + 1;
    while (remaining_seq_length != 0) {
      int32_t doc_id = docs(doc_idx_index);
      int32_t doc_length = sizes(doc_id) - doc_offset;
      remaining_seq_length -= doc_length;
      if (remaining_seq_length <= 0) {
        doc_offset += (int32_t)(remaining_seq_length + doc_length - 1);
        remaining_seq_length = 0;
      } else {
        ++doc_idx_index;
        doc_offset = 0;
      }
    }
    sample_idx[2 * sample_index] = (int32_t)doc_idx_index;
    sample_idx[2 * sample_index + 1] = doc_offset;
    ++sample_index;
  }

  const auto byte_size = (py::ssize_t)sizeof(int32_t);
  return py::array(std::vector<py::ssize_t>{num_samples + 1, 2},
                   std::vector<py::ssize_t>{2 * byte_size, byte_size},
                   sample_idx,
                   py::capsule(sample_idx, [](void* p) {
                     delete[] reinterpret_cast<int32_t*>(p);
                   }));
}

// BERT-style: [num_samples, 3] = (start sentence idx, end sentence idx,
// target sequence length); sentences greedily packed up to max_seq_length,
// with probability short_seq_prob the target length is sampled in
// [2, max_seq_length].
static py::array build_mapping(py::array_t<int64_t> docs_,
                               py::array_t<int32_t> sizes_,
                               int32_t num_epochs, uint64_t max_num_samples,
                               int32_t max_seq_length, double short_seq_prob,
                               uint64_t seed) {
  auto docs = docs_.unchecked<1>();
  auto sizes = sizes_.unchecked<1>();
  const int64_t num_docs = docs_.shape(0) - 1;
  std::mt19937_64 rng(seed);
  std::uniform_real_distribution<double> uniform(0.0, 1.0);
  std::uniform_int_distribution<int32_t> short_len(2, max_seq_length);

  std::vector<int64_t> maps;
  maps.reserve(3 * 65536);
  uint64_t num_samples = 0;
  for (int32_t epoch = 0; epoch < num_epochs; ++epoch) {
    if (num_samples >= max_num_samples) break;
    for (int64_t doc = 0; doc < num_docs; ++doc) {
      if (num_samples >= max_num_samples) break;
      const int64_t sent_start = docs(doc);
      const int64_t sent_end = docs(doc + 1);
      int64_t start = sent_start;
      int32_t target = (uniform(rng) < short_seq_prob)
                           ? short_len(rng) : max_seq_length;
      int64_t accum = 0;
      for (int64_t sent = sent_start; sent < sent_end; ++sent) {
        accum += sizes(sent);
        if (accum >= target && sent > start) {
          maps.push_back(start);
          maps.push_back(sent + 1);
          maps.push_back(target);
          ++num_samples;
          start = sent + 1;
          accum = 0;
          target = (uniform(rng) < short_seq_prob)
                       ? short_len(rng) : max_seq_length;
          if (num_samples >= max_num_samples) break;
        }
      }
      // trailing partial doc: emit if it has >= 2 sentences
      if (num_samples < max_num_samples && sent_end - start >= 2 && accum > 0) {
        maps.push_back(start);
        maps.push_back(sent_end);
        maps.push_back(accum < target ? (int64_t)accum : (int64_t)target);
        ++num_samples;
      }
    }
  }
  // shuffle triplets
  int64_t n = (int64_t)maps.size() / 3;
  for (int64_t i = n - 1; i > 0; --i) {
    std::uniform_int_distribution<int64_t> pick(0, i);
    int64_t j = pick(rng);
    for (int k = 0; k < 3; ++k) std::swap(maps[3 * i + k], maps[3 * j + k]);
  }
  int64_t* out = new int64_t[maps.size()];
  std::copy(maps.begin(), maps.end(), out);
  const auto bs = (py::ssize_t)sizeof(int64_t);
  return py::array(std::vector<py::ssize_t>{n, 3},
                   std::vector<py::ssize_t>{3 * bs, bs}, out,
                   py::capsule(out, [](void* p) {
                     delete[] reinterpret_cast<int64_t*>(p);
                   }));
}

// Weighted blending: assign `size` global samples across datasets so each
// dataset's share tracks its weight; returns (dataset_index[size],
// dataset_sample_index[size]).
static py::tuple build_blending_indices(py::array_t<double> weights_,
                                        int64_t size) {
  auto weights = weights_.unchecked<1>();
  const int32_t n = (int32_t)weights_.shape(0);
  auto* dataset_index = new int8_t[size];
  auto* dataset_sample_index = new int64_t[size];
  std::vector<int64_t> current(n, 0);
  for (int64_t i = 0; i < size; ++i) {
    double max_error = -1.0;
    int32_t pick = 0;
    for (int32_t d = 0; d < n; ++d) {
      double error = weights(d) * (double)(i + 1) - (double)current[d];
      if (error > max_error) {
        max_error = error;
        pick = d;
      }
    }
    dataset_index[i] = (int8_t)pick;
    dataset_sample_index[i] = current[pick];
    ++current[pick];
  }
  auto cap1 = py::capsule(dataset_index, [](void* p) {
    delete[] reinterpret_cast<int8_t*>(p);
  });
  auto cap2 = py::capsule(dataset_sample_index, [](void* p) {
    delete[] reinterpret_cast<int64_t*>(p);
  });
  return py::make_tuple(
      py::array(std::vector<py::ssize_t>{size}, dataset_index, cap1),
      py::array(std::vector<py::ssize_t>{size}, dataset_sample_index, cap2));
}

PYBIND11_MODULE(_helpers, m) {
  m.def("build_sample_idx", &build_sample_idx);
  m.def("build_mapping", &build_mapping);
  m.def("build_blending_indices", &build_blending_indices);
}
