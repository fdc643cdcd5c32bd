// CPU data-pipeline helpers (native equivalent of the reference's
// megatron/data/helpers.cpp, 701 LoC): sample-index construction for the
// GPT dataset and weighted blending indices. Pure C++ loops exposed through
// the same _C extension; the Python callers fall back to numpy when the
// extension is absent (CPU-only test boxes before a build).

#include <torch/extension.h>

#include <cstdint>

// sample i of the epoch-concatenated shuffled documents starts at flat token
// position i*seq_length; returns [num_samples+1, 2] of (doc position in
// doc_idx, offset within that document). Matches the reference's
// build_sample_idx (helpers.cpp:99-162) semantics.
torch::Tensor build_sample_idx(torch::Tensor sizes, torch::Tensor doc_idx,
                               int64_t seq_length, int64_t num_samples) {
  TORCH_CHECK(sizes.dtype() == torch::kInt32);
  TORCH_CHECK(doc_idx.dtype() == torch::kInt32);
  auto sizes_a = sizes.accessor<int32_t, 1>();
  auto doc_a = doc_idx.accessor<int32_t, 1>();
  int64_t n_docs = doc_idx.size(0);

  auto out = torch::empty({num_samples + 1, 2}, torch::kInt64);
  auto out_a = out.accessor<int64_t, 2>();

  int64_t doc_pos = 0;           // position within doc_idx
  int64_t doc_offset = 0;        // token offset within current doc
  out_a[0][0] = 0;
  out_a[0][1] = 0;
  for (int64_t i = 1; i <= num_samples; ++i) {
    int64_t remaining = seq_length;
    while (remaining > 0 && doc_pos < n_docs) {
      int64_t doc_len = sizes_a[doc_a[doc_pos]] - doc_offset;
      if (doc_len > remaining) {
        doc_offset += remaining;
        remaining = 0;
      } else {
        remaining -= doc_len;
        ++doc_pos;
        doc_offset = 0;
      }
    }
    out_a[i][0] = doc_pos < n_docs ? doc_pos : n_docs - 1;
    out_a[i][1] = doc_offset;
  }
  return out;
}

// Greedy weighted mixing: at each step pick the dataset whose sampled
// fraction lags its weight the most (reference helpers.cpp:20-80).
void build_blending_indices(torch::Tensor dataset_index,
                            torch::Tensor dataset_sample_index,
                            torch::Tensor weights, int64_t num_datasets,
                            int64_t size, bool verbose) {
  TORCH_CHECK(dataset_index.dtype() == torch::kUInt8);
  TORCH_CHECK(dataset_sample_index.dtype() == torch::kInt64);
  TORCH_CHECK(weights.dtype() == torch::kFloat64);
  auto di = dataset_index.accessor<uint8_t, 1>();
  auto dsi = dataset_sample_index.accessor<int64_t, 1>();
  auto w = weights.accessor<double, 1>();

  std::vector<int64_t> current(num_datasets, 0);
  for (int64_t i = 0; i < size; ++i) {
    double max_error = -1e18;
    int64_t best = 0;
    for (int64_t d = 0; d < num_datasets; ++d) {
      double error = w[d] * (double)(i + 1) - (double)current[d];
      if (error > max_error) {
        max_error = error;
        best = d;
      }
    }
    di[i] = (uint8_t)best;
    dsi[i] = current[best];
    ++current[best];
  }
}
