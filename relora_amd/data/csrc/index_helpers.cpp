// MI355X-native ReLoRA framework — dataset index-map builders.
//
// pybind11 module `_index_helpers`, the native half of the Megatron-style
// data path (see relora_amd/data/gpt2_dataset.py, blendable.py).  Provides
// the same five entry points as the reference's helpers module
// (reference: peft_pretraining/megatron_dataset/helpers.cpp:749-756), with
// identical signatures, output dtypes/shapes and deterministic RNG
// (std::mt19937 for sample lengths, std::mt19937_64 seeded with seed+1 for
// the Fisher–Yates shuffle) so index maps are bit-reproducible across
// implementations and runs.
//
// Built by relora_amd/ops/build.py as a plain C++ extension (no HIP) —
// index building is host-side, one-shot, rank-0-only work.

#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>

#include <cstdint>
#include <limits>
#include <random>
#include <stdexcept>
#include <vector>

namespace py = pybind11;

namespace {

constexpr int32_t kLongSentenceLen = 512;

// ---------------------------------------------------------------------------
// build_sample_idx — pack consecutive (seq_length+1)-token windows across
// document boundaries.  Row i of the result is (index into doc_idx, starting
// token offset inside that document); sample i spans rows [i, i+1].
// (reference behavior: helpers.cpp:91-259)
// ---------------------------------------------------------------------------

template <typename IdxT>
py::array sample_idx_impl(const py::array_t<int32_t>& sizes_arr,
                          const py::array_t<int32_t>& doc_idx_arr,
                          int32_t seq_length,
                          int32_t num_epochs,
                          int64_t tokens_per_epoch) {
  if (seq_length <= 1 || num_epochs <= 0 || tokens_per_epoch <= 1) {
    throw std::invalid_argument("build_sample_idx: bad seq_length/num_epochs/tokens_per_epoch");
  }
  auto sizes = sizes_arr.template unchecked<1>();
  auto doc_idx = doc_idx_arr.template unchecked<1>();

  // Last token of sample i is the first token of sample i+1, hence the -1
  // (matches _num_epochs in gpt2_dataset.py).
  const int64_t num_samples = (int64_t(num_epochs) * tokens_per_epoch - 1) / seq_length;

  auto* out = new IdxT[2 * (num_samples + 1)];

  int64_t doc_cursor = 0;   // index into doc_idx
  int32_t doc_offset = 0;   // token offset inside the current document
  out[0] = IdxT(doc_cursor);
  out[1] = IdxT(doc_offset);

  for (int64_t s = 1; s <= num_samples; ++s) {
    int32_t remaining = seq_length + 1;
    while (remaining != 0) {
      const int32_t doc_len = sizes[doc_idx[doc_cursor]] - doc_offset;
      if (doc_len >= remaining) {
        // Window ends inside this document; next sample re-reads the last
        // token (the -1), so advance by remaining-1.
        doc_offset += remaining - 1;
        remaining = 0;
      } else {
        remaining -= doc_len;
        ++doc_cursor;
        doc_offset = 0;
      }
    }
    out[2 * s] = IdxT(doc_cursor);
    out[2 * s + 1] = IdxT(doc_offset);
  }

  py::capsule owner(out, [](void* p) { delete[] reinterpret_cast<IdxT*>(p); });
  const int64_t esz = sizeof(IdxT);
  return py::array(std::vector<int64_t>{num_samples + 1, 2},
                   std::vector<int64_t>{2 * esz, esz}, out, owner);
}

// ---------------------------------------------------------------------------
// Sentence-block builders (BERT-style; exported for API parity — unused by
// the ReLoRA causal-LM training path; reference helpers.cpp:272-747).
// Shared scan: walk sentences of each doc, emit a block when the target
// length is reached (and enough sentences remain) or the doc ends.
// ---------------------------------------------------------------------------

// Target length for build_mapping: short with probability 1/short_seq_ratio.
inline int32_t draw_target_len(int32_t short_seq_ratio, int32_t max_length,
                               std::mt19937& gen) {
  const uint32_t r = gen();
  if (r % uint32_t(short_seq_ratio) == 0) return 2 + int32_t(r % uint32_t(max_length - 1));
  return max_length;
}

template <typename IdxT>
void fisher_yates(IdxT* rows, int64_t n, int64_t width, uint64_t seed64) {
  std::mt19937_64 gen(seed64);
  for (int64_t i = n - 1; i > 0; --i) {
    const int64_t j = int64_t(gen() % uint64_t(i + 1));
    for (int64_t c = 0; c < width; ++c) std::swap(rows[width * i + c], rows[width * j + c]);
  }
}

template <typename IdxT>
py::array build_mapping_impl(const py::array_t<int64_t>& docs_arr,
                             const py::array_t<int32_t>& sizes_arr,
                             int32_t num_epochs, uint64_t max_num_samples,
                             int32_t max_seq_length, double short_seq_prob,
                             int32_t seed, bool verbose) {
  if (num_epochs <= 0 || max_seq_length <= 1 || short_seq_prob <= 0.0 ||
      short_seq_prob > 1.0 || seed <= 0) {
    throw std::invalid_argument("build_mapping: bad arguments");
  }
  auto docs = docs_arr.template unchecked<1>();
  auto sizes = sizes_arr.template unchecked<1>();
  const int64_t num_docs = docs_arr.shape(0) - 1;
  const auto short_seq_ratio = int32_t(std::lround(1.0 / short_seq_prob));

  IdxT* maps = nullptr;
  int64_t num_samples = -1;

  // Pass 0 counts (same RNG stream), pass 1 fills.
  for (int pass = 0; pass < 2; ++pass) {
    std::mt19937 gen(seed);
    const bool fill = (pass == 1);
    uint64_t map_index = 0;

    for (int32_t epoch = 0; epoch < num_epochs && map_index < max_num_samples; ++epoch) {
      for (int64_t doc = 0; doc < num_docs; ++doc) {
        const int64_t first = docs[doc], last = docs[doc + 1];
        int64_t remain = last - first;
        if (remain <= 1) continue;
        bool has_long = false;
        for (int64_t s = first; s < last; ++s) {
          if (sizes[s] > kLongSentenceLen) { has_long = true; break; }
        }
        if (has_long) continue;

        int64_t block_start = first;
        int32_t seq_len = 0, num_sent = 0;
        int32_t target = draw_target_len(short_seq_ratio, max_seq_length, gen);
        for (int64_t s = first; s < last; ++s) {
          seq_len += sizes[s];
          ++num_sent;
          --remain;
          if ((seq_len >= target && remain > 1 && num_sent > 1) || remain == 0) {
            if (fill) {
              maps[3 * map_index] = IdxT(block_start);
              maps[3 * map_index + 1] = IdxT(s + 1);
              maps[3 * map_index + 2] = IdxT(target);
            }
            ++map_index;
            block_start = s + 1;
            target = draw_target_len(short_seq_ratio, max_seq_length, gen);
            seq_len = 0;
            num_sent = 0;
          }
        }
      }
    }

    if (!fill) {
      num_samples = int64_t(map_index);
      maps = new IdxT[3 * map_index];
      if (verbose) py::print("build_mapping:", num_samples, "samples");
    }
  }

  fisher_yates(maps, num_samples, 3, uint64_t(seed) + 1);

  py::capsule owner(maps, [](void* p) { delete[] reinterpret_cast<IdxT*>(p); });
  const int64_t esz = sizeof(IdxT);
  return py::array(std::vector<int64_t>{num_samples, 3},
                   std::vector<int64_t>{3 * esz, esz}, maps, owner);
}

template <typename IdxT>
py::array build_blocks_mapping_impl(const py::array_t<int64_t>& docs_arr,
                                    const py::array_t<int32_t>& sizes_arr,
                                    const py::array_t<int32_t>& titles_arr,
                                    int32_t num_epochs, uint64_t max_num_samples,
                                    int32_t max_seq_length, int32_t seed,
                                    bool verbose, bool use_one_sent_blocks) {
  if (num_epochs <= 0 || max_seq_length <= 1 || seed <= 0) {
    throw std::invalid_argument("build_blocks_mapping: bad arguments");
  }
  auto docs = docs_arr.template unchecked<1>();
  auto sizes = sizes_arr.template unchecked<1>();
  auto titles = titles_arr.template unchecked<1>();
  const int64_t num_docs = docs_arr.shape(0) - 1;
  const int32_t min_num_sent = use_one_sent_blocks ? 1 : 2;

  IdxT* maps = nullptr;
  int64_t num_samples = -1;

  for (int pass = 0; pass < 2; ++pass) {
    const bool fill = (pass == 1);
    uint64_t map_index = 0;

    for (int32_t epoch = 0; epoch < num_epochs && map_index < max_num_samples; ++epoch) {
      int32_t block_id = 0;
      for (int64_t doc = 0; doc < num_docs; ++doc) {
        const int64_t first = docs[doc], last = docs[doc + 1];
        const int32_t target = max_seq_length - titles[doc];
        int64_t remain = last - first;
        if (remain < min_num_sent) continue;
        bool has_long = false;
        for (int64_t s = first; s < last; ++s) {
          if (sizes[s] > kLongSentenceLen) { has_long = true; break; }
        }
        if (has_long) continue;

        int64_t block_start = first;
        int32_t seq_len = 0, num_sent = 0;
        for (int64_t s = first; s < last; ++s) {
          seq_len += sizes[s];
          ++num_sent;
          --remain;
          if ((seq_len >= target && remain >= min_num_sent && num_sent >= min_num_sent) ||
              remain == 0) {
            if (fill) {
              maps[4 * map_index] = IdxT(block_start);
              maps[4 * map_index + 1] = IdxT(s + 1);
              maps[4 * map_index + 2] = IdxT(doc);
              maps[4 * map_index + 3] = IdxT(block_id);
            }
            ++map_index;
            ++block_id;
            block_start = s + 1;
            seq_len = 0;
            num_sent = 0;
          }
        }
      }
    }

    if (!fill) {
      num_samples = int64_t(map_index);
      maps = new IdxT[4 * map_index];
      if (verbose) py::print("build_blocks_mapping:", num_samples, "samples");
    }
  }

  fisher_yates(maps, num_samples, 4, uint64_t(seed) + 1);

  py::capsule owner(maps, [](void* p) { delete[] reinterpret_cast<IdxT*>(p); });
  const int64_t esz = sizeof(IdxT);
  return py::array(std::vector<int64_t>{num_samples, 4},
                   std::vector<int64_t>{4 * esz, esz}, maps, owner);
}

// ---------------------------------------------------------------------------
// build_blending_indices — greedy largest-deficit interleaving of weighted
// datasets (reference helpers.cpp:34-89).  Fills the two output arrays in
// place: which dataset each global sample comes from, and the running
// per-dataset sample counter.
// ---------------------------------------------------------------------------

void build_blending_indices(py::array_t<uint8_t>& dataset_index,
                            py::array_t<int64_t>& dataset_sample_index,
                            const py::array_t<double>& weights, int32_t num_datasets,
                            int64_t size, bool verbose) {
  auto didx = dataset_index.mutable_unchecked<1>();
  auto dsidx = dataset_sample_index.mutable_unchecked<1>();
  auto w = weights.unchecked<1>();

  std::vector<int64_t> taken(size_t(num_datasets), 0);

  for (int64_t i = 0; i < size; ++i) {
    // Deficit of dataset d after i draws is w[d]*i - taken[d]; pick the max.
    // (i clamped to >=1 so the first draw follows the weights too.)
    const double n = double(i < 1 ? 1 : i);
    int32_t best = 0;
    double best_err = w[0] * n - double(taken[0]);
    for (int32_t d = 1; d < num_datasets; ++d) {
      const double err = w[d] * n - double(taken[d]);
      if (err > best_err) { best_err = err; best = d; }
    }
    didx[i] = uint8_t(best);
    dsidx[i] = taken[best];
    ++taken[best];
  }

  if (verbose) {
    py::print("blending ratios (input -> achieved):");
    for (int32_t d = 0; d < num_datasets; ++d) {
      py::print("  dataset", d, ":", w[d], "->", double(taken[d]) / double(size));
    }
  }
}

}  // namespace

PYBIND11_MODULE(_index_helpers, m) {
  m.doc() = "Megatron-style dataset index builders (MI355X ReLoRA framework)";
  m.def("build_sample_idx_int32", &sample_idx_impl<int32_t>, py::arg("sizes"),
        py::arg("doc_idx"), py::arg("seq_length"), py::arg("num_epochs"),
        py::arg("tokens_per_epoch"));
  m.def("build_sample_idx_int64", &sample_idx_impl<int64_t>, py::arg("sizes"),
        py::arg("doc_idx"), py::arg("seq_length"), py::arg("num_epochs"),
        py::arg("tokens_per_epoch"));
  m.def("build_blending_indices", &build_blending_indices, py::arg("dataset_index"),
        py::arg("dataset_sample_index"), py::arg("weights"), py::arg("num_datasets"),
        py::arg("size"), py::arg("verbose") = false);
  m.def("build_mapping", [](const py::array_t<int64_t>& docs,
                            const py::array_t<int32_t>& sizes, int num_epochs,
                            uint64_t max_num_samples, int max_seq_length,
                            double short_seq_prob, int seed, bool verbose) {
    if (size_t(sizes.size()) > std::numeric_limits<uint32_t>::max())
      return build_mapping_impl<uint64_t>(docs, sizes, num_epochs, max_num_samples,
                                          max_seq_length, short_seq_prob, seed, verbose);
    return build_mapping_impl<uint32_t>(docs, sizes, num_epochs, max_num_samples,
                                        max_seq_length, short_seq_prob, seed, verbose);
  });
  m.def("build_blocks_mapping",
        [](const py::array_t<int64_t>& docs, const py::array_t<int32_t>& sizes,
           const py::array_t<int32_t>& titles, int num_epochs, uint64_t max_num_samples,
           int max_seq_length, int seed, bool verbose, bool use_one_sent_blocks) {
          if (size_t(sizes.size()) > std::numeric_limits<uint32_t>::max())
            return build_blocks_mapping_impl<uint64_t>(docs, sizes, titles, num_epochs,
                                                       max_num_samples, max_seq_length,
                                                       seed, verbose, use_one_sent_blocks);
          return build_blocks_mapping_impl<uint32_t>(docs, sizes, titles, num_epochs,
                                                     max_num_samples, max_seq_length,
                                                     seed, verbose, use_one_sent_blocks);
        });
}
