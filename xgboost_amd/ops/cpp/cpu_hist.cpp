// Native CPU histogram + partition (OpenMP) — the production CPU path.
//
// Reference behavior: src/tree/hist/histogram.h HistogramBuilder
// (thread-local hists reduced over blocked 2-D space, hist_util.cc
// RowsWiseBuildHistKernel) and common_row_partitioner.h.  Same int32
// gradient / int64 histogram fixed-point scheme as the HIP kernels, so
// results are bit-identical to both the GPU path and the torch oracle.
#include <algorithm>
#include <cstdint>
#include <cstring>
#include <vector>

#ifdef _OPENMP
#include <omp.h>
#endif

namespace {

template <typename BinT>
void HistOneSeg(const BinT* gidx, int n_features, const int32_t* qgpair,
                const int64_t* ridx, int64_t begin, int64_t end,
                const int32_t* cut_ptrs, int64_t* hist /* [n_bins][2] */,
                int n_bins) {
#ifdef _OPENMP
  const int n_threads = omp_get_max_threads();
#else
  const int n_threads = 1;
#endif
  const int64_t n_rows = end - begin;
  if (n_rows <= 0) return;
  const int64_t chunk = (n_rows + n_threads - 1) / n_threads;
  std::vector<std::vector<int64_t>> locals(n_threads);
#pragma omp parallel num_threads(n_threads)
  {
#ifdef _OPENMP
    const int tid = omp_get_thread_num();
#else
    const int tid = 0;
#endif
    auto& local = locals[tid];
    local.assign((size_t)n_bins * 2, 0);
    const int64_t lo = begin + tid * chunk;
    const int64_t hi = std::min(end, lo + chunk);
    for (int64_t i = lo; i < hi; ++i) {
      const int64_t row = ridx[i];
      const int64_t g = qgpair[2 * row];
      const int64_t h = qgpair[2 * row + 1];
      const BinT* rb = gidx + (size_t)row * n_features;
      for (int f = 0; f < n_features; ++f) {
        const int local_bin = (int)rb[f];
        const int width = cut_ptrs[f + 1] - cut_ptrs[f];
        if (local_bin >= width) continue;  // missing sentinel
        const int b = cut_ptrs[f] + local_bin;
        local[2 * b] += g;
        local[2 * b + 1] += h;
      }
    }
  }
  // reduce thread-local hists (parallel over bins)
#pragma omp parallel for schedule(static)
  for (int b = 0; b < n_bins * 2; ++b) {
    int64_t acc = hist[b];
    for (int t = 0; t < n_threads; ++t) {
      acc += locals[t][b];
    }
    hist[b] = acc;
  }
}

}  // namespace

extern "C" {

void gbt_hist_cpu(const uint8_t* gidx8, const uint16_t* gidx16,
                  int n_features, const int32_t* qgpair, const int64_t* ridx,
                  const int64_t* seg_begin, const int64_t* seg_end, int n_segs,
                  const int32_t* cut_ptrs, int64_t* out_hist, int n_bins) {
  for (int s = 0; s < n_segs; ++s) {
    int64_t* hist = out_hist + (size_t)s * n_bins * 2;
    if (gidx8 != nullptr) {
      HistOneSeg<uint8_t>(gidx8, n_features, qgpair, ridx, seg_begin[s],
                          seg_end[s], cut_ptrs, hist, n_bins);
    } else {
      HistOneSeg<uint16_t>(gidx16, n_features, qgpair, ridx, seg_begin[s],
                           seg_end[s], cut_ptrs, hist, n_bins);
    }
  }
}

// Stable two-sided partition of one segment; returns n_left.
// cat_words==0 -> numeric split (local_bin <= split_bin_local goes left);
// else bitset of local bins going RIGHT.
long long gbt_partition_cpu(const uint8_t* gidx8, const uint16_t* gidx16,
                            int n_features, int64_t* ridx, int64_t begin,
                            int64_t end, int feature, int split_bin_local,
                            int default_left, const uint32_t* cat_bits,
                            int cat_words, int fbins, int64_t* scratch) {
  const int64_t n = end - begin;
  int64_t nl = 0, nr = 0;
  for (int64_t i = begin; i < end; ++i) {
    const int64_t row = ridx[i];
    int local;
    if (gidx8 != nullptr) {
      local = (int)gidx8[(size_t)row * n_features + feature];
    } else {
      local = (int)gidx16[(size_t)row * n_features + feature];
    }
    bool left;
    if (local >= fbins) {
      left = default_left != 0;
    } else if (cat_words > 0) {
      const int w = local >> 5;
      const bool in_set =
          (w < cat_words) && ((cat_bits[w] >> (local & 31)) & 1u);
      left = !in_set;
    } else {
      left = local <= split_bin_local;
    }
    if (left) {
      ridx[begin + nl] = row;  // safe: nl <= i - begin
      ++nl;
    } else {
      scratch[nr++] = row;
    }
  }
  memcpy(ridx + begin + nl, scratch, (size_t)nr * sizeof(int64_t));
  (void)n;
  return nl;
}

}  // extern "C"
