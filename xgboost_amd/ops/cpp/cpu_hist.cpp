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

template <typename BinT>
void HistSegSerial(const BinT* gidx, int n_features, const int32_t* qgpair,
                   const int64_t* ridx, int64_t begin, int64_t end,
                   const int32_t* cut_ptrs, int64_t* hist) {
  for (int64_t i = begin; i < end; ++i) {
    const int64_t row = ridx[i];
    const int64_t g = qgpair[2 * row];
    const int64_t h = qgpair[2 * row + 1];
    const BinT* rb = gidx + (size_t)row * n_features;
    for (int f = 0; f < n_features; ++f) {
      const int local_bin = (int)rb[f];
      if (local_bin >= cut_ptrs[f + 1] - cut_ptrs[f]) continue;
      const int b = cut_ptrs[f] + local_bin;
      hist[2 * b] += g;
      hist[2 * b + 1] += h;
    }
  }
}

}  // namespace

extern "C" {

void gbt_hist_cpu(const uint8_t* gidx8, const uint16_t* gidx16,
                  int n_features, const int32_t* qgpair, const int64_t* ridx,
                  const int64_t* seg_begin, const int64_t* seg_end, int n_segs,
                  const int32_t* cut_ptrs, int64_t* out_hist, int n_bins) {
#ifdef _OPENMP
  const int n_threads = omp_get_max_threads();
#else
  const int n_threads = 1;
#endif
  if (n_segs >= n_threads) {
    // deep levels: one thread per segment writing straight into its own
    // output slot — the thread-local-copy scheme would spend more time
    // zeroing and reducing 16B/bin scratch than visiting rows
#pragma omp parallel for schedule(dynamic)
    for (int s = 0; s < n_segs; ++s) {
      int64_t* hist = out_hist + (size_t)s * n_bins * 2;
      if (gidx8 != nullptr) {
        HistSegSerial<uint8_t>(gidx8, n_features, qgpair, ridx,
                               seg_begin[s], seg_end[s], cut_ptrs, hist);
      } else {
        HistSegSerial<uint16_t>(gidx16, n_features, qgpair, ridx,
                                seg_begin[s], seg_end[s], cut_ptrs, hist);
      }
    }
    return;
  }
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

// ---------------------------------------------------------------------------
// Native CPU split evaluation — same semantics/tie rules as the HIP
// EvaluateKernel + SelectBestKernel (evaluate.hip): int64 exact scans,
// fp64 gain in the numpy oracle's operation order.

namespace {

double CpuThresholdL1(double g, double alpha) {
  if (alpha == 0.0) return g;
  double s = (g > 0.0) ? 1.0 : ((g < 0.0) ? -1.0 : 0.0);
  double m = (g < 0 ? -g : g) - alpha;
  if (m < 0.0) m = 0.0;
  return s * m;
}

struct CpuEvalParams {
  double lam, alpha, mds, mcw;
};

double CpuCalcWeight(double g, double h, const CpuEvalParams& p) {
  double w = -CpuThresholdL1(g, p.alpha) / (h + p.lam);
  if (p.mds > 0.0) {
    w = std::min(std::max(w, -p.mds), p.mds);
  }
  return w;
}

double CpuGain(double g, double h, double w, const CpuEvalParams& p) {
  return -(2.0 * g * w + (h + p.lam) * (w * w));
}

struct CpuBest {
  double gain;
  int bin, dir, feature;
  long long lg, lh;
};

}  // namespace

extern "C" void gbt_evaluate_cpu(
    const int64_t* hist /* [k, n_bins, 2] */, int k, int n_bins,
    int n_features, const int32_t* cut_ptrs,
    const int64_t* parent_sums /* [k, 2] */, double g_scale, double h_scale,
    double reg_lambda, double reg_alpha, double max_delta_step,
    double min_child_weight, const int8_t* monotone,
    const double* node_bounds /* [k,2] or null */,
    const uint8_t* feature_mask /* [k, n_features] or null */,
    int64_t* out_best /* [k, 6]: gain-bits, bin, dir, lg, lh, feature */) {
  CpuEvalParams p{reg_lambda, reg_alpha, max_delta_step, min_child_weight};
  const double inv_g = 1.0 / g_scale, inv_h = 1.0 / h_scale;
  for (int node = 0; node < k; ++node) {
    const long long pg = parent_sums[2 * node];
    const long long ph = parent_sums[2 * node + 1];
    const double lo = node_bounds ? node_bounds[2 * node] : -1e300;
    const double hi = node_bounds ? node_bounds[2 * node + 1] : 1e300;
    const double pw0 = CpuCalcWeight(pg * inv_g, ph * inv_h, p);
    const double parent_gain = CpuGain(pg * inv_g, ph * inv_h, pw0, p);
    const int64_t* nh = hist + (size_t)node * n_bins * 2;
    CpuBest best{-1e300, -1, 0, -1, 0, 0};
#pragma omp parallel
    {
      CpuBest mine{-1e300, -1, 0, -1, 0, 0};
#pragma omp for schedule(static) nowait
      for (int f = 0; f < n_features; ++f) {
        if (feature_mask &&
            feature_mask[(size_t)node * n_features + f] == 0) {
          continue;
        }
        const int fb0 = cut_ptrs[f], fb1 = cut_ptrs[f + 1];
        long long fg = 0, fh = 0;
        for (int b = fb0; b < fb1; ++b) {
          fg += nh[2 * b];
          fh += nh[2 * b + 1];
        }
        const long long miss_g = pg - fg, miss_h = ph - fh;
        const int mono = monotone ? (int)monotone[f] : 0;
        for (int dir = 0; dir < 2; ++dir) {
          long long sg = 0, sh = 0;
          const long long ag = dir ? miss_g : 0, ah = dir ? miss_h : 0;
          for (int b = fb0; b < fb1; ++b) {
            sg += nh[2 * b];
            sh += nh[2 * b + 1];
            const long long glq = sg + ag, hlq = sh + ah;
            const long long grq = pg - glq, hrq = ph - hlq;
            const double gl = glq * inv_g, hl = hlq * inv_h;
            const double gr = grq * inv_g, hr = hrq * inv_h;
            double wl = CpuCalcWeight(gl, hl, p);
            double wr = CpuCalcWeight(gr, hr, p);
            wl = std::min(std::max(wl, lo), hi);
            wr = std::min(std::max(wr, lo), hi);
            bool ok = hl >= p.mcw && hr >= p.mcw && hlq > 0 && hrq > 0;
            if (mono > 0) ok = ok && wl <= wr;
            if (mono < 0) ok = ok && wl >= wr;
            if (!ok) continue;
            const double gain = CpuGain(gl, hl, wl, p) +
                                CpuGain(gr, hr, wr, p) - parent_gain;
            // (f, dir, b) iterate ascending within a thread: strict >
            // keeps the first maximum, matching the numpy tie rule
            if (std::isfinite(gain) && gain > mine.gain) {
              mine = CpuBest{gain, b, dir, f, glq, hlq};
            }
          }
        }
      }
#pragma omp critical
      {
        if (mine.feature >= 0 &&
            (best.feature < 0 || mine.gain > best.gain ||
             (mine.gain == best.gain &&
              (mine.feature < best.feature ||
               (mine.feature == best.feature &&
                (mine.dir < best.dir ||
                 (mine.dir == best.dir && mine.bin < best.bin))))))) {
          best = mine;
        }
      }
    }
    int64_t* out = out_best + (size_t)node * 6;
    if (best.feature < 0) {
      out[0] = 0;
      out[1] = -1;
      out[2] = 0;
      out[3] = 0;
      out[4] = 0;
      out[5] = -1;
    } else {
      memcpy(&out[0], &best.gain, sizeof(double));
      out[1] = best.bin;
      out[2] = best.dir;
      out[3] = best.lg;
      out[4] = best.lh;
      out[5] = best.feature;
    }
  }
}
