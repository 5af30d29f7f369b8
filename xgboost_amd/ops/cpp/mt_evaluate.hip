// Multi-target (vector-leaf) split evaluation for CDNA4.
//
// Reference behavior: src/tree/gpu_hist/multi_evaluate_splits.cu
// (ScanHistogramKernel + MT EvaluateSplitsKernel) — gain per bin is the
// SUM of per-target gains; one shared structure, per-target child sums.
//
// One wave64 per (feature, node): loops targets, int64 scan per target,
// accumulates summed gain per bin in registers (bins processed in
// 64-wide chunks; per-chunk the lane owns one bin).  After the argmax,
// a second pass re-derives the chosen bin's per-target left sums.
#include "gbt_kernels.h"

namespace {

__device__ __forceinline__ double MtThresholdL1(double g, double alpha) {
  if (alpha == 0.0) return g;
  double s = (g > 0.0) ? 1.0 : ((g < 0.0) ? -1.0 : 0.0);
  double m = fabs(g) - alpha;
  return s * (m < 0.0 ? 0.0 : m);
}

struct MtParams {
  double lam, alpha, mds, mcw;
};

__device__ __forceinline__ double MtWeight(double g, double h,
                                           const MtParams& p) {
  double w = -MtThresholdL1(g, p.alpha) / (h + p.lam);
  if (p.mds > 0.0) w = fmin(fmax(w, -p.mds), p.mds);
  return w;
}

__device__ __forceinline__ double MtGain(double g, double h, double w,
                                         const MtParams& p) {
  return -(2.0 * g * w + (h + p.lam) * (w * w));
}

// hists layout: [T, n_nodes, n_bins, 2]; parent sums [n_nodes, T, 2]
__global__ __launch_bounds__(64) void MtEvaluateKernel(
    const int64_t* __restrict__ hists, int n_targets, int n_nodes,
    int n_bins, int n_features, const int32_t* __restrict__ cut_ptrs,
    const int64_t* __restrict__ parent_sums,
    const double* __restrict__ g_scales, const double* __restrict__ h_scales,
    double reg_lambda, double reg_alpha, double max_delta_step,
    double min_child_weight, const uint8_t* __restrict__ feature_mask,
    double* __restrict__ out_gain, int32_t* __restrict__ out_bin,
    uint8_t* __restrict__ out_dir) {
  const int node = blockIdx.y;
  const int lane = threadIdx.x;
  MtParams p{reg_lambda, reg_alpha, max_delta_step, min_child_weight};
  for (int f = blockIdx.x; f < n_features; f += gridDim.x) {
    const size_t out_idx = (size_t)node * n_features + f;
    if (feature_mask != nullptr && feature_mask[out_idx] == 0) {
      if (lane == 0) {
        out_gain[out_idx] = -INFINITY;
        out_bin[out_idx] = -1;
      }
      continue;
    }
    const int fb0 = cut_ptrs[f];
    const int fb1 = cut_ptrs[f + 1];

    double best_gain = -INFINITY;
    int best_bin = -1;
    int best_dir = 0;

    double parent_gain = 0.0;
    for (int t = 0; t < n_targets; ++t) {
      const double ig = 1.0 / g_scales[t], ih = 1.0 / h_scales[t];
      const double pg = parent_sums[((size_t)node * n_targets + t) * 2] * ig;
      const double ph =
          parent_sums[((size_t)node * n_targets + t) * 2 + 1] * ih;
      parent_gain += MtGain(pg, ph, MtWeight(pg, ph, p), p);
    }

    for (int dir = 0; dir < 2; ++dir) {
      for (int b0 = fb0; b0 < fb1; b0 += 64) {
        const int b = b0 + lane;
        const bool valid = b < fb1;
        double gain = 0.0;
        double hl_tot = 0.0, hr_tot = 0.0;
        long long hlq_tot = 0, hrq_tot = 0;
        for (int t = 0; t < n_targets; ++t) {
          const int64_t* nh =
              hists + (((size_t)t * n_nodes + node) * n_bins) * 2;
          // per-target inclusive scan of this chunk + carry of previous
          // chunks: recompute prefix from fb0 (bins/feature <= ~1k so the
          // rescan cost is acceptable and keeps the kernel stateless)
          long long sg = 0, sh = 0;
          for (int q = fb0; q <= (valid ? b : b0 - 1); ++q) {
            sg += nh[2 * q];
            sh += nh[2 * q + 1];
          }
          long long fg = 0, fh = 0;
          for (int q = fb0; q < fb1; ++q) {
            fg += nh[2 * q];
            fh += nh[2 * q + 1];
          }
          const long long pgq =
              parent_sums[((size_t)node * n_targets + t) * 2];
          const long long phq =
              parent_sums[((size_t)node * n_targets + t) * 2 + 1];
          const long long missg = pgq - fg, missh = phq - fh;
          const long long glq = sg + (dir ? missg : 0);
          const long long hlq = sh + (dir ? missh : 0);
          const long long grq = pgq - glq, hrq = phq - hlq;
          const double ig = 1.0 / g_scales[t], ih = 1.0 / h_scales[t];
          const double gl = glq * ig, hl = hlq * ih;
          const double gr = grq * ig, hr = hrq * ih;
          const double wl = MtWeight(gl, hl, p);
          const double wr = MtWeight(gr, hr, p);
          gain += MtGain(gl, hl, wl, p) + MtGain(gr, hr, wr, p);
          hl_tot += hl;
          hr_tot += hr;
          hlq_tot += hlq;
          hrq_tot += hrq;
        }
        if (valid) {
          const bool ok = hl_tot >= p.mcw && hr_tot >= p.mcw &&
                          hlq_tot > 0 && hrq_tot > 0;
          const double total = gain - parent_gain;
          // lane-local: ascending (dir, bin) order => strict > keeps the
          // first maximum, matching the numpy oracle's tie rule
          if (ok && isfinite(total) && total > best_gain) {
            best_gain = total;
            best_bin = b;
            best_dir = dir;
          }
        }
      }
    }
    // wave argmax
    for (int off = 32; off > 0; off >>= 1) {
      const double og = __shfl_down(best_gain, off, 64);
      const int ob = __shfl_down(best_bin, off, 64);
      const int od = __shfl_down(best_dir, off, 64);
      if (ob >= 0 && (best_bin < 0 || og > best_gain ||
                      (og == best_gain &&
                       (od < best_dir ||
                        (od == best_dir && ob < best_bin))))) {
        best_gain = og;
        best_bin = ob;
        best_dir = od;
      }
    }
    if (lane == 0) {
      out_gain[out_idx] = best_bin >= 0 ? best_gain : -INFINITY;
      out_bin[out_idx] = best_bin;
      out_dir[out_idx] = (uint8_t)best_dir;
    }
  }
}

}  // namespace

extern "C" void gbt_mt_evaluate(
    const int64_t* hists, int n_targets, int n_nodes, int n_bins,
    int n_features, const int32_t* cut_ptrs, const int64_t* parent_sums,
    const double* g_scales, const double* h_scales, double reg_lambda,
    double reg_alpha, double max_delta_step, double min_child_weight,
    const uint8_t* feature_mask, double* out_gain, int32_t* out_bin,
    uint8_t* out_dir, hipStream_t stream) {
  dim3 grid(n_features > 65535 ? 65535 : n_features, n_nodes);
  hipLaunchKernelGGL(MtEvaluateKernel, grid, dim3(64), 0, stream, hists,
                     n_targets, n_nodes, n_bins, n_features, cut_ptrs,
                     parent_sums, g_scales, h_scales, reg_lambda, reg_alpha,
                     max_delta_step, min_child_weight, feature_mask, out_gain,
                     out_bin, out_dir);
}
