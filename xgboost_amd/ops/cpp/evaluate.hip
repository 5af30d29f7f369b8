// Split evaluation for CDNA4: one wave64 per (feature, node).
//
// Reference behavior: src/tree/gpu_hist/evaluate_splits.cu
// EvaluateSplitsKernel — which hard-codes 32-thread warps; this is a
// wave64 re-design: 64 lanes scan a feature's bins in 64-wide chunks
// with an int64 shuffle-based inclusive scan (exact: quantized sums),
// then compute gains in float64 with the exact operation order of the
// numpy oracle (xgboost_amd/splits.py) so results match bit-for-bit.
#include "gbt_kernels.h"

#include <algorithm>

namespace {

struct Best {
  double gain;
  int bin;       // global bin id
  int dir;       // 1 = missing-left
  long long lg;  // left sums (quantized)
  long long lh;
};

__device__ __forceinline__ bool Better(const Best& a, const Best& b) {
  // match numpy: strictly-greater gain wins; ties keep (dir asc, bin asc)
  if (a.gain != b.gain) return a.gain > b.gain;
  if (a.dir != b.dir) return a.dir < b.dir;
  return a.bin < b.bin;
}

__device__ __forceinline__ double ThresholdL1(double g, double alpha) {
  if (alpha == 0.0) return g;
  double s = (g > 0.0) ? 1.0 : ((g < 0.0) ? -1.0 : 0.0);
  double m = fabs(g) - alpha;
  if (m < 0.0) m = 0.0;
  return s * m;
}

struct Params {
  double lam, alpha, mds, mcw;
  double lo, hi;  // node weight bounds
};

__device__ __forceinline__ double CalcWeight(double g, double h,
                                             const Params& p) {
  double w = -ThresholdL1(g, p.alpha) / (h + p.lam);
  if (p.mds > 0.0) {
    w = fmin(fmax(w, -p.mds), p.mds);
  }
  return w;
}

__device__ __forceinline__ double GainGivenWeight(double g, double h, double w,
                                                  const Params& p) {
  return -(2.0 * g * w + (h + p.lam) * (w * w));
}

__device__ __forceinline__ long long ShflUpLL(long long v, int delta) {
  return __shfl_up(v, delta, 64);
}

}  // namespace

__global__ __launch_bounds__(64) void EvaluateKernel(
    const int64_t* __restrict__ hist, int n_nodes, int n_bins, int n_features,
    const int32_t* __restrict__ cut_ptrs,
    const int64_t* __restrict__ parent_sums,
    const float* __restrict__ maxabs /* null, or [2]: derive scales on
        device so no host sync is needed before the root evaluation */,
    const int32_t* __restrict__ k_dev /* null, or the live node count:
        whole-tree mode launches a worst-case grid */,
    int narrow_max /* >0: skip numeric features with <= this many bins
        (EvaluateNarrowKernel owns them — a 64-lane wave per 2-bin
        one-hot feature wastes 97% of the machine on sparse data) */,
    double g_scale, double h_scale,
    double reg_lambda, double reg_alpha, double max_delta_step,
    double min_child_weight, const int8_t* __restrict__ monotone,
    const double* __restrict__ node_bounds,
    int mask_stride /* n_features = per-node mask rows; 0 = one
                        per-tree row broadcast to every node */,
    const uint8_t* __restrict__ feature_mask,
    const uint8_t* __restrict__ cat_feature, double* __restrict__ out_gain,
    int32_t* __restrict__ out_bin, uint8_t* __restrict__ out_dir,
    int64_t* __restrict__ out_lsum) {
  const int node = blockIdx.y;
  if (k_dev != nullptr && node >= *k_dev) return;
  const int lane = threadIdx.x;
  for (int f = blockIdx.x; f < n_features; f += gridDim.x) {
  const size_t out_idx = (size_t)node * n_features + f;

  if (narrow_max > 0 && cut_ptrs[f + 1] - cut_ptrs[f] <= narrow_max &&
      (cat_feature == nullptr || !cat_feature[f])) {
    continue;  // the narrow kernel writes this slot
  }
  if (feature_mask != nullptr &&
      feature_mask[(size_t)node * mask_stride + f] == 0) {
    if (lane == 0) {
      out_gain[out_idx] = -INFINITY;
      out_bin[out_idx] = -1;
    }
    continue;
  }

  Params p;
  p.lam = reg_lambda;
  p.alpha = reg_alpha;
  p.mds = max_delta_step;
  p.mcw = min_child_weight;
  p.lo = node_bounds ? node_bounds[2 * node] : -INFINITY;
  p.hi = node_bounds ? node_bounds[2 * node + 1] : INFINITY;
  const int mono = monotone ? (int)monotone[f] : 0;
  const bool is_cat = cat_feature && cat_feature[f];

  const int fb0 = cut_ptrs[f];
  const int fb1 = cut_ptrs[f + 1];
  const long long pg = parent_sums[2 * node];
  const long long ph = parent_sums[2 * node + 1];
  if (maxabs != nullptr) {
    // identical derivation to the host/QuantizeKernel formula
    g_scale = maxabs[0] > 0.f ? 1073741824.0 / (double)maxabs[0] : 1.0;
    h_scale = maxabs[1] > 0.f ? 1073741824.0 / (double)maxabs[1] : 1.0;
  }
  const double inv_g = 1.0 / g_scale;
  const double inv_h = 1.0 / h_scale;

  const int64_t* nh = hist + (size_t)node * n_bins * 2;

  // pass 1: feature totals (wave reduce)
  long long fg = 0, fh = 0;
  for (int b = fb0 + lane; b < fb1; b += 64) {
    fg += nh[2 * b];
    fh += nh[2 * b + 1];
  }
  for (int off = 32; off > 0; off >>= 1) {
    fg += __shfl_down(fg, off, 64);
    fh += __shfl_down(fh, off, 64);
  }
  fg = __shfl(fg, 0, 64);
  fh = __shfl(fh, 0, 64);
  const long long miss_g = pg - fg;
  const long long miss_h = ph - fh;

  const double pw = CalcWeight(pg * inv_g, ph * inv_h, p);
  const double parent_gain = GainGivenWeight(pg * inv_g, ph * inv_h, pw, p);

  Best best{-INFINITY, -1, 0, 0, 0};

  for (int dir = 0; dir < 2; ++dir) {
    const long long add_g = dir ? miss_g : 0;
    const long long add_h = dir ? miss_h : 0;
    long long run_g = 0, run_h = 0;
    for (int b0 = fb0; b0 < fb1; b0 += 64) {
      const int b = b0 + lane;
      const bool valid = b < fb1;
      long long sg = valid ? (long long)nh[2 * b] : 0;
      long long sh = valid ? (long long)nh[2 * b + 1] : 0;
      long long glq, hlq;
      if (is_cat) {
        // one-vs-rest: category bin goes RIGHT
        glq = pg - sg - (dir ? 0 : miss_g);
        hlq = ph - sh - (dir ? 0 : miss_h);
      } else {
        // inclusive wave scan (int64: exact)
        for (int off = 1; off < 64; off <<= 1) {
          const long long tg = ShflUpLL(sg, off);
          const long long th = ShflUpLL(sh, off);
          if (lane >= off) {
            sg += tg;
            sh += th;
          }
        }
        sg += run_g;
        sh += run_h;
        glq = sg + add_g;
        hlq = sh + add_h;
      }
      if (valid) {
        const long long grq = pg - glq;
        const long long hrq = ph - hlq;
        const double gl = glq * inv_g;
        const double hl = hlq * inv_h;
        const double gr = grq * inv_g;
        const double hr = hrq * inv_h;
        double wl = CalcWeight(gl, hl, p);
        double wr = CalcWeight(gr, hr, p);
        wl = fmin(fmax(wl, p.lo), p.hi);
        wr = fmin(fmax(wr, p.lo), p.hi);
        // empty-side splits rejected via exact hessian counts; the
        // last-bin + missing-right split (present vs absent) is valid
        bool ok = (hl >= p.mcw) && (hr >= p.mcw) && hlq > 0 && hrq > 0;
        if (mono > 0) ok = ok && (wl <= wr);
        if (mono < 0) ok = ok && (wl >= wr);
        if (ok) {
          const double gain = GainGivenWeight(gl, hl, wl, p)
                              + GainGivenWeight(gr, hr, wr, p) - parent_gain;
          Best cand{gain, b, dir, glq, hlq};
          if (isfinite(gain) && Better(cand, best)) best = cand;
        }
      }
      if (!is_cat) {
        run_g = __shfl(sg, 63, 64);
        run_h = __shfl(sh, 63, 64);
      }
    }
  }

  // wave argmax reduce with full tie key
  for (int off = 32; off > 0; off >>= 1) {
    Best other;
    other.gain = __shfl_down(best.gain, off, 64);
    other.bin = __shfl_down(best.bin, off, 64);
    other.dir = __shfl_down(best.dir, off, 64);
    other.lg = __shfl_down(best.lg, off, 64);
    other.lh = __shfl_down(best.lh, off, 64);
    if (other.bin >= 0 && (best.bin < 0 || Better(other, best))) best = other;
  }
  if (lane == 0) {
    out_gain[out_idx] = best.gain;
    out_bin[out_idx] = best.bin;
    out_dir[out_idx] = (uint8_t)best.dir;
    out_lsum[2 * out_idx] = best.lg;
    out_lsum[2 * out_idx + 1] = best.lh;
  }
  }  // f stride loop
}

// Second stage: per-node argmax over features -> packed [n_nodes, 6]
// (gain bits, bin, dir, left_gq, left_hq, feature).  One wave per node;
// tie rule matches numpy flat argmax: higher gain wins, ties -> lower
// feature index (the per-feature stage already resolved bin/dir ties).
// Scalar evaluation for NARROW numeric features (one-hot scale sparse
// data: millions of 2-bin features).  One THREAD per (node, feature);
// identical fp64 operation order and tie rules as the wave kernel, so
// results are bit-identical — only the parallelization differs.
__global__ __launch_bounds__(256) void EvaluateNarrowKernel(
    const int64_t* __restrict__ hist, int n_nodes, int n_bins,
    int n_features, const int32_t* __restrict__ cut_ptrs,
    const int64_t* __restrict__ parent_sums,
    const float* __restrict__ maxabs,
    const int32_t* __restrict__ k_dev, int narrow_max,
    double g_scale, double h_scale, double reg_lambda, double reg_alpha,
    double max_delta_step, double min_child_weight,
    const int8_t* __restrict__ monotone,
    const double* __restrict__ node_bounds,
    int mask_stride, const uint8_t* __restrict__ feature_mask,
    const uint8_t* __restrict__ cat_feature, double* __restrict__ out_gain,
    int32_t* __restrict__ out_bin, uint8_t* __restrict__ out_dir,
    int64_t* __restrict__ out_lsum) {
  if (maxabs != nullptr) {
    g_scale = maxabs[0] > 0.f ? 1073741824.0 / (double)maxabs[0] : 1.0;
    h_scale = maxabs[1] > 0.f ? 1073741824.0 / (double)maxabs[1] : 1.0;
  }
  const double inv_g = 1.0 / g_scale;
  const double inv_h = 1.0 / h_scale;
  const int kn = (k_dev != nullptr) ? *k_dev : n_nodes;
  const long long total = (long long)kn * n_features;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += stride) {
    const int node = (int)(idx / n_features);
    const int f = (int)(idx % n_features);
    const int fb0 = cut_ptrs[f], fb1 = cut_ptrs[f + 1];
    if (fb1 - fb0 > narrow_max) continue;           // wave kernel's
    if (cat_feature != nullptr && cat_feature[f]) continue;  // slot
    const size_t out_idx = (size_t)node * n_features + f;
    if (feature_mask != nullptr &&
        feature_mask[(size_t)node * mask_stride + f] == 0) {
      out_gain[out_idx] = -INFINITY;
      out_bin[out_idx] = -1;
      continue;
    }
    Params p;
    p.lam = reg_lambda;
    p.alpha = reg_alpha;
    p.mds = max_delta_step;
    p.mcw = min_child_weight;
    p.lo = node_bounds ? node_bounds[2 * node] : -INFINITY;
    p.hi = node_bounds ? node_bounds[2 * node + 1] : INFINITY;
    const int mono = monotone ? (int)monotone[f] : 0;
    const long long pg = parent_sums[2 * node];
    const long long ph = parent_sums[2 * node + 1];
    const int64_t* nh = hist + (size_t)node * n_bins * 2;
    long long fg = 0, fh = 0;
    for (int b = fb0; b < fb1; ++b) {
      fg += nh[2 * b];
      fh += nh[2 * b + 1];
    }
    if (fh == 0 && fg == 0) {
      // feature absent from this node's rows: every candidate split is
      // empty-vs-all and fails the exact hessian-count validity — the
      // common case for one-hot features in deep nodes
      out_gain[out_idx] = -INFINITY;
      out_bin[out_idx] = -1;
      out_dir[out_idx] = 0;
      out_lsum[2 * out_idx] = 0;
      out_lsum[2 * out_idx + 1] = 0;
      continue;
    }
    const long long miss_g = pg - fg;
    const long long miss_h = ph - fh;
    const double pw = CalcWeight(pg * inv_g, ph * inv_h, p);
    const double parent_gain =
        GainGivenWeight(pg * inv_g, ph * inv_h, pw, p);
    Best best{-INFINITY, -1, 0, 0, 0};
    for (int dir = 0; dir < 2; ++dir) {
      const long long add_g = dir ? miss_g : 0;
      const long long add_h = dir ? miss_h : 0;
      long long sg = 0, sh = 0;
      for (int b = fb0; b < fb1; ++b) {
        sg += nh[2 * b];
        sh += nh[2 * b + 1];
        const long long glq = sg + add_g;
        const long long hlq = sh + add_h;
        const long long grq = pg - glq;
        const long long hrq = ph - hlq;
        const double gl = glq * inv_g;
        const double hl = hlq * inv_h;
        const double gr = grq * inv_g;
        const double hr = hrq * inv_h;
        double wl = CalcWeight(gl, hl, p);
        double wr = CalcWeight(gr, hr, p);
        wl = fmin(fmax(wl, p.lo), p.hi);
        wr = fmin(fmax(wr, p.lo), p.hi);
        bool ok = (hl >= p.mcw) && (hr >= p.mcw) && hlq > 0 && hrq > 0;
        if (mono > 0) ok = ok && (wl <= wr);
        if (mono < 0) ok = ok && (wl >= wr);
        if (ok) {
          const double gain = GainGivenWeight(gl, hl, wl, p)
                              + GainGivenWeight(gr, hr, wr, p) - parent_gain;
          Best cand{gain, b, dir, glq, hlq};
          if (isfinite(gain) && Better(cand, best)) best = cand;
        }
      }
    }
    out_gain[out_idx] = best.gain;
    out_bin[out_idx] = best.bin;
    out_dir[out_idx] = (uint8_t)best.dir;
    out_lsum[2 * out_idx] = best.lg;
    out_lsum[2 * out_idx + 1] = best.lh;
  }
}

// Wide-feature-count argmax: 1024 threads per node with an LDS
// cross-wave reduction (one 64-thread wave over 1e6 one-hot features
// was 5.4 ms/launch on the Criteo shape).
__global__ __launch_bounds__(1024) void SelectBestWideKernel(
    const double* __restrict__ gain, const int32_t* __restrict__ bins,
    const uint8_t* __restrict__ dirs, const int64_t* __restrict__ lsum,
    int n_features, int64_t* __restrict__ out_best,
    const int32_t* __restrict__ k_dev) {
  const int node = blockIdx.x;
  if (k_dev != nullptr && node >= *k_dev) return;
  const int lane = threadIdx.x & 63;
  const int wave = (int)threadIdx.x >> 6;
  const size_t base = (size_t)node * n_features;
  double best_gain = -INFINITY;
  int best_f = -1;
  for (int f = (int)threadIdx.x; f < n_features; f += (int)blockDim.x) {
    const double gv = gain[base + f];
    if (gv > best_gain) {
      best_gain = gv;
      best_f = f;
    }
  }
  for (int off = 32; off > 0; off >>= 1) {
    const double og = __shfl_down(best_gain, off, 64);
    const int of = __shfl_down(best_f, off, 64);
    if (of >= 0 && (best_f < 0 || og > best_gain ||
                    (og == best_gain && of < best_f))) {
      best_gain = og;
      best_f = of;
    }
  }
  __shared__ double s_g[1024 / 64];
  __shared__ int s_f[1024 / 64];
  if (lane == 0) {
    s_g[wave] = best_gain;
    s_f[wave] = best_f;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w2 = 1; w2 < (int)blockDim.x / 64; ++w2) {
      const double og = s_g[w2];
      const int of = s_f[w2];
      if (of >= 0 && (best_f < 0 || og > best_gain ||
                      (og == best_gain && of < best_f))) {
        best_gain = og;
        best_f = of;
      }
    }
    int64_t* out = out_best + (size_t)node * 6;
    if (best_f < 0 || !isfinite(best_gain)) {
      out[0] = 0;
      out[1] = -1;
      out[2] = 0;
      out[3] = 0;
      out[4] = 0;
      out[5] = -1;
    } else {
      const size_t idx = base + best_f;
      out[0] = __double_as_longlong(best_gain);
      out[1] = bins[idx];
      out[2] = dirs[idx];
      out[3] = lsum[2 * idx];
      out[4] = lsum[2 * idx + 1];
      out[5] = best_f;
    }
  }
}

__global__ __launch_bounds__(64) void SelectBestKernel(
    const double* __restrict__ gain, const int32_t* __restrict__ bins,
    const uint8_t* __restrict__ dirs, const int64_t* __restrict__ lsum,
    int n_features, int64_t* __restrict__ out_best,
    const int32_t* __restrict__ k_dev) {
  const int node = blockIdx.x;
  if (k_dev != nullptr && node >= *k_dev) return;
  const int lane = threadIdx.x;
  const size_t base = (size_t)node * n_features;
  double best_gain = -INFINITY;
  int best_f = -1;
  for (int f = lane; f < n_features; f += 64) {
    const double gv = gain[base + f];
    if (gv > best_gain) {
      best_gain = gv;
      best_f = f;
    }
  }
  for (int off = 32; off > 0; off >>= 1) {
    const double og = __shfl_down(best_gain, off, 64);
    const int of = __shfl_down(best_f, off, 64);
    if (of >= 0 && (best_f < 0 || og > best_gain ||
                    (og == best_gain && of < best_f))) {
      best_gain = og;
      best_f = of;
    }
  }
  if (lane == 0) {
    int64_t* out = out_best + (size_t)node * 6;
    if (best_f < 0 || !isfinite(best_gain)) {
      out[0] = 0;
      out[1] = -1;
      out[2] = 0;
      out[3] = 0;
      out[4] = 0;
      out[5] = -1;
    } else {
      const size_t idx = base + best_f;
      out[0] = __double_as_longlong(best_gain);
      out[1] = bins[idx];
      out[2] = dirs[idx];
      out[3] = lsum[2 * idx];
      out[4] = lsum[2 * idx + 1];
      out[5] = best_f;
    }
  }
}

extern "C" void gbt_select_best(const double* gain, const int32_t* bins,
                                const uint8_t* dirs, const int64_t* lsum,
                                int n_nodes, int n_features,
                                int64_t* out_best, const int32_t* k_dev,
                                 hipStream_t stream) {
  if (n_features > 4096) {
    hipLaunchKernelGGL(SelectBestWideKernel, dim3(n_nodes), dim3(1024), 0,
                       stream, gain, bins, dirs, lsum, n_features, out_best,
                       k_dev);
    return;
  }
  hipLaunchKernelGGL(SelectBestKernel, dim3(n_nodes), dim3(64), 0, stream,
                     gain, bins, dirs, lsum, n_features, out_best, k_dev);
}

extern "C" void gbt_evaluate(
    const int64_t* hist, int n_nodes, int n_bins, int n_features,
    const int32_t* cut_ptrs, const int64_t* parent_sums,
    const float* maxabs, double g_scale,
    double h_scale, double reg_lambda, double reg_alpha,
    double max_delta_step, double min_child_weight, const int8_t* monotone,
    const double* node_bounds, const uint8_t* feature_mask,
    const uint8_t* cat_feature, double* out_gain, int32_t* out_bin,
    uint8_t* out_dir, int64_t* out_lsum, const int32_t* k_dev,
    int narrow_max, hipStream_t stream) {
  gbt_evaluate_masked(hist, n_nodes, n_bins, n_features, cut_ptrs,
                      parent_sums, maxabs, g_scale, h_scale, reg_lambda,
                      reg_alpha, max_delta_step, min_child_weight, monotone,
                      node_bounds, n_features, feature_mask, cat_feature,
                      out_gain, out_bin, out_dir, out_lsum, k_dev,
                      narrow_max, stream);
}

// mask_stride: n_features = per-node mask rows; 0 = ONE per-tree row
// broadcast to every node (the native driver's colsample_bytree path)
extern "C" void gbt_evaluate_masked(
    const int64_t* hist, int n_nodes, int n_bins, int n_features,
    const int32_t* cut_ptrs, const int64_t* parent_sums,
    const float* maxabs, double g_scale,
    double h_scale, double reg_lambda, double reg_alpha,
    double max_delta_step, double min_child_weight, const int8_t* monotone,
    const double* node_bounds, int mask_stride,
    const uint8_t* feature_mask,
    const uint8_t* cat_feature, double* out_gain, int32_t* out_bin,
    uint8_t* out_dir, int64_t* out_lsum, const int32_t* k_dev,
    int narrow_max, hipStream_t stream) {
  dim3 grid(n_features > 65535 ? 65535 : n_features, n_nodes);
  hipLaunchKernelGGL(EvaluateKernel, grid, dim3(64), 0, stream, hist, n_nodes,
                     n_bins, n_features, cut_ptrs, parent_sums, maxabs, k_dev,
                     narrow_max, g_scale,
                     h_scale, reg_lambda, reg_alpha, max_delta_step,
                     min_child_weight, monotone, node_bounds, mask_stride,
                     feature_mask,
                     cat_feature, out_gain, out_bin, out_dir, out_lsum);
  if (narrow_max > 0) {
    const long long total = (long long)n_nodes * n_features;
    const int blocks =
        (int)std::min<long long>((total + 255) / 256, 16384);
    hipLaunchKernelGGL(EvaluateNarrowKernel, dim3(blocks), dim3(256), 0,
                       stream, hist, n_nodes, n_bins, n_features, cut_ptrs,
                       parent_sums, maxabs, k_dev, narrow_max, g_scale,
                       h_scale, reg_lambda, reg_alpha, max_delta_step,
                       min_child_weight, monotone, node_bounds, mask_stride,
                       feature_mask,
                       cat_feature, out_gain, out_bin, out_dir, out_lsum);
  }
}
