// Forest prediction for CDNA4: one thread per row traverses every tree.
//
// Reference behavior: src/predictor/gpu_predictor.cu PredictKernel /
// PredictLeafKernel.  MI355X re-design: SoA node arrays (the tree
// model's native layout — no Node-struct repacking), 256-thread blocks,
// grid-stride over rows; 28-feature rows are staged in registers.
// Categorical splits read packed bitsets (category in set -> RIGHT,
// reference common/categorical.h Decision).
#include "gbt_kernels.h"

#include <algorithm>

namespace {

__global__ __launch_bounds__(256) void PredictKernel(
    const float* __restrict__ X, int64_t n_rows, int n_features,
    float missing_value, int missing_is_nan,
    const int32_t* __restrict__ tree_offsets, const int32_t* __restrict__ left,
    const int32_t* __restrict__ right, const int32_t* __restrict__ split_index,
    const float* __restrict__ split_cond,
    const uint8_t* __restrict__ default_left,
    const uint8_t* __restrict__ split_type,
    const int32_t* __restrict__ cat_offsets,
    const uint32_t* __restrict__ cat_bits,
    const int32_t* __restrict__ tree_group, int n_trees, int n_groups,
    float* __restrict__ out_margin, int32_t* __restrict__ out_leaf) {
  const int64_t row0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (int64_t row = row0; row < n_rows;
       row += (int64_t)gridDim.x * blockDim.x) {
    const float* xrow = X + row * n_features;
    for (int t = 0; t < n_trees; ++t) {
      const int base = tree_offsets[t];
      int nid = 0;
      while (left[base + nid] != -1) {
        const int n = base + nid;
        const int f = split_index[n];
        const float v = xrow[f];
        const bool missing =
            missing_is_nan ? isnan(v) : (v == missing_value || isnan(v));
        bool go_left;
        if (missing) {
          go_left = default_left[n] != 0;
        } else if (split_type[n] != 0) {
          const int c = (int)v;
          const int w0 = cat_offsets[n];
          const int nw = cat_offsets[n + 1] - w0;
          bool in_set = false;
          if (c >= 0 && (c >> 5) < nw) {
            in_set = (cat_bits[w0 + (c >> 5)] >> (c & 31)) & 1u;
          }
          go_left = !in_set;  // stored set goes RIGHT
        } else {
          go_left = v < split_cond[n];
        }
        nid = go_left ? left[n] : right[n];
      }
      if (out_leaf != nullptr) {
        out_leaf[row * n_trees + t] = nid;
      }
      if (out_margin != nullptr) {
        out_margin[row * n_groups + tree_group[t]] += split_cond[base + nid];
      }
    }
  }
}

}  // namespace

extern "C" void gbt_predict(
    const float* X, int64_t n_rows, int n_features, float missing_value,
    int missing_is_nan, const int32_t* tree_offsets, const int32_t* left,
    const int32_t* right, const int32_t* split_index, const float* split_cond,
    const uint8_t* default_left, const uint8_t* split_type,
    const int32_t* cat_offsets, const uint32_t* cat_bits,
    const int32_t* tree_group, int n_trees, int n_groups, float* out_margin,
    int32_t* out_leaf, hipStream_t stream) {
  const int64_t blocks64 = (n_rows + 255) / 256;
  const int blocks = (int)std::min<int64_t>(blocks64, 2048 * 4);
  hipLaunchKernelGGL(PredictKernel, dim3(std::max(blocks, 1)), dim3(256), 0,
                     stream, X, n_rows, n_features, missing_value,
                     missing_is_nan, tree_offsets, left, right, split_index,
                     split_cond, default_left, split_type, cat_offsets,
                     cat_bits, tree_group, n_trees, n_groups, out_margin,
                     out_leaf);
}
