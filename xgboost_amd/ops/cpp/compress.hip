// Quantization (Ellpack compression) for CDNA4.
//
// Reference behavior: src/data/ellpack_page.cu CompressBinEllpackKernel —
// per (row, feature) binary-search the feature's cut values; here the
// output is the dense-compressed u8/u16 local-bin matrix and the bin
// rule is bin = #cuts <= value (searchsorted side='right'), matching
// HistogramCuts::SearchBin (upper_bound) and quantile.py.
#include "gbt_kernels.h"

#include <algorithm>

namespace {

__device__ __forceinline__ int UpperBound(const float* vals, int n, float v) {
  int lo = 0, hi = n;
  while (lo < hi) {
    const int mid = (lo + hi) >> 1;
    if (vals[mid] <= v) {
      lo = mid + 1;
    } else {
      hi = mid;
    }
  }
  return lo;
}

template <typename BinT>
__global__ __launch_bounds__(256) void CompressKernel(
    const float* __restrict__ X, int64_t n_rows, int n_features,
    const float* __restrict__ cut_values, const int32_t* __restrict__ cut_ptrs,
    const uint8_t* __restrict__ cat_feature, float missing_value,
    int missing_is_nan, BinT* __restrict__ out) {
  const int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t total = n_rows * n_features;
  for (int64_t i = idx; i < total; i += (int64_t)gridDim.x * blockDim.x) {
    const int f = (int)(i % n_features);
    const float v = X[i];
    const int c0 = cut_ptrs[f];
    const int nb = cut_ptrs[f + 1] - c0;
    int bin;
    const bool missing = missing_is_nan ? isnan(v) : (v == missing_value || isnan(v));
    if (missing) {
      bin = nb;  // per-feature sentinel
    } else if (cat_feature != nullptr && cat_feature[f]) {
      int c = (int)v;
      bin = c < 0 ? 0 : (c >= nb ? nb - 1 : c);
    } else {
      bin = UpperBound(cut_values + c0, nb, v);
      if (bin >= nb) bin = nb - 1;
    }
    out[i] = (BinT)bin;
  }
}

}  // namespace

extern "C" void gbt_compress(const float* X, int64_t n_rows, int n_features,
                             const float* cut_values, const int32_t* cut_ptrs,
                             const uint8_t* cat_feature, float missing_value,
                             int missing_is_nan, uint8_t* out8, uint16_t* out16,
                             hipStream_t stream) {
  const int64_t total = n_rows * n_features;
  int blocks = (int)std::min<int64_t>((total + 255) / 256, 65535);
  if (out8 != nullptr) {
    hipLaunchKernelGGL((CompressKernel<uint8_t>), dim3(blocks), dim3(256), 0,
                       stream, X, n_rows, n_features, cut_values, cut_ptrs,
                       cat_feature, missing_value, missing_is_nan, out8);
  } else {
    hipLaunchKernelGGL((CompressKernel<uint16_t>), dim3(blocks), dim3(256), 0,
                       stream, X, n_rows, n_features, cut_values, cut_ptrs,
                       cat_feature, missing_value, missing_is_nan, out16);
  }
}
