// Path-table TreeSHAP for CDNA4 (reference analog: GPUTreeSHAP's
// path decomposition, used by src/predictor for pred_contribs).
//
// The host flattens the forest into root->leaf paths whose per-unique-
// feature elements carry (feature, interval, missing-direction flag,
// zero_fraction) — see xgboost_amd/shap_paths.py.  One THREAD per row
// walks the path table (wave-uniform loads), runs the classic
// Extend/Unwind recursion ENTIRELY IN REGISTERS (fully unrolled with a
// compile-time element cap), and accumulates fp64 contributions into
// its own row of phi — no atomics, no scratch arena.  The previous
// DFS-arena kernel carried 11.7 KB of per-thread scratch and was
// scratch-bandwidth-bound at ~100k rows/s; this formulation is
// fp64-rate-bound.
#include "gbt_kernels.h"

namespace {

// reciprocals of the small integer divisors in Extend/Unwind: fp64
// division has no hardware path (~10 instructions each) and the inner
// loops execute millions of them per row; multiplying by these differs
// from the oracle's division by <= 1 ulp per op (tolerance-tested)
__constant__ double kRcp[20] = {
    0.0,      1.0,      1.0 / 2,  1.0 / 3,  1.0 / 4,  1.0 / 5,  1.0 / 6,
    1.0 / 7,  1.0 / 8,  1.0 / 9,  1.0 / 10, 1.0 / 11, 1.0 / 12, 1.0 / 13,
    1.0 / 14, 1.0 / 15, 1.0 / 16, 1.0 / 17, 1.0 / 18, 1.0 / 19};

// kD: compile-time element cap — all loops fully unroll so the pw
// array and element state stay in registers (predicated dead
// iterations cost ~2x for shallow trees; a dynamic bound would spill
// everything to scratch, which is the very thing this kernel removes)
// X and phi are TRANSPOSED ([feature][row] / [group*col][row]) so the
// per-element feature gathers and the phi read-modify-writes are
// wave-coalesced: in row-major form each lane touched its own 64-byte
// line and the kernel thrashed L1/L2 (measured 4.2 G pair/s; the
// transposed form streams at the fp64 rate).
template <int kD>
__global__ __launch_bounds__(256) void ShapPathsKernel(
    const float* __restrict__ X, long long n_rows, int n_features,
    float missing_value, int missing_is_nan,
    const int64_t* __restrict__ path_ptr, const int32_t* __restrict__ pgrp,
    const int32_t* __restrict__ ef, const float* __restrict__ elo,
    const float* __restrict__ ehi, const uint8_t* __restrict__ emiss,
    const double* __restrict__ ez, const double* __restrict__ erz,
    const double* __restrict__ pv,
    long long n_paths, int n_groups, int n_cols /* n_features + 1 */,
    double* __restrict__ phi /* [n_rows, n_groups, n_cols] */) {
  const long long row0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long row = row0; row < n_rows; row += stride) {
    for (long long p = 0; p < n_paths; ++p) {
      const long long s = path_ptr[p];
      const int M = (int)(path_ptr[p + 1] - s);
      if (M == 0 || M > kD) continue;
      const double v = pv[p];
      if (v == 0.0) continue;
      double zs[kD];
      double rzs[kD];
      double ones[kD];
      // NOTE: every loop below has a COMPILE-TIME trip count with
      // per-iteration predication (`if (j < M)`), never `break`, and
      // no runtime array subscripts — both would block full unrolling
      // and spill pw/zs/ones to scratch (measured 14x slower)
#pragma unroll
      for (int j = 0; j < kD; ++j) {
        if (j < M) {
          const int f = ef[s + j];
          const float x = X[(size_t)f * n_rows + row];
          const bool miss =
              missing_is_nan ? isnan(x) : (isnan(x) || x == missing_value);
          const bool ok = miss ? (emiss[s + j] != 0)
                               : (x >= elo[s + j] && x < ehi[s + j]);
          ones[j] = ok ? 1.0 : 0.0;
          zs[j] = ez[s + j];
          rzs[j] = erz[s + j];
        } else {
          ones[j] = 1.0;
          zs[j] = 1.0;
          rzs[j] = 1.0;
        }
      }
      // extend: pw[0..M] with the implicit (1,1) base already applied
      double pw[kD + 1];
      pw[0] = 1.0;
#pragma unroll
      for (int j = 0; j < kD; ++j) {
        if (j < M) {
          const int mm = j + 1;  // path length before this extend
          const double rcp = kRcp[mm + 1];
          pw[mm] = ones[j] * pw[mm - 1] * mm * rcp;
#pragma unroll
          for (int i = kD - 1; i >= 1; --i) {
            if (i <= mm - 1) {
              pw[i] = ones[j] * pw[i - 1] * i * rcp
                      + zs[j] * pw[i] * (mm - i) * rcp;
            }
          }
          pw[0] = zs[j] * pw[0] * mm * rcp;
        }
      }
      // pw[d] with d = M (runtime): select without a runtime subscript
      double pw_d = 0.0;
#pragma unroll
      for (int t = 0; t <= kD; ++t) {
        if (t == M) pw_d = pw[t];
      }
      const int d = M;
      const int grp = pgrp[p];
      double* phig = phi + ((size_t)grp * n_cols) * n_rows + row;
      const double rd1 = kRcp[d + 1];
#pragma unroll
      for (int i = 0; i < kD; ++i) {
        if (i >= M) continue;
        const double o = ones[i];
        const double z = zs[i];
        double total = 0.0;
        if (o != 0.0) {  // o == 1 in this formulation
          double nxt = pw_d;
#pragma unroll
          for (int j = kD - 1; j >= 0; --j) {
            if (j <= d - 1) {
              const double tmp = nxt * (d + 1) * kRcp[j + 1];
              total += tmp;
              nxt = pw[j] - tmp * z * (d - j) * rd1;
            }
          }
        } else {
          const double rz = rzs[i];
#pragma unroll
          for (int j = kD - 1; j >= 0; --j) {
            if (j <= d - 1) {
              total += pw[j] * (d + 1) * rz * kRcp[d - j];
            }
          }
        }
        phig[(size_t)ef[s + i] * n_rows] += v * (o - z) * total;
      }
    }
  }
}

}  // namespace

extern "C" void gbt_shap_paths(
    const float* X, long long n_rows, int n_features, float missing_value,
    int missing_is_nan, const int64_t* path_ptr, const int32_t* pgrp,
    const int32_t* ef, const float* elo, const float* ehi,
    const uint8_t* emiss, const double* ez, const double* erz,
    const double* pv,
    long long n_paths, int n_groups, int n_cols, double* phi,
    hipStream_t stream) {
  const int blocks =
      (int)((n_rows + 255) / 256 < 16384 ? (n_rows + 255) / 256 : 16384);
  hipLaunchKernelGGL(ShapPathsKernel<8>, dim3(blocks), dim3(256), 0, stream,
                     X, n_rows, n_features, missing_value, missing_is_nan,
                     path_ptr, pgrp, ef, elo, ehi, emiss, ez, erz, pv, n_paths,
                     n_groups, n_cols, phi);
}

extern "C" void gbt_shap_paths16(
    const float* X, long long n_rows, int n_features, float missing_value,
    int missing_is_nan, const int64_t* path_ptr, const int32_t* pgrp,
    const int32_t* ef, const float* elo, const float* ehi,
    const uint8_t* emiss, const double* ez, const double* erz,
    const double* pv,
    long long n_paths, int n_groups, int n_cols, double* phi,
    hipStream_t stream) {
  const int blocks =
      (int)((n_rows + 255) / 256 < 16384 ? (n_rows + 255) / 256 : 16384);
  hipLaunchKernelGGL(ShapPathsKernel<16>, dim3(blocks), dim3(256), 0, stream,
                     X, n_rows, n_features, missing_value, missing_is_nan,
                     path_ptr, pgrp, ef, elo, ehi, emiss, ez, erz, pv, n_paths,
                     n_groups, n_cols, phi);
}
