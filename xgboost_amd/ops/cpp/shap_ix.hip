// Path-table TreeSHAP INTERACTION values for CDNA4.
//
// Reference behavior: src/predictor/interpretability/shap.cu:1068
// (QuadratureShapInteractionTaskKernel) and the CPU conditional
// TreeSHAP (xgboost_amd/shap.py shap_interactions).
//
// Algorithm per (row, path): extend the full element path once
// (pweights pw), then for every element b UNWIND b from pw — giving
// the pweights of the path conditioned on feature b — and accumulate,
// for every other element a,
//   w = 0.5 * v * (one_b - z_b) * (one_a - z_a) * UnwoundSum_a(pw \ b)
// into out[f_a][f_b] AND out[f_b][f_a]; summed over both conditioning
// directions this reproduces the CPU implementation's
// `out[:, j, :] += diff; out[:, :, j] += diff` exactly (validated by
// tests/test_shap.py::test_path_pair_decomposition...).  O(M^3) fp64
// per (row, path) with M <= 16 path elements.
//
// Output is [group][C][C][row] (rows innermost) so each thread owns
// its row slice: coalesced accumulation, no atomics.  Diagonal and
// bias cells are completed on the host from the contribution vector:
// out[i][i] = phi_i - sum_{j != i} out[i][j].
#include "gbt_kernels.h"

namespace {

template <int kD>
__global__ __launch_bounds__(256) void ShapIxKernel(
    const float* __restrict__ X /* [F][n] transposed */, long long n_rows,
    int n_features, float missing_value, int missing_is_nan,
    const int64_t* __restrict__ path_ptr, const int32_t* __restrict__ pgrp,
    const int32_t* __restrict__ ef, const float* __restrict__ elo,
    const float* __restrict__ ehi, const uint8_t* __restrict__ emiss,
    const double* __restrict__ ez, const double* __restrict__ erz,
    const double* __restrict__ pv, long long n_paths, int n_groups,
    int n_cols /* n_features + 1 */,
    double* __restrict__ out /* [groups][C][C][n_rows] */) {
  const long long row0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  const int C = n_cols;
  for (long long row = row0; row < n_rows; row += stride) {
    for (long long p = 0; p < n_paths; ++p) {
      const long long s = path_ptr[p];
      const int M = (int)(path_ptr[p + 1] - s);
      if (M < 2 || M > kD) continue;  // need >= 2 features for a pair
      const double v = pv[p];
      if (v == 0.0) continue;
      double zs[kD], rzs[kD], ones[kD];
      int fs[kD];
      for (int j = 0; j < M; ++j) {
        const int f = ef[s + j];
        fs[j] = f;
        const float x = X[(size_t)f * n_rows + row];
        const bool miss =
            missing_is_nan ? isnan(x) : (isnan(x) || x == missing_value);
        const bool ok = miss ? (emiss[s + j] != 0)
                             : (x >= elo[s + j] && x < ehi[s + j]);
        ones[j] = ok ? 1.0 : 0.0;
        zs[j] = ez[s + j];
        rzs[j] = erz[s + j];
      }
      // extend all M elements (implicit (1,1) base at pw[0])
      double pw[kD + 1];
      pw[0] = 1.0;
      for (int j = 0; j < M; ++j) {
        const int mm = j + 1;
        const double rcp = 1.0 / (mm + 1);
        pw[mm] = ones[j] * pw[mm - 1] * mm * rcp;
        for (int i = mm - 1; i >= 1; --i) {
          pw[i] = ones[j] * pw[i - 1] * i * rcp
                  + zs[j] * pw[i] * (mm - i) * rcp;
        }
        pw[0] = zs[j] * pw[0] * mm * rcp;
      }
      const int grp = pgrp[p];
      double* og = out + (size_t)grp * C * C * n_rows + row;
      const int d = M;
      for (int b = 0; b < M; ++b) {
        const double mult = 0.5 * v * (ones[b] - zs[b]);
        if (mult == 0.0) continue;
        // unwind b: pweights of the path without b (depth d-1)
        double pwb[kD];
        if (ones[b] != 0.0) {
          double nxt = pw[d];
          for (int j = d - 1; j >= 0; --j) {
            const double t = nxt * (d + 1) / (j + 1);
            pwb[j] = t;
            nxt = pw[j] - t * zs[b] * (d - j) / (d + 1);
          }
        } else {
          const double rz = rzs[b];
          for (int j = d - 1; j >= 0; --j) {
            pwb[j] = pw[j] * (d + 1) * rz / (d - j);
          }
        }
        const int d2 = d - 1;
        for (int a = 0; a < M; ++a) {
          if (a == b) continue;
          const double oa = ones[a];
          double total = 0.0;
          if (oa != 0.0) {
            double nxt = pwb[d2];
            for (int j = d2 - 1; j >= 0; --j) {
              const double t = nxt * (d2 + 1) / (j + 1);
              total += t;
              nxt = pwb[j] - t * zs[a] * (d2 - j) / (d2 + 1);
            }
          } else {
            const double rz = rzs[a];
            for (int j = d2 - 1; j >= 0; --j) {
              total += pwb[j] * (d2 + 1) * rz / (d2 - j);
            }
          }
          const double w = mult * (oa - zs[a]) * total;
          if (w != 0.0) {
            og[((size_t)fs[a] * C + fs[b]) * n_rows] += w;
            og[((size_t)fs[b] * C + fs[a]) * n_rows] += w;
          }
        }
      }
    }
  }
}

}  // namespace

extern "C" void gbt_shap_ix(
    const float* X, long long n_rows, int n_features, float missing_value,
    int missing_is_nan, const int64_t* path_ptr, const int32_t* pgrp,
    const int32_t* ef, const float* elo, const float* ehi,
    const uint8_t* emiss, const double* ez, const double* erz,
    const double* pv, long long n_paths, int n_groups, int n_cols,
    double* out, hipStream_t stream) {
  const int blocks =
      (int)((n_rows + 255) / 256 < 16384 ? (n_rows + 255) / 256 : 16384);
  hipLaunchKernelGGL(ShapIxKernel<16>, dim3(blocks), dim3(256), 0, stream, X,
                     n_rows, n_features, missing_value, missing_is_nan,
                     path_ptr, pgrp, ef, elo, ehi, emiss, ez, erz, pv,
                     n_paths, n_groups, n_cols, out);
}
