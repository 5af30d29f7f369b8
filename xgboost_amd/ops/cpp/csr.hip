// Sparse (quantized CSR) kernels for CDNA4.
//
// Row layout: row_ptr int64 [n+1], bin_idx int32 [nnz] of GLOBAL bin
// ids, sorted within each row (column order).  Absent feature = missing
// (reference SparsePage semantics).
//
// Histogram: one thread per row walks its nonzeros with global 64-bit
// atomics (sparse one-hot data has ~tens of nnz/row; LDS privatization
// would thrash for very wide bin spaces like 1e6-column one-hot, so the
// global path with L2-resident hot bins is the right default here).
// Partition: binary search of the split feature's bin range inside the
// row slice, then the same 3-phase block-aggregated scatter as dense.
#include "gbt_kernels.h"

namespace {

__device__ __forceinline__ int FindFeatureBin(const int32_t* bins,
                                              long long s, long long e,
                                              int lo, int hi) {
  // first element >= lo within [s, e); return bin if < hi else -1
  long long a = s, b = e;
  while (a < b) {
    const long long m = (a + b) >> 1;
    if (bins[m] < lo) {
      a = m + 1;
    } else {
      b = m;
    }
  }
  if (a < e && bins[a] < hi) return bins[a];
  return -1;
}

__global__ __launch_bounds__(256) void HistCsrKernel(
    const int64_t* __restrict__ row_ptr, const int32_t* __restrict__ bins,
    const int32_t* __restrict__ qgpair, const int32_t* __restrict__ ridx,
    const BlockTask* __restrict__ tasks, int64_t* __restrict__ out_hist,
    int n_bins) {
  const BlockTask task = tasks[blockIdx.x];
  int64_t* hist = out_hist + (size_t)task.out_slot * n_bins * 2;
  for (int i = task.row_begin + (int)threadIdx.x; i < task.row_end;
       i += blockDim.x) {
    const int row = ridx[i];
    const long long g = qgpair[2 * (size_t)row];
    const long long h = qgpair[2 * (size_t)row + 1];
    const long long s = row_ptr[row], e = row_ptr[row + 1];
    for (long long j = s; j < e; ++j) {
      const int b = bins[j];
      atomicAdd((unsigned long long*)&hist[2 * b], (unsigned long long)g);
      atomicAdd((unsigned long long*)&hist[2 * b + 1],
                (unsigned long long)h);
    }
  }
}

__global__ __launch_bounds__(256) void PartitionCsrKernel(
    const int64_t* __restrict__ row_ptr, const int32_t* __restrict__ bins,
    const int32_t* __restrict__ ridx_in, int32_t* __restrict__ ridx_out,
    const BlockTask* __restrict__ tasks,
    const int32_t* __restrict__ split_feature,
    const int32_t* __restrict__ split_bin_global,
    const uint8_t* __restrict__ default_left,
    const int32_t* __restrict__ cut_ptrs, int32_t* __restrict__ counters) {
  const BlockTask task = tasks[blockIdx.x];
  const int slot = task.out_slot;
  const int feature = split_feature[slot];
  const int sbin = split_bin_global[slot];
  const bool dleft = default_left[slot] != 0;
  const int lo = cut_ptrs[feature];
  const int hi = cut_ptrs[feature + 1];

  const int n_rows = task.row_end - task.row_begin;
  const int chunk = (n_rows + (int)blockDim.x - 1) / (int)blockDim.x;
  const int my_begin = task.row_begin + (int)threadIdx.x * chunk;
  const int my_end = min(my_begin + chunk, task.row_end);

  auto decide = [&](int row) -> bool {
    const int b = FindFeatureBin(bins, row_ptr[row], row_ptr[row + 1], lo, hi);
    if (b < 0) return dleft;
    return b <= sbin;
  };

  int my_left = 0;
  for (int i = my_begin; i < my_end; ++i) {
    my_left += decide(ridx_in[i]) ? 1 : 0;
  }
  const int my_rows = max(my_end - my_begin, 0);
  const int my_right = my_rows - my_left;

  __shared__ int wave_left[4];
  __shared__ int wave_right[4];
  __shared__ int base_l, base_r;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  int scan_l = my_left, scan_r = my_right;
  for (int off = 1; off < 64; off <<= 1) {
    const int tl = __shfl_up(scan_l, off, 64);
    const int tr = __shfl_up(scan_r, off, 64);
    if (lane >= off) {
      scan_l += tl;
      scan_r += tr;
    }
  }
  if (lane == 63) {
    wave_left[wave] = scan_l;
    wave_right[wave] = scan_r;
  }
  __syncthreads();
  int wl_off = 0, wr_off = 0;
  for (int w = 0; w < wave; ++w) {
    wl_off += wave_left[w];
    wr_off += wave_right[w];
  }
  const int excl_l = scan_l - my_left + wl_off;
  const int excl_r = scan_r - my_right + wr_off;
  if (threadIdx.x == (int)blockDim.x - 1) {
    const int tot_l = excl_l + my_left;
    const int tot_r = excl_r + my_right;
    base_l = tot_l ? atomicAdd(&counters[2 * slot], tot_l) : 0;
    base_r = tot_r ? atomicSub(&counters[2 * slot + 1], tot_r) - tot_r : 0;
  }
  __syncthreads();
  int dl = base_l + excl_l;
  int dr = base_r + excl_r;
  for (int i = my_begin; i < my_end; ++i) {
    const int row = ridx_in[i];
    if (decide(row)) {
      ridx_out[dl++] = row;
    } else {
      ridx_out[dr++] = row;
    }
  }
}

}  // namespace

extern "C" {

void gbt_hist_csr(const int64_t* row_ptr, const int32_t* bins,
                  const int32_t* qgpair, const int32_t* ridx,
                  const BlockTask* tasks, int n_tasks, int64_t* out_hist,
                  int n_bins, hipStream_t stream) {
  hipLaunchKernelGGL(HistCsrKernel, dim3(n_tasks), dim3(256), 0, stream,
                     row_ptr, bins, qgpair, ridx, tasks, out_hist, n_bins);
}

void gbt_partition_csr(const int64_t* row_ptr, const int32_t* bins,
                       const int32_t* ridx_in, int32_t* ridx_out,
                       const BlockTask* tasks, int n_tasks,
                       const int32_t* split_feature,
                       const int32_t* split_bin_global,
                       const uint8_t* default_left, const int32_t* cut_ptrs,
                       int32_t* counters, hipStream_t stream) {
  hipLaunchKernelGGL(PartitionCsrKernel, dim3(n_tasks), dim3(256), 0, stream,
                     row_ptr, bins, ridx_in, ridx_out, tasks, split_feature,
                     split_bin_global, default_left, cut_ptrs, counters);
}

}  // extern "C"
