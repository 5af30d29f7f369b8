// Row partition for CDNA4: scatter each node segment's rows into
// [left | right] halves of a double buffer.
//
// Reference behavior: src/tree/gpu_hist/row_partitioner.cuh (cub
// DispatchScan partition + SortPositionCopyKernel).  MI355X re-design:
// three-phase block-aggregated scatter —
//   A) each thread counts left-rows in its contiguous sub-range,
//   B) block scan of per-thread counts (wave64 shfl scan + LDS wave
//      totals), one atomicAdd/atomicSub PER BLOCK reserves the global
//      left/right windows (profiling showed per-wave atomics on the
//      per-node counters serialize: 273us/launch -> ~10us),
//   C) threads re-read their sub-range and write destinations.
// Stable within a block; histogram sums are order-independent anyway
// (int64 fixed point).
#include "gbt_kernels.h"

#ifndef GBT_PART_BLOCK
#define GBT_PART_BLOCK 256
#endif

__device__ __forceinline__ bool DecideLeft(int local_bin, int fbins,
                                           int split_bin_local,
                                           bool default_left,
                                           const uint32_t* cat_bits,
                                           int cat_words) {
  if (local_bin >= fbins) {  // missing
    return default_left;
  }
  if (cat_words > 0) {       // categorical: stored set goes RIGHT
    const int w = local_bin >> 5;
    const bool in_set =
        (w < cat_words) && ((cat_bits[w] >> (local_bin & 31)) & 1u);
    return !in_set;
  }
  return local_bin <= split_bin_local;
}

// gidx_col: optional FEATURE-MAJOR copy of the bin matrix ([F][ld]).
// The partition touches ONE feature per node, so the column layout
// turns its per-row gather from a random 64-byte line across the whole
// row-major matrix (LLC-bandwidth bound, ~4x the roofline) into
// accesses confined to a single n_rows-byte column that sits in cache.
template <typename BinT>
__global__ __launch_bounds__(GBT_PART_BLOCK) void PartitionKernel(
    const BinT* __restrict__ gidx, int n_features,
    const BinT* __restrict__ gidx_col, long long col_ld,
    const int32_t* __restrict__ ridx_in, int32_t* __restrict__ ridx_out,
    const BlockTask* __restrict__ tasks,
    const int32_t* __restrict__ split_feature,
    const int32_t* __restrict__ split_bin_local,
    const uint8_t* __restrict__ default_left,
    const uint32_t* __restrict__ cat_bits,
    const int32_t* __restrict__ cat_bits_offset,
    const int32_t* __restrict__ n_bins_feat,
    int32_t* __restrict__ counters) {
  const BlockTask task = tasks[blockIdx.x];
  if (task.row_begin >= task.row_end) return;  // padded empty task
  const int slot = task.out_slot;
  const int feature = split_feature[slot];
  const int sbin = split_bin_local[slot];
  const bool dleft = default_left[slot] != 0;
  const int fbins = n_bins_feat[feature];
  const uint32_t* cats = nullptr;
  int cat_words = 0;
  if (cat_bits_offset != nullptr) {
    const int c0 = cat_bits_offset[slot];
    cat_words = cat_bits_offset[slot + 1] - c0;
    if (cat_words > 0) cats = cat_bits + c0;
  }

  const int n_rows = task.row_end - task.row_begin;
  const int chunk = (n_rows + (int)blockDim.x - 1) / (int)blockDim.x;
  const int my_begin = task.row_begin + (int)threadIdx.x * chunk;
  const int my_end = min(my_begin + chunk, task.row_end);

  // decision cache: avoid the second gather pass when the task fits
  // the LDS bitmask (64K rows = 8 KiB)
  constexpr int kMaxCacheRows = 65536;
  __shared__ uint32_t decide_bits[kMaxCacheRows / 32];
  const bool use_cache = n_rows <= kMaxCacheRows;
  if (use_cache) {
    const int n_words = (n_rows + 31) / 32;
    for (int i = threadIdx.x; i < n_words; i += blockDim.x) {
      decide_bits[i] = 0u;
    }
    __syncthreads();
  }

  const BinT* col = gidx_col ? gidx_col + (size_t)feature * col_ld
                             : nullptr;
  // phase A: count left in my contiguous sub-range
  int my_left = 0;
  for (int i = my_begin; i < my_end; ++i) {
    const int row = ridx_in[i];
    const int local = col ? (int)col[row]
                          : (int)gidx[(size_t)row * n_features + feature];
    const bool left = DecideLeft(local, fbins, sbin, dleft, cats, cat_words);
    my_left += left ? 1 : 0;
    if (use_cache) {
      const int k = i - task.row_begin;
      // each thread owns a contiguous bit range; no races within a word
      // except at range boundaries -> use atomicOr for safety
      if (left) atomicOr(&decide_bits[k >> 5], 1u << (k & 31));
    }
  }
  const int my_rows = max(my_end - my_begin, 0);
  const int my_right = my_rows - my_left;

  // phase B: block exclusive scan of (left, right) counts
  __shared__ int wave_left[GBT_PART_BLOCK / 64];
  __shared__ int wave_right[GBT_PART_BLOCK / 64];
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
  const int excl_l = scan_l - my_left + wl_off;  // exclusive prefix
  const int excl_r = scan_r - my_right + wr_off;
  if (threadIdx.x == (int)blockDim.x - 1) {
    const int tot_l = excl_l + my_left;
    const int tot_r = excl_r + my_right;
    base_l = tot_l ? atomicAdd(&counters[2 * slot], tot_l) : 0;
    base_r = tot_r ? atomicSub(&counters[2 * slot + 1], tot_r) - tot_r : 0;
  }
  __syncthreads();

  // phase C: scatter (cached decisions when they fit, else re-gather)
  int dl = base_l + excl_l;
  int dr = base_r + excl_r;
  for (int i = my_begin; i < my_end; ++i) {
    const int row = ridx_in[i];
    bool left;
    if (use_cache) {
      const int k = i - task.row_begin;
      left = (decide_bits[k >> 5] >> (k & 31)) & 1u;
    } else {
      const int local = col ? (int)col[row]
                            : (int)gidx[(size_t)row * n_features + feature];
      left = DecideLeft(local, fbins, sbin, dleft, cats, cat_words);
    }
    if (left) {
      ridx_out[dl++] = row;
    } else {
      ridx_out[dr++] = row;
    }
  }
}

extern "C" void gbt_partition(
    const uint8_t* gidx8, const uint16_t* gidx16, int n_features,
    const uint8_t* gidx8_col, const uint16_t* gidx16_col, int64_t col_ld,
    const int32_t* ridx_in, int32_t* ridx_out, const BlockTask* tasks,
    int n_tasks, const int32_t* split_feature, const int32_t* split_bin_local,
    const uint8_t* default_left, const uint32_t* cat_bits,
    const int32_t* cat_bits_offset, const int32_t* n_bins_feat,
    int32_t* counters, hipStream_t stream) {
  if (gidx8 != nullptr) {
    hipLaunchKernelGGL((PartitionKernel<uint8_t>), dim3(n_tasks),
                       dim3(GBT_PART_BLOCK), 0, stream, gidx8, n_features,
                       gidx8_col, (long long)col_ld,
                       ridx_in, ridx_out, tasks, split_feature,
                       split_bin_local, default_left, cat_bits,
                       cat_bits_offset, n_bins_feat, counters);
  } else {
    hipLaunchKernelGGL((PartitionKernel<uint16_t>), dim3(n_tasks),
                       dim3(GBT_PART_BLOCK), 0, stream, gidx16, n_features,
                       gidx16_col, (long long)col_ld,
                       ridx_in, ridx_out, tasks, split_feature,
                       split_bin_local, default_left, cat_bits,
                       cat_bits_offset, n_bins_feat, counters);
  }
}

// Copy partitioned task ranges from the scratch buffer back into the
// primary ridx buffer — one launch replaces per-segment memcpys.
__global__ __launch_bounds__(GBT_PART_BLOCK) void CopyRangesKernel(
    const int32_t* __restrict__ src, int32_t* __restrict__ dst,
    const BlockTask* __restrict__ tasks) {
  const BlockTask task = tasks[blockIdx.x];
  for (int i = task.row_begin + (int)threadIdx.x; i < task.row_end;
       i += blockDim.x) {
    dst[i] = src[i];
  }
}

extern "C" void gbt_copy_ranges(const int32_t* src, int32_t* dst,
                                const BlockTask* tasks, int n_tasks,
                                hipStream_t stream) {
  hipLaunchKernelGGL(CopyRangesKernel, dim3(n_tasks), dim3(GBT_PART_BLOCK), 0,
                     stream, src, dst, tasks);
}

__global__ __launch_bounds__(GBT_PART_BLOCK) void LeafPartitionKernel(
    const int32_t* __restrict__ ridx, const BlockTask* __restrict__ tasks,
    const int32_t* __restrict__ leaf_ids, int32_t* __restrict__ out_pos) {
  const BlockTask task = tasks[blockIdx.x];
  const int leaf = leaf_ids[task.out_slot];
  for (int i = task.row_begin + (int)threadIdx.x; i < task.row_end;
       i += blockDim.x) {
    out_pos[ridx[i]] = leaf;
  }
}

template <typename BinT>
__global__ __launch_bounds__(GBT_PART_BLOCK) void LeafDecideKernel(
    const BinT* __restrict__ gidx, int n_features,
    const BinT* __restrict__ gidx_col, long long col_ld,
    const int32_t* __restrict__ ridx, const BlockTask* __restrict__ tasks,
    const int32_t* __restrict__ split_feature,
    const int32_t* __restrict__ split_bin_local,
    const uint8_t* __restrict__ default_left,
    const int32_t* __restrict__ kids /* [k][2] = (left nid, right nid) */,
    const int32_t* __restrict__ n_bins_feat, int32_t* __restrict__ out_pos) {
  const BlockTask task = tasks[blockIdx.x];
  if (task.row_begin >= task.row_end) return;
  const int slot = task.out_slot;
  const int feature = split_feature[slot];
  const int sbin = split_bin_local[slot];
  const bool dleft = default_left[slot] != 0;
  const int fbins = n_bins_feat[feature];
  const int lnid = kids[2 * slot], rnid = kids[2 * slot + 1];
  const BinT* col = gidx_col ? gidx_col + (size_t)feature * col_ld
                             : nullptr;
  for (int i = task.row_begin + (int)threadIdx.x; i < task.row_end;
       i += blockDim.x) {
    const int row = ridx[i];
    const int local = col ? (int)col[row]
                          : (int)gidx[(size_t)row * n_features + feature];
    const bool left = DecideLeft(local, fbins, sbin, dleft, nullptr, 0);
    out_pos[row] = left ? lnid : rnid;
  }
}

extern "C" void gbt_leaf_decide(const uint8_t* gidx8, const uint16_t* gidx16,
                                int n_features,
                                const uint8_t* gidx8_col,
                                const uint16_t* gidx16_col, int64_t col_ld,
                                const int32_t* ridx,
                                const BlockTask* tasks, int n_tasks,
                                const int32_t* split_feature,
                                const int32_t* split_bin_local,
                                const uint8_t* default_left,
                                const int32_t* kids,
                                const int32_t* n_bins_feat, int32_t* out_pos,
                                hipStream_t stream) {
  if (gidx8 != nullptr) {
    hipLaunchKernelGGL((LeafDecideKernel<uint8_t>), dim3(n_tasks),
                       dim3(GBT_PART_BLOCK), 0, stream, gidx8, n_features,
                       gidx8_col, (long long)col_ld,
                       ridx, tasks, split_feature, split_bin_local,
                       default_left, kids, n_bins_feat, out_pos);
  } else {
    hipLaunchKernelGGL((LeafDecideKernel<uint16_t>), dim3(n_tasks),
                       dim3(GBT_PART_BLOCK), 0, stream, gidx16, n_features,
                       gidx16_col, (long long)col_ld,
                       ridx, tasks, split_feature, split_bin_local,
                       default_left, kids, n_bins_feat, out_pos);
  }
}

extern "C" void gbt_leaf_partition(const int32_t* ridx, const BlockTask* tasks,
                                   int n_tasks, const int32_t* leaf_ids,
                                   int32_t* out_pos, hipStream_t stream) {
  hipLaunchKernelGGL(LeafPartitionKernel, dim3(n_tasks),
                     dim3(GBT_PART_BLOCK), 0, stream, ridx, tasks, leaf_ids,
                     out_pos);
}
