// Row partition for CDNA4: scatter each node segment's rows into
// [left | right] halves of a double buffer.
//
// Reference behavior: src/tree/gpu_hist/row_partitioner.cuh (cub
// DispatchScan partition + SortPositionCopyKernel).  MI355X re-design:
// instead of a stable segmented scan we use an UNSTABLE two-counter
// scatter — wave64 ballot counts the left/right lanes, one pair of
// device atomics per wave reserves the destination slots.  Histogram
// sums are order-independent (int64 fixed point), so stability is not
// required; this is one pass, no scan storage, and the atomic traffic
// is 2 ops per 64 rows.
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

__global__ __launch_bounds__(GBT_PART_BLOCK) void PartitionKernel(
    const uint8_t* __restrict__ gidx8, const uint16_t* __restrict__ gidx16,
    int n_features, const int32_t* __restrict__ ridx_in,
    int32_t* __restrict__ ridx_out, const BlockTask* __restrict__ tasks,
    const int32_t* __restrict__ split_feature,
    const int32_t* __restrict__ split_bin_local,
    const uint8_t* __restrict__ default_left,
    const uint32_t* __restrict__ cat_bits,
    const int32_t* __restrict__ cat_bits_offset,
    const int32_t* __restrict__ n_bins_feat,
    int32_t* __restrict__ counters) {
  const BlockTask task = tasks[blockIdx.x];
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

  const int lane = threadIdx.x & 63;

  for (int base = task.row_begin + (int)threadIdx.x; ; base += blockDim.x) {
    const bool active = base < task.row_end;
    if (__popcll(__ballot(active)) == 0) break;
    int row = -1;
    bool left = false;
    if (active) {
      row = ridx_in[base];
      int local;
      if (gidx8 != nullptr) {
        local = (int)gidx8[(size_t)row * n_features + feature];
      } else {
        local = (int)gidx16[(size_t)row * n_features + feature];
      }
      left = DecideLeft(local, fbins, sbin, dleft, cats, cat_words);
    }
    const unsigned long long left_mask = __ballot(active && left);
    const unsigned long long right_mask = __ballot(active && !left);
    const int n_left = __popcll(left_mask);
    const int n_right = __popcll(right_mask);
    int left_base = 0, right_base = 0;
    // lane 0 reserves slots for the whole wave
    if (lane == 0) {
      if (n_left) left_base = atomicAdd(&counters[2 * slot], n_left);
      if (n_right) right_base = atomicSub(&counters[2 * slot + 1], n_right) - n_right;
    }
    left_base = __shfl(left_base, 0);
    right_base = __shfl(right_base, 0);
    if (active) {
      const unsigned long long lane_lt = (1ULL << lane) - 1;
      if (left) {
        const int rank = __popcll(left_mask & lane_lt);
        ridx_out[left_base + rank] = row;
      } else {
        const int rank = __popcll(right_mask & lane_lt);
        ridx_out[right_base + rank] = row;
      }
    }
  }
}

extern "C" void gbt_partition(
    const uint8_t* gidx8, const uint16_t* gidx16, int n_features,
    const int32_t* ridx_in, int32_t* ridx_out, const BlockTask* tasks,
    int n_tasks, const int32_t* split_feature, const int32_t* split_bin_local,
    const uint8_t* default_left, const uint32_t* cat_bits,
    const int32_t* cat_bits_offset, const int32_t* n_bins_feat,
    int32_t* counters, hipStream_t stream) {
  hipLaunchKernelGGL(PartitionKernel, dim3(n_tasks), dim3(GBT_PART_BLOCK), 0,
                     stream, gidx8, gidx16, n_features, ridx_in, ridx_out,
                     tasks, split_feature, split_bin_local, default_left,
                     cat_bits, cat_bits_offset, n_bins_feat, counters);
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

extern "C" void gbt_leaf_partition(const int32_t* ridx, const BlockTask* tasks,
                                   int n_tasks, const int32_t* leaf_ids,
                                   int32_t* out_pos, hipStream_t stream) {
  hipLaunchKernelGGL(LeafPartitionKernel, dim3(n_tasks),
                     dim3(GBT_PART_BLOCK), 0, stream, ridx, tasks, leaf_ids,
                     out_pos);
}
