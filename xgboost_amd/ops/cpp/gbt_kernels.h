// Shared declarations for the MI355X (gfx950/CDNA4) GBT kernels.
//
// Design notes (see /root/repo/SURVEY.md §2.2 for the reference kernel
// inventory these re-provide, re-designed for CDNA4):
//  - wavefront = 64 lanes; all cross-lane ops use 64-wide shuffles
//  - LDS budget 160 KiB/CU: node histograms are LDS-privatized per
//    workgroup whenever the (feature-group) bin range fits
//  - gradients are int32 fixed point, histograms int64 -> deterministic
//    sums under any atomic order (reference: gpu_hist/quantiser.cuh)
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

// One block-work descriptor: a (node, row-chunk) pair.
// hist kernel: rows ridx[row_begin..row_end) belong to node out_slot.
struct BlockTask {
  int32_t out_slot;   // index into the output hist / segment arrays
  int32_t row_begin;  // offset into ridx
  int32_t row_end;
  int32_t pad;
};

extern "C" {

void gbt_hist(const uint8_t* gidx8, const uint16_t* gidx16,
              int n_features, const int32_t* qgpair /* [n,2] */,
              const int32_t* ridx, const BlockTask* tasks, int n_tasks,
              int64_t* out_hist /* [n_slots, n_bins, 2] */, int n_bins,
              const int32_t* feat_group_start,  // [n_groups+1] feature idx
              const int32_t* bin_group_start,   // [n_groups+1] global bin idx
              int n_groups, int max_group_bins,
              const int32_t* cut_ptrs,          // [n_features+1]
              int use_shared,
              int64_t* node_sums,  // [n_slots,2] or null: per-slot pair sums
              hipStream_t stream);

void gbt_partition(const uint8_t* gidx8, const uint16_t* gidx16,
                   int n_features,
                   const uint8_t* gidx8_col,   // optional feature-major
                   const uint16_t* gidx16_col, // copy ([F][col_ld])
                   int64_t col_ld,
                   const int32_t* ridx_in, int32_t* ridx_out,
                   const BlockTask* tasks, int n_tasks,
                   const int32_t* split_feature,   // [n_slots]
                   const int32_t* split_bin_local, // [n_slots] (-1: cat)
                   const uint8_t* default_left,    // [n_slots]
                   const uint32_t* cat_bits,       // packed bitsets or null
                   const int32_t* cat_bits_offset, // [n_slots+1]
                   const int32_t* n_bins_feat,     // [n_features]
                   int32_t* counters,              // [n_slots, 2] pre-init {seg_start, seg_end}
                   hipStream_t stream);

void gbt_evaluate(const int64_t* hist /* [n_nodes, n_bins, 2] */,
                  int n_nodes, int n_bins, int n_features,
                  const int32_t* cut_ptrs, const int64_t* parent_sums,
                  const float* maxabs,  // null, or [2] device max-abs:
                                        // scales derived in-kernel
                  double g_scale, double h_scale,
                  double reg_lambda, double reg_alpha, double max_delta_step,
                  double min_child_weight,
                  const int8_t* monotone,        // [n_features] or null
                  const double* node_bounds,     // [n_nodes, 2] or null
                  const uint8_t* feature_mask,   // [n_nodes, n_features] or null
                  const uint8_t* cat_feature,    // [n_features] or null
                  double* out_gain,              // [n_nodes, n_features]
                  int32_t* out_bin,              // [n_nodes, n_features]
                  uint8_t* out_dir,              // [n_nodes, n_features]
                  int64_t* out_lsum,             // [n_nodes, n_features, 2]
                  const int32_t* k_dev,          // null, or live node count
                  int narrow_max,  // >0: scalar kernel for narrow features
                  hipStream_t stream);

// mask_stride: n_features = per-node rows, 0 = one per-tree row
void gbt_evaluate_masked(const int64_t* hist, int n_nodes, int n_bins,
                         int n_features, const int32_t* cut_ptrs,
                         const int64_t* parent_sums, const float* maxabs,
                         double g_scale, double h_scale, double reg_lambda,
                         double reg_alpha, double max_delta_step,
                         double min_child_weight, const int8_t* monotone,
                         const double* node_bounds, int mask_stride,
                         const uint8_t* feature_mask,
                         const uint8_t* cat_feature, double* out_gain,
                         int32_t* out_bin, uint8_t* out_dir,
                         int64_t* out_lsum, const int32_t* k_dev,
                         int narrow_max, hipStream_t stream);

void gbt_compress(const float* X, int64_t n_rows, int n_features,
                  const float* cut_values, const int32_t* cut_ptrs,
                  const uint8_t* cat_feature, float missing_value,
                  int missing_is_nan,
                  uint8_t* out8, uint16_t* out16, hipStream_t stream);

void gbt_predict(const float* X, int64_t n_rows, int n_features,
                 float missing_value, int missing_is_nan,
                 // forest arrays, all trees concatenated:
                 const int32_t* tree_offsets,  // [n_trees+1] node offsets
                 const int32_t* left, const int32_t* right,
                 const int32_t* split_index, const float* split_cond,
                 const uint8_t* default_left, const uint8_t* split_type,
                 const int32_t* cat_offsets,   // [total_nodes+1] into cat_bits
                 const uint32_t* cat_bits,     // packed category bitsets
                 const int32_t* tree_group,    // [n_trees]
                 int n_trees, int n_groups,
                 float* out_margin,            // [n_rows, n_groups] preinit
                 int32_t* out_leaf,            // [n_rows, n_trees] or null
                 hipStream_t stream);

void gbt_leaf_decide(const uint8_t* gidx8, const uint16_t* gidx16,
                     int n_features,
                     const uint8_t* gidx8_col, const uint16_t* gidx16_col,
                     int64_t col_ld,
                     const int32_t* ridx,
                     const BlockTask* tasks, int n_tasks,
                     const int32_t* split_feature,
                     const int32_t* split_bin_local,
                     const uint8_t* default_left, const int32_t* kids,
                     const int32_t* n_bins_feat, int32_t* out_pos,
                     hipStream_t stream);

void gbt_leaf_partition(const int32_t* ridx, const BlockTask* tasks,
                        int n_tasks, const int32_t* leaf_ids,
                        int32_t* out_pos, hipStream_t stream);

void gbt_copy_ranges(const int32_t* src, int32_t* dst,
                     const BlockTask* tasks, int n_tasks,
                     hipStream_t stream);

void gbt_select_best(const double* gain, const int32_t* bins,
                     const uint8_t* dirs, const int64_t* lsum, int n_nodes,
                     int n_features, int64_t* out_best,
                     const int32_t* k_dev, hipStream_t stream);

}  // extern "C"
