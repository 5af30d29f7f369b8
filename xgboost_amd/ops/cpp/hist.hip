// Gradient histogram build for CDNA4 (gfx950).
//
// Reference behavior: src/tree/gpu_hist/histogram.cu StHistKernel — but
// re-designed for MI355X rather than translated:
//  - LDS-privatized per-workgroup histogram covering a whole feature
//    group (160 KiB LDS/CU lets one group hold up to ~9K bins at
//    16 B/bin; Higgs 28x256=7168 bins fits in a single group, so the
//    entire node histogram lives in LDS and is flushed once).
//  - 64-wide wavefronts; 1024-thread blocks (16 waves/CU at the
//    1-block LDS occupancy) hide ds_add latency.
//  - int32 quantized gradients accumulated into int64 bins via
//    atomicAdd(u64) — ds_add_u64 on LDS, global_atomic_add_u64 on HBM:
//    deterministic regardless of ordering (two's complement wraparound
//    is exact for signed fixed-point).
//  - blockIdx.x enumerates (node, row-chunk) tasks; blockIdx.y the
//    feature group (grid-filling: tasks are sized so n_tasks >= ~2048
//    when rows allow, covering 256 CUs across 8 XCDs).
#include "gbt_kernels.h"

#include <cstdlib>

#ifndef GBT_HIST_BLOCK
#define GBT_HIST_BLOCK 1024
#endif
#define GBT_HIST_MAX_F 1024  // per-block staged feature metadata cap

template <typename BinT, bool kUseShared>
__global__ __launch_bounds__(GBT_HIST_BLOCK) void HistKernel(
    const BinT* __restrict__ gidx, int n_features,
    const int32_t* __restrict__ qgpair, const int32_t* __restrict__ ridx,
    const BlockTask* __restrict__ tasks,
    int64_t* __restrict__ out_hist, int n_bins,
    const int32_t* __restrict__ feat_group_start,
    const int32_t* __restrict__ bin_group_start,
    const int32_t* __restrict__ cut_ptrs,
    int64_t* __restrict__ node_sums /* [k,2] or null: per-slot
        gradient sums, accumulated by group-0 blocks (used by the
        1-sync native driver for device-chosen siblings) */) {
  const BlockTask task = tasks[blockIdx.x];
  // device-generated task arrays are padded with empty tasks up to the
  // launched grid (task count is not host-known in the 1-sync driver);
  // return before any LDS work — uniform across the block
  if (task.row_begin >= task.row_end) return;
  const int group = blockIdx.y;
  const int f_begin = feat_group_start[group];
  const int f_end = feat_group_start[group + 1];
  const int bin_begin = bin_group_start[group];
  const int bin_end = bin_group_start[group + 1];
  const int group_bins = bin_end - bin_begin;
  const int gf = f_end - f_begin;

  // stage per-feature (start bin - bin_begin, n_bins) in LDS so the hot
  // loop does no global cut_ptrs loads.  On the LDS path the start is
  // group-relative (< 8192) so it PACKS with the width into one 32-bit
  // word — one ds_read per (row, feature) instead of two (the hist
  // kernel is LDS-issue bound; metadata was a third of the traffic)
  __shared__ unsigned s_meta[GBT_HIST_MAX_F];   // (start << 16) | width
  __shared__ int s_start[GBT_HIST_MAX_F];       // global path (wide start)
  __shared__ int s_width[GBT_HIST_MAX_F];
  const bool stage_meta = gf <= GBT_HIST_MAX_F;
  if (stage_meta) {
    for (int f = threadIdx.x; f < gf; f += blockDim.x) {
      const int c0 = cut_ptrs[f_begin + f];
      const int wdt = cut_ptrs[f_begin + f + 1] - c0;
      if (kUseShared) {
        s_meta[f] = ((unsigned)(c0 - bin_begin) << 16) | (unsigned)wdt;
      } else {
        s_start[f] = c0;
        s_width[f] = wdt;
      }
    }
  }

  extern __shared__ unsigned long long smem[];  // [group_bins][2]
  if (kUseShared) {
    for (int i = threadIdx.x; i < group_bins * 2; i += blockDim.x) {
      smem[i] = 0ULL;
    }
  }
  __syncthreads();

  unsigned long long* hist_s = smem;
  int64_t* hist_g = out_hist + (size_t)task.out_slot * n_bins * 2;

  const bool do_sums = node_sums != nullptr && group == 0;
  long long sum_g = 0, sum_h = 0;
  for (int i = task.row_begin + (int)threadIdx.x; i < task.row_end;
       i += blockDim.x) {
    const int row = ridx[i];
    const long long g = qgpair[2 * (size_t)row];
    const long long h = qgpair[2 * (size_t)row + 1];
    if (do_sums) {
      sum_g += g;
      sum_h += h;
    }
    const BinT* rowbins = gidx + (size_t)row * n_features + f_begin;
    if (stage_meta) {
      for (int f = 0; f < gf; ++f) {
        const int local = (int)rowbins[f];
        if (kUseShared) {
          const unsigned m = s_meta[f];
          if (local >= (int)(m & 0xFFFFu)) continue;  // missing sentinel
          const int sbin = (int)(m >> 16) + local;
          atomicAdd(&hist_s[2 * sbin], (unsigned long long)g);
          atomicAdd(&hist_s[2 * sbin + 1], (unsigned long long)h);
        } else {
          if (local >= s_width[f]) continue;
          const int sbin = s_start[f] + local;
          atomicAdd((unsigned long long*)&hist_g[2 * sbin],
                    (unsigned long long)g);
          atomicAdd((unsigned long long*)&hist_g[2 * sbin + 1],
                    (unsigned long long)h);
        }
      }
    } else {
      for (int f = f_begin; f < f_end; ++f) {
        const int local = (int)rowbins[f - f_begin];
        const int c0 = cut_ptrs[f];
        if (local >= cut_ptrs[f + 1] - c0) continue;
        const int gbin = c0 + local;
        if (kUseShared) {
          const int sbin = gbin - bin_begin;
          atomicAdd(&hist_s[2 * sbin], (unsigned long long)g);
          atomicAdd(&hist_s[2 * sbin + 1], (unsigned long long)h);
        } else {
          atomicAdd((unsigned long long*)&hist_g[2 * gbin],
                    (unsigned long long)g);
          atomicAdd((unsigned long long*)&hist_g[2 * gbin + 1],
                    (unsigned long long)h);
        }
      }
    }
  }

  if (kUseShared) {
    __syncthreads();
    for (int i = threadIdx.x; i < group_bins * 2; i += blockDim.x) {
      const unsigned long long v = hist_s[i];
      if (v != 0ULL) {
        atomicAdd((unsigned long long*)&hist_g[2 * bin_begin + i], v);
      }
    }
  }
  if (do_sums) {
    for (int off = 32; off > 0; off >>= 1) {
      sum_g += __shfl_down(sum_g, off, 64);
      sum_h += __shfl_down(sum_h, off, 64);
    }
    __shared__ long long wg[GBT_HIST_BLOCK / 64];
    __shared__ long long wh[GBT_HIST_BLOCK / 64];
    const int lane = (int)threadIdx.x & 63;
    const int wave = (int)threadIdx.x >> 6;
    if (lane == 0) {
      wg[wave] = sum_g;
      wh[wave] = sum_h;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      long long tg = 0, th = 0;
      for (int w = 0; w < (int)blockDim.x / 64; ++w) {
        tg += wg[w];
        th += wh[w];
      }
      if (tg) atomicAdd((unsigned long long*)&node_sums[2 * task.out_slot],
                        (unsigned long long)tg);
      if (th) atomicAdd(
          (unsigned long long*)&node_sums[2 * task.out_slot + 1],
          (unsigned long long)th);
    }
  }
}

// Register-metadata specialization (round-2 ladder): for a single
// feature group with few features (the Higgs-1M shape: 28 features,
// one 7168-bin LDS histogram) the generic kernel issues TWO LDS meta
// reads (s_start/s_width) per (row, feature) on top of the two
// ds_add_u64 — half the LDS issue traffic is metadata.  Here each
// thread caches the packed (start << 16 | width) metadata in
// registers once per block (kGF-bounded unrolled array), and when the
// row stride is dword-aligned the per-row bin bytes are fetched with
// packed 32-bit loads (4 features per load) instead of 28 byte loads.
// Atomic count is unchanged — this attacks the ISSUE-slot bound the
// round-1 PMC ladder identified (bank conflicts were only ~6%).
template <typename BinT, int kGF, bool kVec>
__global__ __launch_bounds__(GBT_HIST_BLOCK) void HistKernelReg(
    const BinT* __restrict__ gidx, int n_features,
    const int32_t* __restrict__ qgpair, const int32_t* __restrict__ ridx,
    const BlockTask* __restrict__ tasks,
    int64_t* __restrict__ out_hist, int n_bins,
    const int32_t* __restrict__ feat_group_start,
    const int32_t* __restrict__ bin_group_start,
    const int32_t* __restrict__ cut_ptrs,
    int64_t* __restrict__ node_sums) {
  const BlockTask task = tasks[blockIdx.x];
  if (task.row_begin >= task.row_end) return;
  const int group = blockIdx.y;
  const int f_begin = feat_group_start[group];
  const int gf = feat_group_start[group + 1] - f_begin;
  const int bin_begin = bin_group_start[group];
  const int group_bins = bin_group_start[group + 1] - bin_begin;

  unsigned meta[kGF];  // (group-rel start << 16) | width, both < 65536
#pragma unroll
  for (int f = 0; f < kGF; ++f) {
    if (f < gf) {
      const int c0 = cut_ptrs[f_begin + f];
      meta[f] = ((unsigned)(c0 - bin_begin) << 16)
                | (unsigned)(cut_ptrs[f_begin + f + 1] - c0);
    } else {
      meta[f] = 0;
    }
  }

  extern __shared__ unsigned long long smem[];
  for (int i = threadIdx.x; i < group_bins * 2; i += blockDim.x) {
    smem[i] = 0ULL;
  }
  __syncthreads();

  int64_t* hist_g = out_hist + (size_t)task.out_slot * n_bins * 2;
  const bool do_sums = node_sums != nullptr && group == 0;
  long long sum_g = 0, sum_h = 0;
  for (int i = task.row_begin + (int)threadIdx.x; i < task.row_end;
       i += blockDim.x) {
    const int row = ridx[i];
    const long long g = qgpair[2 * (size_t)row];
    const long long h = qgpair[2 * (size_t)row + 1];
    if (do_sums) {
      sum_g += g;
      sum_h += h;
    }
    const BinT* rowbins = gidx + (size_t)row * n_features + f_begin;
    if (kVec) {
      // dword-packed bin loads: 4 (u8) / 2 (u16) features per 32-bit
      // load (launcher guarantees rows stay 4-aligned)
      constexpr int kLanes = 4 / (int)sizeof(BinT);
      const unsigned* rowu = (const unsigned*)rowbins;
#pragma unroll
      for (int f4 = 0; f4 < kGF / kLanes; ++f4) {
        if (kLanes * f4 >= gf) break;
        const unsigned packed = rowu[f4];
#pragma unroll
        for (int j = 0; j < kLanes; ++j) {
          const int f = kLanes * f4 + j;
          const unsigned m = meta[f];
          const int local = (int)((packed >> (8 * (int)sizeof(BinT) * j))
                                  & (sizeof(BinT) == 1 ? 0xFFu : 0xFFFFu));
          if (local >= (int)(m & 0xFFFFu)) continue;
          const int sbin = (int)(m >> 16) + local;
          atomicAdd(&smem[2 * sbin], (unsigned long long)g);
          atomicAdd(&smem[2 * sbin + 1], (unsigned long long)h);
        }
      }
    } else {
#pragma unroll
      for (int f = 0; f < kGF; ++f) {
        if (f >= gf) break;
        const unsigned m = meta[f];
        const int local = (int)rowbins[f];
        if (local >= (int)(m & 0xFFFFu)) continue;
        const int sbin = (int)(m >> 16) + local;
        atomicAdd(&smem[2 * sbin], (unsigned long long)g);
        atomicAdd(&smem[2 * sbin + 1], (unsigned long long)h);
      }
    }
  }

  __syncthreads();
  for (int i = threadIdx.x; i < group_bins * 2; i += blockDim.x) {
    const unsigned long long v = smem[i];
    if (v != 0ULL) {
      atomicAdd((unsigned long long*)&hist_g[2 * bin_begin + i], v);
    }
  }
  if (do_sums) {
    for (int off = 32; off > 0; off >>= 1) {
      sum_g += __shfl_down(sum_g, off, 64);
      sum_h += __shfl_down(sum_h, off, 64);
    }
    __shared__ long long wg[GBT_HIST_BLOCK / 64];
    __shared__ long long wh[GBT_HIST_BLOCK / 64];
    const int lane = (int)threadIdx.x & 63;
    const int wave = (int)threadIdx.x >> 6;
    if (lane == 0) {
      wg[wave] = sum_g;
      wh[wave] = sum_h;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      long long tg = 0, th = 0;
      for (int w = 0; w < (int)blockDim.x / 64; ++w) {
        tg += wg[w];
        th += wh[w];
      }
      if (tg) atomicAdd((unsigned long long*)&node_sums[2 * task.out_slot],
                        (unsigned long long)tg);
      if (th) atomicAdd(
          (unsigned long long*)&node_sums[2 * task.out_slot + 1],
          (unsigned long long)th);
    }
  }
}

extern "C" void gbt_hist(const uint8_t* gidx8, const uint16_t* gidx16,
                         int n_features, const int32_t* qgpair,
                         const int32_t* ridx, const BlockTask* tasks,
                         int n_tasks, int64_t* out_hist, int n_bins,
                         const int32_t* feat_group_start,
                         const int32_t* bin_group_start, int n_groups,
                         int max_group_bins, const int32_t* cut_ptrs,
                         int use_shared, int64_t* node_sums,
                         hipStream_t stream) {
  static int block_env = [] {
    const char* e = getenv("GBT_HIST_BLOCK_SIZE");
    int v = e ? atoi(e) : 0;
    return (v >= 64 && v <= 1024) ? (v & ~63) : 0;
  }();
  const int block_size = block_env ? block_env : GBT_HIST_BLOCK;
  // round-2 sweep: the register-metadata kernel peaks at 512 threads
  // (972 vs 918 rounds/s at 1024 — fewer waves contending the LDS
  // atomic pipe, still enough to hide ds latency)
  const int reg_block = block_env ? block_env : 512;
  dim3 grid(n_tasks, n_groups);
  dim3 block(block_size);
  size_t shmem = use_shared ? (size_t)max_group_bins * 2 * sizeof(int64_t) : 0;
  // register-metadata fast path (HistKernelReg): selected by the
  // CALLER through use_shared (it owns the host-side group geometry):
  //   0 = global-atomic fallback, 1 = generic LDS kernel,
  //   2 = register-metadata (every group <= 32 features, u8),
  //   3 = 2 + dword-packed bin loads (stride/group 4-aligned).
  // Env GBT_HIST_REG=0 forces the generic kernel for A/B.
  static int use_reg = [] {
    const char* e = getenv("GBT_HIST_REG");
    return e ? atoi(e) : 1;
  }();
  if (use_reg && use_shared >= 2) {
    dim3 rgrid(n_tasks, n_groups);
    dim3 rblock(reg_block);
    if (gidx8 != nullptr) {
      if (use_shared >= 3) {
        hipLaunchKernelGGL((HistKernelReg<uint8_t, 32, true>), rgrid,
                           rblock, shmem, stream, gidx8, n_features, qgpair,
                           ridx, tasks, out_hist, n_bins, feat_group_start,
                           bin_group_start, cut_ptrs, node_sums);
      } else {
        hipLaunchKernelGGL((HistKernelReg<uint8_t, 32, false>), rgrid,
                           rblock, shmem, stream, gidx8, n_features, qgpair,
                           ridx, tasks, out_hist, n_bins, feat_group_start,
                           bin_group_start, cut_ptrs, node_sums);
      }
    } else {
      if (use_shared >= 3) {
        hipLaunchKernelGGL((HistKernelReg<uint16_t, 32, true>), rgrid,
                           rblock, shmem, stream, gidx16, n_features,
                           qgpair, ridx, tasks, out_hist, n_bins,
                           feat_group_start, bin_group_start, cut_ptrs,
                           node_sums);
      } else {
        hipLaunchKernelGGL((HistKernelReg<uint16_t, 32, false>), rgrid,
                           rblock, shmem, stream, gidx16, n_features,
                           qgpair, ridx, tasks, out_hist, n_bins,
                           feat_group_start, bin_group_start, cut_ptrs,
                           node_sums);
      }
    }
    return;
  }
  if (gidx8 != nullptr) {
    if (use_shared) {
      hipLaunchKernelGGL((HistKernel<uint8_t, true>), grid, block, shmem,
                         stream, gidx8, n_features, qgpair, ridx, tasks,
                         out_hist, n_bins, feat_group_start, bin_group_start,
                         cut_ptrs, node_sums);
    } else {
      hipLaunchKernelGGL((HistKernel<uint8_t, false>), grid, block, 0, stream,
                         gidx8, n_features, qgpair, ridx, tasks, out_hist,
                         n_bins, feat_group_start, bin_group_start, cut_ptrs,
                         node_sums);
    }
  } else {
    if (use_shared) {
      hipLaunchKernelGGL((HistKernel<uint16_t, true>), grid, block, shmem,
                         stream, gidx16, n_features, qgpair, ridx, tasks,
                         out_hist, n_bins, feat_group_start, bin_group_start,
                         cut_ptrs, node_sums);
    } else {
      hipLaunchKernelGGL((HistKernel<uint16_t, false>), grid, block, 0, stream,
                         gidx16, n_features, qgpair, ridx, tasks, out_hist,
                         n_bins, feat_group_start, bin_group_start, cut_ptrs,
                         node_sums);
    }
  }
}

// Fused multi-target histogram (reference MtHistKernel,
// src/tree/gpu_hist/histogram.cu:256): ONE pass over the rows
// accumulates every target's (g, h) — the python fallback launches the
// single-target kernel T times and re-reads the bin matrix per target.
// LDS layout [group_bins][T][2] u64; the caller regroups features so
// T * group_bins * 16 B fits the LDS budget.  Output is target-major:
// out[slot][T][n_bins][2] (the MT grower's stacked-tensor layout).
#define GBT_HIST_MT_MAX_T 8

template <typename BinT>
__global__ __launch_bounds__(512) void HistMtKernel(
    const BinT* __restrict__ gidx, int n_features,
    const int32_t* __restrict__ qg /* [n][T][2] */, int T,
    const int32_t* __restrict__ ridx, const BlockTask* __restrict__ tasks,
    int64_t* __restrict__ out_hist /* [T][n_slots][n_bins][2] */,
    int n_bins, int n_slots,
    const int32_t* __restrict__ feat_group_start,
    const int32_t* __restrict__ bin_group_start,
    const int32_t* __restrict__ cut_ptrs) {
  const BlockTask task = tasks[blockIdx.x];
  if (task.row_begin >= task.row_end) return;
  const int group = blockIdx.y;
  const int f_begin = feat_group_start[group];
  const int f_end = feat_group_start[group + 1];
  const int bin_begin = bin_group_start[group];
  const int group_bins = bin_group_start[group + 1] - bin_begin;
  const int gf = f_end - f_begin;

  __shared__ int s_start[GBT_HIST_MAX_F];
  __shared__ int s_width[GBT_HIST_MAX_F];
  for (int f = threadIdx.x; f < gf; f += blockDim.x) {
    const int c0 = cut_ptrs[f_begin + f];
    s_start[f] = c0 - bin_begin;
    s_width[f] = cut_ptrs[f_begin + f + 1] - c0;
  }
  extern __shared__ unsigned long long smem[];  // [group_bins][T][2]
  const int total = group_bins * T * 2;
  for (int i = threadIdx.x; i < total; i += blockDim.x) smem[i] = 0ULL;
  __syncthreads();

  for (int i = task.row_begin + (int)threadIdx.x; i < task.row_end;
       i += blockDim.x) {
    const int row = ridx[i];
    long long gv[GBT_HIST_MT_MAX_T], hv[GBT_HIST_MT_MAX_T];
    const int32_t* qp = qg + (size_t)row * T * 2;
    for (int t = 0; t < T; ++t) {
      gv[t] = qp[2 * t];
      hv[t] = qp[2 * t + 1];
    }
    const BinT* rowbins = gidx + (size_t)row * n_features + f_begin;
    for (int f = 0; f < gf; ++f) {
      const int local = (int)rowbins[f];
      if (local >= s_width[f]) continue;
      unsigned long long* cell =
          smem + ((size_t)(s_start[f] + local) * T) * 2;
      for (int t = 0; t < T; ++t) {
        atomicAdd(&cell[2 * t], (unsigned long long)gv[t]);
        atomicAdd(&cell[2 * t + 1], (unsigned long long)hv[t]);
      }
    }
  }
  __syncthreads();
  // flush [bin][t][c] -> out[slot][t][bin_begin+bin][c]
  for (int i = threadIdx.x; i < total; i += blockDim.x) {
    const unsigned long long v = smem[i];
    if (v == 0ULL) continue;
    const int c = i & 1;
    const int t = (i >> 1) % T;
    const int bin = (i >> 1) / T;
    atomicAdd((unsigned long long*)&out_hist[
                  (((size_t)t * n_slots + task.out_slot) * n_bins
                   + bin_begin + bin) * 2 + c], v);
  }
}

extern "C" void gbt_hist_mt(const uint8_t* gidx8, const uint16_t* gidx16,
                            int n_features, const int32_t* qg, int T,
                            const int32_t* ridx, const BlockTask* tasks,
                            int n_tasks, int64_t* out_hist, int n_bins,
                            int n_slots,
                            const int32_t* feat_group_start,
                            const int32_t* bin_group_start, int n_groups,
                            int max_group_bins, const int32_t* cut_ptrs,
                            hipStream_t stream) {
  dim3 grid(n_tasks, n_groups);
  dim3 block(512);
  size_t shmem = (size_t)max_group_bins * T * 2 * sizeof(int64_t);
  if (gidx8 != nullptr) {
    hipLaunchKernelGGL((HistMtKernel<uint8_t>), grid, block, shmem, stream,
                       gidx8, n_features, qg, T, ridx, tasks, out_hist,
                       n_bins, n_slots, feat_group_start, bin_group_start,
                       cut_ptrs);
  } else {
    hipLaunchKernelGGL((HistMtKernel<uint16_t>), grid, block, shmem, stream,
                       gidx16, n_features, qg, T, ridx, tasks, out_hist,
                       n_bins, n_slots, feat_group_start, bin_group_start,
                       cut_ptrs);
  }
}
