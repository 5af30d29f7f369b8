// Native tree-grow driver: runs the entire per-tree level loop in C++
// so the only Python work per tree is gradient computation and cache
// update.  (Reference analog: GPUHistMakerDevice::UpdateTree,
// src/tree/updater_gpu_hist.cu:588 — the reference's driver is C++
// too; grower.py's Python driver remains for the feature-rich paths:
// categorical splits, column sampling, interaction constraints,
// lossguide growth, external memory.)
//
// Depthwise growth, numeric splits, optional monotone constraints.
// Produces bit-identical trees to the Python GPU driver: same kernels,
// same double-precision host math, compiled with -ffp-contract=off.
#include "gbt_kernels.h"

#include <hip/hip_runtime.h>

#include <algorithm>
#include <cmath>
#include <cstdlib>
#include <cstring>
#include <vector>

namespace {

#define HIP_CHECK(x)                                    \
  do {                                                  \
    hipError_t err__ = (x);                             \
    if (err__ != hipSuccess) return -(int)err__;        \
  } while (0)

struct PinnedRing {
  static constexpr int kSlots = 6;
  void* host[kSlots] = {};
  void* dev[kSlots] = {};
  size_t cap[kSlots] = {};
  int cur = 0;

  int ensure(int slot, size_t bytes) {
    if (cap[slot] >= bytes) return 0;
    size_t want = 4096;
    while (want < bytes) want <<= 1;
    if (host[slot]) hipHostFree(host[slot]);
    if (dev[slot]) hipFree(dev[slot]);
    HIP_CHECK(hipHostMalloc(&host[slot], want));
    HIP_CHECK(hipMalloc(&dev[slot], want));
    cap[slot] = want;
    return 0;
  }
  int next() {
    int s = cur;
    cur = (cur + 1) % kSlots;
    return s;
  }
};

struct DriverCtx {
  PinnedRing ring;
  void* readback_host = nullptr;
  size_t readback_cap = 0;

  int ensure_readback(size_t bytes) {
    if (readback_cap >= bytes) return 0;
    size_t want = 4096;
    while (want < bytes) want <<= 1;
    if (readback_host) hipHostFree(readback_host);
    HIP_CHECK(hipHostMalloc(&readback_host, want));
    readback_cap = want;
    return 0;
  }
  ~DriverCtx() {
    for (int i = 0; i < PinnedRing::kSlots; ++i) {
      if (ring.host[i]) hipHostFree(ring.host[i]);
      if (ring.dev[i]) hipFree(ring.dev[i]);
    }
    if (readback_host) hipHostFree(readback_host);
  }
};

struct Node {
  int nid;
  int seg_begin, seg_end;
  long long gq, hq;
  int hist_slot;
  double lo, hi;       // monotone weight bounds
  double gain;         // best-split candidate
  int bin, dir, feature;
  long long lgq, lhq;
};

double ThresholdL1(double g, double alpha) {
  if (alpha == 0.0) return g;
  double s = (g > 0.0) ? 1.0 : ((g < 0.0) ? -1.0 : 0.0);
  double m = std::fabs(g) - alpha;
  if (m < 0.0) m = 0.0;
  return s * m;
}

struct HostParams {
  double lam, alpha, mds;
};

double CalcWeight(double g, double h, const HostParams& p) {
  double w = -ThresholdL1(g, p.alpha) / (h + p.lam);
  if (p.mds > 0.0) w = std::fmin(std::fmax(w, -p.mds), p.mds);
  return w;
}

void ChunkTasks(const std::vector<Node*>& nodes, std::vector<BlockTask>* out,
                long long min_rows = 1024, long long target_tasks = 2048) {
  out->clear();
  long long total = 0;
  for (auto* n : nodes) total += n->seg_end - n->seg_begin;
  long long rows_per_task =
      std::max<long long>(min_rows, (total + target_tasks - 1) / target_tasks);
  for (size_t i = 0; i < nodes.size(); ++i) {
    int b = nodes[i]->seg_begin;
    while (b < nodes[i]->seg_end) {
      int e = (int)std::min<long long>(b + rows_per_task, nodes[i]->seg_end);
      out->push_back(BlockTask{(int)i, b, e, 0});
      b = e;
    }
  }
  if (out->empty()) out->push_back(BlockTask{0, 0, 0, 0});
}

__global__ void IotaKernel(int32_t* r, long long n) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n; i += (long long)gridDim.x * blockDim.x) r[i] = (int)i;
}

__global__ void SubtractHistKernel(const int64_t* __restrict__ parents,
                                   const int64_t* __restrict__ built,
                                   int64_t* __restrict__ out,
                                   const int32_t* __restrict__ parent_slot,
                                   int n_bins2, int k) {
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long total = (long long)k * n_bins2;
  for (long long i = idx; i < total; i += (long long)gridDim.x * blockDim.x) {
    const int row = (int)(i / n_bins2);
    const int col = (int)(i % n_bins2);
    out[(long long)row * n_bins2 + col] =
        parents[(long long)parent_slot[row] * n_bins2 + col] -
        built[(long long)row * n_bins2 + col];
  }
}

struct LeafSeg {
  int nid, begin, end;
};

}  // namespace

extern "C" {

void* gbt_driver_create() { return new DriverCtx(); }
void gbt_driver_destroy(void* ctx) { delete (DriverCtx*)ctx; }

typedef void (*AllreduceFn)(long long* dev_ptr, long long n_elems);

// Returns n_nodes (>0) on success, negative on failure.
int gbt_grow_tree(
    void* vctx, const uint8_t* gidx8, const uint16_t* gidx16, int n_features,
    long long n_rows, const int32_t* qgpair, const int32_t* cut_ptrs_dev,
    const float* cut_values_host, const int32_t* cut_ptrs_host,
    const int32_t* n_bins_feat_dev, const int32_t* feat_group_start_dev,
    const int32_t* bin_group_start_dev, int n_groups, int max_group_bins,
    int use_shared, int n_bins,
    // device workspace (torch-allocated)
    int32_t* ridx, int32_t* ridx_out, int64_t* hist_pool_a,
    int64_t* hist_pool_b, double* eval_gain, int32_t* eval_bin,
    uint8_t* eval_dir, int64_t* eval_lsum, int64_t* eval_best,
    int32_t* pos_out, int max_nodes_level,
    // scalars
    double g_scale, double h_scale, long long root_gq, long long root_hq,
    double reg_lambda, double reg_alpha, double max_delta_step,
    double min_child_weight, double gamma, double eta, int max_depth,
    const int8_t* monotone_dev, const int8_t* monotone_host,
    AllreduceFn allreduce,
    // host tree outputs (caller-sized to 2^(max_depth+1))
    int32_t* out_left, int32_t* out_right, int32_t* out_parent,
    int32_t* out_split_index, float* out_split_cond,
    uint8_t* out_default_left, float* out_loss_chg, float* out_sum_hess,
    float* out_base_weight, void* stream_v) {
  DriverCtx* ctx = (DriverCtx*)vctx;
  hipStream_t stream = (hipStream_t)stream_v;
  HostParams p{reg_lambda, reg_alpha, max_delta_step};
  const bool has_mono = monotone_host != nullptr;
  const double inv_g = 1.0 / g_scale, inv_h = 1.0 / h_scale;
  const long long hist_row = (long long)n_bins * 2;

  {
    int blocks = (int)std::min<long long>((n_rows + 255) / 256, 4096);
    hipLaunchKernelGGL(IotaKernel, dim3(blocks), dim3(256), 0, stream, ridx,
                       n_rows);
  }

  int n_tree_nodes = 1;
  out_left[0] = -1;
  out_right[0] = -1;
  out_parent[0] = -1;

  Node root{};
  root.nid = 0;
  root.seg_begin = 0;
  root.seg_end = (int)n_rows;
  root.gq = root_gq;
  root.hq = root_hq;
  root.hist_slot = 0;
  root.lo = -INFINITY;
  root.hi = INFINITY;
  out_base_weight[0] = (float)CalcWeight(root_gq * inv_g, root_hq * inv_h, p);
  out_sum_hess[0] = (float)(root_hq * inv_h);

  // hist wants FEWER, BIGGER tasks than partition: every block pays a
  // full LDS zero+flush of the group histogram, so rows/task must
  // amortize ~2*group_bins atomics (env GBT_HIST_TASKS to tune)
  static long long hist_tasks = [] {
    const char* e = getenv("GBT_HIST_TASKS");
    long long v = e ? atoll(e) : 512;
    return v >= 64 && v <= 16384 ? v : 512;
  }();
  auto build_hists = [&](std::vector<Node*>& nodes, int64_t* pool) -> int {
    std::vector<BlockTask> tasks;
    ChunkTasks(nodes, &tasks, 2048, hist_tasks);
    const int slot = ctx->ring.next();
    size_t bytes = tasks.size() * sizeof(BlockTask);
    if (int e = ctx->ring.ensure(slot, bytes)) return e;
    memcpy(ctx->ring.host[slot], tasks.data(), bytes);
    HIP_CHECK(hipMemcpyAsync(ctx->ring.dev[slot], ctx->ring.host[slot], bytes,
                             hipMemcpyHostToDevice, stream));
    HIP_CHECK(hipMemsetAsync(pool, 0,
                             nodes.size() * hist_row * sizeof(int64_t),
                             stream));
    gbt_hist(gidx8, gidx16, n_features, qgpair, ridx,
             (const BlockTask*)ctx->ring.dev[slot], (int)tasks.size(), pool,
             n_bins, feat_group_start_dev, bin_group_start_dev, n_groups,
             max_group_bins, cut_ptrs_dev, use_shared, stream);
    if (allreduce) {
      allreduce((long long*)pool, (long long)nodes.size() * hist_row);
    }
    return 0;
  };

  auto evaluate = [&](std::vector<Node*>& nodes, const int64_t* hists) -> int {
    const int k = (int)nodes.size();
    const int slot = ctx->ring.next();
    size_t off_ps = 0;
    size_t off_bd = ((size_t)k * 2 * sizeof(int64_t) + 7) & ~7ULL;
    size_t bytes = off_bd + (has_mono ? (size_t)k * 2 * sizeof(double) : 0);
    if (int e = ctx->ring.ensure(slot, bytes)) return e;
    char* h = (char*)ctx->ring.host[slot];
    int64_t* ps = (int64_t*)(h + off_ps);
    for (int i = 0; i < k; ++i) {
      ps[2 * i] = nodes[i]->gq;
      ps[2 * i + 1] = nodes[i]->hq;
    }
    if (has_mono) {
      double* bd = (double*)(h + off_bd);
      for (int i = 0; i < k; ++i) {
        bd[2 * i] = nodes[i]->lo;
        bd[2 * i + 1] = nodes[i]->hi;
      }
    }
    HIP_CHECK(hipMemcpyAsync(ctx->ring.dev[slot], h, bytes,
                             hipMemcpyHostToDevice, stream));
    char* d = (char*)ctx->ring.dev[slot];
    gbt_evaluate(hists, k, n_bins, n_features, cut_ptrs_dev,
                 (const int64_t*)(d + off_ps), g_scale, h_scale, reg_lambda,
                 reg_alpha, max_delta_step, min_child_weight, monotone_dev,
                 has_mono ? (const double*)(d + off_bd) : nullptr, nullptr,
                 nullptr, eval_gain, eval_bin, eval_dir, eval_lsum, stream);
    gbt_select_best(eval_gain, eval_bin, eval_dir, eval_lsum, k, n_features,
                    eval_best, stream);
    size_t rb = (size_t)k * 6 * sizeof(int64_t);
    if (int e = ctx->ensure_readback(rb)) return e;
    HIP_CHECK(hipMemcpyAsync(ctx->readback_host, eval_best, rb,
                             hipMemcpyDeviceToHost, stream));
    HIP_CHECK(hipStreamSynchronize(stream));
    const int64_t* best = (const int64_t*)ctx->readback_host;
    for (int i = 0; i < k; ++i) {
      Node* nd = nodes[i];
      double gain;
      memcpy(&gain, &best[6 * i], sizeof(double));
      nd->bin = (int)best[6 * i + 1];
      nd->dir = (int)best[6 * i + 2];
      nd->lgq = best[6 * i + 3];
      nd->lhq = best[6 * i + 4];
      nd->feature = (int)best[6 * i + 5];
      nd->gain = (nd->bin >= 0 && std::isfinite(gain)) ? gain : -INFINITY;
    }
    return 0;
  };

  auto partition = [&](std::vector<Node*>& nodes,
                       std::vector<int>* left_counts) -> int {
    const int k = (int)nodes.size();
    std::vector<BlockTask> tasks;
    ChunkTasks(nodes, &tasks);
    const int slot = ctx->ring.next();
    size_t off_tasks = 0;
    size_t off_feat = (tasks.size() * sizeof(BlockTask) + 7) & ~7ULL;
    size_t off_sbin = (off_feat + (size_t)k * 4 + 7) & ~7ULL;
    size_t off_dl = (off_sbin + (size_t)k * 4 + 7) & ~7ULL;
    size_t off_cnt = (off_dl + (size_t)k + 7) & ~7ULL;
    size_t bytes = off_cnt + (size_t)k * 2 * sizeof(int32_t);
    if (int e = ctx->ring.ensure(slot, bytes)) return e;
    char* h = (char*)ctx->ring.host[slot];
    memcpy(h + off_tasks, tasks.data(), tasks.size() * sizeof(BlockTask));
    int32_t* feat = (int32_t*)(h + off_feat);
    int32_t* sbin = (int32_t*)(h + off_sbin);
    uint8_t* dl = (uint8_t*)(h + off_dl);
    int32_t* cnt = (int32_t*)(h + off_cnt);
    for (int i = 0; i < k; ++i) {
      Node* nd = nodes[i];
      feat[i] = nd->feature;
      sbin[i] = nd->bin - cut_ptrs_host[nd->feature];
      dl[i] = (uint8_t)nd->dir;
      cnt[2 * i] = nd->seg_begin;
      cnt[2 * i + 1] = nd->seg_end;
    }
    HIP_CHECK(hipMemcpyAsync(ctx->ring.dev[slot], h, bytes,
                             hipMemcpyHostToDevice, stream));
    char* d = (char*)ctx->ring.dev[slot];
    gbt_partition(gidx8, gidx16, n_features, ridx, ridx_out,
                  (const BlockTask*)(d + off_tasks), (int)tasks.size(),
                  (const int32_t*)(d + off_feat),
                  (const int32_t*)(d + off_sbin), (const uint8_t*)(d + off_dl),
                  nullptr, nullptr, n_bins_feat_dev, (int32_t*)(d + off_cnt),
                  stream);
    gbt_copy_ranges(ridx_out, ridx, (const BlockTask*)(d + off_tasks),
                    (int)tasks.size(), stream);
    size_t rb = (size_t)k * 2 * sizeof(int32_t);
    if (int e = ctx->ensure_readback(rb)) return e;
    HIP_CHECK(hipMemcpyAsync(ctx->readback_host, d + off_cnt, rb,
                             hipMemcpyDeviceToHost, stream));
    HIP_CHECK(hipStreamSynchronize(stream));
    const int32_t* fin = (const int32_t*)ctx->readback_host;
    left_counts->resize(k);
    for (int i = 0; i < k; ++i) {
      (*left_counts)[i] = fin[2 * i] - nodes[i]->seg_begin;
    }
    return 0;
  };

  // ---- root ----
  std::vector<Node*> frontier{&root};
  if (int e = build_hists(frontier, hist_pool_a)) return e;
  if (int e = evaluate(frontier, hist_pool_a)) return e;

  std::vector<Node> level_nodes{root};
  std::vector<Node> next_level;
  std::vector<LeafSeg> leaves;
  int64_t* cur_pool = hist_pool_a;
  int64_t* next_pool = hist_pool_b;

  for (int depth = 0; depth < max_depth && !level_nodes.empty(); ++depth) {
    std::vector<Node*> expand;
    for (auto& nd : level_nodes) {
      if (nd.gain > gamma && std::isfinite(nd.gain)) {
        expand.push_back(&nd);
      } else {
        leaves.push_back({nd.nid, nd.seg_begin, nd.seg_end});
      }
    }
    if (expand.empty()) {
      level_nodes.clear();
      break;
    }
    // apply splits on host
    next_level.clear();
    next_level.reserve(2 * expand.size());
    for (Node* nd : expand) {
      const int l = n_tree_nodes, r = n_tree_nodes + 1;
      n_tree_nodes += 2;
      out_left[nd->nid] = l;
      out_right[nd->nid] = r;
      out_left[l] = out_right[l] = -1;
      out_left[r] = out_right[r] = -1;
      out_parent[l] = nd->nid;
      out_parent[r] = nd->nid;
      out_split_index[nd->nid] = nd->feature;
      out_split_cond[nd->nid] = cut_values_host[nd->bin];
      out_default_left[nd->nid] = (uint8_t)nd->dir;
      out_loss_chg[nd->nid] = (float)nd->gain;
      const long long rgq = nd->gq - nd->lgq, rhq = nd->hq - nd->lhq;
      double wl = CalcWeight(nd->lgq * inv_g, nd->lhq * inv_h, p);
      double wr = CalcWeight(rgq * inv_g, rhq * inv_h, p);
      wl = std::fmin(std::fmax(wl, nd->lo), nd->hi);
      wr = std::fmin(std::fmax(wr, nd->lo), nd->hi);
      out_sum_hess[nd->nid] = (float)((nd->lhq + rhq) * inv_h);
      out_base_weight[l] = (float)wl;
      out_base_weight[r] = (float)wr;
      out_sum_hess[l] = (float)(nd->lhq * inv_h);
      out_sum_hess[r] = (float)(rhq * inv_h);
      Node ln{}, rn{};
      ln.nid = l;
      rn.nid = r;
      ln.gq = nd->lgq;
      ln.hq = nd->lhq;
      rn.gq = rgq;
      rn.hq = rhq;
      ln.lo = rn.lo = nd->lo;
      ln.hi = rn.hi = nd->hi;
      ln.gain = rn.gain = -INFINITY;
      ln.bin = rn.bin = -1;
      if (has_mono && nd->feature < n_features) {
        const int c = monotone_host[nd->feature];
        if (c != 0) {
          const double mid = (wl + wr) / 2.0;
          if (c > 0) {
            ln.hi = std::fmin(nd->hi, mid);
            rn.lo = std::fmax(nd->lo, mid);
          } else {
            ln.lo = std::fmax(nd->lo, mid);
            rn.hi = std::fmin(nd->hi, mid);
          }
        }
      }
      next_level.push_back(ln);
      next_level.push_back(rn);
    }
    std::vector<int> left_counts;
    if (int e = partition(expand, &left_counts)) return e;
    for (size_t i = 0; i < expand.size(); ++i) {
      Node& ln = next_level[2 * i];
      Node& rn = next_level[2 * i + 1];
      ln.seg_begin = expand[i]->seg_begin;
      ln.seg_end = expand[i]->seg_begin + left_counts[i];
      rn.seg_begin = ln.seg_end;
      rn.seg_end = expand[i]->seg_end;
    }
    if (depth + 1 >= max_depth) {
      for (auto& nd : next_level) {
        leaves.push_back({nd.nid, nd.seg_begin, nd.seg_end});
      }
      level_nodes.clear();
      break;
    }
    // build smaller sibling, subtract larger
    std::vector<Node*> build;
    std::vector<int32_t> parent_slots;
    std::vector<Node*> subtracted;
    for (size_t i = 0; i < expand.size(); ++i) {
      Node& ln = next_level[2 * i];
      Node& rn = next_level[2 * i + 1];
      Node* small = (ln.seg_end - ln.seg_begin <= rn.seg_end - rn.seg_begin)
                        ? &ln : &rn;
      Node* big = (small == &ln) ? &rn : &ln;
      small->hist_slot = (int)build.size();
      build.push_back(small);
      parent_slots.push_back(expand[i]->hist_slot);
      subtracted.push_back(big);
    }
    if (2 * (int)build.size() > 2 * max_nodes_level) return -9999;
    if (int e = build_hists(build, next_pool)) return e;
    {
      const int k = (int)build.size();
      const int slot = ctx->ring.next();
      size_t bytes = (size_t)k * sizeof(int32_t);
      if (int e = ctx->ring.ensure(slot, bytes)) return e;
      memcpy(ctx->ring.host[slot], parent_slots.data(), bytes);
      HIP_CHECK(hipMemcpyAsync(ctx->ring.dev[slot], ctx->ring.host[slot],
                               bytes, hipMemcpyHostToDevice, stream));
      int64_t* sub_out = next_pool + (long long)k * hist_row;
      const long long total = (long long)k * hist_row;
      int blocks = (int)std::min<long long>((total + 255) / 256, 4096);
      hipLaunchKernelGGL(SubtractHistKernel, dim3(blocks), dim3(256), 0,
                         stream, cur_pool, next_pool, sub_out,
                         (const int32_t*)ctx->ring.dev[slot], (int)hist_row,
                         k);
      for (int i = 0; i < k; ++i) subtracted[i]->hist_slot = k + i;
    }
    std::vector<Node*> eval_nodes;
    eval_nodes.reserve(2 * build.size());
    for (Node* b : build) eval_nodes.push_back(b);
    for (Node* s : subtracted) eval_nodes.push_back(s);
    if (int e = evaluate(eval_nodes, next_pool)) return e;
    level_nodes.swap(next_level);
    std::swap(cur_pool, next_pool);
  }
  for (auto& nd : level_nodes) {
    leaves.push_back({nd.nid, nd.seg_begin, nd.seg_end});
  }

  // leaf values
  for (int nid = 0; nid < n_tree_nodes; ++nid) {
    if (out_left[nid] == -1) {
      out_split_cond[nid] = (float)(out_base_weight[nid] * eta);
    }
  }
  // leaf positions
  {
    std::vector<BlockTask> tasks;
    std::vector<Node> lnodes(leaves.size());
    std::vector<Node*> lptrs(leaves.size());
    for (size_t i = 0; i < leaves.size(); ++i) {
      lnodes[i].seg_begin = leaves[i].begin;
      lnodes[i].seg_end = leaves[i].end;
      lptrs[i] = &lnodes[i];
    }
    ChunkTasks(lptrs, &tasks);
    const int slot = ctx->ring.next();
    size_t off_tasks = 0;
    size_t off_ids = (tasks.size() * sizeof(BlockTask) + 7) & ~7ULL;
    size_t bytes = off_ids + leaves.size() * sizeof(int32_t);
    if (int e = ctx->ring.ensure(slot, bytes)) return e;
    char* h = (char*)ctx->ring.host[slot];
    memcpy(h + off_tasks, tasks.data(), tasks.size() * sizeof(BlockTask));
    int32_t* ids = (int32_t*)(h + off_ids);
    for (size_t i = 0; i < leaves.size(); ++i) ids[i] = leaves[i].nid;
    HIP_CHECK(hipMemcpyAsync(ctx->ring.dev[slot], h, bytes,
                             hipMemcpyHostToDevice, stream));
    char* d = (char*)ctx->ring.dev[slot];
    gbt_leaf_partition(ridx, (const BlockTask*)(d + off_tasks),
                       (int)tasks.size(), (const int32_t*)(d + off_ids),
                       pos_out, stream);
  }
  return n_tree_nodes;
}

}  // extern "C"
