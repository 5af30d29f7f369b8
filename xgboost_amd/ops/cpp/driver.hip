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
//
// ONE host sync per level: the whole phase chain
//   partition -> child-hist task generation (on device, from the
//   partition counters) -> hist build -> sibling subtraction ->
//   split evaluation
// is enqueued in a single burst, then ONE hipStreamSynchronize reads
// back the per-node best splits AND the partition counters together.
// The build-vs-subtract sibling choice uses the hessian sums known
// from the parent's evaluation (exact row counts are not yet on the
// host at enqueue time); histogram subtraction is exact int64, so the
// choice affects only performance, never the resulting tree.
#include "gbt_kernels.h"

#include <hip/hip_runtime.h>

#include <algorithm>
#include <cmath>
#include <cstdio>
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

  // hipGraph replay of the single-GPU whole-tree chain: the enqueue
  // sequence is identical every round (no host decisions mid-chain), so
  // after one capture the ~25 launches/tree replay as ONE graph launch.
  // Valid only while every baked address is unchanged — wt_key records
  // them all and any mismatch forces a recapture.  The chain's host
  // staging lives in a DEDICATED pinned buffer (not the rotating ring)
  // so the baked source addresses stay stable; its contents are
  // rewritten before every launch (replays read them at execution).
  hipGraphExec_t wt_graph = nullptr;
  std::vector<unsigned long long> wt_key;
  // capture-on-second-sight: a key must repeat once before we pay the
  // capture+instantiate cost (fresh per-round contexts — e.g. the gpu
  // approx path rebuilds its ops every iteration — would otherwise
  // capture every round and never replay)
  std::vector<unsigned long long> wt_seen_key;
  void* wtc_host = nullptr;
  void* wtc_dev = nullptr;
  size_t wtc_cap = 0;
  // capture needs a REAL stream: the caller usually passes torch's
  // default stream (0), and capturing a null-stream alias leaves the
  // legacy stream wedged in capture state.  The chain runs on this
  // dedicated stream, ordered behind the caller stream by gevent.
  hipStream_t gstream = nullptr;
  hipEvent_t gevent = nullptr;

  int ensure_gstream() {
    if (gstream == nullptr)
      HIP_CHECK(hipStreamCreateWithFlags(&gstream, hipStreamNonBlocking));
    if (gevent == nullptr)
      HIP_CHECK(hipEventCreateWithFlags(&gevent, hipEventDisableTiming));
    return 0;
  }

  int ensure_readback(size_t bytes) {
    if (readback_cap >= bytes) return 0;
    size_t want = 4096;
    while (want < bytes) want <<= 1;
    if (readback_host) hipHostFree(readback_host);
    HIP_CHECK(hipHostMalloc(&readback_host, want));
    readback_cap = want;
    return 0;
  }
  int ensure_const(size_t bytes) {
    if (wtc_cap >= bytes) return 0;
    size_t want = 4096;
    while (want < bytes) want <<= 1;
    if (wtc_host) hipHostFree(wtc_host);
    if (wtc_dev) hipFree(wtc_dev);
    HIP_CHECK(hipHostMalloc(&wtc_host, want));
    HIP_CHECK(hipMalloc(&wtc_dev, want));
    wtc_cap = want;
    return 0;
  }
  ~DriverCtx() {
    for (int i = 0; i < PinnedRing::kSlots; ++i) {
      if (ring.host[i]) hipHostFree(ring.host[i]);
      if (ring.dev[i]) hipFree(ring.dev[i]);
    }
    if (readback_host) hipHostFree(readback_host);
    if (wt_graph) (void)hipGraphExecDestroy(wt_graph);
    if (wtc_host) hipHostFree(wtc_host);
    if (wtc_dev) hipFree(wtc_dev);
    if (gstream) (void)hipStreamDestroy(gstream);
    if (gevent) (void)hipEventDestroy(gevent);
  }
};

// hipGraph replay of the whole-tree chain (default on; GBT_WT_GRAPH=0
// disables).  Single-GPU only: the distributed chain interleaves RCCL
// enqueues through a host callback, which a capture cannot contain.
bool WtGraphEnabled() {
  static int v = [] {
    const char* e = getenv("GBT_WT_GRAPH");
    return e ? atoi(e) : 1;
  }();
  return v != 0;
}

bool WtGraphDebug() {
  static int v = [] {
    const char* e = getenv("GBT_WT_GRAPH_DEBUG");
    return e ? atoi(e) : 0;
  }();
  return v != 0;
}

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

// device mirror of the host weight math (same doubles, same op order,
// -ffp-contract=off ⇒ the whole-tree replay reproduces device bounds
// bit-exactly)
__device__ double DevCalcWeight(double g, double h, double lam, double alpha,
                                double mds) {
  double t = g;
  if (alpha != 0.0) {
    double s = (g > 0.0) ? 1.0 : ((g < 0.0) ? -1.0 : 0.0);
    double m = fabs(g) - alpha;
    if (m < 0.0) m = 0.0;
    t = s * m;
  }
  double w = -t / (h + lam);
  if (mds > 0.0) w = fmin(fmax(w, -mds), mds);
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

__global__ void ZeroI64Kernel(int64_t* p, long long n) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < n; i += stride) p[i] = 0;
}

__global__ void IotaKernel(int32_t* r, long long n) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n; i += (long long)gridDim.x * blockDim.x) r[i] = (int)i;
}

__global__ void SubtractHistKernel(const int64_t* __restrict__ parents,
                                   const int64_t* __restrict__ built,
                                   int64_t* __restrict__ out,
                                   const int32_t* __restrict__ parent_slot,
                                   int n_bins2, int k,
                                   int64_t* __restrict__ ps,
                                   const int64_t* __restrict__ parent_ps,
                                   const int32_t* __restrict__ kp_dev) {
  if (kp_dev != nullptr) {
    k = *kp_dev;
    out += (long long)k * n_bins2;  // subtracted slots follow the built
  }
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  // piggybacked: subtracted-sibling sums = parent - built (slots k..2k-1)
  if (ps != nullptr && idx < k) {
    ps[2 * (k + idx)] = parent_ps[2 * idx] - ps[2 * idx];
    ps[2 * (k + idx) + 1] = parent_ps[2 * idx + 1] - ps[2 * idx + 1];
  }
  const long long total = (long long)k * n_bins2;
  for (long long i = idx; i < total; i += (long long)gridDim.x * blockDim.x) {
    const int row = (int)(i / n_bins2);
    const int col = (int)(i % n_bins2);
    out[(long long)row * n_bins2 + col] =
        parents[(long long)parent_slot[row] * n_bins2 + col] -
        built[(long long)row * n_bins2 + col];
  }
}

// Generate the child-hist BlockTask array on device: the child
// segments come from the partition counters (still in flight on the
// stream when the host enqueues this), so the host never has to wait
// for the partition before launching the hist build.
//   desc[j] = {parent_begin, parent_end, counter_slot, is_left}
// Child j's segment: left  -> [parent_begin, counters[2*slot])
//                    right -> [counters[2*slot], parent_end)
// The task array is padded with empty tasks up to max_tasks (the
// hist/partition kernels early-return on them); scratch holds
// {begin, end, task_prefix} per child.
__global__ void HistTaskGenKernel(const int32_t* __restrict__ counters,
                                  const int32_t* __restrict__ desc, int k,
                                  long long min_rows, long long target_tasks,
                                  int max_tasks,
                                  int32_t* __restrict__ scratch,
                                  BlockTask* __restrict__ out_tasks,
                                  int64_t* __restrict__ ps /* [2k][2] to
                                      zero, or null */,
                                  const int32_t* __restrict__ kp_dev,
                                  int32_t* __restrict__ seg_out,
                                  /* null, or [2k][2] child segments in
                                     SLOT order: built j, subtracted
                                     k+j (whole-tree mode) */
                                  int zero_n, /* distributed whole-tree:
                                     zero ps up to this host-known
                                     worst-case bound so the padded
                                     fixed-count allreduce sums zeros,
                                     never stale ping-pong garbage */
                                  uint8_t* __restrict__ mode_out
                                  /* null, or [k]: resolved built-is-left
                                     flag per pair — the EXPLICIT record
                                     of which sibling was built (segment
                                     adjacency is ambiguous for nodes
                                     whose local shard is empty) */) {
  if (kp_dev != nullptr) k = *kp_dev;
  if (ps != nullptr) {
    const int zn = max(4 * k, zero_n);
    for (int i = threadIdx.x; i < zn; i += blockDim.x) ps[i] = 0;
  }
  // parallel load phase: thread-0 doing k dependent global loads costs
  // more than the whole rest of the kernel; gather child segments into
  // LDS cooperatively first (k is capped by the driver's level guard)
  __shared__ int s_begin[2048], s_end[2048];
  for (int j = threadIdx.x; j < k; j += blockDim.x) {
    const int pb = desc[4 * j], pe = desc[4 * j + 1];
    const int split = counters[2 * desc[4 * j + 2]];
    // mode: 1 = build left, 0 = build right, 2 = build the smaller
    // child (single-GPU only: local row counts are rank-dependent)
    int mode = desc[4 * j + 3];
    if (mode == 2) mode = (split - pb) <= (pe - split) ? 1 : 0;
    s_begin[j] = mode ? pb : split;
    s_end[j] = mode ? split : pe;
    if (mode_out != nullptr) mode_out[j] = (uint8_t)mode;
    if (seg_out != nullptr) {
      seg_out[2 * j] = s_begin[j];
      seg_out[2 * j + 1] = s_end[j];
      // the subtracted sibling is the complement range
      seg_out[2 * (k + j)] = mode ? split : pb;
      seg_out[2 * (k + j) + 1] = mode ? pe : split;
    }
  }
  __shared__ long long s_rpt;
  __shared__ int s_total_tasks;
  __syncthreads();
  if (threadIdx.x == 0) {
    long long total = 0;
    for (int j = 0; j < k; ++j) {
      scratch[3 * j] = s_begin[j];
      scratch[3 * j + 1] = s_end[j];
      total += s_end[j] - s_begin[j];
    }
    // round rows/task UP to a power of two: all the per-node
    // ceil-divides below become shifts (serial thread-0 64-bit
    // divisions measured ~2x the whole kernel's budget), and the
    // task layout has no effect on results (atomics are
    // order-independent)
    long long rpt0 =
        std::max(min_rows, (total + target_tasks - 1) / target_tasks);
    int shift = 0;
    while ((1LL << shift) < rpt0) ++shift;
    for (;;) {  // defensive: never overflow the task buffer
      long long need = 0;
      for (int j = 0; j < k; ++j) {
        const long long sz = s_end[j] - s_begin[j];
        need += (sz + (1LL << shift) - 1) >> shift;
      }
      if (need <= max_tasks) break;
      ++shift;
    }
    const long long rpt = 1LL << shift;
    int pref = 0;
    for (int j = 0; j < k; ++j) {
      scratch[3 * j + 2] = pref;
      const long long sz = s_end[j] - s_begin[j];
      pref += (int)((sz + rpt - 1) >> shift);
    }
    s_rpt = rpt;
    s_total_tasks = pref;
  }
  __syncthreads();
  const long long rpt = s_rpt;
  const int total_tasks = s_total_tasks;
  for (int t = threadIdx.x; t < max_tasks; t += blockDim.x) {
    if (t >= total_tasks) {
      out_tasks[t] = BlockTask{0, 0, 0, 0};
      continue;
    }
    // binary search: largest j with prefix[j] <= t
    int lo = 0, hi = k - 1;
    while (lo < hi) {
      const int mid = (lo + hi + 1) >> 1;
      if (scratch[3 * mid + 2] <= t) lo = mid; else hi = mid - 1;
    }
    const int j = lo;
    const long long idx = t - scratch[3 * j + 2];
    const int b = s_begin[j] + (int)(idx * rpt);
    const int e = (int)std::min<long long>((long long)b + rpt,
                                           (long long)s_end[j]);
    out_tasks[t] = BlockTask{j, b, e, 0};
  }
}

// Whole-tree mode: apply one level's split decisions entirely on
// device.  Reads the level's best-split records and segments, compacts
// the expanding nodes IN PAIR ORDER (left0, right0, left1, ...) so the
// host replay and the python driver number children identically, and
// emits everything the next phase chain needs: partition args + padded
// task list, counters, task-gen descriptors, and the gathered parent
// pair-sums for the sibling-subtraction derivation.
__global__ __launch_bounds__(256) void ApplyKernel(
    const int64_t* __restrict__ best,    // [kn][6] this level's nodes
    const int32_t* __restrict__ segs,    // [kn][2] slot order
    const int32_t* __restrict__ kn_dev,  // node count of this level
    const int32_t* __restrict__ kp_prev_dev,  // pair count (0 at root)
    const int64_t* __restrict__ ps_prev,      // [kn][2] node pair sums
    double gamma_, const int32_t* __restrict__ cut_ptrs,
    long long min_rows, long long target_tasks, int max_ptasks,
    int32_t* __restrict__ kn_out, int32_t* __restrict__ kp_out,
    int32_t* __restrict__ feat, int32_t* __restrict__ sbin,
    uint8_t* __restrict__ dl, int32_t* __restrict__ cnt,
    BlockTask* __restrict__ out_tasks, int32_t* __restrict__ desc,
    int64_t* __restrict__ parent_ps, int32_t* __restrict__ parent_slot,
    // distributed / monotone extensions (all null / 0 otherwise):
    int choice_global,  // 1: build the smaller-GLOBAL-hessian child so
                        // every rank builds (and allreduces) the SAME
                        // sibling slot; 0: HistTaskGenKernel picks the
                        // smaller LOCAL-row child (single-GPU only)
    const float* __restrict__ maxabs,     // [2] scale derivation (mono)
    const int8_t* __restrict__ monotone,  // [F] or null
    double lam, double alpha, double mds,
    const double* __restrict__ bounds_prev,  // [kn][2] slot order | null
    double* __restrict__ bounds_out,         // [2kb][2] slot order | null
    const uint8_t* __restrict__ mode_prev) { // [kp_prev] built-is-left
                                             // flags of the PREVIOUS
                                             // level's pairs, or null
                                             // (root / legacy adjacency)
  __shared__ int s_order[2048];   // slots in pair order
  __shared__ uint8_t s_flag[2048];
  __shared__ int s_exp[2048];     // expanding slots (pair order)
  __shared__ int s_b[2048], s_e[2048];
  __shared__ int s_kb;
  const int kn = *kn_dev;
  const int kp_prev = *kp_prev_dev;
  // pair-order slot sequence: pair p = slots (p, kp_prev + p); left is
  // the one whose end == the other's begin
  if (kp_prev == 0) {
    if (threadIdx.x == 0) s_order[0] = 0;  // root
  } else {
    for (int p = threadIdx.x; p < kp_prev; p += blockDim.x) {
      const int a = p, b = kp_prev + p;
      // which slot is the LEFT child: from the explicit built-is-left
      // record when available (segment adjacency is ambiguous when this
      // rank's local shard of the pair is empty — distributed mode)
      const bool a_left = mode_prev != nullptr
                              ? (mode_prev[p] != 0)
                              : (segs[2 * a + 1] == segs[2 * b]);
      s_order[2 * p] = a_left ? a : b;
      s_order[2 * p + 1] = a_left ? b : a;
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < kn; i += blockDim.x) {
    const int slot = s_order[i];
    const double gv = __longlong_as_double(best[6 * slot]);
    const long long bin = best[6 * slot + 1];
    // expand iff gain > 0 (validity) AND gain >= gamma (reference
    // driver.h:37 prunes loss_chg < min_split_loss, boundary included)
    s_flag[i] =
        (bin >= 0 && isfinite(gv) && gv > 0.0 && gv >= gamma_) ? 1 : 0;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    int kb = 0;
    for (int i = 0; i < kn; ++i) {
      if (s_flag[i]) s_exp[kb++] = s_order[i];
    }
    s_kb = kb;
    *kp_out = kb;
    *kn_out = 2 * kb;
  }
  __syncthreads();
  const int kb = s_kb;
  for (int j = threadIdx.x; j < kb; j += blockDim.x) {
    const int e = s_exp[j];
    const int f = (int)best[6 * e + 5];
    const int sb = segs[2 * e], se = segs[2 * e + 1];
    feat[j] = f;
    sbin[j] = (int)best[6 * e + 1] - cut_ptrs[f];
    dl[j] = (uint8_t)best[6 * e + 2];
    cnt[2 * j] = sb;
    cnt[2 * j + 1] = se;
    desc[4 * j] = sb;
    desc[4 * j + 1] = se;
    desc[4 * j + 2] = j;
    const long long pgq = ps_prev[2 * e], phq = ps_prev[2 * e + 1];
    const long long lgq = best[6 * e + 3], lhq = best[6 * e + 4];
    int mode = 2;  // device picks the smaller LOCAL child
    if (choice_global) {
      // rank-identical: global hessians come from the (allreduced)
      // histogram's best record, so every rank builds the same slot
      mode = (lhq <= phq - lhq) ? 1 : 0;
    }
    desc[4 * j + 3] = mode;
    parent_ps[2 * j] = pgq;
    parent_ps[2 * j + 1] = phq;
    parent_slot[j] = e;
    s_b[j] = sb;
    s_e[j] = se;
    if (bounds_out != nullptr) {
      // monotone bound propagation (same math as the per-level host
      // path): child bounds in SLOT order — built j, subtracted kb+j
      double lo = bounds_prev ? bounds_prev[2 * e] : -INFINITY;
      double hi = bounds_prev ? bounds_prev[2 * e + 1] : INFINITY;
      double llo = lo, lhi = hi, rlo = lo, rhi = hi;
      const int c = monotone ? (int)monotone[f] : 0;
      if (c != 0) {
        const double gsc =
            maxabs[0] > 0.f ? 1073741824.0 / (double)maxabs[0] : 1.0;
        const double hsc =
            maxabs[1] > 0.f ? 1073741824.0 / (double)maxabs[1] : 1.0;
        const double ig = 1.0 / gsc, ih = 1.0 / hsc;
        double wl = DevCalcWeight(lgq * ig, lhq * ih, lam, alpha, mds);
        double wr = DevCalcWeight((pgq - lgq) * ig, (phq - lhq) * ih, lam,
                                  alpha, mds);
        wl = fmin(fmax(wl, lo), hi);
        wr = fmin(fmax(wr, lo), hi);
        const double mid = (wl + wr) / 2.0;
        if (c > 0) {
          lhi = fmin(hi, mid);
          rlo = fmax(lo, mid);
        } else {
          llo = fmax(lo, mid);
          rhi = fmin(hi, mid);
        }
      }
      const bool built_left = (mode == 1);
      bounds_out[2 * j] = built_left ? llo : rlo;
      bounds_out[2 * j + 1] = built_left ? lhi : rhi;
      bounds_out[2 * (kb + j)] = built_left ? rlo : llo;
      bounds_out[2 * (kb + j) + 1] = built_left ? rhi : lhi;
    }
  }
  // partition tasks over the expand segments, padded to max_ptasks
  __shared__ long long s_rpt;
  __shared__ int s_tt;
  __shared__ int s_pref[2048];
  __syncthreads();
  if (threadIdx.x == 0) {
    long long total = 0;
    for (int j = 0; j < kb; ++j) total += s_e[j] - s_b[j];
    long long rpt0 =
        std::max(min_rows, (total + target_tasks - 1) / target_tasks);
    int shift = 0;
    while ((1LL << shift) < rpt0) ++shift;
    for (;;) {
      long long need = 0;
      for (int j = 0; j < kb; ++j) {
        need += ((long long)(s_e[j] - s_b[j]) + (1LL << shift) - 1) >> shift;
      }
      if (need <= max_ptasks) break;
      ++shift;
    }
    int pref = 0;
    for (int j = 0; j < kb; ++j) {
      s_pref[j] = pref;
      pref += (int)(((long long)(s_e[j] - s_b[j]) + (1LL << shift) - 1)
                    >> shift);
    }
    s_rpt = 1LL << shift;
    s_tt = pref;
  }
  __syncthreads();
  const long long rpt = s_rpt;
  const int tt = s_tt;
  for (int t = threadIdx.x; t < max_ptasks; t += blockDim.x) {
    if (t >= tt || kb == 0) {
      out_tasks[t] = BlockTask{0, 0, 0, 0};
      continue;
    }
    int lo = 0, hi = kb - 1;
    while (lo < hi) {
      const int mid = (lo + hi + 1) >> 1;
      if (s_pref[mid] <= t) lo = mid; else hi = mid - 1;
    }
    const int j = lo;
    const long long idx = t - s_pref[j];
    const int b = s_b[j] + (int)(idx * rpt);
    const int e = (int)std::min<long long>((long long)b + rpt,
                                           (long long)s_e[j]);
    out_tasks[t] = BlockTask{j, b, e, 0};
  }
}

struct LeafSeg {
  int nid, begin, end;
  int parity;  // which ping-pong ridx buffer holds these rows
};

}  // namespace

extern "C" {

void* gbt_driver_create() { return new DriverCtx(); }
void gbt_driver_destroy(void* ctx) { delete (DriverCtx*)ctx; }

typedef void (*AllreduceFn)(long long* dev_ptr, long long n_elems);

// Returns n_nodes (>0) on success, negative on failure.
int gbt_grow_tree(
    void* vctx, const uint8_t* gidx8, const uint16_t* gidx16, int n_features,
    const uint8_t* gidx8_col, const uint16_t* gidx16_col,  // feature-major
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
    int32_t* part_counters,      // unused (counters live in the ring
                                 // slot / whole-tree arena); kept in the
                                 // ABI for wrapper stability
    BlockTask* hist_tasks_dev,   // [hist_tasks_cap]
    int hist_tasks_cap,
    int32_t* tg_scratch,         // [3 * max_nodes_level + 4]
    void* wt_ws,                 // whole-tree record/arg arena, or null
    long long wt_ws_bytes, int wt_max_ptasks,
    const int64_t* root_sums_dev,  // [2] exact quantized (g, h) totals
    const float* maxabs_dev,  // null, or [2] gradient max-abs: scales are
                              // derived on device for the root phase and
                              // on host after the root sync (no separate
                              // max-abs readback sync per round)
    double* out_scales,       // [2] out: the derived (g_scale, h_scale)
    // scalars
    double g_scale, double h_scale,
    double reg_lambda, double reg_alpha, double max_delta_step,
    double min_child_weight, double gamma, double eta, int max_depth,
    const int8_t* monotone_dev, const int8_t* monotone_host,
    const uint8_t* fmask_dev,  // null, or [n_features] colsample_bytree
                               // feature mask (broadcast to every node)
    AllreduceFn allreduce,
    // host tree outputs (caller-sized to 2^(max_depth+1))
    int32_t* out_left, int32_t* out_right, int32_t* out_parent,
    int32_t* out_split_index, float* out_split_cond,
    uint8_t* out_default_left,
    double* out_loss_chg,  // fp64: the grow-policy replay orders its
                           // priority queue by these exact gains
    float* out_sum_hess,
    float* out_base_weight, void* stream_v) {
  DriverCtx* ctx = (DriverCtx*)vctx;
  hipStream_t stream = (hipStream_t)stream_v;
  HostParams p{reg_lambda, reg_alpha, max_delta_step};
  const bool has_mono = monotone_host != nullptr;
  double inv_g = 1.0 / g_scale, inv_h = 1.0 / h_scale;
  const long long hist_row = (long long)n_bins * 2;
  const bool whole_tree =
      wt_ws != nullptr &&
      maxabs_dev != nullptr && max_nodes_level <= 1024 && max_depth >= 2 &&
      wt_max_ptasks > 0;

  if (!whole_tree) {
    int blocks = (int)std::min<long long>((n_rows + 255) / 256, 4096);
    hipLaunchKernelGGL(IotaKernel, dim3(blocks), dim3(256), 0, stream, ridx,
                       n_rows);
  }
  // ping-pong row-index buffers: the partition scatters cur -> alt and
  // the buffers swap, so no copy-back pass is needed.  Rows of a node
  // that stops expanding stay put in whichever buffer was current at
  // that level (later partitions only write other, disjoint ranges),
  // so each leaf records its buffer parity for the final position pass.
  int32_t* cur_ridx = ridx;
  int32_t* alt_ridx = ridx_out;

  int n_tree_nodes = 1;
  out_left[0] = -1;
  out_right[0] = -1;
  out_parent[0] = -1;

  Node root{};
  root.nid = 0;
  root.seg_begin = 0;
  root.seg_end = (int)n_rows;
  root.hist_slot = 0;
  root.lo = -INFINITY;
  root.hi = INFINITY;
  // root.gq/hq arrive with the root-eval readback (root_sums_dev);
  // the evaluator reads the device copy directly

  // hist wants FEWER, BIGGER tasks than partition: every block pays a
  // full LDS zero+flush of the group histogram, so rows/task must
  // amortize ~2*group_bins atomics (env GBT_HIST_TASKS to tune)
  static long long hist_tasks = [] {
    const char* e = getenv("GBT_HIST_TASKS");
    long long v = e ? atoll(e) : 256;  // swept: 256 beats 512 by ~4%
                                       // under whole-tree mode
    return v >= 64 && v <= 16384 ? v : 256;
  }();
  const long long hist_min_rows = 2048;

  // ---- root histogram: tasks host-generated (root segment is known)
  if (!whole_tree) {
    std::vector<Node*> nodes{&root};
    std::vector<BlockTask> tasks;
    ChunkTasks(nodes, &tasks, hist_min_rows, hist_tasks);
    const int slot = ctx->ring.next();
    size_t bytes = tasks.size() * sizeof(BlockTask);
    if (int e = ctx->ring.ensure(slot, bytes)) return e;
    memcpy(ctx->ring.host[slot], tasks.data(), bytes);
    HIP_CHECK(hipMemcpyAsync(ctx->ring.dev[slot], ctx->ring.host[slot], bytes,
                             hipMemcpyHostToDevice, stream));
    HIP_CHECK(hipMemsetAsync(hist_pool_a, 0, hist_row * sizeof(int64_t),
                             stream));
    gbt_hist(gidx8, gidx16, n_features, qgpair, ridx,
             (const BlockTask*)ctx->ring.dev[slot], (int)tasks.size(),
             hist_pool_a, n_bins, feat_group_start_dev, bin_group_start_dev,
             n_groups, max_group_bins, cut_ptrs_dev, use_shared, nullptr,
             stream);
    if (allreduce) allreduce((long long*)hist_pool_a, hist_row);
  }

  // enqueue split evaluation against `hists`; the best-split D2H is
  // enqueued but NOT synced — the caller syncs.  Node sums are staged
  // from `nodes` unless a device-resident sum buffer is given
  // (device-chosen siblings compute their sums on the GPU).
  auto evaluate_enqueue = [&](int k, std::vector<Node*>* nodes,
                              const int64_t* hists, const int64_t* ps_dev,
                              const float* maxabs_eval = nullptr) -> int {
    const int64_t* ps_arg = ps_dev;
    if (ps_dev == nullptr) {
      const int slot = ctx->ring.next();
      size_t off_ps = 0;
      size_t off_bd = ((size_t)k * 2 * sizeof(int64_t) + 7) & ~7ULL;
      size_t bytes = off_bd + (has_mono ? (size_t)k * 2 * sizeof(double) : 0);
      if (int e = ctx->ring.ensure(slot, bytes)) return e;
      char* h = (char*)ctx->ring.host[slot];
      int64_t* ps = (int64_t*)(h + off_ps);
      for (int i = 0; i < k; ++i) {
        ps[2 * i] = (*nodes)[i]->gq;
        ps[2 * i + 1] = (*nodes)[i]->hq;
      }
      if (has_mono) {
        double* bd = (double*)(h + off_bd);
        for (int i = 0; i < k; ++i) {
          bd[2 * i] = (*nodes)[i]->lo;
          bd[2 * i + 1] = (*nodes)[i]->hi;
        }
      }
      HIP_CHECK(hipMemcpyAsync(ctx->ring.dev[slot], h, bytes,
                               hipMemcpyHostToDevice, stream));
      char* d = (char*)ctx->ring.dev[slot];
      ps_arg = (const int64_t*)(d + off_ps);
      if (has_mono) {
        gbt_evaluate_masked(hists, k, n_bins, n_features, cut_ptrs_dev, ps_arg,
                     maxabs_eval,
                     g_scale, h_scale, reg_lambda, reg_alpha, max_delta_step,
                     min_child_weight, monotone_dev,
                     (const double*)(d + off_bd), 0 /*mask_stride*/, fmask_dev, nullptr, eval_gain,
                     eval_bin, eval_dir, eval_lsum, nullptr, 0, stream);
        gbt_select_best(eval_gain, eval_bin, eval_dir, eval_lsum, k,
                        n_features, eval_best, nullptr, stream);
        return 0;
      }
    }
    gbt_evaluate_masked(hists, k, n_bins, n_features, cut_ptrs_dev, ps_arg,
                 maxabs_eval, g_scale,
                 h_scale, reg_lambda, reg_alpha, max_delta_step,
                 min_child_weight, monotone_dev, nullptr, 0 /*mask_stride*/, fmask_dev, nullptr,
                 eval_gain, eval_bin, eval_dir, eval_lsum, nullptr, 0, stream);
    gbt_select_best(eval_gain, eval_bin, eval_dir, eval_lsum, k, n_features,
                    eval_best, nullptr, stream);
    return 0;
  };

  auto parse_best = [&](std::vector<Node*>& nodes, const int64_t* best) {
    for (size_t i = 0; i < nodes.size(); ++i) {
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
  };

  // one D2H burst + ONE sync per level: best splits for `eval_nodes`
  // (may be empty on the final level) + partition counters for
  // `n_expand` nodes
  auto level_sync = [&](int n_eval, int n_expand,
                        const int32_t* cnt_dev,
                        const int64_t** best_out,
                        const int32_t** cnt_out) -> int {
    size_t off_best = 0;
    size_t off_cnt = ((size_t)n_eval * 6 * sizeof(int64_t) + 63) & ~63ULL;
    size_t bytes = off_cnt + (size_t)n_expand * 2 * sizeof(int32_t);
    if (int e = ctx->ensure_readback(bytes)) return e;
    char* h = (char*)ctx->readback_host;
    if (n_eval > 0) {
      HIP_CHECK(hipMemcpyAsync(h + off_best, eval_best,
                               (size_t)n_eval * 6 * sizeof(int64_t),
                               hipMemcpyDeviceToHost, stream));
    }
    if (n_expand > 0) {
      HIP_CHECK(hipMemcpyAsync(h + off_cnt, cnt_dev,
                               (size_t)n_expand * 2 * sizeof(int32_t),
                               hipMemcpyDeviceToHost, stream));
    }
    HIP_CHECK(hipStreamSynchronize(stream));
    *best_out = (const int64_t*)(h + off_best);
    *cnt_out = (const int32_t*)(h + off_cnt);
    return 0;
  };

  std::vector<Node> level_nodes;
  std::vector<Node> next_level;
  std::vector<LeafSeg> leaves;
  int64_t* cur_pool = hist_pool_a;
  int64_t* next_pool = hist_pool_b;

  // ================= WHOLE-TREE MODE =================
  // Single GPU, no monotone bounds: every level's phase chain —
  // apply-splits, partition, task generation, hist, subtraction,
  // evaluation — is enqueued with NO host sync (the expansion decision
  // and all task lists are computed on device); ONE readback at the
  // end returns the per-level best-split and segment records, and the
  // host replays them to build the tree arrays (bit-identical: the
  // replay applies the same comparisons to the same doubles).
  // Distributed (allreduce != null): same single-enqueue chain — the
  // per-level hist/pair-sum allreduces are enqueued with a host-known
  // WORST-CASE padded count (zero slots reduce to zeros), and the
  // sibling build/subtract choice comes from GLOBAL hessians so every
  // rank reduces the same slot layout.  Monotone: fp64 bound
  // propagation runs inside ApplyKernel (scales derived on device).
  // Single-GPU: the fixed launch sequence is captured into a hipGraph
  // once and replayed (one graph launch instead of ~25 enqueues/tree).
  if (whole_tree) {
    const int choice_global = (allreduce != nullptr || has_mono) ? 1 : 0;
    const int pool = 2 * max_nodes_level;
    const int rec_slots = 1 + max_depth * pool;
    size_t off = 0;
    auto carve = [&](size_t bytes) {
      size_t o = off;
      off = (off + bytes + 255) & ~255ULL;
      return o;
    };
    const size_t o_best = carve((size_t)rec_slots * 6 * 8);
    const size_t o_seg = carve((size_t)rec_slots * 2 * 4);
    const size_t o_kn = carve((size_t)(max_depth + 2) * 4);
    const size_t o_kp = carve((size_t)(max_depth + 2) * 4);
    const size_t o_descw = carve((size_t)pool * 16);
    const size_t o_featw = carve((size_t)pool * 4);
    const size_t o_sbinw = carve((size_t)pool * 4);
    const size_t o_dlw = carve((size_t)pool);
    const size_t o_cntw = carve((size_t)pool * 8);
    const size_t o_pt = carve((size_t)wt_max_ptasks * sizeof(BlockTask));
    const size_t o_pps = carve((size_t)pool * 16);
    const size_t o_pslot = carve((size_t)pool * 4);
    const size_t o_psa = carve((size_t)pool * 16);
    const size_t o_psb = carve((size_t)pool * 16);
    const size_t o_bnda = has_mono ? carve((size_t)pool * 16) : 0;
    const size_t o_bndb = has_mono ? carve((size_t)pool * 16) : 0;
    const size_t o_mode = carve((size_t)max_depth * pool);
    if ((long long)off > wt_ws_bytes) return -9997;
    char* w = (char*)wt_ws;
    int64_t* best_rec = (int64_t*)(w + o_best);
    int32_t* seg_rec = (int32_t*)(w + o_seg);
    int32_t* kn_arr = (int32_t*)(w + o_kn);
    int32_t* kp_arr = (int32_t*)(w + o_kp);
    int32_t* d_desc = (int32_t*)(w + o_descw);
    int32_t* d_feat = (int32_t*)(w + o_featw);
    int32_t* d_sbin = (int32_t*)(w + o_sbinw);
    uint8_t* d_dl = (uint8_t*)(w + o_dlw);
    int32_t* d_cnt = (int32_t*)(w + o_cntw);
    BlockTask* d_pt = (BlockTask*)(w + o_pt);
    int64_t* d_pps = (int64_t*)(w + o_pps);
    int32_t* d_pslot = (int32_t*)(w + o_pslot);
    int64_t* ps_bufs[2] = {(int64_t*)(w + o_psa), (int64_t*)(w + o_psb)};
    double* bnd_bufs[2] = {
        has_mono ? (double*)(w + o_bnda) : nullptr,
        has_mono ? (double*)(w + o_bndb) : nullptr};
    uint8_t* d_mode = (uint8_t*)(w + o_mode);  // [max_depth][pool]

    // root tasks: regenerated each call with IDENTICAL content (they
    // depend only on n_rows and the task constants) and staged through
    // the DEDICATED pinned const buffer, so a captured graph can bake
    // the addresses and every replay re-reads the rewritten bytes.
    // Layout: [0..15] kn/kp/seg init ints, then the root BlockTasks.
    std::vector<BlockTask> rtasks;
    {
      std::vector<Node*> rnodes{&root};
      ChunkTasks(rnodes, &rtasks, hist_min_rows, hist_tasks);
    }
    const size_t c_tasks_off = 16;
    if (int e = ctx->ensure_const(c_tasks_off +
                                  rtasks.size() * sizeof(BlockTask)))
      return e;
    {
      int32_t* hh = (int32_t*)ctx->wtc_host;
      hh[0] = 1;
      hh[1] = 0;
      hh[2] = 0;
      hh[3] = (int)n_rows;
      memcpy((char*)ctx->wtc_host + c_tasks_off, rtasks.data(),
             rtasks.size() * sizeof(BlockTask));
    }
    const int wt_max_htasks = (int)std::min<long long>(
        std::min<long long>(n_rows / hist_min_rows, hist_tasks) +
            max_nodes_level + 1,
        (long long)hist_tasks_cap);
    // pre-size the ONE readback so its (graph-baked) address is final
    size_t r_best = 0;
    size_t r_seg = ((size_t)rec_slots * 48 + 63) & ~63ULL;
    size_t r_kp = (r_seg + (size_t)rec_slots * 8 + 63) & ~63ULL;
    size_t r_rs = (r_kp + (size_t)(max_depth + 2) * 4 + 63) & ~63ULL;
    size_t r_ma = r_rs + 16;
    size_t r_mode = (r_ma + 8 + 63) & ~63ULL;
    size_t r_total = r_mode + (size_t)max_depth * pool;
    if (int e = ctx->ensure_readback(r_total)) return e;
    char* rb = (char*)ctx->readback_host;

    // ---- the chain: EVERY launch for the whole tree, no host sync.
    // Ping-pong pointers are locals so a graph REPLAY (which skips this
    // code entirely) leaves the outer state identical.
    // H2D staging: tiny constant uploads, enqueued directly each call
    // (kept OUT of the captured graph — graph-embedded memcpy nodes
    // showed unreliable replay ordering on ROCm)
    auto stage_h2d = [&](hipStream_t cs) -> int {
      char* ch = (char*)ctx->wtc_host;
      HIP_CHECK(hipMemcpyAsync(kn_arr, ch, 4, hipMemcpyHostToDevice,
                               cs));
      HIP_CHECK(hipMemcpyAsync(kp_arr, ch + 4, 4, hipMemcpyHostToDevice,
                               cs));
      HIP_CHECK(hipMemcpyAsync(seg_rec, ch + 8, 8, hipMemcpyHostToDevice,
                               cs));
      HIP_CHECK(hipMemcpyAsync((char*)ctx->wtc_dev + c_tasks_off,
                               ch + c_tasks_off,
                               rtasks.size() * sizeof(BlockTask),
                               hipMemcpyHostToDevice, cs));
      return 0;
    };
    auto enqueue_chain = [&](hipStream_t cs) -> int {
      int32_t* cr = ridx;
      int32_t* ar = ridx_out;
      int64_t* cp = hist_pool_a;
      int64_t* np2 = hist_pool_b;
      {
        int blocks = (int)std::min<long long>((n_rows + 255) / 256, 4096);
        hipLaunchKernelGGL(IotaKernel, dim3(blocks), dim3(256), 0, cs,
                           cr, n_rows);
      }
      {
        // zero via a kernel, not hipMemsetAsync: memset NODES in a
        // captured graph replay out of order on this ROCm
        int zb = (int)std::min<long long>((hist_row + 255) / 256, 4096);
        hipLaunchKernelGGL(ZeroI64Kernel, dim3(zb), dim3(256), 0, cs, cp,
                           hist_row);
      }
      gbt_hist(gidx8, gidx16, n_features, qgpair, cr,
               (const BlockTask*)((char*)ctx->wtc_dev + c_tasks_off),
               (int)rtasks.size(), cp, n_bins, feat_group_start_dev,
               bin_group_start_dev, n_groups, max_group_bins, cut_ptrs_dev,
               use_shared, nullptr, cs);
      if (allreduce) allreduce((long long*)cp, hist_row);
      // root evaluation straight into best_rec[0]
      gbt_evaluate_masked(cp, 1, n_bins, n_features, cut_ptrs_dev,
                   root_sums_dev, maxabs_dev, 0.0, 0.0, reg_lambda,
                   reg_alpha, max_delta_step, min_child_weight,
                   has_mono ? monotone_dev : nullptr, nullptr,
                   0 /*mask_stride*/, fmask_dev,
                   nullptr, eval_gain, eval_bin, eval_dir, eval_lsum,
                   nullptr, 0, cs);
      gbt_select_best(eval_gain, eval_bin, eval_dir, eval_lsum, 1,
                      n_features, best_rec, nullptr, cs);
      const int64_t* ps_prev = root_sums_dev;
      for (int L = 0; L + 1 < max_depth; ++L) {
        const int64_t* bl =
            best_rec + (L == 0 ? 0 : (1 + (size_t)(L - 1) * pool) * 6);
        const int32_t* sl =
            seg_rec + (L == 0 ? 0 : (1 + (size_t)(L - 1) * pool) * 2);
        int64_t* bo = best_rec + (1 + (size_t)L * pool) * 6;
        int32_t* so = seg_rec + (1 + (size_t)L * pool) * 2;
        int64_t* ps_next = ps_bufs[L & 1];
        // level-L capacity: at most 2^L nodes expand, 2^(L+1) children
        const int cap_kids =
            (int)std::min<long long>(2LL << L, (long long)pool);
        hipLaunchKernelGGL(ApplyKernel, dim3(1), dim3(256), 0, cs, bl,
                           sl, kn_arr + L, kp_arr + L, ps_prev, gamma,
                           cut_ptrs_dev,
                           (long long)1024, (long long)2048, wt_max_ptasks,
                           kn_arr + L + 1, kp_arr + L + 1, d_feat, d_sbin,
                           d_dl, d_cnt, d_pt, d_desc, d_pps, d_pslot,
                           choice_global, maxabs_dev,
                           has_mono ? monotone_dev : nullptr,
                           reg_lambda, reg_alpha, max_delta_step,
                           (L == 0 || !has_mono) ? nullptr
                                                 : bnd_bufs[(L - 1) & 1],
                           has_mono ? bnd_bufs[L & 1] : nullptr,
                           L == 0 ? nullptr : d_mode + (size_t)(L - 1) * pool);
        gbt_partition(gidx8, gidx16, n_features, gidx8_col, gidx16_col,
                      n_rows, cr, ar, d_pt,
                      wt_max_ptasks, d_feat, d_sbin, d_dl, nullptr, nullptr,
                      n_bins_feat_dev, d_cnt, cs);
        std::swap(cr, ar);
        hipLaunchKernelGGL(HistTaskGenKernel, dim3(1), dim3(256), 0, cs,
                           d_cnt, d_desc, 0, hist_min_rows, hist_tasks,
                           wt_max_htasks, tg_scratch, hist_tasks_dev,
                           ps_next, kp_arr + L + 1, so,
                           allreduce != nullptr ? 2 * cap_kids : 0,
                           d_mode + (size_t)L * pool);
        {
          const long long zn = (long long)cap_kids * hist_row;
          int zb = (int)std::min<long long>((zn + 255) / 256, 4096);
          hipLaunchKernelGGL(ZeroI64Kernel, dim3(zb), dim3(256), 0, cs,
                             np2, zn);
        }
        gbt_hist(gidx8, gidx16, n_features, qgpair, cr, hist_tasks_dev,
                 wt_max_htasks, np2, n_bins, feat_group_start_dev,
                 bin_group_start_dev, n_groups, max_group_bins,
                 cut_ptrs_dev, use_shared, ps_next, cs);
        if (allreduce) {
          // fixed worst-case counts (host-known, rank-identical): built
          // hist slots 0..cap/2-1 (unused slots are zeros — the memset
          // covers the pool, HistTaskGenKernel zeroed the ps padding)
          // and the built-child pair sums.  Sibling subtraction AFTER
          // the reduce yields global histograms/sums for every child.
          allreduce((long long*)np2,
                    (long long)(cap_kids / 2) * hist_row);
          allreduce((long long*)ps_next, (long long)cap_kids);
        }
        {
          const long long total = (long long)(cap_kids / 2 + 1) * hist_row;
          int blocks = (int)std::min<long long>((total + 255) / 256, 4096);
          hipLaunchKernelGGL(SubtractHistKernel, dim3(blocks), dim3(256), 0,
                             cs, cp, np2, np2, d_pslot,
                             (int)hist_row, 0, ps_next, d_pps,
                             kp_arr + L + 1);
        }
        gbt_evaluate_masked(np2, cap_kids, n_bins, n_features, cut_ptrs_dev,
                     ps_next, maxabs_dev, 0.0, 0.0, reg_lambda, reg_alpha,
                     max_delta_step, min_child_weight,
                     has_mono ? monotone_dev : nullptr,
                     has_mono ? bnd_bufs[L & 1] : nullptr,
                     0 /*mask_stride*/, fmask_dev, nullptr, eval_gain,
                     eval_bin, eval_dir, eval_lsum, kn_arr + L + 1, 0,
                     cs);
        gbt_select_best(eval_gain, eval_bin, eval_dir, eval_lsum, cap_kids,
                        n_features, bo, kn_arr + L + 1, cs);
        ps_prev = ps_next;
        std::swap(cp, np2);
      }
      return 0;
    };
    // D2H readback burst: direct enqueue after the kernels (not a graph
    // node), the ONE sync follows
    auto readback = [&](hipStream_t cs) -> int {
      HIP_CHECK(hipMemcpyAsync(rb + r_best, best_rec,
                               (size_t)rec_slots * 48,
                               hipMemcpyDeviceToHost, cs));
      HIP_CHECK(hipMemcpyAsync(rb + r_seg, seg_rec, (size_t)rec_slots * 8,
                               hipMemcpyDeviceToHost, cs));
      HIP_CHECK(hipMemcpyAsync(rb + r_kp, kp_arr,
                               (size_t)(max_depth + 2) * 4,
                               hipMemcpyDeviceToHost, cs));
      HIP_CHECK(hipMemcpyAsync(rb + r_rs, root_sums_dev, 16,
                               hipMemcpyDeviceToHost, cs));
      HIP_CHECK(hipMemcpyAsync(rb + r_ma, maxabs_dev, 8,
                               hipMemcpyDeviceToHost, cs));
      HIP_CHECK(hipMemcpyAsync(rb + r_mode, d_mode,
                               (size_t)max_depth * pool,
                               hipMemcpyDeviceToHost, cs));
      return 0;
    };

    // hipGraph capture/replay (single-GPU only): the key records every
    // address and scalar the chain bakes at enqueue time — any change
    // forces a recapture, so replay is exactly equivalent.  Capture and
    // launch run on the ctx-owned stream (never the caller's, which is
    // usually the un-capturable default stream), ordered behind the
    // caller's pending work by gevent and joined by the host sync.
    bool enqueued = false;
    if (allreduce == nullptr && WtGraphEnabled() &&
        ctx->ensure_gstream() == 0) {
      auto db = [](double d) {
        unsigned long long u;
        memcpy(&u, &d, 8);
        return u;
      };
      std::vector<unsigned long long> key = {
          (unsigned long long)gidx8, (unsigned long long)gidx16,
          (unsigned long long)gidx8_col, (unsigned long long)gidx16_col,
          (unsigned long long)qgpair, (unsigned long long)cut_ptrs_dev,
          (unsigned long long)n_bins_feat_dev,
          (unsigned long long)feat_group_start_dev,
          (unsigned long long)bin_group_start_dev,
          (unsigned long long)ridx, (unsigned long long)ridx_out,
          (unsigned long long)hist_pool_a, (unsigned long long)hist_pool_b,
          (unsigned long long)eval_gain, (unsigned long long)eval_bin,
          (unsigned long long)eval_dir, (unsigned long long)eval_lsum,
          (unsigned long long)hist_tasks_dev,
          (unsigned long long)tg_scratch, (unsigned long long)wt_ws,
          (unsigned long long)root_sums_dev,
          (unsigned long long)maxabs_dev, (unsigned long long)monotone_dev,
          (unsigned long long)fmask_dev,
          (unsigned long long)ctx->readback_host,
          (unsigned long long)ctx->wtc_host,
          (unsigned long long)ctx->wtc_dev,
          (unsigned long long)n_rows, (unsigned long long)n_features,
          (unsigned long long)n_bins, (unsigned long long)use_shared,
          (unsigned long long)n_groups,
          (unsigned long long)max_group_bins,
          (unsigned long long)max_depth,
          (unsigned long long)max_nodes_level,
          (unsigned long long)wt_max_ptasks,
          (unsigned long long)hist_tasks_cap,
          (unsigned long long)rtasks.size(),
          (unsigned long long)has_mono,
          db(reg_lambda), db(reg_alpha), db(max_delta_step),
          db(min_child_weight), db(gamma)};
      // order the graph stream behind the caller stream (staged
      // gradient/root-sum copies enqueued by the wrapper).  HOST-side
      // wait: a hipStreamWaitEvent dependency is not reliably honored
      // by hipGraphLaunch on the target stream (observed: replays read
      // half-staged gradients), and the whole chain is about to run for
      // ~0.5 ms anyway, so the few-us host block is in the noise.
      HIP_CHECK(hipEventRecord(ctx->gevent, stream));
      HIP_CHECK(hipEventSynchronize(ctx->gevent));
      if (ctx->wt_graph != nullptr && key == ctx->wt_key) {
        if (int e = stage_h2d(ctx->gstream)) return e;
        if (hipGraphLaunch(ctx->wt_graph, ctx->gstream) == hipSuccess) {
          enqueued = true;
          if (WtGraphDebug()) fprintf(stderr, "[wtgraph] replay\n");
        } else {
          (void)hipGetLastError();
          (void)hipGraphExecDestroy(ctx->wt_graph);
          ctx->wt_graph = nullptr;
          ctx->wt_key.clear();
        }
      }
      if (!enqueued && key != ctx->wt_seen_key) {
        // first sighting of this configuration: run direct, remember it
        ctx->wt_seen_key = key;
      } else if (!enqueued) {
        if (int e = stage_h2d(ctx->gstream)) return e;
        if (hipStreamBeginCapture(ctx->gstream,
                                  hipStreamCaptureModeRelaxed) !=
            hipSuccess) {
          (void)hipGetLastError();
          goto graph_done;
        }
        int ec = enqueue_chain(ctx->gstream);
        hipGraph_t g = nullptr;
        hipError_t ce = hipStreamEndCapture(ctx->gstream, &g);
        if (WtGraphDebug())
          fprintf(stderr, "[wtgraph] capture ec=%d end=%d\n", ec, (int)ce);
        if (ec != 0) {
          if (g) (void)hipGraphDestroy(g);
          return ec;
        }
        if (ce == hipSuccess && g != nullptr) {
          if (ctx->wt_graph) {
            (void)hipGraphExecDestroy(ctx->wt_graph);
            ctx->wt_graph = nullptr;
          }
          ctx->wt_key.clear();
          hipError_t ie =
              hipGraphInstantiate(&ctx->wt_graph, g, nullptr, nullptr, 0);
          hipError_t le =
              ie == hipSuccess ? hipGraphLaunch(ctx->wt_graph, ctx->gstream)
                               : hipErrorUnknown;
          if (WtGraphDebug())
            fprintf(stderr, "[wtgraph] inst=%d launch=%d\n", (int)ie,
                    (int)le);
          if (ie == hipSuccess && le == hipSuccess) {
            ctx->wt_key = key;
            enqueued = true;
          } else if (ctx->wt_graph) {
            (void)hipGraphExecDestroy(ctx->wt_graph);
            ctx->wt_graph = nullptr;
          }
          (void)hipGraphDestroy(g);
        } else {
          (void)hipGetLastError();
        }
        // a capture enqueues nothing for execution — every failure path
        // above leaves enqueued=false and falls through to the direct
        // enqueue below
      }
    graph_done:
      if (!enqueued) {
        // safety: never leave the graph stream wedged mid-capture
        hipStreamCaptureStatus st = hipStreamCaptureStatusNone;
        if (hipStreamIsCapturing(ctx->gstream, &st) == hipSuccess &&
            st != hipStreamCaptureStatusNone) {
          hipGraph_t junk = nullptr;
          (void)hipStreamEndCapture(ctx->gstream, &junk);
          if (junk) (void)hipGraphDestroy(junk);
        }
      }
    }
    if (!enqueued) {
      if (WtGraphDebug()) fprintf(stderr, "[wtgraph] direct enqueue\n");
      if (int e = stage_h2d(stream)) return e;
      if (int e = enqueue_chain(stream)) return e;
      if (int e = readback(stream)) return e;
    } else {
      if (int e = readback(ctx->gstream)) return e;
      HIP_CHECK(hipStreamSynchronize(ctx->gstream));
    }
    HIP_CHECK(hipStreamSynchronize(stream));
    // the chain used local ping-pong pointers; reproduce the final
    // parity for the post-sync leaf decide ((max_depth-1) swaps)
    if ((max_depth - 1) & 1) std::swap(cur_ridx, alt_ridx);
    const int64_t* h_best = (const int64_t*)(rb + r_best);
    const int32_t* h_seg = (const int32_t*)(rb + r_seg);
    const int32_t* h_kp = (const int32_t*)(rb + r_kp);
    const int64_t* h_rs = (const int64_t*)(rb + r_rs);
    const float* h_ma = (const float*)(rb + r_ma);
    const uint8_t* h_mode = (const uint8_t*)(rb + r_mode);
    g_scale = h_ma[0] > 0.f ? 1073741824.0 / (double)h_ma[0] : 1.0;
    h_scale = h_ma[1] > 0.f ? 1073741824.0 / (double)h_ma[1] : 1.0;
    inv_g = 1.0 / g_scale;
    inv_h = 1.0 / h_scale;
    if (out_scales != nullptr) {
      out_scales[0] = g_scale;
      out_scales[1] = h_scale;
    }
    root.gq = h_rs[0];
    root.hq = h_rs[1];
    out_base_weight[0] =
        (float)CalcWeight(root.gq * inv_g, root.hq * inv_h, p);
    out_sum_hess[0] = (float)(root.hq * inv_h);
    {
      double gn;
      memcpy(&gn, &h_best[0], 8);
      root.bin = (int)h_best[1];
      root.dir = (int)h_best[2];
      root.lgq = h_best[3];
      root.lhq = h_best[4];
      root.feature = (int)h_best[5];
      root.gain = (root.bin >= 0 && std::isfinite(gn)) ? gn : -INFINITY;
    }
    // ---- host replay of the records ----
    level_nodes.push_back(root);
    for (int depth = 0; depth < max_depth && !level_nodes.empty(); ++depth) {
      const int parity = depth & 1;
      std::vector<Node*> expand;
      for (auto& nd : level_nodes) {
        if (nd.gain > 0.0 && nd.gain >= gamma && std::isfinite(nd.gain)) {
          expand.push_back(&nd);
        } else {
          leaves.push_back({nd.nid, nd.seg_begin, nd.seg_end, parity});
        }
      }
      if (expand.empty()) {
        level_nodes.clear();
        break;
      }
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
        out_loss_chg[nd->nid] = nd->gain;
        const long long rgq = nd->gq - nd->lgq, rhq = nd->hq - nd->lhq;
        double wl = CalcWeight(nd->lgq * inv_g, nd->lhq * inv_h, p);
        double wr = CalcWeight(rgq * inv_g, rhq * inv_h, p);
        // monotone: clamp to this node's bounds and propagate to the
        // children — bit-identical replay of ApplyKernel's fp64 math
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
        ln.gain = rn.gain = -INFINITY;
        ln.bin = rn.bin = -1;
        next_level.push_back(ln);
        next_level.push_back(rn);
      }
      const int kbl = (int)expand.size();
      if (depth + 1 >= max_depth) {
        // final level: direct leaf-position decide (host-staged args)
        std::vector<BlockTask> tasks;
        ChunkTasks(expand, &tasks);
        const int slot = ctx->ring.next();
        size_t offf = (tasks.size() * sizeof(BlockTask) + 7) & ~7ULL;
        size_t offs = (offf + (size_t)kbl * 4 + 7) & ~7ULL;
        size_t offd = (offs + (size_t)kbl * 4 + 7) & ~7ULL;
        size_t offk = (offd + (size_t)kbl + 7) & ~7ULL;
        size_t bytes = offk + (size_t)kbl * 8;
        if (int e = ctx->ring.ensure(slot, bytes)) return e;
        char* h = (char*)ctx->ring.host[slot];
        memcpy(h, tasks.data(), tasks.size() * sizeof(BlockTask));
        int32_t* hf = (int32_t*)(h + offf);
        int32_t* hs = (int32_t*)(h + offs);
        uint8_t* hd = (uint8_t*)(h + offd);
        int32_t* hk = (int32_t*)(h + offk);
        for (int j = 0; j < kbl; ++j) {
          Node* nd = expand[j];
          hf[j] = nd->feature;
          hs[j] = nd->bin - cut_ptrs_host[nd->feature];
          hd[j] = (uint8_t)nd->dir;
          hk[2 * j] = next_level[2 * j].nid;
          hk[2 * j + 1] = next_level[2 * j + 1].nid;
        }
        HIP_CHECK(hipMemcpyAsync(ctx->ring.dev[slot], h, bytes,
                                 hipMemcpyHostToDevice, stream));
        char* d = (char*)ctx->ring.dev[slot];
        gbt_leaf_decide(gidx8, gidx16, n_features, gidx8_col, gidx16_col,
                        n_rows, cur_ridx,
                        (const BlockTask*)d, (int)tasks.size(),
                        (const int32_t*)(d + offf),
                        (const int32_t*)(d + offs),
                        (const uint8_t*)(d + offd),
                        (const int32_t*)(d + offk), n_bins_feat_dev,
                        pos_out, stream);
        level_nodes.clear();
        break;
      }
      if (h_kp[depth + 1] != kbl) return -9991;  // replay divergence
      const int64_t* bo = h_best + (1 + (size_t)depth * pool) * 6;
      const int32_t* so = h_seg + (1 + (size_t)depth * pool) * 2;
      const uint8_t* mo = h_mode + (size_t)depth * pool;
      for (int j = 0; j < kbl; ++j) {
        Node& ln = next_level[2 * j];
        Node& rn = next_level[2 * j + 1];
        const int sA = j, sB = kbl + j;
        const int ab = so[2 * sA], ae = so[2 * sA + 1];
        const int bb2 = so[2 * sB], be = so[2 * sB + 1];
        const bool a_left = mo[j] != 0;  // built-is-left record
        ln.seg_begin = a_left ? ab : bb2;
        ln.seg_end = a_left ? ae : be;
        rn.seg_begin = a_left ? bb2 : ab;
        rn.seg_end = a_left ? be : ae;
        auto parse1 = [&](Node& nd, int slot2) {
          double gn;
          memcpy(&gn, &bo[6 * slot2], 8);
          nd.bin = (int)bo[6 * slot2 + 1];
          nd.dir = (int)bo[6 * slot2 + 2];
          nd.lgq = bo[6 * slot2 + 3];
          nd.lhq = bo[6 * slot2 + 4];
          nd.feature = (int)bo[6 * slot2 + 5];
          nd.gain = (nd.bin >= 0 && std::isfinite(gn)) ? gn : -INFINITY;
        };
        parse1(a_left ? ln : rn, sA);
        parse1(a_left ? rn : ln, sB);
      }
      level_nodes.swap(next_level);
    }
  } else {
  // ================= PER-LEVEL MODE =================
// ---- root evaluation (root-only sync; root sums + max-abs ride
  // along, so neither needs its own host round-trip) ----
  {
    std::vector<Node*> frontier{&root};
    if (int e = evaluate_enqueue(1, nullptr, hist_pool_a, root_sums_dev,
                                 maxabs_dev))
      return e;
    const int rslot = ctx->ring.next();
    if (int e = ctx->ring.ensure(rslot, 4 * sizeof(int64_t))) return e;
    char* rh = (char*)ctx->ring.host[rslot];
    HIP_CHECK(hipMemcpyAsync(rh, root_sums_dev, 2 * sizeof(int64_t),
                             hipMemcpyDeviceToHost, stream));
    if (maxabs_dev != nullptr) {
      HIP_CHECK(hipMemcpyAsync(rh + 2 * sizeof(int64_t), maxabs_dev,
                               2 * sizeof(float), hipMemcpyDeviceToHost,
                               stream));
    }
    const int64_t* best;
    const int32_t* cnt;
    if (int e = level_sync(1, 0, nullptr, &best, &cnt)) return e;
    const int64_t* rs = (const int64_t*)rh;
    root.gq = rs[0];
    root.hq = rs[1];
    if (maxabs_dev != nullptr) {
      const float* ma = (const float*)(rh + 2 * sizeof(int64_t));
      g_scale = ma[0] > 0.f ? 1073741824.0 / (double)ma[0] : 1.0;
      h_scale = ma[1] > 0.f ? 1073741824.0 / (double)ma[1] : 1.0;
      inv_g = 1.0 / g_scale;
      inv_h = 1.0 / h_scale;
      if (out_scales != nullptr) {
        out_scales[0] = g_scale;
        out_scales[1] = h_scale;
      }
    }
    out_base_weight[0] =
        (float)CalcWeight(root.gq * inv_g, root.hq * inv_h, p);
    out_sum_hess[0] = (float)(root.hq * inv_h);
    parse_best(frontier, best);
  }

  level_nodes.push_back(root);

  for (int depth = 0; depth < max_depth && !level_nodes.empty(); ++depth) {
    const int parity = (cur_ridx == ridx) ? 0 : 1;
    std::vector<Node*> expand;
    for (auto& nd : level_nodes) {
      if (nd.gain > 0.0 && nd.gain >= gamma && std::isfinite(nd.gain)) {
        expand.push_back(&nd);
      } else {
        leaves.push_back({nd.nid, nd.seg_begin, nd.seg_end, parity});
      }
    }
    if (expand.empty()) {
      level_nodes.clear();
      break;
    }
    // apply splits on host (child sums/bounds are known from the
    // parents' evaluation; child SEGMENTS arrive at the level sync)
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
      out_loss_chg[nd->nid] = nd->gain;
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
      ln.seg_begin = ln.seg_end = -1;  // set at the level sync
      rn.seg_begin = rn.seg_end = -1;
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
    const int n_expand = (int)expand.size();
    // Sibling to build vs subtract (exact int64 subtraction makes this
    // a performance choice only, never a correctness one):
    //  - single GPU, no monotone bounds: the DEVICE picks the
    //    smaller-row child inside HistTaskGenKernel (exact local
    //    counts) and the hist kernel accumulates its sums; the host
    //    learns the choice from the counters at the level sync.
    //  - distributed or monotone: the HOST picks by global hessian sum
    //    (rank-identical, so every rank builds/allreduces the same
    //    slot layout) and stages sums/bounds as usual.
    const bool last = depth + 1 >= max_depth;
    const bool dev_choice = (allreduce == nullptr) && !has_mono;
    std::vector<Node*> build;
    std::vector<int32_t> parent_slots;
    std::vector<Node*> subtracted;
    std::vector<int32_t> desc;  // [k][4] for HistTaskGenKernel
    if (!last) {
      desc.reserve(4 * n_expand);
      for (int i = 0; i < n_expand; ++i) {
        Node& ln = next_level[2 * i];
        Node& rn = next_level[2 * i + 1];
        parent_slots.push_back(expand[i]->hist_slot);
        desc.push_back(expand[i]->seg_begin);
        desc.push_back(expand[i]->seg_end);
        desc.push_back(i);  // counter slot
        if (dev_choice) {
          desc.push_back(2);  // device picks the smaller child
        } else {
          Node* small = (ln.hq <= rn.hq) ? &ln : &rn;
          Node* big = (small == &ln) ? &rn : &ln;
          small->hist_slot = (int)build.size();
          build.push_back(small);
          subtracted.push_back(big);
          desc.push_back(small == &ln ? 1 : 0);  // is_left
        }
      }
    }
    // capacity guards bound the HIST pool slots and the task-gen
    // LDS arrays — the final level only partitions, so up to
    // 2*max_nodes_level nodes may expand there without either
    if (!last && 2 * n_expand > 2 * max_nodes_level) return -9999;
    if (!last && n_expand > 2048) return -9998;  // task-gen LDS cap
    const int kb = n_expand;
    const bool use_ps = !last && dev_choice;

    auto set_child_segs = [&](const int32_t* fin) {
      for (int i = 0; i < n_expand; ++i) {
        Node& ln = next_level[2 * i];
        Node& rn = next_level[2 * i + 1];
        ln.seg_begin = expand[i]->seg_begin;
        ln.seg_end = fin[2 * i];  // begin + n_left
        rn.seg_begin = ln.seg_end;
        rn.seg_end = expand[i]->seg_end;
      }
    };

    // ---- ONE staging upload for the whole level ----
    // [partition tasks | feat | sbin | default_left | counters | desc |
    //  parent_slots | parent pair-sums]  + (device-only) ps region.
    // Counters live HERE (ring slot): partition atomics update them,
    // the task-gen kernel reads them, and the level sync reads them
    // back — no separate init copy, no persistent buffer.
    std::vector<BlockTask> ptasks;
    ChunkTasks(expand, &ptasks);
    const int k = n_expand;
    const int slot = ctx->ring.next();
    size_t off_feat = (ptasks.size() * sizeof(BlockTask) + 7) & ~7ULL;
    size_t off_sbin = (off_feat + (size_t)k * 4 + 7) & ~7ULL;
    size_t off_dl = (off_sbin + (size_t)k * 4 + 7) & ~7ULL;
    size_t off_cnt = (off_dl + (size_t)k + 7) & ~7ULL;
    size_t off_desc = (off_cnt + (size_t)k * 8 + 7) & ~7ULL;
    size_t desc_bytes = desc.empty() ? (size_t)k * 8 : desc.size() * 4;
    size_t off_pslots = (off_desc + desc_bytes + 7) & ~7ULL;
    size_t off_pps =
        (off_pslots + parent_slots.size() * 4 + 63) & ~63ULL;
    size_t upload_bytes =
        off_pps + (use_ps ? (size_t)kb * 2 * sizeof(int64_t) : 0);
    size_t off_ps = (upload_bytes + 63) & ~63ULL;
    size_t total_bytes =
        off_ps + (use_ps ? (size_t)kb * 4 * sizeof(int64_t) : 0);
    if (int e = ctx->ring.ensure(slot, total_bytes)) return e;
    char* h = (char*)ctx->ring.host[slot];
    memcpy(h, ptasks.data(), ptasks.size() * sizeof(BlockTask));
    {
      int32_t* feat = (int32_t*)(h + off_feat);
      int32_t* sbin = (int32_t*)(h + off_sbin);
      uint8_t* dl = (uint8_t*)(h + off_dl);
      int32_t* cnt = (int32_t*)(h + off_cnt);
      for (int i = 0; i < k; ++i) {
        Node* nd = expand[i];
        feat[i] = nd->feature;
        sbin[i] = nd->bin - cut_ptrs_host[nd->feature];
        dl[i] = (uint8_t)nd->dir;
        cnt[2 * i] = nd->seg_begin;
        cnt[2 * i + 1] = nd->seg_end;
      }
      if (!desc.empty()) {
        memcpy(h + off_desc, desc.data(), desc.size() * 4);
      } else if (last) {
        // final level: (left nid, right nid) per expand node for the
        // direct leaf-position write
        int32_t* kids = (int32_t*)(h + off_desc);
        for (int i = 0; i < k; ++i) {
          kids[2 * i] = next_level[2 * i].nid;
          kids[2 * i + 1] = next_level[2 * i + 1].nid;
        }
      }
      if (!parent_slots.empty()) {
        memcpy(h + off_pslots, parent_slots.data(),
               parent_slots.size() * 4);
      }
      if (use_ps) {
        int64_t* pp = (int64_t*)(h + off_pps);
        for (int i = 0; i < kb; ++i) {
          pp[2 * i] = expand[i]->gq;
          pp[2 * i + 1] = expand[i]->hq;
        }
      }
    }
    HIP_CHECK(hipMemcpyAsync(ctx->ring.dev[slot], h, upload_bytes,
                             hipMemcpyHostToDevice, stream));
    char* d = (char*)ctx->ring.dev[slot];
    int32_t* cnt_dev = (int32_t*)(d + off_cnt);
    int64_t* eval_ps = use_ps ? (int64_t*)(d + off_ps) : nullptr;
    const int64_t* parent_ps_dev =
        use_ps ? (const int64_t*)(d + off_pps) : nullptr;

    if (last) {
      // final level: the children are all leaves — write their
      // positions directly (one decide pass) instead of
      // partition + copy + counter sync + a later leaf sweep
      gbt_leaf_decide(gidx8, gidx16, n_features, gidx8_col, gidx16_col,
                      n_rows, cur_ridx,
                      (const BlockTask*)d,
                      (int)ptasks.size(), (const int32_t*)(d + off_feat),
                      (const int32_t*)(d + off_sbin),
                      (const uint8_t*)(d + off_dl),
                      (const int32_t*)(d + off_desc), n_bins_feat_dev,
                      pos_out, stream);
      level_nodes.clear();
      break;
    }
    gbt_partition(gidx8, gidx16, n_features, gidx8_col, gidx16_col,
                  n_rows, cur_ridx, alt_ridx,
                  (const BlockTask*)d, (int)ptasks.size(),
                  (const int32_t*)(d + off_feat),
                  (const int32_t*)(d + off_sbin),
                  (const uint8_t*)(d + off_dl), nullptr, nullptr,
                  n_bins_feat_dev, cnt_dev, stream);
    std::swap(cur_ridx, alt_ridx);  // children now live in cur_ridx
    // host bound on the device-generated task count
    long long bound_total = 0;
    for (Node* nd : expand) bound_total += nd->seg_end - nd->seg_begin;
    int max_tasks = (int)std::min<long long>(
        std::min<long long>(bound_total / hist_min_rows, hist_tasks) + kb + 1,
        hist_tasks_cap);
    hipLaunchKernelGGL(HistTaskGenKernel, dim3(1), dim3(256), 0, stream,
                       cnt_dev, (const int32_t*)(d + off_desc), kb,
                       hist_min_rows, hist_tasks, max_tasks, tg_scratch,
                       hist_tasks_dev, eval_ps, nullptr, nullptr, 0,
                       nullptr);
    HIP_CHECK(hipMemsetAsync(next_pool, 0,
                             (size_t)kb * hist_row * sizeof(int64_t), stream));
    gbt_hist(gidx8, gidx16, n_features, qgpair, cur_ridx, hist_tasks_dev,
             max_tasks, next_pool, n_bins, feat_group_start_dev,
             bin_group_start_dev, n_groups, max_group_bins, cut_ptrs_dev,
             use_shared, eval_ps, stream);
    if (allreduce) {
      allreduce((long long*)next_pool, (long long)kb * hist_row);
    }
    {
      int64_t* sub_out = next_pool + (long long)kb * hist_row;
      const long long total = (long long)kb * hist_row;
      int blocks = (int)std::min<long long>((total + 255) / 256, 4096);
      hipLaunchKernelGGL(SubtractHistKernel, dim3(blocks), dim3(256), 0,
                         stream, cur_pool, next_pool, sub_out,
                         (const int32_t*)(d + off_pslots), (int)hist_row,
                         kb, eval_ps, parent_ps_dev, nullptr);
      for (int i = 0; i < (int)subtracted.size(); ++i) {
        subtracted[i]->hist_slot = kb + i;
      }
    }
    std::vector<Node*> eval_nodes;
    if (!dev_choice) {
      eval_nodes.reserve(2 * kb);
      for (Node* b : build) eval_nodes.push_back(b);
      for (Node* s : subtracted) eval_nodes.push_back(s);
    }
    if (int e = evaluate_enqueue(2 * kb, dev_choice ? nullptr : &eval_nodes,
                                 next_pool, eval_ps)) return e;
    // ---- the ONE sync for this level ----
    const int64_t* best;
    const int32_t* cnt;
    if (int e = level_sync(2 * kb, n_expand, cnt_dev, &best, &cnt))
      return e;
    set_child_segs(cnt);
    if (dev_choice) {
      // recover the device's smaller-child choice from the counters
      // (same rule as HistTaskGenKernel) and map eval slots to nodes
      eval_nodes.resize(2 * kb);
      for (int i = 0; i < kb; ++i) {
        Node& ln = next_level[2 * i];
        Node& rn = next_level[2 * i + 1];
        const int nl = ln.seg_end - ln.seg_begin;
        const int nr = rn.seg_end - rn.seg_begin;
        Node* small = (nl <= nr) ? &ln : &rn;
        Node* big = (small == &ln) ? &rn : &ln;
        small->hist_slot = i;
        big->hist_slot = kb + i;
        eval_nodes[i] = small;
        eval_nodes[kb + i] = big;
      }
    }
    parse_best(eval_nodes, best);
    level_nodes.swap(next_level);
    std::swap(cur_pool, next_pool);
  }
  {
    const int parity = (cur_ridx == ridx) ? 0 : 1;
    for (auto& nd : level_nodes) {
      leaves.push_back({nd.nid, nd.seg_begin, nd.seg_end, parity});
    }
  }
  }  // end per-level mode

  // leaf values
  for (int nid = 0; nid < n_tree_nodes; ++nid) {
    if (out_left[nid] == -1) {
      out_split_cond[nid] = (float)(out_base_weight[nid] * eta);
    }
  }
  // leaf positions: one sweep per ping-pong parity (a leaf's rows sit
  // in whichever buffer was current when it stopped expanding)
  for (int par = 0; par < 2; ++par) {
    std::vector<const LeafSeg*> group;
    for (const auto& lf : leaves) {
      if (lf.parity == par) group.push_back(&lf);
    }
    if (group.empty()) continue;
    std::vector<BlockTask> tasks;
    std::vector<Node> lnodes(group.size());
    std::vector<Node*> lptrs(group.size());
    for (size_t i = 0; i < group.size(); ++i) {
      lnodes[i].seg_begin = group[i]->begin;
      lnodes[i].seg_end = group[i]->end;
      lptrs[i] = &lnodes[i];
    }
    ChunkTasks(lptrs, &tasks);
    const int slot = ctx->ring.next();
    size_t off_tasks = 0;
    size_t off_ids = (tasks.size() * sizeof(BlockTask) + 7) & ~7ULL;
    size_t bytes = off_ids + group.size() * sizeof(int32_t);
    if (int e = ctx->ring.ensure(slot, bytes)) return e;
    char* h = (char*)ctx->ring.host[slot];
    memcpy(h + off_tasks, tasks.data(), tasks.size() * sizeof(BlockTask));
    int32_t* ids = (int32_t*)(h + off_ids);
    for (size_t i = 0; i < group.size(); ++i) ids[i] = group[i]->nid;
    HIP_CHECK(hipMemcpyAsync(ctx->ring.dev[slot], h, bytes,
                             hipMemcpyHostToDevice, stream));
    char* d = (char*)ctx->ring.dev[slot];
    gbt_leaf_partition(par == 0 ? ridx : ridx_out,
                       (const BlockTask*)(d + off_tasks),
                       (int)tasks.size(), (const int32_t*)(d + off_ids),
                       pos_out, stream);
  }
  return n_tree_nodes;
}

}  // extern "C"
