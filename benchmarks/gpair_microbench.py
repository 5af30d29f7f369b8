"""Microbenchmark for the fused gpair kernel (temporary perf tooling):
isolates whether GpairKernel's ~93us comes from expf (logistic) or the
maxabs reduction / write pattern (both objectives share those)."""
import time

import torch

import xgboost_amd.ops as hip_ops


def main():
    lib = hip_ops.load()
    n = 1_000_000
    dev = "cuda"
    torch.manual_seed(0)
    margin = torch.randn(n, device=dev)
    labels = (torch.rand(n, device=dev) > 0.5).float()
    weights = torch.ones(n, device=dev)
    gh = torch.empty(n, 2, device=dev)
    maxabs = torch.zeros(2, device=dev)

    def run(obj, iters=200):
        for _ in range(20):
            lib.gbt_gpair_fused(obj, hip_ops.ptr(margin),
                                hip_ops.ptr(labels), hip_ops.ptr(weights),
                                1.0, n, hip_ops.ptr(gh),
                                hip_ops.ptr(maxabs), hip_ops.stream())
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            lib.gbt_gpair_fused(obj, hip_ops.ptr(margin),
                                hip_ops.ptr(labels), hip_ops.ptr(weights),
                                1.0, n, hip_ops.ptr(gh),
                                hip_ops.ptr(maxabs), hip_ops.stream())
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters * 1e6

    print(f"logistic (expf):      {run(0):8.1f} us")
    print(f"squarederror (no exp):{run(1):8.1f} us")
    # no-weight variant (skips one 4MB read)
    def run_nw(obj, iters=200):
        for _ in range(20):
            lib.gbt_gpair_fused(obj, hip_ops.ptr(margin),
                                hip_ops.ptr(labels), None, 1.0, n,
                                hip_ops.ptr(gh), hip_ops.ptr(maxabs),
                                hip_ops.stream())
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            lib.gbt_gpair_fused(obj, hip_ops.ptr(margin),
                                hip_ops.ptr(labels), None, 1.0, n,
                                hip_ops.ptr(gh), hip_ops.ptr(maxabs),
                                hip_ops.stream())
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters * 1e6

    print(f"logistic no-weight:   {run_nw(0):8.1f} us")
    print(f"sqerr no-weight:      {run_nw(1):8.1f} us")


if __name__ == "__main__":
    main()
