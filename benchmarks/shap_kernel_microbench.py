"""Microbenchmark for the SHAP path kernel (temporary perf tooling)."""
import time

import numpy as np
import torch

import xgboost_amd.ops as hip_ops


def run(n_rows, n_paths, M, iters=3):
    lib = hip_ops.load()
    dev = "cuda"
    F = 28
    rng = np.random.RandomState(0)
    X = torch.from_numpy(rng.randn(F, n_rows).astype(np.float32)).to(dev)
    pp = np.arange(n_paths + 1, dtype=np.int64) * M
    pg = np.zeros(n_paths, dtype=np.int32)
    ef = rng.randint(0, F, size=n_paths * M).astype(np.int32)
    elo = np.full(n_paths * M, -0.5, np.float32)
    ehi = np.full(n_paths * M, 0.5, np.float32)
    em = np.ones(n_paths * M, np.uint8)
    ez = np.full(n_paths * M, 0.5, np.float64)
    rz = 1.0 / ez
    pv = rng.randn(n_paths).astype(np.float64)
    t = {k: torch.from_numpy(v).to(dev) for k, v in
         dict(pp=pp, pg=pg, ef=ef, elo=elo, ehi=ehi, em=em, ez=ez,
              rz=rz, pv=pv).items()}
    phi = torch.zeros(1, F + 1, n_rows, dtype=torch.float64, device=dev)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        lib.gbt_shap_paths(
            hip_ops.ptr(X), n_rows, F, 0.0, 1, hip_ops.ptr(t["pp"]),
            hip_ops.ptr(t["pg"]), hip_ops.ptr(t["ef"]), hip_ops.ptr(t["elo"]),
            hip_ops.ptr(t["ehi"]), hip_ops.ptr(t["em"]), hip_ops.ptr(t["ez"]),
            hip_ops.ptr(t["rz"]), hip_ops.ptr(t["pv"]), n_paths, 1, F + 1,
            hip_ops.ptr(phi), hip_ops.stream())
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    print(f"rows={n_rows} paths={n_paths} M={M}: {dt*1e3:9.2f} ms "
          f"({n_rows*n_paths/dt/1e9:.2f} G pair/s)")
    return dt


if __name__ == "__main__":
    run(1_000_000, 1000, 6)
    run(1_000_000, 8000, 6)
    run(1_000_000, 32000, 6)
    run(1_000_000, 32000, 1)
    run(100_000, 32000, 6)
