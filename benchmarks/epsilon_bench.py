#!/usr/bin/env python
"""BASELINE config 3: Epsilon-scale dense synthetic (wide: 2000
features), hist, RCCL histogram allreduce when launched under torchrun.

Full config is 5e7 rows x 2000 f32 across 8 GPUs (weak scaling:
6.25e6 rows/GPU = 50 GB f32 + 12.5 GB u8 bins per GPU); --rows scales
it down for single-GPU smoke runs.
"""
import argparse
import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import xgboost_amd as xgb  # noqa: E402
from xgboost_amd import collective  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=6_250_000)  # 5e7 / 8 GPUs
    ap.add_argument("--features", type=int, default=2000)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--max-depth", type=int, default=8)
    ap.add_argument("--max-bin", type=int, default=256)
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    has_gpu = torch.cuda.is_available()
    if has_gpu:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0"))
                              % max(1, torch.cuda.device_count()))
    if world > 1:
        collective.init("nccl" if has_gpu else "gloo")
    n = args.rows if has_gpu else 10_000
    f = args.features if has_gpu else min(args.features, 200)
    # generate on GPU (device ingestion path) to avoid 50 GB host numpy
    gen = torch.Generator(device="cuda" if has_gpu else "cpu")
    gen.manual_seed(99 + rank)
    dev = "cuda" if has_gpu else "cpu"
    # 6.25e6 x 2000 f32 = 50 GB generated and kept ON DEVICE (288 GB
    # HBM); labels computed in chunks to bound the matmul workspace
    X = torch.empty(n, f, device=dev)
    w = torch.randn(f, generator=gen, device=dev) / np.sqrt(f)
    y = torch.empty(n, device=dev)
    chunk = 1_000_000
    for s0 in range(0, n, chunk):
        e0 = min(s0 + chunk, n)
        X[s0:e0] = torch.randn(e0 - s0, f, generator=gen, device=dev)
        y[s0:e0] = (X[s0:e0] @ w
                    + 0.3 * torch.randn(e0 - s0, generator=gen, device=dev))
    y = (y > 0).float().cpu().numpy()
    d = xgb.DMatrix(X if has_gpu else X.cpu().numpy(), label=y)
    bst = xgb.Booster({"objective": "binary:logistic",
                       "max_depth": args.max_depth, "max_bin": args.max_bin,
                       "eta": 0.1, "device": dev, "seed": 5}, cache=[d])
    it = 0
    for _ in range(args.warmup):
        bst.update(d, it)
        it += 1
    if collective.is_distributed():
        collective.barrier()
    if has_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        bst.update(d, it)
        it += 1
    if collective.is_distributed():
        collective.barrier()
    if has_gpu:
        torch.cuda.synchronize()
    el = time.perf_counter() - t0
    if collective.is_distributed():
        el = collective.allreduce_max_scalars([el])[0]
    from xgboost_amd.metrics import create_metric
    auc = create_metric("auc")(bst.predict(d), d.info)
    if rank == 0:
        print(json.dumps({
            "metric": "boosting_rounds_per_sec", "value": args.steps / el,
            "unit": "rounds/s", "n_gpus": world if has_gpu else 0,
            "steps": args.steps, "warmup": args.warmup,
            "ms_per_step": el / args.steps * 1000, "higher_is_better": True,
            "scaling": "weak", "dtype": "fp32", "data": "synthetic",
            "config": {"model": "epsilon-dense", "rows_per_gpu": n,
                       "n_features": f, "max_depth": args.max_depth,
                       "max_bin": args.max_bin,
                       "parallelism": f"dp{world}",
                       "train_auc": round(float(auc), 5)},
        }))
    if collective.is_distributed():
        collective.finalize()


if __name__ == "__main__":
    main()
