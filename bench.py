#!/usr/bin/env python
"""Flagship benchmark — BASELINE.json config: Higgs-1M-shaped synthetic
(1e6 x 28 f32), hist, max_depth=8, max_bin=256, binary:logistic.

Metric: boosting rounds/sec (whole job).  Weak scaling: every rank holds
its own 1M-row shard; histograms are merged with RCCL allreduce over
xGMI (torch.distributed backend "nccl" on ROCm).

Usage (driver contract):
  python bench.py --gpus N --steps K --warmup W
  torchrun --nnodes=1 --nproc-per-node N bench.py --gpus N --steps K --warmup W
"""
import argparse
import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import xgboost_amd as xgb  # noqa: E402
from xgboost_amd import collective  # noqa: E402


def make_higgs_like(n_rows: int, n_features: int = 28, seed: int = 0):
    """Synthetic data with Higgs-like shape and a learnable nonlinear signal."""
    g = torch.Generator().manual_seed(seed)
    X = torch.randn(n_rows, n_features, generator=g)
    w1 = torch.randn(n_features, generator=g) / np.sqrt(n_features)
    w2 = torch.randn(n_features, generator=g) / np.sqrt(n_features)
    logits = X @ w1 + 0.7 * (X * X) @ w2 + 0.5 * X[:, 0] * X[:, 1]
    y = (logits + 0.5 * torch.randn(n_rows, generator=g) > 0).float()
    return X.numpy().astype(np.float32), y.numpy().astype(np.float32)


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--rows", type=int, default=1_000_000)
    ap.add_argument("--features", type=int, default=28)
    ap.add_argument("--max-depth", type=int, default=8)
    ap.add_argument("--max-bin", type=int, default=256)
    args = ap.parse_args()

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    has_gpu = torch.cuda.is_available()
    if has_gpu:
        torch.cuda.set_device(local_rank % max(1, torch.cuda.device_count()))
    if world_size > 1:
        collective.init("nccl" if has_gpu else "gloo")

    n_rows = args.rows if has_gpu else min(args.rows, 100_000)
    X, y = make_higgs_like(n_rows, args.features, seed=1234 + rank)
    dtrain = xgb.DMatrix(X, label=y)

    params = {
        "objective": "binary:logistic",
        "max_depth": args.max_depth,
        "max_bin": args.max_bin,
        "eta": 0.1,
        "tree_method": "hist",
        "device": "cuda" if has_gpu else "cpu",
        "seed": 7,
    }
    bst = xgb.Booster(params, cache=[dtrain])

    it = 0
    for _ in range(args.warmup):
        bst.update(dtrain, it)
        it += 1

    def barrier_sync():
        if collective.is_distributed():
            collective.barrier()
        if has_gpu:
            torch.cuda.synchronize()

    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        bst.update(dtrain, it)
        it += 1
    barrier_sync()
    elapsed = time.perf_counter() - t0
    # MAX over ranks == max elapsed -> min rounds/sec
    if collective.is_distributed():
        elapsed = collective.allreduce_max_scalars([elapsed])[0]

    # correctness: AUC on the training shard (outside timed region)
    from xgboost_amd.metrics import create_metric
    pred = bst.predict(dtrain)
    auc = create_metric("auc")(pred, dtrain.info)

    value = args.steps / elapsed
    if rank == 0:
        out = {
            "metric": "boosting_rounds_per_sec",
            "value": value,
            "unit": "rounds/s",
            "n_gpus": world_size if has_gpu else 0,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": "higgs-1m-hist",
                "rows_per_gpu": n_rows,
                "n_features": args.features,
                "max_depth": args.max_depth,
                "max_bin": args.max_bin,
                "objective": "binary:logistic",
                "parallelism": f"dp{world_size}",
                "train_auc": round(float(auc), 5),
                "device": params["device"],
            },
        }
        print(json.dumps(out))
    if collective.is_distributed():
        collective.finalize()


if __name__ == "__main__":
    main()
