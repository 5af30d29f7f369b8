"""Data-parallel multi-GPU training over RCCL/xGMI.

Launch with one rank per GPU:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 demo/distributed_training.py

Each rank holds a row shard; quantile sketches are merged across ranks
and per-level histograms all-reduced with RCCL (reference analog: dask
distributed training over NCCL/Rabit)."""
import os
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch
import xgboost_amd as xgb
from xgboost_amd import collective

rank = int(os.environ.get("RANK", "0"))
world = int(os.environ.get("WORLD_SIZE", "1"))
if torch.cuda.is_available():
    torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
if world > 1:
    collective.init()  # nccl (RCCL) when GPUs present, else gloo

rng = np.random.RandomState(1234 + rank)   # each rank: its own shard
X = rng.randn(250_000, 28).astype(np.float32)
w = np.random.RandomState(0).randn(28)     # same signal on every rank
y = (X @ w > 0).astype(np.float32)

dtrain = xgb.DMatrix(X, label=y)
bst = xgb.train({"objective": "binary:logistic", "max_depth": 8,
                 "device": "cuda" if torch.cuda.is_available() else "cpu",
                 "seed": 7}, dtrain, 50, verbose_eval=False)
if rank == 0:
    pred = bst.predict(dtrain)
    print("rank0 shard accuracy:", ((pred > 0.5) == y).mean())
    bst.save_model("distributed_model.json")
if world > 1:
    collective.finalize()
