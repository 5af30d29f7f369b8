"""Distributed data-parallel training over torch.distributed gloo
(reference analog: tests/cpp/collective/test_worker.h in-process workers;
our workers are real processes, world_size=2, loopback rendezvous)."""
import os
import pickle
import subprocess
import sys
import tempfile

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = r"""
import os, pickle, sys
import numpy as np
import torch
import torch.distributed as dist
sys.path.insert(0, os.environ["XGB_AMD_REPO"])
import xgboost_amd as xgb
from xgboost_amd import collective

rank = int(os.environ["RANK"])
world = int(os.environ["WORLD_SIZE"])
collective.init("gloo")

rng = np.random.RandomState(0)
n, f = 2000, 8
X = rng.randn(n, f).astype(np.float32)
w = rng.randn(f)
y = (X @ w + 0.3 * rng.randn(n) > 0).astype(np.float32)

# row shard
shard = slice(rank * n // world, (rank + 1) * n // world)
dtrain = xgb.DMatrix(X[shard], label=y[shard])
params = {"objective": "binary:logistic", "max_depth": 4, "eta": 0.3,
          "seed": 7, "debug_synchronize": True}
bst = xgb.train(params, dtrain, 5, verbose_eval=False)

# models must be identical across workers
raw = bytes(bst.save_raw("json"))
ref = collective.broadcast_obj(raw, 0)
assert raw == ref, "model differs across workers"

# predictions on the full data from the distributed model
dfull = xgb.DMatrix(X, label=y)
pred = bst.predict(dfull)
out = os.environ["XGB_AMD_OUT"]
if rank == 0:
    with open(out, "wb") as fh:
        pickle.dump({"pred": pred, "raw": raw}, fh)
dist.barrier()
dist.destroy_process_group()
"""




def _free_port() -> int:
    """OS-assigned free port: fixed ports collide when the suite runs
    under pytest-xdist (parallel workers) or right after a crashed run
    leaves TIME_WAIT sockets."""
    import socket
    with socket.socket() as sock:
        sock.bind(("127.0.0.1", 0))
        return sock.getsockname()[1]


def _run_workers(world_size: int, script: str, env_extra=None) -> None:
    procs = []
    port = _free_port()
    with tempfile.TemporaryDirectory() as td:
        spath = os.path.join(td, "worker.py")
        with open(spath, "w") as fh:
            fh.write(script)
        out = os.path.join(td, "out.pkl")
        for r in range(world_size):
            env = dict(os.environ)
            env.update({
                "RANK": str(r), "WORLD_SIZE": str(world_size),
                "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
                "XGB_AMD_REPO": REPO, "XGB_AMD_OUT": out,
            })
            if env_extra:
                env.update(env_extra)
            procs.append(subprocess.Popen(
                [sys.executable, spath], env=env,
                stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
        outputs = []
        ok = True
        for p in procs:
            stdout, _ = p.communicate(timeout=300)
            outputs.append(stdout.decode())
            ok = ok and p.returncode == 0
        assert ok, "worker failed:\n" + "\n---\n".join(outputs)
        with open(out, "rb") as fh:
            return pickle.load(fh)


def test_two_worker_training_sync():
    res = _run_workers(2, WORKER)
    # distributed model should learn the signal
    rng = np.random.RandomState(0)
    n, f = 2000, 8
    X = rng.randn(n, f).astype(np.float32)
    w = rng.randn(f)
    y = (X @ w + 0.3 * rng.randn(n) > 0).astype(np.float32)
    pred = res["pred"]
    acc = ((pred > 0.5) == y).mean()
    assert acc > 0.8, acc


HIST_WORKER = r"""
import os, sys
import numpy as np
import torch
sys.path.insert(0, os.environ["XGB_AMD_REPO"])
from xgboost_amd import collective
from xgboost_amd.data import DMatrix
from xgboost_amd.quantile import make_cuts, HistogramCuts
from xgboost_amd.data import quantize_dense
from xgboost_amd.backend.cpu import CpuOps, GradQuantizer

collective.init("gloo")
rank = collective.get_rank()
world = collective.get_world_size()

rng = np.random.RandomState(1)
n, f = 1000, 5
X = rng.randn(n, f).astype(np.float32)
gpair_full = torch.tensor(rng.randn(n, 2).astype(np.float32).clip(-1, 1))
gpair_full[:, 1] = gpair_full[:, 1].abs() + 0.1

cuts = make_cuts(X, 32)  # same cuts on all ranks

shard = slice(rank * n // world, (rank + 1) * n // world)
qm = quantize_dense(X[shard], cuts)
ops = CpuOps(qm)
gp = gpair_full[shard]
quant = GradQuantizer(gp)   # max-abs is allreduced -> same scale everywhere
qg = quant.quantize(gp)
m = qg.shape[0]
hist = ops.build_hist(qg, ops.make_ridx(m), [(0, m)])
ops.allreduce_hist(hist)

# oracle: single-process full-data histogram with the same (global) scale
qm_full = quantize_dense(X, cuts)
ops_full = CpuOps(qm_full)
qg_full = quant.quantize(gpair_full)
hist_full = ops_full.build_hist(qg_full, ops_full.make_ridx(n), [(0, n)])

assert torch.equal(hist, hist_full), "allreduced hist != full-data hist"
import torch.distributed as dist
dist.barrier()
dist.destroy_process_group()
"""


def test_hist_allreduce_exact():
    _run_workers_simple(2, HIST_WORKER)


def _run_workers_simple(world_size: int, script: str) -> None:
    procs = []
    port = _free_port()
    with tempfile.TemporaryDirectory() as td:
        spath = os.path.join(td, "worker.py")
        with open(spath, "w") as fh:
            fh.write(script)
        for r in range(world_size):
            env = dict(os.environ)
            env.update({
                "RANK": str(r), "WORLD_SIZE": str(world_size),
                "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
                "XGB_AMD_REPO": REPO,
            })
            procs.append(subprocess.Popen(
                [sys.executable, spath], env=env,
                stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
        outputs = []
        ok = True
        for p in procs:
            stdout, _ = p.communicate(timeout=300)
            outputs.append(stdout.decode())
            ok = ok and p.returncode == 0
        assert ok, "worker failed:\n" + "\n---\n".join(outputs)


def test_collective_compat_surface_single_process():
    """Reference collective API names exist and behave sanely without a
    process group (reference python-package/xgboost/collective.py)."""
    import numpy as np
    from xgboost_amd import collective as c
    a = c.allreduce(np.arange(3.0), c.Op.SUM)
    assert np.array_equal(a, np.arange(3.0))
    assert c.broadcast({"x": 1}, 0) == {"x": 1}
    assert isinstance(c.get_processor_name(), str)
    cfg = c.Config(tracker_host_ip="127.0.0.1")
    assert cfg.tracker_host_ip == "127.0.0.1"
    assert int(c.Op.MAX) == 0 and int(c.Op.SUM) == 2


_SPARSE_SCRIPT = """
import os, pickle, sys
import numpy as np
import torch.distributed as dist
sys.path.insert(0, os.environ["XGB_AMD_REPO"])
import xgboost_amd as xgb
from xgboost_amd import collective
from scipy import sparse as sp

rank = int(os.environ["RANK"])
world = int(os.environ["WORLD_SIZE"])
collective.init("gloo")

rng = np.random.RandomState(0)
n, cols, nnz_row = 3000, 500, 10
rows = np.repeat(np.arange(n), nnz_row)
cidx = rng.randint(0, cols, size=n * nnz_row)
X = sp.csr_matrix((np.ones(n * nnz_row, np.float32), (rows, cidx)),
                  shape=(n, cols))
y = (np.asarray(X[:, :5].sum(axis=1)).ravel() > 0).astype(np.float32)

shard = slice(rank * n // world, (rank + 1) * n // world)
dtrain = xgb.DMatrix(X[shard], label=y[shard])
params = {"objective": "binary:logistic", "max_depth": 4, "eta": 0.3,
          "seed": 3, "debug_synchronize": True}
bst = xgb.train(params, dtrain, 4, verbose_eval=False)
raw = bytes(bst.save_raw("json"))
ref = collective.broadcast_obj(raw, 0)
assert raw == ref, "sparse model differs across workers"
out = os.environ["XGB_AMD_OUT"]
if rank == 0:
    with open(out, "wb") as fh:
        pickle.dump({"raw": raw}, fh)
dist.barrier()
dist.destroy_process_group()
"""


def test_two_process_sparse_training():
    """Distributed quantized-CSR training: the per-rank column
    summaries are allgathered into identical cuts and the CSR histogram
    allreduce produces identical models on every worker (this test
    caught rank-local cuts breaking the collective)."""
    blob = _run_workers(2, _SPARSE_SCRIPT)
    assert len(blob["raw"]) > 100


_EXTMEM_SCRIPT = """
import os, pickle, sys
import numpy as np
import torch.distributed as dist
sys.path.insert(0, os.environ["XGB_AMD_REPO"])
import xgboost_amd as xgb
from xgboost_amd import collective

rank = int(os.environ["RANK"])
collective.init("gloo")


class It(xgb.DataIter):
    def __init__(self):
        super().__init__()
        self.i = 0

    def reset(self):
        self.i = 0

    def next(self, input_data):
        if self.i >= 3:
            return False
        rng = np.random.RandomState(100 * rank + self.i)
        Xb = rng.randn(1500, 6).astype(np.float32)
        yb = (Xb[:, 0] > 0).astype(np.float32)
        input_data(data=Xb, label=yb)
        self.i += 1
        return True


d = xgb.ExtMemQuantileDMatrix(It(), max_bin=64)
params = {"objective": "binary:logistic", "max_depth": 4, "eta": 0.3,
          "seed": 11, "debug_synchronize": True}
bst = xgb.train(params, d, 4, verbose_eval=False)
raw = bytes(bst.save_raw("json"))
ref = collective.broadcast_obj(raw, 0)
assert raw == ref, "extmem model differs across workers"
out = os.environ["XGB_AMD_OUT"]
if rank == 0:
    with open(out, "wb") as fh:
        pickle.dump({"raw": raw}, fh)
dist.barrier()
dist.destroy_process_group()
"""


def test_two_process_extmem_training():
    """Distributed external-memory training: per-rank DataIter batches,
    allgathered sketch summaries, identical models on every worker."""
    blob = _run_workers(2, _EXTMEM_SCRIPT)
    assert len(blob["raw"]) > 100


_MT_SCRIPT = """
import os, pickle, sys
import numpy as np
import torch.distributed as dist
sys.path.insert(0, os.environ["XGB_AMD_REPO"])
import xgboost_amd as xgb
from xgboost_amd import collective

rank = int(os.environ["RANK"])
world = int(os.environ["WORLD_SIZE"])
collective.init("gloo")

rng = np.random.RandomState(0)
n, f = 1600, 6
X = rng.randn(n, f).astype(np.float32)
Y = np.stack([X[:, 0] + 0.1 * rng.randn(n),
              X[:, 1] - X[:, 2] + 0.1 * rng.randn(n)], axis=1).astype(
    np.float32)
shard = slice(rank * n // world, (rank + 1) * n // world)
d = xgb.DMatrix(X[shard], label=Y[shard])
params = {"objective": "reg:squarederror", "max_depth": 4,
          "multi_strategy": "multi_output_tree", "seed": 2,
          "debug_synchronize": True}
bst = xgb.train(params, d, 3, verbose_eval=False)
raw = bytes(bst.save_raw("json"))
ref = collective.broadcast_obj(raw, 0)
assert raw == ref, "multi-target model differs across workers"
out = os.environ["XGB_AMD_OUT"]
if rank == 0:
    with open(out, "wb") as fh:
        pickle.dump({"raw": raw}, fh)
dist.barrier()
dist.destroy_process_group()
"""


def test_two_process_multi_target_training():
    """Distributed vector-leaf trees: per-target hist allreduce +
    summed-gain evaluation give identical models on every worker."""
    blob = _run_workers(2, _MT_SCRIPT)
    assert len(blob["raw"]) > 100


_APPROX_SCRIPT = """
import os, pickle, sys
import numpy as np
import torch.distributed as dist
sys.path.insert(0, os.environ["XGB_AMD_REPO"])
import xgboost_amd as xgb
from xgboost_amd import collective

rank = int(os.environ["RANK"])
world = int(os.environ["WORLD_SIZE"])
collective.init("gloo")

rng = np.random.RandomState(0)
n, f = 2000, 6
X = rng.randn(n, f).astype(np.float32)
y = (X[:, 0] + X[:, 1] ** 2 > 0.5).astype(np.float32)
shard = slice(rank * n // world, (rank + 1) * n // world)
d = xgb.DMatrix(X[shard], label=y[shard])
params = {"objective": "binary:logistic", "tree_method": "approx",
          "max_depth": 4, "seed": 5, "debug_synchronize": True}
bst = xgb.train(params, d, 3, verbose_eval=False)
raw = bytes(bst.save_raw("json"))
ref = collective.broadcast_obj(raw, 0)
assert raw == ref, "approx model differs across workers"
out = os.environ["XGB_AMD_OUT"]
if rank == 0:
    with open(out, "wb") as fh:
        pickle.dump({"raw": raw}, fh)
dist.barrier()
dist.destroy_process_group()
"""


def test_two_process_approx_training():
    """tree_method=approx re-sketches with hessian weights every
    iteration; the weighted summaries must merge across ranks so cuts
    (and models) stay identical."""
    blob = _run_workers(2, _APPROX_SCRIPT)
    assert len(blob["raw"]) > 100


_KITCHEN_SCRIPT = """
import os, pickle, sys
import numpy as np
import torch.distributed as dist
sys.path.insert(0, os.environ["XGB_AMD_REPO"])
import xgboost_amd as xgb
from xgboost_amd import collective

rank = int(os.environ["RANK"])
world = int(os.environ["WORLD_SIZE"])
collective.init("gloo")

rng = np.random.RandomState(0)
n, f = 2400, 8
X = rng.randn(n, f).astype(np.float32)
w = rng.randn(f)
y = (X @ w + 0.3 * rng.randn(n)).astype(np.float32)
shard = slice(rank * n // world, (rank + 1) * n // world)
d = xgb.DMatrix(X[shard], label=y[shard])
params = {"objective": "reg:squarederror", "grow_policy": "lossguide",
          "max_leaves": 24, "max_depth": 0, "colsample_bynode": 0.7,
          "subsample": 0.9, "monotone_constraints": "(1,0,0,0,0,0,0,0)",
          "seed": 9, "debug_synchronize": True}
bst = xgb.train(params, d, 4, verbose_eval=False)
raw = bytes(bst.save_raw("json"))
ref = collective.broadcast_obj(raw, 0)
assert raw == ref, "kitchen-sink model differs across workers"
out = os.environ["XGB_AMD_OUT"]
if rank == 0:
    with open(out, "wb") as fh:
        pickle.dump({"raw": raw}, fh)
dist.barrier()
dist.destroy_process_group()
"""


def test_two_process_lossguide_colsample_monotone():
    """The feature-rich python-driver paths (lossguide heap, column
    sampling, subsample, monotone bounds) must also be bit-identical
    across workers."""
    blob = _run_workers(2, _KITCHEN_SCRIPT)
    assert len(blob["raw"]) > 100


INTERCEPT_WORKER = r"""
import os, pickle, sys
import numpy as np
sys.path.insert(0, os.environ["XGB_AMD_REPO"])
import xgboost_amd as xgb
from xgboost_amd import collective

collective.init("gloo")
rank = collective.get_rank()
world = collective.get_world_size()

rng = np.random.RandomState(5)
n, f = 1200, 4
X = rng.randn(n, f).astype(np.float32)
# labels with a strongly rank-dependent mean: a local fit-stump would
# produce a different intercept on each shard
y = (np.arange(n) / n * 10.0 + rng.rand(n)).astype(np.float32)

shard = slice(rank * n // world, (rank + 1) * n // world)
dtrain = xgb.DMatrix(X[shard], label=y[shard])
params = {"objective": "reg:squarederror", "max_depth": 2, "eta": 0.1}
bst = xgb.train(params, dtrain, 1, verbose_eval=False)
base = float(bst.base_score)

if rank == 0:
    with open(os.environ["XGB_AMD_OUT"], "wb") as fh:
        pickle.dump({"base": base}, fh)
import torch.distributed as dist
dist.barrier()
dist.destroy_process_group()
"""


def test_two_process_global_intercept():
    """Distributed fit-stump must equal the single-process full-data
    intercept (reference GlobalSum in src/tree/fit_stump.cu:46-49)."""
    import xgboost_amd as xgb
    res = _run_workers(2, INTERCEPT_WORKER)
    rng = np.random.RandomState(5)
    n, f = 1200, 4
    X = rng.randn(n, f).astype(np.float32)
    y = (np.arange(n) / n * 10.0 + rng.rand(n)).astype(np.float32)
    solo = xgb.train({"objective": "reg:squarederror", "max_depth": 2,
                      "eta": 0.1}, xgb.DMatrix(X, label=y), 1,
                     verbose_eval=False)
    assert abs(res["base"] - float(solo.base_score)) < 1e-6, (
        res["base"], float(solo.base_score))


TRACKER_WORKER = r"""
import os, sys
import numpy as np
sys.path.insert(0, os.environ["XGB_AMD_REPO"])
import xgboost_amd as xgb
from xgboost_amd import collective

# NO RANK in the env: the worker claims one from the tracker's store
assert "RANK" not in os.environ
collective.init("gloo", timeout=60)
rank = collective.get_rank()
world = collective.get_world_size()
assert world == 2

rng = np.random.RandomState(0)
n, f = 1000, 6
X = rng.randn(n, f).astype(np.float32)
y = (X[:, 0] > 0).astype(np.float32)
shard = slice(rank * n // world, (rank + 1) * n // world)
bst = xgb.train({"objective": "binary:logistic", "max_depth": 3},
                xgb.DMatrix(X[shard], label=y[shard]), 3,
                verbose_eval=False)
raw = bytes(bst.save_raw("json"))
ref = collective.broadcast_obj(raw, 0)
assert raw == ref, "model differs across tracker-assigned ranks"
collective.finalize()
"""


def test_tracker_owned_rendezvous_and_rank_assignment():
    """Builder-owned tracker: workers arrive WITHOUT ranks, claim them
    from the tracker's store (reference tracker.cc rank assignment),
    train in sync, and wait_for() observes completion."""
    import xgboost_amd.tracker as tr
    tracker = tr.RabitTracker(n_workers=2)
    tracker.start()
    args = tracker.worker_args()
    procs = []
    with tempfile.TemporaryDirectory() as td:
        spath = os.path.join(td, "worker.py")
        with open(spath, "w") as fh:
            fh.write(TRACKER_WORKER)
        for _ in range(2):
            env = {k: v for k, v in os.environ.items() if k != "RANK"}
            env.update({str(k): str(v) for k, v in args.items()})
            env["XGB_AMD_REPO"] = REPO
            procs.append(subprocess.Popen(
                [sys.executable, spath], env=env,
                stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
        outs = [p.communicate(timeout=180)[0].decode() for p in procs]
        assert all(p.returncode == 0 for p in procs), "\n---\n".join(outs)
    tracker.wait_for(timeout=30)  # both workers posted completion


FAILURE_WORKER = r"""
import os, sys, time
import numpy as np
import torch
import torch.distributed as dist
sys.path.insert(0, os.environ["XGB_AMD_REPO"])
from xgboost_amd import collective

collective.init("gloo", timeout=10)
rank = collective.get_rank()
t = torch.ones(4)
dist.all_reduce(t)  # both alive: fine
if rank == 1:
    os._exit(17)  # die WITHOUT closing the group
# rank 0: the next collective must fail within the watchdog budget,
# not hang forever (reference watchdog + ncclCommAbort, coll.cu:157)
t0 = time.monotonic()
try:
    for _ in range(100):
        dist.all_reduce(t)
        time.sleep(0.2)
    print("SURVIVED-UNEXPECTEDLY", flush=True)
    sys.exit(3)
except Exception as e:  # noqa: BLE001
    el = time.monotonic() - t0
    print(f"DETECTED after {el:.1f}s: {type(e).__name__}", flush=True)
    assert el < 60, el
    try:
        collective.signal_error(f"peer failure: {type(e).__name__}")
    except RuntimeError:
        pass
    sys.exit(0)
"""


def test_collective_failure_detection():
    """A dead worker must surface as an error on the survivors within
    the watchdog budget instead of hanging the job."""
    procs = []
    port = _free_port()
    with tempfile.TemporaryDirectory() as td:
        spath = os.path.join(td, "worker.py")
        with open(spath, "w") as fh:
            fh.write(FAILURE_WORKER)
        for r in range(2):
            env = dict(os.environ)
            env.update({"RANK": str(r), "WORLD_SIZE": "2",
                        "MASTER_ADDR": "127.0.0.1",
                        "MASTER_PORT": str(port),
                        "XGB_AMD_REPO": REPO})
            procs.append(subprocess.Popen(
                [sys.executable, spath], env=env,
                stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
        out0, _ = procs[0].communicate(timeout=150)
        procs[1].wait(timeout=30)
        assert procs[1].returncode == 17
        assert procs[0].returncode == 0, out0.decode()
        assert b"DETECTED" in out0, out0.decode()


RANK_WORKER = r"""
import os, sys
import numpy as np
sys.path.insert(0, os.environ["XGB_AMD_REPO"])
import xgboost_amd as xgb
from xgboost_amd import collective

collective.init("gloo")
rank = collective.get_rank()
world = collective.get_world_size()
rng = np.random.RandomState(3)
n, f = 2000, 6
X = rng.randn(n, f).astype(np.float32)
y = np.clip((X[:, 0] * 2 + 2).astype(int), 0, 4).astype(np.float32)
qid = np.repeat(np.arange(40), n // 40)
# shard whole GROUPS per rank (ranking groups never straddle workers)
gmask = (qid % world) == rank
d = xgb.DMatrix(X[gmask], label=y[gmask], qid=qid[gmask])
bst = xgb.train({"objective": "rank:ndcg", "max_depth": 4, "eta": 0.3,
                 "debug_synchronize": True}, d, 5, verbose_eval=False)
raw = bytes(bst.save_raw("json"))
ref = collective.broadcast_obj(raw, 0)
assert raw == ref, "ranking model differs across workers"
if rank == 0:
    import pickle
    with open(os.environ["XGB_AMD_OUT"], "wb") as fh:
        pickle.dump({"ok": True}, fh)
import torch.distributed as dist
dist.barrier()
dist.destroy_process_group()
"""


def test_two_process_ranking_training():
    """Group-sharded lambdarank: gradients are group-local, histogram
    sync must still produce identical trees on every worker."""
    _run_workers(2, RANK_WORKER)


DART_WORKER = r"""
import os, pickle, sys
import numpy as np
sys.path.insert(0, os.environ["XGB_AMD_REPO"])
import xgboost_amd as xgb
from xgboost_amd import collective

collective.init("gloo")
rank = collective.get_rank()
world = collective.get_world_size()
rng = np.random.RandomState(3)
n, f = 1600, 5
X = rng.randn(n, f).astype(np.float32)
y = (X[:, 0] - 0.5 * X[:, 1] + 0.1 * rng.randn(n)).astype(np.float32)
sl = slice(rank * n // world, (rank + 1) * n // world)
d = xgb.DMatrix(X[sl], label=y[sl])
bst = xgb.train({"max_depth": 3, "eta": 0.3, "rate_drop": 0.4,
                 "one_drop": True, "seed": 11}, d, 8)
with open(os.environ["XGB_AMD_OUT"] + f".{rank}", "wb") as fh:
    pickle.dump({"dump": bst.get_dump(with_stats=True),
                 "wd": bst.weight_drop}, fh)
collective.finalize()
"""


def test_two_process_dart_identical_models():
    """DART under data-parallel training: the drop set comes from a
    seed-deterministic RNG and tree counts, both rank-identical, so
    every rank must produce the SAME model and weights."""
    port = _free_port()
    with tempfile.TemporaryDirectory() as td:
        spath = os.path.join(td, "worker.py")
        with open(spath, "w") as fh:
            fh.write(DART_WORKER)
        out = os.path.join(td, "out.pkl")
        procs = []
        for r in range(2):
            env = dict(os.environ)
            env.update({
                "RANK": str(r), "WORLD_SIZE": "2",
                "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
                "XGB_AMD_REPO": REPO, "XGB_AMD_OUT": out,
            })
            procs.append(subprocess.Popen(
                [sys.executable, spath], env=env,
                stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
        outputs = []
        ok = True
        for p in procs:
            stdout, _ = p.communicate(timeout=300)
            outputs.append(stdout.decode())
            ok = ok and p.returncode == 0
        assert ok, "worker failed:\n" + "\n---\n".join(outputs)
        with open(out + ".0", "rb") as fh:
            r0 = pickle.load(fh)
        with open(out + ".1", "rb") as fh:
            r1 = pickle.load(fh)
    assert r0["dump"] == r1["dump"]
    assert r0["wd"] == pytest.approx(r1["wd"])
    assert any(w != 1.0 for w in r0["wd"])


def test_tracker_done_and_error_paths():
    """RabitTracker bookkeeping: wait_for returns once every worker
    posts done; a posted error surfaces via .error and makes wait_for
    raise (reference collective tracker error port semantics)."""
    from xgboost_amd.tracker import (RabitTracker, connect_tracker,
                                     post_done, post_error)
    t = RabitTracker(n_workers=2)
    t.start()
    args = t.worker_args()
    stores = [connect_tracker(args["DMLC_TRACKER_URI"],
                              args["DMLC_TRACKER_PORT"], 2, rank=None,
                              timeout_s=30)
              for _ in range(2)]
    ranks = sorted(r for _, r in stores)
    assert ranks == [0, 1]  # tracker-assigned ranks
    for st, _ in stores:
        post_done(st)
    t.wait_for(timeout=30)  # returns, no raise
    assert t.error() is None
    t.free()

    t2 = RabitTracker(n_workers=2)
    t2.start()
    a2 = t2.worker_args()
    s0, _ = connect_tracker(a2["DMLC_TRACKER_URI"],
                            a2["DMLC_TRACKER_PORT"], 2, rank=None,
                            timeout_s=30)
    post_error(s0, "worker exploded")
    with pytest.raises(RuntimeError, match="worker exploded"):
        t2.wait_for(timeout=10)
    assert "exploded" in (t2.error() or "")
    t2.free()


WORLD4_WORKER = r"""
import os, pickle, sys
import numpy as np
sys.path.insert(0, os.environ["XGB_AMD_REPO"])
import xgboost_amd as xgb
from xgboost_amd import collective
collective.init("gloo")
rank, world = collective.get_rank(), collective.get_world_size()
rng = np.random.RandomState(0)
n, f = 2000, 6
X = rng.randn(n, f).astype(np.float32)
y = (X[:, 0] - X[:, 1] > 0).astype(np.float32)
sl = slice(rank * n // world, (rank + 1) * n // world)
bst = xgb.train({"objective": "binary:logistic", "max_depth": 4,
                 "seed": 3, "debug_synchronize": True},
                xgb.DMatrix(X[sl], label=y[sl]), 5)
if rank == 0:
    with open(os.environ["XGB_AMD_OUT"], "wb") as fh:
        pickle.dump(bst.get_dump(), fh)
collective.finalize()
"""


def test_four_process_training():
    """World size 4 (beyond the usual 2): rank layout generality +
    debug_synchronize's per-iteration tree-equality allreduce check."""
    dump = _run_workers(4, WORLD4_WORKER)
    assert len(dump) == 5
