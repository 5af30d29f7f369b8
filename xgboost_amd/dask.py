"""Dask integration stub.

The reference ships xgboost.dask (dask/distributed cluster training).
This image has no dask; the equivalent capability here is one process
per GPU via torchrun (RCCL over xGMI) — see demo/distributed_training.py
and README "Distributed training".  If dask is installed, a thin
adapter could map partitions to ranks; until then importing this module
gives a clear error instead of a silent fallback.
"""


def _unavailable(*_args, **_kwargs):
    raise ImportError(
        "xgboost_amd.dask requires the `dask` package, which is not "
        "installed in this environment.  Use torch.distributed data "
        "parallelism instead: launch one process per GPU with\n"
        "  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \\\n"
        "      --master-addr 127.0.0.1 your_script.py\n"
        "and call xgboost_amd.collective.init() in each process "
        "(see demo/distributed_training.py).")


DaskDMatrix = _unavailable
DaskQuantileDMatrix = _unavailable
train = _unavailable
predict = _unavailable


class DaskXGBClassifier:
    def __init__(self, *a, **k):
        _unavailable()


class DaskXGBRegressor:
    def __init__(self, *a, **k):
        _unavailable()
