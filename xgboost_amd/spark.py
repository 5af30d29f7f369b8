"""PySpark estimators are not available in this build (reference:
python-package/xgboost/spark — a thin layer over the same Booster).

The MI355X-native distributed path is one process per GPU with
torch.distributed over RCCL (see README "Distributed training"):

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \\
        --master-addr 127.0.0.1 your_train_script.py

Each rank shards its rows, builds a DMatrix and calls plain
``xgboost_amd.train`` — histograms are all-reduced per level and the
resulting model is identical on every rank.
"""


def _unavailable(*_args, **_kwargs):
    raise ImportError(
        "pyspark integration is not available in xgboost_amd; use the "
        "torch.distributed launcher documented in xgboost_amd.spark's "
        "module docstring instead.")


SparkXGBClassifier = _unavailable
SparkXGBRegressor = _unavailable
SparkXGBRanker = _unavailable
