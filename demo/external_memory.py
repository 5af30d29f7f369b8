"""External-memory training (reference: demo/guide-python/
external_memory.py): quantized pages in pinned host memory, streamed."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import xgboost_amd as xgb
from xgboost_amd.extmem import DataIter, ExtMemQuantileDMatrix


class BatchIter(DataIter):
    def __init__(self, n_batches=8, rows=50_000, cols=20):
        super().__init__()
        self.n_batches, self.rows, self.cols = n_batches, rows, cols
        self.i = 0

    def reset(self):
        self.i = 0

    def next(self, input_data):
        if self.i >= self.n_batches:
            return False
        rng = np.random.RandomState(self.i)
        X = rng.randn(self.rows, self.cols).astype(np.float32)
        y = (X[:, 0] > 0).astype(np.float32)
        input_data(data=X, label=y)
        self.i += 1
        return True


dtrain = ExtMemQuantileDMatrix(BatchIter(), max_bin=256)
print("rows:", dtrain.num_row(), "pages:", len(dtrain.pages))
bst = xgb.train({"objective": "binary:logistic", "max_depth": 6}, dtrain, 20,
                evals=[(dtrain, "train")], verbose_eval=10)
