"""Vector-leaf multi-target regression (multi_strategy=multi_output_tree,
reference: demo/guide-python/multioutput_regression.py)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import xgboost_amd as xgb

rng = np.random.RandomState(0)
X = rng.randn(4000, 8).astype(np.float32)
W = rng.randn(8, 3)
Y = (X @ W + 0.1 * rng.randn(4000, 3)).astype(np.float32)

d = xgb.DMatrix(X, label=Y)
bst = xgb.train({"objective": "reg:squarederror", "max_depth": 5,
                 "multi_strategy": "multi_output_tree", "eta": 0.3},
                d, 30, evals=[(d, "train")], verbose_eval=10)
print("one tree per round, vector leaves:", len(bst.trees), "trees")
print("pred shape:", bst.predict(d).shape)
