"""SHAP contributions and interactions (reference: gpu SHAP demo)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import xgboost_amd as xgb

rng = np.random.RandomState(0)
X = rng.randn(500, 6).astype(np.float32)
y = (X[:, 0] * X[:, 1] + X[:, 2]).astype(np.float32)
d = xgb.DMatrix(X, label=y)
bst = xgb.train({"objective": "reg:squarederror", "max_depth": 4}, d, 20,
                verbose_eval=False)
contribs = bst.predict(d, pred_contribs=True)          # [n, f+1]
margin = bst.predict(d, output_margin=True)
print("efficiency check:", np.abs(contribs.sum(1) - margin).max())
inter = bst.predict(d, pred_interactions=True)         # [n, f+1, f+1]
print("strongest interaction pair:",
      np.unravel_index(np.abs(inter.mean(0)[:-1, :-1]
                              - np.diag(np.diag(inter.mean(0)[:-1, :-1]))
                              ).argmax(), (6, 6)))
