"""Basic training (reference: demo/guide-python/basic_walkthrough.py)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import xgboost_amd as xgb

rng = np.random.RandomState(7)
X = rng.randn(5000, 10).astype(np.float32)
y = (X[:, 0] * X[:, 1] + X[:, 2] > 0).astype(np.float32)

dtrain = xgb.DMatrix(X[:4000], label=y[:4000])
dtest = xgb.DMatrix(X[4000:], label=y[4000:])

params = {"objective": "binary:logistic", "max_depth": 5, "eta": 0.3,
          "eval_metric": ["logloss", "auc"], "device": "cpu"}
bst = xgb.train(params, dtrain, 30, evals=[(dtrain, "train"), (dtest, "test")],
                verbose_eval=10)
pred = bst.predict(dtest)
print("accuracy:", ((pred > 0.5) == y[4000:]).mean())

bst.save_model("model.json")
bst2 = xgb.Booster(model_file="model.json")
assert np.allclose(bst2.predict(dtest), pred)
print("model round-trip OK")
