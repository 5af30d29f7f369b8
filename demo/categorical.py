"""Categorical features from pandas (reference: demo/guide-python/
categorical.py)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import pandas as pd
import xgboost_amd as xgb

rng = np.random.RandomState(0)
n = 3000
df = pd.DataFrame({
    "color": pd.Categorical(rng.choice(["red", "green", "blue"], n)),
    "size": rng.randn(n).astype(np.float32),
})
y = ((df["color"] == "red").to_numpy() * 2.0
     + df["size"].to_numpy() + 0.1 * rng.randn(n)).astype(np.float32)

d = xgb.DMatrix(df, label=y, enable_categorical=True)
bst = xgb.train({"objective": "reg:squarederror", "max_depth": 4}, d, 20,
                evals=[(d, "train")], verbose_eval=10)
print(bst.get_dump()[0][:400])
