"""Learning to rank with rank:ndcg (reference: demo/rank)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from xgboost_amd.sklearn import XGBRanker

rng = np.random.RandomState(0)
n, f, groups = 2000, 6, 100
X = rng.randn(n, f).astype(np.float32)
qid = np.repeat(np.arange(groups), n // groups)
rel = X[:, 0] + 0.3 * rng.randn(n)
y = np.zeros(n, np.float32)
for q in range(groups):
    m = qid == q
    y[m] = np.argsort(np.argsort(rel[m])) * 4 // m.sum()  # grades 0-3

rk = XGBRanker(n_estimators=30, max_depth=4, learning_rate=0.3)
rk.fit(X, y, qid=qid)
print("trained;", len(rk.get_booster().trees), "trees")
