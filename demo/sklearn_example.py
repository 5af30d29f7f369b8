"""sklearn API (reference: demo/guide-python/sklearn_examples.py)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from xgboost_amd.sklearn import XGBClassifier

rng = np.random.RandomState(0)
X = rng.randn(3000, 10).astype(np.float32)
y = (X[:, :3].sum(1) > 0).astype(int)

clf = XGBClassifier(n_estimators=200, max_depth=4, learning_rate=0.3,
                    early_stopping_rounds=10)
clf.fit(X[:2500], y[:2500], eval_set=[(X[2500:], y[2500:])], verbose=False)
print("best_iteration:", clf.best_iteration)
print("test acc:", clf.score(X[2500:], y[2500:]))
print("top features:", np.argsort(-clf.feature_importances_)[:3])
