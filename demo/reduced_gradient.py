"""Multi-output regression with a REDUCED split gradient.

Experimental class-based objective interface (xgboost_amd.objective):
the full (n, n_targets) gradient values the vector leaves while a
reduced 1-column gradient finds the tree structure — one tree per
iteration instead of n_targets trees.  Reference analog:
demo/guide-python/multioutput_reduced_gradient.py +
XGBoosterTrainOneIterWithSplitGrad.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import numpy as np

import xgboost_amd as xgb
from xgboost_amd.objective import TreeObjective


class ReducedSquaredError(TreeObjective):
    def __call__(self, iteration, y_pred, dtrain):
        y = dtrain.get_label().reshape(y_pred.shape)
        return y_pred - y, np.ones_like(y_pred)

    def split_grad(self, iteration, grad, hess):
        # structure from the target-summed gradient
        return grad.sum(axis=1, keepdims=True), hess.sum(
            axis=1, keepdims=True)


def main():
    rng = np.random.RandomState(0)
    X = rng.randn(4096, 8).astype(np.float32)
    W = rng.randn(8, 3).astype(np.float32)
    Y = (X @ W + 0.1 * rng.randn(4096, 3)).astype(np.float32)
    d = xgb.DMatrix(X, label=Y)
    bst = xgb.train({"max_depth": 5, "eta": 0.3, "num_target": 3,
                     "base_score": 0.0}, d, 32, obj=ReducedSquaredError())
    pred = bst.predict(d)
    rmse = float(np.sqrt(((pred - Y) ** 2).mean()))
    print(f"trees: {len(bst.trees)} (one vector-leaf tree/iter), "
          f"rmse: {rmse:.4f}")


if __name__ == "__main__":
    main()
