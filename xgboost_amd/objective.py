"""Experimental class-based objective interface.

Mirror of the reference's `xgboost.objective` module (python-package/
xgboost/objective.py, added 3.2.0): objectives as classes called with
``(iteration, y_pred, dtrain)`` where ``y_pred`` is the RAW margin, and
``TreeObjective`` optionally supplying a REDUCED gradient for finding
tree structure (``split_grad``) while the full-width gradient values the
leaves (reference c_api.cc:1237 XGBoosterTrainOneIterWithSplitGrad,
gbtree.cc:191 HasValueGrad).  Reduced gradients require vector-leaf
trees: the booster must be configured with ``num_target`` matching the
full gradient's width.

.. warning::
   Experimental, like the reference module: the interface may change.
"""
from abc import ABC, abstractmethod
from typing import Optional, Tuple

__all__ = ["Objective", "TreeObjective"]


class Objective(ABC):
    """Base class for custom objective functions.

    ``__call__(iteration, y_pred, dtrain)`` returns ``(grad, hess)``
    shaped ``(n_samples, n_targets)`` (or flat for single target).
    """

    @abstractmethod
    def __call__(self, iteration: int, y_pred, dtrain) -> Tuple:
        ...


class TreeObjective(Objective):
    """Tree-specific objective: ``split_grad`` may return a reduced
    ``(grad, hess)`` used only to FIND the tree structure; the full
    gradient from ``__call__`` then values the (vector) leaves.  Return
    ``None`` to use the full gradient for both (the default)."""

    def split_grad(self, iteration: int, grad, hess) -> Optional[Tuple]:
        return None
