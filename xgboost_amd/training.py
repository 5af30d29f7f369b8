"""train() / cv() — the top-level training loop.

Reference behavior: python-package/xgboost/training.py:53 (train),
:300+ (cv with folds).
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional, Sequence, Tuple

import numpy as np

from .callback import (CallbackContainer, EarlyStopping, EvaluationMonitor,
                       TrainingCallback)
from .core import Booster
from .data import DMatrix


def train(params: Dict[str, Any], dtrain: DMatrix,
          num_boost_round: int = 10,
          evals: Optional[Sequence[Tuple[DMatrix, str]]] = None,
          obj=None, feval=None, maximize=None,
          early_stopping_rounds: Optional[int] = None,
          evals_result: Optional[dict] = None,
          verbose_eval: Any = True,
          xgb_model: Optional[Booster] = None,
          callbacks: Optional[Sequence[TrainingCallback]] = None,
          custom_metric=None) -> Booster:
    callbacks = list(callbacks) if callbacks else []
    evals = list(evals) if evals else []
    metric_fn = custom_metric or feval

    if early_stopping_rounds is not None and not any(
            isinstance(c, EarlyStopping) for c in callbacks):
        callbacks.append(EarlyStopping(rounds=early_stopping_rounds,
                                       maximize=maximize))
    if verbose_eval:
        period = verbose_eval if isinstance(verbose_eval, int) \
            and not isinstance(verbose_eval, bool) else 1
        callbacks.append(EvaluationMonitor(period=period))

    if xgb_model is not None:
        bst = xgb_model.copy() if isinstance(xgb_model, Booster) else \
            Booster(params, model_file=xgb_model)
        bst.set_param(params)
    else:
        bst = Booster(params, cache=[dtrain] + [d for d, _ in evals])

    cb = CallbackContainer(callbacks, metric=metric_fn)
    bst = cb.before_training(bst)
    if params.get("process_type") == "update":
        start = 0  # updaters revisit existing iterations
        num_boost_round = min(num_boost_round, bst.num_boosted_rounds())
    else:
        start = bst.num_boosted_rounds()
    for i in range(start, start + num_boost_round):
        if cb.before_iteration(bst, i, dtrain, evals):
            break
        bst.update(dtrain, i, fobj=obj)
        if cb.after_iteration(bst, i, dtrain, evals):
            break
    bst = cb.after_training(bst)
    if evals_result is not None:
        evals_result.update(cb.history)
    return bst


class CVPack:
    def __init__(self, dtrain, dtest, params):
        self.dtrain = dtrain
        self.dtest = dtest
        self.watchlist = [(dtrain, "train"), (dtest, "test")]
        self.bst = Booster(params, cache=[dtrain, dtest])

    def update(self, iteration, fobj):
        self.bst.update(self.dtrain, iteration, fobj)

    def eval(self, iteration, feval):
        return self.bst.eval_set(self.watchlist, iteration, feval)


def mknfold(dall: DMatrix, nfold: int, params, seed: int,
            stratified=False, folds=None, shuffle=True,
            fpreproc=None) -> List[CVPack]:
    rng = np.random.RandomState(seed)
    n = dall.num_row()
    if folds is not None:
        splits = folds
    else:
        idx = np.arange(n)
        if stratified and dall.info.labels is not None:
            y = np.asarray(dall.info.labels).reshape(-1)
            splits = _stratified_folds(y, nfold, rng)
        else:
            if shuffle:
                rng.shuffle(idx)
            chunks = np.array_split(idx, nfold)
            splits = [(np.concatenate([c for j, c in enumerate(chunks) if j != i]),
                       chunks[i]) for i in range(nfold)]
    packs = []
    for tr_idx, te_idx in splits:
        dtr, dte = dall.slice(tr_idx), dall.slice(te_idx)
        fparams = params
        if fpreproc is not None:
            # reference semantics (python-package training.py mknfold):
            # per-fold hook returns possibly-new matrices and params
            dtr, dte, fparams = fpreproc(dtr, dte, dict(params))
        packs.append(CVPack(dtr, dte, fparams))
    return packs


def _stratified_folds(y, nfold, rng):
    classes = np.unique(y)
    fold_idx = [[] for _ in range(nfold)]
    for c in classes:
        idx = np.nonzero(y == c)[0]
        rng.shuffle(idx)
        for i, chunk in enumerate(np.array_split(idx, nfold)):
            fold_idx[i].append(chunk)
    test_sets = [np.concatenate(f) for f in fold_idx]
    return [(np.concatenate([t for j, t in enumerate(test_sets) if j != i]),
             test_sets[i]) for i in range(nfold)]


def cv(params, dtrain, num_boost_round=10, nfold=3, stratified=False,
       folds=None, metrics=(), obj=None, feval=None, maximize=None,
       early_stopping_rounds=None, fpreproc=None, as_pandas=True,
       verbose_eval=None, show_stdv=True, seed=0, callbacks=None,
       shuffle=True, custom_metric=None):
    params = dict(params)
    if metrics:
        params["eval_metric"] = list(metrics) if len(list(metrics)) > 1 \
            else list(metrics)[0]
    packs = mknfold(dtrain, nfold, params, seed, stratified, folds, shuffle,
                    fpreproc)
    results: Dict[str, List[float]] = {}
    metric_fn = custom_metric or feval
    best_iter = None
    best_val = None
    stall = 0
    for i in range(num_boost_round):
        for p in packs:
            p.update(i, obj)
        msgs = [p.eval(i, metric_fn) for p in packs]
        # aggregate mean/std per metric
        per_metric: Dict[str, List[float]] = {}
        for msg in msgs:
            for part in msg.split("\t")[1:]:
                k, v = part.rsplit(":", 1)
                per_metric.setdefault(k, []).append(float(v))
        for k, vals in per_metric.items():
            results.setdefault(f"{k}-mean", []).append(float(np.mean(vals)))
            results.setdefault(f"{k}-std", []).append(float(np.std(vals)))
        if verbose_eval:
            print(f"[{i}]\t" + "\t".join(
                f"{k}-mean:{np.mean(v):.5f}" for k, v in per_metric.items()))
        if early_stopping_rounds:
            key = [k for k in per_metric if k.startswith("test")][-1]
            val = float(np.mean(per_metric[key]))
            base = key.rsplit("-", 1)[-1].split("@")[0]
            is_max = base in ("auc", "aucpr", "map", "ndcg", "pre")
            better = (best_val is None or (val > best_val if is_max
                                           else val < best_val))
            if better:
                best_val, best_iter, stall = val, i, 0
            else:
                stall += 1
                if stall >= early_stopping_rounds:
                    for k in results:
                        results[k] = results[k][:best_iter + 1]
                    break
    if as_pandas:
        try:
            import pandas as pd
            return pd.DataFrame.from_dict(results)
        except ImportError:
            pass
    return results
