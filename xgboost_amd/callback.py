"""Training callbacks (reference: python-package/xgboost/callback.py).

TrainingCallback / CallbackContainer / EarlyStopping / EvaluationMonitor /
LearningRateScheduler / TrainingCheckPoint with the reference semantics.
"""
from __future__ import annotations

import os
from typing import Dict, List, Optional, Sequence

from . import collective


class TrainingCallback:
    def before_training(self, model):
        return model

    def after_training(self, model):
        return model

    def before_iteration(self, model, epoch: int, evals_log) -> bool:
        return False

    def after_iteration(self, model, epoch: int, evals_log) -> bool:
        """Return True to stop training."""
        return False


EvalsLog = Dict[str, Dict[str, List[float]]]


class CallbackContainer:
    def __init__(self, callbacks: Sequence[TrainingCallback],
                 metric=None, output_margin: bool = True):
        self.callbacks = list(callbacks)
        self.metric = metric
        self.history: EvalsLog = {}

    def before_training(self, model):
        for c in self.callbacks:
            model = c.before_training(model)
        return model

    def after_training(self, model):
        for c in self.callbacks:
            model = c.after_training(model)
        return model

    def before_iteration(self, model, epoch, dtrain, evals) -> bool:
        return any(c.before_iteration(model, epoch, self.history)
                   for c in self.callbacks)

    def after_iteration(self, model, epoch, dtrain, evals) -> bool:
        if evals:
            msg = model.eval_set(evals, epoch, self.metric)
            parts = msg.split("\t")[1:]
            for part in parts:
                key, val = part.rsplit(":", 1)
                data_name, metric_name = key.split("-", 1)
                self.history.setdefault(data_name, {}).setdefault(
                    metric_name, []).append(float(val))
        return any(c.after_iteration(model, epoch, self.history)
                   for c in self.callbacks)


class EvaluationMonitor(TrainingCallback):
    def __init__(self, rank: int = 0, period: int = 1, show_stdv: bool = False,
                 logger=print):
        self.rank = rank
        self.period = period
        self.logger = logger
        self._latest = None

    def after_iteration(self, model, epoch, evals_log) -> bool:
        if not evals_log:
            return False
        msg = f"[{epoch}]"
        for data, metrics in evals_log.items():
            for metric_name, log in metrics.items():
                msg += f"\t{data}-{metric_name}:{log[-1]:.5f}"
        self._latest = msg
        if collective.get_rank() == self.rank and epoch % self.period == 0:
            self.logger(msg)
        return False

    def after_training(self, model):
        if (collective.get_rank() == self.rank and self._latest is not None
                and self.period != 1):
            self.logger(self._latest)
        return model


class EarlyStopping(TrainingCallback):
    def __init__(self, rounds: int, metric_name: Optional[str] = None,
                 data_name: Optional[str] = None, maximize: Optional[bool] = None,
                 save_best: bool = False, min_delta: float = 0.0):
        self.rounds = rounds
        self.metric_name = metric_name
        self.data_name = data_name
        self.maximize = maximize
        self.save_best = save_best
        self.min_delta = min_delta
        if min_delta < 0:
            raise ValueError("min_delta must be >= 0")
        self.stopping_history: EvalsLog = {}
        self.best_scores: list = []
        self.current_rounds = 0

    def before_training(self, model):
        self.starting_round = model.num_boosted_rounds()
        return model

    def _is_maximize(self, metric_name: str) -> bool:
        if self.maximize is not None:
            return self.maximize
        maximize_metrics = ("auc", "aucpr", "pre", "map", "ndcg",
                            "interval-regression-accuracy")
        base = metric_name.split("@")[0]
        return base in maximize_metrics

    def after_iteration(self, model, epoch, evals_log) -> bool:
        if not evals_log:
            raise ValueError("early stopping requires at least one eval set")
        data_name = self.data_name or list(evals_log.keys())[-1]
        if data_name not in evals_log:
            raise ValueError(f"eval set {data_name} not found")
        metric_name = self.metric_name or list(evals_log[data_name].keys())[-1]
        score = evals_log[data_name][metric_name][-1]
        maximize = self._is_maximize(metric_name)
        if not self.best_scores:
            improved = True
        elif maximize:
            improved = score - self.min_delta > max(self.best_scores)
        else:
            improved = score + self.min_delta < min(self.best_scores)
        if improved:
            self.best_scores.append(score)
            self.current_rounds = 0
            model.best_iteration = epoch
            model.best_score = score
            model.set_attr(best_iteration=str(epoch), best_score=str(score))
        else:
            self.current_rounds += 1
        return self.current_rounds >= self.rounds

    def after_training(self, model):
        if self.save_best and model.best_iteration is not None:
            best = model[: model.best_iteration + 1]
            best.best_iteration = model.best_iteration
            best.best_score = model.best_score
            return best
        return model


class LearningRateScheduler(TrainingCallback):
    def __init__(self, learning_rates):
        if callable(learning_rates):
            self.fn = learning_rates
        else:
            rates = list(learning_rates)
            self.fn = lambda epoch: rates[epoch]

    def before_iteration(self, model, epoch, evals_log) -> bool:
        model.set_param("eta", self.fn(epoch))
        return False


class TrainingCheckPoint(TrainingCallback):
    default_format = "json"

    def __init__(self, directory: str, name: str = "model",
                 as_pickle: bool = False, interval: int = 100):
        self.dir = str(directory)
        self.name = name
        self.as_pickle = as_pickle
        self.interval = interval
        self._epoch = 0

    def after_iteration(self, model, epoch, evals_log) -> bool:
        self._epoch += 1
        if self._epoch % self.interval == 0 and collective.get_rank() == 0:
            ext = "pkl" if self.as_pickle else self.default_format
            path = os.path.join(self.dir, f"{self.name}_{epoch}.{ext}")
            if self.as_pickle:
                import pickle
                with open(path, "wb") as fh:
                    pickle.dump(model, fh)
            else:
                model.save_model(path)
        return False
