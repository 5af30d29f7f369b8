"""Booster — training state machine, prediction, model IO.

Reference behavior: src/learner.cc (LearnerImpl), src/gbm/gbtree.cc
(GBTree::DoBoost/PredictBatch, prediction cache gbtree.h:192), JSON
model schema learner.cc:868-899.

The public API mirrors xgboost's Python Booster so reference users can
switch directly: update/boost/eval_set/predict/save_model/load_model/
dump_model/get_score/attributes/copy/slice.
"""
from __future__ import annotations

import json
import os
from typing import Any, Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch

from . import collective
from .backend.cpu import CpuOps, GradQuantizer
from .data import DMatrix
from .grower import TreeGrower
from .metrics import create_metric
from .objectives import Objective, create_objective
from .params import (TrainParam, canonicalize, check_unknown,
                     make_train_param)
from .tree_model import RegTree

VERSION = (3, 5, 0)


def _resolve_device(params: Dict[str, Any]) -> torch.device:
    dev = str(params.get("device", "cpu"))
    if "device" not in params and params.get("gpu_id") is not None:
        # deprecated reference spelling: gpu_id=N means cuda:N
        dev = f"cuda:{int(params['gpu_id'])}"
    if dev in ("cuda", "gpu"):
        return torch.device("cuda", torch.cuda.current_device()
                            if torch.cuda.is_available() else 0)
    if dev.startswith(("cuda:", "gpu:")):
        return torch.device("cuda", int(dev.split(":")[1]))
    return torch.device("cpu")


class Booster:
    def __init__(self, params: Optional[Dict[str, Any]] = None,
                 cache: Optional[Sequence[DMatrix]] = None,
                 model_file: Optional[str] = None):
        params = dict(params or {})
        self.raw_params = canonicalize(params)
        check_unknown(self.raw_params,
                      bool(self.raw_params.get("validate_parameters", False)))
        booster_kind = self.raw_params.get("booster", "gbtree")
        if booster_kind == "dart":
            booster_kind = "gbtree"  # DART deprecated -> gbtree (learner.cc:225)
        self.booster_kind = booster_kind
        self._linear = None  # GBLinearModel when booster_kind == "gblinear"
        self.tparam: TrainParam = make_train_param(self.raw_params)
        self.device = _resolve_device(self.raw_params)
        obj_name = str(self.raw_params.get("objective", "reg:squarederror"))
        self.objective: Objective = create_objective(obj_name, self.raw_params)
        self.seed = int(self.raw_params.get("seed", 0))
        self.seed_per_iteration = bool(self.raw_params.get("seed_per_iteration", False))

        # DART dropout lives in the tree booster (reference deprecates the
        # `dart` booster name and folds rate_drop/skip_drop/one_drop into
        # gbtree — gbm/gbtree.cc:474 DropTrees, :539 NormalizeTrees)
        def _flag(v) -> bool:
            return str(v).lower() in ("1", "true", "yes")
        rp = self.raw_params
        self.dart_params = {
            "rate_drop": float(rp.get("rate_drop", 0.0)),
            "one_drop": _flag(rp.get("one_drop", 0)),
            "skip_drop": float(rp.get("skip_drop", 0.0)),
            "sample_type": str(rp.get("sample_type", "uniform")),
            "normalize_type": str(rp.get("normalize_type", "tree")),
        }
        self.weight_drop: List[float] = []   # per-tree weights (DART)
        self._idx_drop: List[int] = []       # this iteration's drop set
        self._dart_dc = None                 # dropped-trees margin term
        self._dart_rng = np.random.RandomState((self.seed + 2027) % (1 << 31))

        self.trees: List[RegTree] = []
        self.tree_info: List[int] = []
        self.iteration_indptr: List[int] = [0]
        self.n_features: Optional[int] = None
        self.n_targets = int(self.raw_params.get("num_target", 1))
        self.base_score: Optional[float] = (
            float(self.raw_params["base_score"])
            if "base_score" in self.raw_params else None)
        self._base_score_estimated = "base_score" in self.raw_params
        self.attributes_: Dict[str, str] = {}
        self.feature_names: Optional[List[str]] = None
        self.feature_types: Optional[List[str]] = None
        self.best_iteration: Optional[int] = None
        self.best_score: Optional[float] = None

        self._cache: Dict[int, Tuple[torch.Tensor, int]] = {}  # id(dmat) -> (margin, version)
        self._ops_cache: Dict[int, Any] = {}
        # id(dmat) -> weakref with an eviction callback (see _pin)
        self._cache_refs: Dict[int, Any] = {}
        if model_file is not None:
            self.load_model(model_file)
        if cache:
            for d in cache:
                self._maybe_set_meta(d)

    # ------------------------------------------------------------------
    @property
    def n_outputs(self) -> int:
        return self.objective.n_outputs(self.n_targets)

    def num_boosted_rounds(self) -> int:
        return len(self.iteration_indptr) - 1

    def num_features(self) -> int:
        return self.n_features or 0

    def _pin(self, dmat) -> int:
        """Cache key for a DMatrix that cannot go stale: caches are keyed
        by id(dmat), and CPython reuses ids after collection — so a new
        DMatrix could silently alias a dead one's cached margin/quantized
        matrix (reference avoids this with DMatrixCache weak_ptr eviction,
        include/xgboost/cache.h:26).  A weakref eviction callback clears
        every cache family for the id the moment the DMatrix dies."""
        key = id(dmat)
        refs = self.__dict__.setdefault("_cache_refs", {})
        if key not in refs:
            import weakref
            self_ref = weakref.ref(self)

            def _evict(_, key=key, self_ref=self_ref):
                b = self_ref()
                if b is None:
                    return
                b._cache.pop(key, None)
                b._ops_cache.pop(key, None)
                b.__dict__.get("_fused_cache", {}).pop(key, None)
                b.__dict__.get("_exact_cache", {}).pop(key, None)
                b.__dict__.get("_cache_refs", {}).pop(key, None)

            refs[key] = weakref.ref(dmat, _evict)
        return key

    def _maybe_set_meta(self, dmat: DMatrix) -> None:
        if self.n_features is None:
            self.n_features = dmat.num_col()
        elif self.n_features != dmat.num_col():
            raise ValueError(
                f"feature count mismatch: model has {self.n_features}, "
                f"data has {dmat.num_col()}")
        if self.feature_names is None:
            self.feature_names = dmat.feature_names
        if self.feature_types is None:
            self.feature_types = dmat.feature_types
        if self.n_targets == 1 and dmat.info.labels is not None \
                and dmat.info.labels.ndim == 2 and dmat.info.labels.shape[1] > 1 \
                and self.objective.task == "regression":
            self.n_targets = dmat.info.labels.shape[1]
        if not getattr(self, "cat_categories_", None) and \
                getattr(dmat, "categories_", None):
            self.cat_categories_ = {int(k): list(v)
                                    for k, v in dmat.categories_.items()}

    def _init_base_score(self, dtrain: DMatrix) -> None:
        if self.base_score is not None and self._base_score_estimated:
            return
        if self.base_score is None:
            if dtrain.info.labels is None:
                self.base_score = 0.5
            else:
                self.base_score = float(
                    self.objective.init_estimation(dtrain.info))
                self.base_score = collective.broadcast_obj(self.base_score, 0)
        self._base_score_estimated = True

    def _base_margin_value(self) -> float:
        bs = self.base_score if self.base_score is not None else 0.5
        return float(self.objective.prob_to_margin(bs))

    def _ops_for(self, dmat: DMatrix, hess: Optional[np.ndarray] = None):
        if hess is not None:
            # approx: hessian-weighted cuts regenerated per tree
            # (reference: grow_histmaker/grow_gpu_approx, updater_approx.cc:46)
            if self.device.type == "cuda":
                # fully on-device regen like the reference's device path
                # (updater_gpu_hist.cu:809 ApproxBatch regen): DeviceSketch
                # weighted cuts + gbt_compress, no host round-trip; the
                # sketch handles the distributed merge internally
                from .backend.gpu import GpuOps
                from .data import quantize_dense_device
                from .gpu_sketch import device_cuts
                key = self._pin(dmat)
                xc = self.__dict__.setdefault("_approx_x_cache", {})
                Xd = xc.get(key)
                if Xd is None:
                    dd = (dmat.device_data()
                          if hasattr(dmat, "device_data") else None)
                    Xd = dd if dd is not None else torch.as_tensor(
                        dmat.raw_data(), device=self.device)
                    xc[key] = Xd
                h = torch.as_tensor(hess, device=Xd.device).abs() + 1e-16
                cuts = device_cuts(Xd, self.tparam.max_bin,
                                   missing=dmat.missing, weights=h,
                                   feature_types=dmat.info.feature_types)
                qm = quantize_dense_device(Xd, cuts, dmat.missing)
                return GpuOps(qm)
            from .data import quantize_dense
            from .quantile import make_cuts
            hess = np.asarray(hess)
            if collective.is_distributed():
                # ranks must agree on the re-sketched cuts: merge the
                # weighted per-rank summaries like the in-core sketch
                from .sketch import cuts_from_summaries, summarize_batch
                local = summarize_batch(
                    dmat.raw_data(), dmat.missing,
                    dmat.info.feature_types, self.tparam.max_bin,
                    weights=np.abs(hess) + 1e-16)
                gathered = collective.allgather_obj(local)
                cuts = cuts_from_summaries(
                    gathered, self.tparam.max_bin, dmat.num_col(),
                    dmat.info.feature_types)
            else:
                cuts = make_cuts(dmat.raw_data(), self.tparam.max_bin,
                                 weights=np.abs(hess) + 1e-16,
                                 feature_types=dmat.info.feature_types,
                                 missing=dmat.missing)
            qm = quantize_dense(dmat.raw_data(), cuts, dmat.missing)
            return CpuOps(qm)
        key = self._pin(dmat)
        ops = self._ops_cache.get(key)
        if ops is None:
            from .extmem import ExtMemOps, ExtMemQuantileDMatrix
            if isinstance(dmat, ExtMemQuantileDMatrix):
                ops = ExtMemOps(dmat, self.device)
                self._ops_cache[key] = ops
                return ops
            if getattr(dmat, "_sparse_data", None) is not None:
                from .sparse import CsrCpuOps, CsrGpuOps
                sqm = dmat.sparse_quantized(self.tparam.max_bin)
                ops = (CsrGpuOps(sqm, self.device)
                       if self.device.type == "cuda" else CsrCpuOps(sqm))
                self._ops_cache[key] = ops
                return ops
            from .sketch import sketch_cuts
            max_bin = self.tparam.max_bin
            qm = dmat.quantized(max_bin, sketch_fn=sketch_cuts)
            if self.device.type == "cuda":
                from .backend.gpu import GpuOps
                ops = GpuOps(qm.to(self.device))
            else:
                ops = CpuOps(qm)
            self._ops_cache[key] = ops
        return ops

    # -- DART dropout (reference gbm/gbtree.cc) ------------------------
    def _dart_configured(self) -> bool:
        dp = self.dart_params
        return dp["rate_drop"] != 0.0 or dp["one_drop"] or \
            dp["skip_drop"] != 0.0

    def _dart_track(self) -> bool:
        """CommitModel tracks tree weights once dropout is configured or
        a loaded model carries weights (gbtree.cc:347-356)."""
        return bool(self.weight_drop) or self._dart_configured()

    def _tw(self, t: int) -> float:
        """Per-tree prediction weight (1.0 outside DART)."""
        return self.weight_drop[t] if t < len(self.weight_drop) else 1.0

    def _dart_drop_trees(self) -> List[int]:
        """Select this iteration's drop set (GBTree::DropTrees,
        gbtree.cc:474): uniform or weighted per-tree Bernoulli at
        rate_drop, optional one_drop backstop, skip_drop gate."""
        dp = self.dart_params
        self._idx_drop = []
        if not self._dart_configured() or not self.trees:
            return []
        if not self.weight_drop:
            self.weight_drop = [1.0] * len(self.trees)
        elif len(self.weight_drop) < len(self.trees):
            self.weight_drop.extend(
                [1.0] * (len(self.trees) - len(self.weight_drop)))
        rng = self._dart_rng
        if dp["skip_drop"] > 0.0 and rng.uniform() < dp["skip_drop"]:
            return []
        wd = self.weight_drop
        if dp["sample_type"] == "weighted":
            sw = float(sum(wd))
            for i, w in enumerate(wd):
                if rng.uniform() < dp["rate_drop"] * len(wd) * w / sw:
                    self._idx_drop.append(i)
            if dp["one_drop"] and not self._idx_drop and wd:
                p = np.asarray(wd, np.float64)
                self._idx_drop.append(int(rng.choice(len(wd), p=p / p.sum())))
        else:
            for i in range(len(wd)):
                if rng.uniform() < dp["rate_drop"]:
                    self._idx_drop.append(i)
            if dp["one_drop"] and not self._idx_drop and wd:
                self._idx_drop.append(int(rng.randint(len(wd))))
        return list(self._idx_drop)

    def _dart_new_weight(self) -> float:
        """Weight the trees built this iteration will carry
        (NormalizeTrees, gbtree.cc:539)."""
        if not self._dart_track():
            return 1.0
        k = len(self._idx_drop)
        if k == 0:
            return 1.0
        lr = float(self.tparam.eta)
        if self.dart_params["normalize_type"] == "forest":
            return 1.0 / (1.0 + lr)
        return 1.0 / (k + lr)

    def _dart_commit(self, margin: torch.Tensor, n_new: int) -> None:
        """NormalizeTrees + incremental margin fix-up: the cached margin
        holds sum(w_i * tree_i); scaling the dropped trees by `factor`
        shifts it by (factor - 1) * dropped_contribution."""
        if not self._dart_track():
            return
        n_old = len(self.trees) - n_new
        if len(self.weight_drop) < n_old:
            self.weight_drop.extend([1.0] * (n_old - len(self.weight_drop)))
        k = len(self._idx_drop)
        lr = float(self.tparam.eta)
        if k == 0:
            self.weight_drop.extend([1.0] * n_new)
        else:
            factor = (1.0 / (1.0 + lr)
                      if self.dart_params["normalize_type"] == "forest"
                      else k / (k + lr))
            for i in self._idx_drop:
                self.weight_drop[i] *= factor
            if self._dart_dc is not None:
                margin += (factor - 1.0) * self._dart_dc
            self.weight_drop.extend([self._dart_new_weight()] * n_new)
        self._idx_drop = []
        self._dart_dc = None
        self._forest_dev_cache = {}  # weights changed: drop stale SoA

    def _predict_tree_subset(self, dmat: DMatrix,
                             idxs: List[int]) -> torch.Tensor:
        """sum(w_i * tree_i(X)) over the drop set, shaped like the
        margin cache."""
        n = dmat.num_row()
        out = torch.zeros((n, self.n_outputs), dtype=torch.float32,
                          device=self.device)
        if getattr(dmat, "_sparse_data", None) is not None:
            return self._sparse_margin_add(dmat, out, list(idxs))
        from .extmem import ExtMemQuantileDMatrix
        if isinstance(dmat, ExtMemQuantileDMatrix):
            # quantized-only pages: bin-based traversal per page (the
            # split conds are cut values, so this is exact)
            for pi in range(len(dmat.pages)):
                qm = dmat.page_qm(pi)
                s, e = dmat.page_offsets[pi], dmat.page_offsets[pi + 1]
                gg = qm.global_gidx().cpu().numpy()
                for t in idxs:
                    tree = self.trees[t]
                    pos = tree.predict_leaf_bins(gg, dmat.cuts)
                    vals = tree.split_cond[:tree.n_nodes][pos] * self._tw(t)
                    out[s:e, self.tree_info[t]] += torch.as_tensor(
                        vals, device=out.device)
            return out
        if self.device.type == "cuda" and not any(
                self.trees[t].leaf_values is not None for t in idxs):
            from .backend.gpu import predict_subset_gpu
            return predict_subset_gpu(self, dmat, idxs, out)
        X = dmat.raw_data()
        for t in idxs:
            tree = self.trees[t]
            pos = tree.predict_leaf_np(X, dmat.missing)
            if tree.leaf_values is not None:  # vector leaves
                out += self._tw(t) * torch.as_tensor(
                    tree.leaf_values[:tree.n_nodes][pos],
                    device=out.device)
            else:
                vals = tree.split_cond[:tree.n_nodes][pos] * self._tw(t)
                out[:, self.tree_info[t]] += torch.as_tensor(
                    vals, device=out.device)
        return out

    # -- training ------------------------------------------------------
    def update(self, dtrain: DMatrix, iteration: int,
               fobj=None) -> None:
        self._maybe_set_meta(dtrain)
        self._init_base_score(dtrain)
        if self.tparam.process_type == "update":
            self._update_existing(dtrain, iteration)
            return
        margin = self._cached_margin(dtrain)
        from .monitor import TrainingObserver
        if TrainingObserver.enabled():
            TrainingObserver.observe_predictions(iteration, margin)
        use_margin = margin
        if self.booster_kind != "gblinear" and self._dart_configured():
            # DART: gradients come from the margin with this round's
            # dropped trees removed (GBTree::PredictBatch is_training)
            dropped = self._dart_drop_trees()
            if dropped:
                self._dart_dc = self._predict_tree_subset(dtrain, dropped)
                use_margin = margin - self._dart_dc
        elif fobj is None and self._boost_fused(dtrain, margin, iteration):
            if TrainingObserver.enabled():
                TrainingObserver.observe_tree(iteration, self.trees[-1])
            return
        if fobj is not None:
            # custom objectives receive the RAW margin (reference
            # core.py:2315 predict(output_margin=True, training=True))
            preds = use_margin.detach().cpu().numpy()
            from .objective import Objective as _ClsObj, TreeObjective
            if isinstance(fobj, TreeObjective):
                # full gradient values the leaves; an optional reduced
                # gradient finds the structure (reference core.py:2320)
                vgrad, vhess = fobj(iteration, preds, dtrain)
                sg = fobj.split_grad(iteration, vgrad, vhess)
                if sg is not None:
                    self._boost_split_grad(dtrain, sg[0], sg[1], vgrad,
                                           vhess, iteration)
                    return
                grad, hess = vgrad, vhess
            elif isinstance(fobj, _ClsObj):
                grad, hess = fobj(iteration, preds, dtrain)
            else:
                grad, hess = fobj(np.squeeze(preds), dtrain)
            grad = torch.as_tensor(np.asarray(grad, np.float32),
                                   device=margin.device).view(margin.shape)
            hess = torch.as_tensor(np.asarray(hess, np.float32),
                                   device=margin.device).view(margin.shape)
        else:
            grad, hess = self.objective.get_gradient(use_margin, dtrain.info,
                                                     iteration)
            TrainingObserver.observe_gradient(iteration, grad, hess)
        self.boost_gpair(dtrain, grad, hess, iteration)

    def boost(self, dtrain: DMatrix, iteration: int = 0, grad=None, hess=None) -> None:
        """Boost with user-supplied gradients (XGBoosterBoostOneIter)."""
        self._maybe_set_meta(dtrain)
        if self.base_score is None:
            self.base_score = 0.5
            self._base_score_estimated = True
        dev = self.device
        g = torch.as_tensor(np.asarray(grad, np.float32), device=dev)
        h = torch.as_tensor(np.asarray(hess, np.float32), device=dev)
        n = dtrain.num_row()
        g = g.view(n, -1)
        h = h.view(n, -1)
        self.boost_gpair(dtrain, g, h, iteration)

    def boost_gpair(self, dtrain: DMatrix, grad: torch.Tensor,
                    hess: torch.Tensor, iteration: int) -> None:
        if self.booster_kind == "gblinear":
            self._boost_linear(dtrain, grad, hess, iteration)
            return
        from .data import QuantileDMatrix as _QDM
        if self.tparam.tree_method in ("exact", "approx") and \
                type(dtrain) is _QDM:
            # reference: QuantileDMatrix carries only quantized pages —
            # exact has no SparsePage to walk (iterative_dmatrix.cc:159)
            # and approx would need hessian-weighted cut regeneration
            # (iterative_dmatrix.cc:129 RegenGHist CHECK)
            raise ValueError(
                f"tree_method={self.tparam.tree_method} does not support "
                "QuantileDMatrix; use hist or a plain DMatrix")
        if self.tparam.tree_method == "exact":
            if str(self.raw_params.get(
                    "multi_strategy", "one_output_per_tree")) ==                     "multi_output_tree" and                     (grad.dim() > 1 and grad.shape[1] > 1):
                raise ValueError(
                    "Only the hist tree method is supported for building "
                    "multi-target trees with vector leaf.")
            self._boost_exact(dtrain, grad, hess, iteration)
            return
        ops = self._ops_for(dtrain)
        n = dtrain.num_row()
        n_out = grad.shape[1] if grad.dim() > 1 else 1
        grad = grad.view(n, n_out)
        hess = hess.view(n, n_out)
        margin, version = self._cache[id(dtrain)]
        seed = (self.seed + iteration if not self.seed_per_iteration
                else self.seed + iteration * 2654435761)
        new_trees = 0
        eta_scale = 1.0 / max(1, self.tparam.num_parallel_tree)
        multi_strategy = str(self.raw_params.get("multi_strategy",
                                                 "one_output_per_tree"))
        if multi_strategy == "multi_output_tree" and n_out > 1:
            # reference gbtree.cc:187: vector-leaf trees are hist-only
            if self.tparam.tree_method not in ("auto", "hist"):
                raise ValueError(
                    "Only the hist tree method is supported for building "
                    "multi-target trees with vector leaf.")
            self._boost_multi_target(dtrain, ops, grad, hess, margin, seed)
            return
        is_approx = self.tparam.tree_method == "approx"
        w_new = self._dart_new_weight()
        for k in range(n_out):
            if is_approx:
                ops = self._ops_for(
                    dtrain, hess=hess[:, k].detach())
            for ptree in range(self.tparam.num_parallel_tree):
                gpair = torch.stack([grad[:, k], hess[:, k]], dim=1).contiguous()
                gpair = self._subsample(gpair, seed + 7919 * ptree + 104729 * k)
                gpair = gpair.to(ops.device)
                quantizer = GradQuantizer(gpair)
                qg = quantizer.quantize(gpair)
                tree = RegTree(self.n_features)
                grower = TreeGrower(
                    ops, self._scaled_param(eta_scale), quantizer, n,
                    seed=seed + 31 * ptree + 17 * k,
                    monotone=self._monotone_array(),
                    interaction=self.tparam.interaction_constraints,
                    feature_weights=dtrain.info.feature_weights)
                tree, positions = grower.grow(qg, tree)
                self.trees.append(tree)
                self.tree_info.append(k)
                new_trees += 1
                # update prediction cache from leaf positions
                leaf_np = tree.split_cond[:tree.n_nodes].copy()
                if hasattr(ops, "stager"):
                    (leaf_vals,) = ops.stager.upload([leaf_np])
                else:
                    leaf_vals = torch.as_tensor(leaf_np, device=margin.device)
                add = leaf_vals[positions.to(margin.device).long()]
                margin[:, k] += add if w_new == 1.0 else w_new * add
        self._dart_commit(margin, new_trees)
        self.iteration_indptr.append(self.iteration_indptr[-1] + new_trees)
        self._cache[id(dtrain)] = (margin, len(self.trees))
        from .monitor import TrainingObserver
        if TrainingObserver.enabled() and self.trees:
            TrainingObserver.observe_tree(iteration, self.trees[-1])
        if self.tparam.debug_synchronize:
            collective.check_synchronized(
                json.dumps(self.trees[-1].to_json()).encode(), "tree")

    def _boost_fused(self, dtrain: DMatrix, margin: torch.Tensor,
                     iteration: int) -> bool:
        """GPU fast path: fused gradient+quantize HIP kernel for the hot
        objectives (ops/cpp/gpair.hip); returns False when inapplicable
        (the generic torch path handles those)."""
        if self.device.type != "cuda":
            return False
        obj_id = {"binary:logistic": 0, "reg:squarederror": 1}.get(
            self.objective.name)
        if obj_id is None or self.n_outputs != 1:
            return False
        tp = self.tparam
        if (tp.subsample < 1.0 or tp.num_parallel_tree != 1
                or tp.tree_method not in ("auto", "hist")
                or self.booster_kind == "gblinear"
                or tp.process_type == "update"
                or str(self.raw_params.get("multi_strategy",
                                           "one_output_per_tree"))
                != "one_output_per_tree"):
            return False
        from . import ops as hip_ops
        try:
            lib = hip_ops.load()
        except RuntimeError:
            return False
        if not hasattr(lib, "gbt_gpair_fused"):
            return False
        ops = self._ops_for(dtrain)
        n = dtrain.num_row()
        key = self._pin(dtrain)
        fc = self.__dict__.setdefault("_fused_cache", {})
        ent = fc.get(key)
        if ent is None:
            dev = self.device
            labels = torch.as_tensor(
                np.ascontiguousarray(dtrain.get_label(), np.float32),
                device=dev)
            w = dtrain.info.weights
            weights = (torch.as_tensor(np.ascontiguousarray(w, np.float32),
                                       device=dev) if w is not None else None)
            ent = {"labels": labels, "weights": weights,
                   "gh": torch.empty((n, 2), dtype=torch.float32, device=dev),
                   "qg": torch.empty((n, 2), dtype=torch.int32, device=dev),
                   "maxabs": torch.zeros(2, dtype=torch.float32, device=dev)}
            fc[key] = ent
        ent["maxabs"].zero_()
        spw = float(self.raw_params.get("scale_pos_weight", 1.0))
        lib.gbt_gpair_fused(
            obj_id, hip_ops.ptr(margin.contiguous().view(-1)),
            hip_ops.ptr(ent["labels"]), hip_ops.ptr(ent["weights"]),
            spw, n, hip_ops.ptr(ent["gh"]), hip_ops.ptr(ent["maxabs"]),
            hip_ops.stream())
        m = ent["maxabs"]
        if collective.is_distributed():
            collective.allreduce_max_(m)
        # scales are derived ON DEVICE (QuantizeKernel / root eval) and
        # come back to the host with the tree driver's root sync — this
        # path has no per-round max-abs readback sync at all
        quantizer = GradQuantizer.__new__(GradQuantizer)
        quantizer.g_scale = None
        quantizer.h_scale = None
        quantizer.maxabs_dev = m
        if "rootsum" not in ent:
            ent["rootsum"] = torch.zeros(2, dtype=torch.int64,
                                         device=margin.device)
        else:
            ent["rootsum"].zero_()
        lib.gbt_quantize(hip_ops.ptr(ent["gh"]), n, 0.0, 0.0,
                         hip_ops.ptr(m), hip_ops.ptr(ent["qg"]),
                         hip_ops.ptr(ent["rootsum"]), hip_ops.stream())
        if collective.is_distributed():
            collective.allreduce_sum_(ent["rootsum"])
        quantizer.root_sums_dev = ent["rootsum"]
        seed = (self.seed + iteration if not self.seed_per_iteration
                else self.seed + iteration * 2654435761)
        tree = RegTree(self.n_features)
        grower = TreeGrower(ops, self.tparam, quantizer, n, seed=seed,
                            monotone=self._monotone_array(),
                            interaction=tp.interaction_constraints,
                            feature_weights=dtrain.info.feature_weights)
        tree, positions = grower.grow(ent["qg"], tree)
        self.trees.append(tree)
        self.tree_info.append(0)
        self.iteration_indptr.append(self.iteration_indptr[-1] + 1)
        leaf_np = tree.split_cond[:tree.n_nodes].copy()
        if hasattr(ops, "stager") and hasattr(lib, "gbt_margin_add") \
                and positions.device == margin.device:
            (leaf_vals,) = ops.stager.upload([leaf_np])
            mc = margin.contiguous()
            lib.gbt_margin_add(hip_ops.ptr(mc), hip_ops.ptr(positions),
                               hip_ops.ptr(leaf_vals), n,
                               mc.stride(0), 0, hip_ops.stream())
            if mc.data_ptr() != margin.data_ptr():
                margin.copy_(mc)
        else:
            leaf_vals = torch.as_tensor(leaf_np, device=margin.device)
            margin[:, 0] += leaf_vals[positions.to(margin.device).long()]
        self._cache[key] = (margin, len(self.trees))
        if self.tparam.debug_synchronize:
            collective.check_synchronized(
                json.dumps(self.trees[-1].to_json()).encode(), "tree")
        return True

    def _boost_linear(self, dtrain: DMatrix, grad: torch.Tensor,
                      hess: torch.Tensor, iteration: int) -> None:
        from .linear import GBLinearModel
        # reference gblinear.cc:129: linear boosting has no encoding for
        # categorical inputs
        ft = dtrain.feature_types or []
        if any(t in ("c", "categorical") for t in ft) or \
                getattr(dtrain, "categories_", None):
            raise ValueError("`gblinear` does not support categorical data")
        if self._linear is None:
            self._linear = GBLinearModel(self.n_features, self.n_outputs,
                                         self.raw_params, self.device)
        X = torch.as_tensor(dtrain.raw_data(), device=self.device)
        siw = float(np.sum(dtrain.get_weight()))
        self._linear.update(X, grad, hess, iteration, siw)
        margin = self._linear.predict_margin(X) + self._base_margin_value()
        self.iteration_indptr.append(self.iteration_indptr[-1])
        self._cache[id(dtrain)] = (margin, len(self.trees))

    def _boost_multi_target(self, dtrain: DMatrix, ops, grad, hess,
                            margin, seed: int) -> None:
        """multi_strategy=multi_output_tree: one vector-leaf tree per
        iteration (reference MultiTargetHistMaker)."""
        from .grower import MultiTargetGrower
        n, n_out = grad.shape
        qgpairs, quantizers = [], []
        for k in range(n_out):
            gp = torch.stack([grad[:, k], hess[:, k]], dim=1).contiguous()
            gp = gp.to(ops.device)
            q = GradQuantizer(gp)
            quantizers.append(q)
            qgpairs.append(q.quantize(gp))
        tree = RegTree(self.n_features, n_out)
        grower = MultiTargetGrower(
            ops, self.tparam, quantizers, n, seed,
            feature_weights=dtrain.info.feature_weights)
        tree, positions = grower.grow(qgpairs, tree)
        self.trees.append(tree)
        self.tree_info.append(0)
        self.iteration_indptr.append(self.iteration_indptr[-1] + 1)
        leaf_vals = torch.as_tensor(
            tree.leaf_values[:tree.n_nodes].copy(), device=margin.device)
        w_new = self._dart_new_weight()
        add = leaf_vals[positions.to(margin.device).long()]
        margin += add if w_new == 1.0 else w_new * add
        self._dart_commit(margin, 1)
        self._cache[id(dtrain)] = (margin, len(self.trees))

    def _boost_split_grad(self, dtrain: DMatrix, sgrad, shess, vgrad,
                          vhess, iteration: int) -> None:
        """Reduced-gradient boosting (reference c_api.cc:1237
        XGBoosterTrainOneIterWithSplitGrad + gbtree.cc:191): ONE
        vector-leaf tree per iteration whose STRUCTURE comes from the
        reduced (sgrad, shess) and whose leaf VALUES come from the
        full-width (vgrad, vhess)."""
        dev = self.device
        n = dtrain.num_row()

        def t2(a):
            return torch.as_tensor(np.asarray(a, np.float32),
                                   device=dev).view(n, -1)

        sg, sh = t2(sgrad), t2(shess)
        vg, vh = t2(vgrad), t2(vhess)
        if vg.shape[1] < 2:
            raise ValueError(
                "split_grad requires vector-leaf trees: configure the "
                "booster with num_target matching the full gradient "
                "width (reference gbtree.cc:192)")
        if vg.shape[1] != self.n_outputs:
            raise ValueError(
                f"value gradient width {vg.shape[1]} != model outputs "
                f"{self.n_outputs}")
        if self.tparam.monotone_constraints:
            raise ValueError("monotone constraints are not supported "
                             "with reduced gradients (gbtree.cc:194)")
        margin, _ = self._cache[id(dtrain)]
        ops = self._ops_for(dtrain)
        seed = self.seed + iteration
        if sg.shape[1] == 1:
            gpair = torch.stack([sg[:, 0], sh[:, 0]],
                                dim=1).contiguous().to(ops.device)
            quant = GradQuantizer(gpair)
            qg = quant.quantize(gpair)
            tree = RegTree(self.n_features)
            grower = TreeGrower(ops, self.tparam, quant, n, seed=seed,
                                feature_weights=dtrain.info.feature_weights)
            tree, positions = grower.grow(qg, tree)
        else:
            from .grower import MultiTargetGrower
            qgpairs, quantizers = [], []
            for k in range(sg.shape[1]):
                gp = torch.stack([sg[:, k], sh[:, k]],
                                 dim=1).contiguous().to(ops.device)
                q = GradQuantizer(gp)
                quantizers.append(q)
                qgpairs.append(q.quantize(gp))
            tree = RegTree(self.n_features, sg.shape[1])
            grower = MultiTargetGrower(
                ops, self.tparam, quantizers, n, seed,
                feature_weights=dtrain.info.feature_weights)
            tree, positions = grower.grow(qgpairs, tree)
        # vector leaves from the FULL gradient at the found partition
        # (deterministic sort+cumsum segment sums, no fp64 atomics)
        C = vg.shape[1]
        nn = tree.n_nodes
        pos = positions.to(dev).long()
        ps, pperm = torch.sort(pos)
        bnd = torch.searchsorted(ps, torch.arange(nn + 1, device=dev))
        Gv = torch.zeros((nn, C), dtype=torch.float64, device=dev)
        Hv = torch.zeros((nn, C), dtype=torch.float64, device=dev)
        cs = torch.zeros(n + 1, dtype=torch.float64, device=dev)
        for c in range(C):
            torch.cumsum(vg[:, c].double()[pperm], 0, out=cs[1:])
            Gv[:, c] = cs[bnd[1:]] - cs[bnd[:-1]]
            torch.cumsum(vh[:, c].double()[pperm], 0, out=cs[1:])
            Hv[:, c] = cs[bnd[1:]] - cs[bnd[:-1]]
        lam = self.tparam.reg_lambda
        alpha = self.tparam.reg_alpha
        w = -torch.sign(Gv) * torch.clamp(Gv.abs() - alpha, min=0.0) \
            / (Hv + lam)
        mds = self.tparam.max_delta_step
        if mds > 0:
            w = w.clamp(-mds, mds)
        w = (w * self.tparam.eta).float()
        # upgrade to a vector-leaf tree: values only at leaves
        leaf_mask = torch.as_tensor(tree.left[:nn] == -1, device=dev)
        w = torch.where(leaf_mask[:, None], w, torch.zeros_like(w))
        lv = np.zeros((len(tree.split_cond), C), dtype=np.float32)
        lv[:nn] = w.cpu().numpy()
        tree.n_targets = C
        tree.leaf_values = lv
        self.trees.append(tree)
        self.tree_info.append(0)
        self.iteration_indptr.append(self.iteration_indptr[-1] + 1)
        w_new = self._dart_new_weight()
        add = torch.as_tensor(lv[:nn], device=margin.device)[pos]
        margin += add if w_new == 1.0 else w_new * add
        self._dart_commit(margin, 1)
        self._cache[id(dtrain)] = (margin, len(self.trees))

    def _update_existing(self, dtrain: DMatrix, iteration: int) -> None:
        """process_type=update: run prune/refresh updaters over the trees
        of `iteration` instead of growing new ones (reference gbtree.cc
        process_type handling)."""
        from .updaters import prune_tree, refresh_tree
        if iteration >= self.num_boosted_rounds():
            raise ValueError(
                "process_type=update requires an existing model with at "
                f"least {iteration + 1} boosted rounds")
        updater_seq = (self.tparam.updater or "refresh").split(",")
        margin = self._cached_margin(dtrain)
        grad, hess = self.objective.get_gradient(margin, dtrain.info,
                                                 iteration)
        lo, hi = (self.iteration_indptr[iteration],
                  self.iteration_indptr[iteration + 1])
        X = dtrain.raw_data()
        for t in range(lo, hi):
            k = self.tree_info[t]
            gpair = torch.stack([grad[:, k], hess[:, k]],
                                dim=1).cpu().numpy()
            for upd in updater_seq:
                upd = upd.strip()
                if upd == "prune":
                    self.trees[t] = prune_tree(self.trees[t], self.tparam)
                elif upd == "refresh":
                    refresh_tree(self.trees[t], X, gpair, self.tparam,
                                 dtrain.missing,
                                 refresh_leaf=self.tparam.refresh_leaf)
                else:
                    raise ValueError(
                        f"unsupported updater for process_type=update: {upd}")
        self._cache.clear()  # leaf values changed; rebuild margins lazily

    def _boost_exact(self, dtrain: DMatrix, grad: torch.Tensor,
                     hess: torch.Tensor, iteration: int) -> None:
        """tree_method=exact (reference ColMaker; CPU only like the
        reference — src/tree/updater_colmaker.cc)."""
        from .exact import grow_exact
        if self.device.type == "cuda":
            raise ValueError("tree_method=exact is CPU-only; use hist on GPU")
        # reference updater_colmaker.cc:104-113: the exact method
        # rejects categorical data, external memory and colsample_bynode
        ft = dtrain.feature_types or []
        if any(t in ("c", "categorical") for t in ft) or \
                getattr(dtrain, "categories_", None):
            raise ValueError(
                "Updater `grow_colmaker` or `exact` tree method does not"
                " support categorical data")
        from .extmem import ExtMemQuantileDMatrix
        if isinstance(dtrain, ExtMemQuantileDMatrix):
            raise ValueError(
                "Updater `grow_colmaker` or `exact` tree method doesn't "
                "support external memory training")
        if self.tparam.colsample_bynode != 1.0:
            raise ValueError("column sample by node is not yet supported "
                             "by the exact tree method")
        n = dtrain.num_row()
        n_out = grad.shape[1]
        margin, _ = self._cache[id(dtrain)]
        seed = self.seed + iteration
        cache = self.__dict__.setdefault("_exact_cache", {}).setdefault(
            id(dtrain), {})
        new_trees = 0
        w_new = self._dart_new_weight()
        for k in range(n_out):
            for ptree in range(self.tparam.num_parallel_tree):
                gpair = torch.stack([grad[:, k], hess[:, k]], dim=1)
                gpair = self._subsample(gpair, seed + 7919 * ptree + 11 * k)
                tree = RegTree(self.n_features)
                positions = grow_exact(dtrain.raw_data(),
                                       gpair.cpu().numpy(),
                                       self._scaled_param(
                                           1.0 / max(1, self.tparam.num_parallel_tree)),
                                       tree, dtrain.missing, cache)
                self.trees.append(tree)
                self.tree_info.append(k)
                new_trees += 1
                leaf_vals = torch.as_tensor(
                    tree.split_cond[:tree.n_nodes].copy(), device=margin.device)
                pos_t = torch.as_tensor(positions.astype(np.int64),
                                        device=margin.device)
                add = leaf_vals[pos_t]
                margin[:, k] += add if w_new == 1.0 else w_new * add
        self._dart_commit(margin, new_trees)
        self.iteration_indptr.append(self.iteration_indptr[-1] + new_trees)
        self._cache[id(dtrain)] = (margin, len(self.trees))

    def _scaled_param(self, eta_scale: float) -> TrainParam:
        if eta_scale == 1.0:
            return self.tparam
        import dataclasses as dc
        p = dc.replace(self.tparam)
        p.eta = self.tparam.eta * eta_scale
        return p

    def _monotone_array(self) -> Optional[np.ndarray]:
        mc = self.tparam.monotone_constraints
        if mc is None:
            return None
        if isinstance(mc, str):
            mc = [int(x) for x in mc.strip("()[] ").split(",") if x.strip()]
        arr = np.zeros(self.n_features, dtype=np.int64)
        arr[:len(mc)] = np.asarray(list(mc), dtype=np.int64)[:self.n_features]
        return arr

    def _subsample(self, gpair: torch.Tensor, seed: int) -> torch.Tensor:
        p = self.tparam.subsample
        if p >= 1.0:
            return gpair
        gen = torch.Generator(device="cpu").manual_seed(seed & 0x7FFFFFFF)
        if self.tparam.sampling_method == "gradient_based":
            # MVS-style: keep prob proportional to sqrt(g^2 + lambda*h^2)
            g = gpair[:, 0].cpu()
            h = gpair[:, 1].cpu()
            score = torch.sqrt(g * g + self.tparam.reg_lambda * h * h)
            n = g.numel()
            threshold = _sample_rate_threshold(score, p)
            u = torch.rand(n, generator=gen)
            keep_prob = torch.clamp(score / threshold, max=1.0)
            keep = u < keep_prob
            out = gpair.clone()
            scale = (1.0 / keep_prob.clamp(min=1e-16)).to(gpair.device)
            out[:, 0] = torch.where(keep.to(gpair.device),
                                    gpair[:, 0] * scale, torch.zeros_like(gpair[:, 0]))
            out[:, 1] = torch.where(keep.to(gpair.device),
                                    gpair[:, 1] * scale, torch.zeros_like(gpair[:, 1]))
            return out
        mask = (torch.rand(gpair.shape[0], generator=gen) < p).to(gpair.device)
        out = gpair.clone()
        out[~mask] = 0.0
        return out

    # -- prediction ----------------------------------------------------
    def _cached_margin(self, dmat: DMatrix) -> torch.Tensor:
        key = self._pin(dmat)
        n = dmat.num_row()
        entry = self._cache.get(key)
        if entry is not None and entry[1] == len(self.trees):
            return entry[0]
        margin = self._predict_margin(dmat)
        self._cache[key] = (margin, len(self.trees))
        return margin

    def _predict_margin(self, dmat: DMatrix,
                        iteration_range: Tuple[int, int] = (0, 0)
                        ) -> torch.Tensor:
        n = dmat.num_row()
        out = torch.full((n, self.n_outputs), self._base_margin_value(),
                         dtype=torch.float32, device=self.device)
        if dmat.info.base_margin is not None:
            bm = torch.as_tensor(dmat.info.base_margin, dtype=torch.float32,
                                 device=self.device)
            out = bm.view(n, -1).expand(n, self.n_outputs).clone()
        if self.booster_kind == "gblinear":
            if self._linear is not None:
                X = torch.as_tensor(dmat.raw_data(), device=self.device)
                out = out + self._linear.predict_margin(X)
            return out
        lo, hi = self._tree_range(iteration_range)
        from .extmem import ExtMemQuantileDMatrix
        if isinstance(dmat, ExtMemQuantileDMatrix):
            return self._predict_margin_extmem(dmat, out, lo, hi)
        if getattr(dmat, "_sparse_data", None) is not None:
            return self._predict_margin_sparse(dmat, out, lo, hi)
        train_cats = getattr(self, "cat_categories_", None)
        pred_cats = getattr(dmat, "categories_", None)
        aligned = False
        X = None
        if train_cats and pred_cats and pred_cats != train_cats:
            # predict-frame categories re-coded to the training dictionary
            # (reference encoder/ordinal.h Recode)
            from .data import align_categories
            X = align_categories(dmat.raw_data(), pred_cats, train_cats)
            aligned = True
        has_mt = any(t.leaf_values is not None for t in self.trees[lo:hi])
        if self.device.type == "cuda" and (hi - lo) > 0 and not has_mt \
                and not aligned:
            # device-resident inputs never round-trip to host here
            from .backend.gpu import predict_margin_gpu
            return predict_margin_gpu(self, dmat, out, lo, hi)
        if X is None:
            X = dmat.raw_data()
        for t in range(lo, hi):
            tree = self.trees[t]
            pos = tree.predict_leaf_np(X, dmat.missing)
            w = self._tw(t)
            if tree.leaf_values is not None:
                out += w * torch.as_tensor(
                    tree.leaf_values[:tree.n_nodes][pos], device=out.device)
            else:
                vals = tree.split_cond[:tree.n_nodes][pos]
                out[:, self.tree_info[t]] += w * torch.as_tensor(
                    vals, device=out.device)
        return out

    def _predict_margin_sparse(self, dmat, out: torch.Tensor,
                               lo: int, hi: int) -> torch.Tensor:
        return self._sparse_margin_add(dmat, out, list(range(lo, hi)))

    def _sparse_margin_add(self, dmat, out: torch.Tensor,
                           idxs: List[int]) -> torch.Tensor:
        """Sparse predict: densify ONLY the features used by the trees,
        absent entries become NaN (missing -> default direction)."""
        if not idxs:
            return out  # no trees yet (first margin of training)
        csr = dmat.sparse_data()
        n = csr.shape[0]
        used = sorted(set(
            int(f) for t in idxs
            for nid in range(self.trees[t].n_nodes)
            if not self.trees[t].is_leaf(nid)
            for f in [self.trees[t].split_index[nid]]))
        if not used:
            return out
        col_of = {f: i for i, f in enumerate(used)}
        sub = csr[:, used].tocoo()
        X_sub = np.full((n, len(used)), np.nan, dtype=np.float32)
        X_sub[sub.row, sub.col] = sub.data
        for t in idxs:
            tree = self.trees[t]
            remap = tree.split_index[:tree.n_nodes].copy()
            saved = remap.copy()
            for nid in range(tree.n_nodes):
                if not tree.is_leaf(nid):
                    remap[nid] = col_of[int(saved[nid])]
            tree.split_index[:tree.n_nodes] = remap
            try:
                pos = tree.predict_leaf_np(X_sub, np.nan)
            finally:
                tree.split_index[:tree.n_nodes] = saved
            vals = tree.split_cond[:tree.n_nodes][pos]
            out[:, self.tree_info[t]] += self._tw(t) * torch.as_tensor(
                vals, device=out.device)
        return out

    def _predict_margin_extmem(self, dmat, out: torch.Tensor,
                               lo: int, hi: int) -> torch.Tensor:
        """Bin-based traversal per quantized page (no raw values)."""
        for pi in range(len(dmat.pages)):
            qm = dmat.page_qm(pi)  # loads disk-spilled pages on demand
            s, e = dmat.page_offsets[pi], dmat.page_offsets[pi + 1]
            gg = qm.global_gidx().cpu().numpy()
            for t in range(lo, hi):
                tree = self.trees[t]
                pos = tree.predict_leaf_bins(gg, dmat.cuts)
                vals = tree.split_cond[:tree.n_nodes][pos]
                out[s:e, self.tree_info[t]] += self._tw(t) * torch.as_tensor(
                    vals, device=out.device)
        return out

    def _tree_range(self, iteration_range: Tuple[int, int]) -> Tuple[int, int]:
        lo_it, hi_it = iteration_range
        if hi_it == 0:
            return (self.iteration_indptr[min(lo_it, len(self.iteration_indptr) - 1)],
                    len(self.trees))
        hi_it = min(hi_it, self.num_boosted_rounds())
        return (self.iteration_indptr[lo_it], self.iteration_indptr[hi_it])

    def predict(self, data: DMatrix, output_margin: bool = False,
                pred_leaf: bool = False, pred_contribs: bool = False,
                approx_contribs: bool = False, pred_interactions: bool = False,
                validate_features: bool = True,
                training: bool = False,
                iteration_range: Tuple[int, int] = (0, 0),
                strict_shape: bool = False) -> np.ndarray:
        if not isinstance(data, DMatrix):
            raise TypeError("predict expects a DMatrix; see inplace_predict")
        if validate_features and self.n_features is not None \
                and data.num_col() != self.n_features:
            raise ValueError(
                f"feature mismatch: {data.num_col()} vs {self.n_features}")
        if validate_features and self.feature_names and data.feature_names \
                and list(self.feature_names) != list(data.feature_names):
            # reference core.py _validate_features: names (and order)
            # must match the training frame
            trained = set(self.feature_names)
            given = set(data.feature_names)
            raise ValueError(
                "feature_names mismatch: "
                f"{list(self.feature_names)} vs {list(data.feature_names)}"
                + (f"; missing from data: {sorted(trained - given)}"
                   if trained - given else "")
                + (f"; unexpected in data: {sorted(given - trained)}"
                   if given - trained else ""))
        if pred_leaf:
            lo, hi = self._tree_range(iteration_range)
            X = data.raw_data()
            out = np.stack([self.trees[t].predict_leaf_np(X, data.missing)
                            for t in range(lo, hi)], axis=1)
            out = out.astype(np.float32)
            if strict_shape:
                # reference: (n, n_iterations, n_groups, n_parallel_tree)
                npt = max(1, self.tparam.num_parallel_tree)
                per_iter = max(1, (self.iteration_indptr[1]
                                   - self.iteration_indptr[0]))
                n_iter = (hi - lo) // per_iter
                out = out.reshape(out.shape[0], n_iter,
                                  per_iter // npt, npt)
            return out
        if pred_interactions:
            from .shap import shap_interactions
            out = shap_interactions(self, data, iteration_range)
            if strict_shape and out.ndim == 3:
                out = out[:, None]  # (n, groups=1, ncol+1, ncol+1)
            return out
        if pred_contribs:
            from .shap import shap_values
            out = shap_values(self, data, iteration_range,
                              approx=approx_contribs)
            if strict_shape and out.ndim == 2:
                out = out[:, None]  # (n, groups=1, ncol+1)
            return out
        margin = self._predict_margin(data, iteration_range)
        if output_margin:
            res = margin
        else:
            res = self.objective.pred_transform(margin)
        arr = res.cpu().numpy()
        if not strict_shape and arr.ndim == 2 and arr.shape[1] == 1:
            arr = arr.reshape(-1)
        return arr

    def inplace_predict(self, data, iteration_range=(0, 0),
                        predict_type: str = "value", missing: float = np.nan,
                        validate_features: bool = True, base_margin=None,
                        strict_shape: bool = False) -> np.ndarray:
        """Predict straight from a user array with NO DMatrix
        materialization (reference GBTree::InplacePredict via
        ProxyDMatrix, src/gbm/gbtree.cc / src/data/proxy_dmatrix.h):
        numpy / torch (cpu or cuda) / __cuda_array_interface__ inputs
        are wrapped in a zero-copy proxy and fed directly to the
        predictor kernels; forest device arrays are cached across calls
        so repeated serving calls upload nothing but the rows."""
        from .data import _ProxyMatrix
        proxy = _ProxyMatrix.wrap(data, missing=missing,
                                  base_margin=base_margin,
                                  device=self.device)
        if proxy is None:  # unsupported input shape: full DMatrix path
            d = DMatrix(data, missing=missing, base_margin=base_margin)
            return self.predict(d, output_margin=(predict_type == "margin"),
                                validate_features=validate_features,
                                iteration_range=iteration_range,
                                strict_shape=strict_shape)
        if validate_features and self.n_features is not None \
                and proxy.num_col() != self.n_features:
            raise ValueError(
                f"feature mismatch: {proxy.num_col()} vs {self.n_features}")
        margin = self._predict_margin(proxy, iteration_range)
        res = margin if predict_type == "margin" \
            else self.objective.pred_transform(margin)
        arr = res.cpu().numpy()
        if not strict_shape and arr.ndim == 2 and arr.shape[1] == 1:
            arr = arr.reshape(-1)
        return arr

    # -- evaluation ----------------------------------------------------
    def eval_set(self, evals: Sequence[Tuple[DMatrix, str]],
                 iteration: int = 0, feval=None,
                 output_margin: bool = True) -> str:
        parts = [f"[{iteration}]"]
        metric_names = self._metric_names()
        for dmat, name in evals:
            margin = self._cached_margin(dmat)
            transformed = self.objective.eval_transform(margin)
            # metrics stay device-resident when the margin is on GPU
            # (reference device AUC / elementwise metrics, auc.cu:168);
            # torch-unaware metrics fall back to host inside the wrapper
            tv = transformed.detach()
            if tv.dim() == 2 and tv.shape[1] == 1:
                tv = tv.reshape(-1)
            tnp = None
            for mname in metric_names:
                m = create_metric(mname)
                val = m(tv, dmat.info)
                parts.append(f"{name}-{mname}:{val:.5f}" if abs(val) >= 1e-5
                             else f"{name}-{mname}:{val:g}")
            if feval is not None:
                tnp = tv.cpu().numpy()
                res = feval(tnp, dmat)
                if isinstance(res, list):
                    for mn, v in res:
                        parts.append(f"{name}-{mn}:{v:g}")
                else:
                    mn, v = res
                    parts.append(f"{name}-{mn}:{v:g}")
        return "\t".join(parts)

    def _metric_names(self) -> List[str]:
        em = self.raw_params.get("eval_metric")
        if em is None:
            if self.raw_params.get("disable_default_eval_metric"):
                return []
            dm = self.objective.default_metric
            return [dm]
        if isinstance(em, (list, tuple)):
            return [str(m) for m in em]
        return [str(em)]

    def eval(self, data: DMatrix, name: str = "eval", iteration: int = 0) -> str:
        return self.eval_set([(data, name)], iteration)

    # -- attributes ----------------------------------------------------
    def attr(self, key: str) -> Optional[str]:
        return self.attributes_.get(key)

    def set_attr(self, **kwargs) -> None:
        for k, v in kwargs.items():
            if v is None:
                self.attributes_.pop(k, None)
            else:
                self.attributes_[k] = str(v)

    def attributes(self) -> Dict[str, str]:
        return dict(self.attributes_)

    def set_param(self, params, value=None) -> None:
        if isinstance(params, str):
            params = {params: value}
        elif isinstance(params, (list, tuple)):
            params = dict(params)
        self.raw_params.update(canonicalize(dict(params)))
        self.tparam = make_train_param(self.raw_params)
        self.device = _resolve_device(self.raw_params)

    def load_config(self, config: str) -> None:
        """Apply a configuration produced by save_config (reference
        Learner::LoadConfig, learner.cc:560): parameters only — model
        state is untouched."""
        cfg = json.loads(config)
        learner = cfg.get("learner", {})
        params = {}
        params.update(learner.get("learner_train_param", {}))
        gb = learner.get("gradient_booster", {})
        params.update(gb.get("tree_train_param", {}))
        obj = learner.get("objective", {})
        if obj.get("name"):
            params["objective"] = obj["name"]
        gp = learner.get("generic_param", {})
        if gp.get("device"):
            params["device"] = gp["device"]
        if gp.get("seed"):
            params["seed"] = gp["seed"]
        params.pop("booster", None)
        self.set_param({k: v for k, v in params.items() if v != "None"})

    def get_categories(self, export_to_arrow: bool = False):
        """Per-feature category values recorded at train time
        (reference Booster.get_categories); None when the model was
        trained without categorical features."""
        if export_to_arrow:
            raise NotImplementedError("arrow export is not supported")
        cats = getattr(self, "cat_categories_", None)
        if not cats:
            return None
        names = self.feature_names
        return {(names[f] if names else f"f{f}"): list(v)
                for f, v in cats.items()}

    def save_config(self) -> str:
        """JSON dump of the effective configuration (reference
        Learner::SaveConfig, learner.cc:630)."""
        import dataclasses as dc
        cfg = {
            "learner": {
                "generic_param": {"device": str(self.device),
                                  "seed": str(self.seed)},
                "gradient_booster": {"name": self.booster_kind,
                                     "tree_train_param": {
                                         k: str(v) for k, v in
                                         dc.asdict(self.tparam).items()
                                         if v is not None}},
                "learner_train_param": {
                    "booster": self.booster_kind,
                    "objective": self.objective.name},
                "objective": self.objective.save_config(),
            },
            "version": list(VERSION),
        }
        return json.dumps(cfg)

    # -- model IO ------------------------------------------------------
    def save_model(self, fname: str) -> None:
        j = self._model_to_json()
        if str(fname).endswith(".ubj"):
            from .ubjson import dump_ubjson
            with open(fname, "wb") as fh:
                dump_ubjson(j, fh)
        else:
            with open(fname, "w") as fh:
                json.dump(j, fh)

    def save_raw(self, raw_format: str = "ubj") -> bytearray:
        j = self._model_to_json()
        if raw_format == "json":
            return bytearray(json.dumps(j).encode())
        from .ubjson import dumps_ubjson
        return bytearray(dumps_ubjson(j))

    def load_model(self, fname) -> None:
        if isinstance(fname, (bytes, bytearray)):
            data = bytes(fname)
            j = _parse_model_bytes(data, force_ubj=False)
        else:
            with open(fname, "rb") as fh:
                data = fh.read()
            j = _parse_model_bytes(data,
                                   force_ubj=str(fname).endswith(".ubj"))
        self._model_from_json(j)

    def _model_to_json(self) -> dict:
        if self.booster_kind == "gblinear" and self._linear is not None:
            gb = self._linear.to_json()
        else:
            gb = {
                "model": {
                    "gbtree_model_param": {
                        "num_trees": str(len(self.trees)),
                        "num_parallel_tree": str(self.tparam.num_parallel_tree),
                    },
                    "iteration_indptr": list(self.iteration_indptr),
                    "tree_info": list(self.tree_info),
                    "trees": [t.to_json(i) for i, t in enumerate(self.trees)],
                },
                "name": "gbtree",
            }
            if self.weight_drop:
                # DART tree weights: the reference LOADS this key from the
                # gbtree object (gbtree.cc:456 compat path) but its writer
                # currently drops it; we write it so round-trips keep the
                # weights and reference builds can still read the file
                gb["weight_drop"] = [float(w) for w in self.weight_drop]
        learner = {
            "attributes": dict(self.attributes_),
            "feature_names": self.feature_names or [],
            "feature_types": self.feature_types or [],
            "gradient_booster": gb,
            "learner_model_param": {
                "base_score": f"{self.base_score if self.base_score is not None else 0.5:.9E}",
                "boost_from_average": "1",
                "num_class": str(getattr(self.objective, "num_class", 0)
                                 if self.objective.task == "multiclass" else 0),
                "num_feature": str(self.n_features or 0),
                "num_target": str(self.n_targets),
            },
            "objective": self.objective.save_config(),
        }
        return {"learner": learner, "version": list(VERSION)}

    def _model_from_json(self, j: dict) -> None:
        learner = j["learner"]
        lmp = learner["learner_model_param"]
        self.n_features = int(lmp["num_feature"])
        self.n_targets = max(1, int(lmp.get("num_target", "1")))
        base_score_s = lmp.get("base_score", "5E-1")
        self.base_score = float(base_score_s)
        self._base_score_estimated = True
        obj_cfg = learner["objective"]
        obj_name = obj_cfg["name"]
        obj_params = dict(self.raw_params)
        for v in obj_cfg.values():
            if isinstance(v, dict):
                obj_params.update(v)
        num_class = int(lmp.get("num_class", "0"))
        if num_class > 1:
            obj_params["num_class"] = num_class
        self.objective = create_objective(obj_name, obj_params)
        self.raw_params["objective"] = obj_name
        gb = learner["gradient_booster"]
        if gb.get("name") == "dart":
            # legacy DART format: {"name": "dart", "gbtree": {...},
            # "weight_drop": [...]} (gbtree.cc:452-463)
            wd = gb.get("weight_drop") or []
            gb = dict(gb["gbtree"])
            gb.setdefault("name", "gbtree")
            if wd:
                gb["weight_drop"] = wd
        if gb.get("name") == "gblinear":
            from .linear import GBLinearModel
            self.booster_kind = "gblinear"
            self._linear = GBLinearModel.from_json(
                gb, self.n_features, max(1, num_class, self.n_targets),
                self.raw_params, self.device)
            self.trees = []
            self.tree_info = []
            self.iteration_indptr = [0]
            self.attributes_ = {k: str(v) for k, v in
                                learner.get("attributes", {}).items()}
            self._cache.clear()
            self._ops_cache.clear()
            return
        model = gb["model"]
        self.trees = [RegTree.from_json(t) for t in model["trees"]]
        self.tree_info = [int(x) for x in model["tree_info"]]
        self.weight_drop = [float(w) for w in gb.get("weight_drop", [])]
        if len(self.weight_drop) > len(self.trees):
            raise ValueError("weight_drop longer than the tree list")
        indptr = model.get("iteration_indptr")
        if indptr:
            self.iteration_indptr = [int(x) for x in indptr]
        else:
            n_group = max(1, num_class)
            per_iter = n_group * self.tparam.num_parallel_tree
            self.iteration_indptr = list(
                range(0, len(self.trees) + 1, per_iter))
        self.attributes_ = {k: str(v) for k, v in
                            learner.get("attributes", {}).items()}
        fn = learner.get("feature_names") or []
        ft = learner.get("feature_types") or []
        self.feature_names = list(fn) if fn else None
        self.feature_types = list(ft) if ft else None
        self._cache.clear()
        self._ops_cache.clear()

    def __getstate__(self):
        state = {"raw": bytes(self.save_raw("json")),
                 "params": self.raw_params}
        return state

    def __setstate__(self, state):
        self.__init__(state["params"])
        self.load_model(state["raw"])

    def __copy__(self):
        return self.copy()

    def __deepcopy__(self, memo):
        return self.copy()

    def copy(self) -> "Booster":
        b = Booster(self.raw_params)
        b.load_model(self.save_raw("json"))
        return b

    def __iter__(self):
        """Per-iteration slices (reference core.py Booster.__iter__)."""
        for i in range(self.num_boosted_rounds()):
            yield self[i]

    def __getitem__(self, val) -> "Booster":
        if isinstance(val, int):
            n = self.num_boosted_rounds()
            if val < -n or val >= n:
                raise IndexError(
                    f"index {val} out of range for {n} boosted rounds")
            if val < 0:
                val += n
            val = slice(val, val + 1)
        lo, hi, step = val.indices(self.num_boosted_rounds())
        b = Booster(self.raw_params)
        b.n_features = self.n_features
        b.n_targets = self.n_targets
        b.base_score = self.base_score
        b._base_score_estimated = True
        b.feature_names = self.feature_names
        b.feature_types = self.feature_types
        b.iteration_indptr = [0]
        for it in range(lo, hi, step):
            s, e = self.iteration_indptr[it], self.iteration_indptr[it + 1]
            for t in range(s, e):
                b.trees.append(self.trees[t])
                b.tree_info.append(self.tree_info[t])
                # DART weights follow their trees (reference
                # GBTree::Slice, gbtree.cc:625-631)
                if self.weight_drop:
                    b.weight_drop.append(self._tw(t))
            b.iteration_indptr.append(len(b.trees))
        return b

    def slice(self, begin: int, end: int = 0, step: int = 1) -> "Booster":
        return self[slice(begin, end or self.num_boosted_rounds(), step)]

    # -- introspection -------------------------------------------------
    def _fmap_names(self, fmap: str) -> Optional[List[str]]:
        """Parse a featmap.txt file (reference format:
        '<index>\t<name>\t<type>' per line)."""
        if not fmap or not os.path.exists(fmap):
            return self.feature_names
        names: Dict[int, str] = {}
        with open(fmap) as fh:
            for line in fh:
                parts = line.strip().split("\t")
                if len(parts) >= 2:
                    names[int(parts[0])] = parts[1]
        if not names:
            return self.feature_names
        n = max(names) + 1
        return [names.get(i, f"f{i}") for i in range(n)]

    def get_dump(self, fmap: str = "", with_stats: bool = False,
                 dump_format: str = "text") -> List[str]:
        names = self._fmap_names(fmap)
        return [t.dump(names, with_stats, dump_format) for t in self.trees]

    def dump_model(self, fout, fmap: str = "", with_stats: bool = False,
                   dump_format: str = "text") -> None:
        dumps = self.get_dump(fmap, with_stats, dump_format)
        with open(fout, "w") as fh:
            if dump_format == "json":
                fh.write("[\n" + ",\n".join(dumps) + "\n]")
            else:
                for i, d in enumerate(dumps):
                    fh.write(f"booster[{i}]:\n{d}")

    def get_score(self, fmap: str = "", importance_type: str = "weight"
                  ) -> Dict[str, float]:
        if importance_type not in ("weight", "gain", "cover",
                                   "total_gain", "total_cover"):
            raise ValueError(f"unknown importance_type {importance_type}")
        if self.booster_kind == "gblinear":
            # reference gblinear.cc:210 FeatureScore: the per-feature
            # coefficients (bias excluded); only `weight` is defined
            if importance_type != "weight":
                raise ValueError("gblinear only has `weight` defined "
                                 "for feature importance.")
            if self._linear is None:
                return {}
            names = self._fmap_names(fmap)
            w = self._linear.weights[:-1].detach().cpu().numpy()

            def lname(f):
                return names[f] if names and f < len(names) else f"f{f}"

            if w.shape[1] == 1:
                return {lname(f): float(w[f, 0]) for f in range(w.shape[0])}
            return {lname(f): [float(x) for x in w[f]]
                    for f in range(w.shape[0])}
        counts: Dict[int, float] = {}
        sums: Dict[int, float] = {}
        for tree in self.trees:
            for nid in range(tree.n_nodes):
                if tree.is_leaf(nid):
                    continue
                f = int(tree.split_index[nid])
                counts[f] = counts.get(f, 0.0) + 1.0
                v = (float(tree.loss_chg[nid]) if "gain" in importance_type
                     else float(tree.sum_hess[nid]))
                sums[f] = sums.get(f, 0.0) + v
        names = self._fmap_names(fmap)

        def fname(f):
            return names[f] if names and f < len(names) else f"f{f}"

        if importance_type == "weight":
            return {fname(f): c for f, c in counts.items()}
        if importance_type.startswith("total"):
            return {fname(f): sums[f] for f in sums}
        return {fname(f): sums[f] / counts[f] for f in sums}

    def get_split_value_histogram(self, feature: str, fmap: str = "",
                                  bins=None, as_pandas: bool = True):
        """Histogram of split values for a feature across all trees
        (reference core.py get_split_value_histogram)."""
        names = self.feature_names
        values = []
        for t in self.trees:
            for nid in range(t.n_nodes):
                if t.is_leaf(nid):
                    continue
                f = int(t.split_index[nid])
                fn = names[f] if names and f < len(names) else f"f{f}"
                if fn == feature or f"f{f}" == feature:
                    values.append(float(t.split_cond[nid]))
        values = np.asarray(values, dtype=np.float64)
        nbins = max(min(int(bins) if bins else 10, values.size), 1)
        hist, edges = np.histogram(values, bins=nbins)
        out = np.column_stack([edges[1:], hist])
        if as_pandas:
            try:
                import pandas as pd
                return pd.DataFrame(out, columns=["SplitValue", "Count"])
            except ImportError:
                pass
        return out

    def get_fscore(self, fmap: str = "") -> Dict[str, float]:
        return self.get_score(fmap, "weight")

    def trees_to_dataframe(self, fmap: str = ""):
        """reference core.py trees_to_dataframe (:3259): columns
        Tree/Target/Node/ID/Feature/Split/Yes/No/Missing/Gain/Cover/
        Category (Category holds the go-right category list for
        categorical splits)."""
        import pandas as pd
        rows = []
        for ti, tree in enumerate(self.trees):
            for nid in range(tree.n_nodes):
                leaf = tree.is_leaf(nid)
                is_cat = (not leaf and tree.split_type[nid] == 1)
                cats = (list(map(int, tree.cat_segments.get(nid, [])))
                        if is_cat else None)
                rows.append({
                    "Tree": ti, "Target": 0, "Node": nid,
                    "ID": f"{ti}-{nid}",
                    "Feature": "Leaf" if leaf else
                    (self.feature_names[tree.split_index[nid]]
                     if self.feature_names else f"f{tree.split_index[nid]}"),
                    "Split": None if (leaf or is_cat)
                    else float(tree.split_cond[nid]),
                    "Yes": None if leaf else f"{ti}-{tree.left[nid]}",
                    "No": None if leaf else f"{ti}-{tree.right[nid]}",
                    "Missing": None if leaf else
                    f"{ti}-{tree.left[nid] if tree.default_left[nid] else tree.right[nid]}",
                    "Gain": float(tree.split_cond[nid]) if leaf
                    else float(tree.loss_chg[nid]),
                    "Cover": float(tree.sum_hess[nid]),
                    "Category": cats,
                })
        return pd.DataFrame(rows)

    def reset(self) -> "Booster":
        self._cache.clear()
        self._ops_cache.clear()
        return self


def _parse_model_bytes(data: bytes, force_ubj: bool) -> dict:
    """Both JSON and UBJSON models may begin with b'{' — try JSON text
    first (cheap), fall back to UBJSON."""
    if not force_ubj:
        try:
            return json.loads(data)
        except (UnicodeDecodeError, json.JSONDecodeError):
            pass
    from .ubjson import loads_ubjson
    return loads_ubjson(data)


def _sample_rate_threshold(score: torch.Tensor, p: float) -> float:
    """Threshold u such that sum(min(score/u, 1)) == p*n (reference
    sampler.cu:18 SampleRateDelta), via binary search."""
    n = score.numel()
    target = p * n
    lo = float(score.min()) * 1e-6 + 1e-12
    hi = float(score.max()) / max(p, 1e-6) + 1e-6
    for _ in range(64):
        mid = 0.5 * (lo + hi)
        s = float(torch.clamp(score / mid, max=1.0).sum())
        if s > target:
            lo = mid
        else:
            hi = mid
    return 0.5 * (lo + hi)
