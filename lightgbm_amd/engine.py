"""Training/CV entry points (parity target: reference python-package/lightgbm/engine.py)."""
import collections
import copy

import numpy as np
from pathlib import Path

from . import callback as callback_mod
from .basic import Booster, Dataset

__all__ = ["train", "cv", "CVBooster"]


def train(params, train_set, num_boost_round=100, valid_sets=None, valid_names=None,
          feval=None, init_model=None, keep_training_booster=False, callbacks=None,
          fobj=None):
    """Train a booster (parity: reference engine.py:train)."""
    params = copy.deepcopy(params) if params else {}
    # alias handling for num_boost_round in params
    for alias in ("num_iterations", "num_iteration", "n_iter", "num_tree", "num_trees",
                  "num_round", "num_rounds", "num_boost_round", "n_estimators", "max_iter"):
        if alias in params:
            num_boost_round = int(params.pop(alias))
    if num_boost_round <= 0:
        raise ValueError("num_boost_round must be greater than 0")
    if not isinstance(train_set, Dataset):
        raise TypeError(f"train_set must be a Dataset, not {type(train_set).__name__}")
    first_metric_only = params.get("first_metric_only", False)
    if fobj is not None or callable(params.get("objective")):
        if callable(params.get("objective")):
            fobj = params["objective"]
        params["objective"] = "none"

    cbs = list(callbacks) if callbacks else []
    # early_stopping_round in params spawns the callback
    es_min_delta = 0.0
    for d_alias in ("early_stopping_min_delta",):
        if d_alias in params:
            es_min_delta = float(params.pop(d_alias))
    for alias in ("early_stopping_round", "early_stopping_rounds", "early_stopping",
                  "n_iter_no_change"):
        if alias in params and params[alias]:
            cbs.append(callback_mod.early_stopping(int(params[alias]),
                                                   first_metric_only=first_metric_only,
                                                   min_delta=es_min_delta))
            params.pop(alias)
            break
    if params.get("verbosity", params.get("verbose", 1)) >= 1 and not any(
            getattr(c, "__name__", "") == "_callback" and getattr(c, "order", 0) == 10
            for c in cbs):
        pass  # reference logs every iteration only with log_evaluation callback

    before_cbs = [c for c in cbs if getattr(c, "before_iteration", False)]
    after_cbs = [c for c in cbs if not getattr(c, "before_iteration", False)]
    after_cbs.sort(key=lambda c: getattr(c, "order", 0))

    if not isinstance(train_set, Dataset):
        raise TypeError("train_set must be a Dataset")
    # dataset-relevant params given at train() time flow into construction
    _DATASET_PARAMS = ("max_bin", "min_data_in_bin", "bin_construct_sample_cnt",
                       "use_missing", "zero_as_missing", "feature_pre_filter",
                       "linear_tree", "data_random_seed", "enable_bundle",
                       "max_conflict_rate", "categorical_feature",
                       "forcedbins_filename", "precise_float_parser", "max_bin_by_feature")
    if train_set._handle is None:
        for k in _DATASET_PARAMS:
            if k in params and k not in train_set.params:
                train_set.params[k] = params[k]
    train_set.construct()

    booster = Booster(params=params, train_set=train_set)
    if init_model is not None:
        if isinstance(init_model, Booster):
            init_model = init_model.model_to_string()
            other = Booster(model_str=init_model)
        else:
            other = Booster(model_file=init_model)
        from .basic import _LIB, _safe_call
        _safe_call(_LIB.LGBM_BoosterMerge(booster._handle, other._handle))

    if valid_sets:
        if isinstance(valid_sets, Dataset):
            valid_sets = [valid_sets]
        names = valid_names or [f"valid_{i}" for i in range(len(valid_sets))]
        for vs, name in zip(valid_sets, names):
            if vs is train_set:
                # training data as eval set: use data_idx 0 via eval_train naming
                booster._train_as_valid_name = name
                continue
            # a valid set must share the training data's bin mappers — bind it
            # before construction (reference engine.py does the same); an
            # independently-binned valid set would score trees on wrong bins
            if vs._handle is None and train_set not in vs.get_ref_chain():
                vs.set_reference(train_set)
            vs.construct()
            booster.add_valid(vs, name)

    evaluation_result_list = []
    for i in range(num_boost_round):
        env = callback_mod.CallbackEnv(model=booster, params=params, iteration=i,
                                       begin_iteration=0, end_iteration=num_boost_round,
                                       evaluation_result_list=None)
        for cb in before_cbs:
            cb(env)
        is_finished = booster.update(fobj=fobj)
        evaluation_result_list = []
        if valid_sets or feval is not None:
            if getattr(booster, "_train_as_valid_name", None) is not None:
                for (n, m, v, hb) in booster.eval_train(feval):
                    evaluation_result_list.append((booster._train_as_valid_name, m, v, hb))
            evaluation_result_list.extend(booster.eval_valid(feval))
        env = callback_mod.CallbackEnv(model=booster, params=params, iteration=i,
                                       begin_iteration=0, end_iteration=num_boost_round,
                                       evaluation_result_list=evaluation_result_list)
        try:
            for cb in after_cbs:
                cb(env)
        except callback_mod.EarlyStopException as e:
            booster.best_iteration = e.best_iteration + 1
            evaluation_result_list = e.best_score or []
            break
        if is_finished:
            break

    # record best score
    for r in evaluation_result_list or []:
        name, metric, value = r[0], r[1], r[2]
        booster.best_score.setdefault(name, collections.OrderedDict())[metric] = value
    if not keep_training_booster:
        booster.free_dataset()
    return booster


class CVBooster:
    """Container of per-fold boosters (parity: reference CVBooster)."""

    def __init__(self, model_file=None):
        self.boosters = []
        self.best_iteration = -1
        if model_file is not None:
            text = Path(model_file).read_text()
            for chunk in text.split("!!! cv booster fold separator !!!"):
                chunk = chunk.strip()
                if chunk:
                    self.boosters.append(Booster(model_str=chunk))

    def _append(self, booster):
        self.boosters.append(booster)

    def save_model(self, filename, num_iteration=None, start_iteration=0,
                   importance_type="split"):
        parts = [b.model_to_string(num_iteration=num_iteration,
                                   start_iteration=start_iteration,
                                   importance_type=importance_type)
                 for b in self.boosters]
        Path(filename).write_text("\n!!! cv booster fold separator !!!\n".join(parts))
        return self

    def model_to_string(self, **kwargs):
        return "\n!!! cv booster fold separator !!!\n".join(
            b.model_to_string(**kwargs) for b in self.boosters)

    def __getstate__(self):
        return {"best_iteration": self.best_iteration,
                "_models": [b.model_to_string() for b in self.boosters]}

    def __setstate__(self, state):
        self.best_iteration = state.get("best_iteration", -1)
        self.boosters = [Booster(model_str=m) for m in state.get("_models", [])]

    def __getattr__(self, name):
        if name.startswith("__") and name.endswith("__"):
            raise AttributeError(name)

        def handler_function(*args, **kwargs):
            return [getattr(b, name)(*args, **kwargs) for b in self.boosters]
        return handler_function


def _make_n_folds(full_data, nfold, params, seed, stratified, shuffle):
    num_data = full_data.num_data()
    group = full_data.get_group()
    rng = np.random.RandomState(seed)
    if group is not None:
        # group-aware folds: assign whole queries to folds
        ngroups = len(group)
        gidx = np.arange(ngroups)
        if shuffle:
            rng.shuffle(gidx)
        folds = []
        group_row_start = np.concatenate([[0], np.cumsum(group)])
        for k in range(nfold):
            test_g = gidx[k::nfold]
            mask = np.zeros(num_data, dtype=bool)
            for g in test_g:
                mask[group_row_start[g]:group_row_start[g + 1]] = True
            folds.append((np.where(~mask)[0], np.where(mask)[0]))
        return folds
    idx = np.arange(num_data)
    if stratified:
        label = full_data.get_label()
        folds = []
        pos = idx[label > 0]
        neg = idx[label <= 0]
        if shuffle:
            rng.shuffle(pos)
            rng.shuffle(neg)
        for k in range(nfold):
            test = np.concatenate([pos[k::nfold], neg[k::nfold]])
            mask = np.zeros(num_data, dtype=bool)
            mask[test] = True
            folds.append((idx[~mask], idx[mask]))
        return folds
    if shuffle:
        rng.shuffle(idx)
    folds = []
    for k in range(nfold):
        test = idx[k::nfold]
        mask = np.zeros(num_data, dtype=bool)
        mask[test] = True
        folds.append((np.arange(num_data)[~mask], np.arange(num_data)[mask]))
    return folds


def cv(params, train_set, num_boost_round=100, folds=None, nfold=5, stratified=True,
       shuffle=True, metrics=None, feval=None, init_model=None,
       callbacks=None, eval_train_metric=False, return_cvbooster=False, seed=0,
       fobj=None, fpreproc=None):
    """Cross-validation (parity: reference engine.py:cv)."""
    if num_boost_round <= 0:
        raise ValueError("num_boost_round must be greater than 0")
    if not isinstance(train_set, Dataset):
        raise TypeError(f"train_set must be a Dataset, not {type(train_set).__name__}")
    params = copy.deepcopy(params) if params else {}
    if metrics is not None:
        params["metric"] = metrics
    # callable objective in params (sklearn-style) trains with objective=none
    if fobj is not None or callable(params.get("objective")):
        if callable(params.get("objective")):
            fobj = params["objective"]
        params["objective"] = "none"
    train_set.construct()
    if folds is None:
        obj = params.get("objective", "")
        strat = stratified and obj in ("binary", "multiclass", "multiclassova")
        folds = _make_n_folds(train_set, nfold, params, seed, strat, shuffle)
    elif hasattr(folds, "split"):
        label = train_set.get_label()
        folds = list(folds.split(np.zeros(train_set.num_data()), label))

    cvbooster = CVBooster()
    fold_data = []
    for (train_idx, test_idx) in folds:
        tr = train_set.subset(sorted(train_idx))
        te = train_set.subset(sorted(test_idx))
        fold_params = params
        if fpreproc is not None:
            # per-fold preprocessing hook: (train, test, params) -> same triple
            # (reference engine.py cv fpreproc)
            tr, te, fold_params = fpreproc(tr, te, copy.deepcopy(params))
        bst = Booster(params=fold_params, train_set=tr)
        if init_model is not None:
            # continue every fold booster from the given model (reference
            # cv(init_model=...) semantics)
            from .basic import _LIB, _safe_call
            base = Booster(model_str=init_model.model_to_string()) \
                if isinstance(init_model, Booster) else Booster(model_file=str(init_model))
            _safe_call(_LIB.LGBM_BoosterMerge(bst._handle, base._handle))
        bst.add_valid(te, "valid")
        cvbooster._append(bst)
        fold_data.append((tr, te))

    cbs = list(callbacks) if callbacks else []
    es_cb = None
    for alias in ("early_stopping_round", "early_stopping_rounds", "early_stopping"):
        if alias in params and params[alias]:
            es_cb = callback_mod.early_stopping(int(params[alias]))
            params.pop(alias)
    for c in cbs:
        if getattr(c, "order", None) == 30:
            es_cb = c
    results = collections.defaultdict(list)
    for i in range(num_boost_round):
        agg = collections.defaultdict(list)
        train_agg = collections.defaultdict(list)
        for bst in cvbooster.boosters:
            bst.update(fobj=fobj)
            for (name, metric, value, hb) in bst.eval_valid(feval):
                agg[metric].append((value, hb))
            if eval_train_metric:
                for (name, metric, value, hb) in bst.eval_train(feval):
                    train_agg[metric].append((value, hb))
        line = []
        for prefix, bucket in (("train", train_agg), ("valid", agg)):
            for metric, vals in bucket.items():
                vs = [v for v, _ in vals]
                results[f"{prefix} {metric}-mean"].append(float(np.mean(vs)))
                results[f"{prefix} {metric}-stdv"].append(float(np.std(vs)))
                line.append((("cv_agg"), f"{prefix} {metric}", float(np.mean(vs)),
                             vals[0][1]))
        if es_cb is not None:
            env = callback_mod.CallbackEnv(model=cvbooster, params=params, iteration=i,
                                           begin_iteration=0, end_iteration=num_boost_round,
                                           evaluation_result_list=line)
            try:
                es_cb(env)
            except callback_mod.EarlyStopException as e:
                cvbooster.best_iteration = e.best_iteration + 1
                for k in results:
                    results[k] = results[k][:cvbooster.best_iteration]
                break
    out = dict(results)
    if return_cvbooster:
        out["cvbooster"] = cvbooster
    return out
