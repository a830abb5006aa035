"""scikit-learn estimator wrappers (parity target: reference python-package/lightgbm/sklearn.py)."""

import numpy as np
from pathlib import Path

from . import callback as callback_mod
from .basic import Booster, Dataset, LightGBMError
from .compat import SKLEARN_INSTALLED
from .engine import train as train_fn

__all__ = ["LGBMModel", "LGBMRegressor", "LGBMClassifier", "LGBMRanker"]

if SKLEARN_INSTALLED:
    from sklearn.base import BaseEstimator, ClassifierMixin, RegressorMixin
else:  # minimal shims so the module imports without sklearn
    class BaseEstimator:  # noqa: D401
        pass

    class ClassifierMixin:
        pass

    class RegressorMixin:
        pass


class LGBMModel(BaseEstimator):
    """Base estimator over the migbm Booster."""

    def __init__(self, boosting_type="gbdt", num_leaves=31, max_depth=-1,
                 learning_rate=0.1, n_estimators=100, subsample_for_bin=200000,
                 objective=None, class_weight=None, min_split_gain=0.0,
                 min_child_weight=1e-3, min_child_samples=20, subsample=1.0,
                 subsample_freq=0, colsample_bytree=1.0, reg_alpha=0.0, reg_lambda=0.0,
                 random_state=None, n_jobs=None, importance_type="split", **kwargs):
        self.boosting_type = boosting_type
        self.num_leaves = num_leaves
        self.max_depth = max_depth
        self.learning_rate = learning_rate
        self.n_estimators = n_estimators
        self.subsample_for_bin = subsample_for_bin
        self.objective = objective
        self.class_weight = class_weight
        self.min_split_gain = min_split_gain
        self.min_child_weight = min_child_weight
        self.min_child_samples = min_child_samples
        self.subsample = subsample
        self.subsample_freq = subsample_freq
        self.colsample_bytree = colsample_bytree
        self.reg_alpha = reg_alpha
        self.reg_lambda = reg_lambda
        self.random_state = random_state
        self.n_jobs = n_jobs
        self.importance_type = importance_type
        self._other_params = dict(kwargs)
        self._Booster = None
        self._n_features = None
        self._classes = None
        self._n_classes = 1
        self._evals_result = {}
        self._best_iteration = -1
        self._objective = objective

    def get_params(self, deep=True):
        params = {
            "boosting_type": self.boosting_type, "num_leaves": self.num_leaves,
            "max_depth": self.max_depth, "learning_rate": self.learning_rate,
            "n_estimators": self.n_estimators, "subsample_for_bin": self.subsample_for_bin,
            "objective": self.objective, "class_weight": self.class_weight,
            "min_split_gain": self.min_split_gain, "min_child_weight": self.min_child_weight,
            "min_child_samples": self.min_child_samples, "subsample": self.subsample,
            "subsample_freq": self.subsample_freq, "colsample_bytree": self.colsample_bytree,
            "reg_alpha": self.reg_alpha, "reg_lambda": self.reg_lambda,
            "random_state": self.random_state, "n_jobs": self.n_jobs,
            "importance_type": self.importance_type,
        }
        params.update(self._other_params)
        return params

    def set_params(self, **params):
        for k, v in params.items():
            if hasattr(self, k):
                setattr(self, k, v)
            else:
                self._other_params[k] = v
        return self

    def _make_params(self, default_objective):
        params = self.get_params()
        params.pop("importance_type", None)
        params.pop("n_estimators", None)
        params.pop("class_weight", None)
        obj = params.pop("objective", None) or default_objective
        ren = {
            "boosting_type": "boosting",
            "min_split_gain": "min_gain_to_split",
            "min_child_weight": "min_sum_hessian_in_leaf",
            "min_child_samples": "min_data_in_leaf",
            "subsample": "bagging_fraction",
            "subsample_freq": "bagging_freq",
            "colsample_bytree": "feature_fraction",
            "reg_alpha": "lambda_l1",
            "reg_lambda": "lambda_l2",
            "random_state": "seed",
            "n_jobs": "num_threads",
            "subsample_for_bin": "bin_construct_sample_cnt",
        }
        out = {}
        for k, v in params.items():
            if v is None:
                continue
            out[ren.get(k, k)] = v
        out["objective"] = obj
        if isinstance(out.get("seed"), np.random.RandomState):
            out["seed"] = int(out["seed"].randint(0, 2**31 - 1))
        elif hasattr(np.random, "Generator") and isinstance(out.get("seed"), np.random.Generator):
            out["seed"] = int(out["seed"].integers(0, 2**31 - 1))
        if out.get("bagging_fraction", 1.0) < 1.0 and out.get("bagging_freq", 0) == 0:
            out["bagging_freq"] = 1
        out.setdefault("verbosity", -1)
        return out

    def _fit(self, X, y, default_objective, sample_weight=None, init_score=None,
             group=None, eval_set=None, eval_names=None, eval_sample_weight=None,
             eval_group=None, eval_metric=None, early_stopping_rounds=None,
             callbacks=None, categorical_feature="auto", feature_name="auto",
             init_model=None):
        params = self._make_params(default_objective)
        # sklearn-style CALLABLE objective: (y_true, y_pred[, weight[, group]]) ->
        # (grad, hess); adapted to the engine's (preds, dataset) form
        fobj = None
        if callable(params.get("objective")):
            user_obj = params["objective"]
            params["objective"] = "none"

            def fobj(preds, dataset):
                import inspect
                args = [dataset.get_label(), preds]
                n_par = len(inspect.signature(user_obj).parameters)
                if n_par >= 3:
                    args.append(dataset.get_weight())
                if n_par >= 4:
                    args.append(dataset.get_group())
                g, h = user_obj(*args)
                return np.asarray(g, dtype=np.float32), np.asarray(h, dtype=np.float32)
        feval = None
        if callable(eval_metric):
            user_metric = eval_metric

            def feval(preds, dataset):
                res = user_metric(dataset.get_label(), preds)
                return res if isinstance(res, tuple) else ("metric", float(res), False)
            params["metric"] = "none"
        elif eval_metric is not None:
            params["metric"] = eval_metric
        from .compat import PANDAS_INSTALLED, pd_DataFrame
        if PANDAS_INSTALLED and isinstance(X, pd_DataFrame):
            self._n_features = X.shape[1]  # Dataset handles category-dtype mapping
        else:
            X = np.asarray(X, dtype=np.float64)
            self._n_features = X.shape[1]
        sw = sample_weight
        if self.class_weight is not None and self._classes is not None:
            cw = self.class_weight
            if cw == "balanced":
                counts = np.bincount(y.astype(int), minlength=len(self._classes))
                weights_per_class = len(y) / (len(self._classes) * np.maximum(counts, 1))
            else:
                weights_per_class = np.array([cw.get(c, 1.0) for c in self._classes])
            cw_arr = weights_per_class[y.astype(int)]
            sw = cw_arr if sw is None else np.asarray(sw) * cw_arr
        train_set = Dataset(X, label=y, weight=sw, init_score=init_score, group=group,
                            params=params, categorical_feature=categorical_feature,
                            feature_name=feature_name)
        valid_sets = []
        names = []
        if eval_set is not None:
            if isinstance(eval_set, tuple):
                eval_set = [eval_set]
            for i, (vx, vy) in enumerate(eval_set):
                vw = eval_sample_weight[i] if eval_sample_weight else None
                vg = eval_group[i] if eval_group else None
                if not (PANDAS_INSTALLED and isinstance(vx, pd_DataFrame)):
                    vx = np.asarray(vx, dtype=np.float64)
                valid_sets.append(train_set.create_valid(vx, label=vy, weight=vw,
                                                         group=vg))
                names.append(eval_names[i] if eval_names else f"valid_{i}")
        cbs = list(callbacks) if callbacks else []
        if early_stopping_rounds:
            cbs.append(callback_mod.early_stopping(early_stopping_rounds, verbose=False))
        self._evals_result = {}
        cbs.append(callback_mod.record_evaluation(self._evals_result))
        if init_model is not None and not isinstance(init_model, (str, Path, Booster)):
            init_model = init_model.booster_   # an estimator
        self._Booster = train_fn(params, train_set, num_boost_round=self.n_estimators,
                                 valid_sets=valid_sets or None,
                                 valid_names=names or None, callbacks=cbs,
                                 fobj=fobj, feval=feval, init_model=init_model)
        self._best_iteration = self._Booster.best_iteration
        self._objective = self.objective if callable(self.objective) else params["objective"]
        return self

    def __sklearn_is_fitted__(self):
        return self._Booster is not None

    # ------------------------------------------------------------ properties
    @property
    def booster_(self):
        if self._Booster is None:
            raise LightGBMError("Estimator not fitted")
        return self._Booster

    @property
    def n_features_(self):
        return self._n_features

    @property
    def n_features_in_(self):
        return self._n_features

    @property
    def best_iteration_(self):
        return self._best_iteration

    @property
    def feature_names_in_(self):
        return np.array(self.booster_.feature_name())

    @property
    def n_estimators_(self):
        """actual number of fitted iterations (early stopping aware)."""
        return self._best_iteration if self._best_iteration > 0 \
            else self.booster_.current_iteration()

    @property
    def n_iter_(self):
        return self.n_estimators_

    @property
    def best_score_(self):
        return self.booster_.best_score

    @property
    def evals_result_(self):
        return self._evals_result

    @property
    def feature_importances_(self):
        return self.booster_.feature_importance(importance_type=self.importance_type)

    @property
    def feature_name_(self):
        return self.booster_.feature_name()

    @property
    def objective_(self):
        return self._objective

    def predict(self, X, raw_score=False, start_iteration=0, num_iteration=None,
                pred_leaf=False, pred_contrib=False, **kwargs):
        from .compat import PANDAS_INSTALLED, pd_DataFrame
        if not (PANDAS_INSTALLED and isinstance(X, pd_DataFrame)):
            X = np.asarray(X, dtype=np.float64)
        return self.booster_.predict(X, raw_score=raw_score,
                                     start_iteration=start_iteration,
                                     num_iteration=num_iteration, pred_leaf=pred_leaf,
                                     pred_contrib=pred_contrib)


class LGBMRegressor(RegressorMixin, LGBMModel):
    def fit(self, X, y, sample_weight=None, init_score=None, eval_set=None, **kwargs):
        y = np.asarray(y, dtype=np.float32).ravel()
        return self._fit(X, y, "regression", sample_weight=sample_weight,
                         init_score=init_score, eval_set=eval_set, **kwargs)

    def score(self, X, y, sample_weight=None):
        pred = self.predict(X)
        y = np.asarray(y, dtype=np.float64).ravel()
        u = ((y - pred) ** 2).sum()
        v = ((y - y.mean()) ** 2).sum()
        return 1.0 - u / v if v > 0 else 0.0


class LGBMClassifier(ClassifierMixin, LGBMModel):
    def fit(self, X, y, sample_weight=None, init_score=None, eval_set=None, **kwargs):
        y = np.asarray(y).ravel()
        self._classes = np.unique(y)
        self._n_classes = len(self._classes)
        class_to_idx = {c: i for i, c in enumerate(self._classes)}
        y_enc = np.array([class_to_idx[v] for v in y], dtype=np.float32)
        if self._n_classes <= 2:
            obj = "binary"
            self._other_params.pop("num_class", None)  # classes re-detected each fit
        else:
            obj = "multiclass"
            self._other_params["num_class"] = self._n_classes
        if eval_set is not None:
            if isinstance(eval_set, tuple):
                eval_set = [eval_set]
            eval_set = [(vx, np.array([class_to_idx[v] for v in np.asarray(vy).ravel()],
                                      dtype=np.float32)) for vx, vy in eval_set]
        return self._fit(X, y_enc, obj, sample_weight=sample_weight,
                         init_score=init_score, eval_set=eval_set, **kwargs)

    @property
    def classes_(self):
        return self._classes

    @property
    def n_classes_(self):
        return self._n_classes

    def predict_proba(self, X, raw_score=False, start_iteration=0, num_iteration=None,
                      **kwargs):
        res = super().predict(X, raw_score=raw_score, start_iteration=start_iteration,
                              num_iteration=num_iteration)
        if raw_score:
            return res
        if self._n_classes <= 2:
            res = np.asarray(res).reshape(-1)
            return np.column_stack([1.0 - res, res])
        return res

    def predict(self, X, raw_score=False, start_iteration=0, num_iteration=None,
                pred_leaf=False, pred_contrib=False, **kwargs):
        if raw_score or pred_leaf or pred_contrib:
            return super().predict(X, raw_score=raw_score, start_iteration=start_iteration,
                                   num_iteration=num_iteration, pred_leaf=pred_leaf,
                                   pred_contrib=pred_contrib)
        proba = self.predict_proba(X, start_iteration=start_iteration,
                                   num_iteration=num_iteration)
        return self._classes[np.argmax(proba, axis=1)]

    def score(self, X, y, sample_weight=None):
        return float((self.predict(X) == np.asarray(y).ravel()).mean())


class LGBMRanker(LGBMModel):
    def fit(self, X, y, group=None, sample_weight=None, init_score=None, eval_set=None,
            eval_group=None, eval_at=(1, 2, 3, 4, 5), **kwargs):
        if group is None:
            raise ValueError("LGBMRanker requires group information")
        y = np.asarray(y, dtype=np.float32).ravel()
        self._other_params.setdefault("eval_at", list(eval_at))
        return self._fit(X, y, "lambdarank", sample_weight=sample_weight,
                         init_score=init_score, group=np.asarray(group, dtype=np.int32),
                         eval_set=eval_set, eval_group=eval_group, **kwargs)
