"""Training callbacks (parity target: reference python-package/lightgbm/callback.py).

Callbacks are CLASSES (factory functions return instances) so they pickle —
sklearn estimators carrying callbacks must survive joblib/pickle round-trips
(reference _EarlyStoppingCallback et al.).
"""
import collections
import warnings

__all__ = ["early_stopping", "log_evaluation", "record_evaluation", "reset_parameter",
           "EarlyStopException", "CallbackEnv"]

CallbackEnv = collections.namedtuple(
    "CallbackEnv",
    ["model", "params", "iteration", "begin_iteration", "end_iteration", "evaluation_result_list"])


class EarlyStopException(Exception):
    """Raised by the early_stopping callback to stop training."""

    def __init__(self, best_iteration, best_score):
        super().__init__()
        self.best_iteration = best_iteration
        self.best_score = best_score


class _LogEvaluationCallback:
    order = 10

    def __init__(self, period=1, show_stdv=True):
        self.period = period
        self.show_stdv = show_stdv

    def __call__(self, env):
        if self.period > 0 and env.evaluation_result_list and \
                (env.iteration + 1) % self.period == 0:
            result = "\t".join(
                [f"{name}'s {metric}: {value:g}" if len(r) == 4 else str(r)
                 for r in env.evaluation_result_list
                 for (name, metric, value, _) in [r[:4]]])
            print(f"[{env.iteration + 1}]\t{result}")


def log_evaluation(period=1, show_stdv=True):
    """Log evaluation results every `period` iterations."""
    return _LogEvaluationCallback(period, show_stdv)


class _RecordEvaluationCallback:
    order = 20

    def __init__(self, eval_result):
        if not isinstance(eval_result, dict):
            raise TypeError("eval_result must be a dict")
        self.eval_result = eval_result
        self._started = False

    def _init(self, env):
        self.eval_result.clear()
        for r in env.evaluation_result_list or []:
            name, metric = r[0], r[1]
            self.eval_result.setdefault(name, collections.OrderedDict()).setdefault(metric, [])

    def __call__(self, env):
        if not self._started:
            self._init(env)
            self._started = True
        for r in env.evaluation_result_list or []:
            name, metric, value = r[0], r[1], r[2]
            self.eval_result.setdefault(name, collections.OrderedDict()) \
                .setdefault(metric, []).append(value)


def record_evaluation(eval_result):
    """Record evaluation results into the supplied dict."""
    return _RecordEvaluationCallback(eval_result)


class _ResetParameterCallback:
    order = 10
    before_iteration = True

    def __init__(self, kwargs):
        self.kwargs = kwargs

    def __call__(self, env):
        new_params = {}
        for key, value in self.kwargs.items():
            if isinstance(value, list):
                if len(value) != env.end_iteration - env.begin_iteration:
                    raise ValueError(f"Length of list {key!r} must match num_boost_round")
                new_params[key] = value[env.iteration - env.begin_iteration]
            elif callable(value):
                new_params[key] = value(env.iteration - env.begin_iteration)
            else:
                raise ValueError("value must be a list or callable")
        if new_params:
            env.model.reset_parameter(new_params)
            env.params.update(new_params)


def reset_parameter(**kwargs):
    """Reset parameters on a schedule: value is a list (per iteration) or a callable."""
    return _ResetParameterCallback(kwargs)


class _EarlyStoppingCallback:
    order = 30

    def __init__(self, stopping_rounds, first_metric_only=False, verbose=True,
                 min_delta=0.0):
        self.stopping_rounds = stopping_rounds
        self.first_metric_only = first_metric_only
        self.verbose = verbose
        self.min_delta = min_delta
        self._reset()

    def _reset(self):
        self.best_score = []
        self.best_iter = []
        self.best_score_list = []
        self.higher_better = []
        self.deltas = []
        self.enabled = True
        self.first_metric = ""

    def _init(self, env):
        self._reset()
        self.enabled = bool(env.evaluation_result_list)
        if not self.enabled:
            warnings.warn("Early stopping requires at least one validation set")
            return
        self.first_metric = env.evaluation_result_list[0][1]
        n_metric = len(env.evaluation_result_list)
        self.deltas = list(self.min_delta) if isinstance(self.min_delta, list) \
            else [self.min_delta] * n_metric
        for r in env.evaluation_result_list:
            self.best_iter.append(0)
            self.best_score_list.append(None)
            self.higher_better.append(bool(r[3]))
            self.best_score.append(float("-inf") if r[3] else float("inf"))

    def _improved(self, i, cur):
        if self.higher_better[i]:
            return cur > self.best_score[i] + self.deltas[i]
        return cur < self.best_score[i] - self.deltas[i]

    def __call__(self, env):
        if env.iteration == env.begin_iteration or not self.best_score:
            self._init(env)  # fresh state per training run (instances are reusable)
        if not self.enabled:
            return
        for i, r in enumerate(env.evaluation_result_list):
            name, metric, score = r[0], r[1], r[2]
            if name == "training":
                continue
            if self.first_metric_only and metric != self.first_metric:
                continue
            if self.best_score_list[i] is None or self._improved(i, score):
                self.best_score[i] = score
                self.best_iter[i] = env.iteration
                self.best_score_list[i] = env.evaluation_result_list
            elif env.iteration - self.best_iter[i] >= self.stopping_rounds:
                if self.verbose:
                    print(f"Early stopping, best iteration is: [{self.best_iter[i] + 1}]")
                raise EarlyStopException(self.best_iter[i], self.best_score_list[i])
        if env.iteration == env.end_iteration - 1:
            for i in range(len(self.best_iter)):
                if self.best_score_list[i] is not None:
                    raise EarlyStopException(self.best_iter[i], self.best_score_list[i])


def early_stopping(stopping_rounds, first_metric_only=False, verbose=True, min_delta=0.0):
    """Stop training when a validation metric stops improving."""
    return _EarlyStoppingCallback(stopping_rounds, first_metric_only, verbose, min_delta)
