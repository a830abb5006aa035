"""Training callbacks (parity target: reference python-package/lightgbm/callback.py)."""
import collections

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


def log_evaluation(period=1, show_stdv=True):
    """Log evaluation results every `period` iterations."""
    def _callback(env):
        if period > 0 and env.evaluation_result_list and (env.iteration + 1) % period == 0:
            result = "\t".join(
                [f"{name}'s {metric}: {value:g}" if len(r) == 4 else str(r)
                 for r in env.evaluation_result_list
                 for (name, metric, value, _) in [r[:4]]])
            print(f"[{env.iteration + 1}]\t{result}")
    _callback.order = 10
    return _callback


def record_evaluation(eval_result):
    """Record evaluation results into the supplied dict."""
    if not isinstance(eval_result, dict):
        raise TypeError("eval_result must be a dict")

    def _init(env):
        eval_result.clear()
        for r in env.evaluation_result_list or []:
            name, metric = r[0], r[1]
            eval_result.setdefault(name, collections.OrderedDict()).setdefault(metric, [])

    def _callback(env):
        if not eval_result:
            _init(env)
        for r in env.evaluation_result_list or []:
            name, metric, value = r[0], r[1], r[2]
            eval_result.setdefault(name, collections.OrderedDict()).setdefault(metric, []).append(value)
    _callback.order = 20
    return _callback


def reset_parameter(**kwargs):
    """Reset parameters on a schedule: value is a list (per iteration) or a callable."""
    def _callback(env):
        new_params = {}
        for key, value in kwargs.items():
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
    _callback.before_iteration = True
    _callback.order = 10
    return _callback


def early_stopping(stopping_rounds, first_metric_only=False, verbose=True, min_delta=0.0):
    """Stop training when a validation metric stops improving."""
    best_score = []
    best_iter = []
    best_score_list = []
    cmp_op = []
    enabled = [True]
    first_metric = [""]

    def _init(env):
        enabled[0] = bool(env.evaluation_result_list)
        if not enabled[0]:
            import warnings
            warnings.warn("Early stopping requires at least one validation set")
            return
        best_score.clear(); best_iter.clear(); best_score_list.clear(); cmp_op.clear()
        first_metric[0] = env.evaluation_result_list[0][1]
        n_metric = len(env.evaluation_result_list)
        deltas = [min_delta] * n_metric if not isinstance(min_delta, list) else min_delta
        for i, r in enumerate(env.evaluation_result_list):
            best_iter.append(0)
            best_score_list.append(None)
            if r[3]:  # higher better
                best_score.append(float("-inf"))
                cmp_op.append(lambda cur, best, d=deltas[i]: cur > best + d)
            else:
                best_score.append(float("inf"))
                cmp_op.append(lambda cur, best, d=deltas[i]: cur < best - d)

    def _callback(env):
        if not best_score:
            _init(env)
        if not enabled[0]:
            return
        for i, r in enumerate(env.evaluation_result_list):
            name, metric, score = r[0], r[1], r[2]
            if name == "training":
                continue
            if first_metric_only and metric != first_metric[0]:
                continue
            if best_score_list[i] is None or cmp_op[i](score, best_score[i]):
                best_score[i] = score
                best_iter[i] = env.iteration
                best_score_list[i] = env.evaluation_result_list
            elif env.iteration - best_iter[i] >= stopping_rounds:
                if verbose:
                    print(f"Early stopping, best iteration is: [{best_iter[i] + 1}]")
                raise EarlyStopException(best_iter[i], best_score_list[i])
        if env.iteration == env.end_iteration - 1:
            for i in range(len(best_iter)):
                if best_score_list[i] is not None:
                    raise EarlyStopException(best_iter[i], best_score_list[i])
    _callback.order = 30
    return _callback
