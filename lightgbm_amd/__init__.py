"""lightgbm_amd — MI355X-native gradient boosting with the LightGBM API.

Drop-in `import lightgbm_amd as lgb` replacement for the reference Python package
(python-package/lightgbm/__init__.py parity), backed by lib_migbm.so: a from-scratch
C++/HIP (gfx950) implementation with RCCL-over-xGMI multi-GPU training.
"""
from .basic import Booster, Dataset, LightGBMError, Sequence, register_logger
from .callback import EarlyStopException, early_stopping, log_evaluation, \
    record_evaluation, reset_parameter
from .engine import CVBooster, cv, train

try:
    from .sklearn import LGBMClassifier, LGBMModel, LGBMRanker, LGBMRegressor
    _SKLEARN_EXPORTS = ["LGBMModel", "LGBMRegressor", "LGBMClassifier", "LGBMRanker"]
except ImportError:
    _SKLEARN_EXPORTS = []

try:
    from .dask import DaskLGBMClassifier, DaskLGBMRanker, DaskLGBMRegressor
    _DASK_EXPORTS = ["DaskLGBMRegressor", "DaskLGBMClassifier", "DaskLGBMRanker"]
except ImportError:
    _DASK_EXPORTS = []

try:
    from .plotting import create_tree_digraph, plot_importance, plot_metric, \
        plot_split_value_histogram, plot_tree
    _PLOT_EXPORTS = ["plot_importance", "plot_metric", "plot_tree", "create_tree_digraph",
                     "plot_split_value_histogram"]
except ImportError:
    _PLOT_EXPORTS = []

__version__ = "0.1.0"

__all__ = [
    "Dataset", "Booster", "LightGBMError", "register_logger", "Sequence",
    "train", "cv", "CVBooster",
    "early_stopping", "log_evaluation", "record_evaluation", "reset_parameter",
    "EarlyStopException",
] + _SKLEARN_EXPORTS + _DASK_EXPORTS + _PLOT_EXPORTS
