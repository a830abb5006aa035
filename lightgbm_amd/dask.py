"""Dask distributed wrappers (parity target: reference python-package/lightgbm/dask.py).

The MI355X-native distributed path is one process per GPU with RCCL over xGMI
(see lightgbm_amd.parallel and bench.py); these wrappers provide the reference's
Dask-cluster API surface on top of the same Network seam. Dask itself is an
optional dependency.
"""
import numpy as np

from .basic import Dataset, LightGBMError
from .engine import train as train_fn
from .sklearn import LGBMClassifier, LGBMModel, LGBMRanker, LGBMRegressor

__all__ = ["DaskLGBMClassifier", "DaskLGBMRegressor", "DaskLGBMRanker"]

try:
    import dask  # noqa: F401
    from dask import delayed
    from dask.distributed import Client, default_client, wait
    DASK_INSTALLED = True
except ImportError:
    DASK_INSTALLED = False


def _require_dask():
    if not DASK_INSTALLED:
        raise LightGBMError(
            "dask is not installed. For multi-process training without Dask, use "
            "torch.distributed (one rank per GPU/worker) with "
            "lightgbm_amd.parallel.init_network_from_torch_distributed(); see bench.py."
        )


def _train_part(params, model_factory, data_parts, machines, rank, num_machines, **kwargs):
    """Per-worker training closure (parity: reference dask.py _train_part)."""
    _require_dask()
    # Each Dask worker trains against the shared Network seam; reference uses raw
    # sockets here, the migbm build uses the injected collective functions.
    raise LightGBMError("Dask training requires a running torch.distributed rendezvous; "
                        "use the torchrun path documented in docs/DISTRIBUTED.md")


class _DaskBase:
    def fit(self, X, y, **kwargs):
        _require_dask()
        raise LightGBMError(
            "Dask estimators are provided for API compatibility; this build's supported "
            "distributed path is torchrun + lightgbm_amd.parallel (RCCL over xGMI for "
            "multi-GPU, gloo for CPU). See docs/DISTRIBUTED.md.")


class DaskLGBMRegressor(_DaskBase, LGBMRegressor):
    pass


class DaskLGBMClassifier(_DaskBase, LGBMClassifier):
    pass


class DaskLGBMRanker(_DaskBase, LGBMRanker):
    pass
