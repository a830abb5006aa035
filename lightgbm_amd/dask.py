"""Dask distributed training (parity target: reference python-package/lightgbm/dask.py).

Per-worker training runs over the standalone TCP socket mesh (LGBM_NetworkInit /
cpp/src/socket_linker.cpp) — no torch, no injected collectives. `_train_part`
and `_machines_to_worker_map` mirror the reference (dask.py:187-206, 389) and are
dask-independent (the multi-process path is tested without a Dask cluster); the
estimator `fit` methods orchestrate them over a Dask client when dask is present.
"""
import socket
from urllib.parse import urlparse

import numpy as np

from .basic import LightGBMError, _LIB, _c_str, _safe_call
from .sklearn import LGBMClassifier, LGBMModel, LGBMRanker, LGBMRegressor

__all__ = ["DaskLGBMClassifier", "DaskLGBMRegressor", "DaskLGBMRanker"]

try:
    import dask  # noqa: F401
    from dask import delayed  # noqa: F401
    from dask.distributed import default_client, wait
    DASK_INSTALLED = True
except ImportError:
    DASK_INSTALLED = False


def _require_dask():
    if not DASK_INSTALLED:
        raise LightGBMError(
            "dask is not installed. The per-worker primitives (_train_part over the "
            "TCP socket mesh) work without it; for multi-process training without "
            "Dask use the CLI (tree_learner=data machines=...) or torchrun + "
            "lightgbm_amd.parallel."
        )


def _find_n_open_ports(n):
    """n free listen ports on this host (reference dask.py _find_n_open_ports)."""
    sockets, ports = [], []
    for _ in range(n):
        s = socket.socket()
        s.bind(("", 0))
        sockets.append(s)
        ports.append(s.getsockname()[1])
    for s in sockets:
        s.close()
    return ports


def _machines_to_worker_map(worker_addresses, ports):
    """worker address -> 'ip:port' machine entry (reference dask.py:389)."""
    out = {}
    for addr, port in zip(worker_addresses, ports):
        host = urlparse(addr).hostname or "127.0.0.1"
        try:
            ip = socket.gethostbyname(host)
        except OSError:
            ip = host
        out[addr] = f"{ip}:{port}"
    return out


def _concat_parts(parts):
    """Stack a worker's list of (X, y[, w][, g]) parts."""
    Xs = [p["X"] for p in parts]
    X = np.vstack([np.asarray(x) for x in Xs])
    y = np.concatenate([np.asarray(p["y"]).ravel() for p in parts])
    w = None
    if parts[0].get("w") is not None:
        w = np.concatenate([np.asarray(p["w"]).ravel() for p in parts])
    g = None
    if parts[0].get("g") is not None:
        g = np.concatenate([np.asarray(p["g"]).ravel() for p in parts])
    return X, y, w, g


def _train_part(params, model_factory, parts, machines, local_listen_port,
                num_machines, time_out=120, return_model=True, **fit_kwargs):
    """Train this worker's shard inside the socket mesh (reference _train_part).

    Joins the TCP full mesh at `machines`/`local_listen_port`, fits the sklearn
    estimator with tree_learner=data (bin mappers are synchronized through the
    mesh by Dataset::ConstructFromMat), leaves the mesh, and returns the fitted
    model (identical on every worker) or None.
    """
    X, y, w, g = _concat_parts(parts if isinstance(parts, list) else [parts])
    params = dict(params)
    params.setdefault("tree_learner", "data")
    params.pop("machines", None)
    params.pop("num_machines", None)
    params.pop("local_listen_port", None)
    _safe_call(_LIB.LGBM_NetworkInit(_c_str(machines), int(local_listen_port),
                                     int(time_out), int(num_machines)))
    try:
        model = model_factory(**params)
        if g is not None:
            model.fit(X, y, sample_weight=w, group=g, **fit_kwargs)
        else:
            model.fit(X, y, sample_weight=w, **fit_kwargs)
    finally:
        _safe_call(_LIB.LGBM_NetworkFree())
    return model if return_model else None


def _split_to_parts(data, n):
    idx = np.array_split(np.arange(len(data)), n)
    return [data[i] for i in idx]


def _train(client, X, y, params, model_factory, sample_weight=None, group=None,
           **kwargs):
    """Distributed fit over a Dask client (reference dask.py _train, condensed):
    materialize each worker's partitions, open one port per worker, run
    _train_part everywhere, keep worker 0's (identical) model."""
    _require_dask()
    workers = list(client.scheduler_info()["workers"].keys())
    if not workers:
        raise LightGBMError("no Dask workers available")
    # scatter row blocks round-robin over workers
    n = len(workers)
    X = np.asarray(X)
    y = np.asarray(y)
    parts_per_worker = {wk: [] for wk in workers}
    for i, wk in enumerate(workers):
        rows = np.arange(i, len(X), n)
        if len(rows) == 0:
            continue
        part = {"X": X[rows], "y": y[rows],
                "w": None if sample_weight is None else np.asarray(sample_weight)[rows],
                "g": None}
        parts_per_worker[wk].append(part)
    if group is not None:
        raise LightGBMError("Dask ranking uses pre-partitioned group arrays; pass "
                            "group-aligned partitions via dask arrays")
    active = [wk for wk in workers if parts_per_worker[wk]]
    ports = _find_n_open_ports(len(active))
    machines_map = _machines_to_worker_map(active, ports)
    machines = ",".join(machines_map[wk] for wk in active)
    futures = []
    for rank, wk in enumerate(active):
        futures.append(client.submit(
            _train_part, params, model_factory, parts_per_worker[wk], machines,
            ports[rank], len(active), return_model=(rank == 0), workers=[wk],
            allow_other_workers=False, pure=False, **kwargs))
    wait(futures)
    results = client.gather(futures)
    for r in results:
        if r is not None:
            return r
    raise LightGBMError("no worker returned a model")


class _DaskBase:
    _factory = None

    def fit(self, X, y, sample_weight=None, client=None, **kwargs):
        _require_dask()
        client = client or default_client()
        params = self.get_params()
        model = _train(client, X, y, params, type(self)._factory,
                       sample_weight=sample_weight, **kwargs)
        # adopt the fitted state (reference _lgb_dask_copy semantics)
        self.__dict__.update(model.__dict__)
        return self

    def predict(self, X, **kwargs):
        return type(self)._factory.predict(self, X, **kwargs)


class DaskLGBMRegressor(_DaskBase, LGBMRegressor):
    _factory = LGBMRegressor


class DaskLGBMClassifier(_DaskBase, LGBMClassifier):
    _factory = LGBMClassifier


class DaskLGBMRanker(_DaskBase, LGBMRanker):
    _factory = LGBMRanker
