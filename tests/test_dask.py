"""Dask-module tests. dask itself is not installed in this image, so the
dask-independent primitives (the reference's _train_part / _machines_to_worker_map
logic over the TCP socket mesh) are tested with real multi-process workers; the
client-orchestration layer is import-checked."""
import json
import os
import subprocess
import sys
from pathlib import Path

import numpy as np
import pytest

REPO = Path(__file__).resolve().parent.parent

WORKER = r"""
import json, sys
sys.path.insert(0, sys.argv[1])
import numpy as np
from lightgbm_amd.dask import _train_part
from lightgbm_amd.sklearn import LGBMClassifier

cfg = json.loads(sys.argv[2])
rank = cfg["rank"]
rng = np.random.RandomState(100 + rank)
n = 6000
X = rng.randn(n, 6)
y = (X[:, 0] + 0.5 * X[:, 1] + 0.2 * rng.randn(n) > 0).astype(int)

model = _train_part({"n_estimators": 15, "num_leaves": 31, "verbosity": -1},
                    LGBMClassifier, [{"X": X, "y": y, "w": None, "g": None}],
                    cfg["machines"], cfg["port"], cfg["num_machines"])
text = model.booster_.model_to_string()
open(cfg["out"], "w").write(text)
print("PART_OK", (model.predict(X) == y).mean())
"""


def _free_ports(n):
    import socket
    out = []
    for _ in range(n):
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        out.append(s.getsockname()[1])
        s.close()
    return out


def test_train_part_socket_mesh(tmp_path):
    """The reference _train_part contract: N processes join the TCP mesh, fit
    sklearn estimators with tree_learner=data, and every worker ends with the
    identical model (Dask estimators actually fit)."""
    script = tmp_path / "worker.py"
    script.write_text(WORKER)
    ports = _free_ports(2)
    machines = ",".join(f"127.0.0.1:{p}" for p in ports)
    procs = []
    for r in range(2):
        cfg = {"rank": r, "machines": machines, "port": ports[r],
               "num_machines": 2, "out": str(tmp_path / f"m{r}.txt")}
        procs.append(subprocess.Popen(
            [sys.executable, str(script), str(REPO), json.dumps(cfg)],
            stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True))
    outs = [p.communicate(timeout=240) for p in procs]
    for p, (so, se) in zip(procs, outs):
        assert p.returncode == 0, so[-2000:] + se[-2000:]
        assert "PART_OK" in so
    m0 = (tmp_path / "m0.txt").read_text()
    m1 = (tmp_path / "m1.txt").read_text()
    assert m0[m0.index("Tree=0"):] == m1[m1.index("Tree=0"):]


def test_machines_to_worker_map():
    from lightgbm_amd.dask import _machines_to_worker_map, _find_n_open_ports
    addrs = ["tcp://127.0.0.1:42001", "tcp://127.0.0.1:42002"]
    ports = _find_n_open_ports(2)
    m = _machines_to_worker_map(addrs, ports)
    assert m[addrs[0]] == f"127.0.0.1:{ports[0]}"
    assert m[addrs[1]] == f"127.0.0.1:{ports[1]}"


def test_dask_estimators_importable():
    from lightgbm_amd.dask import (DaskLGBMClassifier, DaskLGBMRegressor,
                                   DaskLGBMRanker, DASK_INSTALLED)
    est = DaskLGBMRegressor(n_estimators=5)
    assert est.get_params()["n_estimators"] == 5
    if not DASK_INSTALLED:
        with pytest.raises(Exception):
            est.fit(np.zeros((10, 2)), np.zeros(10))
