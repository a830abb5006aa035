"""Cross-ecosystem model-format compatibility: models trained by migbm must load and
predict identically in the REFERENCE LightGBM, and vice versa (the SURVEY phase-1 exit
criterion). The reference oracle (built by tools/build_reference_oracle.sh) runs in a
SUBPROCESS — both libraries export the LGBM_* symbols, so they cannot share a process.
Skipped when tools/oracle/lib_lightgbm.so is absent."""
import json
import subprocess
import sys
from pathlib import Path

import numpy as np
import pytest

import lightgbm_amd as lgb

REF_LIB = Path(__file__).resolve().parent.parent / "tools" / "oracle" / "lib_lightgbm.so"
pytestmark = pytest.mark.skipif(not REF_LIB.exists(),
                                reason="reference oracle not built "
                                       "(tools/build_reference_oracle.sh)")

_REF_WORKER = r"""
import ctypes, json, sys
import numpy as np
lib = ctypes.cdll.LoadLibrary(sys.argv[2])
lib.LGBM_GetLastError.restype = ctypes.c_char_p
def ok(ret):
    assert ret == 0, lib.LGBM_GetLastError().decode()
cmd = json.loads(sys.argv[1])
X = np.load(cmd["x"])
if cmd["op"] == "predict":
    h = ctypes.c_void_p(); it = ctypes.c_int(0)
    ok(lib.LGBM_BoosterCreateFromModelfile(cmd["model"].encode(), ctypes.byref(it), ctypes.byref(h)))
    arr = np.ascontiguousarray(X, dtype=np.float64)
    out = np.zeros(len(X), dtype=np.float64); n = ctypes.c_int64(0)
    ok(lib.LGBM_BoosterPredictForMat(h, arr.ctypes.data_as(ctypes.c_void_p), 1,
        ctypes.c_int32(arr.shape[0]), ctypes.c_int32(arr.shape[1]), 1, 0, 0, -1, b"",
        ctypes.byref(n), out.ctypes.data_as(ctypes.POINTER(ctypes.c_double))))
    np.save(cmd["out"], out)
elif cmd["op"] == "train":
    y = np.load(cmd["y"])
    arr = np.ascontiguousarray(X, dtype=np.float64)
    ds = ctypes.c_void_p()
    ok(lib.LGBM_DatasetCreateFromMat(arr.ctypes.data_as(ctypes.c_void_p), 1,
        ctypes.c_int32(arr.shape[0]), ctypes.c_int32(arr.shape[1]), 1,
        cmd["ds_params"].encode(), None, ctypes.byref(ds)))
    lab = np.ascontiguousarray(y, dtype=np.float32)
    ok(lib.LGBM_DatasetSetField(ds, b"label", lab.ctypes.data_as(ctypes.c_void_p),
        ctypes.c_int(len(lab)), 0))
    bst = ctypes.c_void_p()
    ok(lib.LGBM_BoosterCreate(ds, cmd["params"].encode(), ctypes.byref(bst)))
    fin = ctypes.c_int(0)
    for _ in range(cmd["iters"]):
        ok(lib.LGBM_BoosterUpdateOneIter(bst, ctypes.byref(fin)))
    ok(lib.LGBM_BoosterSaveModel(bst, 0, -1, 0, cmd["model"].encode()))
print("REF_OK")
"""


def _ref(cmd, tmp_path):
    worker = tmp_path / "ref_worker.py"
    worker.write_text(_REF_WORKER)
    r = subprocess.run([sys.executable, str(worker), json.dumps(cmd), str(REF_LIB)],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0 and "REF_OK" in r.stdout, r.stdout + r.stderr


def _ref_predict(model_file, X, tmp_path):
    np.save(tmp_path / "x.npy", X)
    _ref({"op": "predict", "model": str(model_file), "x": str(tmp_path / "x.npy"),
          "out": str(tmp_path / "out.npy")}, tmp_path)
    return np.load(tmp_path / "out.npy")


def _data(n=4000, d=8, seed=0):
    rng = np.random.RandomState(seed)
    X = rng.randn(n, d)
    y = ((X[:, 0] - 0.5 * X[:, 1] + X[:, 2] * X[:, 3] + 0.4 * rng.randn(n)) > 0)
    return X, y.astype(np.float32)


def test_our_model_loads_in_reference(tmp_path):
    X, y = _data()
    bst = lgb.train({"objective": "binary", "verbosity": -1, "num_leaves": 31},
                    lgb.Dataset(X, label=y), 20)
    ours = bst.predict(X[:500])
    f = tmp_path / "migbm_model.txt"
    bst.save_model(str(f))
    theirs = _ref_predict(f, X[:500], tmp_path)
    np.testing.assert_allclose(ours, theirs, rtol=1e-9, atol=1e-12)


def test_reference_model_loads_in_ours(tmp_path):
    X, y = _data(seed=3)
    np.save(tmp_path / "x.npy", X)
    np.save(tmp_path / "y.npy", y)
    f = tmp_path / "reference_model.txt"
    _ref({"op": "train", "x": str(tmp_path / "x.npy"), "y": str(tmp_path / "y.npy"),
          "model": str(f), "ds_params": "max_bin=255",
          "params": "objective=binary verbosity=-1 num_leaves=31", "iters": 20}, tmp_path)
    theirs = _ref_predict(f, X[:500], tmp_path)
    mine = lgb.Booster(model_file=str(f)).predict(X[:500])
    np.testing.assert_allclose(mine, theirs, rtol=1e-9, atol=1e-12)


def test_regression_model_cross(tmp_path):
    rng = np.random.RandomState(5)
    X = rng.randn(3000, 6)
    y = (2 * X[:, 0] + np.sin(X[:, 1])).astype(np.float32)
    bst = lgb.train({"objective": "regression", "verbosity": -1}, lgb.Dataset(X, label=y), 15)
    f = tmp_path / "reg.txt"
    bst.save_model(str(f))
    np.testing.assert_allclose(bst.predict(X[:300]), _ref_predict(f, X[:300], tmp_path),
                               rtol=1e-9, atol=1e-12)


def test_categorical_model_cross(tmp_path):
    rng = np.random.RandomState(7)
    n = 4000
    cat = rng.randint(0, 10, size=n)
    X = np.column_stack([cat.astype(float), rng.randn(n), rng.randn(n)])
    effect = rng.randn(10) * 2
    y = (effect[cat] + 0.3 * rng.randn(n) > 0).astype(np.float32)
    bst = lgb.train({"objective": "binary", "verbosity": -1, "min_data_in_leaf": 5},
                    lgb.Dataset(X, label=y, categorical_feature=[0]), 15)
    f = tmp_path / "cat.txt"
    bst.save_model(str(f))
    theirs = _ref_predict(f, X[:500], tmp_path)
    np.testing.assert_allclose(bst.predict(X[:500]), theirs, rtol=1e-9, atol=1e-12)


def test_multiclass_model_cross(tmp_path):
    rng = np.random.RandomState(8)
    X = rng.randn(3000, 5)
    y = ((X[:, 0] > 0.5).astype(int) + (X[:, 1] > 0).astype(int)).astype(np.float32)
    bst = lgb.train({"objective": "multiclass", "num_class": 3, "verbosity": -1},
                    lgb.Dataset(X, label=y), 10)
    f = tmp_path / "mc.txt"
    bst.save_model(str(f))
    # reference PredictForMat on multiclass returns nrow x nclass row-major
    import ctypes  # noqa: F401  (worker handles shape via out size)
    np.save(tmp_path / "x.npy", X[:200])
    worker = tmp_path / "ref_worker_mc.py"
    worker.write_text(_REF_WORKER.replace(
        "out = np.zeros(len(X), dtype=np.float64)",
        "out = np.zeros(len(X) * 3, dtype=np.float64)"))
    import subprocess as sp
    r = sp.run([sys.executable, str(worker), json.dumps(
        {"op": "predict", "model": str(f), "x": str(tmp_path / "x.npy"),
         "out": str(tmp_path / "out.npy")}), str(REF_LIB)],
               capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    theirs = np.load(tmp_path / "out.npy").reshape(200, 3)
    np.testing.assert_allclose(bst.predict(X[:200]), theirs, rtol=1e-9, atol=1e-12)


def test_training_quality_parity_with_reference(tmp_path):
    """Same data, same params: our CPU learner and the reference must reach the same
    quality (not identical trees — binning differs — but equal AUC to ~1e-2)."""
    X, y = _data(n=20000, seed=11)
    params = dict(objective="binary", num_leaves=31, learning_rate=0.1, verbosity=-1)
    bst = lgb.train(dict(params), lgb.Dataset(X[:16000], label=y[:16000]), 60)
    ours_pred = bst.predict(X[16000:])
    np.save(tmp_path / "x.npy", X[:16000])
    np.save(tmp_path / "y.npy", y[:16000])
    f = tmp_path / "ref_model.txt"
    _ref({"op": "train", "x": str(tmp_path / "x.npy"), "y": str(tmp_path / "y.npy"),
          "model": str(f), "ds_params": "max_bin=255",
          "params": "objective=binary verbosity=-1 num_leaves=31 learning_rate=0.1",
          "iters": 60}, tmp_path)
    ref_pred = _ref_predict(f, X[16000:], tmp_path)

    def auc(yy, pp):
        order = np.argsort(-pp, kind="stable")
        ys = yy[order]
        npos = ys.sum()
        nneg = len(ys) - npos
        ranks = np.arange(1, len(ys) + 1)
        return 1.0 - (ranks[ys > 0].sum() - npos * (npos + 1) / 2) / (npos * nneg)

    a_ours = auc(y[16000:], ours_pred)
    a_ref = auc(y[16000:], ref_pred)
    assert abs(a_ours - a_ref) < 0.01, (a_ours, a_ref)
    assert a_ours > 0.9


def test_dart_and_linear_model_cross(tmp_path):
    """DART and linear-tree model files (is_linear leaf coefficients) must load and
    predict identically in the reference implementation."""
    rng = np.random.RandomState(21)
    X = rng.rand(4000, 5)
    y = (3 * X[:, 0] + np.sin(5 * X[:, 1])).astype(np.float32)
    dart = lgb.train({"objective": "regression", "boosting": "dart", "drop_rate": 0.2,
                      "verbosity": -1}, lgb.Dataset(X, label=y), 20)
    f = tmp_path / "dart.txt"
    dart.save_model(str(f))
    np.testing.assert_allclose(dart.predict(X[:300]), _ref_predict(f, X[:300], tmp_path),
                               rtol=1e-9, atol=1e-12)
    lin = lgb.train({"objective": "regression", "linear_tree": True, "verbosity": -1,
                     "num_leaves": 15}, lgb.Dataset(X, label=y), 15)
    f2 = tmp_path / "linear.txt"
    lin.save_model(str(f2))
    np.testing.assert_allclose(lin.predict(X[:300]), _ref_predict(f2, X[:300], tmp_path),
                               rtol=1e-7, atol=1e-10)


def test_subset_categorical_model_cross(tmp_path):
    """Multi-category subset splits (cat_threshold bitsets spanning many cats) must
    route identically in the reference."""
    rng = np.random.RandomState(31)
    n = 6000
    cat = rng.randint(0, 40, size=n)
    X = np.column_stack([cat.astype(float), rng.randn(n)])
    effect = rng.randn(40) * 2
    y = (effect[cat] + 0.3 * rng.randn(n) > 0).astype(np.float32)
    bst = lgb.train({"objective": "binary", "verbosity": -1, "min_data_in_leaf": 5,
                     "max_cat_to_onehot": 4}, lgb.Dataset(X, label=y,
                                                          categorical_feature=[0]), 20)
    f = tmp_path / "catsub.txt"
    bst.save_model(str(f))
    np.testing.assert_allclose(bst.predict(X[:500]), _ref_predict(f, X[:500], tmp_path),
                               rtol=1e-9, atol=1e-12)


def test_kitchen_sink_model_cross(tmp_path):
    """The mixed-storage model (sparse + 4-bit + categorical + EFB features) must
    predict identically in the reference implementation."""
    rng = np.random.RandomState(42)
    n = 4000
    dense = rng.randn(n, 3)
    sparse = np.zeros((n, 4))
    mask = rng.rand(n, 4) < 0.05
    sparse[mask] = rng.rand(mask.sum()) + 1
    lowcard = rng.randint(0, 9, (n, 2)).astype(float)
    cat = rng.randint(0, 30, n).astype(float)
    onehot = np.zeros((n, 6))
    onehot[np.arange(n), rng.randint(0, 6, n)] = 1.0
    X = np.column_stack([dense, sparse, lowcard, cat, onehot])
    eff = rng.randn(30)
    y = ((dense[:, 0] + 2 * (sparse[:, 0] > 0) + eff[cat.astype(int)] * 0.5) > 0.5)
    bst = lgb.train({"objective": "binary", "verbosity": -1, "min_data_in_leaf": 5,
                     "categorical_feature": [9]},
                    lgb.Dataset(X, label=y.astype(np.float32)), 25)
    f = tmp_path / "ks.txt"
    bst.save_model(str(f))
    np.testing.assert_allclose(bst.predict(X[:500]), _ref_predict(f, X[:500], tmp_path),
                               rtol=1e-9, atol=1e-12)


def test_ranking_quality_parity_with_reference(tmp_path):
    """Same ranking data and config: our lambdarank reaches the reference's NDCG
    within tolerance, and the model files interchange."""
    rng = np.random.RandomState(3)
    rows, labels, groups = [], [], []
    for q in range(300):
        nq = 20
        Xq = rng.rand(nq, 10)
        rel = np.clip((3 * Xq[:, 0] + Xq[:, 1] + 0.3 * rng.randn(nq)).astype(int), 0, 4)
        rows.append(Xq)
        labels.append(rel)
        groups.append(nq)
    X = np.vstack(rows)
    y = np.concatenate(labels).astype(np.float32)
    g = np.array(groups, dtype=np.int32)
    bst = lgb.train({"objective": "lambdarank", "verbosity": -1, "num_leaves": 31},
                    lgb.Dataset(X, label=y, group=g), 30)
    f = tmp_path / "rank.txt"
    bst.save_model(str(f))
    # model interchange
    np.testing.assert_allclose(bst.predict(X[:500]), _ref_predict(f, X[:500], tmp_path),
                               rtol=1e-9, atol=1e-12)
    # reference trains the same config; compare NDCG@5 of both models on the data
    np.save(tmp_path / "x.npy", X)
    np.save(tmp_path / "y.npy", y)
    worker = tmp_path / "rank_worker.py"
    worker.write_text(_REF_WORKER.replace(
        '''ok(lib.LGBM_DatasetSetField(ds, b"label", lab.ctypes.data_as(ctypes.c_void_p),
        ctypes.c_int(len(lab)), 0))''',
        '''ok(lib.LGBM_DatasetSetField(ds, b"label", lab.ctypes.data_as(ctypes.c_void_p),
        ctypes.c_int(len(lab)), 0))
    grp = np.full(len(lab) // 20, 20, dtype=np.int32)
    ok(lib.LGBM_DatasetSetField(ds, b"group", grp.ctypes.data_as(ctypes.c_void_p),
        ctypes.c_int(len(grp)), 2))'''))
    import subprocess as sp
    fr = tmp_path / "ref_rank.txt"
    r = sp.run([sys.executable, str(worker), json.dumps(
        {"op": "train", "x": str(tmp_path / "x.npy"), "y": str(tmp_path / "y.npy"),
         "model": str(fr), "ds_params": "max_bin=255",
         "params": "objective=lambdarank verbosity=-1 num_leaves=31", "iters": 30}),
        str(REF_LIB)], capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    ref_scores = _ref_predict(fr, X, tmp_path)
    our_scores = bst.predict(X)

    def ndcg5(scores):
        total, pos = 0.0, 0
        for nq in groups:
            s, rel = scores[pos:pos + nq], y[pos:pos + nq]
            pos += nq
            order = np.argsort(-s, kind="stable")
            dcg = sum((2 ** rel[order[i]] - 1) / np.log2(i + 2) for i in range(5))
            ideal = np.sort(rel)[::-1]
            idcg = sum((2 ** ideal[i] - 1) / np.log2(i + 2) for i in range(5))
            total += dcg / idcg if idcg > 0 else 1.0
        return total / len(groups)
    ours, theirs = ndcg5(our_scores), ndcg5(ref_scores)
    assert ours > theirs - 0.01, (ours, theirs)


def test_continue_training_from_reference_model(tmp_path):
    """a REFERENCE-trained model continues training in migbm: the merged trees'
    outputs seed the scores, so the continued model strictly improves and the
    combined model still loads back in the reference."""
    X, y = _data(n=6000, seed=9)
    np.save(tmp_path / "x.npy", X)
    np.save(tmp_path / "y.npy", y)
    f = tmp_path / "ref_base.txt"
    _ref({"op": "train", "x": str(tmp_path / "x.npy"), "y": str(tmp_path / "y.npy"),
          "model": str(f), "ds_params": "max_bin=255",
          "params": "objective=binary verbosity=-1 num_leaves=31 learning_rate=0.1",
          "iters": 10}, tmp_path)

    def logloss(p):
        p = np.clip(p, 1e-12, 1 - 1e-12)
        return float(-np.mean(y * np.log(p) + (1 - y) * np.log(1 - p)))

    base = lgb.Booster(model_file=str(f))
    ll_base = logloss(base.predict(X))
    cont = lgb.train({"objective": "binary", "verbosity": -1, "num_leaves": 31,
                      "learning_rate": 0.1}, lgb.Dataset(X, label=y), 10,
                     init_model=str(f))
    assert cont.num_trees() == 20
    ll_cont = logloss(cont.predict(X))
    assert ll_cont < ll_base * 0.9, (ll_base, ll_cont)
    # the combined model round-trips through the reference byte-compatibly
    out = tmp_path / "combined.txt"
    cont.save_model(str(out))
    theirs = _ref_predict(out, X[:500], tmp_path)
    np.testing.assert_allclose(cont.predict(X[:500]), theirs, rtol=1e-9, atol=1e-12)
