"""Arrow ingestion tests (parity target: reference tests/python_package_test/test_arrow.py)."""
import numpy as np
import pytest

pa = pytest.importorskip("pyarrow")

import lightgbm_amd as lgb


def test_dataset_from_arrow_table():
    rng = np.random.RandomState(0)
    n = 3000
    cols = {f"f{i}": rng.randn(n) for i in range(5)}
    cols["f2"] = rng.randint(0, 100, size=n).astype(np.int64)  # integer column
    table = pa.table(cols)
    y = (np.asarray(cols["f0"]) > 0).astype(np.float32)
    ds = lgb.Dataset(table, label=y).construct()
    assert ds.num_data() == n
    assert ds.num_feature() == 5
    assert ds.get_feature_name() == ["f0", "f1", "f2", "f3", "f4"]
    bst = lgb.train({"objective": "binary", "verbosity": -1}, ds, 10)
    X = np.column_stack([np.asarray(cols[f"f{i}"], dtype=np.float64) for i in range(5)])
    acc = ((bst.predict(X) > 0.5) == y).mean()
    assert acc > 0.9


def test_arrow_with_nulls():
    rng = np.random.RandomState(1)
    n = 1000
    vals = rng.randn(n)
    mask = rng.rand(n) < 0.1
    arr = pa.array(np.where(mask, np.nan, vals), from_pandas=True)  # nulls from NaN
    table = pa.table({"a": arr, "b": pa.array(rng.randn(n))})
    y = (vals > 0).astype(np.float32)
    ds = lgb.Dataset(table, label=y).construct()
    bst = lgb.train({"objective": "binary", "verbosity": -1}, ds, 5)
    assert np.all(np.isfinite(bst.predict(np.column_stack([vals, rng.randn(n)]))))


def test_arrow_multi_chunk():
    rng = np.random.RandomState(2)
    t1 = pa.table({"x": rng.randn(500), "z": rng.randn(500)})
    t2 = pa.table({"x": rng.randn(700), "z": rng.randn(700)})
    table = pa.concat_tables([t1, t2])
    y = np.zeros(1200, dtype=np.float32)
    ds = lgb.Dataset(table, label=y).construct()
    assert ds.num_data() == 1200


def test_predict_for_arrow():
    """Booster.predict accepts a pyarrow Table (LGBM_BoosterPredictForArrow)."""
    pa = pytest.importorskip("pyarrow")
    rng = np.random.RandomState(3)
    X = rng.rand(500, 4)
    y = (X[:, 0] + X[:, 1] > 1.0).astype(np.float64)
    bst = lgb.train({"objective": "binary", "verbosity": -1}, lgb.Dataset(X, label=y), 10)
    table = pa.table({f"f{i}": X[:, i] for i in range(4)})
    pred_arrow = bst.predict(table)
    np.testing.assert_allclose(pred_arrow, bst.predict(X), rtol=1e-12)
    contrib = bst.predict(table, pred_contrib=True)
    assert contrib.shape == (500, 5)


def test_arrow_metadata_fields():
    """label/weight/group/init_score given as Arrow arrays (ref
    test_dataset_construct_{labels,weights,groups,init_scores_array})."""
    pa = pytest.importorskip("pyarrow")
    rng = np.random.RandomState(10)
    n = 400
    tbl = pa.table({"a": rng.randn(n), "b": rng.randn(n)})
    y = rng.rand(n)
    w = 1.0 + rng.rand(n)
    ds = lgb.Dataset(tbl, label=pa.array(y), weight=pa.array(w),
                     init_score=pa.array(np.zeros(n)))
    ds.construct()
    np.testing.assert_allclose(ds.get_label(), y, rtol=1e-6)
    np.testing.assert_allclose(ds.get_weight(), w, rtol=1e-6)
    rel = (tbl["a"].to_numpy() > 0).astype(np.float64)  # integer relevance
    g = lgb.Dataset(tbl, label=pa.array(rel),
                    group=pa.array(np.array([100, 100, 100, 100], dtype=np.int32)))
    g.construct()
    np.testing.assert_array_equal(g.get_group(), [100, 100, 100, 100])
    bst = lgb.train({"objective": "lambdarank", "verbosity": -1}, g, 3)
    assert bst.num_trees() == 3


def test_arrow_feature_names_and_predict_tasks():
    """feature names come from the table schema; predict accepts tables across
    tasks (ref test_arrow_feature_name_auto / test_predict_*)."""
    pa = pytest.importorskip("pyarrow")
    rng = np.random.RandomState(11)
    n = 600
    tbl = pa.table({"f_one": rng.randn(n), "f_two": rng.randn(n),
                    "f_three": rng.randn(n)})
    Xnp = np.column_stack([tbl[c].to_numpy() for c in tbl.column_names])
    yb = (Xnp[:, 0] > 0).astype(float)
    bb = lgb.train({"objective": "binary", "verbosity": -1},
                   lgb.Dataset(tbl, label=yb), 5)
    assert bb.feature_name() == ["f_one", "f_two", "f_three"]
    np.testing.assert_allclose(bb.predict(tbl), bb.predict(Xnp), rtol=1e-12)
    ym = rng.randint(0, 3, n).astype(float)
    bm = lgb.train({"objective": "multiclass", "num_class": 3, "verbosity": -1},
                   lgb.Dataset(tbl, label=ym), 5)
    assert bm.predict(tbl).shape == (n, 3)
    assert bm.predict(tbl, pred_leaf=True).shape == (n, 15)
