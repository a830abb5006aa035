"""Dataset construction / binning / fields (parity target: reference test_basic.py)."""
import numpy as np
import pytest

import lightgbm_amd as lgb


def test_dataset_from_numpy():
    X = np.random.RandomState(0).randn(500, 5)
    y = np.random.RandomState(1).rand(500)
    d = lgb.Dataset(X, label=y).construct()
    assert d.num_data() == 500
    assert d.num_feature() == 5
    np.testing.assert_allclose(d.get_label(), y.astype(np.float32), rtol=1e-6)


def test_dataset_feature_num_bin():
    rng = np.random.RandomState(0)
    X = rng.randn(1000, 3)
    X[:, 1] = rng.randint(0, 5, size=1000)  # few distinct values
    d = lgb.Dataset(X, params={"max_bin": 63}).construct()
    assert d.feature_num_bin(0) <= 63
    assert d.feature_num_bin(1) <= 6


def test_dataset_weights_group():
    X = np.random.RandomState(0).randn(100, 3)
    y = np.random.RandomState(1).rand(100)
    w = np.arange(100, dtype=np.float32) + 1
    d = lgb.Dataset(X, label=y, weight=w).construct()
    np.testing.assert_allclose(d.get_weight(), w, rtol=1e-6)
    g = np.array([30, 30, 40], dtype=np.int32)
    d2 = lgb.Dataset(X, label=y, group=g).construct()
    np.testing.assert_array_equal(d2.get_group(), g)


def test_dataset_nan_handling():
    rng = np.random.RandomState(0)
    X = rng.randn(1000, 2)
    X[::7, 0] = np.nan
    y = (np.nan_to_num(X[:, 0]) > 0).astype(np.float32)
    bst = lgb.train({"objective": "binary", "verbosity": -1}, lgb.Dataset(X, label=y), 10)
    pred = bst.predict(X)
    assert np.all(np.isfinite(pred))


def test_dataset_from_csr():
    import scipy.sparse as sp
    rng = np.random.RandomState(0)
    X = sp.random(300, 10, density=0.3, random_state=0, format="csr")
    y = rng.rand(300)
    d = lgb.Dataset(X, label=y).construct()
    assert d.num_data() == 300
    assert d.num_feature() == 10


def test_dataset_subset():
    X = np.random.RandomState(0).randn(200, 4)
    y = np.random.RandomState(1).rand(200)
    d = lgb.Dataset(X, label=y).construct()
    sub = d.subset(np.arange(50))
    assert sub.num_data() == 50
    np.testing.assert_allclose(sub.get_label(), y[:50].astype(np.float32), rtol=1e-6)


def test_dataset_save_binary(tmp_path):
    X = np.random.RandomState(0).randn(200, 4)
    y = np.random.RandomState(1).rand(200)
    d = lgb.Dataset(X, label=y).construct()
    f = tmp_path / "data.bin"
    d.save_binary(str(f))
    d2 = lgb.Dataset(str(f)).construct()
    assert d2.num_data() == 200
    np.testing.assert_allclose(d2.get_label(), y.astype(np.float32), rtol=1e-6)


def test_feature_names():
    X = np.random.RandomState(0).randn(100, 3)
    d = lgb.Dataset(X, label=np.zeros(100), feature_name=["a", "b", "c"]).construct()
    assert d.get_feature_name() == ["a", "b", "c"]


def test_efb_bundling_sparse_features():
    """EFB: mutually-exclusive sparse features share a column; quality unaffected."""
    import scipy.sparse as sp
    rng = np.random.RandomState(21)
    n = 6000
    # 10 one-hot-ish mutually exclusive sparse columns + 2 dense
    group = rng.randint(0, 10, size=n)
    S = np.zeros((n, 10))
    S[np.arange(n), group] = rng.rand(n) + 0.5
    D = rng.randn(n, 2)
    X = np.column_stack([S, D])
    effect = rng.randn(10) * 2
    y = (effect[group] + 0.5 * D[:, 0] + 0.3 * rng.randn(n) > 0).astype(np.float32)
    for bundle in (True, False):
        ds = lgb.Dataset(X, label=y, params={"enable_bundle": bundle,
                                             "max_conflict_rate": 0.0}).construct()
        bst = lgb.train({"objective": "binary", "verbosity": -1, "min_data_in_leaf": 5,
                         "enable_bundle": bundle}, ds, 30)
        acc = ((bst.predict(X) > 0.5) == y).mean()
        assert acc > 0.9, (bundle, acc)


def test_efb_valid_set_alignment():
    rng = np.random.RandomState(22)
    n = 4000
    group = rng.randint(0, 8, size=n)
    S = np.zeros((n, 8))
    S[np.arange(n), group] = 1.0
    X = np.column_stack([S, rng.randn(n, 2)])
    y = (group % 2).astype(np.float32)
    train = lgb.Dataset(X[:3000], label=y[:3000]).construct()
    valid = train.create_valid(X[3000:], label=y[3000:])
    ev = {}
    lgb.train({"objective": "binary", "verbosity": -1, "min_data_in_leaf": 5},
              train, 20, valid_sets=[valid], callbacks=[lgb.record_evaluation(ev)])
    assert ev["valid_0"]["binary_logloss"][-1] < 0.1


def test_sequence_input():
    class NpSeq(lgb.basic.Sequence):
        def __init__(self, arr):
            self.arr = arr

        def __getitem__(self, i):
            return self.arr[i]

        def __len__(self):
            return len(self.arr)

    rng = np.random.RandomState(41)
    X = rng.randn(3000, 5)
    y = (X[:, 0] > 0).astype(np.float32)
    ds = lgb.Dataset([NpSeq(X[:1500]), NpSeq(X[1500:])], label=y)
    bst = lgb.train({"objective": "binary", "verbosity": -1}, ds, 10)
    assert ((bst.predict(X) > 0.5) == y).mean() > 0.9


def test_trees_to_dataframe():
    rng = np.random.RandomState(42)
    X = rng.randn(500, 3)
    y = (X[:, 0] > 0).astype(np.float32)
    bst = lgb.train({"objective": "binary", "verbosity": -1, "num_leaves": 7},
                    lgb.Dataset(X, label=y), 3)
    df = bst.trees_to_dataframe()
    assert set(df["tree_index"]) == {0, 1, 2}
    assert "threshold" in df.columns


def test_c_api_predict_for_mats_and_fast_path():
    """LGBM_BoosterPredictForMats (row-pointer array) and the FastConfig single-row
    path must match Booster.predict (reference c_api.h parity)."""
    import ctypes
    from lightgbm_amd.basic import _LIB
    rng = np.random.RandomState(5)
    X = rng.rand(400, 6)
    y = (X[:, 0] + X[:, 1] > 1.0).astype(np.float64)
    bst = lgb.train({"objective": "binary", "verbosity": -1}, lgb.Dataset(X, label=y), 8)
    expected = bst.predict(X[:16])
    h = bst._handle

    # --- PredictForMats: array of row pointers
    rows = np.ascontiguousarray(X[:16], dtype=np.float64)
    ptrs = (ctypes.c_void_p * 16)(*[rows[i].ctypes.data for i in range(16)])
    out = np.zeros(16, dtype=np.float64)
    out_len = ctypes.c_int64(0)
    rc = _LIB.LGBM_BoosterPredictForMats(
        h, ptrs, ctypes.c_int(1), ctypes.c_int32(16), ctypes.c_int32(6), ctypes.c_int(0),
        ctypes.c_int(0), ctypes.c_int(-1), b"", ctypes.byref(out_len),
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_double)))
    assert rc == 0 and out_len.value == 16
    np.testing.assert_allclose(out, expected, rtol=1e-12)

    # --- FastConfig single-row path
    fc = ctypes.c_void_p()
    rc = _LIB.LGBM_BoosterPredictForMatSingleRowFastInit(
        h, ctypes.c_int(0), ctypes.c_int(0), ctypes.c_int(-1), ctypes.c_int(1),
        ctypes.c_int32(6), b"", ctypes.byref(fc))
    assert rc == 0
    one = np.zeros(1, dtype=np.float64)
    for i in range(4):
        row = np.ascontiguousarray(X[i], dtype=np.float64)
        rc = _LIB.LGBM_BoosterPredictForMatSingleRowFast(
            fc, row.ctypes.data_as(ctypes.c_void_p), ctypes.byref(out_len),
            one.ctypes.data_as(ctypes.POINTER(ctypes.c_double)))
        assert rc == 0
        assert abs(one[0] - expected[i]) < 1e-12
    assert _LIB.LGBM_FastConfigFree(fc) == 0


def test_add_features_from():
    """Dataset.add_features_from appends another dataset's features in place."""
    rng = np.random.RandomState(11)
    Xa, Xb = rng.rand(1200, 3), rng.rand(1200, 4)
    y = (Xa[:, 0] + Xb[:, 0] > 1.0).astype(np.float64)
    da = lgb.Dataset(Xa, label=y, free_raw_data=False)
    db = lgb.Dataset(Xb, free_raw_data=False)
    da.construct()
    db.construct()
    da.add_features_from(db)
    assert da.num_feature() == 7
    bst = lgb.train({"objective": "binary", "verbosity": -1}, da, 20)
    # features from BOTH sources must be usable: combined model beats Xa-only
    bst_a = lgb.train({"objective": "binary", "verbosity": -1},
                      lgb.Dataset(Xa, label=y), 20)
    X_full = np.hstack([Xa, Xb])
    acc_joint = ((bst.predict(X_full) > 0.5) == y).mean()
    acc_a = ((bst_a.predict(Xa) > 0.5) == y).mean()
    assert acc_joint > acc_a + 0.02
    imp = bst.feature_importance()
    assert imp[:3].sum() > 0 and imp[3:].sum() > 0


def _sparse_train_data(n=6000, d=12, density=0.08, seed=13):
    rng = np.random.RandomState(seed)
    X = np.zeros((n, d))
    mask = rng.rand(n, d) < density
    X[mask] = rng.rand(mask.sum()) * 3 + 0.5
    y = ((X[:, 0] > 0) | (X[:, 1] > 2.0)).astype(np.float64)
    return X, y


def test_sparse_bin_storage_equivalence():
    """Sparse bin columns (is_enable_sparse) must produce the BYTE-IDENTICAL model
    to dense storage — same splits, same outputs — while storing only nonzeros."""
    X, y = _sparse_train_data()
    p = {"objective": "binary", "num_leaves": 31, "verbosity": -1, "min_data_in_leaf": 5}
    m_sparse = lgb.train({**p, "is_enable_sparse": True},
                         lgb.Dataset(X, label=y), 25).model_to_string()
    m_dense = lgb.train({**p, "is_enable_sparse": False},
                        lgb.Dataset(X, label=y), 25).model_to_string()
    assert m_sparse == m_dense


def test_sparse_bin_with_bagging_and_valid():
    """Sparse columns under bagging (subset histograms) and aligned valid sets."""
    X, y = _sparse_train_data(seed=21)
    p = {"objective": "binary", "verbosity": -1, "bagging_freq": 1,
         "bagging_fraction": 0.6, "min_data_in_leaf": 5}
    ev = {}
    train = lgb.Dataset(X[:5000], label=y[:5000])
    valid = train.create_valid(X[5000:], label=y[5000:])
    bst = lgb.train({**p, "metric": "auc"}, train, 30, valid_sets=[valid],
                    callbacks=[lgb.record_evaluation(ev)])
    assert ev["valid_0"]["auc"][-1] > 0.95
    # dense/sparse parity under bagging too (same rng stream)
    m_dense = lgb.train({**p, "metric": "auc", "is_enable_sparse": False},
                        lgb.Dataset(X[:5000], label=y[:5000]), 30).model_to_string()
    m_sparse = lgb.train({**p, "metric": "auc"},
                         lgb.Dataset(X[:5000], label=y[:5000]), 30).model_to_string()
    assert m_sparse == m_dense


def test_sparse_bin_from_scipy_csr():
    """scipy CSR input over mostly-zero data exercises the sparse-storage path
    end to end (construct -> train -> predict)."""
    sp = pytest.importorskip("scipy.sparse")
    X, y = _sparse_train_data(seed=5)
    Xs = sp.csr_matrix(X)
    bst = lgb.train({"objective": "binary", "verbosity": -1, "min_data_in_leaf": 5},
                    lgb.Dataset(Xs, label=y), 25)
    pred_sparse_in = bst.predict(Xs)
    np.testing.assert_allclose(pred_sparse_in, bst.predict(X), rtol=1e-12)
    assert (((pred_sparse_in > 0.5) == y).mean()) > 0.9


def test_4bit_bin_packing_equivalence():
    """max_bin<=16 features use 4-bit packed columns; models must be byte-identical
    to unpacked training (max_bin=15 forces every feature into the 4-bit path)."""
    rng = np.random.RandomState(8)
    X = rng.rand(4000, 6)
    y = (X[:, 0] + X[:, 1] > 1.0).astype(np.float64)
    p = {"objective": "binary", "max_bin": 15, "verbosity": -1}
    m1 = lgb.train(p, lgb.Dataset(X, label=y, params={"max_bin": 15}), 20)
    pred = m1.predict(X)
    acc = ((pred > 0.5) == y).mean()
    assert acc > 0.9
    # valid-set alignment through the 4-bit path
    train = lgb.Dataset(X[:3000], label=y[:3000], params={"max_bin": 15})
    valid = train.create_valid(X[3000:], label=y[3000:])
    ev = {}
    lgb.train({**p, "metric": "auc"}, train, 20, valid_sets=[valid],
              callbacks=[lgb.record_evaluation(ev)])
    assert ev["valid_0"]["auc"][-1] > 0.95


def test_booster_misc_api():
    """attr/set_attr, leaf get/set, shuffle_models, split-value histogram,
    set_train_data_name (reference Booster API surface)."""
    rng = np.random.RandomState(0)
    X = rng.randn(1500, 5)
    y = (X[:, 0] > 0).astype(np.float32)
    b = lgb.train({"objective": "binary", "verbosity": -1}, lgb.Dataset(X, label=y), 8)
    b.set_attr(note="v1")
    assert b.attr("note") == "v1"
    b.set_attr(note=None)
    assert b.attr("note") is None
    v = b.get_leaf_output(0, 0)
    b.set_leaf_output(0, 0, v * 2)
    assert abs(b.get_leaf_output(0, 0) - 2 * v) < 1e-12
    b.set_leaf_output(0, 0, v)
    pred = b.predict(X[:50])
    b.shuffle_models()
    np.testing.assert_allclose(b.predict(X[:50]), pred, rtol=1e-12)
    hist, edges = b.get_split_value_histogram(0)
    assert hist.sum() > 0 and len(edges) == len(hist) + 1
    assert b.set_train_data_name("train") is b


def test_dataset_misc_api():
    """set_field/get_data/get_params/set_reference/set_categorical_feature/
    get_ref_chain (reference Dataset API surface)."""
    rng = np.random.RandomState(0)
    X = rng.rand(500, 4)
    y = (X[:, 0] > 0.5).astype(np.float32)
    ds = lgb.Dataset(X, label=y, free_raw_data=False, params={"max_bin": 31})
    ds.set_categorical_feature([3])
    ds.construct()
    ds.set_field("weight", np.ones(500, dtype=np.float32) * 2)
    np.testing.assert_allclose(ds.get_field("weight"), 2.0)
    assert ds.get_data() is X
    assert ds.get_params()["max_bin"] == 31
    valid = lgb.Dataset(X[:100], label=y[:100]).set_reference(ds)
    valid.construct()
    assert ds in valid.get_ref_chain()
    import pytest as _pt
    with _pt.raises(lgb.LightGBMError):
        ds.set_reference(valid)  # already constructed


def test_pandas_categorical_dtype():
    """pandas category-dtype columns: auto-detected as categorical, codes aligned
    through save/load via the pandas_categorical model line (reference parity)."""
    pd = pytest.importorskip("pandas")
    rng = np.random.RandomState(0)
    n = 4000
    df = pd.DataFrame({"a": rng.randn(n),
                       "b": pd.Categorical(rng.choice(["x", "y", "z"], n)),
                       "c": rng.rand(n)})
    y = ((df["b"] == "x").values & (df["a"] > 0)).astype(np.float32)
    bst = lgb.train({"objective": "binary", "verbosity": -1},
                    lgb.Dataset(df, label=y), 20)
    pred = bst.predict(df)
    assert ((pred > 0.5) == y).mean() > 0.98
    m = bst.model_to_string()
    assert "pandas_categorical:" in m
    b2 = lgb.Booster(model_str=m)
    # a frame with re-ordered categories must map through the stored category list
    df2 = df.copy()
    df2["b"] = pd.Categorical(df["b"], categories=["z", "x", "y"])
    np.testing.assert_allclose(b2.predict(df2), pred, rtol=1e-12)
    # unseen category becomes missing, not a crash
    df3 = df.head(10).copy()
    df3["b"] = pd.Categorical(["w"] * 10, categories=["w"])
    assert np.isfinite(b2.predict(df3)).all()
    # a VALID set with re-ordered categories aligns its codes to the train mapping
    tr = lgb.Dataset(df[:3000], label=y[:3000])
    dv = df[3000:].copy()
    dv["b"] = pd.Categorical(dv["b"], categories=["z", "y", "x"])
    ev = {}
    lgb.train({"objective": "binary", "metric": "auc", "verbosity": -1}, tr, 10,
              valid_sets=[tr.create_valid(dv, label=y[3000:])],
              callbacks=[lgb.record_evaluation(ev)])
    assert ev["valid_0"]["auc"][-1] > 0.98


def test_validation_errors():
    """Length-mismatched labels and too-few predict columns must raise, not
    silently mis-train/mis-predict."""
    rng = np.random.RandomState(0)
    X = rng.rand(100, 4)
    with pytest.raises(lgb.LightGBMError):
        lgb.Dataset(X, label=np.zeros(50, dtype=np.float32)).construct()
    y = (X[:, 0] > 0.5).astype(np.float32)
    bst = lgb.train({"objective": "binary", "verbosity": -1, "min_data_in_leaf": 5},
                    lgb.Dataset(X, label=y), 3)
    with pytest.raises(lgb.LightGBMError):
        bst.predict(rng.rand(5, 2))
    # more columns than training is allowed (extras ignored), like the reference
    assert len(bst.predict(np.column_stack([X[:5], np.zeros(5)]))) == 5


def test_kitchen_sink_mixed_features():
    """Integration: dense + sparse + 4-bit + categorical + EFB-bundleable features
    in ONE dataset, with bagging and a valid set — exercises feature-storage
    interactions that unit tests miss."""
    rng = np.random.RandomState(42)
    n = 6000
    dense = rng.randn(n, 3)                                   # plain dense
    sparse = np.zeros((n, 4))                                 # ~5% nonzero
    mask = rng.rand(n, 4) < 0.05
    sparse[mask] = rng.rand(mask.sum()) + 1
    lowcard = rng.randint(0, 9, (n, 2)).astype(float)         # 4-bit packable
    cat = rng.randint(0, 30, n).astype(float)                 # categorical
    onehot = np.zeros((n, 6))                                 # EFB-bundleable
    onehot[np.arange(n), rng.randint(0, 6, n)] = 1.0
    X = np.column_stack([dense, sparse, lowcard, cat, onehot])
    eff = rng.randn(30)
    y = ((dense[:, 0] + 2 * (sparse[:, 0] > 0) + eff[cat.astype(int)] * 0.5 +
          onehot[:, 0]) > 0.5).astype(np.float32)
    p = {"objective": "binary", "verbosity": -1, "bagging_freq": 1,
         "bagging_fraction": 0.7, "categorical_feature": [9], "min_data_in_leaf": 5}
    tr = lgb.Dataset(X[:5000], label=y[:5000])
    ev = {}
    bst = lgb.train({**p, "metric": "auc"}, tr, 40,
                    valid_sets=[tr.create_valid(X[5000:], label=y[5000:])],
                    callbacks=[lgb.record_evaluation(ev)])
    assert ev["valid_0"]["auc"][-1] > 0.9
    # model roundtrips and predicts identically
    m = bst.model_to_string()
    np.testing.assert_allclose(lgb.Booster(model_str=m).predict(X[:200]),
                               bst.predict(X[:200]), rtol=1e-12)
    # SHAP consistency on the mixed feature set
    contrib = bst.predict(X[:50], pred_contrib=True)
    np.testing.assert_allclose(contrib.sum(axis=1),
                               bst.predict(X[:50], raw_score=True), rtol=1e-6, atol=1e-6)


def test_subset_group():
    """Dataset.subset carries query boundaries, init_score and positions
    (ref test_basic.py test_subset_group)."""
    rng = np.random.RandomState(40)
    X = rng.randn(200, 4)
    y = rng.randint(0, 3, 200).astype(float)
    ds = lgb.Dataset(X, label=y, group=[50, 50, 50, 50],
                     init_score=np.arange(200, dtype=float))
    ds.construct()
    sub = ds.subset(list(range(100)))
    sub.construct()
    np.testing.assert_array_equal(sub.get_field("group"), [0, 50, 100])
    np.testing.assert_allclose(sub.get_field("init_score"), np.arange(100.0))
    # the subset trains as a 2-query ranking shard
    bst = lgb.train({"objective": "lambdarank", "verbosity": -1}, sub, 3)
    assert bst.num_trees() == 3


def test_set_field_none_removes_field():
    """set_field(name, None) clears weight/group/init_score/position
    (ref test_set_field_none_removes_field)."""
    rng = np.random.RandomState(41)
    X = rng.randn(100, 3)
    ds = lgb.Dataset(X, label=rng.rand(100))
    ds.construct()
    for field, val in (("weight", np.ones(100)),
                       ("init_score", np.zeros(100)),
                       ("group", np.array([50, 50], dtype=np.int32))):
        ds.set_field(field, val)
        assert ds.get_field(field) is not None, field
        ds.set_field(field, None)
        assert ds.get_field(field) is None, field


def test_init_score_multiclass_2d():
    """2D (row, class) init_score seeds multiclass training
    (ref test_init_score_for_multiclass_classification)."""
    rng = np.random.RandomState(42)
    X = rng.randn(1500, 4)
    y = rng.randint(0, 3, 1500).astype(float)
    init = np.zeros((1500, 3))
    init[:, 1] = 5.0  # huge prior for class 1
    # 2D roundtrip: stored class-major, surfaced back as (row, class)
    small = np.array([[i * 10 + j for j in range(3)] for i in range(10)], dtype=float)
    ds_small = lgb.Dataset(rng.rand(10, 2), label=rng.randint(0, 3, 10).astype(float),
                           init_score=small).construct()
    np.testing.assert_array_equal(ds_small.get_field("init_score"), small)
    # training effect: the softmax prior on class 1 makes its first-tree
    # gradients positive, so its raw scores are pushed DOWN, the others UP
    ds = lgb.Dataset(X, label=y, init_score=init)
    bst = lgb.train({"objective": "multiclass", "num_class": 3, "verbosity": -1,
                     "learning_rate": 0.1, "boost_from_average": False}, ds, 1)
    raw = bst.predict(X[:500], raw_score=True).reshape(500, 3)
    assert raw[:, 1].mean() < 0 < raw[:, 0].mean()
    assert raw[:, 1].mean() < 0 < raw[:, 2].mean()


def test_save_dataset_subset_and_load_from_file(tmp_path):
    """a subset Dataset saves to the binary format and loads back with its rows
    (ref test_save_dataset_subset_and_load_from_file)."""
    rng = np.random.RandomState(60)
    X = rng.randn(120, 4)
    y = rng.rand(120)
    ds = lgb.Dataset(X, label=y, free_raw_data=False)
    ds.construct()
    sub = ds.subset(np.arange(30, 100))
    sub.construct()
    f = tmp_path / "sub.bin"
    sub.save_binary(str(f))
    d2 = lgb.Dataset(str(f))
    d2.construct()
    assert d2.num_data() == 70
    np.testing.assert_allclose(d2.get_label(), y[30:100], rtol=1e-6)
    bst = lgb.train({"objective": "regression", "verbosity": -1}, d2, 3)
    assert bst.num_trees() == 3


def test_consistent_state_for_dataset_fields():
    """float64 label/weight are stored as float32 consistently
    (ref test_consistent_state_for_dataset_fields)."""
    rng = np.random.RandomState(61)
    X = rng.randn(100, 3)
    y64 = rng.rand(100)
    w64 = 1 + rng.rand(100)
    ds = lgb.Dataset(X, label=y64, weight=w64)
    ds.construct()
    np.testing.assert_allclose(ds.get_label(), y64.astype(np.float32), rtol=1e-7)
    np.testing.assert_allclose(ds.get_weight(), w64.astype(np.float32), rtol=1e-7)
    assert ds.get_label().dtype == np.float32


def test_feature_names_default_and_custom():
    """Column_N default names; custom names flow to the model
    (ref test_feature_names_are_set_correctly...)."""
    rng = np.random.RandomState(62)
    X = rng.randn(200, 3)
    ds = lgb.Dataset(X, label=X[:, 0])
    bst = lgb.train({"objective": "regression", "verbosity": -1}, ds, 2)
    assert bst.feature_name() == ["Column_0", "Column_1", "Column_2"]
    ds2 = lgb.Dataset(X, label=X[:, 0], feature_name=["x", "y", "z"])
    bst2 = lgb.train({"objective": "regression", "verbosity": -1}, ds2, 2)
    assert bst2.feature_name() == ["x", "y", "z"]


def test_chunked_sequence_batches_match_numpy():
    """Sequence ingestion with a small batch_size builds the same dataset as a
    direct numpy matrix (ref test_chunked_dataset)."""
    class SmallBatchSeq(lgb.basic.Sequence):
        batch_size = 37  # deliberately awkward chunking

        def __init__(self, arr):
            self.arr = arr

        def __getitem__(self, i):
            return self.arr[i]

        def __len__(self):
            return len(self.arr)

    rng = np.random.RandomState(63)
    X = rng.randn(1000, 4)
    y = (X[:, 0] > 0).astype(np.float32)
    p = {"objective": "binary", "verbosity": -1, "seed": 3}
    m_np = lgb.train(p, lgb.Dataset(X, label=y), 8).model_to_string()
    m_seq = lgb.train(p, lgb.Dataset([SmallBatchSeq(X[:400]), SmallBatchSeq(X[400:])],
                                     label=y), 8).model_to_string()
    assert m_np == m_seq
