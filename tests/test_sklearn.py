"""sklearn wrapper tests (parity target: reference test_sklearn.py, core subset)."""
import numpy as np
import pytest

import lightgbm_amd as lgb


def test_regressor():
    rng = np.random.RandomState(0)
    X = rng.randn(3000, 8)
    y = 2 * X[:, 0] + np.sin(X[:, 1]) + 0.1 * rng.randn(3000)
    m = lgb.LGBMRegressor(n_estimators=50, random_state=0)
    m.fit(X, y)
    assert m.score(X, y) > 0.9
    assert m.n_features_in_ == 8
    assert len(m.feature_importances_) == 8


def test_classifier_binary():
    rng = np.random.RandomState(1)
    X = rng.randn(3000, 6)
    y = np.where(X[:, 0] + 0.5 * rng.randn(3000) > 0, "pos", "neg")
    m = lgb.LGBMClassifier(n_estimators=30)
    m.fit(X, y)
    assert set(m.classes_) == {"neg", "pos"}
    proba = m.predict_proba(X[:10])
    assert proba.shape == (10, 2)
    np.testing.assert_allclose(proba.sum(axis=1), 1.0, rtol=1e-9)
    assert m.score(X, y) > 0.9


def test_classifier_multiclass():
    rng = np.random.RandomState(2)
    X = rng.randn(3000, 6)
    y = (X[:, 0] > 0.5).astype(int) + (X[:, 1] > 0).astype(int)
    m = lgb.LGBMClassifier(n_estimators=30)
    m.fit(X, y)
    assert m.n_classes_ == 3
    assert m.predict_proba(X[:5]).shape == (5, 3)
    assert m.score(X, y) > 0.8


def test_classifier_class_weight():
    rng = np.random.RandomState(3)
    X = rng.randn(3000, 4)
    y = (X[:, 0] + 0.8 * rng.randn(3000) > 1.0).astype(int)  # imbalanced
    m = lgb.LGBMClassifier(n_estimators=20, class_weight="balanced")
    m.fit(X, y)
    # balanced weighting should raise recall on the minority class
    pred = m.predict(X)
    recall = (pred[y == 1] == 1).mean()
    assert recall > 0.5


def test_ranker():
    rng = np.random.RandomState(4)
    groups = [20] * 50
    X = rng.randn(sum(groups), 5)
    y = (X[:, 0] + 0.3 * rng.randn(len(X)) > 0.5).astype(int)
    m = lgb.LGBMRanker(n_estimators=20)
    m.fit(X, y, group=groups)
    pred = m.predict(X[:20])
    assert pred.shape == (20,)


def test_early_stopping_fit():
    rng = np.random.RandomState(5)
    X = rng.randn(4000, 6)
    y = X[:, 0] + 0.2 * rng.randn(4000)
    m = lgb.LGBMRegressor(n_estimators=500)
    m.fit(X[:3000], y[:3000], eval_set=[(X[3000:], y[3000:])],
          eval_metric="l2", early_stopping_rounds=5)
    assert m.best_iteration_ > 0
    assert m.best_iteration_ < 500
    assert "valid_0" in m.evals_result_


def test_get_set_params():
    m = lgb.LGBMRegressor(num_leaves=15, custom_thing=7)
    p = m.get_params()
    assert p["num_leaves"] == 15
    assert p["custom_thing"] == 7
    m.set_params(num_leaves=31)
    assert m.get_params()["num_leaves"] == 31


def test_pickle_roundtrip():
    """Booster and sklearn estimators pickle via model-text reconstruction."""
    import pickle
    rng = np.random.RandomState(0)
    X = rng.randn(2000, 6)
    y = (X[:, 0] > 0).astype(int)
    m = lgb.LGBMClassifier(n_estimators=10, verbosity=-1).fit(X, y)
    m2 = pickle.loads(pickle.dumps(m))
    np.testing.assert_array_equal(m2.predict(X[:100]), m.predict(X[:100]))
    np.testing.assert_allclose(m2.predict_proba(X[:100]), m.predict_proba(X[:100]),
                               rtol=1e-12)
    b2 = pickle.loads(pickle.dumps(m.booster_))
    np.testing.assert_allclose(b2.predict(X[:50]), m.booster_.predict(X[:50]), rtol=1e-12)


def test_callable_objective_and_metric():
    """sklearn-style callable objective (y_true, y_pred) -> (grad, hess) and
    callable eval_metric (reference sklearn.py adapters)."""
    rng = np.random.RandomState(0)
    X = rng.randn(2000, 5)
    y = (X[:, 0] > 0).astype(np.float32)

    def logistic_obj(y_true, y_pred):
        p = 1 / (1 + np.exp(-y_pred))
        return (p - y_true).astype(np.float32), (p * (1 - p)).astype(np.float32)

    def acc_metric(y_true, y_pred):
        return "my_acc", float(np.mean((y_pred > 0.5) == y_true)), True

    m = lgb.LGBMClassifier(n_estimators=15, objective=logistic_obj, verbosity=-1)
    m.fit(X, y)
    raw = m.predict(X, raw_score=True).ravel()
    assert (((1 / (1 + np.exp(-raw))) > 0.5) == y).mean() > 0.9
    m2 = lgb.LGBMClassifier(n_estimators=5, verbosity=-1)
    m2.fit(X, y, eval_set=[(X, y)], eval_metric=acc_metric)
    assert "my_acc" in m2.evals_result_["valid_0"]
    assert m2.evals_result_["valid_0"]["my_acc"][-1] > 0.9


def test_sklearn_pandas_categorical():
    """sklearn estimators accept DataFrames with category dtype end to end."""
    pd = pytest.importorskip("pandas")
    rng = np.random.RandomState(0)
    n = 2000
    df = pd.DataFrame({"a": rng.randn(n),
                       "b": pd.Categorical(rng.choice(["x", "y", "z"], n))})
    y = ((df["b"] == "x").values & (df["a"] > 0)).astype(int)
    m = lgb.LGBMClassifier(n_estimators=10, verbosity=-1)
    m.fit(df, y, eval_set=[(df, y)])
    assert (m.predict(df) == y).mean() > 0.98
    assert m.predict_proba(df).shape == (n, 2)


def test_fit_init_model_continuation():
    rng = np.random.RandomState(0)
    X = rng.randn(1000, 4)
    y = (X[:, 0] > 0).astype(int)
    m1 = lgb.LGBMClassifier(n_estimators=5, verbosity=-1).fit(X, y)
    m2 = lgb.LGBMClassifier(n_estimators=5, verbosity=-1).fit(X, y, init_model=m1)
    assert m2.booster_.num_trees() == 10


def test_sklearn_ecosystem_protocols():
    """clone / cross_val_score / GridSearchCV interop."""
    from sklearn.base import clone
    from sklearn.model_selection import GridSearchCV, cross_val_score
    rng = np.random.RandomState(0)
    X = rng.randn(800, 4)
    y = (X[:, 0] > 0).astype(int)
    m = clone(lgb.LGBMClassifier(n_estimators=5, verbosity=-1))
    m.fit(X, y)
    scores = cross_val_score(lgb.LGBMClassifier(n_estimators=5, verbosity=-1), X, y, cv=3)
    assert scores.mean() > 0.9
    gs = GridSearchCV(lgb.LGBMClassifier(verbosity=-1),
                      {"n_estimators": [3, 5], "num_leaves": [7, 15]}, cv=2)
    gs.fit(X, y)
    assert gs.best_score_ > 0.9


def test_feature_importances_type():
    """split vs gain importances differ and respect importance_type (ref test_sklearn.py:778)."""
    rng = np.random.RandomState(7)
    X = rng.randn(1500, 5)
    y = X[:, 0] * 3 + X[:, 1] + 0.1 * rng.randn(1500)
    m = lgb.LGBMRegressor(n_estimators=20, importance_type="split", verbosity=-1).fit(X, y)
    imp_split = m.feature_importances_
    m.set_params(importance_type="gain")
    imp_gain = m.feature_importances_
    assert imp_split.dtype.kind in "iu" or np.allclose(imp_split, imp_split.astype(int))
    assert not np.allclose(imp_split / max(imp_split.sum(), 1),
                           imp_gain / max(imp_gain.sum(), 1e-12))
    assert np.argmax(imp_gain) == 0  # dominant feature carries the gain


def test_objective_resolved_after_fit():
    """objective_ reflects the concrete objective chosen at fit time (ref test_sklearn.py:483)."""
    rng = np.random.RandomState(8)
    X = rng.randn(500, 3)
    r = lgb.LGBMRegressor(n_estimators=3, verbosity=-1).fit(X, X[:, 0])
    assert r.objective_ == "regression"
    c = lgb.LGBMClassifier(n_estimators=3, verbosity=-1).fit(X, (X[:, 0] > 0).astype(int))
    assert c.objective_ == "binary"
    y3 = (X[:, 0] > 0.5).astype(int) + (X[:, 1] > 0).astype(int)
    c3 = lgb.LGBMClassifier(n_estimators=3, verbosity=-1).fit(X, y3)
    assert c3.objective_ == "multiclass"


def test_classifier_fit_detects_classes_every_time():
    """refitting with a different class count must not leak num_class (ref test_sklearn.py:2070)."""
    rng = np.random.RandomState(9)
    X = rng.randn(600, 4)
    y3 = rng.randint(0, 3, 600)
    y2 = rng.randint(0, 2, 600)
    m = lgb.LGBMClassifier(n_estimators=3, verbosity=-1)
    m.fit(X, y3)
    assert m.n_classes_ == 3 and m.predict_proba(X[:5]).shape == (5, 3)
    m.fit(X, y2)
    assert m.n_classes_ == 2 and m.predict_proba(X[:5]).shape == (5, 2)
    m.fit(X, y3)
    assert m.n_classes_ == 3 and m.predict_proba(X[:5]).shape == (5, 3)


def test_actual_number_of_trees():
    """n_estimators trees are built (x num_class for multiclass) (ref test_sklearn.py:1517)."""
    rng = np.random.RandomState(10)
    X = rng.randn(400, 3)
    r = lgb.LGBMRegressor(n_estimators=7, verbosity=-1).fit(X, X[:, 0])
    assert r.booster_.num_trees() == 7
    y3 = rng.randint(0, 3, 400)
    c = lgb.LGBMClassifier(n_estimators=4, verbosity=-1).fit(X, y3)
    assert c.booster_.num_trees() == 12


def test_check_is_fitted():
    """accessing fitted-only attributes before fit raises (ref test_sklearn.py:1528)."""
    m = lgb.LGBMClassifier()
    for attr in ("booster_", "feature_importances_", "best_score_"):
        with pytest.raises(Exception):
            getattr(m, attr)
    assert not m.__sklearn_is_fitted__()
    rng = np.random.RandomState(0)
    X = rng.randn(200, 3)
    m.fit(X, (X[:, 0] > 0).astype(int))
    assert m.__sklearn_is_fitted__()


def test_multiple_eval_metrics():
    """eval_metric accepts a list of metric names (ref test_sklearn.py:1306)."""
    rng = np.random.RandomState(11)
    X = rng.randn(1000, 4)
    y = X[:, 0] + 0.1 * rng.randn(1000)
    m = lgb.LGBMRegressor(n_estimators=5, verbosity=-1)
    m.fit(X, y, eval_set=[(X, y)], eval_metric=["l1", "l2"])
    res = m.evals_result_["valid_0"]
    assert "l1" in res and "l2" in res
    assert len(res["l1"]) == 5


def test_eval_names_and_multiple_eval_sets():
    rng = np.random.RandomState(12)
    X = rng.randn(1500, 4)
    y = X[:, 0] + 0.1 * rng.randn(1500)
    m = lgb.LGBMRegressor(n_estimators=5, verbosity=-1)
    m.fit(X[:1000], y[:1000], eval_set=[(X[1000:1250], y[1000:1250]),
                                        (X[1250:], y[1250:])],
          eval_names=["va", "vb"], eval_metric="l2")
    assert set(m.evals_result_.keys()) == {"va", "vb"}


def test_sample_weight_effect():
    """upweighted rows dominate the fit (ref test_sklearn.py sample_weight paths)."""
    rng = np.random.RandomState(13)
    X = rng.randn(2000, 3)
    # two conflicting labelings; weights pick which one wins
    y = np.where(np.arange(2000) < 1000, (X[:, 0] > 0), (X[:, 0] <= 0)).astype(int)
    w = np.where(np.arange(2000) < 1000, 100.0, 1.0)
    m = lgb.LGBMClassifier(n_estimators=20, verbosity=-1).fit(X, y, sample_weight=w)
    first_half_acc = (m.predict(X[:1000]) == y[:1000]).mean()
    assert first_half_acc > 0.9


def test_predict_leaf_and_contrib_shapes():
    """pred_leaf / pred_contrib through the sklearn surface (ref test_sklearn.py:885)."""
    rng = np.random.RandomState(14)
    X = rng.randn(800, 4)
    y = (X[:, 0] > 0).astype(int)
    m = lgb.LGBMClassifier(n_estimators=6, verbosity=-1).fit(X, y)
    leaves = m.predict(X[:30], pred_leaf=True)
    assert leaves.shape == (30, 6)
    contrib = m.predict(X[:30], pred_contrib=True)
    assert contrib.shape == (30, 5)  # n_features + bias
    raw = m.predict(X[:30], raw_score=True)
    np.testing.assert_allclose(contrib.sum(axis=1), np.asarray(raw).ravel(), rtol=1e-6)


def test_dart_boosting_type():
    rng = np.random.RandomState(15)
    X = rng.randn(1200, 4)
    y = X[:, 0] + 0.1 * rng.randn(1200)
    m = lgb.LGBMRegressor(boosting_type="dart", n_estimators=30, verbosity=-1).fit(X, y)
    assert m.score(X, y) > 0.7
    rf = lgb.LGBMRegressor(boosting_type="rf", n_estimators=20, subsample=0.7,
                           subsample_freq=1, colsample_bytree=0.8, verbosity=-1).fit(X, y)
    assert rf.score(X, y) > 0.5


def test_random_state_object_and_reproducibility():
    """np.random.RandomState accepted as random_state; same seed = same model (ref :739)."""
    rng = np.random.RandomState(16)
    X = rng.randn(1000, 4)
    y = X[:, 0] + 0.3 * rng.randn(1000)
    kw = dict(n_estimators=10, subsample=0.6, subsample_freq=1, verbosity=-1)
    p1 = lgb.LGBMRegressor(random_state=np.random.RandomState(42), **kw).fit(X, y).predict(X)
    p2 = lgb.LGBMRegressor(random_state=np.random.RandomState(42), **kw).fit(X, y).predict(X)
    p3 = lgb.LGBMRegressor(random_state=np.random.RandomState(7), **kw).fit(X, y).predict(X)
    np.testing.assert_array_equal(p1, p2)
    assert not np.array_equal(p1, p3)


def test_ranker_eval_set():
    """eval_set + eval_group records ndcg@k (ref test_sklearn.py:205)."""
    rng = np.random.RandomState(17)
    groups = [25] * 40
    X = rng.randn(sum(groups), 5)
    y = np.clip((X[:, 0] * 2 + rng.randn(len(X)) * 0.5).astype(int), 0, 3)
    m = lgb.LGBMRanker(n_estimators=15, verbosity=-1)
    m.fit(X, y, group=groups, eval_set=[(X, y)], eval_group=[groups], eval_at=[1, 3])
    res = m.evals_result_["valid_0"]
    assert any("ndcg@1" in k for k in res)
    assert any("ndcg@3" in k for k in res)
    last = [v[-1] for v in res.values()]
    assert all(0.0 <= v <= 1.0 for v in last)


def test_first_metric_only_sklearn():
    """first_metric_only early stopping via sklearn fit (ref test_sklearn.py:1358)."""
    rng = np.random.RandomState(18)
    X = rng.randn(3000, 5)
    y = X[:, 0] + 0.3 * rng.randn(3000)
    m = lgb.LGBMRegressor(n_estimators=200, first_metric_only=True, verbosity=-1)
    m.fit(X[:2000], y[:2000], eval_set=[(X[2000:], y[2000:])],
          eval_metric=["l2", "l1"], early_stopping_rounds=5)
    assert 0 < m.best_iteration_ <= 200


def test_nan_handle():
    """NaN features route rows through default bins without error (ref test_sklearn.py:1343)."""
    rng = np.random.RandomState(19)
    X = rng.randn(1000, 4)
    X[rng.rand(1000, 4) < 0.2] = np.nan
    y = np.where(np.isnan(X[:, 0]), 0.0, np.nan_to_num(X[:, 0]))
    m = lgb.LGBMRegressor(n_estimators=20, verbosity=-1).fit(X, y)
    pred = m.predict(X)
    assert np.isfinite(pred).all()
    assert m.score(X, y) > 0.7


def test_multioutput_and_chain_meta_estimators():
    """sklearn meta-estimators compose over our estimators (ref test_sklearn.py:418-483)."""
    sklearn_multi = pytest.importorskip("sklearn.multioutput")
    rng = np.random.RandomState(20)
    X = rng.randn(600, 4)
    Y = np.column_stack([(X[:, 0] > 0).astype(int), (X[:, 1] > 0).astype(int)])
    mo = sklearn_multi.MultiOutputClassifier(lgb.LGBMClassifier(n_estimators=5, verbosity=-1))
    mo.fit(X, Y)
    assert mo.predict(X[:10]).shape == (10, 2)
    chain = sklearn_multi.RegressorChain(lgb.LGBMRegressor(n_estimators=5, verbosity=-1))
    chain.fit(X, np.column_stack([X[:, 0], X[:, 1]]))
    assert chain.predict(X[:10]).shape == (10, 2)


def test_stacking_meta_estimators():
    """StackingClassifier/Regressor over our estimators (ref test_sklearn.py:323-361)."""
    from sklearn.ensemble import StackingClassifier, StackingRegressor
    from sklearn.linear_model import LogisticRegression, Ridge
    rng = np.random.RandomState(21)
    X = rng.randn(600, 4)
    y = (X[:, 0] > 0).astype(int)
    sc = StackingClassifier(estimators=[("l1", lgb.LGBMClassifier(n_estimators=5, verbosity=-1)),
                                        ("l2", lgb.LGBMClassifier(n_estimators=3, num_leaves=7, verbosity=-1))],
                            final_estimator=LogisticRegression(), cv=2)
    sc.fit(X, y)
    assert sc.score(X, y) > 0.8
    sr = StackingRegressor(estimators=[("l1", lgb.LGBMRegressor(n_estimators=5, verbosity=-1))],
                           final_estimator=Ridge(), cv=2)
    sr.fit(X, X[:, 0])
    assert sr.score(X, X[:, 0]) > 0.5


def test_post_fit_attributes():
    """feature_names_in_ / n_estimators_ / n_iter_ (ref sklearn post-fit attrs)."""
    rng = np.random.RandomState(50)
    X = rng.randn(1500, 4)
    y = X[:, 0] + 0.2 * rng.randn(1500)
    m = lgb.LGBMRegressor(n_estimators=30, verbosity=-1).fit(X, y)
    assert list(m.feature_names_in_) == [f"Column_{i}" for i in range(4)]
    assert m.n_estimators_ == 30
    assert m.n_iter_ == 30
    es = lgb.LGBMRegressor(n_estimators=500, verbosity=-1)
    es.fit(X[:1000], y[:1000], eval_set=[(X[1000:], y[1000:])],
           eval_metric="l2", early_stopping_rounds=3)
    assert es.n_estimators_ == es.best_iteration_ < 500


def test_callbacks_and_estimators_picklable_with_joblib():
    """callbacks are class instances and pickle; estimators carrying callbacks
    survive joblib dump/load (ref test_joblib / test_non_serializable_objects)."""
    import pickle
    cbs = [lgb.early_stopping(5), lgb.log_evaluation(1),
           lgb.record_evaluation({}), lgb.reset_parameter(learning_rate=[0.1] * 5)]
    for c in cbs:
        assert pickle.loads(pickle.dumps(c)) is not None
    import joblib
    rng = np.random.RandomState(51)
    X = rng.randn(800, 3)
    y = X[:, 0] + 0.1 * rng.randn(800)
    m = lgb.LGBMRegressor(n_estimators=20, verbosity=-1)
    m.fit(X[:600], y[:600], eval_set=[(X[600:], y[600:])],
          early_stopping_rounds=5, callbacks=[lgb.log_evaluation(0)])
    import io
    buf = io.BytesIO()
    joblib.dump(m, buf)
    buf.seek(0)
    m2 = joblib.load(buf)
    np.testing.assert_allclose(m2.predict(X[:50]), m.predict(X[:50]), rtol=1e-12)
