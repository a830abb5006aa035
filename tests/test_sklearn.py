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
