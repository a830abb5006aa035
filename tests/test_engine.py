"""Training engine tests (parity target: reference test_engine.py, core subset)."""
import numpy as np
import pytest

import lightgbm_amd as lgb


def _binary_data(n=5000, d=10, seed=42):
    rng = np.random.RandomState(seed)
    X = rng.randn(n, d)
    logit = 2 * X[:, 0] - 1.5 * X[:, 1] + X[:, 2] * X[:, 3] + 0.5 * rng.randn(n)
    y = (logit > 0).astype(np.float32)
    return X, y


def _regression_data(n=5000, d=10, seed=7):
    rng = np.random.RandomState(seed)
    X = rng.randn(n, d)
    y = 3 * X[:, 0] + np.sin(X[:, 1]) + 0.1 * rng.randn(n)
    return X, y.astype(np.float32)


def test_binary_auc():
    X, y = _binary_data()
    ev = {}
    bst = lgb.train({"objective": "binary", "metric": "auc", "verbosity": -1},
                    lgb.Dataset(X[:4000], label=y[:4000]), 50,
                    valid_sets=[lgb.Dataset(X[:4000], label=y[:4000]).create_valid(
                        X[4000:], label=y[4000:])],
                    callbacks=[lgb.record_evaluation(ev)])
    assert ev["valid_0"]["auc"][-1] > 0.93
    assert ev["valid_0"]["auc"][-1] > ev["valid_0"]["auc"][0]


def test_regression_l2():
    X, y = _regression_data()
    bst = lgb.train({"objective": "regression", "verbosity": -1},
                    lgb.Dataset(X, label=y), 60)
    pred = bst.predict(X)
    mse = float(np.mean((pred - y) ** 2))
    assert mse < 0.1 * float(np.var(y))


@pytest.mark.parametrize("objective", ["regression_l1", "huber", "fair", "quantile", "mape"])
def test_regression_objectives_run(objective):
    X, y = _regression_data(n=2000)
    y = np.abs(y) + 0.1
    bst = lgb.train({"objective": objective, "verbosity": -1}, lgb.Dataset(X, label=y), 15)
    assert np.all(np.isfinite(bst.predict(X[:50])))


@pytest.mark.parametrize("objective", ["poisson", "gamma", "tweedie"])
def test_positive_objectives_run(objective):
    rng = np.random.RandomState(0)
    X = rng.randn(2000, 5)
    y = np.exp(0.5 * X[:, 0]) + 0.1
    bst = lgb.train({"objective": objective, "verbosity": -1}, lgb.Dataset(X, label=y), 15)
    pred = bst.predict(X[:50])
    assert np.all(pred > 0)


def test_multiclass():
    rng = np.random.RandomState(0)
    X = rng.randn(3000, 6)
    y = (X[:, 0] + 0.3 * rng.randn(3000) > 0.5).astype(int) + \
        (X[:, 1] + 0.3 * rng.randn(3000) > 0).astype(int)
    bst = lgb.train({"objective": "multiclass", "num_class": 3, "verbosity": -1},
                    lgb.Dataset(X, label=y.astype(np.float32)), 30)
    pred = bst.predict(X)
    assert pred.shape == (3000, 3)
    np.testing.assert_allclose(pred.sum(axis=1), 1.0, rtol=1e-6)
    acc = (pred.argmax(axis=1) == y).mean()
    assert acc > 0.7


def test_lambdarank():
    rng = np.random.RandomState(0)
    n_queries = 100
    rows = []
    labels = []
    groups = []
    for q in range(n_queries):
        nq = rng.randint(5, 30)
        Xq = rng.randn(nq, 5)
        rel = (Xq[:, 0] + 0.5 * rng.randn(nq) > 0.5).astype(int) * 2
        rows.append(Xq)
        labels.append(rel)
        groups.append(nq)
    X = np.vstack(rows)
    y = np.concatenate(labels).astype(np.float32)
    ev = {}
    train = lgb.Dataset(X, label=y, group=np.array(groups, dtype=np.int32))
    bst = lgb.train({"objective": "lambdarank", "metric": "ndcg", "eval_at": [5],
                     "verbosity": -1}, train, 30,
                    valid_sets=[train], valid_names=["train"],
                    callbacks=[lgb.record_evaluation(ev)])
    assert ev["train"]["ndcg@5"][-1] > 0.80


def test_early_stopping():
    X, y = _binary_data()
    train = lgb.Dataset(X[:4000], label=y[:4000])
    valid = train.create_valid(X[4000:], label=y[4000:])
    bst = lgb.train({"objective": "binary", "metric": "binary_logloss", "verbosity": -1},
                    train, 500, valid_sets=[valid],
                    callbacks=[lgb.early_stopping(5, verbose=False)])
    assert bst.best_iteration > 0
    assert bst.best_iteration < 500


def test_model_save_load_roundtrip(tmp_path):
    X, y = _binary_data(n=2000)
    bst = lgb.train({"objective": "binary", "verbosity": -1}, lgb.Dataset(X, label=y), 20)
    pred1 = bst.predict(X[:100])
    f = tmp_path / "model.txt"
    bst.save_model(str(f))
    content = f.read_text()
    # model text v4 markers (format parity with the reference)
    assert content.startswith("tree\nversion=v4\n")
    assert "end of trees" in content
    assert "feature_importances:" in content
    assert "parameters:" in content
    bst2 = lgb.Booster(model_file=str(f))
    pred2 = bst2.predict(X[:100])
    np.testing.assert_allclose(pred1, pred2, rtol=1e-12)


def test_continue_training():
    X, y = _binary_data(n=2000)
    train = lgb.Dataset(X, label=y)
    bst1 = lgb.train({"objective": "binary", "verbosity": -1}, train, 10)
    model = bst1.model_to_string()
    train2 = lgb.Dataset(X, label=y)
    bst2 = lgb.train({"objective": "binary", "verbosity": -1}, train2, 10,
                     init_model=bst1)
    assert bst2.num_trees() == 20


def test_custom_objective():
    X, y = _binary_data(n=2000)

    def logloss_obj(preds, train_data):
        labels = train_data.get_label()
        p = 1.0 / (1.0 + np.exp(-preds))
        return (p - labels).astype(np.float32), (p * (1 - p)).astype(np.float32)

    bst = lgb.train({"objective": "none", "verbosity": -1}, lgb.Dataset(X, label=y), 30,
                    fobj=logloss_obj)
    pred_raw = bst.predict(X, raw_score=True)
    p = 1 / (1 + np.exp(-pred_raw))
    acc = ((p > 0.5) == y).mean()
    assert acc > 0.85


def test_custom_metric():
    X, y = _binary_data(n=2000)

    def accuracy(preds, ds):
        labels = ds.get_label()
        return "accuracy", float(((preds > 0.5) == labels).mean()), True

    train = lgb.Dataset(X[:1500], label=y[:1500])
    valid = train.create_valid(X[1500:], label=y[1500:])
    ev = {}
    lgb.train({"objective": "binary", "metric": "none", "verbosity": -1}, train, 10,
              valid_sets=[valid], feval=accuracy, callbacks=[lgb.record_evaluation(ev)])
    assert "accuracy" in ev["valid_0"]
    assert ev["valid_0"]["accuracy"][-1] > 0.8


@pytest.mark.parametrize("boosting", ["dart", "rf"])
def test_other_boosting_modes(boosting):
    X, y = _binary_data(n=3000)
    params = {"objective": "binary", "boosting": boosting, "verbosity": -1}
    if boosting == "rf":
        params.update({"bagging_freq": 1, "bagging_fraction": 0.7})
    bst = lgb.train(params, lgb.Dataset(X, label=y), 20)
    pred = bst.predict(X)
    acc = ((pred > 0.5) == y).mean()
    assert acc > 0.75


def test_bagging_and_feature_fraction():
    X, y = _binary_data()
    bst = lgb.train({"objective": "binary", "bagging_freq": 1, "bagging_fraction": 0.6,
                     "feature_fraction": 0.7, "verbosity": -1},
                    lgb.Dataset(X, label=y), 30)
    acc = ((bst.predict(X) > 0.5) == y).mean()
    assert acc > 0.85


def test_goss():
    X, y = _binary_data()
    bst = lgb.train({"objective": "binary", "data_sample_strategy": "goss",
                     "verbosity": -1}, lgb.Dataset(X, label=y), 30)
    acc = ((bst.predict(X) > 0.5) == y).mean()
    assert acc > 0.85


def test_categorical_feature():
    rng = np.random.RandomState(0)
    n = 4000
    cat = rng.randint(0, 8, size=n)
    X = np.column_stack([cat.astype(float), rng.randn(n)])
    effect = np.array([2.0, -1.0, 0.5, -2.0, 1.5, 0.0, -0.5, 1.0])
    y = (effect[cat] + 0.3 * rng.randn(n) > 0).astype(np.float32)
    bst = lgb.train({"objective": "binary", "verbosity": -1, "min_data_in_leaf": 5},
                    lgb.Dataset(X, label=y, categorical_feature=[0]), 30)
    acc = ((bst.predict(X) > 0.5) == y).mean()
    assert acc > 0.85


def test_feature_importance():
    X, y = _binary_data()
    bst = lgb.train({"objective": "binary", "verbosity": -1}, lgb.Dataset(X, label=y), 20)
    imp_split = bst.feature_importance("split")
    imp_gain = bst.feature_importance("gain")
    assert imp_split.shape == (10,)
    # features 0,1 drive the label; they should dominate gain
    assert np.argsort(imp_gain)[-2:].tolist() in ([0, 1], [1, 0]) or \
        imp_gain[0] + imp_gain[1] > 0.5 * imp_gain.sum()


def test_predict_types():
    X, y = _binary_data(n=1000)
    bst = lgb.train({"objective": "binary", "verbosity": -1, "num_leaves": 7},
                    lgb.Dataset(X, label=y), 5)
    raw = bst.predict(X[:10], raw_score=True)
    prob = bst.predict(X[:10])
    np.testing.assert_allclose(prob, 1 / (1 + np.exp(-raw)), rtol=1e-9)
    leaves = bst.predict(X[:10], pred_leaf=True)
    assert leaves.shape == (10, 5)
    assert leaves.max() < 7
    contrib = bst.predict(X[:10], pred_contrib=True)
    assert contrib.shape == (10, 11)
    np.testing.assert_allclose(contrib.sum(axis=1), raw, rtol=1e-5, atol=1e-5)


def test_cv():
    X, y = _binary_data(n=2000)
    res = lgb.cv({"objective": "binary", "metric": "auc", "verbosity": -1},
                 lgb.Dataset(X, label=y), num_boost_round=10, nfold=3)
    assert "valid auc-mean" in res
    assert len(res["valid auc-mean"]) == 10
    assert res["valid auc-mean"][-1] > 0.85


def test_dump_model_json():
    X, y = _binary_data(n=1000)
    bst = lgb.train({"objective": "binary", "verbosity": -1}, lgb.Dataset(X, label=y), 3)
    d = bst.dump_model()
    assert d["num_class"] == 1
    assert len(d["tree_info"]) == 3
    assert "tree_structure" in d["tree_info"][0]


def test_monotone_constraints():
    rng = np.random.RandomState(0)
    X = rng.rand(3000, 2)
    y = (2 * X[:, 0] + 0.1 * rng.randn(3000)).astype(np.float32)
    bst = lgb.train({"objective": "regression", "monotone_constraints": [1, 0],
                     "verbosity": -1}, lgb.Dataset(X, label=y), 30)
    # predictions must be monotone non-decreasing in feature 0
    xs = np.linspace(0.05, 0.95, 20)
    grid = np.column_stack([xs, np.full(20, 0.5)])
    pred = bst.predict(grid)
    assert np.all(np.diff(pred) >= -1e-9)


def test_monotone_constraints_deep_propagation():
    # deep trees + adversarial interaction: monotonicity must hold at EVERY slice
    # (requires leaf-bound propagation, not just sibling-level rejection)
    rng = np.random.RandomState(7)
    X = rng.rand(6000, 3)
    y = (1.5 * X[:, 0] - 1.2 * X[:, 1] + 2.0 * np.sin(6 * X[:, 2]) +
         0.05 * rng.randn(6000)).astype(np.float32)
    bst = lgb.train({"objective": "regression", "monotone_constraints": [1, -1, 0],
                     "num_leaves": 63, "learning_rate": 0.1, "verbosity": -1},
                    lgb.Dataset(X, label=y), 60)
    xs = np.linspace(0.02, 0.98, 25)
    for other in (0.1, 0.5, 0.9):
        for z in (0.2, 0.8):
            g0 = np.column_stack([xs, np.full(25, other), np.full(25, z)])
            assert np.all(np.diff(bst.predict(g0)) >= -1e-9)     # +1 on f0
            g1 = np.column_stack([np.full(25, other), xs, np.full(25, z)])
            assert np.all(np.diff(bst.predict(g1)) <= 1e-9)      # -1 on f1


def test_weights_affect_training():
    X, y = _binary_data(n=2000)
    w = np.where(y > 0, 10.0, 1.0).astype(np.float32)
    bst = lgb.train({"objective": "binary", "verbosity": -1},
                    lgb.Dataset(X, label=y, weight=w), 20)
    pred = bst.predict(X)
    # heavy positive weights push average prediction up
    assert pred.mean() > y.mean()


def test_linear_tree():
    rng = np.random.RandomState(11)
    X = rng.rand(4000, 4) * 4
    y = (3.0 * X[:, 0] + 2.0 * X[:, 1] + 0.05 * rng.randn(4000)).astype(np.float32)
    const_bst = lgb.train({"objective": "regression", "verbosity": -1, "num_leaves": 4},
                          lgb.Dataset(X, label=y), 20)
    lin_bst = lgb.train({"objective": "regression", "verbosity": -1, "num_leaves": 4,
                         "linear_tree": True}, lgb.Dataset(X, label=y), 20)
    mse_const = float(np.mean((const_bst.predict(X) - y) ** 2))
    mse_lin = float(np.mean((lin_bst.predict(X) - y) ** 2))
    assert mse_lin < 0.5 * mse_const
    s = lin_bst.model_to_string()
    assert "is_linear=1" in s and "leaf_coeff=" in s
    b2 = lgb.Booster(model_str=s)
    np.testing.assert_allclose(b2.predict(X[:50]), lin_bst.predict(X[:50]), rtol=1e-9)


def test_cegb_penalty_reduces_features():
    rng = np.random.RandomState(12)
    X = rng.randn(3000, 10)
    y = (X[:, 0] + 0.3 * X[:, 1] + 0.2 * rng.randn(3000)).astype(np.float32)
    free = lgb.train({"objective": "regression", "verbosity": -1}, lgb.Dataset(X, label=y), 20)
    pen = lgb.train({"objective": "regression", "verbosity": -1,
                     "cegb_penalty_feature_coupled": [0.0] + [1e5] * 9},
                    lgb.Dataset(X, label=y), 20)
    nf_free = int((free.feature_importance() > 0).sum())
    nf_pen = int((pen.feature_importance() > 0).sum())
    assert nf_pen <= nf_free
    assert pen.feature_importance()[0] > 0


def test_interaction_constraints():
    rng = np.random.RandomState(13)
    X = rng.randn(3000, 4)
    y = (X[:, 0] * X[:, 1] + X[:, 2] + 0.1 * rng.randn(3000)).astype(np.float32)
    bst = lgb.train({"objective": "regression", "verbosity": -1,
                     "interaction_constraints": "[0,1],[2,3]"},
                    lgb.Dataset(X, label=y), 20)
    # every branch must stay within one group
    model = bst.dump_model()

    def check(node, path):
        if "leaf_index" in node:
            if path:
                assert set(path) <= {0, 1} or set(path) <= {2, 3}, path
            return
        check(node["left_child"], path + [node["split_feature"]])
        check(node["right_child"], path + [node["split_feature"]])

    for t in model["tree_info"]:
        if "split_index" in t["tree_structure"]:
            check(t["tree_structure"], [])


def test_r2_metric():
    rng = np.random.RandomState(14)
    X = rng.randn(2000, 5)
    y = (X[:, 0] + 0.1 * rng.randn(2000)).astype(np.float32)
    ev = {}
    train = lgb.Dataset(X, label=y)
    lgb.train({"objective": "regression", "metric": "r2", "verbosity": -1}, train, 20,
              valid_sets=[train], valid_names=["train"],
              callbacks=[lgb.record_evaluation(ev)])
    assert ev["train"]["r2"][-1] > 0.9


def test_forced_splits(tmp_path):
    rng = np.random.RandomState(31)
    X = rng.randn(3000, 4)
    y = (X[:, 0] + 0.3 * X[:, 1] + 0.2 * rng.randn(3000)).astype(np.float32)
    fs = tmp_path / "forced.json"
    fs.write_text('{"feature": 3, "threshold": 0.0, '
                  '"left": {"feature": 2, "threshold": 0.5}}')
    bst = lgb.train({"objective": "regression", "verbosity": -1,
                     "forcedsplits_filename": str(fs)}, lgb.Dataset(X, label=y), 5)
    model = bst.dump_model()
    for t in model["tree_info"]:
        root = t["tree_structure"]
        if "split_index" not in root:
            continue
        assert root["split_feature"] == 3
        assert abs(root["threshold"] - 0.0) < 0.2
        left = root["left_child"]
        if "split_index" in left:
            assert left["split_feature"] == 2


def test_position_debias_runs():
    rng = np.random.RandomState(32)
    groups = [25] * 80
    n = sum(groups)
    X = rng.randn(n, 5)
    y = np.clip((X[:, 0] + 0.5 * rng.randn(n) + 1), 0, 3).astype(int).astype(np.float32)
    pos = np.concatenate([np.arange(g) for g in groups]).astype(np.int32)
    ds = lgb.Dataset(X, label=y, group=np.array(groups, dtype=np.int32), position=pos)
    bst = lgb.train({"objective": "lambdarank", "verbosity": -1}, ds, 10)
    assert np.all(np.isfinite(bst.predict(X[:50])))


def test_auc_mu():
    rng = np.random.RandomState(33)
    X = rng.randn(3000, 5)
    y = ((X[:, 0] > 0.5).astype(int) + (X[:, 1] > 0).astype(int)).astype(np.float32)
    ev = {}
    train = lgb.Dataset(X, label=y)
    lgb.train({"objective": "multiclass", "num_class": 3, "metric": "auc_mu",
               "verbosity": -1}, train, 20, valid_sets=[train], valid_names=["t"],
              callbacks=[lgb.record_evaluation(ev)])
    assert ev["t"]["auc_mu"][-1] > 0.9


def test_quantized_training_cpu():
    """use_quantized_grad on CPU: discretized gradients still reach near-baseline
    quality, and quant_train_renew_leaf changes (de-biases) leaf values."""
    X, y = _binary_data(n=4000)
    base = lgb.train({"objective": "binary", "verbosity": -1}, lgb.Dataset(X, label=y), 30)
    q = lgb.train({"objective": "binary", "verbosity": -1, "use_quantized_grad": True,
                   "num_grad_quant_bins": 8}, lgb.Dataset(X, label=y), 30)
    from sklearn.metrics import roc_auc_score
    auc_b = roc_auc_score(y, base.predict(X))
    auc_q = roc_auc_score(y, q.predict(X))
    assert auc_q > auc_b - 0.03
    # predictions must actually differ (quantization is active)
    assert not np.allclose(base.predict(X), q.predict(X))
    r = lgb.train({"objective": "binary", "verbosity": -1, "use_quantized_grad": True,
                   "num_grad_quant_bins": 8, "quant_train_renew_leaf": True},
                  lgb.Dataset(X, label=y), 30)
    assert not np.allclose(q.predict(X), r.predict(X))
    assert roc_auc_score(y, r.predict(X)) > auc_b - 0.03


def test_bagging_by_query():
    """bagging_by_query samples whole query groups for ranking bags."""
    rng = np.random.RandomState(2)
    rows, labels, groups = [], [], []
    for q in range(120):
        nq = rng.randint(5, 20)
        Xq = rng.randn(nq, 5)
        rel = (Xq[:, 0] > 0.3).astype(int)
        rows.append(Xq); labels.append(rel); groups.append(nq)
    X = np.vstack(rows)
    y = np.concatenate(labels).astype(np.float32)
    ev = {}
    train = lgb.Dataset(X, label=y, group=np.array(groups, dtype=np.int32))
    lgb.train({"objective": "lambdarank", "metric": "ndcg", "eval_at": [5],
               "bagging_by_query": True, "bagging_fraction": 0.5, "bagging_freq": 1,
               "verbosity": -1}, train, 25,
              valid_sets=[train], valid_names=["train"],
              callbacks=[lgb.record_evaluation(ev)])
    assert ev["train"]["ndcg@5"][-1] > 0.75


def test_max_depth_enforced():
    X, y = _binary_data()
    bst = lgb.train({"objective": "binary", "max_depth": 3, "num_leaves": 255,
                     "verbosity": -1}, lgb.Dataset(X, label=y), 10)
    d = bst.dump_model()

    def depth(node, cur=0):
        if "leaf_index" in node:
            return cur
        return max(depth(node["left_child"], cur + 1),
                   depth(node["right_child"], cur + 1))
    for t in d["tree_info"]:
        assert depth(t["tree_structure"]) <= 3


def test_min_gain_to_split_prunes():
    X, y = _binary_data()
    free = lgb.train({"objective": "binary", "verbosity": -1},
                     lgb.Dataset(X, label=y), 10)
    strict = lgb.train({"objective": "binary", "min_gain_to_split": 50.0,
                        "verbosity": -1}, lgb.Dataset(X, label=y), 10)

    def count_leaves(d):
        total = 0
        for t in d["tree_info"]:
            total += t["num_leaves"]
        return total
    assert count_leaves(strict.dump_model()) < count_leaves(free.dump_model())


def test_init_score_shifts_training():
    X, y = _binary_data(n=3000)
    base = lgb.train({"objective": "binary", "verbosity": -1},
                     lgb.Dataset(X, label=y), 5)
    init = np.full(3000, 4.0)  # strong positive prior
    shifted = lgb.train({"objective": "binary", "verbosity": -1},
                        lgb.Dataset(X, label=y, init_score=init), 5)
    # raw predictions exclude the init score; the boosted part must differ
    assert not np.allclose(base.predict(X, raw_score=True),
                           shifted.predict(X, raw_score=True))


def test_quantile_objective_hits_quantile():
    rng = np.random.RandomState(0)
    X = rng.rand(8000, 3)
    y = (X[:, 0] * 2 + rng.exponential(1.0, 8000)).astype(np.float32)
    for alpha in (0.2, 0.8):
        bst = lgb.train({"objective": "quantile", "alpha": alpha, "verbosity": -1,
                         "num_leaves": 15}, lgb.Dataset(X, label=y), 80)
        frac_below = (y <= bst.predict(X)).mean()
        assert abs(frac_below - alpha) < 0.08


def test_categorical_with_missing():
    rng = np.random.RandomState(1)
    n = 4000
    cat = rng.randint(0, 6, n).astype(float)
    cat[rng.rand(n) < 0.15] = np.nan
    X = np.column_stack([cat, rng.randn(n)])
    eff = np.array([2.0, -2.0, 1.0, -1.0, 0.5, -0.5])
    y = np.where(np.isnan(cat), 0.3, eff[np.nan_to_num(cat).astype(int)])
    y = (y + 0.2 * rng.randn(n) > 0).astype(np.float32)
    bst = lgb.train({"objective": "binary", "verbosity": -1,
                     "categorical_feature": [0]}, lgb.Dataset(X, label=y), 30)
    acc = ((bst.predict(X) > 0.5) == y).mean()
    assert acc > 0.8


def test_refit_changes_values_not_structure():
    X, y = _binary_data(n=3000, seed=3)
    bst = lgb.train({"objective": "binary", "verbosity": -1},
                    lgb.Dataset(X, label=y), 10)
    X2, y2 = _binary_data(n=3000, seed=99)
    refitted = bst.refit(X2, y2, decay_rate=0.5)
    d1, d2 = bst.dump_model(), refitted.dump_model()
    for t1, t2 in zip(d1["tree_info"], d2["tree_info"]):

        def structure(node):
            if "leaf_index" in node:
                return ("leaf",)
            return (node["split_feature"], round(node["threshold"], 9),
                    structure(node["left_child"]), structure(node["right_child"]))
        assert structure(t1["tree_structure"]) == structure(t2["tree_structure"])
    assert not np.allclose(bst.predict(X), refitted.predict(X))


def test_feature_importance_types_differ():
    X, y = _binary_data()
    bst = lgb.train({"objective": "binary", "verbosity": -1}, lgb.Dataset(X, label=y), 20)
    split_imp = bst.feature_importance(importance_type="split")
    gain_imp = bst.feature_importance(importance_type="gain")
    assert split_imp.dtype.kind in "iu" or np.allclose(split_imp, split_imp.astype(int))
    assert gain_imp.sum() > 0
    # the strongest feature by gain should be one of the true signal features 0..3
    assert int(np.argmax(gain_imp)) in (0, 1, 2, 3)


def test_early_stopping_first_metric_only():
    X, y = _binary_data()
    train = lgb.Dataset(X[:4000], label=y[:4000])
    valid = train.create_valid(X[4000:], label=y[4000:])
    bst = lgb.train({"objective": "binary", "metric": ["binary_logloss", "auc"],
                     "first_metric_only": True, "verbosity": -1}, train, 300,
                    valid_sets=[valid], callbacks=[lgb.early_stopping(5, verbose=False,
                                                                      first_metric_only=True)])
    assert 0 < bst.best_iteration < 300


def test_ndcg_matches_manual():
    """ndcg@k reported by the metric matches a direct computation."""
    rng = np.random.RandomState(0)
    rows, labels, groups = [], [], []
    for q in range(50):
        nq = rng.randint(6, 15)
        Xq = rng.randn(nq, 4)
        rel = rng.randint(0, 3, nq)
        rows.append(Xq); labels.append(rel); groups.append(nq)
    X = np.vstack(rows); y = np.concatenate(labels).astype(np.float32)
    g = np.array(groups, dtype=np.int32)
    train = lgb.Dataset(X, label=y, group=g)
    ev = {}
    bst = lgb.train({"objective": "lambdarank", "metric": "ndcg", "eval_at": [3],
                     "verbosity": -1}, train, 5, valid_sets=[train],
                    valid_names=["t"], callbacks=[lgb.record_evaluation(ev)])
    reported = ev["t"]["ndcg@3"][-1]
    # manual NDCG@3
    scores = bst.predict(X)
    pos = 0
    total = 0.0
    for nq in groups:
        s, rel = scores[pos:pos+nq], y[pos:pos+nq]
        pos += nq
        order = np.argsort(-s, kind="stable")
        dcg = sum((2**rel[order[i]] - 1) / np.log2(i + 2) for i in range(min(3, nq)))
        ideal = np.sort(rel)[::-1]
        idcg = sum((2**ideal[i] - 1) / np.log2(i + 2) for i in range(min(3, nq)))
        total += dcg / idcg if idcg > 0 else 1.0
    np.testing.assert_allclose(reported, total / len(groups), rtol=1e-6)


def test_prediction_early_stopping():
    X, y = _binary_data(n=2000)
    bst = lgb.train({"objective": "binary", "verbosity": -1}, lgb.Dataset(X, label=y), 60)
    exact = bst.predict(X[:200])
    fast = bst.predict(X[:200], pred_early_stop=True, pred_early_stop_freq=10,
                       pred_early_stop_margin=1.5)
    # labels agree even where probabilities differ slightly
    assert (((exact > 0.5) == (fast > 0.5)).mean()) > 0.98


def test_hist_mode_forced_equivalence():
    """force_col_wise and force_row_wise must train models of equal quality
    (different summation order, same splits on well-separated data)."""
    X, y = _binary_data(n=6000)
    col = lgb.train({"objective": "binary", "force_col_wise": True, "verbosity": -1},
                    lgb.Dataset(X, label=y), 20)
    row = lgb.train({"objective": "binary", "force_row_wise": True, "verbosity": -1},
                    lgb.Dataset(X, label=y), 20)
    acc_c = ((col.predict(X) > 0.5) == y).mean()
    acc_r = ((row.predict(X) > 0.5) == y).mean()
    assert acc_c > 0.9 and acc_r > 0.9
    np.testing.assert_allclose(col.predict(X[:200]), row.predict(X[:200]),
                               rtol=1e-6, atol=1e-6)


def test_monotone_penalty_discourages_shallow_monotone_splits():
    rng = np.random.RandomState(0)
    X = rng.rand(4000, 2)
    y = (2 * X[:, 0] + 2 * X[:, 1] + 0.1 * rng.randn(4000)).astype(np.float32)
    p = {"objective": "regression", "monotone_constraints": [1, 0], "verbosity": -1}
    free = lgb.train(p, lgb.Dataset(X, label=y), 10)
    pen = lgb.train({**p, "monotone_penalty": 2.0}, lgb.Dataset(X, label=y), 10)

    def root_feature(b):
        return b.dump_model()["tree_info"][0]["tree_structure"]["split_feature"]
    # with a strong penalty the constrained feature 0 should lose the root split
    assert root_feature(pen) == 1
    assert not np.allclose(free.predict(X[:100]), pen.predict(X[:100]))


def test_dart_modes():
    """uniform_drop / weighted (default) / xgboost_dart_mode all train sane models."""
    X, y = _binary_data(n=3000)
    for extra in ({}, {"uniform_drop": True}, {"xgboost_dart_mode": True}):
        bst = lgb.train({"objective": "binary", "boosting": "dart", "drop_rate": 0.3,
                         "verbosity": -1, **extra}, lgb.Dataset(X, label=y), 25)
        acc = ((bst.predict(X) > 0.5) == y).mean()
        assert acc > 0.8, extra


def test_cv_eval_train_metric_and_plotting_smoke():
    X, y = _binary_data(n=2000)
    r = lgb.cv({"objective": "binary", "metric": "auc", "verbosity": -1},
               lgb.Dataset(X, label=y), 10, nfold=3, eval_train_metric=True,
               return_cvbooster=True)
    assert "train auc-mean" in r and "valid auc-mean" in r
    assert len(r["cvbooster"].boosters) == 3
    # plotting smoke (matplotlib Agg)
    try:
        import matplotlib
        matplotlib.use("Agg")
    except ImportError:
        return
    bst = lgb.train({"objective": "binary", "metric": "auc", "verbosity": -1},
                    lgb.Dataset(X, label=y), 5)
    ev = {}
    lgb.train({"objective": "binary", "metric": "auc", "verbosity": -1},
              lgb.Dataset(X, label=y), 5,
              valid_sets=[lgb.Dataset(X, label=y).create_valid(X, label=y)],
              callbacks=[lgb.record_evaluation(ev)])
    ax = lgb.plot_importance(bst)
    assert ax is not None
    ax2 = lgb.plot_metric(ev)
    assert ax2 is not None
    import pytest as _pt
    with _pt.raises(ImportError):
        lgb.plot_tree(bst, tree_index=0)  # graphviz absent here (reference parity)


def test_multiple_valid_sets_and_names():
    X, y = _binary_data(n=2000)
    tr = lgb.Dataset(X, label=y)
    ev = {}
    lgb.train({"objective": "binary", "metric": "auc", "verbosity": -1}, tr, 5,
              valid_sets=[tr.create_valid(X[:500], label=y[:500]),
                          tr.create_valid(X[500:], label=y[500:])],
              valid_names=["a", "b"], callbacks=[lgb.record_evaluation(ev)])
    assert set(ev.keys()) == {"a", "b"}


def test_early_stopping_min_delta():
    X, y = _binary_data()
    tr = lgb.Dataset(X[:4000], label=y[:4000])
    bst = lgb.train({"objective": "binary", "metric": "auc", "verbosity": -1}, tr, 100,
                    valid_sets=[tr.create_valid(X[4000:], label=y[4000:])],
                    callbacks=[lgb.early_stopping(5, min_delta=0.5, verbose=False)])
    # a 0.5 AUC min_delta is unreachable -> stops at patience
    assert bst.best_iteration <= 6


def test_reset_parameter_callback_changes_lr():
    X, y = _binary_data(n=2000)
    sched = [0.3, 0.2, 0.1, 0.05, 0.02]
    bst = lgb.train({"objective": "binary", "verbosity": -1}, lgb.Dataset(X, label=y),
                    len(sched), callbacks=[lgb.reset_parameter(learning_rate=sched)])
    # shrinkage shows up in per-tree leaf magnitudes: tree0 built at lr 0.3,
    # tree4 at lr 0.02 -> much smaller outputs
    d = bst.dump_model()

    def max_abs_leaf(node):
        if "leaf_index" in node:
            return abs(node["leaf_value"])
        return max(max_abs_leaf(node["left_child"]), max_abs_leaf(node["right_child"]))
    assert max_abs_leaf(d["tree_info"][4]["tree_structure"]) < \
        max_abs_leaf(d["tree_info"][0]["tree_structure"])


@pytest.mark.parametrize("objective", ["cross_entropy", "cross_entropy_lambda"])
def test_xentropy_probabilistic_labels(objective):
    rng = np.random.RandomState(0)
    X = rng.randn(2000, 5)
    yq = np.clip((X[:, 0] > 0) + 0.2 * rng.rand(2000), 0, 1).astype(np.float32)
    bst = lgb.train({"objective": objective, "verbosity": -1}, lgb.Dataset(X, label=yq), 10)
    p = bst.predict(X)
    assert np.all(p >= 0)
    if objective == "cross_entropy":
        assert np.all(p <= 1)  # probability link; the _lambda link is an intensity
    assert np.corrcoef(p, yq)[0, 1] > 0.7


def test_multiclassova():
    rng = np.random.RandomState(0)
    X = rng.randn(3000, 5)
    y = ((X[:, 0] > 0.5).astype(int) + (X[:, 1] > 0).astype(int)).astype(np.float32)
    bst = lgb.train({"objective": "multiclassova", "num_class": 3, "verbosity": -1},
                    lgb.Dataset(X, label=y), 20)
    pred = bst.predict(X)
    assert pred.shape == (3000, 3)
    assert (pred.argmax(axis=1) == y).mean() > 0.7


def test_average_precision_metric():
    X, y = _binary_data(n=3000)
    tr = lgb.Dataset(X, label=y)
    ev = {}
    lgb.train({"objective": "binary", "metric": "average_precision", "verbosity": -1},
              tr, 10, valid_sets=[tr.create_valid(X, label=y)],
              callbacks=[lgb.record_evaluation(ev)])
    ap = ev["valid_0"]["average_precision"]
    assert ap[-1] > 0.9 and ap[-1] >= ap[0] - 1e-9


def test_cvbooster_persistence(tmp_path):
    """CVBooster pickles and saves/loads as a multi-fold model file."""
    import pickle
    X, y = _binary_data(n=1000)
    r = lgb.cv({"objective": "binary", "verbosity": -1}, lgb.Dataset(X, label=y),
               5, nfold=3, return_cvbooster=True)
    cvb = r["cvbooster"]
    cvb2 = pickle.loads(pickle.dumps(cvb))
    assert len(cvb2.boosters) == 3
    f = tmp_path / "cv.txt"
    cvb.save_model(str(f))
    cvb3 = lgb.CVBooster(model_file=str(f))
    a = np.mean([b.predict(X[:50]) for b in cvb.boosters], axis=0)
    b = np.mean([m.predict(X[:50]) for m in cvb3.boosters], axis=0)
    np.testing.assert_allclose(a, b, rtol=1e-12)


def test_forced_splits_respect_monotone(tmp_path):
    """Forced splits clamp into monotone bounds and propagate them (interaction
    of two constraint systems)."""
    import json
    rng = np.random.RandomState(0)
    X = rng.rand(4000, 4)
    y = (2 * X[:, 0] + X[:, 1] > 1.2).astype(np.float32)
    fs = tmp_path / "forced.json"
    fs.write_text(json.dumps({"feature": 0, "threshold": 0.5,
                              "left": {"feature": 1, "threshold": 0.5}}))
    bst = lgb.train({"objective": "binary", "verbosity": -1,
                     "forcedsplits_filename": str(fs),
                     "monotone_constraints": [1, 0, 0, 0],
                     "bagging_freq": 1, "bagging_fraction": 0.7},
                    lgb.Dataset(X, label=y), 20)
    xs = np.linspace(0.05, 0.95, 20)
    for other in (0.2, 0.5, 0.8):
        grid = np.column_stack([xs] + [np.full(20, other)] * 3)
        assert np.all(np.diff(bst.predict(grid)) >= -1e-9)


def test_linear_tree_with_valid_sets():
    """Linear trees score valid sets through raw feature values (regression: this
    used to segfault — valid datasets carried no raw values)."""
    rng = np.random.RandomState(0)
    X = rng.randn(4000, 6)
    y = (2 * X[:, 0] + np.sin(X[:, 1])).astype(np.float32)
    tr = lgb.Dataset(X, label=y)
    ev = {}
    bst = lgb.train({"objective": "regression", "linear_tree": True, "metric": "l2",
                     "verbosity": -1}, tr, 20,
                    valid_sets=[tr.create_valid(X[:800], label=y[:800])],
                    callbacks=[lgb.record_evaluation(ev)])
    assert ev["valid_0"]["l2"][-1] < 0.5 * float(np.var(y))
    # valid eval equals direct prediction MSE (raw values wired correctly)
    mse = float(np.mean((bst.predict(X[:800]) - y[:800]) ** 2))
    assert abs(mse - ev["valid_0"]["l2"][-1]) < 1e-6


def test_deterministic_thread_invariance():
    """deterministic=true: identical trees regardless of num_threads."""
    X, y = _binary_data(n=3000)

    def trees(m):
        return m[m.index("Tree=0"):m.index("end of trees")]
    ms = [trees(lgb.train({"objective": "binary", "verbosity": -1,
                           "deterministic": True, "num_threads": t},
                          lgb.Dataset(X, label=y), 10).model_to_string())
          for t in (1, 4, 8)]
    assert ms[0] == ms[1] == ms[2]


def test_unbound_valid_set_auto_references_train():
    """A valid set passed without reference= is auto-bound to the training
    data's bin mappers by train() (independently-binned valid
    sets silently corrupt eval metrics and early stopping)."""
    X, y = _regression_data()
    tr = lgb.Dataset(X[:4000], label=y[:4000])
    # plain Dataset, NO reference= — subset of the training data
    va = lgb.Dataset(X[:800], label=y[:800])
    ev = {}
    lgb.train({"objective": "regression", "metric": "l2", "verbosity": -1},
              tr, 30, valid_sets=[va], callbacks=[lgb.record_evaluation(ev)])
    l2 = ev["valid_0"]["l2"]
    # on a training subset l2 must improve monotonically-ish and end well below var
    assert l2[-1] < 0.5 * float(np.var(y[:800]))
    assert l2[-1] < l2[0]


def test_misaligned_valid_set_rejected():
    """Adding an independently-constructed (differently-binned) valid set to a
    Booster raises instead of producing garbage metrics."""
    X, y = _regression_data()
    tr = lgb.Dataset(X[:4000], label=y[:4000]).construct()
    va = lgb.Dataset(X[4000:] * 3.7 + 1.0, label=y[4000:]).construct()
    bst = lgb.Booster(params={"objective": "regression", "verbosity": -1},
                      train_set=tr)
    with pytest.raises(lgb.LightGBMError):
        bst.add_valid(va, "bad")


def test_early_stopping_respects_r2_direction():
    """r2 is higher-better; early stopping must not stop while it improves
    (ADVICE r1 medium)."""
    X, y = _regression_data()
    tr = lgb.Dataset(X[:4000], label=y[:4000])
    va = tr.create_valid(X[4000:], label=y[4000:])
    ev = {}
    bst = lgb.train({"objective": "regression", "metric": "r2", "verbosity": -1,
                     "learning_rate": 0.1},
                    tr, 60, valid_sets=[va],
                    callbacks=[lgb.early_stopping(5), lgb.record_evaluation(ev)])
    r2 = ev["valid_0"]["r2"]
    # r2 climbs for many rounds on this problem; a direction bug stops at iter 1
    assert bst.best_iteration > 10
    assert max(r2) > 0.8


def test_save_model_keeps_pandas_categorical(tmp_path):
    """save_model writes the pandas_categorical trailer so a file round-trip
    preserves the training category mapping."""
    pd = pytest.importorskip("pandas")
    rng = np.random.RandomState(3)
    n = 2000
    df = pd.DataFrame({
        "num": rng.randn(n),
        "cat": pd.Categorical(rng.choice(["a", "b", "c", "d"], size=n)),
    })
    y = (df["num"] + (df["cat"].cat.codes % 2) > 0.3).astype(np.float32)
    bst = lgb.train({"objective": "binary", "verbosity": -1},
                    lgb.Dataset(df, label=y), 15)
    p_before = bst.predict(df)
    path = tmp_path / "m.txt"
    bst.save_model(path)
    bst2 = lgb.Booster(model_file=str(path))
    assert bst2.pandas_categorical is not None
    # shuffled category declaration order must still map via the saved trailer
    df2 = df.copy()
    df2["cat"] = pd.Categorical(df["cat"].astype(str),
                                categories=["d", "c", "b", "a"])
    np.testing.assert_allclose(bst2.predict(df2), p_before, rtol=1e-9)


def test_histogram_pool_size_cap():
    """histogram_pool_size caps slot memory: a tiny pool forces LRU eviction and
    subtraction fallbacks but must not change the learned model's quality."""
    X, y = _binary_data(n=20000)
    preds = {}
    for pool in (-1, 0.05):  # unlimited vs ~3-slot pool at 63 bins
        params = {"objective": "binary", "verbosity": -1, "num_leaves": 63,
                  "max_bin": 63, "histogram_pool_size": pool}
        bst = lgb.train(params, lgb.Dataset(X, label=y), 15)
        preds[pool] = bst.predict(X[:4000])
    # identical data + config: pooled histograms recompute instead of subtract
    # (different fp addition order can flip near-tie splits) -> quality parity
    from sklearn.metrics import roc_auc_score
    a_unl = roc_auc_score(y[:4000], preds[-1])
    a_cap = roc_auc_score(y[:4000], preds[0.05])
    assert a_cap > 0.9
    assert abs(a_unl - a_cap) < 5e-3, (a_unl, a_cap)


def test_new_reference_params_accepted():
    """The 8 remaining reference parameter_set entries parse and round-trip."""
    X, y = _binary_data(n=4000)
    params = {"objective": "binary", "verbosity": -1, "num_leaves": 15,
              "saved_feature_importance_type": 1, "precise_float_parser": True,
              "pred_early_stop": True, "pred_early_stop_freq": 5,
              "pred_early_stop_margin": 5.0,
              "lambdarank_position_bias_regularization": 0.1}
    bst = lgb.train(params, lgb.Dataset(X, label=y), 5)
    assert bst.num_trees() == 5


@pytest.mark.parametrize("method", ["intermediate", "advanced"])
def test_monotone_intermediate_policy(method):
    """monotone_constraints_method=intermediate/advanced: output-tight bounds +
    contiguous-leaf re-evaluation. Predictions must be globally
    monotone in the constrained features, and quality must not regress vs basic."""
    rng = np.random.RandomState(0)
    n = 20000
    X = rng.rand(n, 4)
    y = (2.0 * X[:, 0] - 1.5 * X[:, 1] + np.sin(6 * X[:, 2]) +
         0.1 * rng.randn(n)).astype(np.float32)
    preds = {}
    for m in ("basic", method):
        params = {"objective": "regression", "verbosity": -1, "num_leaves": 63,
                  "monotone_constraints": [1, -1, 0, 0],
                  "monotone_constraints_method": m}
        bst = lgb.train(params, lgb.Dataset(X, label=y), 40)
        preds[m] = bst
    # global monotonicity on dense grids across many slices
    xs = np.linspace(0.01, 0.99, 40)
    for other in (0.15, 0.5, 0.85):
        for o2 in (0.2, 0.7):
            g_up = np.column_stack([xs, np.full(40, other), np.full(40, o2),
                                    np.full(40, other)])
            p = preds[method].predict(g_up)
            assert np.all(np.diff(p) >= -1e-9), "increasing constraint violated"
            g_dn = np.column_stack([np.full(40, other), xs, np.full(40, o2),
                                    np.full(40, other)])
            p = preds[method].predict(g_dn)
            assert np.all(np.diff(p) <= 1e-9), "decreasing constraint violated"
    # intermediate's looser child bounds must not hurt accuracy vs basic
    mse_b = float(np.mean((preds["basic"].predict(X) - y) ** 2))
    mse_i = float(np.mean((preds[method].predict(X) - y) ** 2))
    assert mse_i < mse_b * 1.05, (mse_b, mse_i)


def test_forced_bins(tmp_path):
    """forcedbins_filename: forced bin upper bounds are honored exactly
    (reference test_engine.py test_forced_bins semantics)."""
    import json as _json
    rng = np.random.RandomState(0)
    n = 5000
    X = rng.rand(n, 2)
    y = (10 * (X[:, 0] > 0.33) + 5 * (X[:, 1] > 0.66) +
         0.1 * rng.randn(n)).astype(np.float32)
    fb = tmp_path / "forced_bins.json"
    fb.write_text(_json.dumps([
        {"feature": 0, "bin_upper_bound": [0.33, 0.54]},
        {"feature": 1, "bin_upper_bound": [0.66]},
    ]))
    params = {"objective": "regression", "verbosity": -1, "num_leaves": 8,
              "max_bin": 6, "forcedbins_filename": str(fb)}
    bst = lgb.train(params, lgb.Dataset(X, label=y), 30)
    # with only 6 coarse quantile bins the exact 0.33/0.66 steps would be missed;
    # the forced boundaries make the model resolve them precisely
    eps = 1e-4
    p = bst.predict(np.array([[0.33 - eps, 0.5], [0.33 + eps, 0.5]]))
    assert abs(p[1] - p[0]) > 5.0, p  # a ~10-unit jump exactly at 0.33
    q = bst.predict(np.array([[0.5, 0.66 - eps], [0.5, 0.66 + eps]]))
    assert abs(q[1] - q[0]) > 2.5, q
    # and the split thresholds in the model sit exactly on the forced bounds
    d = bst.dump_model()
    thresholds = []
    def walk(node):
        if "split_feature" in node:
            thresholds.append((node["split_feature"], node["threshold"]))
            walk(node["left_child"]); walk(node["right_child"])
    for t in d["tree_info"]:
        walk(t["tree_structure"])
    assert any(f == 0 and abs(thr - 0.33) < 1e-9 for f, thr in thresholds)


def test_max_bin_by_feature():
    """max_bin_by_feature caps per-feature bin counts (reference parity)."""
    rng = np.random.RandomState(0)
    X = rng.rand(3000, 2)
    y = (X[:, 0] + X[:, 1] + 0.1 * rng.randn(3000)).astype(np.float32)
    ds = lgb.Dataset(X, label=y,
                     params={"max_bin": 255, "max_bin_by_feature": [4, 255],
                             "min_data_in_bin": 1}).construct()
    assert ds.num_feature() == 2
    nb0 = ds.feature_num_bin(0)
    nb1 = ds.feature_num_bin(1)
    assert nb0 <= 4, nb0
    assert nb1 > 100, nb1


def test_constant_features_all_objectives():
    """Constant columns are pre-filtered but predictions keep working and the
    model keeps the full feature surface (reference test_constant_features_*)."""
    rng = np.random.RandomState(0)
    X = np.column_stack([rng.randn(2000), np.full(2000, 3.0), rng.randn(2000)])
    for objective, y in [
        ("binary", (X[:, 0] > 0).astype(np.float32)),
        ("regression", (X[:, 0] + 0.1 * rng.randn(2000)).astype(np.float32)),
        ("multiclass", (np.abs(X[:, 0]).astype(int) % 3).astype(np.float32)),
    ]:
        params = {"objective": objective, "verbosity": -1, "num_leaves": 7}
        if objective == "multiclass":
            params["num_class"] = 3
        bst = lgb.train(params, lgb.Dataset(X, label=y), 5)
        assert bst.num_feature() == 3
        p = bst.predict(X[:11])
        assert p.shape[0] == 11
        # the constant column never splits
        assert all(f != 1 for f, _ in _all_split_features(bst))


def _all_split_features(bst):
    out = []
    def walk(node):
        if "split_feature" in node:
            out.append((node["split_feature"], node["threshold"]))
            walk(node["left_child"]); walk(node["right_child"])
    for t in bst.dump_model()["tree_info"]:
        walk(t["tree_structure"])
    return out


def test_boost_from_average_with_single_leaf_trees():
    """All-constant data: every tree is a stump and predictions equal the label
    mean (reference test_boost_from_average_with_single_leaf_trees)."""
    X = np.full((200, 2), 1.0)
    y = np.full(200, 5.0, dtype=np.float32)
    bst = lgb.train({"objective": "regression", "verbosity": -1},
                    lgb.Dataset(X, label=y), 3)
    np.testing.assert_allclose(bst.predict(X[:5]), 5.0, rtol=1e-6)


def test_predict_output_shapes():
    """Prediction output shapes for regression/binary/multiclass and pred_leaf /
    pred_contrib (reference test_predict_*_output_shape)."""
    rng = np.random.RandomState(1)
    X = rng.randn(800, 4)
    yb = (X[:, 0] > 0).astype(np.float32)
    ym = (np.abs(X[:, 1]).astype(int) % 3).astype(np.float32)
    b = lgb.train({"objective": "binary", "verbosity": -1, "num_leaves": 7},
                  lgb.Dataset(X, label=yb), 5)
    assert b.predict(X[:9]).shape == (9,)
    assert b.predict(X[:9], pred_leaf=True).shape == (9, 5)
    assert b.predict(X[:9], pred_contrib=True).shape == (9, 5)  # nf + bias
    m = lgb.train({"objective": "multiclass", "num_class": 3, "verbosity": -1,
                   "num_leaves": 7}, lgb.Dataset(X, label=ym), 4)
    assert m.predict(X[:9]).shape == (9, 3)
    assert m.predict(X[:9], pred_leaf=True).shape == (9, 12)
    assert m.predict(X[:9], pred_contrib=True).shape == (9, 15)


def test_predict_with_start_iteration():
    """start_iteration slices the tree range (reference parity): full prediction
    equals start-slice + head-slice raw scores."""
    X, y = _regression_data(n=4000)
    bst = lgb.train({"objective": "regression", "verbosity": -1, "num_leaves": 15},
                    lgb.Dataset(X, label=y), 20)
    full = bst.predict(X[:50], raw_score=True)
    head = bst.predict(X[:50], raw_score=True, num_iteration=8)
    tail = bst.predict(X[:50], raw_score=True, start_iteration=8, num_iteration=12)
    np.testing.assert_allclose(head + tail, full, rtol=1e-9)


def test_predict_shape_check():
    """Mismatched feature count raises unless predict_disable_shape_check
    (reference parity)."""
    X, y = _binary_data(n=2000)
    bst = lgb.train({"objective": "binary", "verbosity": -1, "num_leaves": 7},
                    lgb.Dataset(X, label=y), 3)
    with pytest.raises(lgb.LightGBMError, match="number of features"):
        bst.predict(X[:5, :6])
    p = bst.predict(X[:5, :6], predict_disable_shape_check=True)
    assert p.shape == (5,)


def test_binning_same_sign():
    """bins split correctly when a feature is all-positive or all-negative
    (ref test_engine.py:3473)."""
    x = np.empty((99, 2))
    x[:, 0] = np.arange(0.01, 1, 0.01)
    x[:, 1] = -np.arange(0.01, 1, 0.01)
    y = np.arange(0.01, 1, 0.01)
    params = {"objective": "regression_l1", "max_bin": 5, "num_leaves": 2,
              "min_data_in_leaf": 1, "verbosity": -1, "seed": 0}
    est = lgb.train(params, lgb.Dataset(x, label=y), num_boost_round=20)
    new_x = np.zeros((3, 2))
    new_x[:, 0] = [-1, 0, 1]
    p = est.predict(new_x)
    assert p[0] == pytest.approx(p[1])       # -1 and 0 both below the positive range
    assert p[1] != pytest.approx(p[2])
    new_x = np.zeros((3, 2))
    new_x[:, 1] = [-1, 0, 1]
    p = est.predict(new_x)
    assert p[0] != pytest.approx(p[1])
    assert p[1] == pytest.approx(p[2])       # 0 and 1 both above the negative range


def test_sliced_data():
    """non-contiguous numpy/CSR slices train identically to contiguous copies
    (ref test_engine.py:2064)."""
    from scipy.sparse import csr_matrix
    rng = np.random.RandomState(5)

    def train_pred(features, labels):
        ds = lgb.Dataset(features, label=labels)
        gbm = lgb.train({"objective": "binary", "verbosity": -1, "min_data_in_leaf": 5},
                        ds, num_boost_round=10)
        return gbm.predict(features)

    n = 100
    features = rng.uniform(size=(n, 5))
    labels = np.append(np.ones(25, dtype=np.float32), np.zeros(75, dtype=np.float32))
    origin = train_pred(features, labels)
    sliced_labels = np.column_stack((labels, np.ones(n, dtype=np.float32)))[:, 0]
    np.testing.assert_allclose(origin, train_pred(features, sliced_labels))
    stacked = np.column_stack((np.ones(n), np.ones(n), features, np.ones(n), np.ones(n)))
    stacked = np.vstack([np.ones((2, 9)), stacked, np.ones((2, 9))])
    sliced = stacked[2:102, 2:7]
    assert np.all(sliced == features)
    np.testing.assert_allclose(origin, train_pred(sliced, sliced_labels))
    sliced_csr = csr_matrix(stacked)[2:102, 2:7]
    np.testing.assert_allclose(origin, train_pred(sliced_csr, sliced_labels))


def test_non_ascii_feature_names():
    rng = np.random.RandomState(6)
    X = rng.randn(300, 3)
    y = X[:, 0]
    names = ["F_零", "F_一", "渋谷"]
    ds = lgb.Dataset(X, label=y, feature_name=names)
    bst = lgb.train({"objective": "regression", "verbosity": -1}, ds, num_boost_round=5)
    assert bst.feature_name() == names
    s = bst.model_to_string()
    bst2 = lgb.Booster(model_str=s)
    assert bst2.feature_name() == names
    np.testing.assert_allclose(bst2.predict(X[:20]), bst.predict(X[:20]), rtol=1e-12)


def test_get_split_value_histogram():
    """histogram of split thresholds per feature (ref test_engine.py get_split_value_histogram)."""
    rng = np.random.RandomState(7)
    X = rng.randn(2000, 4)
    y = 3 * X[:, 0] + X[:, 1] + 0.1 * rng.randn(2000)
    bst = lgb.train({"objective": "regression", "verbosity": -1},
                    lgb.Dataset(X, label=y), num_boost_round=20)
    counts, edges = bst.get_split_value_histogram(0)
    assert counts.sum() > 0
    assert len(edges) == len(counts) + 1
    # splits on the dominant feature outnumber a weak feature's
    c3, _ = bst.get_split_value_histogram(3)
    assert counts.sum() >= c3.sum()
    # by name
    cn, _ = bst.get_split_value_histogram("Column_0")
    assert cn.sum() == counts.sum()


def test_dataset_subset_training():
    """Dataset.subset trains on the row subset; parent raw data retained
    (ref test_engine.py test_init_with_subset)."""
    rng = np.random.RandomState(8)
    X = rng.uniform(size=(50, 2))
    y = np.array([1] * 25 + [0] * 25, dtype=np.float32)
    full = lgb.Dataset(X, y, free_raw_data=False)
    idx1 = np.sort(rng.choice(np.arange(50), 30, replace=False))
    sub1 = full.subset(idx1)
    idx2 = np.sort(rng.choice(np.arange(50), 20, replace=False))
    sub2 = full.subset(idx2)
    params = {"objective": "binary", "verbosity": -1, "min_data_in_leaf": 3}
    m1 = lgb.train(params, sub1, num_boost_round=10, keep_training_booster=True)
    m2 = lgb.train(params, sub2, num_boost_round=10, init_model=m1)
    assert m2.num_trees() == 20
    assert full.get_data().shape[0] == 50
    assert sub1.num_data() == 30
    assert sub2.num_data() == 20


def test_monotone_advanced_precise_mode():
    """monotone_constraints_method=advanced (monotone precise): per-threshold
    piecewise bounds recomputed from the live tree (reference
    AdvancedLeafConstraints). Must stay globally monotone, must actually differ
    from the intermediate policy, and the extra split freedom must not hurt fit."""
    rng = np.random.RandomState(3)
    n = 30000
    X = rng.rand(n, 4)
    y = (3.0 * X[:, 0] ** 2 - 2.0 * X[:, 1] + 1.5 * np.sin(7 * X[:, 2]) * X[:, 0] +
         0.1 * rng.randn(n)).astype(np.float32)
    models = {}
    for m in ("basic", "intermediate", "advanced"):
        params = {"objective": "regression", "verbosity": -1, "num_leaves": 127,
                  "learning_rate": 0.15, "min_data_in_leaf": 5,
                  "monotone_constraints": [1, -1, 0, 0],
                  "monotone_constraints_method": m}
        models[m] = lgb.train(params, lgb.Dataset(X, label=y), 50)
    # dense monotonicity audit over random slices
    xs = np.linspace(0.01, 0.99, 60)
    slice_rng = np.random.RandomState(17)
    for _ in range(12):
        o = slice_rng.rand(3)
        g_up = np.column_stack([xs, np.full(60, o[0]), np.full(60, o[1]),
                                np.full(60, o[2])])
        assert np.all(np.diff(models["advanced"].predict(g_up)) >= -1e-9)
        g_dn = np.column_stack([np.full(60, o[0]), xs, np.full(60, o[1]),
                                np.full(60, o[2])])
        assert np.all(np.diff(models["advanced"].predict(g_dn)) <= 1e-9)
    # the precise policy must be a distinct tree-growth policy
    assert not np.allclose(models["advanced"].predict(X[:2000]),
                           models["intermediate"].predict(X[:2000]))
    # and its looser-but-exact bounds must fit at least as well as intermediate
    mse = {m: float(np.mean((b.predict(X) - y) ** 2)) for m, b in models.items()}
    assert mse["advanced"] <= mse["intermediate"] * 1.02, mse


def test_missing_value_handle_dense_default():
    """NaN rows learn their own default direction (ref test_engine.py:~test_missing_value_handle)."""
    rng = np.random.RandomState(21)
    X = np.zeros((100, 1))
    y = np.zeros(100)
    trues = rng.choice(100, 20, replace=False)
    X[trues, 0] = np.nan
    y[trues] = 1
    ds = lgb.Dataset(X, label=y)
    res = {}
    bst = lgb.train({"objective": "regression", "metric": "l2", "verbosity": -1,
                     "boost_from_average": False}, ds, 20,
                    valid_sets=[lgb.Dataset(X, label=y)],
                    callbacks=[lgb.record_evaluation(res)])
    mse = float(np.mean((bst.predict(X) - y) ** 2))
    assert mse < 0.005
    assert res["valid_0"]["l2"][-1] == pytest.approx(mse)


def test_missing_value_handle_more_na():
    """majority-NaN column: NaN becomes the most-frequent bin yet still separates."""
    rng = np.random.RandomState(22)
    X = np.ones((100, 1))
    y = np.ones(100)
    trues = rng.choice(100, 80, replace=False)
    X[trues, 0] = np.nan
    y[trues] = 0
    bst = lgb.train({"objective": "regression", "metric": "l2", "verbosity": -1,
                     "boost_from_average": False}, lgb.Dataset(X, label=y), 20)
    assert float(np.mean((bst.predict(X) - y) ** 2)) < 0.005


def test_missing_value_handle_na():
    """one split separates {0..3, NaN} from {4..7} with NaN default-left."""
    x = [0, 1, 2, 3, 4, 5, 6, 7, np.nan]
    y = [1, 1, 1, 1, 0, 0, 0, 0, 1]
    X = np.array(x).reshape(-1, 1)
    params = {"objective": "regression", "verbosity": -1, "boost_from_average": False,
              "min_data_in_leaf": 1, "num_leaves": 2, "learning_rate": 1,
              "min_data_in_bin": 1, "zero_as_missing": False}
    bst = lgb.train(params, lgb.Dataset(X, label=np.array(y, dtype=np.float64)), 1)
    np.testing.assert_allclose(bst.predict(X), y, atol=1e-9)


def test_missing_value_handle_zero():
    """zero_as_missing=True routes both 0 and NaN through the missing bin."""
    x = [0, 1, 2, 3, 4, 5, 6, 7, np.nan]
    y = [0, 1, 1, 1, 0, 0, 0, 0, 0]
    X = np.array(x).reshape(-1, 1)
    params = {"objective": "regression", "verbosity": -1, "boost_from_average": False,
              "min_data_in_leaf": 1, "num_leaves": 2, "learning_rate": 1,
              "min_data_in_bin": 1, "zero_as_missing": True}
    bst = lgb.train(params, lgb.Dataset(X, label=np.array(y, dtype=np.float64)), 1)
    np.testing.assert_allclose(bst.predict(X), y, atol=1e-9)


def test_missing_value_handle_none():
    """use_missing=False: NaN is treated exactly like 0."""
    x = [0, 1, 2, 3, 4, 5, 6, 7, np.nan]
    y = [0, 1, 1, 1, 0, 0, 0, 0, 0]
    X = np.array(x).reshape(-1, 1)
    params = {"objective": "regression", "verbosity": -1, "boost_from_average": False,
              "min_data_in_leaf": 1, "num_leaves": 2, "learning_rate": 1,
              "min_data_in_bin": 1, "use_missing": False}
    bst = lgb.train(params, lgb.Dataset(X, label=np.array(y, dtype=np.float64)), 1)
    pred = bst.predict(X)
    assert pred[-1] == pytest.approx(pred[0])  # NaN row == 0 row
    from sklearn.metrics import roc_auc_score
    assert roc_auc_score(y, pred) > 0.8


def test_predict_stump():
    """num_leaves>=2 is unreachable on pure-noise root: stump tree predicts the
    boosted average everywhere (ref test_predict_stump / boost_from_average)."""
    rng = np.random.RandomState(23)
    X = rng.randn(500, 3)
    y = np.full(500, 3.7)
    bst = lgb.train({"objective": "regression", "verbosity": -1},
                    lgb.Dataset(X, label=y), 5)
    pred = bst.predict(rng.randn(20, 3))
    np.testing.assert_allclose(pred, 3.7, rtol=1e-6)
    d = bst.dump_model()
    assert d["tree_info"][0]["num_leaves"] == 1


def test_small_max_bin():
    """max_bin=2 trains and still separates a two-sided signal (ref test_small_max_bin)."""
    rng = np.random.RandomState(24)
    X = rng.randn(2000, 3)
    y = (X[:, 0] > 0).astype(np.float64)
    # even bin counts place a quantile boundary at the median (~0 here); an odd
    # count puts boundaries at the tertiles, capping accuracy near 0.83
    for mb, floor in ((2, 0.95), (3, 0.80), (4, 0.95)):
        bst = lgb.train({"objective": "binary", "verbosity": -1, "max_bin": mb},
                        lgb.Dataset(X, label=y), 20)
        pred = bst.predict(X)
        assert ((pred > 0.5) == y).mean() > floor, mb


def test_equal_predict_from_row_major_and_col_major():
    """C-order and F-order float64 matrices produce identical datasets and
    predictions (ref test_equal_predict_from_row_major_and_col_major_data)."""
    rng = np.random.RandomState(25)
    Xc = np.ascontiguousarray(rng.randn(2000, 6))
    Xf = np.asfortranarray(Xc)
    y = Xc[:, 0] + 0.1 * rng.randn(2000)
    pc = lgb.train({"objective": "regression", "verbosity": -1},
                   lgb.Dataset(Xc, label=y), 10).predict(Xc)
    pf = lgb.train({"objective": "regression", "verbosity": -1},
                   lgb.Dataset(Xf, label=y), 10).predict(Xf)
    np.testing.assert_allclose(pc, pf, rtol=1e-12)


def test_model_size_large_roundtrip():
    """a многи-tree model string round-trips exactly (ref test_model_size)."""
    rng = np.random.RandomState(26)
    X = rng.randn(3000, 10)
    y = X @ rng.randn(10) + 0.1 * rng.randn(3000)
    bst = lgb.train({"objective": "regression", "verbosity": -1, "num_leaves": 63},
                    lgb.Dataset(X, label=y), 100)
    s = bst.model_to_string()
    assert len(s) > 100_000
    bst2 = lgb.Booster(model_str=s)
    np.testing.assert_allclose(bst2.predict(X[:200]), bst.predict(X[:200]), rtol=1e-12)
    assert bst2.num_trees() == 100


def test_sparse_train_predict_contrib_consistency():
    """CSR/CSC train + predict + contribs agree with dense (ref test_contribs_sparse)."""
    from scipy.sparse import csc_matrix, csr_matrix
    rng = np.random.RandomState(27)
    X = rng.randn(2000, 6)
    X[rng.rand(2000, 6) < 0.6] = 0.0
    y = (X[:, 0] + X[:, 1] > 0).astype(np.float64)
    p_dense = lgb.train({"objective": "binary", "verbosity": -1},
                        lgb.Dataset(X, label=y), 15).predict(X)
    for sp in (csr_matrix, csc_matrix):
        bst = lgb.train({"objective": "binary", "verbosity": -1},
                        lgb.Dataset(sp(X), label=y), 15)
        np.testing.assert_allclose(bst.predict(sp(X)), p_dense, rtol=1e-10)
        contrib = bst.predict(sp(X), pred_contrib=True)
        assert np.asarray(contrib).shape == (2000, 7)
        raw = bst.predict(sp(X), raw_score=True)
        np.testing.assert_allclose(np.asarray(contrib).sum(axis=1),
                                   np.asarray(raw).ravel(), rtol=1e-6)


def test_validate_features():
    """predict(validate_features=True) rejects renamed DataFrame columns
    (ref test_validate_features)."""
    pd = pytest.importorskip("pandas")
    rng = np.random.RandomState(28)
    df = pd.DataFrame(rng.randn(500, 3), columns=["a", "b", "c"])
    y = df["a"] > 0
    bst = lgb.train({"objective": "binary", "verbosity": -1},
                    lgb.Dataset(df, label=y.astype(float)), 5)
    bst.predict(df, validate_features=True)  # matching names pass
    bad = df.rename(columns={"b": "bad_name"})
    with pytest.raises(Exception, match="[Ff]eature|name"):
        bst.predict(bad, validate_features=True)


def test_cv_fpreproc():
    """cv(fpreproc=...) runs the per-fold hook (ref test_fpreproc)."""
    rng = np.random.RandomState(29)
    X = rng.randn(900, 4)
    y = X[:, 0] + 0.1 * rng.randn(900)
    seen = []

    def fpreproc(tr, te, params):
        seen.append(1)
        params["learning_rate"] = 0.05
        return tr, te, params

    res = lgb.cv({"objective": "regression", "metric": "l2", "verbosity": -1},
                 lgb.Dataset(X, label=y), num_boost_round=5, nfold=3,
                 fpreproc=fpreproc)
    assert len(seen) == 3
    assert len(res["valid l2-mean"]) == 5


def test_categorical_handle_exact():
    """pure-categorical signal fits exactly with per-category leaves
    (ref test_categorical_handle)."""
    n = 400
    rng = np.random.RandomState(30)
    x = rng.randint(0, 8, n).astype(np.float64)
    lut = rng.randn(8)
    y = lut[x.astype(int)]
    ds = lgb.Dataset(x.reshape(-1, 1), label=y, categorical_feature=[0])
    bst = lgb.train({"objective": "regression", "verbosity": -1, "min_data_in_leaf": 1,
                     "min_data_per_group": 1, "cat_smooth": 1e-3, "cat_l2": 0.0,
                     "learning_rate": 1.0, "num_leaves": 16}, ds, 10)
    pred = bst.predict(x.reshape(-1, 1))
    np.testing.assert_allclose(pred, y, atol=1e-3)


def test_continue_train_dart_and_multiclass():
    """init_model continuation under dart and multiclass (ref test_continue_train_*)."""
    rng = np.random.RandomState(31)
    X = rng.randn(1500, 4)
    y3 = rng.randint(0, 3, 1500)
    p = {"objective": "multiclass", "num_class": 3, "verbosity": -1}
    m1 = lgb.train(p, lgb.Dataset(X, label=y3.astype(float)), 5)
    m2 = lgb.train(p, lgb.Dataset(X, label=y3.astype(float)), 5, init_model=m1)
    assert m2.num_trees() == 30  # (5+5) iterations x 3 classes
    yd = X[:, 0] + 0.1 * rng.randn(1500)
    pd_ = {"objective": "regression", "boosting": "dart", "verbosity": -1}
    d1 = lgb.train(pd_, lgb.Dataset(X, label=yd), 5)
    d2 = lgb.train(pd_, lgb.Dataset(X, label=yd), 5, init_model=d1)
    assert d2.num_trees() == 10


def test_refit_one_tree_variants():
    """refit keeps structure, renews outputs, for 1-tree reg/binary/multiclass
    (ref test_refit_with_one_tree_*)."""
    rng = np.random.RandomState(32)
    X = rng.randn(1000, 4)
    cases = [
        ({"objective": "regression", "verbosity": -1}, X[:, 0]),
        ({"objective": "binary", "verbosity": -1}, (X[:, 0] > 0).astype(float)),
        ({"objective": "multiclass", "num_class": 3, "verbosity": -1},
         rng.randint(0, 3, 1000).astype(float)),
    ]
    for params, y in cases:
        bst = lgb.train(params, lgb.Dataset(X, label=y), 1)
        y2 = np.roll(y, 137)
        new = bst.refit(X, y2, decay_rate=0.5)
        assert new.num_trees() == bst.num_trees()
        d_old = bst.dump_model()["tree_info"][0]["tree_structure"]
        d_new = new.dump_model()["tree_info"][0]["tree_structure"]
        if "split_feature" in d_old:
            assert d_old["split_feature"] == d_new["split_feature"]
        assert not np.allclose(bst.predict(X[:50]), new.predict(X[:50]))


def test_dataset_reference_chain():
    """valid sets created from a valid set still bin against the root reference
    (ref test_reference_chain)."""
    rng = np.random.RandomState(33)
    X = rng.randn(1200, 3)
    y = X[:, 0] + 0.1 * rng.randn(1200)
    tr = lgb.Dataset(X[:600], label=y[:600])
    v1 = tr.create_valid(X[600:900], label=y[600:900])
    v2 = v1.create_valid(X[900:], label=y[900:])
    res = {}
    lgb.train({"objective": "regression", "metric": "l2", "verbosity": -1}, tr, 10,
              valid_sets=[v1, v2], valid_names=["v1", "v2"],
              callbacks=[lgb.record_evaluation(res)])
    assert "v1" in res and "v2" in res
    assert res["v1"]["l2"][-1] < res["v1"]["l2"][0]


def test_all_expected_params_written_to_model_text():
    """the saved model's parameters block reflects the training config and
    survives reload (ref test_all_expected_params_are_written_out_to_model_text)."""
    rng = np.random.RandomState(34)
    X = rng.randn(500, 3)
    bst = lgb.train({"objective": "regression", "verbosity": -1, "num_leaves": 7,
                     "learning_rate": 0.2, "lambda_l2": 0.5,
                     "bagging_fraction": 0.8, "bagging_freq": 2},
                    lgb.Dataset(X, label=X[:, 0]), 3)
    s = bst.model_to_string()
    assert "parameters:" in s
    for frag in ("[objective: regression]", "[num_leaves: 7]", "[learning_rate: 0.2]",
                 "[lambda_l2: 0.5]", "[bagging_fraction: 0.8]", "[bagging_freq: 2]",
                 "[boosting: gbdt]", "[tree_learner: serial]"):
        assert frag in s, frag
    # reloaded booster exposes the stored parameter string
    bst2 = lgb.Booster(model_str=s)
    np.testing.assert_allclose(bst2.predict(X[:20]), bst.predict(X[:20]), rtol=1e-12)


def test_multi_error_top_k():
    """multi_error@k counts a row correct if the true class is in the top k
    (ref test_multi_class_error)."""
    rng = np.random.RandomState(35)
    X = rng.randn(2000, 4)
    y = rng.randint(0, 4, 2000).astype(float)
    res1, res2 = {}, {}
    ds = lgb.Dataset(X, label=y)
    lgb.train({"objective": "multiclass", "num_class": 4, "metric": "multi_error",
               "verbosity": -1}, ds, 5, valid_sets=[lgb.Dataset(X, label=y)],
              callbacks=[lgb.record_evaluation(res1)])
    lgb.train({"objective": "multiclass", "num_class": 4, "metric": "multi_error",
               "multi_error_top_k": 2, "verbosity": -1},
              lgb.Dataset(X, label=y), 5, valid_sets=[lgb.Dataset(X, label=y)],
              callbacks=[lgb.record_evaluation(res2)])
    e1 = res1["valid_0"]["multi_error"][-1]
    e2 = res2["valid_0"]["multi_error@2"][-1]
    assert 0.0 <= e2 <= e1 <= 1.0      # top-2 error can only be lower


def test_prediction_early_stopping_multiclass():
    """pred_early_stop on multiclass prediction stays close to the exact result
    (ref test_multiclass_prediction_early_stopping)."""
    rng = np.random.RandomState(36)
    X = rng.randn(3000, 6)
    y = (X[:, 0] > 0.5).astype(int) + (X[:, 1] > 0).astype(int)
    bst = lgb.train({"objective": "multiclass", "num_class": 3, "verbosity": -1},
                    lgb.Dataset(X, label=y.astype(float)), 50)
    exact = np.argmax(bst.predict(X[:500]), axis=1)
    fast = np.argmax(bst.predict(X[:500], pred_early_stop=True,
                                 pred_early_stop_freq=5,
                                 pred_early_stop_margin=1.5), axis=1)
    assert (exact == fast).mean() > 0.95


def test_categorical_handle_na():
    """categorical column with NaN rows routes NaN to the non-bitset side
    (ref test_categorical_handle_na)."""
    x = np.array([0, 1, 2, 2, 1, 0, np.nan, 1, 2, np.nan]).reshape(-1, 1)
    lut = {0: 1.0, 1: 2.0, 2: 3.0}
    y = np.array([lut.get(v, 0.0) if not np.isnan(v) else 0.0 for v in x[:, 0]])
    ds = lgb.Dataset(x, label=y, categorical_feature=[0])
    bst = lgb.train({"objective": "regression", "verbosity": -1, "min_data_in_leaf": 1,
                     "min_data_per_group": 1, "cat_smooth": 1e-3, "cat_l2": 0.0,
                     "learning_rate": 1.0, "num_leaves": 8, "min_data_in_bin": 1,
                     "max_cat_to_onehot": 1}, ds, 10)
    pred = bst.predict(x)
    np.testing.assert_allclose(pred, y, atol=1e-2)


def test_pandas_nullable_dtypes():
    """pandas nullable Int64/Float64/boolean columns train and predict
    (ref test_pandas_nullable_dtypes)."""
    pd = pytest.importorskip("pandas")
    rng = np.random.RandomState(37)
    n = 1000
    df = pd.DataFrame({
        "a": pd.array(rng.randint(0, 10, n), dtype="Int64"),
        "b": pd.array(rng.randn(n), dtype="Float64"),
        "c": pd.array(rng.rand(n) > 0.5, dtype="boolean"),
    })
    df.loc[:20, "a"] = pd.NA
    y = (df["b"].astype(float).to_numpy() > 0).astype(float)
    bst = lgb.train({"objective": "binary", "verbosity": -1},
                    lgb.Dataset(df, label=y), 10)
    pred = bst.predict(df)
    assert ((pred > 0.5) == y).mean() > 0.9


def test_cegb_scaling_equalities():
    """cegb_penalty_split scaled with tradeoff leaves trees unchanged when both
    scale together (ref test_cegb_scaling_equalities semantics)."""
    rng = np.random.RandomState(38)
    X = rng.randn(2000, 4)
    y = X[:, 0] + 0.2 * rng.randn(2000)
    def model(tradeoff, split_pen):
        return lgb.train({"objective": "regression", "verbosity": -1,
                          "cegb_tradeoff": tradeoff, "cegb_penalty_split": split_pen},
                         lgb.Dataset(X, label=y), 10).model_to_string()
    # tradeoff*penalty identical => identical trees
    assert model(1.0, 0.5) == model(2.0, 0.25)
    # different effective penalty => different trees
    assert model(1.0, 0.5) != model(1.0, 2.0)


def test_reset_params_metric_boosting():
    """reset_parameter callback + Booster.reset_parameter change live params
    (ref test_reset_params_works_with_metric_num_class_and_boosting)."""
    rng = np.random.RandomState(39)
    X = rng.randn(1000, 4)
    y = X[:, 0] + 0.1 * rng.randn(1000)
    lrs = []
    def track(env):
        lrs.append(float(env.params.get("learning_rate", -1)))
    track.order = 20
    lgb.train({"objective": "regression", "verbosity": -1, "learning_rate": 0.1},
              lgb.Dataset(X, label=y), 5,
              callbacks=[lgb.reset_parameter(learning_rate=lambda i: 0.1 * (0.5 ** i)),
                         track])
    assert len(lrs) == 5
    assert lrs[0] > lrs[-1]


def test_predict_output_shapes():
    """prediction output shapes across tasks (ref test_predict_*_output_shape)."""
    rng = np.random.RandomState(40)
    X = rng.randn(600, 5)
    yr = X[:, 0]
    br = lgb.train({"objective": "regression", "verbosity": -1}, lgb.Dataset(X, label=yr), 4)
    assert br.predict(X).shape == (600,)
    assert br.predict(X, pred_leaf=True).shape == (600, 4)
    assert br.predict(X, pred_contrib=True).shape == (600, 6)
    yb = (X[:, 0] > 0).astype(float)
    bb = lgb.train({"objective": "binary", "verbosity": -1}, lgb.Dataset(X, label=yb), 4)
    assert bb.predict(X).shape == (600,)
    ym = rng.randint(0, 3, 600).astype(float)
    bm = lgb.train({"objective": "multiclass", "num_class": 3, "verbosity": -1},
                   lgb.Dataset(X, label=ym), 4)
    assert bm.predict(X).shape == (600, 3)
    assert bm.predict(X, pred_leaf=True).shape == (600, 12)
    assert bm.predict(X, pred_contrib=True).shape == (600, 3 * 6)


def test_default_metric_per_objective():
    """each objective evaluates with its matching default metric; metric='None'
    disables evaluation (ref test_metrics / test_default_objective_and_metric)."""
    rng = np.random.RandomState(41)
    X = rng.randn(500, 3)
    cases = [("regression", X[:, 0], "l2"), ("regression_l1", X[:, 0], "l1"),
             ("binary", (X[:, 0] > 0).astype(float), "binary_logloss"),
             ("huber", X[:, 0], "huber"), ("quantile", X[:, 0], "quantile"),
             ("mape", np.abs(X[:, 0]) + 1, "mape"),
             ("poisson", np.abs(X[:, 0]), "poisson"),
             ("gamma", np.abs(X[:, 0]) + 0.1, "gamma"),
             ("tweedie", np.abs(X[:, 0]), "tweedie"),
             ("cross_entropy", (X[:, 0] > 0).astype(float), "cross_entropy")]
    for obj, y, want in cases:
        res = {}
        lgb.train({"objective": obj, "verbosity": -1}, lgb.Dataset(X, label=y), 2,
                  valid_sets=[lgb.Dataset(X, label=y)],
                  callbacks=[lgb.record_evaluation(res)])
        assert want in res["valid_0"], (obj, list(res["valid_0"]))
    res = {}
    yb = (X[:, 0] > 0).astype(float)
    lgb.train({"objective": "binary", "metric": "None", "verbosity": -1},
              lgb.Dataset(X, label=yb), 2, valid_sets=[lgb.Dataset(X, label=yb)],
              callbacks=[lgb.record_evaluation(res)])
    assert res == {}


def test_objective_aliases_equivalent():
    """objective aliases train identically to the canonical name
    (ref test_objective_aliases)."""
    rng = np.random.RandomState(42)
    X = rng.randn(800, 3)
    y = X[:, 0] + 0.1 * rng.randn(800)
    groups = {"regression": ["mean_squared_error", "mse", "l2", "rmse"],
              "regression_l1": ["mae", "mean_absolute_error", "l1"],
              "multiclass": ["softmax"]}
    for canon, aliases in groups.items():
        yy = rng.randint(0, 3, 800).astype(float) if canon == "multiclass" else y
        extra = {"num_class": 3} if canon == "multiclass" else {}
        base = lgb.train({"objective": canon, "verbosity": -1, **extra},
                         lgb.Dataset(X, label=yy), 5).model_to_string()
        for a in aliases:
            m = lgb.train({"objective": a, "verbosity": -1, **extra},
                          lgb.Dataset(X, label=yy), 5).model_to_string()
            assert m == base, (canon, a)


def test_goss_boosting_and_strategy_equivalent():
    """boosting=goss (back-compat) == data_sample_strategy=goss
    (ref test_goss_boosting_and_strategy_equivalent)."""
    rng = np.random.RandomState(43)
    X = rng.randn(3000, 4)
    y = X[:, 0] + 0.2 * rng.randn(3000)
    a = lgb.train({"objective": "regression", "boosting": "goss", "verbosity": -1,
                   "seed": 5}, lgb.Dataset(X, label=y), 10).model_to_string()
    b = lgb.train({"objective": "regression", "boosting": "gbdt",
                   "data_sample_strategy": "goss", "verbosity": -1, "seed": 5},
                  lgb.Dataset(X, label=y), 10).model_to_string()
    assert a == b


def test_boost_from_average_with_single_leaf_trees():
    """stump-only models still boost from the label average
    (ref test_boost_from_average_with_single_leaf_trees)."""
    rng = np.random.RandomState(44)
    X = rng.randn(500, 2)
    y = np.full(500, 7.5) + 0.001 * rng.randn(500)
    bst = lgb.train({"objective": "regression", "verbosity": -1, "min_data_in_leaf": 400},
                    lgb.Dataset(X, label=y), 5)
    np.testing.assert_allclose(bst.predict(X[:10]), 7.5, rtol=1e-3)


def test_mape_with_bagging():
    """mape objective under rf/bagging keeps positive predictions
    (ref test_mape_for_specific_boosting_types)."""
    rng = np.random.RandomState(45)
    X = rng.randn(3000, 4)
    y = np.abs(X[:, 0]) + 1.0 + 0.05 * rng.randn(3000)
    bst = lgb.train({"objective": "mape", "verbosity": -1, "bagging_fraction": 0.8,
                     "bagging_freq": 1, "seed": 3}, lgb.Dataset(X, label=y), 30)
    pred = bst.predict(X)
    assert (pred > 0).all()
    assert float(np.mean(np.abs(pred - y) / y)) < 0.3


def test_forced_split_feature_indices(tmp_path):
    """forcedsplits_filename: the forced feature is the root split of every tree
    (ref test_forced_split_feature_indices semantics)."""
    import json as _json
    rng = np.random.RandomState(46)
    X = rng.randn(2000, 4)
    y = X[:, 0] + X[:, 3] + 0.1 * rng.randn(2000)
    fs = tmp_path / "forced.json"
    fs.write_text(_json.dumps({"feature": 3, "threshold": 0.0}))
    bst = lgb.train({"objective": "regression", "verbosity": -1,
                     "forcedsplits_filename": str(fs)}, lgb.Dataset(X, label=y), 5)
    d = bst.dump_model()
    for t in d["tree_info"]:
        root = t["tree_structure"]
        assert root["split_feature"] == 3
        assert root["threshold"] == pytest.approx(0.0, abs=0.2)


def test_parameters_loaded_from_model_file(tmp_path):
    """Booster(model_file) exposes the saved parameters block via .params
    (ref test_parameters_are_loaded_from_model_file)."""
    rng = np.random.RandomState(47)
    X = rng.randn(400, 3)
    bst = lgb.train({"objective": "regression", "verbosity": -1, "num_leaves": 9,
                     "lambda_l1": 0.4, "bagging_freq": 2, "bagging_fraction": 0.9},
                    lgb.Dataset(X, label=X[:, 0]), 3)
    f = tmp_path / "m.txt"
    bst.save_model(str(f))
    b2 = lgb.Booster(model_file=str(f))
    assert b2.params["objective"] == "regression"
    assert b2.params["num_leaves"] == 9
    assert b2.params["lambda_l1"] == 0.4
    assert b2.params["bagging_freq"] == 2
    b3 = lgb.Booster(model_str=bst.model_to_string())
    assert b3.params["num_leaves"] == 9


def test_train_cv_informative_errors():
    """train/cv raise informative errors for bad inputs
    (ref test_train_and_cv_raise_informative_error_*)."""
    rng = np.random.RandomState(48)
    X = rng.randn(100, 3)
    ds = lgb.Dataset(X, label=X[:, 0])
    with pytest.raises(ValueError, match="num_boost_round"):
        lgb.train({"verbosity": -1}, ds, num_boost_round=0)
    with pytest.raises(ValueError, match="num_boost_round"):
        lgb.cv({"verbosity": -1}, ds, num_boost_round=-5)
    with pytest.raises(TypeError, match="Dataset"):
        lgb.train({"verbosity": -1}, np.zeros((5, 2)), 5)
    with pytest.raises(TypeError, match="Dataset"):
        lgb.cv({"verbosity": -1}, [1, 2, 3], 5)


def test_is_unbalance_and_scale_pos_weight():
    """is_unbalance / scale_pos_weight raise minority-class recall
    (ref binary objective weighting behavior)."""
    rng = np.random.RandomState(49)
    X = rng.randn(5000, 4)
    y = (X[:, 0] + 0.8 * rng.randn(5000) > 1.3).astype(float)  # ~10% positives
    base = lgb.train({"objective": "binary", "verbosity": -1},
                     lgb.Dataset(X, label=y), 20)
    unb = lgb.train({"objective": "binary", "is_unbalance": True, "verbosity": -1},
                    lgb.Dataset(X, label=y), 20)
    spw = lgb.train({"objective": "binary", "scale_pos_weight": 8.0, "verbosity": -1},
                    lgb.Dataset(X, label=y), 20)
    def recall(b):
        return float(((b.predict(X) > 0.5) & (y == 1)).sum() / max((y == 1).sum(), 1))
    assert recall(unb) > recall(base)
    assert recall(spw) > recall(base)
    # mean predicted probability rises with positive upweighting
    assert unb.predict(X).mean() > base.predict(X).mean()


def test_pandas_sparse_dtype():
    """pandas SparseArray columns train and predict (ref test_pandas_sparse)."""
    pd = pytest.importorskip("pandas")
    rng = np.random.RandomState(50)
    df = pd.DataFrame({
        "a": pd.arrays.SparseArray(np.where(rng.rand(800) < 0.8, 0.0, rng.randn(800))),
        "b": rng.randn(800)})
    y = (df["b"] > 0).astype(float)
    bst = lgb.train({"objective": "binary", "verbosity": -1}, lgb.Dataset(df, label=y), 5)
    assert ((bst.predict(df) > 0.5) == y).mean() > 0.95


def test_categorical_non_zero_inputs():
    """categories need not start at 0 nor be contiguous
    (ref test_categorical_non_zero_inputs)."""
    rng = np.random.RandomState(51)
    x = rng.choice([5, 7, 11, 400], 500).astype(float).reshape(-1, 1)
    lut = {5: 1.0, 7: -2.0, 11: 3.0, 400: 0.5}
    y = np.array([lut[int(v)] for v in x[:, 0]])
    bst = lgb.train({"objective": "regression", "verbosity": -1, "min_data_in_leaf": 1,
                     "learning_rate": 1.0, "cat_l2": 0.0, "cat_smooth": 1e-3},
                    lgb.Dataset(x, label=y, categorical_feature=[0]), 10)
    np.testing.assert_allclose(bst.predict(x), y, atol=1e-3)


def test_register_logger_captures_native_logs():
    """register_logger redirects native Warning/Info lines into a Python logger
    (ref test_utilities.py test_register_logger)."""
    import logging
    records = []
    logger = logging.getLogger("migbm_capture_test")
    logger.setLevel(logging.DEBUG)

    class _H(logging.Handler):
        def emit(self, r):
            records.append(r.getMessage())
    h = _H()
    logger.addHandler(h)
    try:
        lgb.register_logger(logger)
        X = np.zeros((100, 2))  # all-constant features emit a Warning
        # verbosity on the DATASET too: the warning fires at construction time
        # and the log level must already allow warnings then
        lgb.train({"objective": "regression", "verbosity": 0},
                  lgb.Dataset(X, label=np.zeros(100), params={"verbosity": 0}), 1)
        assert any("trivial" in m for m in records), records
    finally:
        logger.removeHandler(h)
        # restore native stderr logging for the rest of the suite
        import ctypes
        from lightgbm_amd.basic import _LIB
        _LIB.LGBM_RegisterLogCallback(ctypes.cast(None, ctypes.CFUNCTYPE(None, ctypes.c_char_p)))


def test_degenerate_inputs_error_gracefully():
    """degenerate inputs either train or raise informative LightGBMError —
    never crash (reference error-contract behaviors)."""
    rng = np.random.RandomState(52)
    # these succeed
    lgb.train({"objective": "regression", "verbosity": -1, "min_data_in_leaf": 1,
               "min_data_in_bin": 1}, lgb.Dataset(np.ones((1, 2)), label=np.ones(1)), 2)
    lgb.train({"objective": "regression", "verbosity": -1},
              lgb.Dataset(rng.randn(100, 1), label=rng.rand(100)), 2)
    b = lgb.train({"objective": "regression", "verbosity": -1},
                  lgb.Dataset(rng.randn(100, 2), label=rng.rand(100)), 2)
    assert b.predict(np.zeros((0, 2))).shape == (0,)
    # these raise with informative messages
    from lightgbm_amd.basic import LightGBMError
    with pytest.raises(LightGBMError, match="non-negative"):
        lgb.train({"objective": "poisson", "verbosity": -1},
                  lgb.Dataset(rng.randn(100, 2), label=-np.ones(100)), 2)
    with pytest.raises(LightGBMError, match="Length of label"):
        lgb.Dataset(rng.randn(100, 2), label=np.zeros(50)).construct()
    with pytest.raises(LightGBMError, match="finite"):
        lgb.train({"objective": "regression", "verbosity": -1},
                  lgb.Dataset(rng.randn(100, 2), label=np.full(100, np.inf)), 2)
    with pytest.raises(LightGBMError, match="out of range"):
        lgb.train({"objective": "multiclass", "num_class": 3, "verbosity": -1},
                  lgb.Dataset(rng.randn(100, 2), label=np.full(100, 7.0)), 2)


def test_rank_xendcg_quality():
    """rank_xendcg reaches lambdarank-class NDCG on a clean ranking signal
    (ref test_xendcg)."""
    rng = np.random.RandomState(53)
    qsizes = [20] * 120
    n = sum(qsizes)
    X = rng.rand(n, 6)
    rel = np.clip((3 * X[:, 0] + 0.5 * rng.randn(n)).astype(int), 0, 3)
    res = {}
    ds = lgb.Dataset(X, label=rel.astype(float), group=qsizes)
    vs = lgb.Dataset(X, label=rel.astype(float), group=qsizes, reference=ds) \
        if hasattr(lgb.Dataset, "reference") else ds.create_valid(X, label=rel.astype(float), group=qsizes)
    lgb.train({"objective": "rank_xendcg", "metric": "ndcg", "eval_at": [5],
               "verbosity": -1, "objective_seed": 7}, ds, 40,
              valid_sets=[vs], callbacks=[lgb.record_evaluation(res)])
    ndcg = res["valid_0"]["ndcg@5"][-1]
    assert ndcg > 0.85, ndcg


def test_multiclass_rf():
    """random forest boosting with multiclass softmax (ref test_multiclass_rf)."""
    rng = np.random.RandomState(54)
    X = rng.randn(4000, 5)
    y = (X[:, 0] > 0.5).astype(int) + (X[:, 1] > 0).astype(int)
    bst = lgb.train({"objective": "multiclass", "num_class": 3, "boosting": "rf",
                     "bagging_fraction": 0.7, "bagging_freq": 1, "verbosity": -1,
                     "num_leaves": 31}, lgb.Dataset(X, label=y.astype(float)), 30)
    pred = bst.predict(X)
    assert pred.shape == (4000, 3)
    np.testing.assert_allclose(pred.sum(axis=1), 1.0, rtol=1e-6)
    assert (np.argmax(pred, axis=1) == y).mean() > 0.8


def test_cv_with_init_model_and_callable_objective():
    """cv(init_model=...) continues every fold from the model; a callable
    objective in params drives cv like train (ref test_cv_works_with_init_model
    / test_objective_callable_cv_*)."""
    rng = np.random.RandomState(55)
    X = rng.randn(900, 4)
    y = X[:, 0] + 0.1 * rng.randn(900)
    ds = lgb.Dataset(X, label=y, free_raw_data=False)
    base = lgb.train({"objective": "regression", "verbosity": -1}, ds, 5)
    res = lgb.cv({"objective": "regression", "metric": "l2", "verbosity": -1},
                 ds, num_boost_round=5, nfold=3, init_model=base,
                 return_cvbooster=True)
    for b in res["cvbooster"].boosters:
        assert b.num_trees() == 10  # 5 merged + 5 trained
    # continued folds start from the base model's error level, not from zero
    assert res["valid l2-mean"][0] < float(np.var(y))

    def l2_obj(preds, dataset):
        g = (preds - dataset.get_label()).astype(np.float32)
        return g, np.ones_like(g)
    yb = (X[:, 0] > 0).astype(float)
    res2 = lgb.cv({"objective": l2_obj, "metric": "l2", "verbosity": -1},
                  lgb.Dataset(X, label=y), num_boost_round=5, nfold=3)
    assert len(res2["valid l2-mean"]) == 5
    assert res2["valid l2-mean"][-1] < res2["valid l2-mean"][0]


def test_continue_training_equals_straight_run():
    """init_model continuation is numerically identical to uninterrupted
    training: merged trees' outputs are folded into the training scores
    (was silently restarting gradients from zero)."""
    rng = np.random.RandomState(56)
    X = rng.randn(3000, 4)
    cases = [
        ({"objective": "regression", "learning_rate": 0.2}, X[:, 0] + 0.1 * rng.randn(3000)),
        ({"objective": "binary", "learning_rate": 0.2}, (X[:, 0] > 0).astype(float)),
        ({"objective": "multiclass", "num_class": 3, "learning_rate": 0.2},
         rng.randint(0, 3, 3000).astype(float)),
    ]
    for extra, y in cases:
        p = {**extra, "verbosity": -1}
        straight = lgb.train(p, lgb.Dataset(X, label=y), 10)
        half = lgb.train(p, lgb.Dataset(X, label=y), 5)
        cont = lgb.train(p, lgb.Dataset(X, label=y), 5, init_model=half)
        np.testing.assert_allclose(cont.predict(X[:500]), straight.predict(X[:500]),
                                   rtol=1e-9, err_msg=str(extra))


def test_feature_fraction_bynode_cpu():
    """per-node column sampling changes trees but keeps quality
    (ref test_node_level_subcol)."""
    rng = np.random.RandomState(57)
    X = rng.randn(4000, 8)
    y = X[:, 0] + X[:, 1] + 0.1 * rng.randn(4000)
    full = lgb.train({"objective": "regression", "verbosity": -1, "seed": 1},
                     lgb.Dataset(X, label=y), 30)
    sub = lgb.train({"objective": "regression", "verbosity": -1, "seed": 1,
                     "feature_fraction_bynode": 0.5}, lgb.Dataset(X, label=y), 30)
    assert sub.model_to_string() != full.model_to_string()
    mse_sub = float(np.mean((sub.predict(X) - y) ** 2))
    mse_full = float(np.mean((full.predict(X) - y) ** 2))
    assert mse_sub < mse_full * 3.0
    # reproducible under the same seed
    sub2 = lgb.train({"objective": "regression", "verbosity": -1, "seed": 1,
                      "feature_fraction_bynode": 0.5}, lgb.Dataset(X, label=y), 30)
    assert sub2.model_to_string() == sub.model_to_string()


def test_sample_strategy_with_boosting_combos():
    """GOSS composes with dart and with plain gbdt; bagging composes with dart
    (ref test_sample_strategy_with_boosting)."""
    rng = np.random.RandomState(58)
    X = rng.randn(3000, 5)
    y = X[:, 0] + 0.2 * rng.randn(3000)
    combos = [
        {"boosting": "dart", "data_sample_strategy": "goss"},
        {"boosting": "gbdt", "data_sample_strategy": "goss"},
        {"boosting": "dart", "bagging_fraction": 0.7, "bagging_freq": 1},
    ]
    for extra in combos:
        bst = lgb.train({"objective": "regression", "verbosity": -1, "seed": 2, **extra},
                        lgb.Dataset(X, label=y), 20)
        mse = float(np.mean((bst.predict(X) - y) ** 2))
        assert mse < float(np.var(y)) * 0.5, (extra, mse)


def test_early_stopping_via_global_params_with_min_delta():
    """early_stopping_round + early_stopping_min_delta in params drive stopping
    without an explicit callback (ref test_early_stopping_via_global_params /
    _min_delta_via_global_params)."""
    rng = np.random.RandomState(59)
    X = rng.randn(4000, 5)
    y = X[:, 0] + 0.3 * rng.randn(4000)
    tr = lgb.Dataset(X[:3000], label=y[:3000])
    va = tr.create_valid(X[3000:], label=y[3000:])
    res = {}
    bst = lgb.train({"objective": "regression", "metric": "l2", "verbosity": -1,
                     "early_stopping_round": 5}, tr, 300, valid_sets=[va],
                    callbacks=[lgb.record_evaluation(res)])
    assert 0 < bst.best_iteration < 300
    # a large min_delta stops much earlier: tiny improvements no longer count
    res2 = {}
    bst2 = lgb.train({"objective": "regression", "metric": "l2", "verbosity": -1,
                      "early_stopping_round": 5, "early_stopping_min_delta": 0.05},
                     tr, 300, valid_sets=[va], callbacks=[lgb.record_evaluation(res2)])
    assert bst2.best_iteration <= bst.best_iteration
    assert len(res2["valid_0"]["l2"]) < len(res["valid_0"]["l2"])
