"""GPU (MI355X) tests: HIP learner parity vs the CPU oracle. All marked gpu."""
import numpy as np
import pytest

import lightgbm_amd as lgb

pytestmark = pytest.mark.gpu


def _binary_data(n=200_000, d=28, seed=3):
    rng = np.random.RandomState(seed)
    X = rng.randn(n, d).astype(np.float32)
    logit = 1.2 * X[:, 0] - 0.8 * X[:, 1] + 0.9 * X[:, 2] * X[:, 3] + 0.5 * X[:, 4]
    y = (logit + 1.0 * rng.randn(n) > 0).astype(np.float32)
    return X, y


def _auc(y, p):
    order = np.argsort(-p, kind="stable")
    ys = y[order]
    n_pos = ys.sum()
    n_neg = len(ys) - n_pos
    ranks = np.arange(1, len(ys) + 1)
    return 1.0 - (ranks[ys > 0].sum() - n_pos * (n_pos + 1) / 2) / (n_pos * n_neg)


def test_gpu_trains_binary():
    X, y = _binary_data()
    params = {"objective": "binary", "device_type": "gpu", "max_bin": 63,
              "num_leaves": 63, "min_data_in_leaf": 1, "min_sum_hessian_in_leaf": 100,
              "verbosity": 0}
    bst = lgb.train(params, lgb.Dataset(X, label=y), 20)
    assert bst.num_trees() == 20
    pred = bst.predict(X[:20000])
    assert np.all(np.isfinite(pred))
    assert _auc(y[:20000], pred) > 0.75


def test_gpu_cpu_parity_auc():
    """CPU vs HIP learner must land at near-identical quality (ref test_dual.py)."""
    X, y = _binary_data(n=100_000)
    Xv, yv = _binary_data(n=50_000, seed=77)
    aucs = {}
    preds = {}
    for dev in ("cpu", "gpu"):
        params = {"objective": "binary", "device_type": dev, "max_bin": 63,
                  "num_leaves": 63, "min_data_in_leaf": 1,
                  "min_sum_hessian_in_leaf": 100, "verbosity": 0}
        bst = lgb.train(params, lgb.Dataset(X, label=y), 30)
        p = bst.predict(Xv)
        aucs[dev] = _auc(yv, p)
        preds[dev] = p
    # same algorithm, fp32-histogram device vs fp64 host: quality parity
    assert abs(aucs["cpu"] - aucs["gpu"]) < 2e-3, aucs
    # and per-row probabilities should be close on average
    assert np.mean(np.abs(preds["cpu"] - preds["gpu"])) < 0.02


def test_gpu_regression():
    rng = np.random.RandomState(0)
    X = rng.randn(100_000, 10).astype(np.float32)
    y = (3 * X[:, 0] + np.sin(X[:, 1]) + 0.1 * rng.randn(100_000)).astype(np.float32)
    params = {"objective": "regression", "device_type": "gpu", "verbosity": 0,
              "num_leaves": 63}
    bst = lgb.train(params, lgb.Dataset(X, label=y), 30)
    pred = bst.predict(X[:10000])
    mse = float(np.mean((pred - y[:10000]) ** 2))
    assert mse < 0.15 * float(np.var(y))


def test_gpu_model_text_roundtrip(tmp_path):
    X, y = _binary_data(n=50_000)
    params = {"objective": "binary", "device_type": "gpu", "verbosity": 0,
              "num_leaves": 31}
    bst = lgb.train(params, lgb.Dataset(X, label=y), 5)
    f = tmp_path / "gpu_model.txt"
    bst.save_model(str(f))
    bst2 = lgb.Booster(model_file=str(f))
    np.testing.assert_allclose(bst.predict(X[:1000]), bst2.predict(X[:1000]), rtol=1e-12)


def test_gpu_bagging():
    X, y = _binary_data(n=100_000)
    params = {"objective": "binary", "device_type": "gpu", "verbosity": 0,
              "num_leaves": 63, "bagging_freq": 1, "bagging_fraction": 0.5}
    bst = lgb.train(params, lgb.Dataset(X, label=y), 10)
    pred = bst.predict(X[:10000])
    assert _auc(y[:10000], pred) > 0.7


def test_gpu_weights():
    X, y = _binary_data(n=50_000)
    w = np.where(y > 0, 5.0, 1.0).astype(np.float32)
    params = {"objective": "binary", "device_type": "gpu", "verbosity": 0,
              "num_leaves": 31}
    bst = lgb.train(params, lgb.Dataset(X, label=y, weight=w), 10)
    assert bst.predict(X[:5000]).mean() > y.mean()


def test_gpu_quantized_grad():
    """use_quantized_grad: packed-int histograms must stay at quality parity."""
    X, y = _binary_data(n=150_000)
    Xv, yv = _binary_data(n=50_000, seed=55)
    aucs = {}
    for q in (False, True):
        params = {"objective": "binary", "device_type": "gpu", "max_bin": 63,
                  "num_leaves": 63, "min_data_in_leaf": 1,
                  "min_sum_hessian_in_leaf": 100, "verbosity": 0,
                  "use_quantized_grad": q}
        bst = lgb.train(params, lgb.Dataset(X, label=y), 30)
        aucs[q] = _auc(yv, bst.predict(Xv))
    assert abs(aucs[True] - aucs[False]) < 5e-3, aucs


def test_gpu_lambdarank():
    """Device lambdarank gradients: quality parity with the CPU objective."""
    rng = np.random.RandomState(9)
    groups = rng.randint(20, 120, size=400)
    n = int(groups.sum())
    X = rng.randn(n, 20).astype(np.float32)
    rel = np.clip((X[:, 0] * 1.5 + 0.5 * rng.randn(n) + 1.2), 0, 4).astype(int)
    y = rel.astype(np.float32)
    ndcgs = {}
    for dev in ("cpu", "gpu"):
        params = {"objective": "lambdarank", "metric": "ndcg", "eval_at": [10],
                  "device_type": dev, "num_leaves": 31, "verbosity": 0}
        ev = {}
        train = lgb.Dataset(X, label=y, group=groups.astype(np.int32))
        lgb.train(params, train, 30, valid_sets=[train], valid_names=["t"],
                  callbacks=[lgb.record_evaluation(ev)])
        ndcgs[dev] = ev["t"]["ndcg@10"][-1]
    assert ndcgs["gpu"] > 0.85, ndcgs
    assert abs(ndcgs["gpu"] - ndcgs["cpu"]) < 0.02, ndcgs


def test_gpu_rccl_world1_comm():
    """End-to-end RCCL path on one GPU: unique-id -> ncclCommInitRank(world=1) ->
    device-resident training with the comm ACTIVE (every ncclAllReduce in the split
    loop actually executes) -> identical quality to comm-free training. De-risks the
    multi-GPU scaling run, which only differs by world size."""
    import ctypes
    from lightgbm_amd.basic import _LIB
    buf = ctypes.create_string_buffer(256)
    size = ctypes.c_int(0)
    assert _LIB.LGBM_GPUGetUniqueId(buf, ctypes.byref(size)) == 0
    assert size.value > 0
    assert _LIB.LGBM_GPUNetworkInit(ctypes.c_int(1), ctypes.c_int(0),
                                    bytes(buf.raw[:size.value])) == 0
    try:
        rng = np.random.RandomState(0)
        X = rng.rand(60000, 10)
        y = (X[:, 0] + X[:, 1] > 1.0).astype(np.float32)
        params = {"objective": "binary", "device_type": "cuda", "num_leaves": 63,
                  "tree_learner": "data", "verbosity": -1}
        bst = lgb.train(params, lgb.Dataset(X, label=y), 15)
        from sklearn.metrics import roc_auc_score
        assert roc_auc_score(y, bst.predict(X)) > 0.9
    finally:
        assert _LIB.LGBM_GPUNetworkFree() == 0


def test_gpu_monotone_constraints():
    """monotone constraints run IN the device split loop (per-leaf bound array +
    midpoint propagation in k_finalize); predictions must be globally monotone."""
    rng = np.random.RandomState(0)
    X = rng.rand(20000, 3)
    y = (2 * X[:, 0] + 0.1 * rng.randn(20000)).astype(np.float32)
    bst = lgb.train({"objective": "regression", "device_type": "cuda",
                     "monotone_constraints": [1, 0, 0], "num_leaves": 63,
                     "verbosity": -1}, lgb.Dataset(X, label=y), 40)
    xs = np.linspace(0.02, 0.98, 30)
    for other in (0.1, 0.5, 0.9):
        grid = np.column_stack([xs, np.full(30, other), np.full(30, other)])
        assert np.all(np.diff(bst.predict(grid)) >= -1e-9)
    # and the fallback path still exists for a genuinely unimplemented feature
    bst2 = lgb.train({"objective": "regression", "device_type": "cuda",
                      "linear_tree": True, "verbosity": -1},
                     lgb.Dataset(X, label=y), 5)
    assert np.isfinite(bst2.predict(X[:10])).all()


def test_gpu_goss():
    """GOSS on GPU: gradients computed on host (device boosting disabled for
    gradient-dependent sampling), amplified correctly, trained on device."""
    rng = np.random.RandomState(0)
    X = rng.randn(60000, 10)
    y = (2 * X[:, 0] - X[:, 1] > 0).astype(np.float32)
    bst = lgb.train({"objective": "binary", "device_type": "cuda",
                     "data_sample_strategy": "goss", "verbosity": -1},
                    lgb.Dataset(X, label=y), 25)
    from sklearn.metrics import roc_auc_score
    assert roc_auc_score(y, bst.predict(X)) > 0.95


def test_gpu_dart():
    """DART on GPU: device scores synced around the host-side drop/renormalize."""
    rng = np.random.RandomState(0)
    X = rng.randn(50000, 8)
    y = (X[:, 0] + 0.5 * X[:, 1] > 0).astype(np.float32)
    bst = lgb.train({"objective": "binary", "device_type": "cuda", "boosting": "dart",
                     "drop_rate": 0.3, "verbosity": -1}, lgb.Dataset(X, label=y), 25)
    from sklearn.metrics import roc_auc_score
    auc = roc_auc_score(y, bst.predict(X))
    assert auc > 0.95
    # cross-check against CPU DART quality (same config)
    cpu = lgb.train({"objective": "binary", "boosting": "dart", "drop_rate": 0.3,
                     "verbosity": -1}, lgb.Dataset(X, label=y), 25)
    assert abs(auc - roc_auc_score(y, cpu.predict(X))) < 0.02


def test_gpu_extra_trees_and_bynode():
    """extra_trees + feature_fraction_bynode run in the device loop via the
    sync-free hash sampler; models train to reasonable quality and differ from
    the unsampled model."""
    rng = np.random.RandomState(0)
    X = rng.randn(60000, 10)
    y = (2 * X[:, 0] - X[:, 1] + 0.3 * rng.randn(60000) > 0).astype(np.float32)
    from sklearn.metrics import roc_auc_score
    base = lgb.train({"objective": "binary", "device_type": "cuda", "verbosity": -1},
                     lgb.Dataset(X, label=y), 25)
    et = lgb.train({"objective": "binary", "device_type": "cuda", "extra_trees": True,
                    "verbosity": -1}, lgb.Dataset(X, label=y), 25)
    bn = lgb.train({"objective": "binary", "device_type": "cuda",
                    "feature_fraction_bynode": 0.5, "verbosity": -1},
                   lgb.Dataset(X, label=y), 25)
    for m in (et, bn):
        assert roc_auc_score(y, m.predict(X)) > 0.9
        assert not np.allclose(m.predict(X[:100]), base.predict(X[:100]))


def test_gpu_categorical_sorted_subset():
    """Categorical features with 4 < cats <= 64 use the device wave64
    sorted-subset scan; quality must match the CPU learner's subset splits."""
    rng = np.random.RandomState(0)
    n = 60000
    cat = rng.randint(0, 24, n).astype(float)
    X = np.column_stack([cat, rng.randn(n)])
    eff = rng.randn(24) * 1.5
    y = (eff[cat.astype(int)] + 0.4 * rng.randn(n) > 0).astype(np.float32)
    from sklearn.metrics import roc_auc_score
    p = {"objective": "binary", "categorical_feature": [0], "verbosity": -1,
         "num_leaves": 31}
    gpu = lgb.train({**p, "device_type": "cuda"}, lgb.Dataset(X, label=y), 25)
    cpu = lgb.train(p, lgb.Dataset(X, label=y), 25)
    auc_gpu = roc_auc_score(y, gpu.predict(X))
    auc_cpu = roc_auc_score(y, cpu.predict(X))
    assert auc_gpu > 0.85
    assert abs(auc_gpu - auc_cpu) < 0.02
    # the model must actually contain multi-category subset splits
    d = gpu.dump_model()

    def has_subset(node):
        if "leaf_index" in node:
            return False
        if node.get("decision_type") == "==" and \
                len(str(node.get("threshold", "")).split("||")) > 1:
            return True
        return has_subset(node["left_child"]) or has_subset(node["right_child"])
    assert any(has_subset(t["tree_structure"]) for t in d["tree_info"])


def test_gpu_multiclass():
    """Multiclass on GPU: per-class device score buffers; host softmax gradients,
    device histograms/partition per class tree."""
    rng = np.random.RandomState(0)
    X = rng.randn(60000, 8)
    y = ((X[:, 0] > 0.5).astype(int) + (X[:, 1] > 0).astype(int)).astype(np.float32)
    bst = lgb.train({"objective": "multiclass", "num_class": 3, "device_type": "cuda",
                     "verbosity": -1}, lgb.Dataset(X, label=y), 25)
    pred = bst.predict(X)
    assert pred.shape == (60000, 3)
    np.testing.assert_allclose(pred.sum(axis=1), 1.0, rtol=1e-6)
    acc = (pred.argmax(axis=1) == y).mean()
    assert acc > 0.85


def test_gpu_interaction_constraints():
    """Interaction constraints enforced inside the device loop via per-leaf
    branch-feature bitmasks: no tree may mix features across groups."""
    rng = np.random.RandomState(0)
    X = rng.randn(40000, 4)
    y = (X[:, 0] * X[:, 1] + X[:, 2] * X[:, 3] > 0).astype(np.float32)
    bst = lgb.train({"objective": "binary", "device_type": "cuda", "num_leaves": 31,
                     "interaction_constraints": "[0,1],[2,3]", "verbosity": -1},
                    lgb.Dataset(X, label=y), 15)
    d = bst.dump_model()

    def feats(node, acc):
        if "leaf_index" in node:
            return
        acc.add(node["split_feature"])
        feats(node["left_child"], acc)
        feats(node["right_child"], acc)
    for t in d["tree_info"]:
        used = set()
        feats(t["tree_structure"], used)
        assert used <= {0, 1} or used <= {2, 3}, used


def test_gpu_path_smooth():
    """path_smooth runs in the device gain scan (LeafStat.parent_out chain);
    smoothed models differ from unsmoothed and keep quality."""
    rng = np.random.RandomState(0)
    X = rng.randn(50000, 8)
    y = (X[:, 0] + 0.5 * X[:, 1] + 0.3 * rng.randn(50000)).astype(np.float32)
    base = lgb.train({"objective": "regression", "device_type": "cuda",
                      "verbosity": -1}, lgb.Dataset(X, label=y), 20)
    sm = lgb.train({"objective": "regression", "device_type": "cuda",
                    "path_smooth": 10.0, "verbosity": -1}, lgb.Dataset(X, label=y), 20)
    assert not np.allclose(base.predict(X[:100]), sm.predict(X[:100]))
    r2 = 1 - np.mean((sm.predict(X) - y) ** 2) / np.var(y)
    assert r2 > 0.8


def _run_dist_gpu(tmp_path, mode, world=2, extra=()):
    """torchrun `world` ranks on GPU 0 through the device data-parallel learner."""
    import subprocess, sys, os, json, random
    from pathlib import Path
    out = tmp_path / mode
    out.mkdir(parents=True, exist_ok=True)
    env = dict(os.environ)
    env["MIGBM_DIST_HIST"] = mode
    env["HIP_VISIBLE_DEVICES"] = env.get("HIP_VISIBLE_DEVICES", "0").split(",")[0]
    port = random.randint(20000, 59000)
    worker = Path(__file__).parent / "helpers" / "gpu_dist_worker.py"
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           f"--nproc-per-node={world}", "--master-addr", "127.0.0.1",
           "--master-port", str(port), str(worker), "--out-dir", str(out)] + list(extra)
    r = subprocess.run(cmd, env=env, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, f"worker failed:\n{r.stdout[-3000:]}\n{r.stderr[-3000:]}"
    models = [(out / f"model_rank{i}.txt").read_text() for i in range(world)]
    res = json.loads((out / "result.json").read_text())
    return models, res


def test_gpu_dist_world2_allreduce(tmp_path):
    """Two ranks share GPU 0 (host-seam transport, identical device code path to
    RCCL): full-histogram allreduce mode. Every rank must build the byte-identical
    model, and quality must match a single-process run."""
    models, res = _run_dist_gpu(tmp_path, "allreduce")
    assert models[0] == models[1]
    assert res["num_trees"] == 10
    assert res["auc_full"] > 0.80


def test_gpu_dist_world2_reduce_scatter(tmp_path):
    """Reduce-scatter mode: per-rank feature-block ownership, owner-only gain
    scan, winner allgather + device argmax. Models identical across ranks and
    quality parity with the allreduce mode."""
    m_rs, res_rs = _run_dist_gpu(tmp_path, "reduce_scatter")
    assert m_rs[0] == m_rs[1]
    assert res_rs["auc_full"] > 0.80
    # ownership must not change WHAT is learned, only who scans what: compare
    # against the allreduce mode at identical seeds/config
    m_ar, res_ar = _run_dist_gpu(tmp_path, "allreduce")
    assert abs(res_rs["auc_full"] - res_ar["auc_full"]) < 2e-3


def test_gpu_dist_world3_reduce_scatter(tmp_path):
    """Odd world size exercises unbalanced feature-block ownership."""
    models, res = _run_dist_gpu(tmp_path, "reduce_scatter", world=3)
    assert models[0] == models[1] == models[2]
    assert res["auc_full"] > 0.80


@pytest.mark.parametrize("objective,make_label,pred_tol", [
    ("regression_l1", "reg", 0.05),
    ("huber", "reg", 0.05),
    ("fair", "reg", 0.05),
    ("quantile", "reg", 0.08),
    ("mape", "pos", 0.10),
    ("poisson", "pos", 0.10),
    ("gamma", "pos", 0.10),
    ("tweedie", "pos", 0.10),
    ("cross_entropy", "prob", 0.03),
    ("cross_entropy_lambda", "prob", 0.05),
])
def test_gpu_device_objective_parity(objective, make_label, pred_tol):
    """Every device gradient kernel must track the CPU oracle of the same
    objective (device objective breadth). Compares predictions
    on held-out rows between device_type=cpu and gpu at identical config."""
    rng = np.random.RandomState(5)
    n, d = 60_000, 10
    X = rng.randn(n, d).astype(np.float32)
    base = 1.5 * X[:, 0] + np.sin(X[:, 1]) + 0.5 * X[:, 2] * X[:, 3]
    if make_label == "reg":
        y = (base + 0.2 * rng.randn(n)).astype(np.float32)
    elif make_label == "pos":
        y = np.exp(0.4 * base + 0.1 * rng.randn(n)).astype(np.float32)
    else:  # prob
        y = (1.0 / (1.0 + np.exp(-base))).astype(np.float32)
    preds = {}
    for dev in ("cpu", "gpu"):
        params = {"objective": objective, "device_type": dev, "max_bin": 63,
                  "num_leaves": 31, "min_data_in_leaf": 20, "verbosity": 0,
                  "metric": "none"}
        bst = lgb.train(params, lgb.Dataset(X, label=y), 25)
        preds[dev] = bst.predict(X[:10000])
    scale = max(1e-3, float(np.abs(preds["cpu"]).mean()))
    rel = np.abs(preds["cpu"] - preds["gpu"]).mean() / scale
    assert rel < pred_tol, (objective, rel)


def test_gpu_multiclass_device_objective():
    """multiclass softmax gradients now run on device (all classes at once);
    quality parity with the CPU learner."""
    rng = np.random.RandomState(0)
    n, d, k = 60_000, 8, 4
    X = rng.randn(n, d).astype(np.float32)
    logits = np.stack([X[:, i] + 0.5 * X[:, (i + 1) % d] for i in range(k)], axis=1)
    y = np.argmax(logits + 0.5 * rng.randn(n, k), axis=1).astype(np.float32)
    accs = {}
    for dev in ("cpu", "gpu"):
        params = {"objective": "multiclass", "num_class": k, "device_type": dev,
                  "max_bin": 63, "num_leaves": 31, "verbosity": 0, "metric": "none"}
        bst = lgb.train(params, lgb.Dataset(X, label=y), 15)
        p = bst.predict(X[:10000]).reshape(-1, k)
        accs[dev] = float((np.argmax(p, axis=1) == y[:10000]).mean())
    assert accs["gpu"] > 0.7, accs  # cpu lands ~0.735 on this task
    assert abs(accs["cpu"] - accs["gpu"]) < 0.02, accs


def test_gpu_device_metric_eval_matches_host():
    """Train-metric eval must produce the same numbers through the device
    pointwise reducer as through the host path (score download)."""
    X, y = _binary_data(n=80_000)
    params = {"objective": "binary", "device_type": "gpu", "max_bin": 63,
              "num_leaves": 63, "verbosity": 0,
              "metric": ["binary_logloss", "binary_error", "l2"]}
    tr = lgb.Dataset(X, label=y)
    ev = {}
    bst = lgb.train(params, tr, 10, valid_sets=[tr], valid_names=["training"],
                    callbacks=[lgb.record_evaluation(ev)])
    # recompute each metric from downloaded predictions (raw scores -> sigmoid)
    p = bst.predict(X)
    ll = -(y * np.log(np.clip(p, 1e-12, None)) +
           (1 - y) * np.log(np.clip(1 - p, 1e-12, None))).mean()
    err = ((p > 0.5) != (y > 0)).mean()
    l2 = ((y - p) ** 2).mean()
    assert abs(ev["training"]["binary_logloss"][-1] - ll) < 1e-6
    assert abs(ev["training"]["binary_error"][-1] - err) < 1e-9
    assert abs(ev["training"]["l2"][-1] - l2) < 1e-6


def test_gpu_l1_renew_parity():
    """Device percentile renewal (binary-search order statistics) must match the
    CPU median renewal: same leaf outputs -> near-identical predictions."""
    rng = np.random.RandomState(11)
    n = 50_000
    X = rng.randn(n, 8).astype(np.float32)
    y = (2 * X[:, 0] + np.abs(X[:, 1]) + 0.3 * rng.standard_cauchy(n)).astype(np.float32)
    preds = {}
    for dev in ("cpu", "gpu"):
        params = {"objective": "regression_l1", "device_type": dev, "max_bin": 63,
                  "num_leaves": 31, "min_data_in_leaf": 50, "verbosity": 0,
                  "metric": "none"}
        bst = lgb.train(params, lgb.Dataset(X, label=y), 20)
        preds[dev] = bst.predict(X[:10000])
    rel = np.abs(preds["cpu"] - preds["gpu"]).mean() / max(0.3, np.abs(preds["cpu"]).mean())
    assert rel < 0.05, rel


def test_gpu_weighted_quantile_renew():
    """Weighted percentile renewal path (weights present + quantile alpha)."""
    rng = np.random.RandomState(3)
    n = 40_000
    X = rng.randn(n, 6).astype(np.float32)
    y = (X[:, 0] * 2 + 0.5 * rng.randn(n)).astype(np.float32)
    w = rng.uniform(0.5, 2.0, size=n).astype(np.float32)
    preds = {}
    for dev in ("cpu", "gpu"):
        params = {"objective": "quantile", "alpha": 0.8, "device_type": dev,
                  "max_bin": 63, "num_leaves": 31, "min_data_in_leaf": 50,
                  "verbosity": 0, "metric": "none"}
        bst = lgb.train(params, lgb.Dataset(X, label=y, weight=w), 20)
        preds[dev] = bst.predict(X[:10000])
    rel = np.abs(preds["cpu"] - preds["gpu"]).mean() / max(0.3, np.abs(preds["cpu"]).mean())
    assert rel < 0.08, rel


def test_gpu_kernel_hist_vs_fp64_oracle():
    """Kernel-level: the device k_hist output must match the fp64 host histogram
    oracle bin-for-bin (VERDICT r1 #3 — a wrong-by-one-bin kernel bug fails here,
    not just a 1e-3 AUC drift)."""
    import ctypes
    from lightgbm_amd.basic import _LIB, _c_str
    for seed, max_bin, n in [(0, 63, 100_000), (1, 255, 60_000), (2, 16, 40_000)]:
        rng = np.random.RandomState(seed)
        X = rng.randn(n, 12).astype(np.float32)
        # inject NaNs to cover the missing-bin path
        X[rng.rand(n, 12) < 0.02] = np.nan
        y = (X[:, 0] > 0).astype(np.float32)
        ds = lgb.Dataset(X, label=y, params={"max_bin": max_bin}).construct()
        g = rng.randn(n).astype(np.float32)
        h = rng.uniform(0.5, 2.0, n).astype(np.float32)
        err = ctypes.c_double(1e9)
        rc = _LIB.MIGBM_DebugDeviceRootHist(
            ds._handle, _c_str(f"max_bin={max_bin} num_leaves=31"),
            g.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
            h.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
            ctypes.byref(err))
        assert rc == 0
        # fp32 atomics vs fp64 sums over <=100k rows: tight bound
        assert err.value < 2e-3, (seed, max_bin, err.value)


def test_gpu_kernel_tree_structure_parity():
    """Kernel-level: CPU (fp64) and device (fp32 hist, fp64 gain) learners must
    pick the IDENTICAL split structure (feature, bin threshold, default side)
    and the identical exact leaf counts on a well-separated task — covers
    k_best_feat, k_hist_subtract and the partition kernels end to end."""
    rng = np.random.RandomState(42)
    n = 50_000
    X = rng.randn(n, 8).astype(np.float32)
    y = (1.5 * X[:, 0] - X[:, 1] + 0.7 * X[:, 2] * X[:, 3] +
         0.3 * rng.randn(n) > 0).astype(np.float32)
    dumps = {}
    for dev in ("cpu", "gpu"):
        params = {"objective": "binary", "device_type": dev, "max_bin": 63,
                  "num_leaves": 8, "min_data_in_leaf": 100, "verbosity": 0,
                  "metric": "none"}
        bst = lgb.train(params, lgb.Dataset(X, label=y), 3)
        dumps[dev] = bst.dump_model()
    for t_cpu, t_gpu in zip(dumps["cpu"]["tree_info"], dumps["gpu"]["tree_info"]):
        def walk(node, acc):
            if "split_feature" in node:
                acc.append((node["split_feature"], round(node["threshold"], 9),
                            node["default_left"], node["internal_count"]))
                walk(node["left_child"], acc)
                walk(node["right_child"], acc)
            else:
                acc.append(("leaf", node["leaf_count"]))
        a, b = [], []
        walk(t_cpu["tree_structure"], a)
        walk(t_gpu["tree_structure"], b)
        assert a == b


def test_gpu_num_gpu_single_process():
    """Single-process multi-GPU (num_gpu=2) without any torchrun rendezvous
    (reference NCCLGBDT parity). On a 1-GPU box both shard
    contexts run on device 0 through the in-process clique transport — the
    orchestration (sharding, per-shard learners, histogram/winner sync, score
    gather) is identical to the multi-device RCCL clique."""
    X, y = _binary_data(n=120_000, d=12)
    preds = {}
    for ng in (1, 2):
        params = {"objective": "binary", "device_type": "gpu", "max_bin": 63,
                  "num_leaves": 31, "min_data_in_leaf": 20, "verbosity": 0,
                  "num_gpu": ng, "metric": "none"}
        bst = lgb.train(params, lgb.Dataset(X, label=y), 10)
        assert bst.num_trees() == 10
        preds[ng] = bst.predict(X[:20000])
    # same data, same bin mappers: num_gpu=2 must match num_gpu=1 quality
    a1, a2 = _auc(y[:20000], preds[1]), _auc(y[:20000], preds[2])
    assert a2 > 0.8, (a1, a2)
    assert abs(a1 - a2) < 5e-3, (a1, a2)


def test_gpu_num_gpu_l2_and_metrics():
    """num_gpu=2 with a renewing objective is rejected? No — l1 renew syncs
    through the clique; device train metrics aggregate across shards."""
    rng = np.random.RandomState(4)
    n = 80_000
    X = rng.randn(n, 8).astype(np.float32)
    y = (2 * X[:, 0] + np.sin(X[:, 1]) + 0.2 * rng.randn(n)).astype(np.float32)
    ev = {}
    params = {"objective": "regression_l1", "device_type": "gpu", "max_bin": 63,
              "num_leaves": 31, "verbosity": 0, "num_gpu": 2, "metric": "l1"}
    tr = lgb.Dataset(X, label=y)
    bst = lgb.train(params, tr, 15, valid_sets=[tr], valid_names=["training"],
                    callbacks=[lgb.record_evaluation(ev)])
    p = bst.predict(X)
    mae = float(np.abs(p - y).mean())
    # device metric (summed over shards) must equal the host-recomputed value
    assert abs(ev["training"]["l1"][-1] - mae) < 1e-6
    assert mae < 0.5 * float(np.abs(y - y.mean()).mean())


def test_gpu_large_bins_uint16():
    """max_bin > 256 runs on device with uint16 bins (the GPU
    previously hard-errored at num_bin>256). Parity with the CPU learner at
    max_bin=1023, plus the kernel-level fp64 hist oracle at 16-bit bins."""
    import ctypes
    from lightgbm_amd.basic import _LIB, _c_str
    rng = np.random.RandomState(8)
    n = 60_000
    X = rng.randn(n, 10).astype(np.float32)
    y = (1.3 * X[:, 0] - 0.7 * X[:, 1] + 0.5 * X[:, 2] * X[:, 3] +
         0.8 * rng.randn(n) > 0).astype(np.float32)
    aucs = {}
    for dev in ("cpu", "gpu"):
        params = {"objective": "binary", "device_type": dev, "max_bin": 1023,
                  "num_leaves": 63, "min_data_in_leaf": 20, "verbosity": 0,
                  "metric": "none"}
        bst = lgb.train(params, lgb.Dataset(X, label=y), 20)
        aucs[dev] = _auc(y[:20000], bst.predict(X[:20000]))
    assert aucs["gpu"] > 0.8
    assert abs(aucs["cpu"] - aucs["gpu"]) < 3e-3, aucs
    # kernel probe: device uint16 histogram vs fp64 host oracle
    ds = lgb.Dataset(X, label=y, params={"max_bin": 1023}).construct()
    g = rng.randn(n).astype(np.float32)
    h = rng.uniform(0.5, 2.0, n).astype(np.float32)
    err = ctypes.c_double(1e9)
    rc = _LIB.MIGBM_DebugDeviceRootHist(
        ds._handle, _c_str("max_bin=1023 num_leaves=31"),
        g.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
        h.ctypes.data_as(ctypes.POINTER(ctypes.c_float)), ctypes.byref(err))
    assert rc == 0
    assert err.value < 2e-3, err.value


def test_gpu_categorical_large_cardinality():
    """Sorted-subset categorical splits beyond 64 bins on device (VERDICT r1 #6:
    previously degraded to one-hot): a 200-category feature must reach CPU-parity
    quality (the CPU oracle runs the same sorted scan)."""
    rng = np.random.RandomState(1)
    n = 80_000
    ncat = 200
    cat = rng.randint(0, ncat, size=n)
    cat_effect = rng.randn(ncat) * 1.5
    X = np.column_stack([cat.astype(np.float32),
                         rng.randn(n).astype(np.float32)])
    y = (cat_effect[cat] + 0.5 * X[:, 1] + 0.5 * rng.randn(n) > 0).astype(np.float32)
    aucs = {}
    for dev in ("cpu", "gpu"):
        params = {"objective": "binary", "device_type": dev, "max_bin": 255,
                  "num_leaves": 31, "min_data_in_leaf": 20, "verbosity": 0,
                  "metric": "none", "max_cat_threshold": 64,
                  "categorical_feature": [0]}
        bst = lgb.train(params, lgb.Dataset(X, label=y), 15)
        aucs[dev] = _auc(y[:20000], bst.predict(X[:20000]))
    # a one-hot-only device would land far below the CPU's sorted-subset quality
    assert aucs["gpu"] > 0.85, aucs
    assert abs(aucs["cpu"] - aucs["gpu"]) < 0.01, aucs


def test_gpu_use_dp_fp64_histograms():
    """gpu_use_dp=true: fp64 histogram accumulation end to end (reference dp
    parity mode). The device histogram then matches the
    fp64 host oracle to ~1e-12 instead of the fp32 ~1e-3, and training quality
    matches the sp mode."""
    import ctypes
    from lightgbm_amd.basic import _LIB, _c_str
    rng = np.random.RandomState(2)
    n = 100_000
    X = rng.randn(n, 12).astype(np.float32)
    y = (1.2 * X[:, 0] - 0.8 * X[:, 1] + X[:, 2] * X[:, 3] +
         rng.randn(n) > 0).astype(np.float32)
    ds = lgb.Dataset(X, label=y, params={"max_bin": 63}).construct()
    g = rng.randn(n).astype(np.float32)
    h = rng.uniform(0.5, 2.0, n).astype(np.float32)
    errs = {}
    for dp in ("false", "true"):
        err = ctypes.c_double(1e9)
        rc = _LIB.MIGBM_DebugDeviceRootHist(
            ds._handle, _c_str(f"max_bin=63 num_leaves=31 gpu_use_dp={dp}"),
            g.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
            h.ctypes.data_as(ctypes.POINTER(ctypes.c_float)), ctypes.byref(err))
        assert rc == 0
        errs[dp] = err.value
    assert errs["true"] < 1e-10, errs   # fp64 = oracle precision
    assert errs["false"] < 2e-3, errs   # fp32 stays within the sp envelope
    aucs = {}
    for dp in (False, True):
        params = {"objective": "binary", "device_type": "gpu", "max_bin": 63,
                  "num_leaves": 63, "verbosity": 0, "gpu_use_dp": dp,
                  "metric": "none"}
        bst = lgb.train(params, lgb.Dataset(X, label=y), 15)
        aucs[dp] = _auc(y[:20000], bst.predict(X[:20000]))
    assert aucs[True] > 0.8
    assert abs(aucs[True] - aucs[False]) < 2e-3, aucs


def test_gpu_random_forest():
    """boosting=rf on the device learner (round 1 fell back to CPU): quality
    parity with the CPU RF at identical config."""
    X, y = _binary_data(n=80_000, d=10)
    aucs = {}
    for dev in ("cpu", "gpu"):
        params = {"objective": "binary", "boosting": "rf", "device_type": dev,
                  "bagging_freq": 1, "bagging_fraction": 0.6, "num_leaves": 63,
                  "verbosity": 0, "metric": "none"}
        bst = lgb.train(params, lgb.Dataset(X, label=y), 20)
        aucs[dev] = _auc(y[:20000], bst.predict(X[:20000]))
    assert aucs["gpu"] > 0.8, aucs
    assert abs(aucs["cpu"] - aucs["gpu"]) < 5e-3, aucs


def test_gpu_max_depth_enforced():
    """max_depth must bound the device-built trees (was silently ignored on GPU
    before round 2 — the depth now lives in the device LeafStat)."""
    X, y = _binary_data(n=60_000, d=10)
    for md in (3, 6):
        params = {"objective": "binary", "device_type": "gpu", "num_leaves": 255,
                  "max_depth": md, "verbosity": 0, "metric": "none",
                  "min_data_in_leaf": 1}
        bst = lgb.train(params, lgb.Dataset(X, label=y), 5)
        d = bst.dump_model()
        def depth(node):
            if "split_feature" not in node:
                return 0
            return 1 + max(depth(node["left_child"]), depth(node["right_child"]))
        for t in d["tree_info"]:
            assert depth(t["tree_structure"]) <= md, (md, depth(t["tree_structure"]))
        # and a depth-md tree has at most 2^md leaves
        for t in d["tree_info"]:
            assert t["num_leaves"] <= 2 ** md


def test_gpu_monotone_penalty():
    """monotone_penalty now runs in the device gain scan (was a loud CPU
    fallback): monotonicity holds and penalized models differ from unpenalized."""
    rng = np.random.RandomState(0)
    X = rng.rand(30000, 3)
    y = (2 * X[:, 0] + np.sin(5 * X[:, 1]) + 0.1 * rng.randn(30000)).astype(np.float32)
    preds = {}
    for pen in (0.0, 2.0):
        params = {"objective": "regression", "device_type": "cuda",
                  "monotone_constraints": [1, 0, 0], "monotone_penalty": pen,
                  "num_leaves": 63, "verbosity": -1, "metric": "none"}
        bst = lgb.train(params, lgb.Dataset(X, label=y), 30)
        xs = np.linspace(0.02, 0.98, 30)
        for other in (0.2, 0.8):
            grid = np.column_stack([xs, np.full(30, other), np.full(30, other)])
            assert np.all(np.diff(bst.predict(grid)) >= -1e-9)
        preds[pen] = bst.predict(X[:3000])
    assert np.abs(preds[0.0] - preds[2.0]).max() > 1e-6  # penalty changed the model


def test_gpu_init_score_honored():
    """Per-row init_score must seed the device score buffer (was silently
    dropped on GPU). Parity: GPU model trained on residuals equals the CPU one
    at quality level."""
    rng = np.random.RandomState(5)
    n = 50_000
    X = rng.randn(n, 8).astype(np.float32)
    y = (2 * X[:, 0] + X[:, 1] + 0.3 * rng.randn(n)).astype(np.float32)
    init = (2 * X[:, 0]).astype(np.float64)  # a strong known component
    preds = {}
    for dev in ("cpu", "gpu"):
        params = {"objective": "regression", "device_type": dev, "num_leaves": 31,
                  "verbosity": 0, "metric": "none"}
        ds = lgb.Dataset(X, label=y, init_score=init)
        bst = lgb.train(params, ds, 15)
        preds[dev] = bst.predict(X[:5000])
    # with init_score absorbed, the trees model y - init ~= X1: if the GPU
    # ignored init_score its trees would re-learn 2*X0 and diverge sharply
    diff = np.abs(preds["cpu"] - preds["gpu"]).mean()
    assert diff < 0.05, diff
    resid_target = y[:5000] - init[:5000]
    mse = float(np.mean((preds["gpu"] - resid_target) ** 2))
    assert mse < 0.5 * float(np.var(resid_target)), mse


def test_gpu_cegb():
    """Cost-effective gradient boosting runs in the device gain scan (was a
    loud CPU fallback): coupled penalties steer the model to fewer features,
    matching the CPU learner's behavior."""
    rng = np.random.RandomState(0)
    n = 40_000
    X = rng.randn(n, 6).astype(np.float32)
    # two redundant informative features; CEGB coupled penalty should make the
    # model reuse one instead of paying for both
    y = (X[:, 0] + 0.95 * X[:, 1] + 0.2 * rng.randn(n) > 0).astype(np.float32)
    used = {}
    for dev in ("cpu", "gpu"):
        params = {"objective": "binary", "device_type": dev, "num_leaves": 31,
                  "verbosity": 0, "metric": "none", "cegb_tradeoff": 1.0,
                  "cegb_penalty_feature_coupled": [50.0] * 6}
        bst = lgb.train(params, lgb.Dataset(X, label=y), 10)
        imp = bst.feature_importance()
        used[dev] = int((imp > 0).sum())
        assert _auc(y[:10000], bst.predict(X[:10000])) > 0.8
    # both learners concentrate on few features under the coupled penalty
    assert used["gpu"] <= 3, used
    assert abs(used["cpu"] - used["gpu"]) <= 1, used


def test_gpu_refit():
    """Booster.refit on a device-trained model (scores downloaded for the
    gradient re-derivation)."""
    X, y = _binary_data(n=30_000, d=8)
    params = {"objective": "binary", "device_type": "gpu", "num_leaves": 31,
              "verbosity": 0, "metric": "none"}
    bst = lgb.train(params, lgb.Dataset(X, label=y), 8)
    y2 = 1.0 - y  # refit onto flipped labels shifts outputs
    new_bst = bst.refit(X, y2)
    p_old = bst.predict(X[:2000])
    p_new = new_bst.predict(X[:2000])
    # refit_decay_rate=0.9 keeps most of the old outputs: CPU shows ~0.03 mean
    # delta on this setup — require a comparable, clearly-nonzero shift
    assert np.abs(p_old - p_new).mean() > 0.01


def test_gpu_forced_splits(tmp_path):
    """forcedsplits_filename drives the device split loop (was a loud CPU
    fallback): the first splits follow the forced JSON exactly."""
    import json as _json
    rng = np.random.RandomState(0)
    n = 40_000
    X = rng.rand(n, 4).astype(np.float32)
    y = (X[:, 0] + 0.5 * X[:, 1] + 0.1 * rng.randn(n) > 0.8).astype(np.float32)
    fs = tmp_path / "forced.json"
    fs.write_text(_json.dumps(
        {"feature": 0, "threshold": 0.5,
         "left": {"feature": 1, "threshold": 0.25}}))
    params = {"objective": "binary", "device_type": "gpu", "num_leaves": 15,
              "verbosity": 0, "metric": "none",
              "forcedsplits_filename": str(fs)}
    bst = lgb.train(params, lgb.Dataset(X, label=y), 3)
    d = bst.dump_model()
    for t in d["tree_info"]:
        root = t["tree_structure"]
        assert root["split_feature"] == 0
        assert abs(root["threshold"] - 0.5) < 0.02
        left = root["left_child"]
        assert left["split_feature"] == 1
        assert abs(left["threshold"] - 0.25) < 0.02
    assert _auc(y[:10000], bst.predict(X[:10000])) > 0.8


def test_gpu_linear_tree():
    """linear_tree on the device learner (was the last CPU fallback): per-leaf
    Gram matrices accumulate on GPU, tiny Cholesky solves on host, linear-aware
    device score update. Quality parity with the CPU linear-tree learner."""
    rng = np.random.RandomState(7)
    n = 60_000
    X = rng.rand(n, 4).astype(np.float32)
    # piecewise-LINEAR target: linear trees fit it far better than constants
    y = (np.where(X[:, 0] > 0.5, 3 * X[:, 1], -2 * X[:, 1]) +
         0.05 * rng.randn(n)).astype(np.float32)
    mses = {}
    preds = {}
    for dev in ("cpu", "gpu"):
        params = {"objective": "regression", "device_type": dev, "num_leaves": 15,
                  "linear_tree": True, "learning_rate": 0.5, "verbosity": 0,
                  "metric": "none"}
        bst = lgb.train(params, lgb.Dataset(X, label=y), 30)
        preds[dev] = bst.predict(X[:10000])
        mses[dev] = float(np.mean((preds[dev] - y[:10000]) ** 2))
    var = float(np.var(y))
    assert mses["gpu"] < 0.01 * var, (mses, var)  # linear fit => near-perfect
    assert mses["gpu"] < mses["cpu"] * 1.5 + 1e-5, mses
    assert np.abs(preds["cpu"] - preds["gpu"]).mean() < 0.05
    # and a constant-leaf model is clearly worse on this target
    params_c = {"objective": "regression", "device_type": "gpu", "num_leaves": 15,
                "learning_rate": 0.5, "verbosity": 0, "metric": "none"}
    bst_c = lgb.train(params_c, lgb.Dataset(X, label=y), 30)
    mse_c = float(np.mean((bst_c.predict(X[:10000]) - y[:10000]) ** 2))
    assert mses["gpu"] < 0.5 * mse_c, (mses["gpu"], mse_c)


def test_gpu_interaction_constraints_wide():
    """>64-feature interaction constraints on device (4-word branch masks):
    every root->leaf path must stay within one feature group."""
    rng = np.random.RandomState(5)
    nf = 100
    X = rng.randn(30000, nf)
    y = (X[:, 0] * X[:, 1] + X[:, 70] * X[:, 90] > 0).astype(np.float32)
    g1 = ",".join(str(i) for i in range(50))
    g2 = ",".join(str(i) for i in range(50, nf))
    bst = lgb.train({"objective": "binary", "device_type": "cuda", "num_leaves": 31,
                     "interaction_constraints": f"[{g1}],[{g2}]", "verbosity": -1},
                    lgb.Dataset(X, label=y), 12)
    d = bst.dump_model()

    def paths(node, acc, out):
        if "leaf_index" in node:
            out.append(set(acc))
            return
        paths(node["left_child"], acc + [node["split_feature"]], out)
        paths(node["right_child"], acc + [node["split_feature"]], out)
    lo, hi = set(range(50)), set(range(50, nf))
    n_split_trees = 0
    for t in d["tree_info"]:
        out = []
        paths(t["tree_structure"], [], out)
        for p in out:
            if p:
                n_split_trees += 1
                assert p <= lo or p <= hi, p
    assert n_split_trees > 0


def test_gpu_missing_only_split():
    """device scan allows the all-numeric-vs-NaN split: a constant column whose
    only signal is missingness must still separate (mirrors the CPU fix)."""
    rng = np.random.RandomState(6)
    n = 20000
    X = np.zeros((n, 2))
    X[:, 1] = rng.randn(n) * 0.01  # noise column so the dataset isn't degenerate
    y = np.zeros(n, dtype=np.float32)
    nan_rows = rng.choice(n, n // 5, replace=False)
    X[nan_rows, 0] = np.nan
    y[nan_rows] = 1.0
    bst = lgb.train({"objective": "regression", "device_type": "cuda",
                     "verbosity": -1, "boost_from_average": False},
                    lgb.Dataset(X, label=y), 20)
    mse = float(np.mean((bst.predict(X) - y) ** 2))
    assert mse < 0.005, mse


def test_gpu_bagging_categorical_oob():
    """OOB score replay must route categorical splits via bin-space masks:
    GPU bagging+categorical quality must match the CPU learner's."""
    rng = np.random.RandomState(7)
    n = 40000
    cat = rng.randint(0, 12, n).astype(np.float64)
    lut = rng.randn(12) * 2
    X = np.column_stack([cat, rng.randn(n)])
    y = (lut[cat.astype(int)] + 0.3 * X[:, 1] + 0.3 * rng.randn(n) > 0).astype(np.float32)
    aucs = {}
    for dev in ("cpu", "cuda"):
        bst = lgb.train({"objective": "binary", "device_type": dev, "verbosity": -1,
                         "bagging_fraction": 0.6, "bagging_freq": 1, "seed": 7,
                         "categorical_feature": "0"},
                        lgb.Dataset(X, label=y, categorical_feature=[0]), 30)
        from sklearn.metrics import roc_auc_score
        aucs[dev] = roc_auc_score(y, bst.predict(X))
    assert aucs["cuda"] > aucs["cpu"] - 0.01, aucs


def test_gpu_linear_tree_bagging():
    """linear_tree + bagging on device: OOB rows get linear-leaf score updates
    (previously a host fallback). Linear fit must beat constant leaves."""
    rng = np.random.RandomState(8)
    n = 30000
    X = rng.rand(n, 3)
    y = (3.0 * X[:, 0] + np.where(X[:, 1] > 0.5, 2.0 * X[:, 2], -2.0 * X[:, 2]) +
         0.05 * rng.randn(n)).astype(np.float32)
    common = {"objective": "regression", "device_type": "cuda", "verbosity": -1,
              "num_leaves": 8, "learning_rate": 0.5, "bagging_fraction": 0.6,
              "bagging_freq": 1, "seed": 3}
    lin = lgb.train({**common, "linear_tree": True}, lgb.Dataset(X, label=y), 30)
    const = lgb.train(common, lgb.Dataset(X, label=y), 30)
    mse_lin = float(np.mean((lin.predict(X) - y) ** 2))
    mse_const = float(np.mean((const.predict(X) - y) ** 2))
    assert mse_lin < mse_const * 0.8, (mse_lin, mse_const)


def test_gpu_categorical_nan_bin():
    """NaN categorical values occupy the reserved bin 0 on device too: a model
    must be able to separate missing from category 0."""
    rng = np.random.RandomState(9)
    n = 20000
    cat = rng.randint(0, 5, n).astype(np.float64)
    nan_rows = rng.choice(n, n // 4, replace=False)
    cat[nan_rows] = np.nan
    X = np.column_stack([cat, rng.randn(n) * 0.01])
    lut = np.array([1.0, 2.0, 3.0, 4.0, 5.0])
    y = np.where(np.isnan(cat), -2.0, lut[np.nan_to_num(cat).astype(int)]).astype(np.float32)
    bst = lgb.train({"objective": "regression", "device_type": "cuda",
                     "verbosity": -1, "learning_rate": 0.5, "num_leaves": 15,
                     "min_data_in_leaf": 1, "cat_l2": 0.0, "cat_smooth": 1e-3},
                    lgb.Dataset(X, label=y, categorical_feature=[0]), 30)
    mse = float(np.mean((bst.predict(X) - y) ** 2))
    assert mse < 0.01, mse


def test_gpu_continue_training_equals_straight():
    """init_model continuation on the device learner: merged-tree scores are
    uploaded to the GPU so new gradients continue from them — predictions match
    an uninterrupted run."""
    rng = np.random.RandomState(10)
    X = rng.randn(60000, 6).astype(np.float64)
    y = (X[:, 0] + 0.5 * X[:, 1] + 0.4 * rng.randn(60000) > 0).astype(np.float32)
    p = {"objective": "binary", "device_type": "cuda", "verbosity": -1,
         "learning_rate": 0.2, "num_leaves": 31}
    straight = lgb.train(p, lgb.Dataset(X, label=y), 10)
    half = lgb.train(p, lgb.Dataset(X, label=y), 5)
    cont = lgb.train(p, lgb.Dataset(X, label=y), 5, init_model=half)
    ps, pc = straight.predict(X[:2000]), cont.predict(X[:2000])
    # fp32 histogram accumulation order differs run to run; demand agreement
    # at the quality level and close pointwise tracking
    from sklearn.metrics import roc_auc_score
    assert abs(roc_auc_score(y[:2000], ps) - roc_auc_score(y[:2000], pc)) < 0.005
    assert np.mean(np.abs(ps - pc)) < 0.02, float(np.mean(np.abs(ps - pc)))
