"""CLI application tests (parity target: reference test_consistency.py — CLI-config
examples vs Python API on the same data)."""
import subprocess
from pathlib import Path

import numpy as np
import pytest

import lightgbm_amd as lgb

REPO = Path(__file__).resolve().parent.parent
CLI = REPO / "lightgbm_amd" / "bin" / "migbm"
EXAMPLE = REPO / "examples" / "binary_classification"


@pytest.fixture(scope="module")
def cli_model(tmp_path_factory):
    out = tmp_path_factory.mktemp("cli")
    model = out / "model.txt"
    r = subprocess.run(
        [str(CLI), f"config={EXAMPLE/'train.conf'}", f"data={EXAMPLE/'binary.train'}",
         f"valid_data={EXAMPLE/'binary.test'}", f"output_model={model}",
         "num_trees=30", "verbosity=-1"],
        capture_output=True, text=True, timeout=300, cwd=str(out))
    assert r.returncode == 0, r.stderr
    return model


def _load_tsv(path):
    data = np.loadtxt(path, delimiter="\t")
    return data[:, 1:], data[:, 0]


def test_cli_trains_and_predicts(cli_model, tmp_path):
    assert cli_model.exists()
    result = tmp_path / "pred.txt"
    r = subprocess.run(
        [str(CLI), "task=predict", f"data={EXAMPLE/'binary.test'}",
         f"input_model={cli_model}", f"output_result={result}"],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr
    preds = np.loadtxt(result)
    X, y = _load_tsv(EXAMPLE / "binary.test")
    assert len(preds) == len(y)
    acc = ((preds > 0.5) == y).mean()
    assert acc > 0.75


def test_cli_python_consistency(cli_model):
    """The Python API must produce the same predictions from the CLI-trained model."""
    X, y = _load_tsv(EXAMPLE / "binary.test")
    bst = lgb.Booster(model_file=str(cli_model))
    pred_py = bst.predict(X)
    # CLI predict on the same file
    import subprocess, tempfile
    with tempfile.TemporaryDirectory() as td:
        result = Path(td) / "pred.txt"
        subprocess.run([str(CLI), "task=predict", f"data={EXAMPLE/'binary.test'}",
                        f"input_model={cli_model}", f"output_result={result}"],
                       capture_output=True, timeout=300, check=True)
        pred_cli = np.loadtxt(result)
    np.testing.assert_allclose(pred_py, pred_cli, rtol=1e-10)


def test_cli_python_same_training():
    """Training via Python on the example files reaches similar quality."""
    X, y = _load_tsv(EXAMPLE / "binary.train")
    Xt, yt = _load_tsv(EXAMPLE / "binary.test")
    bst = lgb.train({"objective": "binary", "num_leaves": 63, "min_data_in_leaf": 50,
                     "min_sum_hessian_in_leaf": 5.0, "verbosity": -1},
                    lgb.Dataset(X, label=y), 30)
    acc = ((bst.predict(Xt) > 0.5) == yt).mean()
    assert acc > 0.75


def test_convert_model_cpp_codegen(tmp_path):
    """convert_model_language=cpp emits standalone C++ whose compiled predictions
    match Booster.predict bit-for-bit (incl. missing + categorical handling)."""
    import subprocess
    rng = np.random.RandomState(3)
    X = rng.rand(1500, 5)
    X[:, 3] = rng.randint(0, 8, 1500)              # categorical
    X[rng.rand(1500) < 0.1, 1] = np.nan            # missing
    y = ((X[:, 0] + (X[:, 3] % 3 == 1) + np.nan_to_num(X[:, 1])) > 1.4).astype(np.float64)
    bst = lgb.train({"objective": "binary", "num_leaves": 15, "verbosity": -1,
                     "categorical_feature": [3]}, lgb.Dataset(X, label=y), 12)
    model_file = tmp_path / "m.txt"
    bst.save_model(str(model_file))
    gen = tmp_path / "gen.cpp"
    subprocess.run([str(CLI), "task=convert_model", f"input_model={model_file}",
                    "convert_model_language=cpp", f"convert_model={gen}"],
                   capture_output=True, timeout=300, check=True)
    src = gen.read_text()
    assert "PredictTree0" in src and "void Predict(" in src
    # compile with a tiny main and compare predictions
    main = tmp_path / "main.cpp"
    main.write_text(src + r"""
#include <cstdio>
int main() {
    double row[MIGBM_NUM_FEATURES]; double out[MIGBM_NUM_CLASSES];
    while (std::fscanf(stdin, "%lf %lf %lf %lf %lf", row, row+1, row+2, row+3, row+4) == 5) {
        Predict(row, out);
        std::printf("%.17g\n", out[0]);
    }
    return 0;
}
""")
    exe = tmp_path / "pred"
    subprocess.run(["g++", "-O1", "-o", str(exe), str(main)], check=True, timeout=300)
    rows = "\n".join(" ".join("nan" if np.isnan(v) else "%.17g" % v for v in r) for r in X[:200])
    res = subprocess.run([str(exe)], input=rows, capture_output=True, text=True,
                         timeout=300, check=True)
    pred_gen = np.array([float(t) for t in res.stdout.split()])
    np.testing.assert_allclose(pred_gen, bst.predict(X[:200]), rtol=1e-12)


def test_cli_arg_overrides_config_alias():
    """An argv param must override its ALIAS form in the config file
    (num_iterations on argv vs num_trees in train.conf)."""
    import subprocess, tempfile
    with tempfile.TemporaryDirectory() as td:
        out = Path(td) / "m.txt"
        subprocess.run([str(CLI), "config=train.conf", "num_iterations=3",
                        f"output_model={out}"],
                       cwd=EXAMPLE, capture_output=True, timeout=300, check=True)
        assert out.read_text().count("Tree=") == 3


def test_cli_refit(cli_model, tmp_path):
    """task=refit re-derives leaf values on new data and saves a working model."""
    import subprocess
    out = tmp_path / "refit.txt"
    subprocess.run([str(CLI), "task=refit", f"data={EXAMPLE/'binary.train'}",
                    f"input_model={cli_model}", f"output_model={out}"],
                   capture_output=True, timeout=300, check=True)
    X, y = _load_tsv(EXAMPLE / "binary.test")
    bst = lgb.Booster(model_file=str(out))
    acc = ((bst.predict(X) > 0.5) == y).mean()
    assert acc > 0.7


def test_cli_predict_variants(cli_model, tmp_path):
    """predict_raw_score / predict_leaf_index / predict_contrib output shapes."""
    import subprocess
    ntrees = lgb.Booster(model_file=str(cli_model)).num_trees()
    for flag, ncols in (("predict_raw_score", 1), ("predict_leaf_index", ntrees),
                        ("predict_contrib", 29)):
        out = tmp_path / f"{flag}.txt"
        subprocess.run([str(CLI), "task=predict", f"data={EXAMPLE/'binary.test'}",
                        f"input_model={cli_model}", f"output_result={out}",
                        f"{flag}=true"], capture_output=True, timeout=300, check=True)
        first = out.read_text().splitlines()[0].split("\t")
        assert len(first) == ncols, (flag, len(first))


def test_position_sidecar_loading(tmp_path):
    """A <file>.position sidecar feeds position-debiased lambdarank training."""
    import subprocess
    rng = np.random.RandomState(0)
    rows, qsizes, positions = [], [], []
    for q in range(80):
        nq = 10
        Xq = rng.rand(nq, 4)
        rel = (Xq[:, 0] > 0.5).astype(int)
        for i in range(nq):
            rows.append(f"{rel[i]}\t" + "\t".join(f"{v:.6f}" for v in Xq[i]))
            positions.append(i)
        qsizes.append(nq)
    train = tmp_path / "rank.train"
    train.write_text("\n".join(rows) + "\n")
    (tmp_path / "rank.train.query").write_text("\n".join(map(str, qsizes)) + "\n")
    (tmp_path / "rank.train.position").write_text("\n".join(map(str, positions)) + "\n")
    out = tmp_path / "m.txt"
    r = subprocess.run([str(CLI), "task=train", "objective=lambdarank",
                        f"data={train}", "num_trees=10", f"output_model={out}"],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    assert out.exists()
    # positions were seen -> debiasing engaged (auto when positions exist)
    assert "position debiasing enabled" in (r.stdout + r.stderr)


def test_cli_binary_dataset_roundtrip(tmp_path):
    """save_binary writes <data>.bin; training from the .bin reproduces the exact
    same trees as training from text."""
    import subprocess, shutil
    work = tmp_path / "ex"
    shutil.copytree(EXAMPLE, work)
    m1, m2 = tmp_path / "m1.txt", tmp_path / "m2.txt"
    subprocess.run([str(CLI), "config=train.conf", "save_binary=true", "num_trees=5",
                    f"output_model={m1}"], cwd=work, capture_output=True,
                   timeout=300, check=True)
    assert (work / "binary.train.bin").exists()
    subprocess.run([str(CLI), "config=train.conf", "data=binary.train.bin",
                    "num_trees=5", f"output_model={m2}"], cwd=work,
                   capture_output=True, timeout=300, check=True)

    def trees(p):
        t = p.read_text()
        return t[t.index("Tree=0"):t.index("end of trees")]
    assert trees(m1) == trees(m2)


def test_cli_column_specs(tmp_path):
    """weight_column / group_column / ignore_column with header + name: resolution."""
    import subprocess
    rng = np.random.RandomState(0)
    n = 1200
    lines = ["label\tw\tqid\tf0\tf1\tjunk"]
    qid = np.repeat(np.arange(60), 20)
    for i in range(n):
        f0, f1 = rng.rand(), rng.rand()
        y = int(f0 > 0.5)
        lines.append(f"{y}\t{1.0 + (i % 3)}\t{qid[i]}\t{f0:.6f}\t{f1:.6f}\t999")
    data = tmp_path / "t.tsv"
    data.write_text("\n".join(lines) + "\n")
    out = tmp_path / "m.txt"
    r = subprocess.run([str(CLI), "task=train", "objective=lambdarank", f"data={data}",
                        "header=true", "label_column=name:label",
                        "weight_column=name:w", "group_column=name:qid",
                        "ignore_column=name:junk", "num_trees=5",
                        f"output_model={out}"],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    txt = out.read_text()
    # only f0/f1 remain as features
    assert "max_feature_idx=1" in txt


@pytest.mark.parametrize("example", ["binary_classification", "regression",
                                     "multiclass_classification", "lambdarank"])
def test_bundled_examples_run(example, tmp_path):
    """Every bundled CLI example trains and (where configured) predicts."""
    import subprocess, shutil
    src = REPO / "examples" / example
    work = tmp_path / example
    shutil.copytree(src, work)
    r = subprocess.run([str(CLI), "config=train.conf", "num_trees=10"], cwd=work,
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    assert list(work.glob("*model.txt")), "no model file written"
    if (work / "predict.conf").exists():
        r2 = subprocess.run([str(CLI), "config=predict.conf"], cwd=work,
                            capture_output=True, text=True, timeout=300)
        assert r2.returncode == 0, r2.stdout + r2.stderr


def test_convert_model_cpp_linear_trees(tmp_path):
    """C++ codegen emits linear-leaf expressions (const + coeffs), matching the
    Booster bit-for-bit."""
    import subprocess
    rng = np.random.RandomState(0)
    X = rng.rand(2000, 3)
    y = (3 * X[:, 0] + np.sin(4 * X[:, 1])).astype(np.float32)
    bst = lgb.train({"objective": "regression", "linear_tree": True, "verbosity": -1},
                    lgb.Dataset(X, label=y), 5)
    m = tmp_path / "lin.txt"
    bst.save_model(str(m))
    gen = tmp_path / "gen.cpp"
    subprocess.run([str(CLI), "task=convert_model", f"input_model={m}",
                    "convert_model_language=cpp", f"convert_model={gen}"],
                   check=True, capture_output=True, timeout=300)
    main = tmp_path / "main.cpp"
    main.write_text(gen.read_text() + r"""
#include <cstdio>
int main(){double r[3],o[1];
 while(std::scanf("%lf %lf %lf",r,r+1,r+2)==3){Predict(r,o);std::printf("%.17g\n",o[0]);}
 return 0;}
""")
    exe = tmp_path / "pred"
    subprocess.run(["g++", "-O1", "-o", str(exe), str(main)], check=True, timeout=300)
    rows = "\n".join(" ".join("%.17g" % v for v in r) for r in X[:100])
    res = subprocess.run([str(exe)], input=rows, capture_output=True, text=True,
                         check=True, timeout=300)
    gen_pred = np.array([float(t) for t in res.stdout.split()])
    np.testing.assert_allclose(gen_pred, bst.predict(X[:100]), rtol=1e-10)


def test_cli_distributed_socket_mesh(tmp_path):
    """Distributed CPU training over the standalone TCP socket mesh — NO torch,
    no injected collectives: N CLI processes, localhost machine list (reference
    tests/distributed/_test_distributed.py mockup). Models must be identical
    across workers and predict well."""
    import socket
    from concurrent.futures import ThreadPoolExecutor

    def free_port():
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        p = s.getsockname()[1]
        s.close()
        return p

    n_workers = 2
    ports = [free_port() for _ in range(n_workers)]
    machines = ",".join(f"127.0.0.1:{p}" for p in ports)

    rng = np.random.RandomState(3)
    n = 6000
    X = rng.randn(n, 6)
    y = (X[:, 0] + 0.5 * X[:, 1] + 0.2 * rng.randn(n) > 0).astype(int)
    train = tmp_path / "train.tsv"
    np.savetxt(train, np.column_stack([y, X]), delimiter="\t", fmt="%.8g")

    def run_worker(i):
        model = tmp_path / f"model{i}.txt"
        r = subprocess.run(
            [str(CLI), "task=train", f"data={train}", "objective=binary",
             "tree_learner=data", f"num_machines={n_workers}",
             f"machines={machines}", f"local_listen_port={ports[i]}",
             "time_out=60", "num_trees=15", "num_leaves=31", "verbosity=-1",
             f"output_model={model}"],
            capture_output=True, text=True, timeout=240)
        return i, r, model

    with ThreadPoolExecutor(n_workers) as ex:
        results = list(ex.map(run_worker, range(n_workers)))
    models = []
    for i, r, model in results:
        assert r.returncode == 0, f"worker {i}:\n{r.stdout[-2000:]}\n{r.stderr[-2000:]}"
        models.append(model.read_text())
    trees = [m[m.index("Tree=0"):m.index("end of trees")] for m in models]
    assert trees[0] == trees[1]

    # quality: load in the Python package and predict
    bst = lgb.Booster(model_file=str(results[0][2]))
    acc = ((bst.predict(X) > 0.5) == y).mean()
    assert acc > 0.93, acc


def test_cli_python_consistency(tmp_path):
    """CLI config training and Python API training on the same example data
    produce the same model predictions (ref tests/python_package_test/
    test_consistency.py semantics)."""
    import shutil
    ex = Path(__file__).resolve().parent.parent / "examples" / "binary_classification"
    work = tmp_path / "w"
    shutil.copytree(ex, work)
    subprocess.run([str(CLI), "config=train.conf", "num_trees=20",
                    "metric_freq=100"], cwd=work, check=True, timeout=300,
                   capture_output=True)
    model_file = work / "LightGBM_model.txt"
    assert model_file.exists()
    cli_bst = lgb.Booster(model_file=str(model_file))

    # same data through the Python API with the same params
    data = np.loadtxt(work / "binary.train")
    y, X = data[:, 0], data[:, 1:]
    params = {"objective": "binary", "metric": ["binary_logloss", "auc"],
              "max_bin": 255, "num_leaves": 63, "learning_rate": 0.1,
              "min_data_in_leaf": 50, "min_sum_hessian_in_leaf": 5.0,
              "verbosity": -1}
    py_bst = lgb.train(params, lgb.Dataset(X, label=y), 20)

    test = np.loadtxt(work / "binary.test")
    Xt, yt = test[:, 1:], test[:, 0]
    p_cli = cli_bst.predict(Xt)
    p_py = py_bst.predict(Xt)
    np.testing.assert_allclose(p_cli, p_py, rtol=1e-9)


def test_cli_snapshot_and_continue(tmp_path):
    """snapshot_freq writes model.txt.snapshot_iter_N checkpoints; input_model
    continues training from a checkpoint (ref Application/GBDT::Train)."""
    import shutil
    ex = Path(__file__).resolve().parent.parent / "examples" / "binary_classification"
    work = tmp_path / "w"
    shutil.copytree(ex, work)
    subprocess.run([str(CLI), "config=train.conf", "num_trees=10", "snapshot_freq=5",
                    "metric_freq=100"], cwd=work, check=True, timeout=300,
                   capture_output=True)
    snap = work / "LightGBM_model.txt.snapshot_iter_5"
    assert snap.exists()
    assert lgb.Booster(model_file=str(snap)).num_trees() == 5
    # continue from the snapshot
    subprocess.run([str(CLI), "config=train.conf", "num_trees=5",
                    f"input_model={snap}", "output_model=cont.txt",
                    "metric_freq=100"], cwd=work, check=True, timeout=300,
                   capture_output=True)
    cont = lgb.Booster(model_file=str(work / "cont.txt"))
    assert cont.num_trees() == 10
