#!/usr/bin/env python3
"""CPU head-to-head: migbm vs the reference oracle (tools/oracle/lib_lightgbm.so)
on the Higgs-shaped 1Mx28 binary config. Both run the identical data/params via
their C APIs. Usage: python tools/cpu_headtohead.py [--rows N] [--iters K]
[--hard] — --hard forces full 255-leaf trees (min_data_in_leaf=1, min_sum_hessian=1).
"""
import argparse
import ctypes
import sys
import time
from pathlib import Path

import numpy as np

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))


def make_data(n, d, seed=1234):
    rng = np.random.RandomState(seed)
    X = rng.randn(n, d).astype(np.float64)
    logit = (1.2 * X[:, 0] - 0.8 * X[:, 1] + 0.9 * X[:, 2] * X[:, 3] +
             0.6 * np.sin(2 * X[:, 4]) + 0.45 * X[:, 5])
    y = (logit + 1.1 * rng.randn(n) > 0).astype(np.float32)
    return np.ascontiguousarray(X), y


def run_lib(libpath, X, y, params, iters, warmup=3):
    lib = ctypes.CDLL(str(libpath))
    ds = ctypes.c_void_p()
    pstr = " ".join(f"{k}={v}" for k, v in params.items()).encode()
    rc = lib.LGBM_DatasetCreateFromMat(
        X.ctypes.data_as(ctypes.c_void_p), ctypes.c_int(1),  # C_API_DTYPE_FLOAT64
        ctypes.c_int32(X.shape[0]), ctypes.c_int32(X.shape[1]), ctypes.c_int(1),
        pstr, None, ctypes.byref(ds))
    assert rc == 0
    rc = lib.LGBM_DatasetSetField(ds, b"label", y.ctypes.data_as(ctypes.c_void_p),
                                  ctypes.c_int(len(y)), ctypes.c_int(0))
    assert rc == 0
    bst = ctypes.c_void_p()
    rc = lib.LGBM_BoosterCreate(ds, pstr, ctypes.byref(bst))
    assert rc == 0
    fin = ctypes.c_int(0)
    for _ in range(warmup):
        lib.LGBM_BoosterUpdateOneIter(bst, ctypes.byref(fin))
    # best-of-3 blocks: the container CPU is shared, min filters interference
    dt = float("inf")
    for _ in range(3):
        t0 = time.time()
        for _ in range(iters):
            lib.LGBM_BoosterUpdateOneIter(bst, ctypes.byref(fin))
        dt = min(dt, (time.time() - t0) / iters)
    # leaf count of the last tree (did we hit the full-tree hard target?)
    lib.LGBM_BoosterFree(bst)
    lib.LGBM_DatasetFree(ds)
    return dt * 1000.0


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=1_000_000)
    ap.add_argument("--features", type=int, default=28)
    ap.add_argument("--iters", type=int, default=10)
    ap.add_argument("--hard", action="store_true")
    ap.add_argument("--objective", default="binary")
    args = ap.parse_args()

    X, y = make_data(args.rows, args.features)
    params = {
        "objective": args.objective,
        "max_bin": 63,
        "num_leaves": 255,
        "learning_rate": 0.1,
        "verbosity": -1,
        "metric": "none",
        "num_threads": 8,
    }
    if args.hard:
        params["min_data_in_leaf"] = 1
        params["min_sum_hessian_in_leaf"] = 1
    else:
        params["min_data_in_leaf"] = 1
        params["min_sum_hessian_in_leaf"] = 100

    ours = REPO / "lightgbm_amd" / "lib" / "lib_migbm.so"
    ref = REPO / "tools" / "oracle" / "lib_lightgbm.so"
    for name, lib in [("migbm", ours), ("reference", ref)]:
        if not lib.exists():
            print(f"{name}: missing {lib}")
            continue
        ms = run_lib(lib, X, y, params, args.iters)
        print(f"{name:10s}: {ms:7.1f} ms/iter  ({'hard' if args.hard else 'easy'} target, "
              f"{args.rows} rows)")


if __name__ == "__main__":
    main()
