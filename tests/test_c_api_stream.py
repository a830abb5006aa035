"""Streaming push C API tests (parity target: reference tests/cpp_tests/test_stream.cpp
and c_api_test/test_.py — exercised from ctypes)."""
import ctypes

import numpy as np

import lightgbm_amd as lgb
from lightgbm_amd.basic import _LIB, _safe_call, _c_str


def _sampled_column_dataset(X, params="max_bin=63"):
    n, d = X.shape
    col_ptrs = (ctypes.POINTER(ctypes.c_double) * d)()
    idx_ptrs = (ctypes.POINTER(ctypes.c_int) * d)()
    keep = []
    num_per_col = (ctypes.c_int * d)()
    for c in range(d):
        vals = np.ascontiguousarray(X[:, c], dtype=np.float64)
        idxs = np.arange(n, dtype=np.int32)
        keep.append((vals, idxs))
        col_ptrs[c] = vals.ctypes.data_as(ctypes.POINTER(ctypes.c_double))
        idx_ptrs[c] = idxs.ctypes.data_as(ctypes.POINTER(ctypes.c_int))
        num_per_col[c] = n
    out = ctypes.c_void_p()
    _safe_call(_LIB.LGBM_DatasetCreateFromSampledColumn(
        col_ptrs, idx_ptrs, ctypes.c_int32(d), num_per_col, ctypes.c_int32(n),
        ctypes.c_int32(n), ctypes.c_int64(n), _c_str(params), ctypes.byref(out)))
    return out


def test_push_rows_roundtrip():
    rng = np.random.RandomState(0)
    n, d = 2000, 6
    X = rng.randn(n, d)
    y = (X[:, 0] > 0).astype(np.float32)
    handle = _sampled_column_dataset(X)
    _safe_call(_LIB.LGBM_DatasetInitStreaming(handle, 0, 0, 0, 1, 1, 1))
    # push in two chunks
    half = n // 2
    for start in (0, half):
        chunk = np.ascontiguousarray(X[start:start + half], dtype=np.float64)
        _safe_call(_LIB.LGBM_DatasetPushRows(
            handle, chunk.ctypes.data_as(ctypes.c_void_p), ctypes.c_int(1),
            ctypes.c_int32(half), ctypes.c_int32(d), ctypes.c_int32(start)))
    _safe_call(_LIB.LGBM_DatasetMarkFinished(handle))
    ds = lgb.Dataset(None)
    ds._handle = handle
    ds.set_label(y)
    bst = lgb.train({"objective": "binary", "verbosity": -1}, ds, 10)
    acc = ((bst.predict(X) > 0.5) == y).mean()
    assert acc > 0.9


def test_serialized_reference_roundtrip():
    rng = np.random.RandomState(1)
    n, d = 1000, 4
    X = rng.randn(n, d)
    base = lgb.Dataset(X, label=np.zeros(n, dtype=np.float32),
                       params={"max_bin": 31}).construct()
    out_len = ctypes.c_int64(0)
    _safe_call(_LIB.LGBM_DatasetSerializeReferenceToBinary(
        base._handle, ctypes.c_int64(0), ctypes.byref(out_len), None))
    buf = ctypes.create_string_buffer(out_len.value)
    _safe_call(_LIB.LGBM_DatasetSerializeReferenceToBinary(
        base._handle, out_len, ctypes.byref(out_len), buf))
    # rebuild an empty dataset from the serialized reference and push the same rows
    out = ctypes.c_void_p()
    _safe_call(_LIB.LGBM_DatasetCreateFromSerializedReference(
        buf, ctypes.c_int32(out_len.value), ctypes.c_int64(n), ctypes.c_int32(1),
        _c_str(""), ctypes.byref(out)))
    arr = np.ascontiguousarray(X, dtype=np.float64)
    _safe_call(_LIB.LGBM_DatasetPushRows(
        out, arr.ctypes.data_as(ctypes.c_void_p), ctypes.c_int(1), ctypes.c_int32(n),
        ctypes.c_int32(d), ctypes.c_int32(0)))
    nd = ctypes.c_int32(0)
    _safe_call(_LIB.LGBM_DatasetGetNumData(out, ctypes.byref(nd)))
    assert nd.value == n
    _safe_call(_LIB.LGBM_DatasetFree(out))


def test_push_rows_by_csr():
    import scipy.sparse as sp
    rng = np.random.RandomState(2)
    n, d = 1000, 8
    Xs = sp.random(n, d, density=0.4, random_state=2, format="csr")
    X = Xs.toarray()
    y = (X[:, 0] > 0.3).astype(np.float32)
    handle = _sampled_column_dataset(X)
    indptr = np.ascontiguousarray(Xs.indptr, dtype=np.int32)
    indices = np.ascontiguousarray(Xs.indices, dtype=np.int32)
    values = np.ascontiguousarray(Xs.data, dtype=np.float64)
    _safe_call(_LIB.LGBM_DatasetPushRowsByCSR(
        handle, indptr.ctypes.data_as(ctypes.c_void_p), ctypes.c_int(2),
        indices.ctypes.data_as(ctypes.POINTER(ctypes.c_int32)),
        values.ctypes.data_as(ctypes.c_void_p), ctypes.c_int(1),
        ctypes.c_int64(len(indptr)), ctypes.c_int64(len(values)),
        ctypes.c_int64(d), ctypes.c_int64(0)))
    ds = lgb.Dataset(None)
    ds._handle = handle
    ds.set_label(y)
    bst = lgb.train({"objective": "binary", "verbosity": -1}, ds, 10)
    assert ((bst.predict(X) > 0.5) == y).mean() > 0.85
