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
    # reference ABI: library-owned ByteBuffer + per-byte getter
    bb = ctypes.c_void_p()
    out_len = ctypes.c_int32(0)
    _safe_call(_LIB.LGBM_DatasetSerializeReferenceToBinary(
        base._handle, ctypes.byref(bb), ctypes.byref(out_len)))
    assert out_len.value > 0
    buf = ctypes.create_string_buffer(out_len.value)
    v = ctypes.c_uint8(0)
    for i in range(out_len.value):
        _safe_call(_LIB.LGBM_ByteBufferGetAt(bb, ctypes.c_int32(i), ctypes.byref(v)))
        buf[i] = v.value
    _safe_call(_LIB.LGBM_ByteBufferFree(bb))
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


def test_new_c_api_surface():
    """GetMaxThreads / SetLastError / ValidateFeatureNames / GetLoadedParam /
    PredictForCSC / PredictSparseOutput (reference c_api.h parity)."""
    import scipy.sparse as sp
    rng = np.random.RandomState(9)
    X = rng.rand(600, 5)
    y = (X[:, 0] + X[:, 1] > 1.0).astype(np.float64)
    bst = lgb.train({"objective": "binary", "verbosity": -1},
                    lgb.Dataset(X, label=y, feature_name=[f"f{i}" for i in range(5)]), 10)
    h = bst._handle

    out = ctypes.c_int(0)
    assert _LIB.LGBM_GetMaxThreads(ctypes.byref(out)) == 0 and out.value >= 1
    assert _LIB.LGBM_SetLastError(b"custom err") == 0
    assert _LIB.LGBM_GetLastError() is not None

    names = (ctypes.c_char_p * 5)(*[f"f{i}".encode() for i in range(5)])
    assert _LIB.LGBM_BoosterValidateFeatureNames(
        h, ctypes.cast(names, ctypes.POINTER(ctypes.c_char_p)), ctypes.c_int(5)) == 0
    bad = (ctypes.c_char_p * 5)(*[b"x"] * 5)
    assert _LIB.LGBM_BoosterValidateFeatureNames(
        h, ctypes.cast(bad, ctypes.POINTER(ctypes.c_char_p)), ctypes.c_int(5)) != 0

    buf = ctypes.create_string_buffer(1 << 16)
    out_len = ctypes.c_int64(0)
    assert _LIB.LGBM_BoosterGetLoadedParam(
        h, ctypes.c_int64(len(buf)), ctypes.byref(out_len), buf) == 0
    import json
    params = json.loads(buf.value.decode())
    assert params.get("objective") == "binary"

    # CSC predict matches dense predict
    Xc = sp.csc_matrix(X[:50])
    expected = bst.predict(X[:50])
    res = np.zeros(50)
    assert _LIB.LGBM_BoosterPredictForCSC(
        h, Xc.indptr.astype(np.int32).ctypes.data_as(ctypes.c_void_p), ctypes.c_int(2),
        Xc.indices.astype(np.int32).ctypes.data_as(ctypes.POINTER(ctypes.c_int32)),
        Xc.data.astype(np.float64).ctypes.data_as(ctypes.c_void_p), ctypes.c_int(1),
        ctypes.c_int64(len(Xc.indptr)), ctypes.c_int64(Xc.nnz), ctypes.c_int64(50),
        ctypes.c_int(0), ctypes.c_int(0), ctypes.c_int(-1), b"",
        ctypes.byref(out_len), res.ctypes.data_as(ctypes.POINTER(ctypes.c_double))) == 0
    np.testing.assert_allclose(res, expected, rtol=1e-12)

    # sparse SHAP output reconstructs the dense contribs
    Xr = sp.csr_matrix(X[:20])
    dense_contrib = bst.predict(X[:20], pred_contrib=True)
    o_indptr = ctypes.c_void_p()
    o_indices = ctypes.POINTER(ctypes.c_int32)()
    o_data = ctypes.c_void_p()
    assert _LIB.LGBM_BoosterPredictSparseOutput(
        h, Xr.indptr.astype(np.int32).ctypes.data_as(ctypes.c_void_p), ctypes.c_int(2),
        Xr.indices.astype(np.int32).ctypes.data_as(ctypes.POINTER(ctypes.c_int32)),
        Xr.data.astype(np.float64).ctypes.data_as(ctypes.c_void_p), ctypes.c_int(1),
        ctypes.c_int64(len(Xr.indptr)), ctypes.c_int64(Xr.nnz), ctypes.c_int64(5),
        ctypes.c_int(3), ctypes.c_int(0), ctypes.c_int(-1), b"", ctypes.c_int(0),
        ctypes.byref(out_len), ctypes.byref(o_indptr), ctypes.byref(o_indices),
        ctypes.byref(o_data)) == 0
    nnz = out_len.value
    indptr_arr = np.ctypeslib.as_array(ctypes.cast(o_indptr,
                                       ctypes.POINTER(ctypes.c_int64)), shape=(21,))
    idx_arr = np.ctypeslib.as_array(o_indices, shape=(max(nnz, 1),))
    val_arr = np.ctypeslib.as_array(ctypes.cast(o_data,
                                    ctypes.POINTER(ctypes.c_double)), shape=(max(nnz, 1),))
    recon = np.zeros((20, 6))
    for r in range(20):
        for k in range(indptr_arr[r], indptr_arr[r + 1]):
            recon[r, idx_arr[k]] = val_arr[k]
    np.testing.assert_allclose(recon, dense_contrib, rtol=1e-10, atol=1e-12)
    assert _LIB.LGBM_BoosterFreePredictSparse(o_indptr, o_indices, o_data,
                                              ctypes.c_int(3), ctypes.c_int(1)) == 0


def test_push_rows_by_csr_with_metadata():
    """Streaming CSR push with per-row label/weight/query metadata."""
    import scipy.sparse as sp
    rng = np.random.RandomState(4)
    n, d = 800, 6
    X = rng.rand(n, d)
    y = (X[:, 0] > 0.5).astype(np.float32)
    w = rng.rand(n).astype(np.float32) + 0.5
    qid = np.repeat(np.arange(40, dtype=np.int32), 20)
    handle = _sampled_column_dataset(X)
    Xs = sp.csr_matrix(X)
    half = n // 2
    for start, stop in ((0, half), (half, n)):
        sub = Xs[start:stop]
        indptr = np.ascontiguousarray(sub.indptr, dtype=np.int32)
        indices = np.ascontiguousarray(sub.indices, dtype=np.int32)
        vals = np.ascontiguousarray(sub.data, dtype=np.float64)
        rc = _LIB.LGBM_DatasetPushRowsByCSRWithMetadata(
            handle, indptr.ctypes.data_as(ctypes.c_void_p), ctypes.c_int(2),
            indices.ctypes.data_as(ctypes.POINTER(ctypes.c_int32)),
            vals.ctypes.data_as(ctypes.c_void_p), ctypes.c_int(1),
            ctypes.c_int64(len(indptr)), ctypes.c_int64(sub.nnz), ctypes.c_int64(start),
            y[start:stop].ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
            w[start:stop].ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
            None,
            qid[start:stop].ctypes.data_as(ctypes.POINTER(ctypes.c_int32)),
            ctypes.c_int32(0))
        assert rc == 0
    assert _LIB.LGBM_DatasetMarkFinished(handle) == 0
    ds = lgb.Dataset.__new__(lgb.Dataset)
    ds._handle = handle
    ds.params = {}
    ds._free_handle = True
    got_w = ds.get_field("weight")
    np.testing.assert_allclose(got_w, w, rtol=1e-6)
    grp = ds.get_field("group")
    assert grp is not None and len(grp) == 40 + 1 or len(grp) == 40
