"""ctypes bindings: Dataset / Booster over the LGBM_* C ABI of lib_migbm.so.

Capability parity target: reference python-package/lightgbm/basic.py (Dataset lazy
construction from numpy/pandas/scipy/file, Booster train/predict/save). Fresh
implementation sized to the implemented C API surface.
"""
import ctypes
import json
import os
from pathlib import Path

import numpy as np

from .compat import PANDAS_INSTALLED, SCIPY_INSTALLED, pd_DataFrame, pd_Series, scipy_sparse
from .libpath import find_lib_path

__all__ = ["Dataset", "Booster", "LightGBMError", "register_logger", "Sequence"]

_DTYPE_F32, _DTYPE_F64, _DTYPE_I32, _DTYPE_I64 = 0, 1, 2, 3

# Metrics whose C++ factor_to_bigger_better is -1 (higher value = better model);
# must stay in sync with cpp/src/metric.cpp. Used by early stopping.
_HIGHER_BETTER_METRICS = frozenset({
    "auc", "auc_mu", "average_precision", "ndcg", "map",
    "mean_average_precision", "r2",
})
_PREDICT_NORMAL, _PREDICT_RAW, _PREDICT_LEAF, _PREDICT_CONTRIB = 0, 1, 2, 3


class LightGBMError(Exception):
    """Error from the native library."""


def _load_lib():
    lib = ctypes.cdll.LoadLibrary(find_lib_path()[0])
    lib.LGBM_GetLastError.restype = ctypes.c_char_p
    return lib


_LIB = _load_lib()

_LOG_CALLBACK = None


def register_logger(logger, info_method_name="info", warning_method_name="warning"):
    """Redirect native logging into a Python logger."""
    global _LOG_CALLBACK
    cb_type = ctypes.CFUNCTYPE(None, ctypes.c_char_p)

    def _cb(msg):
        text = msg.decode("utf-8", errors="replace").rstrip("\n")
        if "[Warning]" in text:
            getattr(logger, warning_method_name)(text)
        else:
            getattr(logger, info_method_name)(text)

    _LOG_CALLBACK = cb_type(_cb)  # keep alive
    _safe_call(_LIB.LGBM_RegisterLogCallback(_LOG_CALLBACK))


def _safe_call(ret):
    if ret != 0:
        raise LightGBMError(_LIB.LGBM_GetLastError().decode("utf-8"))


def _c_str(string):
    return ctypes.c_char_p(str(string).encode("utf-8"))


def _param_dict_to_str(params):
    if not params:
        return ""
    pairs = []
    for k, v in params.items():
        if isinstance(v, (list, tuple, set)):
            pairs.append(f"{k}={','.join(map(str, v))}")
        elif isinstance(v, bool):
            pairs.append(f"{k}={'true' if v else 'false'}")
        elif v is None:
            continue
        else:
            pairs.append(f"{k}={v}")
    return " ".join(pairs)


def _pandas_to_float64(df, pandas_categorical=None):
    """DataFrame -> float64 matrix, mapping category dtypes to their codes
    (reference _data_from_pandas semantics). When `pandas_categorical` (a list of
    category-value lists, in categorical-column order) is given, codes are aligned
    to it — the training-time mapping — otherwise the mapping is recorded and
    returned. Returns (array, pandas_categorical)."""
    import pandas as pd
    bad = [str(c) for c in df.columns if str(df[c].dtype) == "object"]
    if bad:
        raise LightGBMError(
            "DataFrame.dtypes must be int, float, bool or category; found object "
            f"columns {bad} — cast them with .astype('category') first")
    cat_cols = [c for c in df.columns if str(df[c].dtype) == "category"]
    recorded = pandas_categorical is not None
    if pandas_categorical is None:
        pandas_categorical = [list(df[c].cat.categories) for c in cat_cols]
    # nullable extension dtypes (Int64/Float64/boolean): pd.NA -> NaN column-wise
    ext_cols = [c for c in df.columns
                if c not in cat_cols and hasattr(df[c].dtype, "na_value")]
    if ext_cols:
        df = df.copy()
        for c in ext_cols:
            df[c] = df[c].to_numpy(dtype=np.float64, na_value=np.nan)
    if not cat_cols:
        return np.ascontiguousarray(df.to_numpy(), dtype=np.float64), pandas_categorical
    df = df.copy()
    for i, c in enumerate(cat_cols):
        cats = pandas_categorical[i] if i < len(pandas_categorical) else \
            list(df[c].cat.categories)
        codes = pd.Categorical(df[c], categories=cats).codes.astype(np.float64)
        codes[codes < 0] = np.nan  # unseen category / NaN -> missing
        df[c] = codes
    _ = recorded
    return np.ascontiguousarray(df.to_numpy(dtype=np.float64)), pandas_categorical


def _to_2d_float64(data, pandas_categorical=None):
    if PANDAS_INSTALLED and isinstance(data, pd_DataFrame):
        arr, _ = _pandas_to_float64(data, pandas_categorical)
        if arr.ndim == 1:
            arr = arr.reshape(1, -1)
        return arr
    arr = np.asarray(data)
    if arr.ndim == 1:
        arr = arr.reshape(1, -1)
    return np.ascontiguousarray(arr, dtype=np.float64)


def _np_float32(data):
    if PANDAS_INSTALLED and isinstance(data, (pd_Series, pd_DataFrame)):
        data = data.values
    return np.ascontiguousarray(np.asarray(data).ravel(), dtype=np.float32)


class Sequence:
    """Generic random-access data source (parity: reference basic.py Sequence ABC).

    Subclass and implement __getitem__ (row -> 1D numpy array) and __len__; pass
    instances (or a list of them) as Dataset data. batch_size controls the chunk
    size used while streaming rows into the dataset.
    """

    batch_size = 4096

    def __getitem__(self, idx):
        raise NotImplementedError("Sequence subclasses must implement __getitem__")

    def __len__(self):
        raise NotImplementedError("Sequence subclasses must implement __len__")


def _is_sequence_input(data):
    if isinstance(data, Sequence):
        return True
    return isinstance(data, list) and len(data) > 0 and all(
        isinstance(s, Sequence) for s in data)


def _create_dataset_from_seqs(seqs, param_str, ref_handle):
    if isinstance(seqs, Sequence):
        seqs = [seqs]
    total = sum(len(s) for s in seqs)
    first = np.asarray(seqs[0][0], dtype=np.float64).ravel()
    ncol = len(first)
    # sample rows for binning (bounded), then create + push in batches
    sample_rows = []
    stride = max(1, total // 200_000)
    for s in seqs:
        for i in range(0, len(s), stride):
            sample_rows.append(np.asarray(s[i], dtype=np.float64).ravel())
    sample = np.vstack(sample_rows)
    out = ctypes.c_void_p()
    if ref_handle is not None:
        raise LightGBMError("Sequence input with a reference dataset is not supported yet")
    # build mappers from the sample via the sampled-column API
    col_ptrs = (ctypes.POINTER(ctypes.c_double) * ncol)()
    idx_ptrs = (ctypes.POINTER(ctypes.c_int) * ncol)()
    keep = []
    nper = (ctypes.c_int * ncol)()
    n_s = sample.shape[0]
    for c in range(ncol):
        vals = np.ascontiguousarray(sample[:, c])
        idxs = np.arange(n_s, dtype=np.int32)
        keep.append((vals, idxs))
        col_ptrs[c] = vals.ctypes.data_as(ctypes.POINTER(ctypes.c_double))
        idx_ptrs[c] = idxs.ctypes.data_as(ctypes.POINTER(ctypes.c_int))
        nper[c] = n_s
    _safe_call(_LIB.LGBM_DatasetCreateFromSampledColumn(
        col_ptrs, idx_ptrs, ctypes.c_int32(ncol), nper, ctypes.c_int32(n_s),
        ctypes.c_int32(total), ctypes.c_int64(total), _c_str(param_str),
        ctypes.byref(out)))
    # push rows in batches
    start = 0
    for s in seqs:
        bs = getattr(s, "batch_size", 4096) or 4096
        for b0 in range(0, len(s), bs):
            rows = [np.asarray(s[i], dtype=np.float64).ravel()
                    for i in range(b0, min(b0 + bs, len(s)))]
            chunk = np.ascontiguousarray(np.vstack(rows))
            _safe_call(_LIB.LGBM_DatasetPushRows(
                out, chunk.ctypes.data_as(ctypes.c_void_p), ctypes.c_int(_DTYPE_F64),
                ctypes.c_int32(chunk.shape[0]), ctypes.c_int32(ncol),
                ctypes.c_int32(start)))
            start += chunk.shape[0]
    _safe_call(_LIB.LGBM_DatasetMarkFinished(out))
    return out


def _is_pyarrow_table(data):
    try:
        import pyarrow as pa
        return isinstance(data, pa.Table)
    except ImportError:
        return False


def _export_arrow_table(table):
    """Export a pyarrow Table over the Arrow C data interface.
    Returns (n_chunks, chunks_array, schema) ctypes objects."""
    import pyarrow as pa  # noqa: F401

    class _ArrowArray(ctypes.Structure):
        _fields_ = [("length", ctypes.c_int64), ("null_count", ctypes.c_int64),
                    ("offset", ctypes.c_int64), ("n_buffers", ctypes.c_int64),
                    ("n_children", ctypes.c_int64), ("buffers", ctypes.c_void_p),
                    ("children", ctypes.c_void_p), ("dictionary", ctypes.c_void_p),
                    ("release", ctypes.c_void_p), ("private_data", ctypes.c_void_p)]

    class _ArrowSchema(ctypes.Structure):
        _fields_ = [("format", ctypes.c_char_p), ("name", ctypes.c_char_p),
                    ("metadata", ctypes.c_char_p), ("flags", ctypes.c_int64),
                    ("n_children", ctypes.c_int64), ("children", ctypes.c_void_p),
                    ("dictionary", ctypes.c_void_p), ("release", ctypes.c_void_p),
                    ("private_data", ctypes.c_void_p)]

    batches = table.to_batches()
    if not batches:
        raise ValueError("empty pyarrow Table")
    chunks = (_ArrowArray * len(batches))()
    schema = _ArrowSchema()
    for i, batch in enumerate(batches):
        if i == 0:
            batch._export_to_c(ctypes.addressof(chunks[i]), ctypes.addressof(schema))
        else:
            batch._export_to_c(ctypes.addressof(chunks[i]))
    return len(batches), chunks, schema


def _create_dataset_from_arrow(table, param_str, ref_handle):
    """Export record batches via the Arrow C data interface and build the dataset."""
    import pyarrow as pa  # noqa: F401

    class _ArrowArray(ctypes.Structure):
        _fields_ = [("length", ctypes.c_int64), ("null_count", ctypes.c_int64),
                    ("offset", ctypes.c_int64), ("n_buffers", ctypes.c_int64),
                    ("n_children", ctypes.c_int64), ("buffers", ctypes.c_void_p),
                    ("children", ctypes.c_void_p), ("dictionary", ctypes.c_void_p),
                    ("release", ctypes.c_void_p), ("private_data", ctypes.c_void_p)]

    class _ArrowSchema(ctypes.Structure):
        _fields_ = [("format", ctypes.c_char_p), ("name", ctypes.c_char_p),
                    ("metadata", ctypes.c_char_p), ("flags", ctypes.c_int64),
                    ("n_children", ctypes.c_int64), ("children", ctypes.c_void_p),
                    ("dictionary", ctypes.c_void_p), ("release", ctypes.c_void_p),
                    ("private_data", ctypes.c_void_p)]

    n_chunks, chunks, schema = _export_arrow_table(table)
    out = ctypes.c_void_p()
    _safe_call(_LIB.LGBM_DatasetCreateFromArrow(
        ctypes.c_int64(n_chunks), chunks, ctypes.byref(schema), _c_str(param_str),
        ref_handle, ctypes.byref(out)))
    return out


class Dataset:
    """Binned training dataset (parity: reference lgb.Dataset)."""

    def __init__(self, data, label=None, reference=None, weight=None, group=None,
                 init_score=None, feature_name="auto", categorical_feature="auto",
                 params=None, free_raw_data=True, position=None):
        self.data = data
        self.label = label
        self.reference = reference
        self.weight = weight
        self.group = group
        self.init_score = init_score
        self.position = position
        self.feature_name = feature_name
        self.categorical_feature = categorical_feature
        self.params = dict(params) if params else {}
        self.free_raw_data = free_raw_data
        self.pandas_categorical = None
        self._handle = None
        self.used_indices = None
        self._predictor = None

    # ------------------------------------------------------------ construction
    @property
    def handle(self):
        return self._handle

    def construct(self):
        if self._handle is not None:
            return self
        if self.reference is not None:
            self.reference.construct()
        params = dict(self.params)
        cat = self._resolve_categorical(params)
        if cat:
            params["categorical_feature"] = ",".join(str(c) for c in cat)
        param_str = _param_dict_to_str(params)
        ref_handle = self.reference._handle if self.reference is not None else None

        if _is_pyarrow_table(self.data):
            self._handle = _create_dataset_from_arrow(self.data, param_str, ref_handle)
        elif _is_sequence_input(self.data):
            self._handle = _create_dataset_from_seqs(self.data, param_str, ref_handle)
        elif isinstance(self.data, (str, Path)):
            out = ctypes.c_void_p()
            _safe_call(_LIB.LGBM_DatasetCreateFromFile(
                _c_str(str(self.data)), _c_str(param_str), ref_handle, ctypes.byref(out)))
            self._handle = out
        elif SCIPY_INSTALLED and scipy_sparse is not None and scipy_sparse.issparse(self.data):
            csr = self.data.tocsr()
            out = ctypes.c_void_p()
            indptr = np.ascontiguousarray(csr.indptr, dtype=np.int32)
            indices = np.ascontiguousarray(csr.indices, dtype=np.int32)
            values = np.ascontiguousarray(csr.data, dtype=np.float64)
            _safe_call(_LIB.LGBM_DatasetCreateFromCSR(
                indptr.ctypes.data_as(ctypes.c_void_p), ctypes.c_int(_DTYPE_I32),
                indices.ctypes.data_as(ctypes.POINTER(ctypes.c_int32)),
                values.ctypes.data_as(ctypes.c_void_p), ctypes.c_int(_DTYPE_F64),
                ctypes.c_int64(len(indptr)), ctypes.c_int64(len(values)),
                ctypes.c_int64(csr.shape[1]), _c_str(param_str), ref_handle,
                ctypes.byref(out)))
            self._handle = out
        else:
            if PANDAS_INSTALLED and isinstance(self.data, pd_DataFrame):
                # a valid set aligns its category codes to the reference's mapping
                ref_cats = getattr(self.reference, "pandas_categorical", None) \
                    if self.reference is not None else None
                arr, self.pandas_categorical = _pandas_to_float64(self.data, ref_cats)
            else:
                arr = _to_2d_float64(self.data)
            out = ctypes.c_void_p()
            _safe_call(_LIB.LGBM_DatasetCreateFromMat(
                arr.ctypes.data_as(ctypes.c_void_p), ctypes.c_int(_DTYPE_F64),
                ctypes.c_int32(arr.shape[0]), ctypes.c_int32(arr.shape[1]),
                ctypes.c_int(1), _c_str(param_str), ref_handle, ctypes.byref(out)))
            self._handle = out

        # fields
        if self.label is not None:
            self.set_label(self.label)
        if self.weight is not None:
            self.set_weight(self.weight)
        if self.group is not None:
            self.set_group(self.group)
        if self.init_score is not None:
            self.set_init_score(self.init_score)
        if self.position is not None:
            self.set_position(self.position)
        # feature names
        names = self._resolve_feature_names()
        if names is not None:
            arr_names = (ctypes.c_char_p * len(names))(*[n.encode("utf-8") for n in names])
            _safe_call(_LIB.LGBM_DatasetSetFeatureNames(
                self._handle, arr_names, ctypes.c_int(len(names))))
        if self.free_raw_data:
            self.data = None
        return self

    def _resolve_feature_names(self):
        if isinstance(self.feature_name, (list, tuple)):
            return list(self.feature_name)
        if self.feature_name == "auto" and PANDAS_INSTALLED and isinstance(self.data, pd_DataFrame):
            return [str(c) for c in self.data.columns]
        return None

    def _resolve_categorical(self, params):
        cf = self.categorical_feature
        if cf is None or cf == "auto":
            if PANDAS_INSTALLED and isinstance(self.data, pd_DataFrame):
                cats = [i for i, d in enumerate(self.data.dtypes)
                        if str(d) in ("category", "object")]
                return cats
            return []
        out = []
        names = self._resolve_feature_names()
        for c in cf:
            if isinstance(c, str):
                if names and c in names:
                    out.append(names.index(c))
            else:
                out.append(int(c))
        return out

    def __del__(self):
        try:
            if self._handle is not None:
                _safe_call(_LIB.LGBM_DatasetFree(self._handle))
                self._handle = None
        except Exception:
            pass

    # ------------------------------------------------------------ fields
    def _set_float_field(self, name, data):
        self.construct()
        arr = _np_float32(data)
        _safe_call(_LIB.LGBM_DatasetSetField(
            self._handle, _c_str(name), arr.ctypes.data_as(ctypes.c_void_p),
            ctypes.c_int(len(arr)), ctypes.c_int(_DTYPE_F32)))

    def set_field(self, field_name, data):
        """Generic field setter (reference parity). data=None clears the field."""
        if data is None and field_name != "label":
            _safe_call(_LIB.LGBM_DatasetSetField(
                self._handle, field_name.encode("utf-8"), None,
                ctypes.c_int32(0), ctypes.c_int(0)))
            return self
        if field_name in ("label", "weight"):
            self._set_float_field(field_name, data)
        elif field_name == "group":
            self.set_group(data)
        elif field_name == "init_score":
            self.set_init_score(data)
        elif field_name == "position":
            self.set_position(data)
        else:
            raise LightGBMError(f"Unknown field {field_name}")
        return self

    def get_position(self):
        return self.get_field("position")

    def get_data(self):
        """Return the raw data this Dataset was built from (if still referenced)."""
        if getattr(self, "free_raw_data", True) and self._handle is not None \
                and self.data is None:
            raise LightGBMError("Raw data was freed (construct with free_raw_data=False)")
        return self.data

    def get_params(self):
        return dict(self.params)

    def set_feature_name(self, feature_name):
        self.feature_name = list(feature_name)
        return self

    def set_categorical_feature(self, categorical_feature):
        if self._handle is not None:
            raise LightGBMError(
                "set_categorical_feature must be called before construct()")
        self.categorical_feature = categorical_feature
        return self

    def set_reference(self, reference):
        if self._handle is not None:
            raise LightGBMError("set_reference must be called before construct()")
        self.reference = reference
        return self

    def get_ref_chain(self, ref_limit=100):
        chain = []
        node = self
        while node is not None and len(chain) < ref_limit:
            chain.append(node)
            node = getattr(node, "reference", None)
        return set(chain)

    def set_label(self, label):
        self.label = label
        if self._handle is not None:
            self._set_float_field("label", label)
        return self

    def set_weight(self, weight):
        self.weight = weight
        if self._handle is not None and weight is not None:
            self._set_float_field("weight", weight)
        return self

    def set_group(self, group):
        self.group = group
        if self._handle is not None and group is not None:
            arr = np.ascontiguousarray(np.asarray(group).ravel(), dtype=np.int32)
            _safe_call(_LIB.LGBM_DatasetSetField(
                self._handle, _c_str("group"), arr.ctypes.data_as(ctypes.c_void_p),
                ctypes.c_int(len(arr)), ctypes.c_int(_DTYPE_I32)))
        return self

    def set_position(self, position):
        self.position = position
        if self._handle is not None and position is not None:
            arr = np.ascontiguousarray(np.asarray(position).ravel(), dtype=np.int32)
            _safe_call(_LIB.LGBM_DatasetSetField(
                self._handle, _c_str("position"), arr.ctypes.data_as(ctypes.c_void_p),
                ctypes.c_int(len(arr)), ctypes.c_int(_DTYPE_I32)))
        return self

    def set_init_score(self, init_score):
        self.init_score = init_score
        if self._handle is not None and init_score is not None:
            a = np.asarray(init_score)
            # 2D (row, class) flattens class-major: the engine stores scores as
            # num_data*num_class with class as the outer index (reference order="F")
            arr = np.ascontiguousarray(
                a.ravel(order="F") if a.ndim == 2 else a.ravel(), dtype=np.float64)
            _safe_call(_LIB.LGBM_DatasetSetField(
                self._handle, _c_str("init_score"), arr.ctypes.data_as(ctypes.c_void_p),
                ctypes.c_int(len(arr)), ctypes.c_int(_DTYPE_F64)))
        return self

    def get_field(self, name):
        self.construct()
        out_len = ctypes.c_int(0)
        out_ptr = ctypes.c_void_p()
        out_type = ctypes.c_int(0)
        _safe_call(_LIB.LGBM_DatasetGetField(
            self._handle, _c_str(name), ctypes.byref(out_len), ctypes.byref(out_ptr),
            ctypes.byref(out_type)))
        if not out_ptr.value or out_len.value == 0:
            return None
        n = out_len.value
        if out_type.value == _DTYPE_F32:
            return np.ctypeslib.as_array(ctypes.cast(out_ptr, ctypes.POINTER(ctypes.c_float)),
                                         shape=(n,)).copy()
        if out_type.value == _DTYPE_F64:
            arr = np.ctypeslib.as_array(ctypes.cast(out_ptr, ctypes.POINTER(ctypes.c_double)),
                                        shape=(n,)).copy()
            if name == "init_score":
                nrow = self.num_data()
                if nrow and n > nrow and n % nrow == 0:
                    # stored class-major; surface as (row, class) like the reference
                    arr = arr.reshape(n // nrow, nrow).T
            return arr
        return np.ctypeslib.as_array(ctypes.cast(out_ptr, ctypes.POINTER(ctypes.c_int32)),
                                     shape=(n,)).copy()

    def get_label(self):
        return self.get_field("label")

    def get_weight(self):
        return self.get_field("weight")

    def get_group(self):
        b = self.get_field("group")
        if b is None:
            return None
        return np.diff(b)

    def get_init_score(self):
        return self.get_field("init_score")

    def num_data(self):
        self.construct()
        out = ctypes.c_int32(0)
        _safe_call(_LIB.LGBM_DatasetGetNumData(self._handle, ctypes.byref(out)))
        return out.value

    def num_feature(self):
        self.construct()
        out = ctypes.c_int32(0)
        _safe_call(_LIB.LGBM_DatasetGetNumFeature(self._handle, ctypes.byref(out)))
        return out.value

    def add_features_from(self, other):
        """Add features of `other` Dataset to this one (both must be constructed
        and have the same number of rows). Reference parity: Dataset.add_features_from."""
        self.construct()
        other.construct()
        _safe_call(_LIB.LGBM_DatasetAddFeaturesFrom(self._handle, other._handle))
        if isinstance(self.feature_name, list) and isinstance(other.feature_name, list):
            self.feature_name = self.feature_name + other.feature_name
        return self

    def feature_num_bin(self, feature):
        self.construct()
        out = ctypes.c_int32(0)
        _safe_call(_LIB.LGBM_DatasetGetFeatureNumBin(
            self._handle, ctypes.c_int(feature), ctypes.byref(out)))
        return out.value

    def save_binary(self, filename):
        self.construct()
        _safe_call(_LIB.LGBM_DatasetSaveBinary(self._handle, _c_str(str(filename))))
        return self

    def subset(self, used_indices, params=None):
        self.construct()
        # sorted order is required downstream (query-boundary slicing in
        # Dataset::Subset counts rows per query in index order)
        used = np.sort(np.asarray(used_indices).ravel()).astype(np.int32)
        used = np.ascontiguousarray(used)
        out = ctypes.c_void_p()
        _safe_call(_LIB.LGBM_DatasetGetSubset(
            self._handle, used.ctypes.data_as(ctypes.POINTER(ctypes.c_int32)),
            ctypes.c_int32(len(used)), _c_str(_param_dict_to_str(params or {})),
            ctypes.byref(out)))
        ds = Dataset(None, params=params or dict(self.params))
        ds._handle = out
        ds.used_indices = used
        ds.reference = self
        return ds

    def create_valid(self, data, label=None, weight=None, group=None, init_score=None,
                     params=None, position=None):
        return Dataset(data, label=label, reference=self, weight=weight, group=group,
                       init_score=init_score, params=params or dict(self.params),
                       position=position)

    def get_feature_name(self):
        self.construct()
        return _get_string_buffer(_LIB.LGBM_DatasetGetFeatureNames, self._handle)


def _get_string_buffer(fn, handle):
    n = ctypes.c_int(0)
    buf_len = ctypes.c_size_t(0)
    # probe pass
    dummy = (ctypes.c_char_p * 1)(ctypes.addressof(ctypes.create_string_buffer(1)))
    fn(handle, ctypes.c_int(0), ctypes.byref(n), ctypes.c_size_t(0),
       ctypes.byref(buf_len), dummy)
    if n.value == 0:
        return []
    bufs = [ctypes.create_string_buffer(buf_len.value) for _ in range(n.value)]
    arr = (ctypes.c_char_p * n.value)(*[ctypes.addressof(b) for b in bufs])
    out_n = ctypes.c_int(0)
    out_len = ctypes.c_size_t(0)
    _safe_call(fn(handle, ctypes.c_int(n.value), ctypes.byref(out_n), buf_len,
                  ctypes.byref(out_len), arr))
    return [b.value.decode("utf-8") for b in bufs]


class Booster:
    """Gradient-boosting model handle (parity: reference lgb.Booster)."""

    def __init__(self, params=None, train_set=None, model_file=None, model_str=None):
        self.params = dict(params) if params else {}
        self._handle = None
        self._train_set = None
        self._valid_sets = []
        self.best_iteration = -1
        self.best_score = {}
        self._name_valid_sets = []
        self._network_initialized = False
        self.pandas_categorical = None
        if train_set is not None:
            if not isinstance(train_set, Dataset):
                raise TypeError("train_set must be a Dataset")
            train_set.construct()
            out = ctypes.c_void_p()
            _safe_call(_LIB.LGBM_BoosterCreate(
                train_set._handle, _c_str(_param_dict_to_str(self.params)),
                ctypes.byref(out)))
            self._handle = out
            self._train_set = train_set
            self.pandas_categorical = getattr(train_set, "pandas_categorical", None)
        elif model_file is not None:
            out = ctypes.c_void_p()
            out_iters = ctypes.c_int(0)
            _safe_call(_LIB.LGBM_BoosterCreateFromModelfile(
                _c_str(str(model_file)), ctypes.byref(out_iters), ctypes.byref(out)))
            self._handle = out
            self.best_iteration = -1
            # trailing pandas_categorical line (reference format)
            try:
                with open(model_file, "rb") as fh:
                    tail = fh.read()[-65536:].decode("utf-8", errors="replace")
                for line in reversed(tail.splitlines()):
                    if line.startswith("pandas_categorical:"):
                        self.pandas_categorical = json.loads(line[len("pandas_categorical:"):])
                        break
            except (OSError, ValueError):
                pass
        elif model_str is not None:
            self.model_from_string(model_str)
        else:
            raise TypeError("Need train_set, model_file or model_str")
        if train_set is None:
            self._load_params_from_model()

    def _load_params_from_model(self):
        """Populate self.params from the loaded model's parameters block
        (reference LGBM_BoosterGetLoadedParam semantics)."""
        try:
            out_len = ctypes.c_int64(0)
            _safe_call(_LIB.LGBM_BoosterGetLoadedParam(
                self._handle, ctypes.c_int64(0), ctypes.byref(out_len), None))
            buf = ctypes.create_string_buffer(out_len.value + 1)
            _safe_call(_LIB.LGBM_BoosterGetLoadedParam(
                self._handle, ctypes.c_int64(len(buf)), ctypes.byref(out_len), buf))
            loaded = json.loads(buf.value.decode("utf-8"))
            for k, v in loaded.items():
                self.params.setdefault(k, v)
        except (LightGBMError, ValueError):
            pass

    @property
    def handle(self):
        return self._handle

    def __getstate__(self):
        """Pickle as the model text (reference Booster pickling semantics):
        native handle and dataset references are dropped, the model is
        reconstructed from its string form on unpickle."""
        state = self.__dict__.copy()
        if state.get("_handle") is not None:
            state["_model_str"] = self.model_to_string(num_iteration=-1)
        state["_handle"] = None
        state["_train_set"] = None
        state["_valid_sets"] = []
        state["_kept_refs"] = []
        return state

    def __setstate__(self, state):
        model_str = state.pop("_model_str", None)
        self.__dict__.update(state)
        if model_str is not None:
            self.model_from_string(model_str)

    def __del__(self):
        try:
            if self._handle is not None:
                _safe_call(_LIB.LGBM_BoosterFree(self._handle))
                self._handle = None
        except Exception:
            pass

    # ------------------------------------------------------------ training
    def add_valid(self, data, name):
        data.construct()
        _safe_call(_LIB.LGBM_BoosterAddValidData(self._handle, data._handle))
        self._valid_sets.append(data)
        self._name_valid_sets.append(name)
        return self

    def reset_parameter(self, params):
        self.params.update(params)
        _safe_call(_LIB.LGBM_BoosterResetParameter(
            self._handle, _c_str(_param_dict_to_str(params))))
        return self

    def update(self, train_set=None, fobj=None):
        if train_set is not None and train_set is not self._train_set:
            train_set.construct()
            _safe_call(_LIB.LGBM_BoosterResetTrainingData(self._handle, train_set._handle))
            self._train_set = train_set
        if fobj is None:
            is_finished = ctypes.c_int(0)
            _safe_call(_LIB.LGBM_BoosterUpdateOneIter(self._handle, ctypes.byref(is_finished)))
            return is_finished.value == 1
        grad, hess = fobj(self.__inner_predict(0), self._train_set)
        return self.__boost(grad, hess)

    def __boost(self, grad, hess):
        grad = np.ascontiguousarray(np.asarray(grad).ravel(order="F"), dtype=np.float32)
        hess = np.ascontiguousarray(np.asarray(hess).ravel(order="F"), dtype=np.float32)
        is_finished = ctypes.c_int(0)
        _safe_call(_LIB.LGBM_BoosterUpdateOneIterCustom(
            self._handle, grad.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
            hess.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
            ctypes.byref(is_finished)))
        return is_finished.value == 1

    def rollback_one_iter(self):
        _safe_call(_LIB.LGBM_BoosterRollbackOneIter(self._handle))
        return self

    def current_iteration(self):
        out = ctypes.c_int(0)
        _safe_call(_LIB.LGBM_BoosterGetCurrentIteration(self._handle, ctypes.byref(out)))
        return out.value

    def num_model_per_iteration(self):
        out = ctypes.c_int(0)
        _safe_call(_LIB.LGBM_BoosterNumModelPerIteration(self._handle, ctypes.byref(out)))
        return out.value

    def num_trees(self):
        out = ctypes.c_int(0)
        _safe_call(_LIB.LGBM_BoosterNumberOfTotalModel(self._handle, ctypes.byref(out)))
        return out.value

    def num_feature(self):
        out = ctypes.c_int(0)
        _safe_call(_LIB.LGBM_BoosterGetNumFeature(self._handle, ctypes.byref(out)))
        return out.value

    def feature_name(self):
        return _get_string_buffer(_LIB.LGBM_BoosterGetFeatureNames, self._handle)

    def upper_bound(self):
        out = ctypes.c_double(0)
        _safe_call(_LIB.LGBM_BoosterGetUpperBoundValue(self._handle, ctypes.byref(out)))
        return out.value

    def lower_bound(self):
        out = ctypes.c_double(0)
        _safe_call(_LIB.LGBM_BoosterGetLowerBoundValue(self._handle, ctypes.byref(out)))
        return out.value

    # ------------------------------------------------------------ evaluation
    def _eval_names(self):
        return _get_string_buffer(_LIB.LGBM_BoosterGetEvalNames, self._handle)

    def num_data(self):
        if self._train_set is not None:
            return self._train_set.num_data()
        raise LightGBMError("num_data requires a training dataset")

    def set_attr(self, **kwargs):
        """Store free-form string attributes on the Booster (reference parity)."""
        attrs = getattr(self, "_attrs", {})
        for k, v in kwargs.items():
            if v is None:
                attrs.pop(k, None)
            else:
                attrs[k] = str(v)
        self._attrs = attrs
        return self

    def attr(self, key):
        return getattr(self, "_attrs", {}).get(key)

    def set_train_data_name(self, name):
        self._train_data_name = name
        return self

    def shuffle_models(self, start_iteration=0, end_iteration=-1):
        _safe_call(_LIB.LGBM_BoosterShuffleModels(
            self._handle, ctypes.c_int(start_iteration), ctypes.c_int(end_iteration)))
        return self

    def get_leaf_output(self, tree_id, leaf_id):
        out = ctypes.c_double(0.0)
        _safe_call(_LIB.LGBM_BoosterGetLeafValue(
            self._handle, ctypes.c_int(tree_id), ctypes.c_int(leaf_id), ctypes.byref(out)))
        return out.value

    def set_leaf_output(self, tree_id, leaf_id, value):
        _safe_call(_LIB.LGBM_BoosterSetLeafValue(
            self._handle, ctypes.c_int(tree_id), ctypes.c_int(leaf_id),
            ctypes.c_double(value)))
        return self

    def get_split_value_histogram(self, feature, bins=None):
        """Histogram of split threshold values used for `feature` across the model."""
        d = self.dump_model()
        values = []

        def walk(node):
            if "leaf_index" in node:
                return
            f = node.get("split_feature")
            name = self.feature_name()[f] if isinstance(f, int) else f
            if f == feature or name == feature:
                values.append(node["threshold"])
            walk(node["left_child"])
            walk(node["right_child"])
        for t in d["tree_info"]:
            walk(t["tree_structure"])
        values = np.asarray(values, dtype=np.float64)
        if bins is None:
            bins = max(1, min(len(values), 32))
        return np.histogram(values, bins=bins)

    def eval_train(self, feval=None):
        return self.__inner_eval("training", 0, feval)

    def eval_valid(self, feval=None):
        out = []
        for i in range(len(self._valid_sets)):
            out.extend(self.__inner_eval(self._name_valid_sets[i], i + 1, feval))
        return out

    def eval(self, data, name, feval=None):
        idx = None
        for i, v in enumerate(self._valid_sets):
            if v is data:
                idx = i + 1
        if idx is None and data is self._train_set:
            idx = 0
        if idx is None:
            raise ValueError("Data must be added with add_valid first")
        return self.__inner_eval(name, idx, feval)

    def __inner_eval(self, name, data_idx, feval=None):
        names = self._eval_names()
        out = []
        if names:
            res = np.zeros(len(names), dtype=np.float64)
            out_len = ctypes.c_int(0)
            _safe_call(_LIB.LGBM_BoosterGetEval(
                self._handle, ctypes.c_int(data_idx), ctypes.byref(out_len),
                res.ctypes.data_as(ctypes.POINTER(ctypes.c_double))))
            higher_better = [n in _HIGHER_BETTER_METRICS or
                             n.startswith(("auc", "ndcg@", "map@")) for n in names]
            for i in range(out_len.value):
                out.append((name, names[i], res[i], higher_better[i]))
        if feval is not None:
            ds = self._train_set if data_idx == 0 else self._valid_sets[data_idx - 1]
            preds = self.__inner_predict(data_idx)
            for fe_fn in (feval if isinstance(feval, (list, tuple)) else [feval]):
                fe = fe_fn(preds, ds)
                if isinstance(fe, list):
                    for (fn, fv, fb) in fe:
                        out.append((name, fn, fv, fb))
                else:
                    fn, fv, fb = fe
                    out.append((name, fn, fv, fb))
        return out

    def __inner_predict(self, data_idx):
        n = ctypes.c_int64(0)
        _safe_call(_LIB.LGBM_BoosterGetNumPredict(self._handle, ctypes.c_int(data_idx),
                                                  ctypes.byref(n)))
        res = np.zeros(n.value, dtype=np.float64)
        out_len = ctypes.c_int64(0)
        _safe_call(_LIB.LGBM_BoosterGetPredict(
            self._handle, ctypes.c_int(data_idx), ctypes.byref(out_len),
            res.ctypes.data_as(ctypes.POINTER(ctypes.c_double))))
        return res

    # ------------------------------------------------------------ prediction
    def predict(self, data, start_iteration=0, num_iteration=None, raw_score=False,
                pred_leaf=False, pred_contrib=False, validate_features=False, **kwargs):
        if num_iteration is None:
            num_iteration = self.best_iteration if self.best_iteration > 0 else -1
        if validate_features and PANDAS_INSTALLED and isinstance(data, pd_DataFrame):
            model_names = self.feature_name()
            data_names = [str(c) for c in data.columns]
            if data_names != model_names:
                raise LightGBMError(
                    f"Feature names mismatch: model expects {model_names}, "
                    f"data has {data_names}")
        pred_param = _param_dict_to_str(
            {k: v for k, v in kwargs.items()
             if k in ("pred_early_stop", "pred_early_stop_freq",
                      "pred_early_stop_margin", "predict_disable_shape_check")})
        if not kwargs.get("predict_disable_shape_check", False):
            ncol = getattr(data, "shape", (0, 0))[1] if hasattr(data, "shape") and \
                len(getattr(data, "shape", ())) == 2 else None
            if ncol is not None and ncol < self.num_feature():
                raise LightGBMError(
                    f"The number of features in data ({ncol}) is not the same as it "
                    f"was in training data ({self.num_feature()}).\n"
                    "You can set ``predict_disable_shape_check=true`` to discard "
                    "this error, but please be aware what you are doing.")
        ptype = _PREDICT_NORMAL
        if raw_score:
            ptype = _PREDICT_RAW
        if pred_leaf:
            ptype = _PREDICT_LEAF
        if pred_contrib:
            ptype = _PREDICT_CONTRIB
        if SCIPY_INSTALLED and scipy_sparse is not None and scipy_sparse.issparse(data):
            csr = data.tocsr()
            nrow = csr.shape[0]
            n = ctypes.c_int64(0)
            _safe_call(_LIB.LGBM_BoosterCalcNumPredict(
                self._handle, ctypes.c_int(nrow), ctypes.c_int(ptype),
                ctypes.c_int(start_iteration), ctypes.c_int(num_iteration),
                ctypes.byref(n)))
            res = np.zeros(n.value, dtype=np.float64)
            out_len = ctypes.c_int64(0)
            indptr = np.ascontiguousarray(csr.indptr, dtype=np.int32)
            indices = np.ascontiguousarray(csr.indices, dtype=np.int32)
            vals = np.ascontiguousarray(csr.data, dtype=np.float64)
            _safe_call(_LIB.LGBM_BoosterPredictForCSR(
                self._handle, indptr.ctypes.data_as(ctypes.c_void_p),
                ctypes.c_int(_DTYPE_I32),
                indices.ctypes.data_as(ctypes.POINTER(ctypes.c_int32)),
                vals.ctypes.data_as(ctypes.c_void_p), ctypes.c_int(_DTYPE_F64),
                ctypes.c_int64(len(indptr)), ctypes.c_int64(csr.nnz),
                ctypes.c_int64(csr.shape[1]), ctypes.c_int(ptype),
                ctypes.c_int(start_iteration), ctypes.c_int(num_iteration), _c_str(""),
                ctypes.byref(out_len),
                res.ctypes.data_as(ctypes.POINTER(ctypes.c_double))))
            per_row = out_len.value // nrow if nrow else 0
            if per_row > 1:
                res = res.reshape(nrow, per_row)
            if pred_leaf:
                res = res.astype(np.int32)
            return res
        if _is_pyarrow_table(data):
            nrow = data.num_rows
            n = ctypes.c_int64(0)
            _safe_call(_LIB.LGBM_BoosterCalcNumPredict(
                self._handle, ctypes.c_int(nrow), ctypes.c_int(ptype),
                ctypes.c_int(start_iteration), ctypes.c_int(num_iteration),
                ctypes.byref(n)))
            res = np.zeros(n.value, dtype=np.float64)
            out_len = ctypes.c_int64(0)
            n_chunks, chunks, schema = _export_arrow_table(data)
            _safe_call(_LIB.LGBM_BoosterPredictForArrow(
                self._handle, ctypes.c_int64(n_chunks), chunks, ctypes.byref(schema),
                ctypes.c_int(ptype), ctypes.c_int(start_iteration),
                ctypes.c_int(num_iteration), _c_str(""), ctypes.byref(out_len),
                res.ctypes.data_as(ctypes.POINTER(ctypes.c_double))))
            per_row = out_len.value // nrow if nrow else 0
            if per_row > 1:
                res = res.reshape(nrow, per_row)
            if pred_leaf:
                res = res.astype(np.int32)
            return res
        arr = _to_2d_float64(data, getattr(self, "pandas_categorical", None))
        nrow, ncol = arr.shape
        n = ctypes.c_int64(0)
        _safe_call(_LIB.LGBM_BoosterCalcNumPredict(
            self._handle, ctypes.c_int(nrow), ctypes.c_int(ptype),
            ctypes.c_int(start_iteration), ctypes.c_int(num_iteration), ctypes.byref(n)))
        res = np.zeros(n.value, dtype=np.float64)
        out_len = ctypes.c_int64(0)
        _safe_call(_LIB.LGBM_BoosterPredictForMat(
            self._handle, arr.ctypes.data_as(ctypes.c_void_p), ctypes.c_int(_DTYPE_F64),
            ctypes.c_int32(nrow), ctypes.c_int32(ncol), ctypes.c_int(1),
            ctypes.c_int(ptype), ctypes.c_int(start_iteration), ctypes.c_int(num_iteration),
            _c_str(pred_param), ctypes.byref(out_len),
            res.ctypes.data_as(ctypes.POINTER(ctypes.c_double))))
        per_row = out_len.value // nrow if nrow else 0
        if per_row > 1:
            res = res.reshape(nrow, per_row)
        if pred_leaf:
            res = res.astype(np.int32)
        return res

    def refit(self, data, label, decay_rate=0.9, **kwargs):
        leaf_preds = self.predict(data, pred_leaf=True)
        if leaf_preds.ndim == 1:
            leaf_preds = leaf_preds.reshape(-1, 1)
        nrow, ncol = leaf_preds.shape
        new_params = dict(self.params)
        new_params["refit_decay_rate"] = decay_rate
        train_set = Dataset(data, label=label, params=new_params)
        new_booster = Booster(new_params, train_set)
        # copy model
        _safe_call(_LIB.LGBM_BoosterMerge(new_booster._handle, self._handle))
        arr = np.ascontiguousarray(leaf_preds.astype(np.int32))
        _safe_call(_LIB.LGBM_BoosterRefit(
            new_booster._handle, arr.ctypes.data_as(ctypes.POINTER(ctypes.c_int32)),
            ctypes.c_int32(nrow), ctypes.c_int32(ncol)))
        return new_booster

    # ------------------------------------------------------------ serialization
    def save_model(self, filename, num_iteration=None, start_iteration=0,
                   importance_type="split"):
        # write via model_to_string so the pandas_categorical trailer survives a
        # file round-trip (reference Booster.save_model does the same)
        text = self.model_to_string(num_iteration=num_iteration,
                                    start_iteration=start_iteration,
                                    importance_type=importance_type)
        with open(str(filename), "w") as f:
            f.write(text)
        return self

    def model_to_string(self, num_iteration=None, start_iteration=0,
                        importance_type="split"):
        imp = 0 if importance_type == "split" else 1
        if num_iteration is None:
            num_iteration = self.best_iteration if self.best_iteration > 0 else -1
        out_len = ctypes.c_int64(0)
        _safe_call(_LIB.LGBM_BoosterSaveModelToString(
            self._handle, ctypes.c_int(start_iteration), ctypes.c_int(num_iteration),
            ctypes.c_int(imp), ctypes.c_int64(0), ctypes.byref(out_len), None))
        buf = ctypes.create_string_buffer(out_len.value)
        _safe_call(_LIB.LGBM_BoosterSaveModelToString(
            self._handle, ctypes.c_int(start_iteration), ctypes.c_int(num_iteration),
            ctypes.c_int(imp), out_len, ctypes.byref(out_len), buf))
        text = buf.value.decode("utf-8")
        if getattr(self, "pandas_categorical", None) is not None:
            text += "\npandas_categorical:" + json.dumps(self.pandas_categorical) + "\n"
        return text

    def model_from_string(self, model_str):
        idx = model_str.rfind("pandas_categorical:")
        if idx >= 0:
            try:
                self.pandas_categorical = json.loads(
                    model_str[idx + len("pandas_categorical:"):].splitlines()[0])
            except ValueError:
                pass
            model_str = model_str[:idx]
        if self._handle is not None:
            _safe_call(_LIB.LGBM_BoosterFree(self._handle))
        out = ctypes.c_void_p()
        out_iters = ctypes.c_int(0)
        _safe_call(_LIB.LGBM_BoosterLoadModelFromString(
            _c_str(model_str), ctypes.byref(out_iters), ctypes.byref(out)))
        self._handle = out
        return self

    def dump_model(self, num_iteration=None, start_iteration=0, importance_type="split"):
        imp = 0 if importance_type == "split" else 1
        if num_iteration is None:
            num_iteration = self.best_iteration if self.best_iteration > 0 else -1
        out_len = ctypes.c_int64(0)
        _safe_call(_LIB.LGBM_BoosterDumpModel(
            self._handle, ctypes.c_int(start_iteration), ctypes.c_int(num_iteration),
            ctypes.c_int(imp), ctypes.c_int64(0), ctypes.byref(out_len), None))
        buf = ctypes.create_string_buffer(out_len.value)
        _safe_call(_LIB.LGBM_BoosterDumpModel(
            self._handle, ctypes.c_int(start_iteration), ctypes.c_int(num_iteration),
            ctypes.c_int(imp), out_len, ctypes.byref(out_len), buf))
        return json.loads(buf.value.decode("utf-8"))

    def feature_importance(self, importance_type="split", iteration=None):
        imp = 0 if importance_type == "split" else 1
        if iteration is None:
            iteration = -1
        nf = self.num_feature()
        res = np.zeros(nf, dtype=np.float64)
        _safe_call(_LIB.LGBM_BoosterFeatureImportance(
            self._handle, ctypes.c_int(iteration), ctypes.c_int(imp),
            res.ctypes.data_as(ctypes.POINTER(ctypes.c_double))))
        if importance_type == "split":
            return res.astype(np.int64)
        return res

    def trees_to_dataframe(self):
        """Model structure as a pandas DataFrame (parity: reference Booster method)."""
        from .compat import PANDAS_INSTALLED
        if not PANDAS_INSTALLED:
            raise ImportError("pandas is required for trees_to_dataframe")
        import pandas as pd
        model = self.dump_model()
        rows = []

        feat_names = self.feature_name()

        def walk(tree_index, node, parent=None, depth=1):
            if "leaf_index" in node:
                rows.append(dict(tree_index=tree_index, node_depth=depth,
                                 node_index=f"{tree_index}-L{node['leaf_index']}",
                                 left_child=None, right_child=None,
                                 parent_index=parent, split_feature=None,
                                 split_gain=None, threshold=None, decision_type=None,
                                 missing_direction=None, missing_type=None,
                                 value=node["leaf_value"], count=node.get("leaf_count"),
                                 weight=node.get("leaf_weight")))
                return
            ni = f"{tree_index}-S{node['split_index']}"

            def child_index(c):
                return (f"{tree_index}-L{c['leaf_index']}" if "leaf_index" in c
                        else f"{tree_index}-S{c['split_index']}")
            f = node["split_feature"]
            rows.append(dict(tree_index=tree_index, node_depth=depth, node_index=ni,
                             left_child=child_index(node["left_child"]),
                             right_child=child_index(node["right_child"]),
                             parent_index=parent,
                             split_feature=feat_names[f] if isinstance(f, int) and
                             f < len(feat_names) else f,
                             split_gain=node.get("split_gain"),
                             threshold=node["threshold"],
                             decision_type=node["decision_type"],
                             missing_direction="left" if node.get("default_left")
                             else "right",
                             missing_type=node.get("missing_type"),
                             value=node.get("internal_value"),
                             count=node.get("internal_count"),
                             weight=node.get("internal_weight")))
            walk(tree_index, node["left_child"], ni, depth + 1)
            walk(tree_index, node["right_child"], ni, depth + 1)

        for t in model["tree_info"]:
            walk(t["tree_index"], t["tree_structure"])
        return pd.DataFrame(rows)

    def free_dataset(self):
        # The native GBDT keeps raw pointers into the Dataset; keep the Python objects
        # alive for the Booster's lifetime (the reference uses shared_ptr ownership in
        # the C++ Booster for the same reason).
        self._kept_refs = getattr(self, "_kept_refs", [])
        if self._train_set is not None:
            self._kept_refs.append(self._train_set)
        self._kept_refs.extend(self._valid_sets)
        return self

    def free_network(self):
        _safe_call(_LIB.LGBM_NetworkFree())
        self._network_initialized = False
        return self

    def set_network(self, machines=None, local_listen_port=12400, listen_time_out=120,
                    num_machines=1):
        from .parallel import init_network_from_torch_distributed
        init_network_from_torch_distributed()
        self._network_initialized = True
        return self
