"""DMatrix — the framework's data container for training and prediction.

Replaces ``xgb.DMatrix`` (the reference's native boundary at
data_utils.py:309-313,361,384,453). Holds features as either a dense float32
matrix (missing encoded as NaN) or a scipy CSR matrix, plus per-row label /
weight / base-margin vectors. Quantization into the on-GPU bin matrix is the
trainer's job (models/quantize.py) so that one DMatrix can be shared across
boosters with different ``max_bin``.

Construction accepts:
  * numpy 2-D arrays / scipy sparse matrices,
  * file or directory paths, optionally with URI-style parameters:
        "/path/data?format=csv&label_column=0&delimiter=,&weight_column=1"
        "/path/data?format=libsvm"
    (the same URI surface the reference forwards to xgboost,
    data_utils.py:309-313,361).
"""
import os
from urllib.parse import parse_qs

import numpy as np
import scipy.sparse as sp

from ..toolkit import exceptions as exc


def _is_data_file_name(path, name):
    if not os.path.isfile(os.path.join(path, name)):
        return False
    if name.startswith(".") or name.startswith("_"):
        return False
    if ".cache" in name and ("dtrain" in name or "dval" in name):
        return False
    return True


def _list_data_files(path):
    if os.path.isfile(path):
        return [path]
    files = [os.path.join(path, f) for f in sorted(os.listdir(path)) if _is_data_file_name(path, f)]
    if not files:
        raise exc.UserError(f"No data files found under {path}")
    return files


def _parse_csv_native(files, delimiter):
    import torch  # noqa: F401

    from ..ops import _smxgb_hip as K

    text = "".join(open(f, "r", errors="ignore").read() for f in files)
    return K.parse_csv(text, delimiter, 0).numpy()


def _parse_csv_files(files, delimiter=",", label_column=0, weight_column=None):
    try:
        data = _parse_csv_native(files, delimiter)
    except ImportError:
        import pandas as pd

        frames = [
            pd.read_csv(f, sep=delimiter, header=None, dtype=np.float32, na_values=[""], skip_blank_lines=True)
            for f in files
        ]
        data = pd.concat(frames, axis=0, ignore_index=True).to_numpy(dtype=np.float32)
    ncol = data.shape[1]
    label = None
    weight = None
    cols = list(range(ncol))
    if label_column is not None and ncol > label_column:
        label = data[:, label_column]
        cols.remove(label_column)
    if weight_column is not None and ncol > weight_column:
        weight = data[:, weight_column]
        cols.remove(weight_column)
    features = data[:, cols]
    return features, label, weight


def _parse_libsvm_line(line):
    parts = line.split()
    if not parts:
        return None
    head = parts[0].split(":")
    label = float(head[0])
    weight = float(head[1]) if len(head) == 2 else None
    qid = None
    indices = []
    values = []
    for tok in parts[1:]:
        idx, _, val = tok.partition(":")
        if idx == "qid":
            qid = int(val)
            continue
        indices.append(int(idx))
        values.append(float(val))
    return label, weight, qid, indices, values


def _parse_libsvm_files_native(files):
    """Multi-threaded C++ libsvm parser (ops/csrc/text_parsers.cpp)."""
    import torch  # noqa: F401  (extension links against torch)

    from ..ops import _smxgb_hip as K

    text = "".join(open(f, "r", errors="ignore").read() for f in files)
    values, indices, indptr, labels, weights, qids, ncol = K.parse_libsvm(text, 0)
    csr = sp.csr_matrix(
        (values.numpy(), indices.numpy(), indptr.numpy()),
        shape=(len(labels), max(int(ncol[0]), 0)),
    )
    return (
        csr,
        labels.numpy(),
        weights.numpy() if weights.numel() else None,
        qids.numpy() if qids.numel() else None,
    )


def _parse_libsvm_files(files):
    """Parse libsvm files (with optional <label>:<weight> extension) to CSR."""
    try:
        return _parse_libsvm_files_native(files)
    except ImportError:
        pass  # extension not built: pure-Python fallback below
    labels = []
    weights = []
    qids = []
    data = []
    indices = []
    indptr = [0]
    any_weight = False
    any_qid = False
    for path in files:
        with open(path, "r", errors="ignore") as f:
            for line in f:
                line = line.strip()
                if not line or line.startswith("#"):
                    continue
                parsed = _parse_libsvm_line(line)
                if parsed is None:
                    continue
                label, weight, qid, idx, val = parsed
                labels.append(label)
                if weight is not None:
                    any_weight = True
                if qid is not None:
                    any_qid = True
                weights.append(weight if weight is not None else 1.0)
                qids.append(qid if qid is not None else 0)
                indices.extend(idx)
                data.extend(val)
                indptr.append(len(indices))
    ncol = (max(indices) + 1) if indices else 0
    csr = sp.csr_matrix(
        (np.asarray(data, dtype=np.float32), np.asarray(indices, dtype=np.int64), np.asarray(indptr, dtype=np.int64)),
        shape=(len(labels), ncol),
    )
    return (
        csr,
        np.asarray(labels, dtype=np.float32),
        np.asarray(weights, dtype=np.float32) if any_weight else None,
        np.asarray(qids, dtype=np.int64) if any_qid else None,
    )


def _parse_parquet_files(files):
    import pyarrow.parquet as pq

    arrays = []
    for f in files:
        table = pq.read_table(f)
        frame = table.to_pandas()
        arrays.append(np.asarray(frame, dtype=np.float32))
    data = np.vstack(arrays)
    return data[:, 1:], data[:, 0], None


def _parse_recordio_files(files):
    from .recordio_protobuf import read_recordio_protobuf

    buf = b"".join(open(f, "rb").read() for f in files)
    features, labels = read_recordio_protobuf(buf)
    return features, labels, None


def _load_uri(uri):
    """Load 'path?format=...&k=v' into (features, label, weight[, qid])."""
    path, _, query = uri.partition("?")
    params = {k: v[0] for k, v in parse_qs(query).items()}
    fmt = params.get("format")
    files = _list_data_files(path)
    if fmt == "csv":
        return _parse_csv_files(
            files,
            delimiter=params.get("delimiter", ","),
            label_column=int(params["label_column"]) if "label_column" in params else None,
            weight_column=int(params["weight_column"]) if "weight_column" in params else None,
        )
    if fmt == "libsvm" or fmt is None:
        csr, labels, weights, qids = _parse_libsvm_files(files)
        return csr, labels, weights, qids
    if fmt == "parquet":
        return _parse_parquet_files(files)
    if fmt == "recordio-protobuf":
        return _parse_recordio_files(files)
    raise exc.UserError(f"Unknown DMatrix format: {fmt}")


class DMatrix:
    """Feature matrix + per-row metadata (label, weight, base margin)."""

    def __init__(
        self,
        data,
        label=None,
        weight=None,
        missing=None,
        feature_names=None,
        nthread=None,
        base_margin=None,
        qid=None,
        group=None,
    ):
        self.missing = np.nan if missing is None else missing
        self.feature_names = feature_names
        self._dense = None
        self._csr = None

        self._group = None
        if isinstance(data, str):
            loaded = _load_uri(data)
            if len(loaded) == 4:
                features, file_label, file_weight, file_qid = loaded
                if file_qid is not None:
                    self.set_qid(file_qid)
            else:
                features, file_label, file_weight = loaded
            if label is None:
                label = file_label
            if weight is None:
                weight = file_weight
            data = features

        # pandas DataFrame/Series: numeric conversion + column names
        if hasattr(data, "to_numpy") and hasattr(data, "columns"):
            if self.feature_names is None:
                self.feature_names = [str(c) for c in data.columns]
            data = data.to_numpy(dtype=np.float32)
        if label is not None and hasattr(label, "to_numpy"):
            label = label.to_numpy(dtype=np.float32)

        if sp.issparse(data):
            self._csr = data.tocsr().astype(np.float32)
        else:
            arr = np.asarray(data, dtype=np.float32)
            if arr.ndim == 1:
                arr = arr.reshape(1, -1)
            if arr.ndim != 2:
                raise exc.UserError(f"DMatrix expects 2-D data, got shape {arr.shape}")
            if not (isinstance(self.missing, float) and np.isnan(self.missing)):
                arr = arr.copy()
                arr[arr == self.missing] = np.nan
            self._dense = np.ascontiguousarray(arr)

        self._label = None if label is None else np.asarray(label, dtype=np.float32).reshape(-1)
        self._weight = None if weight is None else np.asarray(weight, dtype=np.float32).reshape(-1)
        self._base_margin = None if base_margin is None else np.asarray(base_margin, dtype=np.float32)

        if qid is not None:
            self.set_qid(np.asarray(qid))
        if group is not None:
            self.set_group(group)

        if self._label is not None and len(self._label) != self.num_row():
            raise exc.UserError(
                f"Label length {len(self._label)} does not match number of rows {self.num_row()}"
            )

    # -- shape ------------------------------------------------------------
    def num_row(self):
        return self._dense.shape[0] if self._dense is not None else self._csr.shape[0]

    def num_col(self):
        return self._dense.shape[1] if self._dense is not None else self._csr.shape[1]

    @property
    def is_sparse(self):
        return self._csr is not None

    # -- metadata ---------------------------------------------------------
    def get_label(self):
        return self._label if self._label is not None else np.array([], dtype=np.float32)

    def set_label(self, label):
        self._label = np.asarray(label, dtype=np.float32).reshape(-1)

    def get_weight(self):
        return self._weight if self._weight is not None else np.array([], dtype=np.float32)

    def set_weight(self, weight):
        self._weight = None if weight is None else np.asarray(weight, dtype=np.float32).reshape(-1)

    def get_base_margin(self):
        return self._base_margin

    def set_base_margin(self, margin):
        self._base_margin = None if margin is None else np.asarray(margin, dtype=np.float32)

    def set_group(self, group):
        """Per-query group sizes (ranking objectives)."""
        self._group = None if group is None else np.asarray(group, dtype=np.int64)

    def get_group(self):
        return self._group

    def set_qid(self, qid):
        """Set groups from per-row query ids (consecutive runs)."""
        qid = np.asarray(qid)
        if qid.size == 0:
            self._group = None
            return
        change = np.nonzero(np.diff(qid))[0]
        bounds = np.concatenate([[0], change + 1, [len(qid)]])
        self._group = np.diff(bounds).astype(np.int64)

    def get_float_info(self, name):
        if name == "label":
            return self.get_label()
        if name == "weight":
            return self.get_weight()
        if name == "base_margin":
            return self._base_margin
        return getattr(self, "_info", {}).get(name)

    def set_float_info(self, name, value):
        if name == "label":
            self.set_label(value)
        elif name == "weight":
            self.set_weight(value)
        elif name == "base_margin":
            self.set_base_margin(value)
        else:
            if not hasattr(self, "_info"):
                self._info = {}
            self._info[name] = np.asarray(value, dtype=np.float32).reshape(-1)

    # -- views ------------------------------------------------------------
    def to_dense(self):
        """Dense float32 view (NaN = missing). Materializes sparse data.

        Entries ABSENT from a sparse matrix are missing (NaN), matching
        xgboost's sparse semantics — explicit stored zeros stay 0.
        """
        if self._dense is not None:
            return self._dense
        csr = self._csr
        n, f = csr.shape
        dense = np.full((n, f), np.nan, dtype=np.float32)
        rows = np.repeat(np.arange(n), np.diff(csr.indptr))
        dense[rows, csr.indices] = csr.data
        return dense

    def csr(self):
        if self._csr is not None:
            return self._csr
        return sp.csr_matrix(np.nan_to_num(self._dense, nan=0.0))

    def slice(self, rindex):
        """Row-subset DMatrix (used by k-fold CV, train.py:409-451 parity).
        Carries label/weight/base_margin; per-query groups are dropped
        (row subsets break group alignment — xgboost does the same)."""
        rindex = np.asarray(rindex, dtype=np.int64)
        data = self._dense[rindex] if self._dense is not None else self._csr[rindex]
        out = DMatrix(
            data,
            label=self._label[rindex] if self._label is not None else None,
            weight=self._weight[rindex] if self._weight is not None else None,
            base_margin=self._base_margin[rindex] if self._base_margin is not None else None,
            feature_names=self.feature_names,
        )
        return out

    def __repr__(self):
        kind = "sparse" if self.is_sparse else "dense"
        return f"DMatrix({self.num_row()}x{self.num_col()}, {kind})"


class DeviceDMatrix:
    """DMatrix backed by tensors that already live on the training device.

    Used for GPU-resident synthetic data (bench.py) and any caller that
    builds features directly in HBM — avoids the host round-trip of the
    numpy-backed DMatrix. Exposes the subset of the DMatrix surface the
    trainer consumes.
    """

    def __init__(self, data, label=None, weight=None, feature_names=None, base_margin=None):
        self._data = data
        self._label = label
        self._weight = weight
        self._base_margin = base_margin
        self.feature_names = feature_names

    def num_row(self):
        return self._data.shape[0]

    def num_col(self):
        return self._data.shape[1]

    def to_dense(self):
        return self._data

    def get_label(self):
        import torch

        return self._label if self._label is not None else torch.zeros(0)

    def get_weight(self):
        import torch

        return self._weight if self._weight is not None else torch.zeros(0)

    def get_base_margin(self):
        return self._base_margin
