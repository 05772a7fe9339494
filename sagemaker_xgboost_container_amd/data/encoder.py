"""Inference payload decoders: request body -> DMatrix.

Parity: reference encoder.py:31-142 (csv/libsvm/recordio-protobuf decoders,
json_to_jsonlines, MIME-dispatching decode).
"""
import csv as csv_module
import io
import json
import logging

import numpy as np

from ..constants import xgb_content_types
from ..toolkit import exceptions as exc
from .dmatrix import DMatrix
from .recordio_protobuf import read_recordio_protobuf

_MIME_CSV = "text/csv"


def csv_to_dmatrix(input, dtype=None):
    """CSV payload (no label column) -> DMatrix. Empty fields become NaN.

    The serving hot path uses the extension's multithreaded C++ tokenizer
    (a 1000x28 payload parses in ~0.3 ms vs ~6 ms through pandas); pandas
    remains the fallback when the extension is absent.
    """
    csv_string = input.decode() if isinstance(input, bytes) else input
    sniff_delimiter = csv_module.Sniffer().sniff(csv_string.split("\n")[0][:512]).delimiter
    delimiter = "," if sniff_delimiter.isalnum() else sniff_delimiter
    logging.debug("Determined delimiter of CSV input is '%s'", delimiter)
    try:
        from ..ops import _smxgb_hip as K

        # serving payloads are small: thread-spawn cost exceeds parse time
        # (measured 11.6 ms p50 with all-core threads vs ~0.5 ms single-
        # threaded at 217 KB) — one thread per ~4 MB of payload, capped
        nthreads = min(8, max(1, len(csv_string) >> 22))
        arr = K.parse_csv(csv_string, delimiter, nthreads).numpy()
        return DMatrix(arr)
    except ImportError:
        pass
    import io

    import pandas as pd

    frame = pd.read_csv(
        io.StringIO(csv_string),
        sep=delimiter,
        header=None,
        dtype=np.float64 if dtype in (float, np.float64) else np.float32,
        na_values=[""],
        skip_blank_lines=True,
    )
    return DMatrix(frame.to_numpy(dtype=dtype or np.float32))


def libsvm_to_dmatrix(string_like):
    """LIBSVM payload (no labels expected) -> dense DMatrix.

    Standard 1-based indices are shifted to 0-based when no index 0 appears.
    """
    if isinstance(string_like, (bytes, bytearray)):
        string_like = string_like.decode("utf-8")

    rows = []
    for line in string_like.strip().split("\n"):
        row = {}
        for token in line.strip().split():
            if ":" in token:
                idx, _, val = token.partition(":")
                row[int(idx)] = float(val)
        rows.append(row)

    if not rows or not any(rows):
        return DMatrix(np.empty((0, 0), dtype=np.float32))

    min_idx = min(idx for row in rows for idx in row)
    offset = 1 if min_idx >= 1 else 0
    max_col = max(idx for row in rows for idx in row) - offset + 1
    data = np.zeros((len(rows), max_col), dtype=np.float32)
    for i, row in enumerate(rows):
        for idx, val in row.items():
            data[i, idx - offset] = val
    return DMatrix(data)


def recordio_protobuf_to_dmatrix(string_like):
    features, labels = read_recordio_protobuf(bytes(string_like))
    return DMatrix(features, label=labels)


def npy_to_dmatrix(bytes_like):
    """Numpy .npy payload -> DMatrix. allow_pickle stays False (the
    hardening the reference ships as docker/.../patches/decoder.py)."""
    stream = io.BytesIO(bytes(bytes_like))
    return DMatrix(np.load(stream, allow_pickle=False))


_dmatrix_decoders_map = {
    _MIME_CSV: csv_to_dmatrix,
    xgb_content_types.LIBSVM: libsvm_to_dmatrix,
    xgb_content_types.X_LIBSVM: libsvm_to_dmatrix,
    xgb_content_types.X_RECORDIO_PROTOBUF: recordio_protobuf_to_dmatrix,
    "application/x-npy": npy_to_dmatrix,
}


def json_to_jsonlines(json_data):
    """{'key': [entries...]} -> one JSON entry per line (bytes)."""
    resp_dict = json_data if isinstance(json_data, dict) else json.loads(json_data)
    if len(resp_dict.keys()) != 1:
        raise ValueError("JSON response is not compatible for conversion to jsonlines.")
    bio = io.BytesIO()
    for value in resp_dict.values():
        for entry in value:
            bio.write(bytes(json.dumps(entry) + "\n", "UTF-8"))
    return bio.getvalue()


def decode(obj, content_type):
    """Decode a request payload into a DMatrix by MIME content type."""
    media_content_type = content_type.split(";")[0].strip().lower()
    try:
        decoder = _dmatrix_decoders_map[media_content_type]
    except KeyError:
        raise exc.UserError(f"Unsupported content type: {media_content_type}")
    return decoder(obj)
