"""RecordIO-framed aialgs-protobuf codec (reader AND writer).

The SageMaker "application/x-recordio-protobuf" format is a stream of
records, each framed as:

    uint32 magic = 0xCED7230A | uint32 length | <length bytes> | pad to 4

where the payload is an ``aialgs.data.Record`` protobuf message:

    message Float32Tensor { repeated float  values = 1 [packed]; repeated uint64 keys = 2 [packed]; repeated uint64 shape = 3 [packed]; }
    message Float64Tensor { repeated double values = 1 [packed]; repeated uint64 keys = 2 [packed]; repeated uint64 shape = 3 [packed]; }
    message Int32Tensor   { repeated int32  values = 1 [packed]; repeated uint64 keys = 2 [packed]; repeated uint64 shape = 3 [packed]; }
    message Value  { oneof value { Float32Tensor float32_tensor = 2; Float64Tensor float64_tensor = 3; Int32Tensor int32_tensor = 7; } }
    message Record { map<string, Value> features = 1; map<string, Value> label = 2; string uid = 3; string metadata = 4; string configuration = 5; }

This module implements the wire format directly (hand-rolled varint
encode/decode) so the framework has no dependency on generated _pb2 stubs.

Parity: reference recordio_protobuf.py (framing, magic 0xCED7230A, dense +
sparse-CSR decode) and serve_utils' response encoding.
"""
import struct

import numpy as np
from scipy.sparse import csr_matrix
from scipy.sparse import vstack as scipy_vstack

RECORDIO_MAGIC = 0xCED7230A

# --------------------------------------------------------------------------
# protobuf wire-format primitives
# --------------------------------------------------------------------------

_WT_VARINT = 0
_WT_I64 = 1
_WT_LEN = 2
_WT_I32 = 5


def _read_varint(buf, pos):
    result = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not b & 0x80:
            return result, pos
        shift += 7
        if shift > 63:
            raise ValueError("varint too long")


def _write_varint(out, value):
    while True:
        b = value & 0x7F
        value >>= 7
        if value:
            out.append(b | 0x80)
        else:
            out.append(b)
            return


def _skip_field(buf, pos, wire_type):
    if wire_type == _WT_VARINT:
        _, pos = _read_varint(buf, pos)
    elif wire_type == _WT_I64:
        pos += 8
    elif wire_type == _WT_LEN:
        n, pos = _read_varint(buf, pos)
        pos += n
    elif wire_type == _WT_I32:
        pos += 4
    else:
        raise ValueError(f"unsupported wire type {wire_type}")
    return pos


def _iter_fields(buf):
    """Yield (field_number, wire_type, value_bytes_or_int) over a message."""
    pos = 0
    end = len(buf)
    while pos < end:
        tag, pos = _read_varint(buf, pos)
        field, wire_type = tag >> 3, tag & 7
        if wire_type == _WT_VARINT:
            value, pos = _read_varint(buf, pos)
            yield field, wire_type, value
        elif wire_type == _WT_LEN:
            n, pos = _read_varint(buf, pos)
            yield field, wire_type, buf[pos : pos + n]
            pos += n
        elif wire_type == _WT_I64:
            yield field, wire_type, buf[pos : pos + 8]
            pos += 8
        elif wire_type == _WT_I32:
            yield field, wire_type, buf[pos : pos + 4]
            pos += 4
        else:
            raise ValueError(f"unsupported wire type {wire_type}")


def _decode_packed(data, dtype, scalar_wire):
    """Decode a packed repeated numeric field into a numpy array."""
    if scalar_wire == _WT_I32:
        return np.frombuffer(data, dtype=dtype)
    if scalar_wire == _WT_I64:
        return np.frombuffer(data, dtype=dtype)
    # packed varints
    values = []
    pos = 0
    while pos < len(data):
        v, pos = _read_varint(data, pos)
        values.append(v)
    return np.asarray(values, dtype=dtype)


def _decode_tensor(buf, value_dtype, value_wire):
    """Decode a {Float32,Float64,Int32}Tensor message body."""
    values = np.array([], dtype=value_dtype)
    keys = None
    shape = None
    for field, wt, data in _iter_fields(buf):
        if field == 1:  # values
            if wt == _WT_LEN:
                values = _decode_packed(data, value_dtype, value_wire)
            else:  # unpacked scalar
                values = np.append(values, np.frombuffer(bytes(data), dtype=value_dtype))
        elif field == 2:  # keys (packed uint64 varints)
            keys = _decode_packed(data, np.uint64, _WT_VARINT) if wt == _WT_LEN else None
        elif field == 3:  # shape
            shape = _decode_packed(data, np.uint64, _WT_VARINT).tolist() if wt == _WT_LEN else None
    return values, keys, shape


def _decode_value(buf):
    """Decode a Value message -> (values, keys, shape) or (None, None, None)."""
    for field, wt, data in _iter_fields(buf):
        if wt != _WT_LEN:
            continue
        if field == 2:  # float32_tensor
            return _decode_tensor(data, np.float32, _WT_I32)
        if field == 3:  # float64_tensor
            return _decode_tensor(data, np.float64, _WT_I64)
        if field == 7:  # int32_tensor
            return _decode_tensor(data, np.int32, _WT_VARINT)
    return None, None, None


def _decode_map_entry(buf):
    """Decode a map<string, Value> entry -> (key, value_bytes)."""
    key = None
    value = None
    for field, wt, data in _iter_fields(buf):
        if field == 1:
            key = bytes(data).decode("utf-8")
        elif field == 2:
            value = data
    return key, value


def _decode_record(buf):
    """Decode a Record message -> (features_dict, label_dict)."""
    features = {}
    label = {}
    for field, wt, data in _iter_fields(buf):
        if wt != _WT_LEN:
            continue
        if field == 1:
            k, v = _decode_map_entry(data)
            if k is not None and v is not None:
                features[k] = v
        elif field == 2:
            k, v = _decode_map_entry(data)
            if k is not None and v is not None:
                label[k] = v
    return features, label


# --------------------------------------------------------------------------
# RecordIO framing
# --------------------------------------------------------------------------


def iter_recordio(buf):
    """Yield the payload bytes of each RecordIO-framed record in ``buf``."""
    offset = 0
    total = len(buf)
    while offset < total:
        if offset + 8 > total:
            break
        magic, length = struct.unpack_from("<II", buf, offset)
        if magic != RECORDIO_MAGIC:
            raise ValueError(f"Invalid RecordIO magic at offset {offset}")
        offset += 8
        padded = (length + 3) & ~3
        if offset + length > total:
            raise ValueError(f"Truncated record at offset {offset}")
        yield buf[offset : offset + length]
        offset += padded


def _frame_record(payload):
    pad = (-len(payload)) % 4
    return struct.pack("<II", RECORDIO_MAGIC, len(payload)) + payload + b"\x00" * pad


# --------------------------------------------------------------------------
# Public API
# --------------------------------------------------------------------------


def read_recordio_protobuf(buf):
    """Decode a RecordIO-protobuf byte buffer.

    Returns ``(features, labels)``: features is a dense ndarray or scipy CSR
    matrix (when any record carries sparse keys), labels an ndarray or None.
    """
    dense_rows = []
    sparse_rows = []
    labels = []
    any_sparse = False

    for payload in iter_recordio(bytes(buf)):
        features, label = _decode_record(payload)
        if "values" in features:
            values, keys, shape = _decode_value(features["values"])
            if values is None and keys is None:
                continue
            if keys is not None:
                any_sparse = True
                if shape:
                    ncols = int(shape[0])
                elif len(keys):
                    ncols = int(keys.max()) + 1
                else:
                    ncols = 1
                sparse_rows.append(
                    csr_matrix(
                        (values.astype(np.float32), keys.astype(np.int64), [0, len(keys)]),
                        shape=(1, ncols),
                    )
                )
                dense_rows.append(None)
            elif shape and not len(values):
                # empty sparse row encoded with shape only
                any_sparse = True
                sparse_rows.append(csr_matrix((1, int(shape[0])), dtype=np.float32))
                dense_rows.append(None)
            else:
                dense_rows.append(values.astype(np.float32).reshape(1, -1))
                sparse_rows.append(None)
        else:
            continue

        if "values" in label:
            lv, _, _ = _decode_value(label["values"])
            if lv is not None:
                labels.append(lv)

    if not dense_rows:
        raise ValueError("No records found in RecordIO-Protobuf data")

    if any_sparse:
        ncols = max(row.shape[1] for row in sparse_rows if row is not None)
        rows = []
        for dense, sparse in zip(dense_rows, sparse_rows):
            if sparse is not None:
                if sparse.shape[1] != ncols:
                    sparse = csr_matrix((sparse.data, sparse.indices, sparse.indptr), shape=(1, ncols))
                rows.append(sparse)
            else:
                rows.append(csr_matrix(dense, shape=(1, ncols)))
        features_out = scipy_vstack(rows).tocsr()
    else:
        features_out = np.vstack(dense_rows)

    labels_out = np.concatenate(labels, axis=None) if labels else None
    return features_out, labels_out


def _encode_float32_tensor(values, keys=None, shape=None):
    body = bytearray()
    values = np.asarray(values, dtype="<f4")
    if values.size:
        packed = values.tobytes()
        _write_varint(body, (1 << 3) | _WT_LEN)
        _write_varint(body, len(packed))
        body += packed
    if keys is not None and len(keys):
        kb = bytearray()
        for k in keys:
            _write_varint(kb, int(k))
        _write_varint(body, (2 << 3) | _WT_LEN)
        _write_varint(body, len(kb))
        body += kb
    if shape is not None:
        sb = bytearray()
        for s in shape:
            _write_varint(sb, int(s))
        _write_varint(body, (3 << 3) | _WT_LEN)
        _write_varint(body, len(sb))
        body += sb
    return bytes(body)


def _encode_value(tensor_body):
    out = bytearray()
    _write_varint(out, (2 << 3) | _WT_LEN)  # float32_tensor
    _write_varint(out, len(tensor_body))
    out += tensor_body
    return bytes(out)


def _encode_map_entry(field, key, value_body):
    entry = bytearray()
    kb = key.encode("utf-8")
    _write_varint(entry, (1 << 3) | _WT_LEN)
    _write_varint(entry, len(kb))
    entry += kb
    _write_varint(entry, (2 << 3) | _WT_LEN)
    _write_varint(entry, len(value_body))
    entry += value_body
    out = bytearray()
    _write_varint(out, (field << 3) | _WT_LEN)
    _write_varint(out, len(entry))
    out += entry
    return bytes(out)


def write_recordio_protobuf(feature_map, label_map=None):
    """Encode ONE Record with the given {name: ndarray} maps, RecordIO-framed."""
    payload = bytearray()
    for name, values in (feature_map or {}).items():
        payload += _encode_map_entry(1, name, _encode_value(_encode_float32_tensor(values)))
    for name, values in (label_map or {}).items():
        payload += _encode_map_entry(2, name, _encode_value(_encode_float32_tensor(values)))
    return _frame_record(bytes(payload))


def write_recordio_rows(rows_iter):
    """Encode many Records (one per row dict of {map_name: {key: values}})."""
    out = bytearray()
    for features, label in rows_iter:
        out += write_recordio_protobuf(features, label)
    return bytes(out)
