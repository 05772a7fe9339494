"""UBJSON codec (Universal Binary JSON, spec draft 12).

XGBoost >= 1.6 saves Boosters in UBJSON by default when the file name has
no `.json` extension; this module lets the Booster load such files (and
write them). Supports the full value set xgboost emits: sized/typed
(optimized) containers, all integer widths, float32/64, strings.
"""
import struct

import numpy as np

_INT_TYPES = {
    b"i": ("b", 1),
    b"U": ("B", 1),
    b"I": (">h", 2),
    b"l": (">i", 4),
    b"L": (">q", 8),
}
_FLOAT_TYPES = {b"d": (">f", 4), b"D": (">d", 8)}


class _Reader:
    def __init__(self, buf):
        self.buf = buf
        self.pos = 0

    def byte(self):
        b = self.buf[self.pos : self.pos + 1]
        self.pos += 1
        return b

    def peek(self):
        return self.buf[self.pos : self.pos + 1]

    def read(self, n):
        out = self.buf[self.pos : self.pos + n]
        self.pos += n
        return out

    def int_value(self, marker):
        fmt, size = _INT_TYPES[marker]
        return struct.unpack(fmt, self.read(size))[0]

    def length(self):
        marker = self.byte()
        if marker not in _INT_TYPES:
            raise ValueError(f"UBJSON: invalid length marker {marker!r}")
        return self.int_value(marker)

    def string(self):
        return self.read(self.length()).decode("utf-8")

    def value(self, marker=None):
        if marker is None:
            marker = self.byte()
        if marker in _INT_TYPES:
            return self.int_value(marker)
        if marker in _FLOAT_TYPES:
            fmt, size = _FLOAT_TYPES[marker]
            return struct.unpack(fmt, self.read(size))[0]
        if marker == b"S":
            return self.string()
        if marker == b"C":
            return self.read(1).decode("latin-1")
        if marker == b"T":
            return True
        if marker == b"F":
            return False
        if marker in (b"Z", b"N"):
            return None
        if marker == b"[":
            return self.array()
        if marker == b"{":
            return self.obj()
        if marker == b"H":  # high-precision number -> float
            return float(self.string())
        raise ValueError(f"UBJSON: unknown marker {marker!r} at {self.pos}")

    def _container_header(self):
        item_type = None
        count = None
        if self.peek() == b"$":
            self.byte()
            item_type = self.byte()
        if self.peek() == b"#":
            self.byte()
            count = self.length()
        elif item_type is not None:
            raise ValueError("UBJSON: typed container without count")
        return item_type, count

    def array(self):
        item_type, count = self._container_header()
        if count is not None and item_type is not None:
            if item_type in _INT_TYPES:
                fmt, size = _INT_TYPES[item_type]
                if size == 1:
                    dtype = np.int8 if item_type == b"i" else np.uint8
                    return np.frombuffer(self.read(count), dtype=dtype).tolist()
                raw = self.read(count * size)
                return list(struct.unpack(f">{count}{fmt[-1]}", raw))
            if item_type in _FLOAT_TYPES:
                fmt, size = _FLOAT_TYPES[item_type]
                raw = self.read(count * size)
                return np.frombuffer(raw, dtype=f">{'f' if size == 4 else 'd'}{''}").astype(
                    np.float32 if size == 4 else np.float64
                ).tolist()
            if item_type in (b"T", b"F"):
                return [item_type == b"T"] * count
            if item_type == b"Z":
                return [None] * count
            return [self.value(item_type) for _ in range(count)]
        out = []
        if count is not None:
            for _ in range(count):
                out.append(self.value())
            return out
        while self.peek() != b"]":
            out.append(self.value())
        self.byte()
        return out

    def obj(self):
        item_type, count = self._container_header()
        out = {}
        if count is not None:
            for _ in range(count):
                key = self.string()
                out[key] = self.value(item_type) if item_type else self.value()
            return out
        while self.peek() != b"}":
            key = self.string()
            out[key] = self.value()
        self.byte()
        return out


def loads(buf):
    return _Reader(bytes(buf)).value()


# ---------------------------------------------------------------------------


def _write_length(out, n):
    if n < 256:
        out += b"U" + struct.pack("B", n)
    elif n < 2**31:
        out += b"l" + struct.pack(">i", n)
    else:
        out += b"L" + struct.pack(">q", n)


def _write_string_body(out, s):
    raw = s.encode("utf-8")
    _write_length(out, len(raw))
    out += raw


def _dump(out, value):
    if value is None:
        out += b"Z"
    elif value is True:
        out += b"T"
    elif value is False:
        out += b"F"
    elif isinstance(value, (int, np.integer)):
        v = int(value)
        if -(2**31) <= v < 2**31:
            out += b"l" + struct.pack(">i", v)
        else:
            out += b"L" + struct.pack(">q", v)
    elif isinstance(value, (float, np.floating)):
        out += b"D" + struct.pack(">d", float(value))
    elif isinstance(value, str):
        out += b"S"
        _write_string_body(out, value)
    elif isinstance(value, dict):
        out += b"{"
        for k, v in value.items():
            _write_string_body(out, str(k))
            _dump(out, v)
        out += b"}"
    elif isinstance(value, (list, tuple, np.ndarray)):
        out += b"["
        for v in list(value):
            _dump(out, v)
        out += b"]"
    else:
        raise TypeError(f"UBJSON: cannot encode {type(value)}")


def dumps(value):
    out = bytearray()
    _dump(out, value)
    return bytes(out)
