"""UBJSON (Universal Binary JSON, draft-12) encoder/decoder.

XGBoost >= 1.6 serializes models as UBJSON for ``.ubj`` files and is the
binary model format the reference's user contract exposes through
``bst.save_model("model.xgb")`` (reference README.md:78) and Ray Tune's
``load_model`` path (reference xgboost_ray/tune.py:130-156).  This module
implements the subset of the spec XGBoost's strict reader understands so
models written here load in stock XGBoost and vice versa:

* objects ``{`` .. ``}`` with length-prefixed keys (no ``S`` marker on
  keys, per spec),
* strings ``S`` + int-typed length,
* integers ``i``/``U``/``I``/``l``/``L`` (smallest-fit on encode, all
  accepted on decode),
* floats ``d`` (f32) / ``D`` (f64) and high-precision ``H``,
* booleans ``T``/``F``, null ``Z``, no-op ``N``,
* strongly-typed sized containers ``[$<type>#<count>`` for numeric
  arrays (XGBoost's F32Array/F64Array/I32Array/I64Array/U8Array
  spellings), including sized-only (``#`` without ``$``) containers and
  sized/typed objects on decode.

Standalone: no third-party dependency, pure struct packing.
"""

import struct
from typing import Any, List, Tuple

import numpy as np

__all__ = ["dumps", "loads"]

_INT_MARKS = {
    b"i": ("<b", 1),
    b"U": ("<B", 1),
    b"I": ("<h", 2),
    b"l": ("<i", 4),
    b"L": ("<q", 8),
}

_TYPED_DTYPES = {
    b"i": np.dtype("<i1"),
    b"U": np.dtype("<u1"),
    b"I": np.dtype("<i2"),
    b"l": np.dtype("<i4"),
    b"L": np.dtype("<i8"),
    b"d": np.dtype("<f4"),
    b"D": np.dtype("<f8"),
}


# ---------------------------------------------------------------------------
# encoding
# ---------------------------------------------------------------------------


def _enc_int(out: List[bytes], v: int):
    """Smallest-fit integer per the UBJSON spec recommendation."""
    if -128 <= v <= 127:
        out.append(b"i" + struct.pack("<b", v))
    elif 0 <= v <= 255:
        out.append(b"U" + struct.pack("<B", v))
    elif -32768 <= v <= 32767:
        out.append(b"I" + struct.pack("<h", v))
    elif -(2**31) <= v <= 2**31 - 1:
        out.append(b"l" + struct.pack("<i", v))
    elif -(2**63) <= v <= 2**63 - 1:
        out.append(b"L" + struct.pack("<q", v))
    else:
        # out-of-range: high-precision number (decimal string)
        s = str(v).encode()
        out.append(b"H")
        _enc_int(out, len(s))
        out.append(s)


def _enc_str_body(out: List[bytes], s: str):
    b = s.encode("utf-8")
    _enc_int(out, len(b))
    out.append(b)


def _typed_array_marker(seq) -> bytes:
    """``$``-marker for a homogeneous numeric list, or b"" if mixed."""
    if len(seq) == 0:
        return b""
    all_int = True
    all_float = True
    lo, hi = 0, 0
    for x in seq:
        if isinstance(x, bool) or not isinstance(x, (int, float)):
            return b""
        if isinstance(x, int):
            all_float = False
            lo = min(lo, x)
            hi = max(hi, x)
        else:
            all_int = False
    if all_int:
        if -(2**31) <= lo and hi <= 2**31 - 1:
            return b"l"
        if -(2**63) <= lo and hi <= 2**63 - 1:
            return b"L"
        return b""
    if all_float:
        # f64 always: f32 would truncate gblinear weights / split values
        return b"D"
    return b""


def _enc_typed_payload(out: List[bytes], mark: bytes, seq):
    out.append(np.asarray(seq, dtype=_TYPED_DTYPES[mark]).tobytes())


def _enc_ndarray(out: List[bytes], a: np.ndarray):
    a = np.ascontiguousarray(a).reshape(-1)
    kind_map = {
        "f4": b"d", "f8": b"D",
        "i1": b"i", "u1": b"U", "i2": b"I", "i4": b"l", "i8": b"L",
    }
    key = a.dtype.str.lstrip("<>|=")
    if key == "b1":
        a = a.astype("<u1")
        key = "u1"
    if key not in kind_map:
        _enc_value(out, a.tolist())
        return
    mark = kind_map[key]
    out.append(b"[$" + mark + b"#")
    _enc_int(out, a.size)
    out.append(a.astype(_TYPED_DTYPES[mark], copy=False).tobytes())


def _enc_value(out: List[bytes], v: Any):
    if v is None:
        out.append(b"Z")
    elif v is True:
        out.append(b"T")
    elif v is False:
        out.append(b"F")
    elif isinstance(v, (int, np.integer)):
        _enc_int(out, int(v))
    elif isinstance(v, (float, np.floating)):
        out.append(b"D" + struct.pack("<d", float(v)))
    elif isinstance(v, str):
        out.append(b"S")
        _enc_str_body(out, v)
    elif isinstance(v, (bytes, bytearray)):
        # raw bytes as a U8 typed array
        out.append(b"[$U#")
        _enc_int(out, len(v))
        out.append(bytes(v))
    elif isinstance(v, np.ndarray):
        _enc_ndarray(out, v)
    elif isinstance(v, dict):
        out.append(b"{")
        for k, item in v.items():
            _enc_str_body(out, str(k))
            _enc_value(out, item)
        out.append(b"}")
    elif isinstance(v, (list, tuple)):
        mark = _typed_array_marker(v)
        if mark:
            out.append(b"[$" + mark + b"#")
            _enc_int(out, len(v))
            _enc_typed_payload(out, mark, v)
        else:
            out.append(b"[")
            for item in v:
                _enc_value(out, item)
            out.append(b"]")
    else:
        raise TypeError(f"UBJSON cannot encode {type(v)!r}")


def dumps(doc: Any) -> bytes:
    out: List[bytes] = []
    _enc_value(out, doc)
    return b"".join(out)


# ---------------------------------------------------------------------------
# decoding
# ---------------------------------------------------------------------------


class _Reader:
    __slots__ = ("buf", "pos")

    def __init__(self, buf: bytes):
        self.buf = buf
        self.pos = 0

    def take(self, n: int) -> bytes:
        b = self.buf[self.pos : self.pos + n]
        if len(b) != n:
            raise ValueError("UBJSON: truncated input")
        self.pos += n
        return b

    def marker(self) -> bytes:
        # skip no-ops
        while True:
            m = self.take(1)
            if m != b"N":
                return m


def _dec_int(r: _Reader, m: bytes) -> int:
    try:
        fmt, n = _INT_MARKS[m]
    except KeyError:
        raise ValueError(f"UBJSON: expected integer marker, got {m!r}")
    return struct.unpack(fmt, r.take(n))[0]


def _dec_length(r: _Reader) -> int:
    n = _dec_int(r, r.marker())
    if n < 0:
        raise ValueError("UBJSON: negative length")
    return n


def _dec_str_body(r: _Reader) -> str:
    return r.take(_dec_length(r)).decode("utf-8")


def _dec_container_header(r: _Reader) -> Tuple[bytes, int]:
    """Parse optional ``$type`` / ``#count``; returns (type or b'', count
    or -1) and leaves the reader at the first element."""
    typ, count = b"", -1
    m = r.marker()
    if m == b"$":
        typ = r.take(1)
        m = r.marker()
        if m != b"#":
            raise ValueError("UBJSON: typed container requires a count")
        count = _dec_length(r)
        return typ, count
    if m == b"#":
        count = _dec_length(r)
        return typ, count
    r.pos -= 1  # not a header: rewind the marker
    return typ, count


def _dec_value(r: _Reader, m: bytes) -> Any:
    if m in _INT_MARKS:
        return _dec_int(r, m)
    if m == b"d":
        return struct.unpack("<f", r.take(4))[0]
    if m == b"D":
        return struct.unpack("<d", r.take(8))[0]
    if m == b"H":
        s = r.take(_dec_length(r)).decode()
        try:
            return int(s)
        except ValueError:
            return float(s)
    if m == b"S":
        return _dec_str_body(r)
    if m == b"C":
        return r.take(1).decode("latin-1")
    if m == b"T":
        return True
    if m == b"F":
        return False
    if m == b"Z":
        return None
    if m == b"[":
        return _dec_array(r)
    if m == b"{":
        return _dec_object(r)
    raise ValueError(f"UBJSON: unknown marker {m!r} at {r.pos - 1}")


def _dec_array(r: _Reader) -> list:
    typ, count = _dec_container_header(r)
    if typ:
        if typ in _TYPED_DTYPES:
            dt = _TYPED_DTYPES[typ]
            arr = np.frombuffer(
                r.take(count * dt.itemsize), dtype=dt
            )
            return arr.tolist()
        if typ == b"C":
            return list(r.take(count).decode("latin-1"))
        if typ in (b"T", b"F"):
            return [typ == b"T"] * count
        if typ == b"Z":
            return [None] * count
        # typed array of strings/containers: elements carry no marker
        return [_dec_value(r, typ) for _ in range(count)]
    if count >= 0:
        return [_dec_value(r, r.marker()) for _ in range(count)]
    out = []
    while True:
        m = r.marker()
        if m == b"]":
            return out
        out.append(_dec_value(r, m))


def _dec_object(r: _Reader) -> dict:
    typ, count = _dec_container_header(r)
    out = {}
    if count >= 0:
        for _ in range(count):
            key = _dec_str_body(r)
            out[key] = (
                _dec_value(r, typ) if typ else _dec_value(r, r.marker())
            )
        return out
    while True:
        m = r.marker()
        if m == b"}":
            return out
        r.pos -= 1
        key = _dec_str_body(r)
        out[key] = _dec_value(r, r.marker())


def loads(data: bytes) -> Any:
    r = _Reader(bytes(data))
    val = _dec_value(r, r.marker())
    return val
