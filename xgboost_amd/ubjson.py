"""Minimal UBJSON (draft-12) encoder/decoder.

Reference behavior: src/common/json.cc UBJSON reader/writer used for
.ubj model files; supports the optimized typed-array containers the
reference emits (`[$<type>#<count>`) so reference-written models load.
"""
from __future__ import annotations

import struct
from io import BytesIO
from typing import Any, BinaryIO

import numpy as np


def dump_ubjson(obj: Any, fh: BinaryIO) -> None:
    fh.write(dumps_ubjson(obj))


def dumps_ubjson(obj: Any) -> bytes:
    out = BytesIO()
    _write(obj, out)
    return out.getvalue()


def load_ubjson(fh: BinaryIO) -> Any:
    return loads_ubjson(fh.read())


def loads_ubjson(data: bytes) -> Any:
    val, _ = _read(memoryview(data), 0)
    return val


def _write_int(v: int, out: BytesIO) -> None:
    if -128 <= v <= 127:
        out.write(b"i" + struct.pack(">b", v))
    elif 0 <= v <= 255:
        out.write(b"U" + struct.pack(">B", v))
    elif -32768 <= v <= 32767:
        out.write(b"I" + struct.pack(">h", v))
    elif -2 ** 31 <= v <= 2 ** 31 - 1:
        out.write(b"l" + struct.pack(">i", v))
    else:
        out.write(b"L" + struct.pack(">q", v))


def _write_str_payload(s: str, out: BytesIO) -> None:
    b = s.encode("utf-8")
    _write_int(len(b), out)
    out.write(b)


def _write(obj: Any, out: BytesIO) -> None:
    if obj is None:
        out.write(b"Z")
    elif obj is True:
        out.write(b"T")
    elif obj is False:
        out.write(b"F")
    elif isinstance(obj, str):
        out.write(b"S")
        _write_str_payload(obj, out)
    elif isinstance(obj, (int, np.integer)):
        _write_int(int(obj), out)
    elif isinstance(obj, (float, np.floating)):
        out.write(b"D" + struct.pack(">d", float(obj)))
    elif isinstance(obj, dict):
        out.write(b"{")
        for k, v in obj.items():
            _write_str_payload(str(k), out)
            _write(v, out)
        out.write(b"}")
    elif isinstance(obj, np.ndarray):
        _write_typed_array(obj, out)
    elif isinstance(obj, (list, tuple)):
        out.write(b"[")
        for v in obj:
            _write(v, out)
        out.write(b"]")
    else:
        raise TypeError(f"cannot UBJSON-encode {type(obj)}")


_NP_TO_MARK = {
    np.dtype(np.float32): (b"d", ">f4"),
    np.dtype(np.float64): (b"D", ">f8"),
    np.dtype(np.int8): (b"i", ">i1"),
    np.dtype(np.uint8): (b"U", ">u1"),
    np.dtype(np.int16): (b"I", ">i2"),
    np.dtype(np.int32): (b"l", ">i4"),
    np.dtype(np.int64): (b"L", ">i8"),
}


def _write_typed_array(arr: np.ndarray, out: BytesIO) -> None:
    arr = np.ascontiguousarray(arr).reshape(-1)
    if arr.dtype not in _NP_TO_MARK:
        _write(arr.tolist(), out)
        return
    mark, be = _NP_TO_MARK[arr.dtype]
    out.write(b"[$" + mark + b"#")
    _write_int(arr.size, out)
    out.write(arr.astype(be).tobytes())


_SIZES = {b"i": 1, b"U": 1, b"I": 2, b"l": 4, b"L": 8, b"d": 4, b"D": 8}
_FMTS = {b"i": ">b", b"U": ">B", b"I": ">h", b"l": ">i", b"L": ">q",
         b"d": ">f", b"D": ">d"}
_NPT = {b"i": np.int8, b"U": np.uint8, b"I": np.int16, b"l": np.int32,
        b"L": np.int64, b"d": np.float32, b"D": np.float64}


def _read_int(buf, pos):
    t = bytes(buf[pos:pos + 1])
    pos += 1
    if t not in _FMTS or t in (b"d", b"D"):
        raise ValueError(f"expected integer marker, got {t!r}")
    size = _SIZES[t]
    (v,) = struct.unpack(_FMTS[t], buf[pos:pos + size])
    return int(v), pos + size


def _read_str_payload(buf, pos):
    n, pos = _read_int(buf, pos)
    s = bytes(buf[pos:pos + n]).decode("utf-8")
    return s, pos + n


def _read(buf, pos, marker=None):
    t = marker if marker is not None else bytes(buf[pos:pos + 1])
    if marker is None:
        pos += 1
    if t == b"Z":
        return None, pos
    if t == b"T":
        return True, pos
    if t == b"F":
        return False, pos
    if t == b"N":  # no-op
        return _read(buf, pos)
    if t in _FMTS:
        size = _SIZES[t]
        (v,) = struct.unpack(_FMTS[t], buf[pos:pos + size])
        if t in (b"d", b"D"):
            return float(v), pos + size
        return int(v), pos + size
    if t == b"C":
        return bytes(buf[pos:pos + 1]).decode(), pos + 1
    if t == b"S":
        return _read_str_payload(buf, pos)
    if t == b"[":
        return _read_array(buf, pos)
    if t == b"{":
        return _read_object(buf, pos)
    raise ValueError(f"unknown UBJSON marker {t!r} at {pos}")


def _read_array(buf, pos):
    elem_type = None
    count = None
    if bytes(buf[pos:pos + 1]) == b"$":
        elem_type = bytes(buf[pos + 1:pos + 2])
        pos += 2
        if bytes(buf[pos:pos + 1]) != b"#":
            raise ValueError("typed array requires count")
    if bytes(buf[pos:pos + 1]) == b"#":
        pos += 1
        count, pos = _read_int(buf, pos)
    if elem_type is not None and elem_type in _NPT:
        size = _SIZES[elem_type]
        arr = np.frombuffer(buf[pos:pos + count * size],
                            dtype=np.dtype(_NPT[elem_type]).newbyteorder(">"))
        return arr.astype(_NPT[elem_type]).tolist(), pos + count * size
    out = []
    if count is not None:
        for _ in range(count):
            v, pos = _read(buf, pos, marker=elem_type)
            out.append(v)
        return out, pos
    while bytes(buf[pos:pos + 1]) != b"]":
        v, pos = _read(buf, pos)
        out.append(v)
    return out, pos + 1


def _read_object(buf, pos):
    elem_type = None
    count = None
    if bytes(buf[pos:pos + 1]) == b"$":
        elem_type = bytes(buf[pos + 1:pos + 2])
        pos += 2
    if bytes(buf[pos:pos + 1]) == b"#":
        pos += 1
        count, pos = _read_int(buf, pos)
    out = {}
    if count is not None:
        for _ in range(count):
            k, pos = _read_str_payload(buf, pos)
            v, pos = _read(buf, pos, marker=elem_type)
            out[k] = v
        return out, pos
    while bytes(buf[pos:pos + 1]) != b"}":
        k, pos = _read_str_payload(buf, pos)
        v, pos = _read(buf, pos)
        out[k] = v
    return out, pos + 1
