"""Schema-driven protobuf messages with binary wire-format encode/decode.

Replaces the reference's generated protobuf C++ classes
(/root/reference/src/caffe/proto/caffe.proto) with a small, dependency-free
implementation keyed off spec.py. Binary output is byte-compatible with
proto2 serialization of the same schema (fields emitted in field-number
order), so .caffemodel files interoperate.
"""

from __future__ import annotations

import struct
from typing import Any, Dict, List

import numpy as np

from . import spec

_WIRE_VARINT = 0
_WIRE_64BIT = 1
_WIRE_LEN = 2
_WIRE_32BIT = 5

_VARINT_KINDS = {"int32", "int64", "uint32", "uint64", "bool"}


def _wire_type(kind: str) -> int:
    if kind in _VARINT_KINDS or kind.startswith("enum:"):
        return _WIRE_VARINT
    if kind == "float":
        return _WIRE_32BIT
    if kind == "double":
        return _WIRE_64BIT
    return _WIRE_LEN  # string, bytes, msg


def _encode_varint(out: bytearray, value: int) -> None:
    if value < 0:
        value &= (1 << 64) - 1  # two's-complement, 10 bytes (proto2 int32/64)
    while True:
        b = value & 0x7F
        value >>= 7
        if value:
            out.append(b | 0x80)
        else:
            out.append(b)
            return


def _decode_varint(buf: memoryview, pos: int):
    result = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not (b & 0x80):
            return result, pos
        shift += 7
        if shift > 70:
            raise ValueError("malformed varint")


def _signed32(v: int) -> int:
    v &= (1 << 64) - 1
    v &= 0xFFFFFFFFFFFFFFFF
    if v >= 1 << 63:
        v -= 1 << 64
    return int(v)


class Message:
    """One protobuf message instance. Fields are attributes.

    Optional scalar fields read as their declared default when unset;
    repeated fields read as lists (packed float repeated fields may hold a
    numpy float32 array for efficiency).
    """

    __slots__ = ("_type", "_fields", "_values")

    def __init__(self, type_name: str, **kwargs):
        fields = spec.MESSAGES.get(type_name)
        if fields is None:
            raise KeyError(f"unknown message type {type_name!r}")
        object.__setattr__(self, "_type", type_name)
        object.__setattr__(self, "_fields", fields)
        object.__setattr__(self, "_values", {})
        for k, v in kwargs.items():
            setattr(self, k, v)

    # -- attribute protocol ------------------------------------------------
    def __getattr__(self, name: str):
        fields = object.__getattribute__(self, "_fields")
        if name not in fields:
            raise AttributeError(f"{self._type} has no field {name!r}")
        values = object.__getattribute__(self, "_values")
        if name in values:
            return values[name]
        num, kind, label, default = fields[name]
        if label in ("rep", "packed"):
            lst: List[Any] = []
            values[name] = lst  # autovivify so callers can append
            return lst
        if kind.startswith("msg:") and default is None:
            return None
        return default

    def __setattr__(self, name: str, value: Any) -> None:
        fields = object.__getattribute__(self, "_fields")
        if name not in fields:
            raise AttributeError(f"{self._type} has no field {name!r}")
        num, kind, label, default = fields[name]
        if kind.startswith("enum:") and isinstance(value, str):
            value = spec.ENUMS[kind[5:]][value]
        object.__getattribute__(self, "_values")[name] = value

    # -- helpers -----------------------------------------------------------
    @property
    def type_name(self) -> str:
        return self._type

    def has(self, name: str) -> bool:
        v = object.__getattribute__(self, "_values").get(name)
        if v is None:
            return False
        num, kind, label, default = self._fields[name]
        if label in ("rep", "packed"):
            return len(v) > 0
        return True

    def clear(self, name: str) -> None:
        object.__getattribute__(self, "_values").pop(name, None)

    def add(self, field_, **kwargs) -> "Message":
        """Append a new submessage to a repeated message field."""
        num, kind, label, default = self._fields[field_]
        assert kind.startswith("msg:") and label == "rep"
        m = Message(kind[4:], **kwargs)
        getattr(self, field_).append(m)
        return m

    def ensure(self, name: str) -> "Message":
        """Get-or-create an optional submessage field."""
        num, kind, label, default = self._fields[name]
        assert kind.startswith("msg:")
        v = object.__getattribute__(self, "_values").get(name)
        if v is None:
            v = Message(kind[4:])
            object.__getattribute__(self, "_values")[name] = v
        return v

    def enum_name(self, field: str) -> str:
        num, kind, label, default = self._fields[field]
        assert kind.startswith("enum:")
        table = spec.ENUMS[kind[5:]]
        val = getattr(self, field)
        for sym, v in table.items():
            if v == val:
                return sym
        raise ValueError(f"{field}: unknown enum value {val}")

    def copy(self) -> "Message":
        return Message.decode(self._type, self.encode())

    def merge_from(self, other: "Message") -> None:
        """proto2 MergeFrom: singular set fields overwrite (submessages merge
        recursively), repeated fields concatenate."""
        assert other._type == self._type
        for name, v in object.__getattribute__(other, "_values").items():
            num, kind, label, default = self._fields[name]
            if label in ("rep", "packed"):
                cur = getattr(self, name)
                if isinstance(cur, np.ndarray) or isinstance(v, np.ndarray):
                    merged = np.concatenate(
                        [np.asarray(cur, dtype=np.float32).ravel(),
                         np.asarray(v, dtype=np.float32).ravel()])
                    object.__getattribute__(self, "_values")[name] = merged
                else:
                    cur.extend(v)
            elif kind.startswith("msg:"):
                mine = object.__getattribute__(self, "_values").get(name)
                if mine is None:
                    object.__getattribute__(self, "_values")[name] = v.copy()
                else:
                    mine.merge_from(v)
            else:
                object.__getattribute__(self, "_values")[name] = v

    def __repr__(self) -> str:
        from .text_format import to_text
        return f"<{self._type}\n{to_text(self)}>"

    # -- binary encode -----------------------------------------------------
    def encode(self) -> bytes:
        out = bytearray()
        values = object.__getattribute__(self, "_values")
        # emit in field-number order for canonical output
        items = sorted(
            ((self._fields[n][0], n, v) for n, v in values.items()),
            key=lambda t: t[0])
        for num, name, v in items:
            _, kind, label, _ = self._fields[name]
            if label == "packed":
                arr = np.asarray(v, dtype=np.float32).ravel()
                if arr.size == 0:
                    continue
                _encode_varint(out, (num << 3) | _WIRE_LEN)
                raw = arr.tobytes()  # little-endian f32 == wire format
                _encode_varint(out, len(raw))
                out += raw
            elif label == "rep":
                for item in (v.tolist() if isinstance(v, np.ndarray) else v):
                    self._encode_one(out, num, kind, item)
            else:
                self._encode_one(out, num, kind, v)
        return bytes(out)

    @staticmethod
    def _encode_one(out: bytearray, num: int, kind: str, v: Any) -> None:
        wt = _wire_type(kind)
        _encode_varint(out, (num << 3) | wt)
        if wt == _WIRE_VARINT:
            _encode_varint(out, int(v))
        elif wt == _WIRE_32BIT:
            out += struct.pack("<f", float(v))
        elif wt == _WIRE_64BIT:
            out += struct.pack("<d", float(v))
        else:
            if isinstance(v, Message):
                raw = v.encode()
            elif isinstance(v, str):
                raw = v.encode("utf-8")
            else:
                raw = bytes(v)
            _encode_varint(out, len(raw))
            out += raw

    # -- binary decode -----------------------------------------------------
    @staticmethod
    def decode(type_name: str, data) -> "Message":
        msg = Message(type_name)
        msg._merge_wire(memoryview(bytes(data)), 0, len(data))
        return msg

    def _merge_wire(self, buf: memoryview, pos: int, end: int) -> None:
        by_num = {f[0]: (n, f[1], f[2]) for n, f in self._fields.items()}
        values = object.__getattribute__(self, "_values")
        while pos < end:
            tag, pos = _decode_varint(buf, pos)
            num, wt = tag >> 3, tag & 7
            field = by_num.get(num)
            if field is None:
                pos = _skip(buf, pos, wt)
                continue
            name, kind, label = field
            if wt == _WIRE_VARINT:
                raw, pos = _decode_varint(buf, pos)
                if kind in ("int32", "int64"):
                    val: Any = _signed32(raw)
                elif kind == "bool":
                    val = bool(raw)
                else:
                    val = raw
                self._store(values, name, label, val)
            elif wt == _WIRE_32BIT:
                val = struct.unpack_from("<f", buf, pos)[0]
                pos += 4
                self._store(values, name, label, val)
            elif wt == _WIRE_64BIT:
                val = struct.unpack_from("<d", buf, pos)[0]
                pos += 8
                self._store(values, name, label, val)
            elif wt == _WIRE_LEN:
                ln, pos = _decode_varint(buf, pos)
                chunk = buf[pos:pos + ln]
                pos += ln
                if kind.startswith("msg:"):
                    sub = Message(kind[4:])
                    sub._merge_wire(buf, pos - ln, pos)
                    self._store(values, name, label, sub)
                elif kind == "string":
                    self._store(values, name, label,
                                bytes(chunk).decode("utf-8", "replace"))
                elif kind == "bytes":
                    self._store(values, name, label, bytes(chunk))
                elif kind == "float":
                    # packed repeated floats
                    arr = np.frombuffer(bytes(chunk), dtype="<f4")
                    cur = values.get(name)
                    if cur is not None and len(cur):
                        arr = np.concatenate(
                            [np.asarray(cur, np.float32).ravel(), arr])
                    values[name] = arr
                else:
                    # packed varints
                    p2, e2 = pos - ln, pos
                    while p2 < e2:
                        raw, p2 = _decode_varint(buf, p2)
                        if kind in ("int32", "int64"):
                            raw = _signed32(raw)
                        self._store(values, name, label, raw)
            else:
                raise ValueError(f"bad wire type {wt}")

    @staticmethod
    def _store(values: Dict[str, Any], name: str, label: str, val: Any) -> None:
        if label in ("rep", "packed"):
            cur = values.get(name)
            if cur is None:
                values[name] = [val]
            elif isinstance(cur, np.ndarray):
                values[name] = np.append(cur, np.float32(val))
            else:
                cur.append(val)
        else:
            values[name] = val


def _skip(buf: memoryview, pos: int, wt: int) -> int:
    if wt == _WIRE_VARINT:
        _, pos = _decode_varint(buf, pos)
    elif wt == _WIRE_32BIT:
        pos += 4
    elif wt == _WIRE_64BIT:
        pos += 8
    elif wt == _WIRE_LEN:
        ln, pos = _decode_varint(buf, pos)
        pos += ln
    else:
        raise ValueError(f"cannot skip wire type {wt}")
    return pos
