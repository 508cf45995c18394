"""Minimal protobuf (proto2) wire + runtime, written from scratch.

The reference (yahoo/CaffeOnSpark) configures everything through `caffe.proto`
(see reference caffe-distri/Makefile:46-48 and SURVEY.md §2.5) and stores
checkpoints as binary protos (`.caffemodel` / `.solverstate`).  We need exact
wire-format compatibility without a protoc binary, so this module implements
the proto2 wire format (varint / fixed32 / fixed64 / length-delimited) and a
small message runtime driven by declarative field specs.

No code is taken from protobuf or the reference; this is a clean-room
implementation of the public wire format spec.
"""

from __future__ import annotations

import struct
from typing import Any, Dict, List, Optional, Tuple

# wire types
WT_VARINT = 0
WT_FIXED64 = 1
WT_LEN = 2
WT_FIXED32 = 5

_SCALAR_WIRE = {
    "int32": WT_VARINT, "int64": WT_VARINT, "uint32": WT_VARINT,
    "uint64": WT_VARINT, "sint32": WT_VARINT, "sint64": WT_VARINT,
    "bool": WT_VARINT, "enum": WT_VARINT,
    "fixed64": WT_FIXED64, "sfixed64": WT_FIXED64, "double": WT_FIXED64,
    "fixed32": WT_FIXED32, "sfixed32": WT_FIXED32, "float": WT_FIXED32,
    "string": WT_LEN, "bytes": WT_LEN, "message": WT_LEN,
}


def _encode_varint(buf: bytearray, value: int) -> None:
    if value < 0:
        value &= (1 << 64) - 1  # two's complement, 64-bit
    while True:
        b = value & 0x7F
        value >>= 7
        if value:
            buf.append(b | 0x80)
        else:
            buf.append(b)
            return


def _decode_varint(data: bytes, pos: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        b = data[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not (b & 0x80):
            return result, pos
        shift += 7
        if shift >= 70:
            raise ValueError("varint too long")


def _zigzag_encode(v: int) -> int:
    return (v << 1) ^ (v >> 63)


def _zigzag_decode(v: int) -> int:
    return (v >> 1) ^ -(v & 1)


class EnumType:
    """A proto enum: bidirectional name<->number map."""

    def __init__(self, name: str, values: Dict[str, int]):
        self.name = name
        self.by_name = dict(values)
        self.by_number = {v: k for k, v in values.items()}
        for k, v in values.items():
            setattr(self, k, v)

    def __contains__(self, name: str) -> bool:
        return name in self.by_name


class Field:
    __slots__ = ("number", "name", "type", "repeated", "default", "packed",
                 "msg_type", "enum_type")

    def __init__(self, number: int, name: str, type: str, *, repeated: bool = False,
                 default: Any = None, packed: bool = False,
                 msg_type: Any = None, enum_type: Optional[EnumType] = None):
        self.number = number
        self.name = name
        self.type = type
        self.repeated = repeated
        self.packed = packed
        self.msg_type = msg_type       # class or lazy string resolved later
        self.enum_type = enum_type
        if default is None and not repeated:
            default = {
                "float": 0.0, "double": 0.0, "bool": False,
                "string": "", "bytes": b"",
            }.get(type, 0 if type != "message" else None)
        self.default = default


class MessageMeta(type):
    def __new__(mcls, name, bases, ns):
        cls = super().__new__(mcls, name, bases, ns)
        fields: List[Field] = ns.get("FIELDS", [])
        cls._by_number = {f.number: f for f in fields}
        cls._by_name = {f.name: f for f in fields}
        return cls


class Message(metaclass=MessageMeta):
    FIELDS: List[Field] = []

    def __init__(self, **kwargs):
        self._values: Dict[str, Any] = {}
        for k, v in kwargs.items():
            setattr(self, k, v)

    # -- attribute access ---------------------------------------------------
    def __getattr__(self, name: str):
        # only called when not found normally
        if name.startswith("_"):
            raise AttributeError(name)
        f = type(self)._by_name.get(name)
        if f is None:
            raise AttributeError(f"{type(self).__name__} has no field {name!r}")
        vals = self.__dict__.setdefault("_values", {})
        if name in vals:
            return vals[name]
        if f.repeated:
            lst: List[Any] = []
            vals[name] = lst  # auto-vivify so callers can append
            return lst
        if f.type == "message":
            inst = f.msg_type()
            vals[name] = inst  # auto-vivify nested message (proto2-ish ergonomics)
            return inst
        return f.default

    def __setattr__(self, name: str, value: Any):
        if name.startswith("_"):
            object.__setattr__(self, name, value)
            return
        f = type(self)._by_name.get(name)
        if f is None:
            raise AttributeError(f"{type(self).__name__} has no field {name!r}")
        if f.repeated and not isinstance(value, list) and \
                not (hasattr(value, "dtype") and hasattr(value, "tobytes")):
            value = list(value)  # ndarrays pass through for the fast path
        self._values[name] = value

    def has_field(self, name: str) -> bool:
        v = self._values.get(name)
        if v is None:
            return False
        if isinstance(v, list):
            return len(v) > 0
        if isinstance(v, Message):
            # mirror the serializer: a nested message auto-vivified by a
            # mere attribute read (and still content-free) is NOT present
            # — otherwise reading solver.net_param anywhere would flip
            # Solver._resolve_net_param into building an empty net
            return bool(v._present())
        return True

    def clear_field(self, name: str) -> None:
        self._values.pop(name, None)

    def __eq__(self, other):
        if type(self) is not type(other):
            return NotImplemented
        return self._present() == other._present()

    def _present(self):
        out = {}
        for k, v in self._values.items():
            if isinstance(v, list):
                if v:
                    out[k] = [e._present() if isinstance(e, Message) else e
                              for e in v]
            elif isinstance(v, Message):
                p = v._present()
                if p or not type(self)._by_name[k].repeated:
                    out[k] = p
            else:
                out[k] = v
        return out

    def __repr__(self):
        from . import text_format
        return f"<{type(self).__name__}\n{text_format.dumps(self)}>"

    # -- binary encode ------------------------------------------------------
    def SerializeToString(self) -> bytes:
        buf = bytearray()
        for f in type(self).FIELDS:
            name = f.name
            if name not in self._values:
                continue
            v = self._values[name]
            if f.repeated:
                if not (len(v) if hasattr(v, "__len__") else True):
                    continue
                if f.packed:
                    if f.type in ("float", "double"):
                        # numpy fast path: packed floats are exactly the
                        # little-endian array bytes (snapshots carry
                        # millions of weights)
                        import numpy as _np
                        dt = "<f4" if f.type == "float" else "<f8"
                        payload = _np.asarray(v, dtype=dt).tobytes()
                    else:
                        payload = bytearray()
                        for item in v:
                            _encode_scalar(payload, f, item)
                    _encode_varint(buf, (f.number << 3) | WT_LEN)
                    _encode_varint(buf, len(payload))
                    buf += payload
                else:
                    for item in v:
                        _encode_field(buf, f, item)
            else:
                if isinstance(v, Message) and not v._values and f.type == "message":
                    # auto-vivified but untouched nested message: skip
                    continue
                _encode_field(buf, f, v)
        return bytes(buf)

    # -- binary decode ------------------------------------------------------
    @classmethod
    def FromString(cls, data: bytes) -> "Message":
        msg = cls()
        msg.MergeFromString(data)
        return msg

    def MergeFromString(self, data: bytes) -> None:
        pos = 0
        n = len(data)
        cls = type(self)
        while pos < n:
            tag, pos = _decode_varint(data, pos)
            fnum, wt = tag >> 3, tag & 7
            f = cls._by_number.get(fnum)
            if f is None:
                pos = _skip_field(data, pos, wt)
                continue
            if wt == WT_LEN and f.type not in ("string", "bytes", "message"):
                # packed repeated scalars
                length, pos = _decode_varint(data, pos)
                end = pos + length
                if f.type in ("float", "double"):
                    import numpy as _np
                    dt = "<f4" if f.type == "float" else "<f8"
                    arr = _np.frombuffer(data[pos:end], dtype=dt)
                    cur = self._values.get(f.name)
                    if cur is None or (hasattr(cur, "__len__")
                                       and len(cur) == 0):
                        self._values[f.name] = arr  # ndarray: list-like
                    else:
                        self._values[f.name] = _np.concatenate(
                            [_np.asarray(cur, dtype=dt), arr])
                    pos = end
                    continue
                lst = getattr(self, f.name)
                while pos < end:
                    val, pos = _decode_scalar(data, pos, f)
                    lst.append(val)
                continue
            val, pos = _decode_wire(data, pos, wt, f)
            if f.repeated:
                getattr(self, f.name).append(val)
            else:
                if f.type == "message" and f.name in self._values:
                    self._values[f.name]._merge(val)
                else:
                    setattr(self, f.name, val)

    def _merge(self, other: "Message") -> None:
        for k, v in other._values.items():
            f = type(self)._by_name[k]
            if f.repeated:
                if f.type == "message":
                    getattr(self, k).extend(e.clone() for e in v)
                else:
                    getattr(self, k).extend(v)
            elif f.type == "message":
                if k in self._values:
                    self._values[k]._merge(v)
                else:
                    setattr(self, k, v.clone())
            else:
                setattr(self, k, v)

    def CopyFrom(self, other: "Message") -> None:
        self._values = {}
        self._merge(other)

    def clone(self):
        c = type(self)()
        c.CopyFrom(self)
        return c


def _encode_scalar(buf: bytearray, f: Field, v: Any) -> None:
    t = f.type
    if t in ("int32", "int64", "uint32", "uint64", "enum"):
        _encode_varint(buf, int(v))
    elif t == "bool":
        _encode_varint(buf, 1 if v else 0)
    elif t in ("sint32", "sint64"):
        _encode_varint(buf, _zigzag_encode(int(v)))
    elif t == "float":
        buf += struct.pack("<f", float(v))
    elif t == "double":
        buf += struct.pack("<d", float(v))
    elif t in ("fixed32", "sfixed32"):
        buf += struct.pack("<i" if t == "sfixed32" else "<I", int(v))
    elif t in ("fixed64", "sfixed64"):
        buf += struct.pack("<q" if t == "sfixed64" else "<Q", int(v))
    else:
        raise TypeError(f"not a scalar: {t}")


def _encode_field(buf: bytearray, f: Field, v: Any) -> None:
    wt = _SCALAR_WIRE[f.type]
    _encode_varint(buf, (f.number << 3) | wt)
    t = f.type
    if t == "string":
        b = v.encode("utf-8") if isinstance(v, str) else bytes(v)
        _encode_varint(buf, len(b))
        buf += b
    elif t == "bytes":
        b = bytes(v)
        _encode_varint(buf, len(b))
        buf += b
    elif t == "message":
        b = v.SerializeToString()
        _encode_varint(buf, len(b))
        buf += b
    else:
        _encode_scalar(buf, f, v)


def _decode_scalar(data: bytes, pos: int, f: Field) -> Tuple[Any, int]:
    t = f.type
    if t in ("int32", "int64"):
        v, pos = _decode_varint(data, pos)
        if v >= 1 << 63:
            v -= 1 << 64
        return v, pos
    if t in ("uint32", "uint64", "enum"):
        return _decode_varint(data, pos)
    if t == "bool":
        v, pos = _decode_varint(data, pos)
        return bool(v), pos
    if t in ("sint32", "sint64"):
        v, pos = _decode_varint(data, pos)
        return _zigzag_decode(v), pos
    if t == "float":
        return struct.unpack_from("<f", data, pos)[0], pos + 4
    if t == "double":
        return struct.unpack_from("<d", data, pos)[0], pos + 8
    if t == "fixed32":
        return struct.unpack_from("<I", data, pos)[0], pos + 4
    if t == "sfixed32":
        return struct.unpack_from("<i", data, pos)[0], pos + 4
    if t == "fixed64":
        return struct.unpack_from("<Q", data, pos)[0], pos + 8
    if t == "sfixed64":
        return struct.unpack_from("<q", data, pos)[0], pos + 8
    raise TypeError(f"not a scalar: {t}")


def _decode_wire(data: bytes, pos: int, wt: int, f: Field) -> Tuple[Any, int]:
    if wt == WT_LEN:
        length, pos = _decode_varint(data, pos)
        chunk = data[pos:pos + length]
        pos += length
        if f.type == "string":
            return chunk.decode("utf-8", errors="replace"), pos
        if f.type == "bytes":
            return chunk, pos
        if f.type == "message":
            return f.msg_type.FromString(chunk), pos
        raise ValueError(f"unexpected LEN wire for field {f.name}")
    return _decode_scalar(data, pos, f)


def _skip_field(data: bytes, pos: int, wt: int) -> int:
    if wt == WT_VARINT:
        _, pos = _decode_varint(data, pos)
        return pos
    if wt == WT_FIXED64:
        return pos + 8
    if wt == WT_FIXED32:
        return pos + 4
    if wt == WT_LEN:
        length, pos = _decode_varint(data, pos)
        return pos + length
    raise ValueError(f"cannot skip wire type {wt}")
