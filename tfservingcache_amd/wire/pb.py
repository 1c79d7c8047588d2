"""Minimal protobuf (proto3) wire-format codec.

This environment has no protoc / grpcio-tools, so the TF-Serving wire
messages are hand-declared (see messages.py) on top of this codec. The
codec implements the proto3 binary wire format:

  https://protobuf.dev/programming-guides/encoding/

Design goals:
  * byte-compatible with real TF Serving clients (field numbers taken from
    the public tensorflow/serving protos; cross-validated in
    tests/test_wire.py against python-protobuf dynamic messages),
  * unknown-field preservation, so proxied messages survive a
    decode->re-encode hop without byte loss (the reference proxies at
    message level: /root/reference/pkg/tfservingproxy/tfservingproxy.go:168-244),
  * no dependency on generated code.
"""
from __future__ import annotations

import struct
from typing import Any, List, Optional, Tuple

WIRE_VARINT = 0
WIRE_FIXED64 = 1
WIRE_LEN = 2
WIRE_FIXED32 = 5

_f32 = struct.Struct("<f")
_f64 = struct.Struct("<d")
_u32 = struct.Struct("<I")
_u64 = struct.Struct("<Q")


def write_varint(buf: bytearray, value: int) -> None:
    if value < 0:
        value &= (1 << 64) - 1  # negative int32/int64 -> 10-byte varint
    while True:
        b = value & 0x7F
        value >>= 7
        if value:
            buf.append(b | 0x80)
        else:
            buf.append(b)
            return


def read_varint(data: bytes, pos: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        if pos >= len(data):
            raise ValueError("truncated varint")
        b = data[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not (b & 0x80):
            return result, pos
        shift += 7
        if shift >= 70:
            raise ValueError("varint too long")


def _zigzag_decode_signed(value: int, bits: int) -> int:
    # plain (non-zigzag) signed interpretation of a varint
    if value >= (1 << (bits - 1)):
        value -= 1 << bits
    return value


def write_tag(buf: bytearray, field_no: int, wire_type: int) -> None:
    write_varint(buf, (field_no << 3) | wire_type)


def skip_field(data: bytes, pos: int, wire_type: int) -> int:
    if wire_type == WIRE_VARINT:
        _, pos = read_varint(data, pos)
    elif wire_type == WIRE_FIXED64:
        pos += 8
    elif wire_type == WIRE_LEN:
        n, pos = read_varint(data, pos)
        if pos + n > len(data):
            raise ValueError("truncated length-delimited field")
        pos += n
    elif wire_type == WIRE_FIXED32:
        pos += 4
    else:
        raise ValueError(f"unsupported wire type {wire_type}")
    if pos > len(data):
        raise ValueError("truncated field")
    return pos


# ---------------------------------------------------------------------------
# Field kinds. Each kind knows its wire type, how to encode one value and how
# to decode one value. Scalars use proto3 semantics (default values omitted).
# ---------------------------------------------------------------------------

class Field:
    __slots__ = ("name", "number", "kind", "msg_cls", "repeated", "packed",
                 "map_value", "default")

    def __init__(self, name, number, kind, msg_cls=None, repeated=False,
                 packed=False, map_value=None):
        self.name = name
        self.number = number
        self.kind = kind           # int32,int64,uint32,uint64,bool,enum,
                                   # float,double,string,bytes,message,map
        self.msg_cls = msg_cls     # for kind == message / map with message val
        self.repeated = repeated
        self.packed = packed
        self.map_value = map_value # 'message' | 'int64' | 'string' for maps
        if repeated or kind == "map":
            self.default = list if kind != "map" else dict
        elif kind == "message":
            self.default = lambda: None
        elif kind in ("string",):
            self.default = lambda: ""
        elif kind in ("bytes",):
            self.default = lambda: b""
        elif kind in ("float", "double"):
            self.default = lambda: 0.0
        elif kind == "bool":
            self.default = lambda: False
        else:
            self.default = lambda: 0


_VARINT_KINDS = {"int32", "int64", "uint32", "uint64", "bool", "enum"}


def _encode_scalar(buf: bytearray, kind: str, number: int, value: Any) -> None:
    if kind in _VARINT_KINDS:
        write_tag(buf, number, WIRE_VARINT)
        write_varint(buf, int(value))
    elif kind == "float":
        write_tag(buf, number, WIRE_FIXED32)
        buf += _f32.pack(value)
    elif kind == "fixed32":
        write_tag(buf, number, WIRE_FIXED32)
        buf += _u32.pack(int(value) & 0xFFFFFFFF)
    elif kind == "double":
        write_tag(buf, number, WIRE_FIXED64)
        buf += _f64.pack(value)
    elif kind == "string":
        raw = value.encode("utf-8") if isinstance(value, str) else value
        write_tag(buf, number, WIRE_LEN)
        write_varint(buf, len(raw))
        buf += raw
    elif kind == "bytes":
        write_tag(buf, number, WIRE_LEN)
        write_varint(buf, len(value))
        buf += value
    else:
        raise ValueError(f"bad scalar kind {kind}")


def _decode_scalar(kind: str, data: bytes, pos: int, wire_type: int):
    if wire_type == WIRE_VARINT:
        v, pos = read_varint(data, pos)
        if kind == "int32":
            v = _zigzag_decode_signed(v, 64)  # int32 negatives sign-extended
        elif kind == "int64":
            v = _zigzag_decode_signed(v, 64)
        elif kind == "bool":
            v = bool(v)
        return v, pos
    if wire_type == WIRE_FIXED32:
        if pos + 4 > len(data):
            raise ValueError("truncated fixed32 field")
        if kind == "float":
            return _f32.unpack_from(data, pos)[0], pos + 4
        return _u32.unpack_from(data, pos)[0], pos + 4
    if wire_type == WIRE_FIXED64:
        if pos + 8 > len(data):
            raise ValueError("truncated fixed64 field")
        if kind == "double":
            return _f64.unpack_from(data, pos)[0], pos + 8
        return _u64.unpack_from(data, pos)[0], pos + 8
    if wire_type == WIRE_LEN:
        n, pos = read_varint(data, pos)
        if pos + n > len(data):
            raise ValueError("truncated length-delimited field")
        if kind == "string":
            raw = data[pos:pos + n]
            pos += n
            return raw.decode("utf-8", errors="surrogateescape"), pos
        # bytes fields are returned zero-copy for large payloads
        # (tensor_content on the predict hot path): a memoryview keeps
        # the source buffer alive and feeds np.frombuffer directly
        if n > 4096:
            raw = memoryview(data)[pos:pos + n]
        else:
            raw = data[pos:pos + n]
        pos += n
        return raw, pos
    raise ValueError(f"wire type {wire_type} for kind {kind}")


class MessageMeta(type):
    def __new__(mcls, name, bases, ns):
        cls = super().__new__(mcls, name, bases, ns)
        fields: List[Field] = []
        for spec in ns.get("FIELDS", ()):
            fields.append(Field(*spec[:3], **(spec[3] if len(spec) > 3 else {})))
        cls._fields = fields
        cls._by_number = {f.number: f for f in fields}
        cls._by_name = {f.name: f for f in fields}
        slots = [f.name for f in fields] + ["_unknown"]
        return cls


class Message(metaclass=MessageMeta):
    """Base class. Subclasses declare FIELDS = [(name, number, kind, opts?)]."""
    FIELDS: List[tuple] = []

    def __init__(self, **kwargs):
        for f in self._fields:
            if f.name in kwargs:
                setattr(self, f.name, kwargs.pop(f.name))
            else:
                setattr(self, f.name, f.default())
        self._unknown = b""
        if kwargs:
            raise TypeError(f"unknown fields for {type(self).__name__}: {list(kwargs)}")

    # -- encode ------------------------------------------------------------
    def encode(self) -> bytes:
        buf = bytearray()
        self.encode_into(buf)
        return bytes(buf)

    def encode_into(self, buf: bytearray) -> None:
        for f in self._fields:
            value = getattr(self, f.name)
            if f.kind == "map":
                for k, v in value.items():
                    entry = bytearray()
                    _encode_scalar(entry, "string", 1, k)
                    if f.map_value == "message":
                        sub = bytearray()
                        v.encode_into(sub)
                        write_tag(entry, 2, WIRE_LEN)
                        write_varint(entry, len(sub))
                        entry += sub
                    elif f.map_value in _VARINT_KINDS:
                        write_tag(entry, 2, WIRE_VARINT)
                        write_varint(entry, int(v))
                    else:
                        _encode_scalar(entry, f.map_value, 2, v)
                    write_tag(buf, f.number, WIRE_LEN)
                    write_varint(buf, len(entry))
                    buf += entry
            elif f.repeated:
                if not value:
                    continue
                if f.packed and f.kind in _VARINT_KINDS:
                    body = bytearray()
                    for v in value:
                        write_varint(body, int(v))
                    write_tag(buf, f.number, WIRE_LEN)
                    write_varint(buf, len(body))
                    buf += body
                elif f.packed and f.kind == "float":
                    write_tag(buf, f.number, WIRE_LEN)
                    write_varint(buf, 4 * len(value))
                    for v in value:
                        buf += _f32.pack(v)
                elif f.packed and f.kind == "double":
                    write_tag(buf, f.number, WIRE_LEN)
                    write_varint(buf, 8 * len(value))
                    for v in value:
                        buf += _f64.pack(v)
                elif f.kind == "message":
                    for v in value:
                        sub = bytearray()
                        v.encode_into(sub)
                        write_tag(buf, f.number, WIRE_LEN)
                        write_varint(buf, len(sub))
                        buf += sub
                else:
                    for v in value:
                        _encode_scalar(buf, f.kind, f.number, v)
            elif f.kind == "message":
                if value is not None:
                    sub = bytearray()
                    value.encode_into(sub)
                    write_tag(buf, f.number, WIRE_LEN)
                    write_varint(buf, len(sub))
                    buf += sub
            else:
                # proto3: skip default values
                if f.kind in ("string",) and value == "":
                    continue
                if f.kind == "bytes" and value == b"":
                    continue
                if f.kind in _VARINT_KINDS and int(value) == 0:
                    continue
                if f.kind in ("float", "double") and value == 0.0:
                    continue
                _encode_scalar(buf, f.kind, f.number, value)
        if self._unknown:
            buf += self._unknown

    # -- decode ------------------------------------------------------------
    @classmethod
    def decode(cls, data: bytes, pos: int = 0, end: Optional[int] = None):
        msg = cls()
        if end is None:
            end = len(data)
        unknown = None
        while pos < end:
            tag, pos = read_varint(data, pos)
            field_no = tag >> 3
            wire_type = tag & 7
            f = cls._by_number.get(field_no)
            if f is None:
                start = pos
                pos = skip_field(data, pos, wire_type)
                if unknown is None:
                    unknown = bytearray()
                write_tag(unknown, field_no, wire_type)
                unknown += data[start:pos]
                continue
            if f.kind == "map":
                n, pos = read_varint(data, pos)
                if pos + n > end:
                    raise ValueError("truncated map entry")
                entry_end = pos + n
                k, v = "", None
                epos = pos
                while epos < entry_end:
                    etag, epos = read_varint(data, epos)
                    eno, ewt = etag >> 3, etag & 7
                    if eno == 1:
                        k, epos = _decode_scalar("string", data, epos, ewt)
                    elif eno == 2:
                        if f.map_value == "message":
                            vn, epos = read_varint(data, epos)
                            v = f.msg_cls.decode(data, epos, epos + vn)
                            epos += vn
                        else:
                            v, epos = _decode_scalar(f.map_value, data, epos, ewt)
                    else:
                        epos = skip_field(data, epos, ewt)
                getattr(msg, f.name)[k] = v
                pos = entry_end
            elif f.kind == "message":
                n, pos = read_varint(data, pos)
                if pos + n > end:
                    raise ValueError("truncated nested message")
                sub = f.msg_cls.decode(data, pos, pos + n)
                pos += n
                if f.repeated:
                    getattr(msg, f.name).append(sub)
                else:
                    setattr(msg, f.name, sub)
            elif f.repeated:
                if wire_type == WIRE_LEN and f.kind in _VARINT_KINDS:
                    n, pos = read_varint(data, pos)
                    stop = pos + n
                    lst = getattr(msg, f.name)
                    while pos < stop:
                        v, pos = read_varint(data, pos)
                        if f.kind in ("int32", "int64"):
                            v = _zigzag_decode_signed(v, 64)
                        elif f.kind == "bool":
                            v = bool(v)
                        lst.append(v)
                elif wire_type == WIRE_LEN and f.kind == "float":
                    n, pos = read_varint(data, pos)
                    if pos + n > len(data):
                        raise ValueError("truncated packed float field")
                    lst = getattr(msg, f.name)
                    for i in range(n // 4):
                        lst.append(_f32.unpack_from(data, pos + 4 * i)[0])
                    pos += n
                elif wire_type == WIRE_LEN and f.kind == "double":
                    n, pos = read_varint(data, pos)
                    if pos + n > len(data):
                        raise ValueError("truncated packed double field")
                    lst = getattr(msg, f.name)
                    for i in range(n // 8):
                        lst.append(_f64.unpack_from(data, pos + 8 * i)[0])
                    pos += n
                else:
                    v, pos = _decode_scalar(f.kind, data, pos, wire_type)
                    getattr(msg, f.name).append(v)
            else:
                v, pos = _decode_scalar(f.kind, data, pos, wire_type)
                setattr(msg, f.name, v)
        if unknown:
            msg._unknown = bytes(unknown)
        return msg

    # -- misc --------------------------------------------------------------
    def __repr__(self):
        parts = []
        for f in self._fields:
            v = getattr(self, f.name)
            if v in (None, "", b"", 0, 0.0, False) or v == [] or v == {}:
                continue
            parts.append(f"{f.name}={v!r}")
        return f"{type(self).__name__}({', '.join(parts)})"

    def __eq__(self, other):
        if type(self) is not type(other):
            return NotImplemented
        return self.encode() == other.encode()

    def __hash__(self):
        return hash(self.encode())
