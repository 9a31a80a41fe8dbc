"""Minimal protobuf (proto3) wire-format codec.

grpc_tools is not part of the image, so the kubelet contracts
(deviceplugin v1beta1, podresources v1alpha1) are implemented directly on the
wire format: declarative message specs compiled once into encode/decode
closures. The messages involved are tiny (an Allocate request is a handful of
strings), so this is also the fastest path available to Python — no descriptor
pool, no reflection per call, and gRPC is given plain ``bytes`` callables as
(de)serializers.

Supported field kinds: string, bytes, bool, int32/int64 (varint), message,
map<string,string>; any field may be repeated.
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional, Tuple

WIRE_VARINT = 0
WIRE_I64 = 1
WIRE_LEN = 2
WIRE_I32 = 5


def encode_varint(n: int) -> bytes:
    if n < 0:
        n += 1 << 64  # proto3 negative int32/int64 -> 10-byte varint
    out = bytearray()
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def decode_varint(buf: bytes, pos: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not b & 0x80:
            return result, pos
        shift += 7
        if shift >= 70:
            raise ValueError("varint too long")


def _skip(buf: bytes, pos: int, wire: int) -> int:
    if wire == WIRE_VARINT:
        _, pos = decode_varint(buf, pos)
        return pos
    if wire == WIRE_LEN:
        n, pos = decode_varint(buf, pos)
        return pos + n
    if wire == WIRE_I64:
        return pos + 8
    if wire == WIRE_I32:
        return pos + 4
    raise ValueError(f"unsupported wire type {wire}")


class MessageSpec:
    """Declarative spec: fields = [(number, name, kind, submsg_or_None, repeated)]."""

    def __init__(self, name: str, fields: List[Tuple[int, str, str, Optional["MessageSpec"], bool]]):
        self.name = name
        self.fields = fields
        self.by_number = {f[0]: f for f in fields}
        self.field_names = [f[1] for f in fields]

    # ---- encoding ----
    def encode(self, obj: Dict[str, Any]) -> bytes:
        out = bytearray()
        for number, name, kind, sub, repeated in self.fields:
            val = obj.get(name)
            if val is None:
                continue
            key_len = encode_varint(number << 3 | WIRE_LEN)
            key_varint = encode_varint(number << 3 | WIRE_VARINT)
            if kind == "map_str_str":
                for k, v in val.items():
                    entry = bytearray()
                    if k:
                        kb = k.encode()
                        entry += b"\x0a" + encode_varint(len(kb)) + kb
                    if v:
                        vb = v.encode()
                        entry += b"\x12" + encode_varint(len(vb)) + vb
                    out += key_len + encode_varint(len(entry)) + entry
                continue
            items = val if repeated else [val]
            for item in items:
                if kind == "string":
                    if item or repeated:
                        b = item.encode()
                        out += key_len + encode_varint(len(b)) + b
                elif kind == "bytes":
                    if item or repeated:
                        out += key_len + encode_varint(len(item)) + item
                elif kind == "bool":
                    if item or repeated:
                        out += key_varint + (b"\x01" if item else b"\x00")
                elif kind in ("int32", "int64"):
                    if item or repeated:
                        out += key_varint + encode_varint(item)
                elif kind == "message":
                    b = sub.encode(item)
                    out += key_len + encode_varint(len(b)) + b
                else:
                    raise ValueError(f"unknown kind {kind}")
        return bytes(out)

    # ---- decoding ----
    def decode(self, buf: bytes) -> Dict[str, Any]:
        obj: Dict[str, Any] = {}
        for number, name, kind, sub, repeated in self.fields:
            if kind == "map_str_str":
                obj[name] = {}
            elif repeated:
                obj[name] = []
            elif kind == "string":
                obj[name] = ""
            elif kind == "bytes":
                obj[name] = b""
            elif kind == "bool":
                obj[name] = False
            elif kind in ("int32", "int64"):
                obj[name] = 0
            else:
                obj[name] = None
        pos, end = 0, len(buf)
        while pos < end:
            tag, pos = decode_varint(buf, pos)
            number, wire = tag >> 3, tag & 7
            f = self.by_number.get(number)
            if f is None:
                pos = _skip(buf, pos, wire)
                continue
            _, name, kind, sub, repeated = f
            if kind == "map_str_str":
                n, pos = decode_varint(buf, pos)
                entry = buf[pos : pos + n]
                pos += n
                k = v = ""
                epos = 0
                while epos < len(entry):
                    etag, epos = decode_varint(entry, epos)
                    ln, epos = decode_varint(entry, epos)
                    s = entry[epos : epos + ln].decode()
                    epos += ln
                    if etag >> 3 == 1:
                        k = s
                    else:
                        v = s
                obj[name][k] = v
                continue
            if kind in ("string", "bytes", "message"):
                n, pos = decode_varint(buf, pos)
                raw = buf[pos : pos + n]
                pos += n
                if kind == "string":
                    item: Any = raw.decode()
                elif kind == "bytes":
                    item = raw
                else:
                    item = sub.decode(raw)
            elif kind == "bool":
                v_int, pos = decode_varint(buf, pos)
                item = bool(v_int)
            elif kind in ("int32", "int64"):
                item, pos = decode_varint(buf, pos)
                bits = 32 if kind == "int32" else 64
                if item >= 1 << (bits - 1):
                    item -= 1 << bits if bits == 64 else 1 << 64
            else:
                raise ValueError(f"unknown kind {kind}")
            if repeated:
                obj[name].append(item)
            else:
                obj[name] = item
        return obj
