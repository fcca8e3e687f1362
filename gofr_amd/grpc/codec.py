"""Protobuf wire-format codec (proto3 subset), descriptor-driven.

The reference ships protoc-generated marshal code
(examples/grpc-server/grpc/hello.pb.go). Here messages are plain dicts
plus a MessageDesc describing field numbers/types — no codegen, no
protobuf runtime. Wire types implemented: varint (int32/int64/uint/bool/
enum), length-delimited (string/bytes/message/packed), fixed32/fixed64.

The GPU fast path uses k_varint_spans (native/hip/gofr_kernels.hip) to
decode the field tag/span table for a whole request batch; this module
is the host codec and the kernel's golden model.
"""

from __future__ import annotations

import struct
from typing import Any, Optional


class MessageDesc:
    """fields: {field_number: (name, type, sub_desc_or_None)}.
    type in: 'string','bytes','int32','int64','uint32','uint64','bool',
    'float','double','fixed32','fixed64','message','repeated_*'."""

    def __init__(self, name: str, fields: dict):
        self.name = name
        self.fields = fields
        self.by_name = {v[0]: (k, v[1], v[2] if len(v) > 2 else None)
                        for k, v in fields.items()}


def write_varint(out: bytearray, v: int) -> None:
    if v < 0:
        v += 1 << 64
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return


def read_varint(buf: bytes, pos: int) -> tuple[int, int]:
    result = 0
    shift = 0
    while True:
        if pos >= len(buf):
            raise ValueError("truncated varint")
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not b & 0x80:
            return result, pos
        shift += 7
        if shift > 63:
            raise ValueError("varint too long")


def _zigzag(v: int) -> int:
    return (v << 1) ^ (v >> 63)


def encode_field(out: bytearray, num: int, ftype: str, value: Any,
                 sub: Optional[MessageDesc]) -> None:
    base = ftype.replace("repeated_", "")
    values = value if ftype.startswith("repeated_") else [value]
    for v in values:
        if base in ("int32", "int64", "uint32", "uint64", "bool", "enum"):
            write_varint(out, num << 3 | 0)
            write_varint(out, int(v))
        elif base == "sint32" or base == "sint64":
            write_varint(out, num << 3 | 0)
            write_varint(out, _zigzag(int(v)))
        elif base in ("string", "bytes"):
            data = v.encode("utf-8") if isinstance(v, str) else bytes(v)
            write_varint(out, num << 3 | 2)
            write_varint(out, len(data))
            out.extend(data)
        elif base == "message":
            payload = encode_message(v, sub)
            write_varint(out, num << 3 | 2)
            write_varint(out, len(payload))
            out.extend(payload)
        elif base == "double":
            write_varint(out, num << 3 | 1)
            out.extend(struct.pack("<d", float(v)))
        elif base == "fixed64":
            write_varint(out, num << 3 | 1)
            out.extend(struct.pack("<Q", int(v)))
        elif base == "float":
            write_varint(out, num << 3 | 5)
            out.extend(struct.pack("<f", float(v)))
        elif base == "fixed32":
            write_varint(out, num << 3 | 5)
            out.extend(struct.pack("<I", int(v)))
        else:
            raise ValueError(f"unsupported field type {ftype}")


def encode_message(msg: dict, desc: MessageDesc) -> bytes:
    out = bytearray()
    for name, (num, ftype, sub) in desc.by_name.items():
        if name not in msg or msg[name] is None:
            continue
        v = msg[name]
        # proto3 default-value elision for scalars
        if not ftype.startswith("repeated_"):
            if v == "" and ftype == "string":
                continue
            if v == b"" and ftype == "bytes":
                continue
            if v == 0 and ftype not in ("string", "bytes", "message"):
                continue
        encode_field(out, num, ftype, v, sub)
    return bytes(out)


def decode_message(buf: bytes, desc: MessageDesc) -> dict:
    msg: dict = {}
    # defaults
    for num, f in desc.fields.items():
        name, ftype = f[0], f[1]
        if ftype.startswith("repeated_"):
            msg[name] = []
        elif ftype == "string":
            msg[name] = ""
        elif ftype == "bytes":
            msg[name] = b""
        elif ftype == "message":
            msg[name] = None
        elif ftype in ("float", "double"):
            msg[name] = 0.0
        else:
            msg[name] = 0
    pos = 0
    while pos < len(buf):
        tag, pos = read_varint(buf, pos)
        num, wt = tag >> 3, tag & 7
        f = desc.fields.get(num)
        if wt == 0:
            v, pos = read_varint(buf, pos)
        elif wt == 1:
            v = buf[pos:pos + 8]
            pos += 8
        elif wt == 2:
            ln, pos = read_varint(buf, pos)
            v = buf[pos:pos + ln]
            pos += ln
        elif wt == 5:
            v = buf[pos:pos + 4]
            pos += 4
        else:
            raise ValueError(f"unsupported wire type {wt}")
        if f is None:
            continue  # unknown field: skip
        name, ftype = f[0], f[1]
        sub = f[2] if len(f) > 2 else None
        base = ftype.replace("repeated_", "")
        if base == "string":
            v = v.decode("utf-8")
        elif base == "bytes":
            v = bytes(v)
        elif base == "message":
            v = decode_message(v, sub)
        elif base in ("sint32", "sint64"):
            v = (v >> 1) ^ -(v & 1)
        elif base == "int32":
            v = v - (1 << 64) if v >= (1 << 63) else v
            v = int(struct.unpack("<i", struct.pack("<I", v & 0xFFFFFFFF))[0])
        elif base == "int64":
            v = v - (1 << 64) if v >= (1 << 63) else v
        elif base == "bool":
            v = bool(v)
        elif base == "double":
            v = struct.unpack("<d", v)[0]
        elif base == "float":
            v = struct.unpack("<f", v)[0]
        elif base == "fixed64":
            v = struct.unpack("<Q", v)[0]
        elif base == "fixed32":
            v = struct.unpack("<I", v)[0]
        if ftype.startswith("repeated_"):
            msg[name].append(v)
        else:
            msg[name] = v
    return msg


# ---- the example service descriptors (hello.proto equivalent;
# reference: examples/grpc-server/grpc/hello.proto:4-14) --------------------
HELLO_REQUEST = MessageDesc("HelloRequest", {1: ("name", "string")})
HELLO_RESPONSE = MessageDesc("HelloResponse", {1: ("message", "string")})
