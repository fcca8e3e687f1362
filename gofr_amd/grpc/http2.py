"""Minimal HTTP/2 framing + HPACK for the gRPC transport.

From-scratch subset sufficient for gRPC unary calls: frame read/write
(DATA, HEADERS, SETTINGS, PING, WINDOW_UPDATE, RST_STREAM, GOAWAY),
HPACK static table + dynamic table, integer/string primitives. Huffman
string DEcoding is implemented (clients commonly huffman-encode); our
encoder always emits raw literals (legal per RFC 7541).
"""

from __future__ import annotations

import struct

PREFACE = b"PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n"

FT_DATA = 0x0
FT_HEADERS = 0x1
FT_RST_STREAM = 0x3
FT_SETTINGS = 0x4
FT_PING = 0x6
FT_GOAWAY = 0x7
FT_WINDOW_UPDATE = 0x8
FT_CONTINUATION = 0x9

FLAG_END_STREAM = 0x1
FLAG_END_HEADERS = 0x4
FLAG_ACK = 0x1
FLAG_PADDED = 0x8
FLAG_PRIORITY = 0x20


def pack_frame(ftype: int, flags: int, stream_id: int, payload: bytes) -> bytes:
    return (len(payload).to_bytes(3, "big") + bytes([ftype, flags]) +
            struct.pack(">I", stream_id & 0x7FFFFFFF) + payload)


def read_frame(read_exact):
    hdr = read_exact(9)
    length = int.from_bytes(hdr[:3], "big")
    ftype, flags = hdr[3], hdr[4]
    stream_id = struct.unpack(">I", hdr[5:9])[0] & 0x7FFFFFFF
    payload = read_exact(length) if length else b""
    return ftype, flags, stream_id, payload


# ---------------------------------------------------------------------------
# HPACK (RFC 7541)
# ---------------------------------------------------------------------------

STATIC_TABLE = [
    (":authority", ""), (":method", "GET"), (":method", "POST"),
    (":path", "/"), (":path", "/index.html"), (":scheme", "http"),
    (":scheme", "https"), (":status", "200"), (":status", "204"),
    (":status", "206"), (":status", "304"), (":status", "400"),
    (":status", "404"), (":status", "500"), ("accept-charset", ""),
    ("accept-encoding", "gzip, deflate"), ("accept-language", ""),
    ("accept-ranges", ""), ("accept", ""), ("access-control-allow-origin", ""),
    ("age", ""), ("allow", ""), ("authorization", ""), ("cache-control", ""),
    ("content-disposition", ""), ("content-encoding", ""),
    ("content-language", ""), ("content-length", ""), ("content-location", ""),
    ("content-range", ""), ("content-type", ""), ("cookie", ""), ("date", ""),
    ("etag", ""), ("expect", ""), ("expires", ""), ("from", ""), ("host", ""),
    ("if-match", ""), ("if-modified-since", ""), ("if-none-match", ""),
    ("if-range", ""), ("if-unmodified-since", ""), ("last-modified", ""),
    ("link", ""), ("location", ""), ("max-forwards", ""),
    ("proxy-authenticate", ""), ("proxy-authorization", ""), ("range", ""),
    ("referer", ""), ("refresh", ""), ("retry-after", ""), ("server", ""),
    ("set-cookie", ""), ("strict-transport-security", ""),
    ("transfer-encoding", ""), ("user-agent", ""), ("vary", ""), ("via", ""),
    ("www-authenticate", ""),
]

# RFC 7541 Appendix B huffman code table: (code, bit-length) per symbol
# 0..256 (256 = EOS). Encoded compactly as (length, code) pairs.
_HUFF = [
    (13, 0x1ff8), (23, 0x7fffd8), (28, 0xfffffe2), (28, 0xfffffe3),
    (28, 0xfffffe4), (28, 0xfffffe5), (28, 0xfffffe6), (28, 0xfffffe7),
    (28, 0xfffffe8), (24, 0xffffea), (30, 0x3ffffffc), (28, 0xfffffe9),
    (28, 0xfffffea), (30, 0x3ffffffd), (28, 0xfffffeb), (28, 0xfffffec),
    (28, 0xfffffed), (28, 0xfffffee), (28, 0xfffffef), (28, 0xffffff0),
    (28, 0xffffff1), (28, 0xffffff2), (30, 0x3ffffffe), (28, 0xffffff3),
    (28, 0xffffff4), (28, 0xffffff5), (28, 0xffffff6), (28, 0xffffff7),
    (28, 0xffffff8), (28, 0xffffff9), (28, 0xffffffa), (28, 0xffffffb),
    (6, 0x14), (10, 0x3f8), (10, 0x3f9), (12, 0xffa), (13, 0x1ff9),
    (6, 0x15), (8, 0xf8), (11, 0x7fa), (10, 0x3fa), (10, 0x3fb), (8, 0xf9),
    (11, 0x7fb), (8, 0xfa), (6, 0x16), (6, 0x17), (6, 0x18), (5, 0x0),
    (5, 0x1), (5, 0x2), (6, 0x19), (6, 0x1a), (6, 0x1b), (6, 0x1c),
    (6, 0x1d), (6, 0x1e), (6, 0x1f), (7, 0x5c), (8, 0xfb), (15, 0x7ffc),
    (6, 0x20), (12, 0xffb), (10, 0x3fc), (13, 0x1ffa), (6, 0x21), (7, 0x5d),
    (7, 0x5e), (7, 0x5f), (7, 0x60), (7, 0x61), (7, 0x62), (7, 0x63),
    (7, 0x64), (7, 0x65), (7, 0x66), (7, 0x67), (7, 0x68), (7, 0x69),
    (7, 0x6a), (7, 0x6b), (7, 0x6c), (7, 0x6d), (7, 0x6e), (7, 0x6f),
    (7, 0x70), (7, 0x71), (7, 0x72), (8, 0xfc), (7, 0x73), (8, 0xfd),
    (13, 0x1ffb), (19, 0x7fff0), (13, 0x1ffc), (14, 0x3ffc), (6, 0x22),
    (15, 0x7ffd), (5, 0x3), (6, 0x23), (5, 0x4), (6, 0x24), (5, 0x5),
    (6, 0x25), (6, 0x26), (6, 0x27), (5, 0x6), (7, 0x74), (7, 0x75),
    (6, 0x28), (6, 0x29), (6, 0x2a), (5, 0x7), (6, 0x2b), (7, 0x76),
    (6, 0x2c), (5, 0x8), (5, 0x9), (6, 0x2d), (7, 0x77), (7, 0x78),
    (7, 0x79), (7, 0x7a), (7, 0x7b), (15, 0x7ffe), (11, 0x7fc), (14, 0x3ffd),
    (13, 0x1ffd), (28, 0xffffffc), (20, 0xfffe6), (22, 0x3fffd2),
    (20, 0xfffe7), (20, 0xfffe8), (22, 0x3fffd3), (22, 0x3fffd4),
    (22, 0x3fffd5), (23, 0x7fffd9), (22, 0x3fffd6), (23, 0x7fffda),
    (23, 0x7fffdb), (23, 0x7fffdc), (23, 0x7fffdd), (23, 0x7fffde),
    (24, 0xffffeb), (23, 0x7fffdf), (24, 0xffffec), (24, 0xffffed),
    (22, 0x3fffd7), (23, 0x7fffe0), (24, 0xffffee), (23, 0x7fffe1),
    (23, 0x7fffe2), (23, 0x7fffe3), (23, 0x7fffe4), (21, 0x1fffdc),
    (22, 0x3fffd8), (23, 0x7fffe5), (22, 0x3fffd9), (23, 0x7fffe6),
    (23, 0x7fffe7), (24, 0xffffef), (22, 0x3fffda), (21, 0x1fffdd),
    (20, 0xfffe9), (22, 0x3fffdb), (22, 0x3fffdc), (23, 0x7fffe8),
    (23, 0x7fffe9), (21, 0x1fffde), (23, 0x7fffea), (22, 0x3fffdd),
    (22, 0x3fffde), (24, 0xfffff0), (21, 0x1fffdf), (22, 0x3fffdf),
    (23, 0x7fffeb), (23, 0x7fffec), (21, 0x1fffe0), (21, 0x1fffe1),
    (22, 0x3fffe0), (21, 0x1fffe2), (23, 0x7fffed), (22, 0x3fffe1),
    (23, 0x7fffee), (23, 0x7fffef), (20, 0xfffea), (22, 0x3fffe2),
    (22, 0x3fffe3), (22, 0x3fffe4), (23, 0x7ffff0), (22, 0x3fffe5),
    (22, 0x3fffe6), (23, 0x7ffff1), (26, 0x3ffffe0), (26, 0x3ffffe1),
    (20, 0xfffeb), (19, 0x7fff1), (22, 0x3fffe7), (23, 0x7ffff2),
    (22, 0x3fffe8), (25, 0x1ffffec), (26, 0x3ffffe2), (26, 0x3ffffe3),
    (26, 0x3ffffe4), (27, 0x7ffffde), (27, 0x7ffffdf), (26, 0x3ffffe5),
    (24, 0xfffff1), (25, 0x1ffffed), (19, 0x7fff2), (21, 0x1fffe3),
    (26, 0x3ffffe6), (27, 0x7ffffe0), (27, 0x7ffffe1), (26, 0x3ffffe7),
    (27, 0x7ffffe2), (24, 0xfffff2), (21, 0x1fffe4), (21, 0x1fffe5),
    (26, 0x3ffffe8), (26, 0x3ffffe9), (28, 0xffffffd), (27, 0x7ffffe3),
    (27, 0x7ffffe4), (27, 0x7ffffe5), (20, 0xfffec), (24, 0xfffff3),
    (20, 0xfffed), (21, 0x1fffe6), (22, 0x3fffe9), (21, 0x1fffe7),
    (21, 0x1fffe8), (23, 0x7ffff3), (22, 0x3fffea), (22, 0x3fffeb),
    (25, 0x1ffffee), (25, 0x1ffffef), (24, 0xfffff4), (24, 0xfffff5),
    (26, 0x3ffffea), (23, 0x7ffff4), (26, 0x3ffffeb), (27, 0x7ffffe6),
    (26, 0x3ffffec), (26, 0x3ffffed), (27, 0x7ffffe7), (27, 0x7ffffe8),
    (27, 0x7ffffe9), (27, 0x7ffffea), (27, 0x7ffffeb), (28, 0xffffffe),
    (27, 0x7ffffec), (27, 0x7ffffed), (27, 0x7ffffee), (27, 0x7ffffef),
    (27, 0x7fffff0), (26, 0x3ffffee), (30, 0x3fffffff),
]

_HUFF_DECODE: dict[tuple[int, int], int] = {
    (bits, code): sym for sym, (bits, code) in enumerate(_HUFF)}


def huffman_decode(data: bytes) -> bytes:
    out = bytearray()
    code = 0
    bits = 0
    for byte in data:
        for k in range(7, -1, -1):
            code = (code << 1) | ((byte >> k) & 1)
            bits += 1
            sym = _HUFF_DECODE.get((bits, code))
            if sym is not None:
                if sym == 256:
                    raise ValueError("EOS in huffman string")
                out.append(sym)
                code = 0
                bits = 0
            elif bits > 30:
                raise ValueError("bad huffman code")
    # residual bits must be a prefix of EOS (all ones), <= 7 bits
    if bits > 7 or code != (1 << bits) - 1:
        raise ValueError("bad huffman padding")
    return bytes(out)


class HpackDecoder:
    def __init__(self, max_table: int = 4096):
        self.dynamic: list[tuple[str, str]] = []
        self.max_table = max_table
        self.size = 0

    def _entry(self, idx: int) -> tuple[str, str]:
        if idx <= 0:
            raise ValueError("HPACK index 0")
        if idx <= len(STATIC_TABLE):
            return STATIC_TABLE[idx - 1]
        didx = idx - len(STATIC_TABLE) - 1
        if didx >= len(self.dynamic):
            raise ValueError("HPACK index out of range")
        return self.dynamic[didx]

    def _add(self, name: str, value: str) -> None:
        self.dynamic.insert(0, (name, value))
        self.size += len(name) + len(value) + 32
        while self.size > self.max_table and self.dynamic:
            n, v = self.dynamic.pop()
            self.size -= len(n) + len(v) + 32

    @staticmethod
    def _read_int(data: bytes, pos: int, prefix: int) -> tuple[int, int]:
        mask = (1 << prefix) - 1
        v = data[pos] & mask
        pos += 1
        if v < mask:
            return v, pos
        shift = 0
        while True:
            b = data[pos]
            pos += 1
            v += (b & 0x7F) << shift
            shift += 7
            if not b & 0x80:
                return v, pos

    def _read_str(self, data: bytes, pos: int) -> tuple[str, int]:
        huff = bool(data[pos] & 0x80)
        ln, pos = self._read_int(data, pos, 7)
        raw = data[pos:pos + ln]
        pos += ln
        if huff:
            raw = huffman_decode(raw)
        return raw.decode("utf-8", "surrogateescape"), pos

    def decode(self, data: bytes) -> list[tuple[str, str]]:
        headers = []
        pos = 0
        while pos < len(data):
            b = data[pos]
            if b & 0x80:  # indexed
                idx, pos = self._read_int(data, pos, 7)
                headers.append(self._entry(idx))
            elif b & 0x40:  # literal with incremental indexing
                idx, pos = self._read_int(data, pos, 6)
                name = self._entry(idx)[0] if idx else None
                if name is None:
                    name, pos = self._read_str(data, pos)
                value, pos = self._read_str(data, pos)
                self._add(name, value)
                headers.append((name, value))
            elif b & 0x20:  # dynamic table size update
                _, pos = self._read_int(data, pos, 5)
            else:  # literal without/never indexing
                idx, pos = self._read_int(data, pos, 4)
                name = self._entry(idx)[0] if idx else None
                if name is None:
                    name, pos = self._read_str(data, pos)
                value, pos = self._read_str(data, pos)
                headers.append((name, value))
        return headers


class HpackEncoder:
    """Emits literal-without-indexing fields only (always legal)."""

    @staticmethod
    def _put_str(out: bytearray, raw: bytes) -> None:
        # string length as an HPACK 7-bit-prefix integer, no huffman
        if len(raw) < 0x7F:
            out.append(len(raw))
        else:
            out.append(0x7F)
            rest = len(raw) - 0x7F
            while rest >= 0x80:
                out.append((rest & 0x7F) | 0x80)
                rest >>= 7
            out.append(rest)
        out.extend(raw)

    @staticmethod
    def encode(headers: list[tuple[str, str]]) -> bytes:
        out = bytearray()
        for name, value in headers:
            out.append(0x00)  # literal without indexing, new name
            HpackEncoder._put_str(out, name.encode())
            HpackEncoder._put_str(out, value.encode())
        return bytes(out)
