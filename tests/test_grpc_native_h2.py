"""HTTP/2 edge coverage for the NATIVE h2c ingress: huffman-encoded
HPACK strings, incremental-indexing (dynamic table) reuse, and
CONTINUATION-split header blocks — the C++ decoder paths the standard
client (raw literals, single HEADERS) never exercises."""

import socket
import struct

import pytest

import gofr_amd
from gofr_amd.config import MapConfig
from gofr_amd.grpc import http2 as h2
from gofr_amd.grpc.codec import HELLO_REQUEST, encode_message
from gofr_amd.grpc.server import GRPCServer, ServiceDesc


class HelloImpl:
    def SayHello(self, ctx, req):
        return {"message": f"Hello {req.get('name') or 'World'}!"}


def huffman_encode(data: bytes) -> bytes:
    """RFC 7541 §5.2 huffman encoding (test-side; the server's C++
    decoder is under test)."""
    bits = 0
    nbits = 0
    out = bytearray()
    for b in data:
        ln, code = h2._HUFF[b]
        bits = (bits << ln) | code
        nbits += ln
        while nbits >= 8:
            out.append((bits >> (nbits - 8)) & 0xFF)
            nbits -= 8
    if nbits:
        pad = 8 - nbits
        out.append(((bits << pad) | ((1 << pad) - 1)) & 0xFF)
    return bytes(out)


def hpack_str(raw: bytes, huffman: bool) -> bytes:
    if huffman:
        enc = huffman_encode(raw)
        assert len(enc) < 127
        return bytes([0x80 | len(enc)]) + enc
    assert len(raw) < 127
    return bytes([len(raw)]) + raw


def hpack_literal_incremental(name: bytes, value: bytes,
                              huffman: bool) -> bytes:
    # 0x40: literal with incremental indexing, new name
    return b"\x40" + hpack_str(name, huffman) + hpack_str(value, huffman)


def hpack_indexed(idx: int) -> bytes:
    assert idx < 127
    return bytes([0x80 | idx])


REQ_HEADERS = [(b":method", b"POST"), (b":scheme", b"http"),
               (b":path", b"/hello.HelloService/SayHello"),
               (b":authority", b"localhost"),
               (b"content-type", b"application/grpc"),
               (b"te", b"trailers")]


@pytest.fixture()
def native_srv():
    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    desc = ServiceDesc("hello.HelloService",
                       {"SayHello": (HELLO_REQUEST, HELLO_RESPONSE)},
                       gpu_methods={"SayHello": "hello_echo"})
    app.RegisterService(desc, HelloImpl())
    srv = GRPCServer(app, 0, batch_window_us=2000, native=True)
    srv.start()
    yield srv
    srv.stop()


from gofr_amd.grpc.codec import HELLO_RESPONSE  # noqa: E402


class RawH2:
    """Byte-level h2 client for crafting specific frame shapes."""

    def __init__(self, port):
        self.s = socket.create_connection(("127.0.0.1", port), timeout=10)
        self.s.sendall(h2.PREFACE)
        self.s.sendall(h2.pack_frame(h2.FT_SETTINGS, 0, 0, b""))
        self.buf = bytearray()
        self.dec = h2.HpackDecoder()

    def read_exact(self, n):
        while len(self.buf) < n:
            chunk = self.s.recv(65536)
            if not chunk:
                raise ConnectionError("closed")
            self.buf.extend(chunk)
        out = bytes(self.buf[:n])
        del self.buf[:n]
        return out

    def roundtrip(self, sid, frames, want_name):
        for f in frames:
            self.s.sendall(f)
        msg = None
        status = -1
        while True:
            ft, fl, fsid, payload = h2.read_frame(self.read_exact)
            if ft == h2.FT_SETTINGS and not fl & h2.FLAG_ACK:
                self.s.sendall(h2.pack_frame(h2.FT_SETTINGS,
                                             h2.FLAG_ACK, 0, b""))
            if fsid != sid:
                continue
            if ft == h2.FT_HEADERS:
                hdrs = dict(self.dec.decode(payload))
                if "grpc-status" in hdrs:
                    status = int(hdrs["grpc-status"])
                if fl & h2.FLAG_END_STREAM:
                    return msg, status
            elif ft == h2.FT_DATA and len(payload) >= 5:
                mlen = struct.unpack(">I", payload[1:5])[0]
                from gofr_amd.grpc.codec import decode_message
                msg = decode_message(payload[5:5 + mlen], HELLO_RESPONSE)

    def close(self):
        self.s.close()


def grpc_frame(name):
    payload = encode_message({"name": name} if name else {},
                             HELLO_REQUEST)
    return bytes([0]) + struct.pack(">I", len(payload)) + payload


def test_huffman_and_dynamic_table(native_srv):
    c = RawH2(native_srv.port)
    # request 1: every header literal-with-incremental-indexing,
    # huffman-encoded strings -> exercises the C++ huffman decoder AND
    # populates the dynamic table
    blk = b"".join(hpack_literal_incremental(n, v, huffman=True)
                   for n, v in REQ_HEADERS)
    frames = [h2.pack_frame(h2.FT_HEADERS, h2.FLAG_END_HEADERS, 1, blk),
              h2.pack_frame(h2.FT_DATA, h2.FLAG_END_STREAM, 1,
                            grpc_frame("huff"))]
    msg, status = c.roundtrip(1, frames, "huff")
    assert status == 0 and msg == {"message": "Hello huff!"}
    # request 2: the SAME headers via dynamic-table indexed refs
    # (entries 62..67, most-recent-first -> te is 62, :method is 67)
    blk2 = b"".join(hpack_indexed(67 - i) for i in range(len(REQ_HEADERS)))
    frames = [h2.pack_frame(h2.FT_HEADERS, h2.FLAG_END_HEADERS, 3, blk2),
              h2.pack_frame(h2.FT_DATA, h2.FLAG_END_STREAM, 3,
                            grpc_frame("dyn"))]
    msg, status = c.roundtrip(3, frames, "dyn")
    assert status == 0 and msg == {"message": "Hello dyn!"}
    c.close()


def test_continuation_split_headers(native_srv):
    c = RawH2(native_srv.port)
    blk = b"".join(hpack_literal_incremental(n, v, huffman=False)
                   for n, v in REQ_HEADERS)
    half = len(blk) // 2
    frames = [
        h2.pack_frame(h2.FT_HEADERS, 0, 1, blk[:half]),  # no END_HEADERS
        h2.pack_frame(h2.FT_CONTINUATION, h2.FLAG_END_HEADERS, 1,
                      blk[half:]),
        h2.pack_frame(h2.FT_DATA, h2.FLAG_END_STREAM, 1,
                      grpc_frame("cont")),
    ]
    msg, status = c.roundtrip(1, frames, "cont")
    assert status == 0 and msg == {"message": "Hello cont!"}
    c.close()


def test_padded_headers_and_data(native_srv):
    c = RawH2(native_srv.port)
    blk = b"".join(hpack_literal_incremental(n, v, huffman=False)
                   for n, v in REQ_HEADERS)
    pad = 7
    hdr_payload = bytes([pad]) + blk + b"\0" * pad
    data = grpc_frame("padded")
    data_payload = bytes([pad]) + data + b"\0" * pad
    frames = [
        h2.pack_frame(h2.FT_HEADERS,
                      h2.FLAG_END_HEADERS | h2.FLAG_PADDED, 1,
                      hdr_payload),
        h2.pack_frame(h2.FT_DATA, h2.FLAG_END_STREAM | h2.FLAG_PADDED, 1,
                      data_payload),
    ]
    msg, status = c.roundtrip(1, frames, "padded")
    assert status == 0 and msg == {"message": "Hello padded!"}
    c.close()
