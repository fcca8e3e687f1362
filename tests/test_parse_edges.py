"""Edge-case pins for the r2 parse semantics (one-header-line-per-lane
with a 63-line cap, chunk-granular classification stop at the first
\r\n\r\n, bare-LF blank lines) — the CPU mirror is byte-equality-pinned
against the kernel on GPU; these tests pin the MIRROR's behavior so a
refactor can't silently shift both sides together."""

import numpy as np

from gofr_amd import ops
from gofr_amd.http.router import Router


def parse_one(raw: bytes):
    r = Router()
    r.add("GET", "/a", lambda c: None)
    trie = r.compile()
    tab = np.asarray([[ops.HK_HOST, 0, 0, 200]], np.int32).reshape(-1)
    buf = np.frombuffer(raw, np.uint8).copy()
    fields = ops.cpu_parse_route(buf, np.asarray([0], np.int64),
                                 np.asarray([len(raw)], np.int32),
                                 trie, tab)
    return fields[0], buf


def test_max_62_headers_with_blank_at_63_parse():
    """The one-line-per-lane budget examines lines 1..63: 62 headers
    + the blank line at 63 is the densest fully-parsed shape."""
    hdrs = "".join(f"H{i}: v{i}\r\n" for i in range(61))
    raw = (f"GET /a HTTP/1.1\r\nContent-Length: 3\r\n{hdrs}"
           "\r\nxyz").encode()
    f, _ = parse_one(raw)
    assert f[ops.FI_CLEN] == 3
    assert not (f[ops.FI_FLAGS] & ops.FL_ERR_PARSE)
    assert raw[f[ops.FI_BODY_OFF]:f[ops.FI_BODY_OFF] +
               f[ops.FI_BODY_LEN]] == b"xyz"


def test_beyond_63_headers_routes_to_host():
    """>63 header lines before the blank: beyond the one-line-per-lane
    budget — the request routes to the host parser (r1 silently
    truncated the body here)."""
    hdrs = "".join(f"H{i}: v{i}\r\n" for i in range(80))
    raw = (f"GET /a HTTP/1.1\r\n{hdrs}\r\nbody").encode()
    f, _ = parse_one(raw)
    assert f[ops.FI_FLAGS] & ops.FL_NEEDS_HOST
    assert f[ops.FI_KIND] == ops.HK_HOST


def test_header_after_64th_line_routes_to_host():
    """A Content-Length buried past the 63-line budget: the request
    routes to the host parser rather than being half-parsed."""
    hdrs = "".join(f"H{i}: v{i}\r\n" for i in range(70))
    raw = (f"GET /a HTTP/1.1\r\n{hdrs}Content-Length: 5\r\n"
           "\r\nhello").encode()
    f, _ = parse_one(raw)
    assert f[ops.FI_FLAGS] & ops.FL_NEEDS_HOST


def test_bare_lf_blank_line_ends_headers():
    raw = b"GET /a HTTP/1.1\nHost: h\n\nBODY"
    f, _ = parse_one(raw)
    assert raw[f[ops.FI_BODY_OFF]:].startswith(b"BODY")


def test_crlfcrlf_inside_body_not_rescanned():
    """The classification stop at the FIRST \r\n\r\n: a second blank
    sequence inside the body must not affect header spans."""
    body = b"AAAA\r\n\r\nBBBB"
    raw = (b"GET /a HTTP/1.1\r\nContent-Length: " +
           str(len(body)).encode() + b"\r\n\r\n" + body)
    f, _ = parse_one(raw)
    got = raw[f[ops.FI_BODY_OFF]:f[ops.FI_BODY_OFF] + f[ops.FI_BODY_LEN]]
    assert got == body


def test_no_terminating_blank_line_is_parse_error():
    raw = b"GET /a HTTP/1.1\r\nHost: h\r\n"
    f, _ = parse_one(raw)
    assert f[ops.FI_FLAGS] & ops.FL_ERR_PARSE


def test_duplicate_headers_last_wins():
    raw = (b"GET /a HTTP/1.1\r\nConnection: close\r\n"
           b"Connection: keep-alive\r\nContent-Length: 1\r\n"
           b"Content-Length: 2\r\n\r\nxy")
    f, _ = parse_one(raw)
    assert f[ops.FI_FLAGS] & ops.FL_KEEP_ALIVE
    assert f[ops.FI_CLEN] == 2


def test_http10_default_close_with_opt_in():
    f, _ = parse_one(b"GET /a HTTP/1.0\r\nHost: h\r\n\r\n")
    assert not (f[ops.FI_FLAGS] & ops.FL_KEEP_ALIVE)
    f, _ = parse_one(b"GET /a HTTP/1.0\r\nConnection: keep-alive\r\n\r\n")
    assert f[ops.FI_FLAGS] & ops.FL_KEEP_ALIVE


def test_pct_path_decode_rewrites_in_place():
    raw = b"GET /%61 HTTP/1.1\r\nHost: h\r\n\r\n"
    f, buf = parse_one(raw)
    got = bytes(buf[f[ops.FI_PATH_OFF]:
                    f[ops.FI_PATH_OFF] + f[ops.FI_PATH_LEN]])
    assert got == b"/a"
    assert f[ops.FI_ROUTE] == 0  # decoded path matches the /a route
    assert not (f[ops.FI_FLAGS] & ops.FL_NEEDS_HOST)
