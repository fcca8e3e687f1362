"""Property-based tests (hypothesis) for the byte-level codecs whose
GPU kernels are pinned to these mirrors by byte-equality tests:
- gzip_static_mirror: any body must round-trip through zlib's gunzip
  (the mirror is the golden model of deflate_gzip_wave);
- the protobuf codec: encode(decode) identity over descriptor-driven
  messages (golden model of k_varint_spans' span table)."""

import gzip

from hypothesis import given, settings, strategies as st

from gofr_amd.grpc.codec import MessageDesc, decode_message, encode_message
from gofr_amd.ops import gzip_static_mirror

# GOFR_FUZZ_EXAMPLES raises the example count for deep campaigns
# (CI/default stays fast at 120)
import os  # noqa: E402

SET = settings(max_examples=int(os.environ.get("GOFR_FUZZ_EXAMPLES",
                                               "120")),
               deadline=None)


@SET
@given(st.binary(min_size=1, max_size=3000))
def test_gzip_mirror_roundtrips_any_bytes(data):
    gz = gzip_static_mirror(data)
    if gz is None:
        return  # over cap: serving path falls back to identity
    assert gzip.decompress(gz) == data


@SET
@given(st.text(max_size=200), st.integers(0, 50))
def test_gzip_mirror_roundtrips_repetitive_json(s, rep):
    # repetitive JSON-shaped payloads exercise the LZ77 match path
    data = (b'{"data":' + (s.encode("utf-8", "surrogatepass")
                           or b"x") * (rep + 1) + b"}")
    gz = gzip_static_mirror(data)
    if gz is None:
        return
    assert gzip.decompress(gz) == data


MSG = MessageDesc("M", {
    1: ("name", "string"),
    2: ("count", "int64"),
    3: ("flag", "bool"),
    4: ("blob", "bytes"),
    5: ("ratio", "double"),
})


@SET
@given(st.text(max_size=100),
       st.integers(min_value=-(2 ** 63), max_value=2 ** 63 - 1),
       st.booleans(),
       st.binary(max_size=100),
       st.floats(allow_nan=False, allow_infinity=False))
def test_protobuf_codec_roundtrip(name, count, flag, blob, ratio):
    msg = {"name": name, "count": count, "flag": flag, "blob": blob,
           "ratio": ratio}
    wire = encode_message(msg, MSG)
    out = decode_message(wire, MSG)
    # proto3 semantics: zero values are omitted from the wire and read
    # back as defaults
    assert out.get("name", "") == name
    assert out.get("count", 0) == count
    assert out.get("flag", False) == flag
    assert out.get("blob", b"") == blob
    assert out.get("ratio", 0.0) == ratio


@SET
@given(st.lists(st.tuples(
    st.text(alphabet=st.characters(min_codepoint=97, max_codepoint=122),
            min_size=1, max_size=300),
    st.text(max_size=400)), min_size=0, max_size=8))
def test_hpack_roundtrip(headers):
    from gofr_amd.grpc.http2 import HpackDecoder, HpackEncoder
    wire = HpackEncoder.encode(headers)
    out = HpackDecoder().decode(wire)
    assert out == headers


SEG = st.sampled_from(["a", "b", "c", "{x}", "{y}"])


@SET
@given(st.lists(st.tuples(st.sampled_from(["GET", "POST"]),
                          st.lists(SEG, min_size=1, max_size=3)),
                min_size=1, max_size=6, unique_by=lambda r: (r[0],
                                                            tuple(r[1]))),
       st.booleans(),
       st.lists(st.lists(st.sampled_from(["a", "b", "c", "d"]),
                         max_size=4), min_size=1, max_size=8))
def test_router_table_matches_python_match(routes, add_prefix, paths):
    """The compiled flat table (what k_parse_route walks) must encode
    the same decisions as Router.match for ANY route set — the
    hypothesis generalization of the fixed fuzz in test_router."""
    from gofr_amd.http.request import METHOD_IDS
    from gofr_amd.http.router import Router

    r = Router()
    seen_patterns = set()
    for method, segs in routes:
        pattern = "/" + "/".join(segs)
        if (method, pattern) in seen_patterns:
            continue
        seen_patterns.add((method, pattern))
        try:
            r.add(method, pattern, lambda c: None)
        except ValueError:
            continue  # conflicting param names at one level etc.
    if add_prefix:
        r.add_prefix("GET", "/", lambda c: None)
    t = r.compile()

    def table_match(method, path):
        node = 0
        mid = METHOD_IDS[method]
        best_prefix = t["node_prefix"][0]
        for seg in [s for s in path.strip("/").split("/") if s]:
            sb = seg.encode()
            nxt = -1
            f, c = t["node_child_first"][node], t["node_child_count"][node]
            for ci in range(f, f + c):
                off, ln = t["child_seg_off"][ci], t["child_seg_len"][ci]
                if bytes(t["seg_blob"][off:off + ln]) == sb:
                    nxt = t["child_node"][ci]
                    break
            if nxt < 0 and t["node_param"][node] >= 0:
                nxt = t["node_param"][node]
            if nxt < 0:
                return int(best_prefix)
            node = nxt
            if t["node_prefix"][node] >= 0:
                best_prefix = t["node_prefix"][node]
        rid = t["node_route"][node * 8 + mid]
        return int(rid) if rid >= 0 else int(best_prefix)

    for segs in paths:
        path = "/" + "/".join(segs)
        for method in ("GET", "POST"):
            route, _, _ = r.match(method, path)
            want = route.route_id if route else -1
            assert table_match(method, path) == want, (method, path)


@SET
@given(st.lists(st.binary(min_size=1, max_size=200), min_size=0,
                max_size=6),
       st.booleans())
def test_chunked_decode_roundtrip(chunks, with_trailer):
    # chunks are >=1 byte: a zero-size chunk IS the stream terminator
    from gofr_amd.http.request import decode_chunked
    wire = b""
    for c in chunks:
        wire += f"{len(c):x}".encode() + b"\r\n" + c + b"\r\n"
    wire += b"0\r\n"
    if with_trailer:
        wire += b"X-Trailer: v\r\n"
    wire += b"\r\n"
    assert decode_chunked(wire) == b"".join(chunks)


@SET
@given(st.sampled_from(["GET", "POST", "PUT", "DELETE", "PATCH",
                        "OPTIONS", "HEAD"]),
       st.lists(st.sampled_from(["a", "bb", "ccc"]), max_size=3),
       st.one_of(st.none(), st.sampled_from(["k=v", "q=a+b&x=1"])),
       st.binary(max_size=120))
def test_parse_mirror_agrees_with_host_parser(method, segs, query, body):
    """cpu_parse_route (the kernel's golden model) and
    parse_request_bytes (the host parser) must agree on method, path,
    query and body spans for any well-formed request."""
    import numpy as np

    from gofr_amd import ops
    from gofr_amd.http.request import METHOD_IDS, parse_request_bytes
    from gofr_amd.http.router import Router

    path = "/" + "/".join(segs)
    target = path + (f"?{query}" if query else "")
    head = (f"{method} {target} HTTP/1.1\r\nHost: h\r\n"
            + (f"Content-Length: {len(body)}\r\n" if body else "")
            + "\r\n")
    raw = head.encode() + body

    r = Router()
    r.add("GET", "/a", lambda c: None)
    trie = r.compile()
    tab = np.asarray([[ops.HK_HOST, 0, 0, 200]], np.int32).reshape(-1)
    buf = np.frombuffer(raw, np.uint8)
    fields = ops.cpu_parse_route(buf, np.asarray([0], np.int64),
                                 np.asarray([len(raw)], np.int32),
                                 trie, tab)
    f = fields[0]
    req = parse_request_bytes(raw)
    assert f[ops.FI_METHOD] == METHOD_IDS[method]
    got_path = raw[f[ops.FI_PATH_OFF]:
                   f[ops.FI_PATH_OFF] + f[ops.FI_PATH_LEN]].decode()
    assert got_path == req.path == path
    got_q = raw[f[ops.FI_QUERY_OFF]:
                f[ops.FI_QUERY_OFF] + f[ops.FI_QUERY_LEN]].decode()
    assert got_q == (query or "")
    got_body = raw[f[ops.FI_BODY_OFF]:
                   f[ops.FI_BODY_OFF] + f[ops.FI_BODY_LEN]]
    assert bytes(got_body) == req.body == body


INNER = MessageDesc("Inner", {1: ("id", "int32"), 2: ("tag", "string")})
OUTER = MessageDesc("Outer", {
    1: ("items", "repeated_message", INNER),
    2: ("names", "repeated_string"),
    3: ("kind", "sint64"),
})


@SET
@given(st.lists(st.tuples(st.integers(-(2 ** 31), 2 ** 31 - 1),
                          st.text(max_size=30)), max_size=5),
       st.lists(st.text(max_size=20), max_size=5),
       st.integers(-(2 ** 62), 2 ** 62))
def test_protobuf_nested_repeated_roundtrip(items, names, kind):
    msg = {"items": [{"id": i, "tag": t} for i, t in items],
           "names": names, "kind": kind}
    out = decode_message(encode_message(msg, OUTER), OUTER)
    assert [(d["id"], d["tag"]) for d in out["items"]] == items
    assert out["names"] == names
    assert out["kind"] == kind


# ---- round-2 mirrors: JSON field scan, pct decode, query lookup ------------

_JKEY = st.text(alphabet="abcdefghijklmnop_-0123456789", min_size=1,
                max_size=12)
_JVAL = st.recursive(
    st.one_of(st.none(), st.booleans(),
              st.integers(min_value=-10**12, max_value=10**12),
              st.floats(allow_nan=False, allow_infinity=False,
                        width=32),
              st.text(max_size=30)),
    lambda children: st.one_of(
        st.lists(children, max_size=4),
        st.dictionaries(_JKEY, children, max_size=4)),
    max_leaves=8)


@SET
@given(st.dictionaries(_JKEY, _JVAL, min_size=0, max_size=8),
       st.sampled_from([(",", ":"), (", ", ": ")]))
def test_json_top_fields_extracts_true_values(obj, seps):
    """json_top_fields_py's raw value spans must json-decode to the
    true field values for any serializable object body (the device
    scan is byte-equality-pinned against this mirror)."""
    import json as _json

    from gofr_amd import ops
    body = _json.dumps(obj, separators=seps).encode()
    fields = ops.json_top_fields_py(body)
    if len(obj) > ops.MAX_JSON_FIELDS:
        return  # scan caps at 8 fields (callers splice empty beyond)
    got = {}
    for key, vs, vl in fields:
        got[key.decode()] = _json.loads(body[vs:vs + vl])
    assert got == obj


@SET
@given(st.lists(st.one_of(
    st.binary(min_size=1, max_size=4).filter(lambda b: b"%" not in b),
    st.integers(min_value=0, max_value=255).map(
        lambda v: f"%{v:02x}".encode()),
    st.integers(min_value=0, max_value=255).map(
        lambda v: f"%{v:02X}".encode())), max_size=12))
def test_pct_decode_matches_unquote_on_valid_escapes(parts):
    from urllib.parse import unquote_to_bytes

    from gofr_amd import ops
    raw = b"".join(parts)
    got = ops.pct_decode(raw)
    assert got == unquote_to_bytes(raw)


@SET
@given(st.binary(max_size=24))
def test_pct_decode_rejects_exactly_invalid_escapes(raw):
    """None iff some '%' is not followed by two hex digits (the kernel
    then routes the request to the host parser's unquote leniency)."""
    from gofr_amd import ops
    hexd = set(b"0123456789abcdefABCDEF")
    valid = True
    i = 0
    while i < len(raw):
        if raw[i] == 0x25:
            if i + 2 >= len(raw) or raw[i + 1] not in hexd \
                    or raw[i + 2] not in hexd:
                valid = False
                break
            i += 3
        else:
            i += 1
    got = ops.pct_decode(raw)
    assert (got is not None) == valid


@SET
@given(st.lists(st.tuples(
    st.text(alphabet="abcxyz_123", min_size=1, max_size=6),
    st.text(alphabet="abc+%20xyz", max_size=8)), max_size=6),
    st.text(alphabet="abcxyz_123", min_size=1, max_size=6))
def test_q_find_first_value_semantics(pairs, probe):
    """q_find returns the FIRST value for a key (reference
    Param semantics: http/request.go:28-30) and None for misses."""
    from gofr_amd import ops
    q = "&".join(f"{k}={v}" for k, v in pairs).encode()
    got = ops.q_find_py(q, probe.encode())
    want = next((v.encode() for k, v in pairs if k == probe), None)
    assert got == want


@SET
@given(st.lists(st.sampled_from(
    ["plain", "b%20c", "%41%42", "x%2Fy", "pct%25", "u%C3%A9", "%ff"]),
    min_size=0, max_size=4),
    st.one_of(st.none(), st.sampled_from(["k=%20v", "a=b+c&d="])))
def test_parse_mirror_percent_decode_agrees_with_unquote(segs, query):
    """The in-place path decode (kernel mirror) must byte-match
    urllib's unquote for any valid-escape path, with the query span
    left RAW (decoded only at splice time)."""
    from urllib.parse import unquote_to_bytes

    import numpy as np

    from gofr_amd import ops
    from gofr_amd.http.router import Router

    path = "/" + "/".join(segs)
    target = path + (f"?{query}" if query else "")
    raw = f"GET {target} HTTP/1.1\r\nHost: h\r\n\r\n".encode()
    r = Router()
    r.add("GET", "/plain", lambda c: None)
    trie = r.compile()
    tab = np.asarray([[ops.HK_HOST, 0, 0, 200]], np.int32).reshape(-1)
    buf = np.frombuffer(raw, np.uint8).copy()  # decode mutates in place
    fields = ops.cpu_parse_route(buf, np.asarray([0], np.int64),
                                 np.asarray([len(raw)], np.int32),
                                 trie, tab)
    f = fields[0]
    got_path = bytes(buf[f[ops.FI_PATH_OFF]:
                         f[ops.FI_PATH_OFF] + f[ops.FI_PATH_LEN]])
    assert got_path == unquote_to_bytes(path)
    assert not (f[ops.FI_FLAGS] & ops.FL_NEEDS_HOST)
    got_q = bytes(buf[f[ops.FI_QUERY_OFF]:
                      f[ops.FI_QUERY_OFF] + f[ops.FI_QUERY_LEN]])
    assert got_q == (query or "").encode()  # query stays raw


_VALTXT = st.text(
    alphabet=st.characters(min_codepoint=0x20, max_codepoint=0x7E,
                           blacklist_characters="/?#& %+"),
    min_size=0, max_size=12)


@SET
@given(_VALTXT, _VALTXT,
       st.dictionaries(_JKEY, _JVAL, min_size=0, max_size=4))
def test_template_engine_dispatch_parity_fuzz(pid, qval, body_obj):
    """The HK_TEMPLATE mirror (engine) and the handler's Python body
    (CPU transport dispatch) must render identical bytes for arbitrary
    path params, query values and JSON bodies."""
    import json as _json

    import gofr_amd
    from gofr_amd import handlers
    from gofr_amd.config import MapConfig
    from gofr_amd.engine import BatchEngine
    from gofr_amd.http.request import parse_request_bytes
    from gofr_amd.server import dispatch

    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.GET("/user/{id}", handlers.template_json(
        '{"data":{"id":"', ("path", 0), '","q":"', ("query", "q"),
        '"}}'))
    app.POST("/order", handlers.template_json(
        '{"data":{"f":', ("jfield", "f"), "}}"))
    eng = BatchEngine(app)
    body = _json.dumps({"f": body_obj}, separators=(",", ":")).encode()
    raws = [
        (f"GET /user/{pid or 'x'}?q={qval} HTTP/1.1\r\n"
         "Host: h\r\n\r\n").encode(),
        (f"POST /order HTTP/1.1\r\nHost: h\r\n"
         f"Content-Type: application/json\r\n"
         f"Content-Length: {len(body)}\r\n\r\n").encode() + body,
    ]
    outs = eng.process(list(raws))
    for raw, out in zip(raws, outs):
        _, _, ebody = out.partition(b"\r\n\r\n")
        resp = dispatch(app, parse_request_bytes(raw))
        assert resp.body == ebody, (raw, resp.body, ebody)
