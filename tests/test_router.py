"""Router trie tests + GPU-compile layout sanity (CPU side)."""

import numpy as np

from gofr_amd.http.router import Router


def _h(name):
    def handler(ctx):
        return name
    handler.__name__ = name
    return handler


def test_static_and_param_match():
    r = Router()
    r.add("GET", "/greet", _h("greet"))
    r.add("GET", "/user/{id}", _h("user"))
    r.add("POST", "/user/{id}", _h("user_post"))
    r.add("GET", "/user/{id}/orders/{oid}", _h("orders"))

    route, params, st = r.match("GET", "/greet")
    assert st == 200 and route.handler(None) == "greet" and params == {}

    route, params, st = r.match("GET", "/user/42")
    assert st == 200 and params == {"id": "42"}

    route, params, st = r.match("POST", "/user/42")
    assert route.handler(None) == "user_post"

    route, params, st = r.match("GET", "/user/7/orders/99")
    assert st == 200 and params == {"id": "7", "oid": "99"}


def test_static_wins_over_param():
    r = Router()
    r.add("GET", "/user/{id}", _h("param"))
    r.add("GET", "/user/me", _h("static"))
    route, params, st = r.match("GET", "/user/me")
    assert route.handler(None) == "static" and params == {}
    route, params, st = r.match("GET", "/user/other")
    assert route.handler(None) == "param" and params == {"id": "other"}


def test_not_found_and_method_fallthrough():
    r = Router()
    r.add("GET", "/a", _h("a"))
    _, _, st = r.match("GET", "/missing")
    assert st == 404
    # wrong method with no catch-all -> 405
    _, _, st = r.match("POST", "/a")
    assert st == 405
    # with a catch-all installed, both fall to it (gorilla/mux behavior:
    # PathPrefix("/") catches non-matching methods too — gofr.go:104-107)
    r.add_prefix("GET", "/", _h("catch"))
    route, _, st = r.match("POST", "/a")
    assert st == 200 and route.handler(None) == "catch"
    route, _, st = r.match("GET", "/missing")
    assert route.handler(None) == "catch"


def test_root_route():
    r = Router()
    r.add("GET", "/", _h("root"))
    route, _, st = r.match("GET", "/")
    assert st == 200 and route.handler(None) == "root"


def test_compile_layout():
    r = Router()
    r.add("GET", "/greet", _h("g"))
    r.add("GET", "/user/{id}", _h("u"))
    r.add("POST", "/user/{id}", _h("up"))
    r.add_prefix("GET", "/", _h("c"))
    t = r.compile()
    assert t["n_nodes"] >= 3
    assert t["node_route"].dtype == np.int32
    # root node: children 'greet' and 'user'
    assert t["node_child_count"][0] == 2
    # root prefix route installed
    assert t["node_prefix"][0] == 3
    # walk 'user' -> param child
    segs = bytes(t["seg_blob"]).decode()
    assert "greet" in segs and "user" in segs


def test_compile_matches_python_match_on_fuzz():
    """The flat table must encode the same decisions as match() — walked
    here in pure Python as the kernel's algorithm golden test."""
    from gofr_amd.http.request import METHOD_IDS
    r = Router()
    r.add("GET", "/a/b/c", _h("abc"))
    r.add("GET", "/a/{x}/c", _h("axc"))
    r.add("POST", "/a/b", _h("ab"))
    r.add("GET", "/z", _h("z"))
    r.add_prefix("GET", "/", _h("catch"))
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

    for method, path in [("GET", "/a/b/c"), ("GET", "/a/q/c"),
                         ("POST", "/a/b"), ("GET", "/z"), ("GET", "/nope"),
                         ("POST", "/z"), ("GET", "/a/b"), ("GET", "/a")]:
        route, _, st = r.match(method, path)
        want = route.route_id if route else -1
        assert table_match(method, path) == want, (method, path)


def test_strict_slash_false_parity():
    """gorilla/mux StrictSlash(false): "/path/" is not "/path"
    (reference http/router.go:17); prefix routes still match."""
    from gofr_amd.http.router import Router
    r = Router()
    r.add("GET", "/path", lambda c: "x")
    r.add_prefix("GET", "/files", lambda c: "f")
    route, params, st = r.match("GET", "/path")
    assert route is not None and st == 200
    route, params, st = r.match("GET", "/path/")
    assert route is None and st == 404
    route, params, st = r.match("GET", "/files/a/")
    assert route is not None  # prefix routes still catch trailing /
    route, params, st = r.match("GET", "/")
    assert st in (200, 404)  # root unaffected by the rule
