"""HTTP router: segment radix-trie with `{param}` captures.

Role of the reference's gorilla/mux wrapper (pkg/gofr/http/router.go:13-33):
routes are registered per-method with patterns like `/user/{id}`; matching
walks path segments. Unlike the reference (which delegates to mux's regexp
tree), this router is a radix trie designed to compile to a flat table the
GPU route-match kernel walks (native/hip/gofr_kernels.hip, k_parse_route):
  - nodes laid out breadth-first; literal children contiguous per node
  - segment literals in one byte blob (LDS-staged on device)
  - per-node [method] -> route id leaf table
The Python `match` below is the golden model the kernel is tested against.
"""

from __future__ import annotations

from typing import Any, Callable, Optional

import numpy as np

from .request import METHOD_IDS

# padded to 8 so the per-node method table is pow2-strided for the kernel
N_METHODS = 8
assert len(METHOD_IDS) <= N_METHODS


class _Node:
    __slots__ = ("children", "param_child", "param_name", "routes", "prefix_route")

    def __init__(self):
        self.children: dict[str, _Node] = {}
        self.param_child: Optional[_Node] = None
        self.param_name: str = ""
        # method id -> route id
        self.routes: dict[int, int] = {}
        # route id for PathPrefix-style catch-all rooted here (-1 = none)
        self.prefix_route: int = -1


class Route:
    __slots__ = ("method", "pattern", "handler", "route_id", "param_names",
                 "is_prefix")

    def __init__(self, method: str, pattern: str, handler: Callable,
                 route_id: int, param_names: list[str], is_prefix: bool):
        self.method = method
        self.pattern = pattern
        self.handler = handler
        self.route_id = route_id
        self.param_names = param_names
        self.is_prefix = is_prefix


class Router:
    """Trie router. add() registers, match() resolves.

    match returns (route, params, status): status 200 when found, 405 when
    the path exists but the method doesn't, 404 otherwise (the App installs
    a catch-all route that renders the 404 body, mirroring
    pkg/gofr/gofr.go:104-107).
    """

    def __init__(self):
        self.root = _Node()
        self.routes: list[Route] = []

    def add(self, method: str, pattern: str, handler: Callable,
            is_prefix: bool = False) -> Route:
        method = method.upper()
        if method not in METHOD_IDS:
            raise ValueError(f"unsupported method {method}")
        node = self.root
        param_names: list[str] = []
        segs = [s for s in pattern.strip("/").split("/") if s != ""]
        for seg in segs:
            if seg.startswith("{") and seg.endswith("}"):
                name = seg[1:-1]
                if node.param_child is None:
                    node.param_child = _Node()
                    node.param_name = name
                param_names.append(node.param_name)
                node = node.param_child
            else:
                node = node.children.setdefault(seg, _Node())
        rid = len(self.routes)
        route = Route(method, pattern, handler, rid, param_names, is_prefix)
        self.routes.append(route)
        if is_prefix:
            node.prefix_route = rid
        else:
            node.routes[METHOD_IDS[method]] = rid
        return route

    def add_prefix(self, method: str, prefix: str, handler: Callable) -> Route:
        """PathPrefix-style catch-all (reference: gofr.go:104-107)."""
        return self.add(method, prefix, handler, is_prefix=True)

    def match(self, method: str, path: str):
        mid = METHOD_IDS.get(method.upper(), -1)
        node = self.root
        params: dict[str, str] = {}
        best_prefix = self.root.prefix_route
        segs = [s for s in path.strip("/").split("/") if s != ""]
        # gorilla/mux StrictSlash(false) parity (http/router.go:17):
        # "/path/" is NOT "/path" — a trailing slash only reaches
        # prefix (catch-all) routes
        found_node = not (len(path) > 1 and path.endswith("/"))
        for seg in segs:
            nxt = node.children.get(seg)
            if nxt is None and node.param_child is not None:
                params[node.param_name] = seg
                nxt = node.param_child
            if nxt is None:
                found_node = False
                break
            node = nxt
            if node.prefix_route >= 0:
                best_prefix = node.prefix_route
        if found_node and node.routes:
            rid = node.routes.get(mid)
            if rid is not None:
                return self.routes[rid], params, 200
            # path exists, method doesn't
            if best_prefix >= 0:
                return self.routes[best_prefix], params, 200
            return None, params, 405
        if best_prefix >= 0:
            return self.routes[best_prefix], params, 200
        return None, params, 404

    # -- GPU compilation -----------------------------------------------------
    def compile(self):
        """Flatten the trie into numpy arrays for the device kernel.

        Layout (all int32 unless noted):
          seg_blob  uint8[*]   concatenated literal segment bytes
          node_child_first[n]  index into child arrays
          node_child_count[n]
          child_seg_off/child_seg_len/child_node[m]   per literal child
          node_param[n]        child node id or -1 (param segment)
          node_prefix[n]       route id of catch-all rooted here or -1
          node_route[n*8]      per-method route id or -1
        """
        nodes: list[_Node] = []
        index: dict[int, int] = {}

        def visit(nd: _Node):
            index[id(nd)] = len(nodes)
            nodes.append(nd)

        # BFS
        queue = [self.root]
        while queue:
            nd = queue.pop(0)
            visit(nd)
            for seg in sorted(nd.children):
                queue.append(nd.children[seg])
            if nd.param_child is not None:
                queue.append(nd.param_child)

        n = len(nodes)
        seg_blob = bytearray()
        child_first = np.zeros(n, np.int32)
        child_count = np.zeros(n, np.int32)
        child_seg_off: list[int] = []
        child_seg_len: list[int] = []
        child_node: list[int] = []
        node_param = np.full(n, -1, np.int32)
        node_prefix = np.full(n, -1, np.int32)
        node_route = np.full(n * N_METHODS, -1, np.int32)

        for i, nd in enumerate(nodes):
            child_first[i] = len(child_node)
            for seg in sorted(nd.children):
                child_seg_off.append(len(seg_blob))
                sb = seg.encode("utf-8")
                seg_blob.extend(sb)
                child_seg_len.append(len(sb))
                child_node.append(index[id(nd.children[seg])])
            child_count[i] = len(child_node) - child_first[i]
            if nd.param_child is not None:
                node_param[i] = index[id(nd.param_child)]
            node_prefix[i] = nd.prefix_route
            for mid, rid in nd.routes.items():
                node_route[i * N_METHODS + mid] = rid

        return {
            "seg_blob": np.frombuffer(bytes(seg_blob) or b"\0", np.uint8).copy(),
            "node_child_first": child_first,
            "node_child_count": child_count,
            "child_seg_off": np.asarray(child_seg_off or [0], np.int32),
            "child_seg_len": np.asarray(child_seg_len or [0], np.int32),
            "child_node": np.asarray(child_node or [0], np.int32),
            "node_param": node_param,
            "node_prefix": node_prefix,
            "node_route": node_route,
            "n_nodes": n,
        }
