"""GPU-native handler library.

Handlers carry a `__gofr_gpu__` spec so the batch engine can execute them
entirely on-device (gofr_amd/engine); each also has a working Python body
so the same app runs identically on the CPU transport. Routes whose
handlers have no spec run through the host trampoline: the GPU still
parses, routes and serializes the batch, and only the handler body runs
in Python between the two kernels.
"""

from __future__ import annotations

import json


def echo_json(ctx):
    """Bind the JSON body and return it — the 1 KB JSON echo of
    BASELINE.json config 2. GPU: HK_ECHO_JSON (zero-copy envelope around
    the request body)."""
    return ctx.Bind()


echo_json.__gofr_gpu__ = ("echo_json",)


def static_json(data):
    """Return a fixed JSON-serializable value. GPU: HK_STATIC with the
    envelope precomputed into the handler blob."""
    def handler(ctx):
        return data
    handler.__gofr_gpu__ = ("static", json.dumps(
        {"data": data}, separators=(",", ":"), ensure_ascii=False
    ).encode("utf-8"), 200)
    return handler


def static_raw(body: bytes, status: int = 200):
    """Serve precomputed envelope bytes as-is (advanced)."""
    def handler(ctx):
        from .http.response import Raw
        return Raw(json.loads(body))
    handler.__gofr_gpu__ = ("static", body, status)
    return handler


def template_json(*pieces, status: int = 200):
    """Template-substitution handler: the response body is a compiled
    piece sequence rendered fully on-device (HK_TEMPLATE in k_respond —
    no host trampoline for /user/{id}-class routes). Pieces:

      "literal text"          verbatim bytes (carry your own envelope)
      ("path", i)             i-th path param, JSON-string-escaped
      ("path_raw", i)         i-th path param verbatim
      ("query", "key")        query param: %XX/'+'-decoded + escaped
      ("query_raw", "key")    query param: decoded, not escaped
      ("jfield", "key")       top-level JSON body field, raw value span
      ("jfield_str", "key")   ... string content (quotes stripped)

    Missing query/body fields splice as empty. The Python body renders
    the identical bytes for the CPU transport (shared ops mirrors).
    Reference handler shape: /root/reference/examples/http-server/
    main.go:14-29 (param routes the r1 engine sent to the trampoline).
    """
    from . import ops
    from .http.response import File

    def handler(ctx):
        out = bytearray()
        jfields = None
        body = ctx.request.body
        for piece in pieces:
            if isinstance(piece, (str, bytes)):
                out += piece.encode("utf-8") if isinstance(piece, str) \
                    else piece
                continue
            op, arg = piece[0], piece[1]
            if op in ("path", "path_raw"):
                vals = list(ctx.request.path_params.values())
                v = vals[arg].encode("utf-8") if arg < len(vals) else b""
                out += ops.splice_py(v, ops.TM_JESC if op == "path"
                                     else 0)
            elif op in ("query", "query_raw"):
                q = ctx.request.query_string.encode("latin-1")
                val = ops.q_find_py(q, arg.encode("utf-8"))
                if val is not None:
                    mode = (ops.TM_PCT | ops.TM_JESC) if op == "query" \
                        else ops.TM_PCT
                    out += ops.splice_py(val, mode)
            elif op in ("jfield", "jfield_str"):
                if jfields is None:
                    jfields = ops.json_top_fields_py(body)
                for key, vs, vl in jfields:
                    if key == arg.encode("utf-8"):
                        if op == "jfield_str" and vl >= 2 and \
                                body[vs] == 0x22:
                            vs, vl = vs + 1, vl - 2
                        out += ops.splice_py(body[vs:vs + vl], 0)
                        break
        return File(bytes(out), "application/json")

    handler.__gofr_gpu__ = ("template", list(pieces), status)
    return handler


def redis_json(prefix: str = "", param: int = 0):
    """Redis-backed read route: GET <prefix><path param> where the
    stored value is JSON text. The GPU engine's host trampoline serves
    the WHOLE batch with ONE pipelined MGET round trip (the
    __gofr_batch__ hook — VERDICT r1 item 9: a DB-touching handler
    must not serialize the batch command-by-command). Single-request
    transports (dispatch) use the per-call client path.
    Reference handler shape: /root/reference/examples/http-server/
    main.go:31-38 (redis get by path param)."""
    from .errors import GofrError, KeyNotFoundError
    from .http.response import File

    def _body(v):
        raw = v if isinstance(v, bytes) else str(v).encode("utf-8")
        return b'{"data":' + raw + b"}"

    def handler(ctx):
        if ctx.Redis is None:
            raise GofrError("redis not configured")
        vals = list(ctx.request.path_params.values())
        key = prefix + (vals[param] if param < len(vals) else "")
        v = ctx.Redis.Get(key)
        if v is None:
            raise KeyNotFoundError()
        return File(_body(v), "application/json")

    def batch(app, requests):
        """One MGET round trip for the whole trampoline batch,
        returned PACKED (single blob + int32 table) so the engine
        skips the per-row Python emit loop: reply spans come from the
        native RESP parser and the envelopes assemble with one join.
        Falls back to the per-row protocol when the native extension
        is unavailable."""
        import numpy as np
        redis = app.container.redis
        n = len(requests)
        keys = []
        for req in requests:
            vals = list(req.path_params.values())
            keys.append(prefix + (vals[param] if param < len(vals)
                                  else ""))
        if redis is None:
            err = b'{"error":{"message":"redis not configured"}}'
            return [(500, err, "application/json")] * n
        try:
            reply, offs, lens = redis.mget_spans(keys)
        except ImportError:
            replies = redis.MGet(keys)
            out = []
            for v in replies:
                if v is None:
                    out.append(
                        (404,
                         b'{"error":{"message":"key not found"}}',
                         "application/json"))
                else:
                    out.append((200, _body(v), "application/json"))
            return out
        MISS = b'{"error":{"message":"key not found"}}'
        OPEN, CLOSE = b'{"data":', b"}"
        parts = []
        tab = np.zeros((n, 4), np.int32)
        pos = 0
        for i in range(n):
            ln = int(lens[i])
            if ln < 0:
                parts.append(MISS)
                tab[i] = (pos, len(MISS), 404, 0)
                pos += len(MISS)
            else:
                o = int(offs[i])
                parts.append(OPEN)
                parts.append(reply[o:o + ln])
                parts.append(CLOSE)
                blen = len(OPEN) + ln + 1
                tab[i] = (pos, blen, 200, 0)
                pos += blen
        return ("packed", b"".join(parts), tab)

    def batch_fields(app, reqs, offs, fields, rows):
        """Fields-level fast path: keys come straight from the GPU's
        decoded param-0 spans (no host request parse); one MGET via
        the native RESP parser; packed envelope assembly."""
        import numpy as np

        from . import ops
        redis = app.container.redis
        if redis is None:
            raise GofrError("redis not configured")
        pfx = prefix
        F2 = np.asarray(fields).reshape(-1, ops.NF)
        rows_a = np.asarray(rows, np.int64)
        koffs = (np.asarray(offs)[rows_a] +
                 F2[rows_a, ops.FI_PARAM0 + 2 * param]).tolist()
        klens = F2[rows_a, ops.FI_PARAM0 + 2 * param + 1].tolist()
        raw = np.asarray(reqs).tobytes()  # plain-bytes slicing is ~4x
        keys = [pfx + raw[o:o + ln].decode("latin-1")
                for o, ln in zip(koffs, klens)]
        reply, roffs, rlens = redis.mget_spans(keys)
        MISS = b'{"error":{"message":"key not found"}}'
        OPEN, CLOSE = b'{"data":', b"}"
        # vectorized table: statuses/lengths/offsets via numpy; the
        # only per-row Python is the parts list build
        nl = len(rows)
        rlens_l = rlens.tolist()
        roffs_l = roffs.tolist()
        blens = np.where(rlens < 0, len(MISS),
                         rlens + (len(OPEN) + 1)).astype(np.int32)
        pos_col = np.zeros(nl, np.int32)
        if nl > 1:
            np.cumsum(blens[:-1], out=pos_col[1:])
        tab = np.zeros((nl, 4), np.int32)
        tab[:, 0] = pos_col
        tab[:, 1] = blens
        tab[:, 2] = np.where(rlens < 0, 404, 200)
        parts = []
        for i in range(nl):
            ln = rlens_l[i]
            if ln < 0:
                parts.append(MISS)
            else:
                o = roffs_l[i]
                parts.append(OPEN)
                parts.append(reply[o:o + ln])
                parts.append(CLOSE)
        return ("packed", b"".join(parts), tab)

    handler.__gofr_batch__ = batch
    handler.__gofr_batch_fields__ = batch_fields
    return handler


def kv_json(store: dict):
    """Device KV-store read handler keyed by the route's first path
    param (HK_KV): the store compiles into an open-addressing table +
    value blob resident in HBM; k_respond probes it and splices the
    pre-wrapped {"data":...} envelope — the /user/{name} redis-get
    analog (reference examples/http-server/main.go:31-38) with the
    lookup on-device. Miss -> 404 {"error":{"message":"key not found"}}.
    """
    from .errors import KeyNotFoundError
    from .http.response import File

    def handler(ctx):
        vals = list(ctx.request.path_params.values())
        key = vals[0] if vals else ""
        if key in store:
            body = (b'{"data":' +
                    json.dumps(store[key], separators=(",", ":"),
                               ensure_ascii=False).encode("utf-8") + b"}")
            return File(body, "application/json")
        raise KeyNotFoundError()

    handler.__gofr_gpu__ = ("kv", store)
    return handler
