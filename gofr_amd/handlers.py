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
