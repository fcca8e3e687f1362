"""HTTP server example — mirror of reference examples/http-server/main.go.

Routes: /hello (query param), /error, /redis, /trace, /mysql, plus the
framework defaults (/.well-known/health, /favicon.ico, catch-all 404).
Config comes from ./configs/.env (same key set as the reference).
"""

import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))

import gofr_amd as gofr  # noqa: E402
from gofr_amd.errors import GofrError  # noqa: E402


def hello_handler(c):
    name = c.Param("name")
    if not name:
        c.Log("Name came empty")
        name = "World"
    return f"Hello {name}!"


def error_handler(c):
    raise GofrError("some error occurred")


def redis_handler(c):
    if c.Redis is None:
        raise GofrError("redis not configured")
    try:
        val = c.Redis.Get("test")
    except Exception as e:  # key missing is not an error (reference note)
        raise GofrError(str(e)) from e
    return val or ""


def trace_handler(c):
    with c.Trace("traceHandler"):
        with c.Trace("some-sample-work"):
            time.sleep(0.001)  # simulate workload
        # ping redis 5 times (the reference does them concurrently)
        if c.Redis is not None:
            import threading
            ts = [threading.Thread(target=c.Redis.Ping) for _ in range(5)]
            for t in ts:
                t.start()
            for t in ts:
                t.join()
        svc = c.GetHTTPService("anotherService")
        if svc is not None:
            resp = svc.Get(c, "redis", None)
            return resp.body.decode("utf-8", "replace")
    return "ok"


def mysql_handler(c):
    if c.DB is None:
        raise GofrError("db not configured")
    row = c.DB.QueryRow("select 2+2")
    return row[0] if row else None


def build_app():
    a = gofr.New()
    a.AddHTTPService("anotherService", "http://localhost:9000")
    a.GET("/hello", hello_handler)
    a.GET("/error", error_handler)
    a.GET("/redis", redis_handler)
    a.GET("/trace", trace_handler)
    a.GET("/mysql", mysql_handler)
    return a


if __name__ == "__main__":
    build_app().Run()
