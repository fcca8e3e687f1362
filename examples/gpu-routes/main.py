"""GPU-native route showcase: every route here executes fully
on-device when served by the MI355X batch engine — no Python in the
request path. The same app runs identically on the CPU transport (each
handler carries a working Python body).

Routes (handler kind in parentheses):
  GET  /user/{id}      template splice of the path param (HK_TEMPLATE)
  GET  /greet?name=x   query-param splice, %XX/'+'-decoded (HK_TEMPLATE)
  POST /order          JSON body field binding — ctx.Bind()-class
                       handler reading 3 named fields (HK_TEMPLATE)
  GET  /plans/{tier}   device KV-store lookup in HBM (HK_KV)
  POST /echo           zero-copy JSON echo (HK_ECHO_JSON)
  GET  /cached/{key}   redis-backed read: the engine serves a whole
                       batch with ONE pipelined MGET (host batch hook)
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))

import gofr_amd as gofr  # noqa: E402
from gofr_amd import handlers  # noqa: E402

PLANS = {
    "free": {"tier": "free", "rps": 100},
    "pro": {"tier": "pro", "rps": 10_000},
    "enterprise": {"tier": "enterprise", "rps": 1_000_000},
}


def build_app():
    a = gofr.New()
    a.GET("/user/{id}", handlers.template_json(
        '{"data":{"id":"', ("path", 0), '"}}'))
    a.GET("/greet", handlers.template_json(
        '{"data":"Hello ', ("query", "name"), '!"}'))
    a.POST("/order", handlers.template_json(
        '{"data":{"item":', ("jfield", "item"),
        ',"qty":', ("jfield", "qty"),
        ',"note":"', ("jfield_str", "note"), '"}}'))
    a.GET("/plans/{tier}", handlers.kv_json(PLANS))
    a.POST("/echo", handlers.echo_json)
    a.GET("/cached/{key}", handlers.redis_json(prefix="cache:"))
    return a


if __name__ == "__main__":
    build_app().Run()
