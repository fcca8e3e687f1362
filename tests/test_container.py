"""Container (DI) — reference container/container_test.go semantics:
conditional datasource bring-up, health aggregation, service lookup."""

import gofr_amd
from gofr_amd.config import MapConfig
from gofr_amd.container import NewContainer
from gofr_amd.datasource import STATUS_UP


def test_no_hosts_no_datasources():
    c = NewContainer(MapConfig({"LOG_LEVEL": "FATAL"}))
    assert c.redis is None and c.db is None
    assert c.Health() == {}  # nothing to report (container.go:26-38)


def test_sqlite_db_brought_up_and_health():
    c = NewContainer(MapConfig({"LOG_LEVEL": "FATAL",
                                "DB_DIALECT": "sqlite",
                                "DB_HOST": ":memory:",
                                "DB_NAME": ":memory:"}))
    assert c.db is not None
    h = c.Health()
    assert h["sql"]["status"] == STATUS_UP


def test_redis_connect_failure_not_fatal():
    # container.go:60-64: failure logged, startup continues
    c = NewContainer(MapConfig({"LOG_LEVEL": "FATAL",
                                "REDIS_HOST": "127.0.0.1",
                                "REDIS_PORT": "1"}))
    assert c is not None  # no exception


def test_get_http_service():
    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.AddHTTPService("orders", "http://localhost:9999")
    svc = app.container.GetHTTPService("orders")
    assert svc is not None and svc.address == "http://localhost:9999"
    assert app.container.GetHTTPService("nope") is None


def test_logger_embedding():
    # Context embeds the container; ctx.Info/Error reach the logger
    c = NewContainer(MapConfig({"LOG_LEVEL": "FATAL"}))
    c.Info("x")  # must not raise (logger pass-through, context.go:21)
    c.Errorf("e %d", 1)
