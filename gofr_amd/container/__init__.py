"""Dependency-injection container.

Reference: pkg/gofr/container/container.go:19-95 — the Container embeds
the Logger, holds named downstream HTTP services, Redis, and DB; brings
datasources up conditionally from config (connect failures logged, never
fatal); aggregates per-datasource health.
"""

from __future__ import annotations

from .. import logging as gofr_logging
from ..datasource import redis as redis_ds
from ..datasource import sql as sql_ds


class Container:
    """Shared per-app concerns handed (via Context) to every handler."""

    def __init__(self, config, logger=None, tracer=None):
        self.config = config
        self.logger = logger or gofr_logging.new_logger_from_env(config)
        self.tracer = tracer
        self.services: dict[str, object] = {}
        # conditional bring-up — reference: container.go:48-86
        self.redis = redis_ds.new_client(config, logger=self.logger,
                                         tracer=tracer)
        self.db = sql_ds.new_db(config, logger=self.logger)

    # -- health aggregation — reference: container.go:26-38 ------------------
    def Health(self) -> dict:
        report: dict[str, object] = {}
        if self.redis is not None:
            report["redis"] = self.redis.HealthCheck()
        if self.db is not None:
            report["sql"] = self.db.HealthCheck()
        return report

    health = Health

    # -- reference: container.go:93-95 ---------------------------------------
    def GetHTTPService(self, name: str):
        return self.services.get(name)

    get_http_service = GetHTTPService

    # logger pass-through so ctx.Log/Info/Error work via embedding
    def Log(self, *a):
        self.logger.Info(*a)

    def Logf(self, fmt, *a):
        self.logger.Infof(fmt, *a)

    def Info(self, *a):
        self.logger.Info(*a)

    def Infof(self, fmt, *a):
        self.logger.Infof(fmt, *a)

    def Debug(self, *a):
        self.logger.Debug(*a)

    def Debugf(self, fmt, *a):
        self.logger.Debugf(fmt, *a)

    def Error(self, *a):
        self.logger.Error(*a)

    def Errorf(self, fmt, *a):
        self.logger.Errorf(fmt, *a)

    def Warn(self, *a):
        self.logger.Warn(*a)

    def Warnf(self, fmt, *a):
        self.logger.Warnf(fmt, *a)

    def close(self):
        if self.redis is not None:
            self.redis.close()
        if self.db is not None:
            self.db.close()


def NewContainer(config, tracer=None) -> Container:
    """Reference: container/container.go:40-89."""
    return Container(config, tracer=tracer)
