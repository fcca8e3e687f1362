"""Per-request Context handed to every handler.

Reference: pkg/gofr/context.go:12-70 — Context embeds context.Context,
the transport Request, and the *Container; adds Trace() and Bind().
Here: attribute delegation to the request and container gives handlers
ctx.Param / ctx.PathParam / ctx.Bind / ctx.HostName / ctx.Redis / ctx.DB /
ctx.GetHTTPService / ctx.Info... exactly like the embedded-struct access
in Go.
"""

from __future__ import annotations

from typing import Optional


class Context:
    def __init__(self, request, container, responder=None, span=None):
        self.request = request
        self.container = container
        self.responder = responder
        self.span = span  # handler span (reference: handler.go:34)

    # -- request surface ------------------------------------------------------
    def Param(self, key: str) -> str:
        return self.request.Param(key)

    def PathParam(self, key: str) -> str:
        return self.request.PathParam(key)

    def Bind(self, into=None):
        return self.request.Bind(into)

    def HostName(self) -> str:
        return self.request.HostName()

    # -- container surface ----------------------------------------------------
    @property
    def Redis(self):
        return self.container.redis

    @property
    def DB(self):
        return self.container.db

    @property
    def Config(self):
        return self.container.config

    @property
    def logger(self):
        return self.container.logger

    def GetHTTPService(self, name: str):
        return self.container.GetHTTPService(name)

    # -- tracing — reference: context.go:45-50 --------------------------------
    def Trace(self, name: str):
        tracer = self.container.tracer
        if tracer is None:
            from .trace import noop_tracer
            tracer = noop_tracer()
        return tracer.start_span(name, parent=self.span)

    # -- logging pass-through -------------------------------------------------
    def Log(self, *a):
        self.container.logger.Info(*a)

    def Logf(self, fmt, *a):
        self.container.logger.Infof(fmt, *a)

    def Info(self, *a):
        self.container.logger.Info(*a)

    def Infof(self, fmt, *a):
        self.container.logger.Infof(fmt, *a)

    def Debug(self, *a):
        self.container.logger.Debug(*a)

    def Debugf(self, fmt, *a):
        self.container.logger.Debugf(fmt, *a)

    def Error(self, *a):
        self.container.logger.Error(*a)

    def Errorf(self, fmt, *a):
        self.container.logger.Errorf(fmt, *a)

    def Warn(self, *a):
        self.container.logger.Warn(*a)

    def Warnf(self, fmt, *a):
        self.container.logger.Warnf(fmt, *a)

    # pythonic aliases
    param = Param
    path_param = PathParam
    bind = Bind
    host_name = HostName
    trace = Trace
    get_http_service = GetHTTPService

    @property
    def redis(self):
        return self.container.redis

    @property
    def db(self):
        return self.container.db


def new_context(request, container, responder=None,
                span: Optional[object] = None) -> Context:
    """Reference: context.go:63-70 newContext."""
    return Context(request, container, responder=responder, span=span)
