"""Application object: the public gofr API surface.

Reference: pkg/gofr/gofr.go — `New()` reads config, builds the container,
initializes tracing and the servers; `GET/PUT/POST/DELETE` register
routes; `Run()` installs the default routes (health, favicon, catch-all)
and serves; `NewCMD()` builds the CLI-mode app; `RegisterService` /
`AddHTTPService` / `SubCommand` mirror gofr.go:42-46, 139-149, 181-183.

Engine selection: by default `Run()` serves through the CPU transport
(gofr_amd/server.py). With GOFR_ENGINE=gpu (and a visible MI355X) the
HTTP data plane runs on the GPU batch engine (gofr_amd/engine): the C++
epoll ingress stages raw request bytes into pinned rings, the CDNA4
kernels parse/route/serialize whole batches, and Python handlers are
invoked only for routes without a GPU fast path.
"""

from __future__ import annotations

import os
import threading

from . import config as config_mod
from .cmd import CMD
from .container import NewContainer
from .context import Context  # noqa: F401  (re-exported for typing)
from .errors import MissingFileError
from .http.response import File
from .http.router import Router
from .server import HTTPServer
from .service import NewHTTPService
from .trace import Tracer

_STATIC_DIR = os.path.join(os.path.dirname(__file__), "static")


def _health_handler(ctx):
    """Reference: handler.go:38-40."""
    return ctx.container.Health()


def _favicon_handler(ctx):
    """Reference: handler.go:42-49."""
    try:
        with open(os.path.join(_STATIC_DIR, "favicon.ico"), "rb") as f:
            return File(f.read(), "image/x-icon")
    except OSError as e:
        raise MissingFileError() from e


def _catch_all_handler(ctx):
    """Reference: handler.go:51-53 — 404 via the missing-file error."""
    raise MissingFileError()


class App:
    def __init__(self, cmd_mode: bool = False, config=None):
        self.config = config or self._read_config()
        app_name = self.config.GetOrDefault("APP_NAME", "gofr-app")
        # tracing — reference: gofr.go:185-211 (exporter only if TRACER_HOST)
        tracer_host = self.config.Get("TRACER_HOST")
        tracer_port = int(self.config.GetOrDefault(
            "TRACER_PORT", str(config_mod.DEFAULT_TRACER_PORT)))
        self.tracer = Tracer(app_name=app_name, exporter_host=tracer_host,
                             exporter_port=tracer_port)
        self.container = NewContainer(self.config, tracer=self.tracer)
        self.router = Router()
        self.cmd = CMD() if cmd_mode else None
        self.http_port = int(self.config.GetOrDefault(
            "HTTP_PORT", str(config_mod.DEFAULT_HTTP_PORT)))
        self.grpc_port = int(self.config.GetOrDefault(
            "GRPC_PORT", str(config_mod.DEFAULT_GRPC_PORT)))
        self.http_registered = False
        self.grpc_registered = False
        self._grpc_services: list = []
        self._defaults_installed = False
        self.http_server = None
        self.grpc_server = None
        self.engine = None  # GPU batch engine, created in Run when enabled
        self.auth_secret = None   # enable_auth()
        self.gzip_min_size = None  # enable_gzip()

    @staticmethod
    def _read_config():
        """Reference: gofr.go:129-136 — ./configs/.env if present."""
        folder = "./configs" if os.path.isdir("./configs") else "."
        return config_mod.EnvFile(folder)

    # -- route registration — reference: gofr.go:152-177 ---------------------
    def GET(self, pattern: str, handler):
        self._add("GET", pattern, handler)

    def POST(self, pattern: str, handler):
        self._add("POST", pattern, handler)

    def PUT(self, pattern: str, handler):
        self._add("PUT", pattern, handler)

    def DELETE(self, pattern: str, handler):
        self._add("DELETE", pattern, handler)

    def PATCH(self, pattern: str, handler):
        self._add("PATCH", pattern, handler)

    def _add(self, method: str, pattern: str, handler):
        self.http_registered = True
        self.router.add(method, pattern, handler)

    # decorator sugar (no reference analog; idiomatic Python)
    def get(self, pattern: str):
        def deco(fn):
            self.GET(pattern, fn)
            return fn
        return deco

    def post(self, pattern: str):
        def deco(fn):
            self.POST(pattern, fn)
            return fn
        return deco

    def put(self, pattern: str):
        def deco(fn):
            self.PUT(pattern, fn)
            return fn
        return deco

    def delete(self, pattern: str):
        def deco(fn):
            self.DELETE(pattern, fn)
            return fn
        return deco

    def patch(self, pattern: str):
        def deco(fn):
            self.PATCH(pattern, fn)
            return fn
        return deco

    # -- gRPC — reference: gofr.go:42-46 -------------------------------------
    def RegisterService(self, service, impl=None):
        """Register a gRPC service implementation (see gofr_amd.grpc)."""
        self.grpc_registered = True
        self._grpc_services.append((service, impl))

    # -- downstream services — reference: gofr.go:139-149 ---------------------
    def AddHTTPService(self, name: str, address: str):
        if name in self.container.services:
            self.container.logger.Errorf(
                "service %s already registered", name)
        self.container.services[name] = NewHTTPService(
            address, logger=self.container.logger, tracer=self.tracer)

    # -- middlewares (BASELINE config 4: auth + gzip + log) -------------------
    def enable_auth(self, secret: bytes):
        """HMAC-SHA256 bearer auth on every non-OPTIONS request:
        Authorization: HMAC <hex of HMAC-SHA256(secret, "METHOD path")>.
        GPU engine: k_auth kernel; CPU transport: checked in dispatch."""
        self.auth_secret = bytes(secret)
        return self

    def enable_etag(self):
        """ETag middleware: every response carries a strong validator
        computed by the MFMA batched body hash (k_respond's
        mfma_etag_wave; model: ops.etag_u32)."""
        self.etag_on = True
        return self

    def enable_request_log(self, sample_every: int = 4096):
        """Batch-aware access-log middleware: the engine emits one
        BatchLog aggregate per processed batch plus every Nth request
        as a full RequestLog record (parsing 22M+ full per-request log
        lines/s is neither writable nor readable; sampling preserves
        the reference's observability signal — middleware/logger.go:
        24-63 — at batch rates)."""
        self.request_log_every = max(1, int(sample_every))
        return self

    def enable_gzip(self, min_size: int = 256):
        """gzip-compress JSON responses when the request advertises
        Accept-Encoding: gzip and the body is >= min_size bytes."""
        self.gzip_min_size = int(min_size)
        return self

    # -- CLI — reference: gofr.go:181-183 -------------------------------------
    def SubCommand(self, pattern: str, handler):
        if self.cmd is None:
            self.cmd = CMD()
        self.cmd.add_route(pattern, handler)

    # -- default routes — reference: gofr.go:102-107 ---------------------------
    def install_default_routes(self) -> None:
        if self._defaults_installed:
            return
        self._defaults_installed = True
        self.router.add("GET", "/.well-known/health", _health_handler)
        self.router.add("GET", "/favicon.ico", _favicon_handler)
        self.router.add_prefix("GET", "/", _catch_all_handler)

    # -- serve — reference: gofr.go:90-126 -------------------------------------
    def Run(self, block: bool = True) -> None:
        if self.cmd is not None and self.cmd.routes:
            self.cmd.run(self.container)
            return
        threads = []
        if self.http_registered or not self.grpc_registered:
            self.install_default_routes()
            engine_kind = (os.environ.get("GOFR_ENGINE") or
                           self.config.GetOrDefault("GOFR_ENGINE", "cpu"))
            if engine_kind == "gpu":
                from .engine import GPUServer
                self.http_server = GPUServer(self, self.http_port)
            else:
                self.http_server = HTTPServer(self, self.http_port)
            self.http_server.start()
            self.container.logger.Infof(
                "HTTP server listening on :%d (engine=%s)",
                self.http_port, engine_kind)
        if self.grpc_registered:
            from .grpc.server import GRPCServer
            self.grpc_server = GRPCServer(self, self.grpc_port)
            self.grpc_server.start()
            self.container.logger.Infof("gRPC server listening on :%d",
                                        self.grpc_port)
        if block:
            # Graceful shutdown on SIGTERM/SIGINT — a conscious fix of
            # the reference's block-forever Run (gofr.go:125 wg.Wait()
            # with no signal handling; SURVEY.md §5 "no graceful
            # shutdown"): in-flight batches complete, sockets close,
            # datasources disconnect.
            import signal
            ev = threading.Event()
            prev = None
            if threading.current_thread() is threading.main_thread():
                prev = signal.signal(signal.SIGTERM,
                                     lambda *_: ev.set())
            try:
                ev.wait()  # blocks until SIGTERM (or KeyboardInterrupt)
            except KeyboardInterrupt:
                pass
            finally:
                if prev is not None:
                    signal.signal(signal.SIGTERM, prev)
                self.container.logger.Infof("shutting down")
                self.shutdown()

    run = Run

    def shutdown(self) -> None:
        if self.http_server is not None:
            self.http_server.stop()
        if self.grpc_server is not None:
            self.grpc_server.stop()
        self.container.close()


def New(config=None) -> App:
    """Reference: gofr.go:49-73."""
    return App(cmd_mode=False, config=config)


def NewCMD(config=None) -> App:
    """Reference: gofr.go:76-87."""
    return App(cmd_mode=True, config=config)
