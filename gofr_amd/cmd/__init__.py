"""CLI application mode.

Reference: pkg/gofr/cmd.go:12-70 + pkg/gofr/cmd/{request,responder}.go —
non-flag argv words are concatenated into a command string matched
against registered patterns AS REGEXES in registration order; flags
-k / -k=v / --k=v become params (bare flag -> "true"); handler data goes
to stdout, errors to stderr; no match prints "No Command Found!" on
stderr.

Quirk preserved: '-'-prefixed args are stripped
before the command string is built, so a route registered as "-route"
can never match (reference cmd.go:33-41, proven by cmd_test.go:162-179).
"""

from __future__ import annotations

import re
import sys

from ..context import new_context
from ..errors import CommandNotFoundError, GofrError


class CMDRequest:
    """Reference: pkg/gofr/cmd/request.go:25-114."""

    def __init__(self, args: list[str]):
        self.params: dict[str, str] = {}
        self.args = list(args)
        for raw in args:
            if not raw.startswith("-"):
                continue
            body = raw.lstrip("-")
            if not body:
                continue
            if "=" in body:
                k, _, v = body.partition("=")
                if k:
                    self.params[k] = v
            else:
                self.params[body] = "true"

    def Param(self, key: str) -> str:
        return self.params.get(key, "")

    def PathParam(self, key: str) -> str:
        return self.params.get(key, "")

    def Bind(self, into):
        """Reflection bind into object attributes (request.go:87-114):
        only str/bool/int attribute types are converted."""
        for k, v in self.params.items():
            if not hasattr(into, k):
                continue
            cur = getattr(into, k)
            if isinstance(cur, bool):
                setattr(into, k, v.lower() == "true")
            elif isinstance(cur, int):
                try:
                    setattr(into, k, int(v))
                except ValueError:
                    pass
            else:
                setattr(into, k, v)
        return into

    def HostName(self) -> str:
        import socket
        return socket.gethostname()

    param = Param
    path_param = PathParam
    bind = Bind
    host_name = HostName

    # no JSON body / headers in CMD mode
    def header(self, key: str) -> str:
        return ""

    @property
    def client_ip(self) -> str:
        return ""


class CMDResponder:
    """data -> stdout, err -> stderr. Reference: cmd/responder.go:10-19."""

    def __init__(self, out=None, err=None):
        self.out = out or sys.stdout
        self.err = err or sys.stderr

    def Respond(self, data, err) -> None:
        if err is not None:
            print(str(err), file=self.err)
        if data is not None:
            print(data, file=self.out)

    respond = Respond


class CMD:
    """Route table + regex dispatch. Reference: pkg/gofr/cmd.go:12-63."""

    def __init__(self):
        self.routes: list[tuple[str, object]] = []

    def add_route(self, pattern: str, handler) -> None:
        self.routes.append((pattern, handler))

    def _handler(self, command: str):
        # first regex match wins, in registration order (cmd.go:54-63)
        for pattern, h in self.routes:
            try:
                if re.fullmatch(pattern, command) or re.match(pattern, command):
                    return h
            except re.error:
                if pattern == command:
                    return h
        return None

    def run(self, container, argv: list[str] | None = None,
            responder: CMDResponder | None = None) -> int:
        """Reference: cmd.go:27-52."""
        argv = list(sys.argv[1:] if argv is None else argv)
        words = [a for a in argv if not a.startswith("-")]
        command = " ".join(words)
        responder = responder or CMDResponder()
        h = self._handler(command)
        req = CMDRequest(argv)
        ctx = new_context(req, container, responder=responder)
        if h is None:
            responder.Respond(None, CommandNotFoundError())
            return 1
        try:
            result = h(ctx)
            if (isinstance(result, tuple) and len(result) == 2 and
                    (result[1] is None or
                     isinstance(result[1], BaseException))):
                data, err = result
            else:
                data, err = result, None
        except GofrError as e:
            data, err = None, e
        except Exception as e:  # noqa: BLE001
            data, err = None, e
        responder.Respond(data, err)
        return 0 if err is None else 1
