"""gRPC server example — mirror of reference examples/grpc-server."""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))

import gofr_amd as gofr  # noqa: E402
from gofr_amd.grpc import ServiceDesc  # noqa: E402
from gofr_amd.grpc.codec import HELLO_REQUEST, HELLO_RESPONSE  # noqa: E402


class Server:
    """Reference: examples/grpc-server/grpc/server.go:12-21."""

    def SayHello(self, ctx, req):
        name = req.get("name") or "World"
        return {"message": f"Hello {name}!"}


HELLO_SERVICE = ServiceDesc("hello.Hello", {
    "SayHello": (HELLO_REQUEST, HELLO_RESPONSE),
}, gpu_methods={"SayHello": "hello_echo"})  # batched GPU codec path


def build_app():
    app = gofr.New()
    app.RegisterService(HELLO_SERVICE, Server())
    return app


if __name__ == "__main__":
    build_app().Run()
