"""CLI example — mirror of reference examples/sample-cmd/main.go."""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))

import gofr_amd as gofr  # noqa: E402


def build_app():
    app = gofr.NewCMD()
    app.SubCommand("hello", lambda c: "Hello World!")
    app.SubCommand("params", lambda c: f"Hello {c.Param('name')}!")
    return app


if __name__ == "__main__":
    build_app().Run()
