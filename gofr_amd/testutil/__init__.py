"""Test utilities.

Reference: pkg/gofr/testutil/os.go:8-36 (stdout/stderr capture around a
closure) and testutil/mock_logger.go (level-stamped capture logger).
"""

from __future__ import annotations

import contextlib
import io

from .. import logging as gofr_logging


def stdout_output_for_func(fn) -> str:
    """Run fn, return what it wrote to stdout."""
    buf = io.StringIO()
    with contextlib.redirect_stdout(buf):
        fn()
    return buf.getvalue()


def stderr_output_for_func(fn) -> str:
    buf = io.StringIO()
    with contextlib.redirect_stderr(buf):
        fn()
    return buf.getvalue()


# Go-style aliases (reference: StdoutOutputForFunc / StderrOutputForFunc)
StdoutOutputForFunc = stdout_output_for_func
StderrOutputForFunc = stderr_output_for_func


class MockLogger(gofr_logging.Logger):
    """Captures log lines in-memory, JSON mode, at DEBUG level."""

    def __init__(self, level: int = gofr_logging.DEBUG):
        self.out_buf = io.StringIO()
        self.err_buf = io.StringIO()
        super().__init__(level=level, out=self.out_buf, err=self.err_buf,
                         force_json=True)

    @property
    def stdout(self) -> str:
        return self.out_buf.getvalue()

    @property
    def stderr(self) -> str:
        return self.err_buf.getvalue()
