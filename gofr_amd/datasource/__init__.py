"""Datasource layer: health contract + reduced logger interface.

Reference: pkg/gofr/datasource/health.go:3-11 (UP/DOWN status types) and
pkg/gofr/datasource/logger.go:9-16 (reduced Logger interface so
datasources don't import the logging package — same layering kept here:
datasource modules only call .Debugf/.Errorf/.log_record on whatever
logger-like object they're handed).
"""

from __future__ import annotations

STATUS_UP = "UP"
STATUS_DOWN = "DOWN"


class Health(dict):
    """Health report: {"status": UP|DOWN, "details": {...}}."""

    def __init__(self, status: str, details: dict | None = None):
        super().__init__(status=status, details=details or {})
