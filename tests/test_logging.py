"""Logging tests — mirror reference logging/logger_test.go (capture +
level filtering) and level_test.go (level parsing/colors)."""

import json

from gofr_amd import logging as gl
from gofr_amd.testutil import MockLogger


def test_level_from_string():
    # table-driven, as CONTRIBUTING.md mandates for the reference
    cases = [("DEBUG", gl.DEBUG), ("debug", gl.DEBUG), ("INFO", gl.INFO),
             ("NOTICE", gl.NOTICE), ("WARN", gl.WARN), ("ERROR", gl.ERROR),
             ("FATAL", gl.FATAL), ("bogus", gl.INFO), ("", gl.INFO)]
    for s, want in cases:
        assert gl.level_from_string(s) == want, s


def test_color_for_status_code():
    assert gl.color_for_status_code(200) == 34
    assert gl.color_for_status_code(404) == 220
    assert gl.color_for_status_code(500) == 160
    assert gl.color_for_status_code(302) == 37


def test_level_filtering_and_streams():
    lg = MockLogger(level=gl.INFO)
    lg.Debugf("hidden %d", 1)
    lg.Infof("shown %d", 2)
    lg.Errorf("bad %s", "thing")
    assert "hidden" not in lg.stdout
    out = [json.loads(line) for line in lg.stdout.splitlines()]
    assert out[0]["level"] == "INFO"
    assert out[0]["message"] == "shown 2"
    # ERROR+ goes to stderr (reference logger.go:60-64)
    err = [json.loads(line) for line in lg.stderr.splitlines()]
    assert err[0]["level"] == "ERROR"
    assert err[0]["message"] == "bad thing"


def test_typed_record_json():
    from gofr_amd.datasource.redis import QueryLog
    lg = MockLogger(level=gl.DEBUG)
    lg.debug_record(QueryLog("GET k", 12.5))
    rec = json.loads(lg.stdout.splitlines()[0])
    assert rec["message"]["query"] == "GET k"
    assert rec["message"]["datasource"] == "redis"


def test_silent_logger_discards():
    lg = gl.NewSilentLogger()
    lg.Errorf("nothing")  # must not raise or print


def test_fatal_exits():
    import pytest
    lg = MockLogger(level=gl.DEBUG)
    with pytest.raises(SystemExit):
        lg.Fatalf("die")
    assert "die" in lg.stderr


def test_all_record_types_pretty_and_json():
    """Every typed record renders in both output modes (the
    reference's per-record pretty formats, logging/logger.go:106-131)."""
    import io

    import gofr_amd.logging as gl
    from gofr_amd.datasource.redis import QueryLog
    from gofr_amd.datasource.sql import Log as SqlLog
    from gofr_amd.engine import BatchLog
    from gofr_amd.grpc.server import RPCLog
    from gofr_amd.http.middleware import RequestLog
    from gofr_amd.service import ErrorLog, Log as SvcLog

    records = [
        RequestLog("abc123", "2026-01-01T00:00:00Z", 150.0, "GET",
                   "curl", "1.2.3.4", "/x?q=1", 200),
        SqlLog("Query", "SELECT 1", 42.0),
        QueryLog("GET key", 7.0),
        SvcLog("abc123", 200, 99.0, "http://svc/x", "GET"),
        ErrorLog("abc123", 99.0, "http://svc/x", "GET", "boom"),
        RPCLog("abc123", "2026-01-01T00:00:00Z", 10.0, "/hello.Hello/Say"),
        BatchLog(32768, 100, 200, 1.25),
    ]
    # JSON mode: every record serializes
    buf = io.StringIO()
    log = gl.Logger(level=gl.DEBUG, out=buf, err=buf, force_json=True)
    for r in records:
        log.log_record(gl.INFO, r)
    import json as _json
    lines = [ln for ln in buf.getvalue().splitlines() if ln.strip()]
    assert len(lines) == len(records)
    for ln in lines:
        _json.loads(ln)
    # pretty mode: every record renders without raising
    buf2 = io.StringIO()
    log2 = gl.Logger(level=gl.DEBUG, out=buf2, err=buf2, force_json=False)
    for r in records:
        log2.log_record(gl.INFO, r)
    out = buf2.getvalue()
    assert "/x?q=1" in out and "SELECT 1" in out and "GET key" in out
    assert "/hello.Hello/Say" in out and "32768" in out
