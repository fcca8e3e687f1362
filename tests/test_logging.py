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
