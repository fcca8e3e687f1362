"""CLI mode (reference cmd_test.go:14-205 + cmd/request_test.go:9-44):
regex route dispatch over non-flag argv words, flag parsing, reflection
Bind, stdout/stderr split, and the preserved '-route' quirk."""

import io

import gofr_amd
from gofr_amd.cmd import CMDRequest, CMDResponder
from gofr_amd.config import MapConfig
from gofr_amd.testutil import (StderrOutputForFunc, StdoutOutputForFunc)


def new_cmd_app():
    return gofr_amd.NewCMD(config=MapConfig({"LOG_LEVEL": "FATAL"}))


def run(app, argv):
    out, err = io.StringIO(), io.StringIO()
    rc = app.cmd.run(app.container, argv,
                     responder=CMDResponder(out=out, err=err))
    return rc, out.getvalue(), err.getvalue()


def test_dispatch_and_stdout():
    app = new_cmd_app()
    app.SubCommand("hello", lambda ctx: "Hello World!")
    rc, out, err = run(app, ["hello"])
    assert rc == 0 and out.strip() == "Hello World!" and err == ""


def test_regex_match_registration_order():
    # reference cmd.go:54-63: first regex match wins in order
    app = new_cmd_app()
    app.SubCommand("log.*", lambda ctx: "log-star")
    app.SubCommand("login", lambda ctx: "exact-login")
    rc, out, _ = run(app, ["login"])
    assert out.strip() == "log-star"


def test_multiword_command():
    app = new_cmd_app()
    app.SubCommand("user add", lambda ctx: "added " + ctx.Param("name"))
    rc, out, _ = run(app, ["user", "add", "-name=ada"])
    assert out.strip() == "added ada"


def test_flag_parsing_forms():
    # reference cmd/request.go:36-60: -k / -k=v / --k=v, bare flag = true
    r = CMDRequest(["-a", "-b=2", "--c=three", "positional", "-", "--"])
    assert r.Param("a") == "true"
    assert r.Param("b") == "2"
    assert r.Param("c") == "three"
    assert r.Param("missing") == ""


def test_bind_reflection():
    # reference cmd/request.go:87-114: String/Bool/Int attr conversion
    class Opts:
        def __init__(self):
            self.name = ""
            self.count = 0
            self.force = False

    r = CMDRequest(["-name=ada", "-count=3", "-force"])
    o = r.Bind(Opts())
    assert o.name == "ada" and o.count == 3 and o.force is True


def test_no_command_found_stderr():
    # reference cmd.go:21-25,46-49
    app = new_cmd_app()
    app.SubCommand("known", lambda ctx: "ok")
    rc, out, err = run(app, ["unknown"])
    assert rc == 1 and out == "" and "No Command Found!" in err


def test_error_goes_to_stderr():
    app = new_cmd_app()

    def boom(ctx):
        raise ValueError("bad input")

    app.SubCommand("boom", boom)
    rc, out, err = run(app, ["boom"])
    assert rc == 1 and "bad input" in err and out == ""


def test_dash_route_quirk_preserved():
    # '-'-prefixed argv words are stripped before the
    # command string is built, so a route registered as "-route" can
    # never match (reference cmd.go:33-41, cmd_test.go:162-179)
    app = new_cmd_app()
    app.SubCommand("-route", lambda ctx: "never")
    rc, out, err = run(app, ["-route"])
    assert rc == 1 and "No Command Found!" in err


def test_capture_helpers():
    # testutil/os.go:8-36 parity
    assert StdoutOutputForFunc(lambda: print("to-out")) == "to-out\n"
    import sys
    assert StderrOutputForFunc(
        lambda: print("to-err", file=sys.stderr)) == "to-err\n"
