"""Config system tests — mirror reference config/godotenv_test.go patterns."""

import os

from gofr_amd.config import EnvFile, MapConfig, load_dotenv


def test_env_file_loads_and_reads(tmp_path):
    (tmp_path / ".env").write_text(
        "APP_NAME=demo\n"
        "# comment line\n"
        "export HTTP_PORT=9099\n"
        "QUOTED=\"hello world\"\n"
        "SINGLE='sq'\n"
        "INLINE=val # trailing comment\n"
        "BROKENLINE\n")
    for k in ("APP_NAME", "HTTP_PORT", "QUOTED", "SINGLE", "INLINE"):
        os.environ.pop(k, None)
    cfg = EnvFile(str(tmp_path))
    assert cfg.Get("APP_NAME") == "demo"
    assert cfg.Get("HTTP_PORT") == "9099"
    assert cfg.Get("QUOTED") == "hello world"
    assert cfg.Get("SINGLE") == "sq"
    assert cfg.Get("INLINE") == "val"
    assert cfg.Get("MISSING") == ""
    assert cfg.GetOrDefault("MISSING", "x") == "x"
    assert cfg.GetOrDefault("APP_NAME", "x") == "demo"


def test_env_does_not_override_existing(tmp_path):
    (tmp_path / ".env").write_text("PRESET_KEY=file\n")
    os.environ["PRESET_KEY"] = "env"
    try:
        load_dotenv(str(tmp_path / ".env"))
        assert os.environ["PRESET_KEY"] == "env"
        load_dotenv(str(tmp_path / ".env"), override=True)
        assert os.environ["PRESET_KEY"] == "file"
    finally:
        del os.environ["PRESET_KEY"]


def test_missing_env_file_is_fine(tmp_path):
    cfg = EnvFile(str(tmp_path / "nope"))
    assert cfg.Get("ANYTHING") == ""


def test_map_config():
    cfg = MapConfig({"A": "1"})
    assert cfg.Get("A") == "1"
    assert cfg.GetOrDefault("B", "z") == "z"
