"""Configuration system.

Mirrors the reference's config layer (`pkg/gofr/config/config.go:3-6`,
`pkg/gofr/config/godotenv.go:9-33`): a two-method `Config` interface
(`Get`, `GetOrDefault`) with a single env-file implementation that loads
`./configs/.env` into the process environment and then reads `os.environ`.

The key set and defaults are the reference's:
APP_NAME, HTTP_PORT, GRPC_PORT, LOG_LEVEL, REDIS_HOST/REDIS_PORT,
DB_HOST/DB_USER/DB_PASSWORD/DB_PORT/DB_NAME, TRACER_HOST/TRACER_PORT.
"""

from __future__ import annotations

import os
from typing import Optional


class Config:
    """Abstract config: Get(key) -> str | "", GetOrDefault(key, default)."""

    def Get(self, key: str) -> str:
        raise NotImplementedError

    def GetOrDefault(self, key: str, default: str) -> str:
        v = self.Get(key)
        return v if v else default

    # pythonic aliases
    get = Get
    get_or_default = GetOrDefault


def _parse_env_line(line: str):
    """Parse one KEY=VALUE line of a .env file.

    Supports comments (#), export prefix, single/double quotes —
    the subset of godotenv behavior the reference relies on
    (reference: pkg/gofr/config/godotenv.go:18-22 delegates to godotenv).
    """
    line = line.strip()
    if not line or line.startswith("#"):
        return None
    if line.startswith("export "):
        line = line[len("export "):].lstrip()
    if "=" not in line:
        return None
    key, _, val = line.partition("=")
    key = key.strip()
    val = val.strip()
    if len(val) >= 2 and val[0] == val[-1] and val[0] in ("'", '"'):
        val = val[1:-1]
    else:
        # strip trailing inline comment for unquoted values
        if " #" in val:
            val = val.split(" #", 1)[0].rstrip()
    if not key:
        return None
    return key, val


def load_dotenv(path: str, override: bool = False) -> bool:
    """Load KEY=VALUE pairs from `path` into os.environ.

    Like godotenv.Load: existing environment variables win unless
    override is set. Returns True if the file existed and was read.
    """
    try:
        with open(path, "r", encoding="utf-8") as f:
            lines = f.readlines()
    except OSError:
        return False
    for line in lines:
        kv = _parse_env_line(line)
        if kv is None:
            continue
        k, v = kv
        if override or k not in os.environ:
            os.environ[k] = v
    return True


class EnvFile(Config):
    """Env-file config: loads `<folder>/.env`, reads from os.environ.

    Reference: pkg/gofr/config/godotenv.go:9-33 (NewEnvFile + Get/GetOrDefault).
    """

    def __init__(self, folder: str):
        self.folder = folder
        load_dotenv(os.path.join(folder, ".env"))

    def Get(self, key: str) -> str:
        return os.environ.get(key, "")


class MapConfig(Config):
    """In-memory config for tests and embedding (no reference analog;
    the reference tests pass testutil configs similarly)."""

    def __init__(self, values: Optional[dict] = None):
        self.values = dict(values or {})

    def Get(self, key: str) -> str:
        return str(self.values.get(key, ""))


# Default ports — reference: pkg/gofr/default.go:3-6 and
# pkg/gofr/container/default.go:3-6, pkg/gofr/gofr.go:187.
DEFAULT_HTTP_PORT = 8000
DEFAULT_GRPC_PORT = 9000
DEFAULT_REDIS_PORT = 6379
DEFAULT_MYSQL_PORT = 3306
DEFAULT_TRACER_PORT = 9411
