"""gofr_amd — an MI355X-native microservice framework with GoFr's API.

A brand-new implementation of the capabilities of JigarJoshi04/gofr
(the Go microservice framework) built MI355X-first:
  - public API & semantics: gofr.New()/NewCMD(), app.GET/POST/PUT/DELETE,
    app.Run(), app.SubCommand, app.RegisterService, app.AddHTTPService,
    handler(ctx) with the {"data"/"error"} JSON envelope, config layout
    `configs/.env` with the reference's key set
  - data plane: a request-batch engine — C++ epoll ingress staging raw
    request bytes into pinned rings, hand-written CDNA4 HIP kernels
    (gfx950) for batched HTTP parse, radix-trie route match, JSON
    envelope serialization, gzip, HMAC auth and protobuf varint decode,
    RCCL all-to-all over xGMI for multi-GPU request sharding
  - control plane on CPU: config, logging, tracing, DI container,
    Redis/SQL datasources, inter-service HTTP client, CLI mode.
"""

from .app import App, New, NewCMD  # noqa: F401
from .context import Context  # noqa: F401
from .errors import (CommandNotFoundError, GofrError,  # noqa: F401
                     MissingFileError)
from .http.response import File, Raw  # noqa: F401

__version__ = "0.1.0"
