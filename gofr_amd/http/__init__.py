"""HTTP layer: request/responder abstractions, trie router, middlewares.

Reference analog: pkg/gofr/http/ (router, request, responder, response/,
middleware/).
"""

from .request import METHOD_IDS, Request, parse_request_bytes  # noqa: F401
from .responder import (JSON_CT, Responder, ResponseWriter,  # noqa: F401
                        envelope_bytes, reason_phrase)
from .response import File, Raw  # noqa: F401
from .router import Router  # noqa: F401
