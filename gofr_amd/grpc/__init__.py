"""gRPC layer: HTTP/2 transport, protobuf codec, unary dispatch.

Reference analog: pkg/gofr/grpc.go (server wrapping google.golang.org/grpc
with recovery + logging interceptors) and grpc/log.go (RPCLog). Here the
whole stack is from scratch: HTTP/2 framing + HPACK (gofr_amd/grpc/http2),
a descriptor-driven protobuf wire codec (codec.py), and the unary server
(server.py). The batched varint-decode kernel for the GPU path lives in
native/hip/gofr_kernels.hip (k_varint_spans).
"""

from .codec import MessageDesc, decode_message, encode_message  # noqa: F401
from .server import GRPCServer, ServiceDesc  # noqa: F401
