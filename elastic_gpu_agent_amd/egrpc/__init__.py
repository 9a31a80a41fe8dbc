"""egrpc — purpose-built minimal gRPC-over-HTTP/2 transport (unix sockets).

Serves the kubelet device-plugin surface and speaks to kubelet sockets with
~10× lower per-RPC latency than grpcio's Python bindings on this host class.
Wire-interop with real gRPC stacks is continuously verified in the test
suite (grpcio client ↔ egrpc server and the reverse).
"""
from .client import Channel  # noqa: F401
from .core import ConnectionClosed, EgrpcError  # noqa: F401
from .core import (  # noqa: F401
    CANCELLED,
    DEADLINE_EXCEEDED,
    FAILED_PRECONDITION,
    INTERNAL,
    INVALID_ARGUMENT,
    NOT_FOUND,
    OK,
    RESOURCE_EXHAUSTED,
    UNAVAILABLE,
    UNKNOWN,
)
from .server import Method, Server, ServerContext, unary_stream, unary_unary  # noqa: F401
