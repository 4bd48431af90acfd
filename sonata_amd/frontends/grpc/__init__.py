"""gRPC frontend: wire-compatible with the reference sonata_grpc service
(crates/frontends/grpc/proto/sonata_grpc.proto, server main.rs)."""

from .proto import MESSAGES, SERVICE_NAME  # noqa: F401
from .server import SonataGrpcService, create_server, serve  # noqa: F401
