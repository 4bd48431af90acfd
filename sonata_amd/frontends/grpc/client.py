"""Minimal sonata_grpc client over generic channel methods (no generated
stubs needed).  Useful for tests and as a reference client."""

from __future__ import annotations

import grpc

from .proto import MESSAGES, RPCS, SERVICE_NAME


class SonataGrpcClient:
    def __init__(self, target: str):
        self.channel = grpc.insecure_channel(target)
        self._methods = {}
        for name, (req_t, resp_t, streaming) in RPCS.items():
            path = f"/{SERVICE_NAME}/{name}"
            resp_cls = MESSAGES[resp_t]
            if streaming:
                self._methods[name] = self.channel.unary_stream(
                    path,
                    request_serializer=lambda m: m.SerializeToString(),
                    response_deserializer=resp_cls.FromString,
                )
            else:
                self._methods[name] = self.channel.unary_unary(
                    path,
                    request_serializer=lambda m: m.SerializeToString(),
                    response_deserializer=resp_cls.FromString,
                )

    def close(self):
        self.channel.close()

    def __getattr__(self, name):
        if name in RPCS:
            return self._methods[name]
        raise AttributeError(name)
