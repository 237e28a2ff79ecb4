"""Registry service + transparent controller proxy.

Counterpart of the reference's pkg/oim-registry/registry.go:
  - SetValue/GetValues with TLS-CN authorization (registry.go:100-145)
  - transparent L7 proxying of every non-Registry method to the
    controller named by the ``controllerid`` metadata key
    (registry.go:157-204, spec.md:65-73), with per-request dialing so a
    controller restart never strands the registry.

Authorization model (CN = TLS client common name):
  - ``user.admin`` may set any value and proxy anywhere
  - ``controller.<id>`` may set only ``<id>/address``
  - ``host.<id>`` may proxy only to controller ``<id>``
  - plaintext servers (tests) skip authorization
"""

from __future__ import annotations

from typing import Optional, Sequence

import grpc

from .. import spec
from ..common import split_registry_path
from ..common.server import NonBlockingGRPCServer, grpc_target
from ..common.tlsutil import (
    TLSConfig,
    channel_options_for_peer,
    load_tls_channel_credentials,
    peer_common_name,
)
from ..common.tracing import LogServerInterceptor
from ..log import from_context
from .db import MemRegistryDB, RegistryDB

ADMIN_CN = "user.admin"


class Registry(spec.RegistryServicer):
    """The Registry service + streamDirector-equivalent proxy logic."""

    def __init__(
        self,
        db: Optional[RegistryDB] = None,
        tls: Optional[TLSConfig] = None,
        proxy_tls: Optional[TLSConfig] = None,
    ):
        self.db = db if db is not None else MemRegistryDB()
        self.tls = tls  # server-side credentials; None = plaintext (tests)
        # Credentials used for outgoing proxy dials; defaults to the
        # server's own keypair (the registry authenticates to
        # controllers as component.registry).
        self.proxy_tls = proxy_tls if proxy_tls is not None else tls

    # --- Registry service ---------------------------------------------------

    def _peer_name(self, context) -> str:
        if self.tls is None:
            return ADMIN_CN  # plaintext test mode: everything allowed
        return peer_common_name(context)

    def SetValue(self, request, context):
        try:
            elements = split_registry_path(request.value.path)
        except ValueError as exc:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT, str(exc))
        peer = self._peer_name(context)
        # Authorization (registry.go:100-111): admin, or the controller
        # itself updating its own <id>/address.
        allowed = peer == ADMIN_CN or (
            peer.startswith("controller.")
            and len(elements) == 2
            and elements[0] == peer[len("controller."):]
            and elements[1] == "address"
        )
        if not allowed:
            context.abort(
                grpc.StatusCode.PERMISSION_DENIED,
                f"{peer!r} may not set {request.value.path!r}",
            )
        self.db.store(elements, request.value.value)
        return spec.SetValueReply()

    def GetValues(self, request, context):
        prefix: Sequence[str] = []
        if request.path:
            try:
                prefix = split_registry_path(request.path)
            except ValueError as exc:
                context.abort(grpc.StatusCode.INVALID_ARGUMENT, str(exc))
        reply = spec.GetValuesReply()
        for elements, value in self.db.list(prefix):
            reply.values.add(path="/".join(elements), value=value)
        return reply

    # --- transparent proxy --------------------------------------------------

    def connect(self, method: str, context) -> grpc.Channel:
        """Resolve + authorize a proxied call; returns an open channel.

        Mirrors streamDirector.Connect (registry.go:157-204).
        """
        if method.startswith(f"/{spec.REGISTRY_SERVICE}/"):
            context.abort(
                grpc.StatusCode.UNIMPLEMENTED,
                f"unknown Registry method {method}",
            )
        metadata = dict(context.invocation_metadata())
        controller_id = metadata.get(spec.CONTROLLER_ID_KEY, "")
        if not controller_id:
            context.abort(
                grpc.StatusCode.UNIMPLEMENTED,
                "no controllerid metadata, method unknown to registry",
            )
        peer = self._peer_name(context)
        if peer != ADMIN_CN and peer != f"host.{controller_id}":
            context.abort(
                grpc.StatusCode.PERMISSION_DENIED,
                f"{peer!r} may not access controller {controller_id!r}",
            )
        try:
            elements = split_registry_path(controller_id + "/address")
        except ValueError as exc:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT, str(exc))
        address = self.db.lookup(elements)
        if not address:
            context.abort(
                grpc.StatusCode.UNAVAILABLE,
                f"controller {controller_id!r} not registered",
            )
        target = grpc_target(address)
        if self.proxy_tls is not None:
            creds = load_tls_channel_credentials(self.proxy_tls)
            # Pin the expected server name controller.<id>
            # (registry.go:193-195).
            options = channel_options_for_peer(f"controller.{controller_id}")
            return grpc.secure_channel(target, creds, options=options)
        return grpc.insecure_channel(target)


class _ProxyHandler(grpc.GenericRpcHandler):
    """UnknownServiceHandler analog: splices unknown methods through.

    Uses identity (de)serializers so payload bytes pass through
    untouched — the registry never needs the controller's message types
    (the vgough/grpc-proxy codec trick, reference registry.go:248-261).
    """

    def __init__(self, registry: Registry, handled_prefix: str):
        self.registry = registry
        self.handled_prefix = handled_prefix

    def service(self, handler_call_details):
        method = handler_call_details.method
        if method.startswith(self.handled_prefix):
            return None  # let the real Registry handlers take it

        def proxy(request_iterator, context):
            channel = self.registry.connect(method, context)
            try:
                multi = channel.stream_stream(
                    method,
                    request_serializer=lambda b: b,
                    response_deserializer=lambda b: b,
                )
                # Forward caller metadata (including controllerid).
                metadata = [
                    (k, v)
                    for k, v in context.invocation_metadata()
                    if not k.startswith("grpc-")
                ]
                try:
                    for response in multi(
                        request_iterator,
                        metadata=metadata,
                        timeout=context.time_remaining(),
                    ):
                        yield response
                except grpc.RpcError as err:
                    context.abort(err.code(), err.details())
            finally:
                channel.close()

        return grpc.stream_stream_rpc_method_handler(
            proxy,
            request_deserializer=lambda b: b,
            response_serializer=lambda b: b,
        )


class RegistryServer:
    """Wires Registry + proxy into a NonBlockingGRPCServer."""

    def __init__(self, endpoint: str, registry: Registry):
        self.registry = registry
        self.server = NonBlockingGRPCServer(
            endpoint=endpoint,
            tls=registry.tls,
            interceptors=[LogServerInterceptor()],
        )

    def start(self) -> None:
        def register(server: grpc.Server):
            spec.add_registry_to_server(self.registry, server)
            server.add_generic_rpc_handlers(
                (_ProxyHandler(self.registry, f"/{spec.REGISTRY_SERVICE}/"),)
            )

        self.server.start(register)
        from_context().info("registry started", endpoint=self.server.addr())

    def addr(self) -> str:
        return self.server.addr()

    def stop(self) -> None:
        self.server.stop()

    def run(self) -> None:
        def register(server: grpc.Server):
            spec.add_registry_to_server(self.registry, server)
            server.add_generic_rpc_handlers(
                (_ProxyHandler(self.registry, f"/{spec.REGISTRY_SERVICE}/"),)
            )

        self.server.run(register)
