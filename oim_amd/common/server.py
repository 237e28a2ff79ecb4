"""Non-blocking gRPC server wrapper (reference pkg/oim-common/server.go).

Endpoints use the reference's URL convention (server.go:28-40):
``unix:///abs/path.sock``, ``tcp://host:port`` (also tcp4/tcp6).
``tcp://:0`` binds an ephemeral port; :meth:`NonBlockingGRPCServer.addr`
reports the bound address for tests (server.go:104-115).
"""

from __future__ import annotations

import os
import signal
import threading
from concurrent import futures
from typing import Optional, Sequence, Tuple

import grpc

from ..log import from_context
from .tlsutil import TLSConfig, load_tls_server_credentials


def parse_endpoint(endpoint: str) -> Tuple[str, str]:
    """Split an endpoint into (scheme, address); raises on unknown schemes."""
    for scheme in ("unix", "tcp", "tcp4", "tcp6"):
        prefix = scheme + "://"
        if endpoint.startswith(prefix):
            address = endpoint[len(prefix):]
            if not address:
                raise ValueError(f"missing address in endpoint {endpoint!r}")
            return scheme, address
    raise ValueError(f"unsupported endpoint scheme: {endpoint!r}")


def grpc_target(endpoint: str) -> str:
    """Endpoint -> grpc dial target."""
    scheme, address = parse_endpoint(endpoint)
    if scheme == "unix":
        return "unix:" + address
    if scheme == "tcp6":
        return "ipv6:" + address
    return address


class NonBlockingGRPCServer:
    """Start/wait/stop lifecycle around a grpc.Server.

    Interceptors and service registration callables are supplied by the
    component (registry/controller/CSI driver); TLS is optional so unit
    tests can run over plaintext unix sockets.
    """

    def __init__(
        self,
        endpoint: str,
        tls: Optional[TLSConfig] = None,
        interceptors: Sequence[grpc.ServerInterceptor] = (),
        max_workers: int = 16,
    ):
        self.endpoint = endpoint
        self.tls = tls
        self.interceptors = list(interceptors)
        self.max_workers = max_workers
        self._server: Optional[grpc.Server] = None
        self._bound_port: Optional[int] = None
        self._cleanup_socket: Optional[str] = None
        self._stopped = threading.Event()

    def start(self, *register) -> None:
        """Create the server, call each ``register(server)``, and serve."""
        if self._server is not None:
            raise RuntimeError("server already started")
        server = grpc.server(
            futures.ThreadPoolExecutor(max_workers=self.max_workers),
            interceptors=self.interceptors,
            options=[("grpc.so_reuseport", 0)],
        )
        for fn in register:
            fn(server)
        scheme, address = parse_endpoint(self.endpoint)
        if scheme == "unix":
            # Remove a stale socket from an earlier run (server.go:57-66).
            if os.path.exists(address):
                os.unlink(address)
            target = "unix:" + address
            self._cleanup_socket = address
        else:
            target = address
        if self.tls is not None:
            creds = load_tls_server_credentials(self.tls)
            port = server.add_secure_port(target, creds)
        else:
            port = server.add_insecure_port(target)
        if scheme != "unix" and port == 0:
            raise RuntimeError(f"failed to bind {self.endpoint}")
        self._bound_port = port
        self._server = server
        server.start()
        from_context().info("listening", endpoint=self.endpoint, port=port)

    def addr(self) -> str:
        """The bound address, resolving an ephemeral port (tcp://:0)."""
        scheme, address = parse_endpoint(self.endpoint)
        if scheme == "unix":
            return self.endpoint
        host, _, port = address.rpartition(":")
        if port in ("0", "") and self._bound_port:
            port = str(self._bound_port)
        host = host or "127.0.0.1"
        return f"{scheme}://{host}:{port}"

    def wait(self) -> None:
        if self._server is not None:
            self._server.wait_for_termination()

    def stop(self, grace: float = 5.0) -> None:
        if self._server is not None:
            self._server.stop(grace).wait()
            self._server = None
        if self._cleanup_socket and os.path.exists(self._cleanup_socket):
            os.unlink(self._cleanup_socket)
        self._stopped.set()

    def force_stop(self) -> None:
        self.stop(grace=0.0)

    def run(self, *register) -> None:
        """start() + wait(), stopping cleanly on SIGINT/SIGTERM."""
        self.start(*register)

        def handler(signum, frame):
            from_context().info("terminating", signal=signum)
            threading.Thread(target=self.stop, daemon=True).start()

        signal.signal(signal.SIGINT, handler)
        signal.signal(signal.SIGTERM, handler)
        self.wait()
