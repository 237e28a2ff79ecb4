"""Mutual-TLS material + common-name pinning (reference grpc.go:77-137).

Certificate naming convention (kept from the reference's CA layout):
every component has a keypair ``<role>.<name>.crt`` / ``<role>.<name>.key``
whose certificate CommonName (and SAN DNS entry) is ``<role>.<name>`` —
e.g. ``component.registry``, ``controller.host-0``, ``host.host-0``,
``user.admin``.  Clients pin the expected server name; servers require a
client certificate signed by the shared CA and authorize on its CN
(reference registry.go:67-111, 180-195).
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Optional, Tuple

import grpc


@dataclass
class TLSConfig:
    """Paths to CA plus this component's keypair and the peer to expect."""

    ca: str
    key: str
    cert: Optional[str] = None
    peer_name: str = ""  # expected server CN when dialing; "" accepts any CA-signed peer

    def resolve_cert(self) -> str:
        if self.cert:
            return self.cert
        root, ext = os.path.splitext(self.key)
        if ext == ".key":
            return root + ".crt"
        raise ValueError(f"cannot derive certificate path from key path {self.key!r}")

    def read(self) -> Tuple[bytes, bytes, bytes]:
        with open(self.ca, "rb") as f:
            ca = f.read()
        with open(self.resolve_cert(), "rb") as f:
            cert = f.read()
        with open(self.key, "rb") as f:
            key = f.read()
        return ca, cert, key


def load_tls_server_credentials(config: TLSConfig) -> grpc.ServerCredentials:
    """Server credentials requiring a CA-signed client certificate."""
    ca, cert, key = config.read()
    return grpc.ssl_server_credentials(
        [(key, cert)],
        root_certificates=ca,
        require_client_auth=True,
    )


def load_tls_channel_credentials(config: TLSConfig) -> grpc.ChannelCredentials:
    ca, cert, key = config.read()
    return grpc.ssl_channel_credentials(
        root_certificates=ca, private_key=key, certificate_chain=cert
    )


def channel_options_for_peer(peer_name: str):
    """Channel options pinning the expected server certificate name.

    gRPC validates the server certificate against the target name; the
    override lets us dial an IP/socket while still enforcing the
    certificate CN/SAN (reference registry.go:193-195 pins
    ``controller.<id>`` when proxying).
    """
    if not peer_name:
        return []
    return [("grpc.ssl_target_name_override", peer_name)]


def peer_common_name(context: grpc.ServicerContext) -> str:
    """CN of the authenticated TLS client, "" when unauthenticated.

    Reference getPeer (registry.go:67-82).
    """
    auth = context.auth_context()
    for key in ("x509_common_name",):
        values = auth.get(key)
        if values:
            return values[0].decode()
    # grpcio may only expose the SAN list; our certs carry CN as SAN too.
    values = auth.get("x509_subject_alternative_name")
    if values:
        return values[0].decode()
    return ""
