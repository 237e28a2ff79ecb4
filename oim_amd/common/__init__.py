"""Shared infrastructure (reference pkg/oim-common).

  - server:  NonBlockingGRPCServer equivalent + endpoint parsing
  - tlsutil: mutual-TLS material loading + common-name pinning
  - pci:     PCI BDF parse/merge/pretty with 0xFFFF wildcards
  - paths:   registry key path split/join with sanitization
  - tracing: gRPC payload-logging interceptors + formatters
  - util:    block-size probing, child-process monitor
"""

from .server import NonBlockingGRPCServer, parse_endpoint
from .tlsutil import TLSConfig, load_tls_server_credentials, load_tls_channel_credentials
from .pci import PCIAddress, parse_bdf_string, complete_pci_address, pretty_pci_address
from .paths import join_registry_path, split_registry_path, RegistryPathError

__all__ = [
    "NonBlockingGRPCServer",
    "parse_endpoint",
    "TLSConfig",
    "load_tls_server_credentials",
    "load_tls_channel_credentials",
    "PCIAddress",
    "parse_bdf_string",
    "complete_pci_address",
    "pretty_pci_address",
    "join_registry_path",
    "split_registry_path",
    "RegistryPathError",
]
