"""OIM registry: key/value DB + authenticated gRPC + transparent proxy.

Counterpart of the reference's pkg/oim-registry.  On an MI355X node the
8 GPUs register as 8 independent controllers (``<id>/address`` +
``<id>/pci`` per GPU); clients reach any of them through this one
registry endpoint by setting the ``controllerid`` metadata key.
"""

from .db import RegistryDB, MemRegistryDB, FileRegistryDB
from .registry import Registry, RegistryServer

__all__ = [
    "RegistryDB",
    "MemRegistryDB",
    "FileRegistryDB",
    "Registry",
    "RegistryServer",
]
