"""OIM CSI driver (reference pkg/oim-csi-driver)."""

from .backend import OIMBackend
from .driver import OIMDriver, IdentityServer, ControllerServer, NodeServer, make_params_mapper
from .local import LocalBackend
from .remote import RemoteBackend, malloc_params
from .cephemu import ceph_csi_params, EMULATIONS
from .mount import Mounter, OsExec, FakeExec

__all__ = [
    "OIMBackend",
    "OIMDriver",
    "IdentityServer",
    "ControllerServer",
    "NodeServer",
    "LocalBackend",
    "RemoteBackend",
    "malloc_params",
    "ceph_csi_params",
    "EMULATIONS",
    "make_params_mapper",
    "Mounter",
    "OsExec",
    "FakeExec",
]
