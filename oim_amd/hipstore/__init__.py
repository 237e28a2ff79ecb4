"""Client bindings for hipstored's JSON-RPC socket.

Counterpart of the reference's pkg/spdk (client.go + spdk.go): a
JSON-RPC 2.0 client over a Unix socket plus typed wrappers for every
method the control plane invokes. Wire-compatible with SPDK's RPC
plane, so this client also drives a stock SPDK daemon.
"""

from .client import Client, RpcError, ERROR_INVALID_PARAMS, ERROR_INTERNAL
from .api import (
    BDev,
    NBDDisk,
    VHostController,
    SCSITarget,
    SCSILun,
    get_bdevs,
    delete_bdev,
    construct_malloc_bdev,
    construct_aio_bdev,
    construct_rbd_bdev,
    start_nbd_disk,
    get_nbd_disks,
    stop_nbd_disk,
    construct_vhost_scsi_controller,
    add_vhost_scsi_lun,
    remove_vhost_scsi_target,
    remove_vhost_controller,
    get_vhost_controllers,
    perf_run,
    BDevIostat,
    get_bdevs_iostat,
)

__all__ = [
    "Client",
    "RpcError",
    "ERROR_INVALID_PARAMS",
    "ERROR_INTERNAL",
    "BDev",
    "NBDDisk",
    "VHostController",
    "SCSITarget",
    "SCSILun",
    "get_bdevs",
    "delete_bdev",
    "construct_malloc_bdev",
    "construct_aio_bdev",
    "construct_rbd_bdev",
    "start_nbd_disk",
    "get_nbd_disks",
    "stop_nbd_disk",
    "construct_vhost_scsi_controller",
    "add_vhost_scsi_lun",
    "remove_vhost_scsi_target",
    "remove_vhost_controller",
    "get_vhost_controllers",
    "perf_run",
    "BDevIostat",
    "get_bdevs_iostat",
]
