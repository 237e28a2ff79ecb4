"""Typed wrappers over the hipstored RPC methods (reference pkg/spdk/spdk.go).

Parameter and result field names are the SPDK wire contract; see
docs/spec.md "hipstored JSON-RPC surface".
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

from .client import Client


@dataclass
class BDev:
    name: str
    product_name: str
    uuid: str
    block_size: int
    num_blocks: int
    claimed: bool
    driver_specific: dict = field(default_factory=dict)

    @property
    def size_bytes(self) -> int:
        return self.block_size * self.num_blocks


@dataclass
class NBDDisk:
    bdev_name: str
    nbd_device: str


@dataclass
class SCSILun:
    lun: int
    bdev_name: str


@dataclass
class SCSITarget:
    target_name: str
    id: int
    scsi_dev_num: int
    luns: List[SCSILun]


@dataclass
class VHostController:
    controller: str
    cpumask: str
    scsi_targets: List[SCSITarget]


def get_bdevs(client: Client, name: str = "") -> List[BDev]:
    params = {"name": name} if name else {}
    result = client.invoke("get_bdevs", params)
    return [
        BDev(
            name=b["name"],
            product_name=b["product_name"],
            uuid=b.get("uuid", ""),
            block_size=b["block_size"],
            num_blocks=b["num_blocks"],
            claimed=b.get("claimed", False),
            driver_specific=b.get("driver_specific", {}),
        )
        for b in result
    ]


def delete_bdev(client: Client, name: str) -> None:
    client.invoke("delete_bdev", {"name": name})


def construct_malloc_bdev(
    client: Client, num_blocks: int, block_size: int, name: str = "",
    uuid: str = ""
) -> str:
    params = {"num_blocks": num_blocks, "block_size": block_size}
    if name:
        params["name"] = name
    if uuid:
        params["uuid"] = uuid
    return client.invoke("construct_malloc_bdev", params)


def construct_aio_bdev(client: Client, name: str, filename: str,
                       block_size: int = 512) -> str:
    """File-backed bdev (SPDK aio): data survives daemon restarts."""
    return client.invoke("construct_aio_bdev", {
        "name": name, "filename": filename, "block_size": block_size})


def construct_rbd_bdev(
    client: Client,
    pool_name: str,
    rbd_name: str,
    block_size: int,
    name: str = "",
    user_id: str = "",
    config: Optional[Dict[str, str]] = None,
) -> str:
    params = {
        "pool_name": pool_name,
        "rbd_name": rbd_name,
        "block_size": block_size,
    }
    if name:
        params["name"] = name
    if user_id:
        params["user_id"] = user_id
    if config:
        params["config"] = config
    return client.invoke("construct_rbd_bdev", params)


def start_nbd_disk(client: Client, bdev_name: str, nbd_device: str) -> None:
    client.invoke("start_nbd_disk", {"bdev_name": bdev_name, "nbd_device": nbd_device})


def get_nbd_disks(client: Client) -> List[NBDDisk]:
    result = client.invoke("get_nbd_disks")
    return [NBDDisk(bdev_name=d["bdev_name"], nbd_device=d["nbd_device"]) for d in result]


def stop_nbd_disk(client: Client, nbd_device: str) -> None:
    client.invoke("stop_nbd_disk", {"nbd_device": nbd_device})


def construct_vhost_scsi_controller(
    client: Client, controller: str, cpumask: str = ""
) -> None:
    params = {"ctrlr": controller}
    if cpumask:
        params["cpumask"] = cpumask
    client.invoke("construct_vhost_scsi_controller", params)


def add_vhost_scsi_lun(
    client: Client, controller: str, scsi_target_num: int, bdev_name: str
) -> None:
    client.invoke(
        "add_vhost_scsi_lun",
        {"ctrlr": controller, "scsi_target_num": scsi_target_num,
         "bdev_name": bdev_name},
    )


def remove_vhost_scsi_target(
    client: Client, controller: str, scsi_target_num: int
) -> None:
    client.invoke(
        "remove_vhost_scsi_target",
        {"ctrlr": controller, "scsi_target_num": scsi_target_num},
    )


def remove_vhost_controller(client: Client, controller: str) -> None:
    client.invoke("remove_vhost_controller", {"ctrlr": controller})


def get_vhost_controllers(client: Client) -> List[VHostController]:
    result = client.invoke("get_vhost_controllers")
    controllers = []
    for c in result:
        targets = []
        for t in (c.get("backend_specific", {}).get("scsi") or []):
            luns = [SCSILun(lun=l["id"], bdev_name=l["bdev_name"])
                    for l in t.get("luns", [])]
            targets.append(
                SCSITarget(
                    target_name=t.get("target_name", ""),
                    id=t.get("id", 0),
                    scsi_dev_num=t.get("scsi_dev_num", 0),
                    luns=luns,
                )
            )
        controllers.append(
            VHostController(
                controller=c["ctrlr"],
                cpumask=c.get("cpumask", ""),
                scsi_targets=targets,
            )
        )
    return controllers


@dataclass
class BDevIostat:
    name: str
    num_read_ops: int
    num_write_ops: int
    num_unmap_ops: int
    bytes_read: int
    bytes_written: int


def construct_striped_malloc_bdev(client: Client, name: str,
                                  num_blocks: int, block_size: int,
                                  stripe_size_kb: int = 128,
                                  count: int = 2,
                                  devices: Optional[List[int]] = None) -> str:
    """Striped malloc bdev across GPUs (oim-amd extension)."""
    params = {"name": name, "num_blocks": num_blocks,
              "block_size": block_size, "stripe_size_kb": stripe_size_kb}
    if devices:
        params["devices"] = devices
    else:
        params["count"] = count
    return client.invoke("construct_striped_malloc_bdev", params)


def construct_replicated_malloc_bdev(client: Client, name: str,
                                     num_blocks: int, block_size: int,
                                     count: int = 2,
                                     devices: Optional[List[int]] = None
                                     ) -> str:
    """Replicated malloc bdev across GPUs (oim-amd extension)."""
    params = {"name": name, "num_blocks": num_blocks,
              "block_size": block_size}
    if devices:
        params["devices"] = devices
    else:
        params["count"] = count
    return client.invoke("construct_replicated_malloc_bdev", params)


def bdev_clone(client: Client, src: str, name: str) -> str:
    """Device-side clone (HBM rates; xGMI across GPUs)."""
    return client.invoke("bdev_clone", {"src": src, "name": name})


def resize_malloc_bdev(client: Client, name: str, size: int) -> int:
    """Offline resize to `size` bytes; returns the new block count."""
    return client.invoke("resize_malloc_bdev",
                         {"name": name, "size": size})["num_blocks"]


def nvmf_create_target(client: Client, listen_addr: str = "127.0.0.1",
                       port: int = 0, subnqn: str = "",
                       bdevs: Optional[List[str]] = None,
                       digests: bool = True) -> Dict:
    params = {"listen_addr": listen_addr, "port": port, "digests": digests,
              "bdevs": bdevs or []}
    if subnqn:
        params["subnqn"] = subnqn
    return client.invoke("nvmf_create_target", params)


def nvmf_delete_target(client: Client, subnqn: str) -> None:
    client.invoke("nvmf_delete_target", {"subnqn": subnqn})


def save_config(client: Client) -> Dict:
    return client.invoke("save_config")


def load_config(client: Client, config: Dict) -> int:
    """Replays a save_config snapshot; returns entries applied."""
    return client.invoke("load_config", config)


def get_bdevs_iostat(client: Client, name: str = "") -> List[BDevIostat]:
    params = {"name": name} if name else {}
    result = client.invoke("get_bdevs_iostat", params)
    return [
        BDevIostat(
            name=b["name"],
            num_read_ops=b["num_read_ops"],
            num_write_ops=b["num_write_ops"],
            num_unmap_ops=b["num_unmap_ops"],
            bytes_read=b["bytes_read"],
            bytes_written=b["bytes_written"],
        )
        for b in result["bdevs"]
    ]


def perf_run(
    client: Client,
    bdev_name: str,
    workload: str = "randread",
    io_size: int = 4096,
    queue_depth: int = 32,
    num_queues: int = 1,
    seconds: float = 2.0,
    max_ios: int = 0,
) -> dict:
    return client.invoke(
        "perf_run",
        {
            "bdev_name": bdev_name,
            "workload": workload,
            "io_size": io_size,
            "queue_depth": queue_depth,
            "num_queues": num_queues,
            "seconds": seconds,
            "max_ios": max_ios,
        },
    )
