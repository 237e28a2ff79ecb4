"""The per-card agent driving hipstored (reference pkg/oim-controller).

One controller runs per MI355X GPU ("card"): it serves the oim.v0
Controller API, translates MapVolume/UnmapVolume/Provision into
hipstored RPC calls, and self-registers its address (and its GPU's PCI
BDF) with the registry on a timer so a wiped registry heals itself
(reference controller.go:411-468, README.md:147-151).

Semantics kept from the reference (controller.go:55-278):
  - MapVolume is idempotent: an existing LUN for the volume returns the
    same target; Malloc volumes must be pre-provisioned; Ceph volumes
    are created on first map.
  - UnmapVolume removes the SCSI target and deletes the bdev only when
    it is NOT a Malloc bdev (product_name check) — Malloc data survives
    Map/Unmap cycles.
  - ProvisionMallocBDev: size>0 creates (AlreadyExists on size
    mismatch), size==0 deletes; CheckMallocBDev -> NOT_FOUND.
"""

from __future__ import annotations

import threading
from contextlib import contextmanager
from typing import Dict, Optional

import grpc

from .. import hipstore, spec
from ..common.server import NonBlockingGRPCServer, grpc_target
from ..common.tlsutil import (
    TLSConfig,
    channel_options_for_peer,
    load_tls_channel_credentials,
)
from ..common.tracing import LogClientInterceptor, LogServerInterceptor
from ..common.pci import parse_bdf_string
from ..log import from_context

MAX_SCSI_TARGETS = 8  # matches hipstored / reference vhost_scsi default


class _KeyedMutex:
    """Per-key locking (reference keymutex, controller.go:44-51)."""

    def __init__(self):
        self._guard = threading.Lock()
        self._locks: Dict[str, threading.Lock] = {}

    @contextmanager
    def locked(self, key: str):
        with self._guard:
            lock = self._locks.setdefault(key, threading.Lock())
        lock.acquire()
        try:
            yield
        finally:
            lock.release()


class Controller(spec.ControllerServicer):
    def __init__(
        self,
        controller_id: str,
        hipstored_socket: str,
        vhost_controller: str = "vhost.0",
        vm_vhost_device: str = "",
        controller_address: str = "",
        registry_address: str = "",
        registry_delay: float = 60.0,
        tls: Optional[TLSConfig] = None,
        pci_address: str = "",
    ):
        self.controller_id = controller_id
        self.hipstored_socket = hipstored_socket
        self.vhost_controller = vhost_controller
        self.vm_vhost_device = vm_vhost_device
        self.controller_address = controller_address
        self.registry_address = registry_address
        self.registry_delay = registry_delay
        self.tls = tls
        self.pci_address = pci_address  # this GPU's BDF, for <id>/pci
        self._volume_mutex = _KeyedMutex()
        self._stop_event = threading.Event()
        self._register_thread: Optional[threading.Thread] = None
        self._ensure_vhost_lock = threading.Lock()

    # --- hipstored plumbing -------------------------------------------------

    def _client(self) -> hipstore.Client:
        return hipstore.Client(self.hipstored_socket)

    def _ensure_vhost_controller(self, client: hipstore.Client) -> None:
        """Create our SCSI controller object on first use (idempotent)."""
        with self._ensure_vhost_lock:
            controllers = hipstore.get_vhost_controllers(client)
            if not any(c.controller == self.vhost_controller for c in controllers):
                hipstore.construct_vhost_scsi_controller(client, self.vhost_controller)

    def _reply_pci(self) -> spec.PCIAddress:
        """PCI address of this card for MapVolumeReply.

        Priority: explicit --vm-vhost-device (VM deployments), else the
        GPU's own BDF, else all-unset 0xFFFF (reference spec.md:141-143).
        """
        source = self.vm_vhost_device or self.pci_address
        if source:
            try:
                addr = parse_bdf_string(source)
                return spec.PCIAddress(
                    domain=addr.domain, bus=addr.bus,
                    device=addr.device, function=addr.function,
                )
            except ValueError:
                from_context().warn("bad PCI address", value=source)
        return spec.PCIAddress(domain=0xFFFF, bus=0xFFFF, device=0xFFFF,
                               function=0xFFFF)

    # --- Controller service -------------------------------------------------

    def MapVolume(self, request, context):
        volume_id = request.volume_id
        if not volume_id:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT, "empty volume ID")
        with self._volume_mutex.locked(volume_id):
            with self._client() as client:
                # Reuse-or-create the bdev (controller.go:75-93).
                try:
                    bdevs = hipstore.get_bdevs(client, volume_id)
                except hipstore.RpcError as err:
                    if not err.is_not_found():
                        context.abort(grpc.StatusCode.INTERNAL, str(err))
                    bdevs = []
                if not bdevs:
                    which = request.WhichOneof("params")
                    if which == "malloc":
                        context.abort(
                            grpc.StatusCode.NOT_FOUND,
                            f"no existing MallocBDev with name {volume_id} found",
                        )
                    elif which == "ceph":
                        self._map_ceph(client, volume_id, request.ceph, context)
                    else:
                        context.abort(
                            grpc.StatusCode.INVALID_ARGUMENT,
                            "unsupported or missing volume parameters",
                        )
                self._ensure_vhost_controller(client)
                # Idempotency scan (controller.go:99-125): already a LUN?
                controllers = hipstore.get_vhost_controllers(client)
                targets_used = set()
                for ctrl in controllers:
                    if ctrl.controller != self.vhost_controller:
                        continue
                    for target in ctrl.scsi_targets:
                        targets_used.add(target.scsi_dev_num)
                        for lun in target.luns:
                            if lun.bdev_name == volume_id:
                                return spec.MapVolumeReply(
                                    pci_address=self._reply_pci(),
                                    scsi_disk=spec.SCSIDisk(
                                        target=target.scsi_dev_num, lun=0),
                                )
                # Attach to the first free target 0..7 (controller.go:
                # 131-148). Concurrent MapVolume calls for DIFFERENT
                # volumes hold different keyed mutexes and can race for
                # the same free target; "occupied" just means another
                # volume won it — try the next one.
                last_err = None
                for target_num in range(MAX_SCSI_TARGETS):
                    if target_num in targets_used:
                        continue
                    try:
                        hipstore.add_vhost_scsi_lun(
                            client, self.vhost_controller, target_num, volume_id)
                    except hipstore.RpcError as err:
                        if "occupied" in err.message:
                            last_err = err
                            continue
                        context.abort(grpc.StatusCode.INTERNAL, str(err))
                    return spec.MapVolumeReply(
                        pci_address=self._reply_pci(),
                        scsi_disk=spec.SCSIDisk(target=target_num, lun=0),
                    )
                if last_err is not None:
                    context.abort(
                        grpc.StatusCode.RESOURCE_EXHAUSTED,
                        f"no free SCSI target on {self.vhost_controller}: "
                        f"{last_err}")
                context.abort(
                    grpc.StatusCode.RESOURCE_EXHAUSTED,
                    f"no free SCSI target on {self.vhost_controller}",
                )

    def _map_ceph(self, client, volume_id, ceph, context):
        """construct_rbd_bdev from CephParams (controller.go:280-297)."""
        config = {}
        if ceph.monitors:
            config["mon_host"] = ceph.monitors
        if ceph.secret:
            config["key"] = ceph.secret
        try:
            hipstore.construct_rbd_bdev(
                client,
                pool_name=ceph.pool,
                rbd_name=ceph.image,
                block_size=512,
                name=volume_id,
                user_id=ceph.user_id,
                config=config,
            )
        except hipstore.RpcError as err:
            context.abort(grpc.StatusCode.INTERNAL, f"ConstructRBDBDev: {err}")

    def UnmapVolume(self, request, context):
        volume_id = request.volume_id
        if not volume_id:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT, "empty volume ID")
        with self._volume_mutex.locked(volume_id):
            with self._client() as client:
                # Find and remove the SCSI target (controller.go:159-201).
                try:
                    controllers = hipstore.get_vhost_controllers(client)
                except hipstore.RpcError as err:
                    context.abort(grpc.StatusCode.INTERNAL, str(err))
                for ctrl in controllers:
                    if ctrl.controller != self.vhost_controller:
                        continue
                    for target in ctrl.scsi_targets:
                        if any(l.bdev_name == volume_id for l in target.luns):
                            try:
                                hipstore.remove_vhost_scsi_target(
                                    client, self.vhost_controller,
                                    target.scsi_dev_num)
                            except hipstore.RpcError as err:
                                if not err.is_not_found():
                                    context.abort(grpc.StatusCode.INTERNAL,
                                                  str(err))
                # Delete the bdev unless it is a Malloc bdev
                # (controller.go:203-209: Malloc data survives unmap).
                try:
                    bdevs = hipstore.get_bdevs(client, volume_id)
                    if bdevs and bdevs[0].product_name != "Malloc disk":
                        hipstore.delete_bdev(client, volume_id)
                except hipstore.RpcError as err:
                    if not err.is_not_found():
                        context.abort(grpc.StatusCode.INTERNAL, str(err))
        return spec.UnmapVolumeReply()

    def ProvisionMallocBDev(self, request, context):
        name = request.bdev_name
        size = request.size
        if not name:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT, "empty bdev name")
        if size < 0 or size % 512 != 0:
            context.abort(
                grpc.StatusCode.INVALID_ARGUMENT,
                f"size {size} must be a non-negative multiple of 512",
            )
        with self._volume_mutex.locked(name):
            with self._client() as client:
                try:
                    bdevs = hipstore.get_bdevs(client, name)
                except hipstore.RpcError as err:
                    if not err.is_not_found():
                        context.abort(grpc.StatusCode.INTERNAL, str(err))
                    bdevs = []
                if size > 0:
                    if bdevs:
                        if bdevs[0].size_bytes != size:
                            context.abort(
                                grpc.StatusCode.ALREADY_EXISTS,
                                f"bdev {name} exists with size "
                                f"{bdevs[0].size_bytes}, requested {size}",
                            )
                        return spec.ProvisionMallocBDevReply()
                    try:
                        hipstore.construct_malloc_bdev(
                            client, num_blocks=size // 512, block_size=512,
                            name=name)
                    except hipstore.RpcError as err:
                        context.abort(grpc.StatusCode.INTERNAL,
                                      f"ConstructMallocBDev: {err}")
                else:
                    if bdevs:
                        try:
                            hipstore.delete_bdev(client, name)
                        except hipstore.RpcError as err:
                            if not err.is_not_found():
                                context.abort(grpc.StatusCode.INTERNAL,
                                              f"DeleteBDev: {err}")
        return spec.ProvisionMallocBDevReply()

    def CheckMallocBDev(self, request, context):
        name = request.bdev_name
        with self._volume_mutex.locked(name):
            with self._client() as client:
                try:
                    bdevs = hipstore.get_bdevs(client, name)
                except hipstore.RpcError as err:
                    if err.is_not_found():
                        bdevs = []
                    else:
                        context.abort(grpc.StatusCode.INTERNAL, str(err))
                if not bdevs:
                    context.abort(grpc.StatusCode.NOT_FOUND,
                                  f"BDev {name} not found")
        return spec.CheckMallocBDevReply()

    def CloneMallocBDev(self, request, context):
        """oim-amd extension (docs/spec.md): device-side bdev clone —
        backs CSI snapshots in remote mode. Idempotent on an existing
        destination of the same geometry."""
        source, dest = request.source, request.dest
        if not source or not dest:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          "source and dest are required")
        with self._volume_mutex.locked(dest):
            with self._client() as client:
                try:
                    src_bdevs = hipstore.get_bdevs(client, source)
                except hipstore.RpcError as err:
                    if err.is_not_found():
                        context.abort(grpc.StatusCode.NOT_FOUND,
                                      f"BDev {source} not found")
                    context.abort(grpc.StatusCode.INTERNAL, str(err))
                try:
                    existing = hipstore.get_bdevs(client, dest)
                except hipstore.RpcError as err:
                    if not err.is_not_found():
                        context.abort(grpc.StatusCode.INTERNAL, str(err))
                    existing = []
                if existing:
                    same = (existing[0].block_size == src_bdevs[0].block_size
                            and existing[0].num_blocks
                            == src_bdevs[0].num_blocks)
                    if not same:
                        context.abort(
                            grpc.StatusCode.ALREADY_EXISTS,
                            f"BDev {dest} exists with different geometry")
                    return spec.CloneMallocBDevReply()
                try:
                    client.invoke("bdev_clone", {"src": source, "name": dest})
                except hipstore.RpcError as err:
                    context.abort(grpc.StatusCode.INTERNAL, str(err))
        return spec.CloneMallocBDevReply()

    def ResizeMallocBDev(self, request, context):
        """oim-amd extension (docs/spec.md): offline volume expansion."""
        name, size = request.bdev_name, request.size
        if not name or size <= 0 or size % 512 != 0:
            context.abort(
                grpc.StatusCode.INVALID_ARGUMENT,
                "bdev_name and a positive 512-multiple size are required")
        with self._volume_mutex.locked(name):
            with self._client() as client:
                try:
                    bdevs = hipstore.get_bdevs(client, name)
                except hipstore.RpcError as err:
                    if err.is_not_found():
                        context.abort(grpc.StatusCode.NOT_FOUND,
                                      f"BDev {name} not found")
                    context.abort(grpc.StatusCode.INTERNAL, str(err))
                if size <= bdevs[0].size_bytes:
                    # grow-only (CSI expansion must never shrink);
                    # idempotent for the current size and below
                    return spec.ResizeMallocBDevReply()
                try:
                    client.invoke("resize_malloc_bdev",
                                  {"name": name, "size": size})
                except hipstore.RpcError as err:
                    if "busy" in str(err):
                        context.abort(grpc.StatusCode.FAILED_PRECONDITION,
                                      str(err))
                    context.abort(grpc.StatusCode.INTERNAL, str(err))
        return spec.ResizeMallocBDevReply()

    def ListMallocBDevs(self, request, context):
        """oim-amd extension (docs/spec.md): enumerate malloc bdevs."""
        reply = spec.ListMallocBDevsReply()
        with self._client() as client:
            for bdev in hipstore.get_bdevs(client):
                if request.prefix and not bdev.name.startswith(request.prefix):
                    continue
                info = reply.bdevs.add()
                info.name = bdev.name
                info.size = bdev.size_bytes
                info.block_size = bdev.block_size
                info.product_name = bdev.product_name
        return reply

    def GetIOStats(self, request, context):
        """oim-amd extension (docs/spec.md): per-bdev I/O counters."""
        reply = spec.GetIOStatsReply()
        with self._client() as client:
            for stat in hipstore.get_bdevs_iostat(client,
                                                  request.bdev_name):
                entry = reply.bdevs.add()
                entry.name = stat.name
                entry.num_read_ops = stat.num_read_ops
                entry.num_write_ops = stat.num_write_ops
                entry.num_unmap_ops = stat.num_unmap_ops
                entry.bytes_read = stat.bytes_read
                entry.bytes_written = stat.bytes_written
        return reply

    # --- self-registration --------------------------------------------------

    def register(self) -> None:
        """One registration pass: dial the registry anew and SetValue
        <id>/address (+ <id>/pci when known) — reference
        controller.go:448-468."""
        if not self.registry_address or not self.controller_address:
            return
        target = grpc_target(self.registry_address)
        if self.tls is not None:
            creds = load_tls_channel_credentials(self.tls)
            options = channel_options_for_peer("component.registry")
            channel = grpc.secure_channel(target, creds, options=options)
        else:
            channel = grpc.insecure_channel(target)
        channel = grpc.intercept_channel(channel, LogClientInterceptor())
        try:
            stub = spec.RegistryStub(channel)
            stub.SetValue(
                spec.SetValueRequest(value=spec.Value(
                    path=f"{self.controller_id}/address",
                    value=self.controller_address)),
                timeout=10,
            )
            if self.pci_address:
                # MI355X extension: the card knows its own GPU BDF, so
                # it registers <id>/pci itself instead of waiting for an
                # admin oimctl call.
                try:
                    stub.SetValue(
                        spec.SetValueRequest(value=spec.Value(
                            path=f"{self.controller_id}/pci",
                            value=self.pci_address)),
                        timeout=10,
                    )
                except grpc.RpcError as err:
                    # Permitted only for admin certs; controllers with a
                    # controller.<id> cert fall back to admin-set pci.
                    if err.code() != grpc.StatusCode.PERMISSION_DENIED:
                        raise
        finally:
            channel.close()

    def start(self) -> None:
        """Start the periodic self-registration loop
        (controller.go:411-443: first tick immediately)."""
        if self._register_thread is not None:
            return

        def loop():
            while not self._stop_event.is_set():
                try:
                    self.register()
                except grpc.RpcError as err:
                    from_context().warn(
                        "registration failed",
                        registry=self.registry_address,
                        error=str(err.code()),
                    )
                if self._stop_event.wait(self.registry_delay):
                    return

        self._stop_event.clear()
        self._register_thread = threading.Thread(target=loop, daemon=True)
        self._register_thread.start()

    def stop(self) -> None:
        self._stop_event.set()
        if self._register_thread is not None:
            self._register_thread.join()
            self._register_thread = None


class ControllerServer:
    """Serves the Controller API (reference controller.go:485-501)."""

    def __init__(self, endpoint: str, controller: Controller):
        self.controller = controller
        self.server = NonBlockingGRPCServer(
            endpoint=endpoint,
            tls=controller.tls,
            interceptors=[LogServerInterceptor()],
        )

    def start(self) -> None:
        self.server.start(
            lambda s: spec.add_controller_to_server(self.controller, s))

    def addr(self) -> str:
        return self.server.addr()

    def stop(self) -> None:
        self.controller.stop()
        self.server.stop()

    def run(self) -> None:
        """start() + block until SIGINT/SIGTERM."""
        self.server.run(
            lambda s: spec.add_controller_to_server(self.controller, s))
