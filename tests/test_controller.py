"""Controller tests against a real (CPU-mode) hipstored daemon.

Counterpart of the reference's pkg/oim-controller/controller_test.go:
self-registration/re-registration (:43-149) and Map/Unmap/Provision
idempotency against the real data daemon (:151-341).
"""

import time

import grpc
import pytest

from oim_amd import hipstore, spec
from oim_amd.common.server import grpc_target
from oim_amd.controller import Controller, ControllerServer
from oim_amd.registry import MemRegistryDB, Registry, RegistryServer

from fixtures import hipstored  # noqa: F401


class FakeContext:
    """Minimal grpc context for direct servicer calls."""

    def __init__(self):
        self.code = None
        self.details = None

    def abort(self, code, details):
        self.code = code
        self.details = details
        raise _Abort(code, details)

    def invocation_metadata(self):
        return ()

    def auth_context(self):
        return {}


class _Abort(Exception):
    def __init__(self, code, details):
        super().__init__(f"{code}: {details}")
        self.code = code
        self.details = details


@pytest.fixture
def controller(hipstored):  # noqa: F811
    return Controller(
        controller_id="host-0",
        hipstored_socket=hipstored.socket_path,
        vm_vhost_device="0000:00:15.0",
    )


def provision(controller, name, size):
    return controller.ProvisionMallocBDev(
        spec.ProvisionMallocBDevRequest(bdev_name=name, size=size),
        FakeContext(),
    )


class TestProvision:
    def test_create_check_delete(self, controller):
        provision(controller, "vol1", 1024 * 1024)
        controller.CheckMallocBDev(
            spec.CheckMallocBDevRequest(bdev_name="vol1"), FakeContext())
        # idempotent re-provision, same size
        provision(controller, "vol1", 1024 * 1024)
        # size mismatch -> ALREADY_EXISTS (controller.go:234-238)
        with pytest.raises(_Abort) as excinfo:
            provision(controller, "vol1", 2048 * 1024)
        assert excinfo.value.code == grpc.StatusCode.ALREADY_EXISTS
        # delete via size 0, idempotent
        provision(controller, "vol1", 0)
        provision(controller, "vol1", 0)
        with pytest.raises(_Abort) as excinfo:
            controller.CheckMallocBDev(
                spec.CheckMallocBDevRequest(bdev_name="vol1"), FakeContext())
        assert excinfo.value.code == grpc.StatusCode.NOT_FOUND

    def test_invalid_size(self, controller):
        with pytest.raises(_Abort) as excinfo:
            provision(controller, "bad", 1000)  # not multiple of 512
        assert excinfo.value.code == grpc.StatusCode.INVALID_ARGUMENT


class TestMapUnmap:
    def test_malloc_map_requires_provisioned(self, controller):
        with pytest.raises(_Abort) as excinfo:
            controller.MapVolume(
                spec.MapVolumeRequest(volume_id="ghost",
                                      malloc=spec.MallocParams()),
                FakeContext(),
            )
        assert excinfo.value.code == grpc.StatusCode.NOT_FOUND

    def test_map_idempotent(self, controller):
        provision(controller, "volm", 1024 * 1024)
        req = spec.MapVolumeRequest(volume_id="volm", malloc=spec.MallocParams())
        reply1 = controller.MapVolume(req, FakeContext())
        reply2 = controller.MapVolume(req, FakeContext())
        assert reply1.scsi_disk.target == reply2.scsi_disk.target
        assert reply1.scsi_disk.lun == 0
        assert reply1.pci_address.bus == 0x00
        assert reply1.pci_address.device == 0x15
        # unmap: Malloc bdev survives (controller.go:203-209)
        controller.UnmapVolume(
            spec.UnmapVolumeRequest(volume_id="volm"), FakeContext())
        controller.CheckMallocBDev(
            spec.CheckMallocBDevRequest(bdev_name="volm"), FakeContext())
        # unmap again: idempotent
        controller.UnmapVolume(
            spec.UnmapVolumeRequest(volume_id="volm"), FakeContext())
        provision(controller, "volm", 0)

    def test_map_two_volumes_distinct_targets(self, controller):
        provision(controller, "va", 1024 * 1024)
        provision(controller, "vb", 1024 * 1024)
        ra = controller.MapVolume(
            spec.MapVolumeRequest(volume_id="va", malloc=spec.MallocParams()),
            FakeContext())
        rb = controller.MapVolume(
            spec.MapVolumeRequest(volume_id="vb", malloc=spec.MallocParams()),
            FakeContext())
        assert ra.scsi_disk.target != rb.scsi_disk.target

    def test_ceph_map_creates_and_unmap_deletes(self, controller, hipstored):  # noqa: F811
        # Monitors present => the daemon speaks the RADOS wire protocol
        # to that endpoint (round 2); stand up the loopback cluster the
        # way a real deployment has a reachable Ceph cluster.
        with hipstore.Client(hipstored.socket_path) as client:
            info = client.invoke("rados_cluster_start",
                                 {"arena_mb": 16, "object_mb": 1})
        req = spec.MapVolumeRequest(
            volume_id="ceph-vol",
            ceph=spec.CephParams(user_id="admin", secret="k",
                                 monitors=info["mon_host"], pool="rbd",
                                 image="img"),
        )
        reply = controller.MapVolume(req, FakeContext())
        assert reply.scsi_disk.lun == 0
        # bdev now exists (as an RBD-emulation disk)
        controller.CheckMallocBDev(
            spec.CheckMallocBDevRequest(bdev_name="ceph-vol"), FakeContext())
        controller.UnmapVolume(
            spec.UnmapVolumeRequest(volume_id="ceph-vol"), FakeContext())
        # non-Malloc bdev is deleted on unmap
        with pytest.raises(_Abort) as excinfo:
            controller.CheckMallocBDev(
                spec.CheckMallocBDevRequest(bdev_name="ceph-vol"),
                FakeContext())
        assert excinfo.value.code == grpc.StatusCode.NOT_FOUND


class TestRegistration:
    def test_self_registration_heals_db(self, hipstored, tmp_path):  # noqa: F811
        registry = Registry(db=MemRegistryDB())
        reg_server = RegistryServer(f"unix://{tmp_path}/reg.sock", registry)
        reg_server.start()
        try:
            controller = Controller(
                controller_id="host-0",
                hipstored_socket=hipstored.socket_path,
                controller_address="tcp://127.0.0.1:9000",
                registry_address=reg_server.addr(),
                registry_delay=0.2,
                pci_address="0000:c1:00.0",
            )
            controller.start()
            try:
                deadline = time.time() + 10
                while time.time() < deadline:
                    if registry.db.lookup(["host-0", "address"]):
                        break
                    time.sleep(0.05)
                assert registry.db.lookup(["host-0", "address"]) == \
                    "tcp://127.0.0.1:9000"
                # MI355X extension: the GPU BDF self-registers too
                # (follows address within the same register() pass)
                deadline = time.time() + 10
                while time.time() < deadline:
                    if registry.db.lookup(["host-0", "pci"]):
                        break
                    time.sleep(0.05)
                assert registry.db.lookup(["host-0", "pci"]) == "0000:c1:00.0"
                # wipe the DB: the loop re-registers (controller_test.go:107-127)
                registry.db.store(["host-0", "address"], "")
                deadline = time.time() + 10
                while time.time() < deadline:
                    if registry.db.lookup(["host-0", "address"]):
                        break
                    time.sleep(0.05)
                assert registry.db.lookup(["host-0", "address"]) is not None
            finally:
                controller.stop()
            # after stop, no more updates
            registry.db.store(["host-0", "address"], "")
            time.sleep(0.5)
            assert registry.db.lookup(["host-0", "address"]) is None
        finally:
            reg_server.stop()


class TestEndToEnd:
    def test_proxied_provision_and_map(self, hipstored, tmp_path):  # noqa: F811
        """Full control path: client -> registry proxy -> controller ->
        hipstored (the reference's signature flow)."""
        registry = Registry(db=MemRegistryDB())
        reg_server = RegistryServer(f"unix://{tmp_path}/reg.sock", registry)
        reg_server.start()
        controller = Controller(
            controller_id="gpu-0",
            hipstored_socket=hipstored.socket_path,
        )
        ctrl_server = ControllerServer(f"unix://{tmp_path}/ctrl.sock", controller)
        ctrl_server.start()
        try:
            registry.db.store(["gpu-0", "address"], f"unix://{tmp_path}/ctrl.sock")
            with grpc.insecure_channel(grpc_target(reg_server.addr())) as channel:
                stub = spec.ControllerStub(channel)
                metadata = ((spec.CONTROLLER_ID_KEY, "gpu-0"),)
                stub.ProvisionMallocBDev(
                    spec.ProvisionMallocBDevRequest(bdev_name="pvc-1",
                                                    size=8 * 1024 * 1024),
                    metadata=metadata, timeout=10)
                reply = stub.MapVolume(
                    spec.MapVolumeRequest(volume_id="pvc-1",
                                          malloc=spec.MallocParams()),
                    metadata=metadata, timeout=10)
                assert reply.scsi_disk.lun == 0
                stub.UnmapVolume(
                    spec.UnmapVolumeRequest(volume_id="pvc-1"),
                    metadata=metadata, timeout=10)
                stub.CheckMallocBDev(
                    spec.CheckMallocBDevRequest(bdev_name="pvc-1"),
                    metadata=metadata, timeout=10)
        finally:
            ctrl_server.stop()
            reg_server.stop()
