"""Concurrency properties: idempotency under concurrent retry
(reference controller_test.go:260-304) and parallel volume churn
through the full control plane."""

import random
import threading

import grpc
import pytest

from oim_amd import spec
from oim_amd.common.server import grpc_target
from oim_amd.controller import Controller, ControllerServer
from oim_amd.registry import MemRegistryDB, Registry, RegistryServer

from fixtures import hipstored  # noqa: F401


@pytest.fixture
def stack(hipstored, tmp_path):  # noqa: F811
    registry = Registry(db=MemRegistryDB())
    reg_server = RegistryServer(f"unix://{tmp_path}/reg.sock", registry)
    reg_server.start()
    controller = Controller(controller_id="c0",
                            hipstored_socket=hipstored.socket_path)
    ctrl_server = ControllerServer(f"unix://{tmp_path}/ctrl.sock", controller)
    ctrl_server.start()
    registry.db.store(["c0", "address"], f"unix://{tmp_path}/ctrl.sock")
    channel = grpc.insecure_channel(grpc_target(reg_server.addr()))
    stub = spec.ControllerStub(channel)
    yield stub
    channel.close()
    ctrl_server.stop()
    reg_server.stop()


METADATA = ((spec.CONTROLLER_ID_KEY, "c0"),)


class TestConcurrentRetry:
    def test_concurrent_provision_same_volume(self, stack):
        """N threads racing the same ProvisionMallocBDev must all
        succeed and leave exactly one bdev."""
        errors = []

        def provision():
            try:
                stack.ProvisionMallocBDev(
                    spec.ProvisionMallocBDevRequest(bdev_name="race-1",
                                                    size=1 << 20),
                    metadata=METADATA, timeout=30)
            except grpc.RpcError as err:
                errors.append(err)

        threads = [threading.Thread(target=provision) for _ in range(8)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        assert not errors
        stack.CheckMallocBDev(
            spec.CheckMallocBDevRequest(bdev_name="race-1"),
            metadata=METADATA, timeout=30)
        stack.ProvisionMallocBDev(
            spec.ProvisionMallocBDevRequest(bdev_name="race-1", size=0),
            metadata=METADATA, timeout=30)

    def test_concurrent_map_same_volume(self, stack):
        """Racing MapVolume calls must converge on ONE SCSI target."""
        stack.ProvisionMallocBDev(
            spec.ProvisionMallocBDevRequest(bdev_name="race-2", size=1 << 20),
            metadata=METADATA, timeout=30)
        targets = []
        errors = []

        def map_volume():
            try:
                reply = stack.MapVolume(
                    spec.MapVolumeRequest(volume_id="race-2",
                                          malloc=spec.MallocParams()),
                    metadata=METADATA, timeout=30)
                targets.append(reply.scsi_disk.target)
            except grpc.RpcError as err:
                errors.append(err)

        threads = [threading.Thread(target=map_volume) for _ in range(8)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        assert not errors
        assert len(set(targets)) == 1
        stack.UnmapVolume(spec.UnmapVolumeRequest(volume_id="race-2"),
                          metadata=METADATA, timeout=30)
        stack.ProvisionMallocBDev(
            spec.ProvisionMallocBDevRequest(bdev_name="race-2", size=0),
            metadata=METADATA, timeout=30)

    def test_parallel_volume_churn(self, stack):
        """Independent volumes provisioned/mapped/unmapped in parallel."""
        errors = []

        def churn(i):
            rng = random.Random(i)
            try:
                for round_ in range(3):
                    name = f"churn-{i}"
                    stack.ProvisionMallocBDev(
                        spec.ProvisionMallocBDevRequest(
                            bdev_name=name, size=(1 + rng.randrange(4)) << 20),
                        metadata=METADATA, timeout=30)
                    reply = stack.MapVolume(
                        spec.MapVolumeRequest(volume_id=name,
                                              malloc=spec.MallocParams()),
                        metadata=METADATA, timeout=30)
                    assert reply.scsi_disk.lun == 0
                    stack.UnmapVolume(
                        spec.UnmapVolumeRequest(volume_id=name),
                        metadata=METADATA, timeout=30)
                    stack.ProvisionMallocBDev(
                        spec.ProvisionMallocBDevRequest(bdev_name=name, size=0),
                        metadata=METADATA, timeout=30)
            except Exception as exc:  # noqa: BLE001
                errors.append(exc)

        # 6 volumes < 8 SCSI targets so churn never exhausts the bus.
        threads = [threading.Thread(target=churn, args=(i,)) for i in range(6)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        assert not errors, errors[:3]


class TestCloneAndResizeRaces:
    def test_concurrent_clone_same_dest(self, stack):
        """N clients race to clone the same source into the same dest:
        exactly one copy happens; all see success or ALREADY_EXISTS
        with matching geometry — never a corrupt half-state."""
        stack.ProvisionMallocBDev(
            spec.ProvisionMallocBDevRequest(bdev_name="clsrc", size=1 << 20),
            metadata=METADATA, timeout=30)
        errors = []

        def clone():
            try:
                stack.CloneMallocBDev(
                    spec.CloneMallocBDevRequest(source="clsrc", dest="cldst"),
                    metadata=METADATA, timeout=30)
            except grpc.RpcError as exc:
                errors.append(exc.code())

        threads = [threading.Thread(target=clone) for _ in range(8)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        # keyed mutex serializes: every call is idempotent success
        assert errors == []
        stack.CheckMallocBDev(
            spec.CheckMallocBDevRequest(bdev_name="cldst"),
            metadata=METADATA, timeout=30)
        for name in ("clsrc", "cldst"):
            stack.ProvisionMallocBDev(
                spec.ProvisionMallocBDevRequest(bdev_name=name, size=0),
                metadata=METADATA, timeout=30)

    def test_concurrent_resize_same_volume(self, stack):
        stack.ProvisionMallocBDev(
            spec.ProvisionMallocBDevRequest(bdev_name="rsz", size=1 << 20),
            metadata=METADATA, timeout=30)
        errors = []

        def grow(size):
            try:
                stack.ResizeMallocBDev(
                    spec.ResizeMallocBDevRequest(bdev_name="rsz", size=size),
                    metadata=METADATA, timeout=30)
            except grpc.RpcError as exc:
                errors.append(exc.code())

        threads = [threading.Thread(target=grow, args=((i + 2) << 20,))
                   for i in range(6)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        assert errors == []  # serialized; each lands a consistent size
        stack.ProvisionMallocBDev(
            spec.ProvisionMallocBDevRequest(bdev_name="rsz", size=0),
            metadata=METADATA, timeout=30)

    def test_clone_missing_source_race(self, stack):
        """Clone of a volume deleted mid-flight fails NOT_FOUND, never
        crashes or creates an orphan dest."""
        with pytest.raises(grpc.RpcError) as excinfo:
            stack.CloneMallocBDev(
                spec.CloneMallocBDevRequest(source="nope", dest="orphan"),
                metadata=METADATA, timeout=30)
        assert excinfo.value.code() == grpc.StatusCode.NOT_FOUND
        with pytest.raises(grpc.RpcError) as excinfo:
            stack.CheckMallocBDev(
                spec.CheckMallocBDevRequest(bdev_name="orphan"),
                metadata=METADATA, timeout=30)
        assert excinfo.value.code() == grpc.StatusCode.NOT_FOUND


class TestConfigSnapshotRaces:
    def test_save_config_during_churn(self, stack, hipstored):  # noqa: F811
        """save_config stays valid while volumes churn concurrently."""
        from oim_amd import hipstore

        stop = threading.Event()
        errors = []

        def churn(i):
            try:
                while not stop.is_set():
                    stack.ProvisionMallocBDev(
                        spec.ProvisionMallocBDevRequest(
                            bdev_name=f"churn-{i}", size=1 << 20),
                        metadata=METADATA, timeout=30)
                    stack.ProvisionMallocBDev(
                        spec.ProvisionMallocBDevRequest(
                            bdev_name=f"churn-{i}", size=0),
                        metadata=METADATA, timeout=30)
            except grpc.RpcError as exc:
                errors.append(exc)

        threads = [threading.Thread(target=churn, args=(i,))
                   for i in range(3)]
        for t in threads:
            t.start()
        try:
            with hipstore.Client(hipstored.socket_path) as client:
                for _ in range(30):
                    config = client.invoke("save_config")
                    assert {s["subsystem"] for s in config["subsystems"]} \
                        == {"rados", "bdev", "vhost", "nvmf", "nbd", "ublk"}
                    for sub in config["subsystems"]:
                        for entry in sub["config"]:
                            assert "method" in entry
        finally:
            stop.set()
            for t in threads:
                t.join()
        assert not errors
