"""CSI driver tests: sysfs discovery, services, mock e2e over the full
control plane (reference nodeserver_test.go, oim-driver_test.go).
"""

import os
import threading
import time

import grpc
import pytest

from oim_amd import spec
from oim_amd.common.pci import parse_bdf_string
from oim_amd.common.server import grpc_target
from oim_amd.controller import Controller, ControllerServer
from oim_amd.csidriver import (
    FakeExec,
    LocalBackend,
    Mounter,
    OIMDriver,
    RemoteBackend,
    ceph_csi_params,
    make_params_mapper,
)
from oim_amd.registry import MemRegistryDB, Registry, RegistryServer
from oim_amd.spec import csi_v1 as csi
from oim_amd.spec.rpc_csi import CSIControllerStub, CSIIdentityStub, CSINodeStub

from fixtures import hipstored  # noqa: F401

VIRTIO_TARGET = ("../../devices/pci0000:00/0000:00:15.0/virtio2/host2/"
                 "target2:0:{t}/2:0:{t}:{l}/block/{name}")


def add_sysfs_device(sysfs, major_minor, target, lun, name):
    os.makedirs(sysfs, exist_ok=True)
    os.symlink(VIRTIO_TARGET.format(t=target, l=lun, name=name),
               os.path.join(sysfs, major_minor))


class TestFindDevice:
    def make_backend(self, tmp_path):
        return RemoteBackend(
            registry_address="unix:///unused.sock",
            controller_id="host-0",
            sysfs_block_dir=str(tmp_path / "block"),
            dev_dir=str(tmp_path / "dev"),
            device_timeout=2.0,
        )

    def test_finds_matching_device(self, tmp_path):
        backend = self.make_backend(tmp_path)
        add_sysfs_device(tmp_path / "block", "8:32", 3, 0, "sdb")
        add_sysfs_device(tmp_path / "block", "8:48", 4, 0, "sdc")
        pci = parse_bdf_string("0000:00:15.0")
        assert backend.find_device(pci, 3, 0) == ("8:32", "sdb")
        assert backend.find_device(pci, 4, 0) == ("8:48", "sdc")
        assert backend.find_device(pci, 5, 0) is None
        # wrong PCI address: no match
        assert backend.find_device(parse_bdf_string("0000:00:16.0"), 3, 0) is None

    def test_wait_sees_delayed_symlink(self, tmp_path):
        """Device appears 0.5s after the wait starts
        (reference nodeserver_test.go delayed-symlink case)."""
        backend = self.make_backend(tmp_path)
        pci = parse_bdf_string("0000:00:15.0")

        def later():
            time.sleep(0.5)
            add_sysfs_device(tmp_path / "block", "8:32", 1, 0, "sdb")

        thread = threading.Thread(target=later)
        thread.start()
        try:
            path = backend.wait_for_device(pci, 1, 0)
        finally:
            thread.join()
        assert path.endswith("/sdb")
        st = os.stat(path)
        assert os.major(st.st_rdev) == 8 and os.minor(st.st_rdev) == 32

    def test_wait_times_out(self, tmp_path):
        backend = self.make_backend(tmp_path)
        backend.device_timeout = 0.4
        with pytest.raises(TimeoutError):
            backend.wait_for_device(parse_bdf_string("0000:00:15.0"), 1, 0)


class TestCephEmulation:
    def test_param_mapping(self):
        request = spec.MapVolumeRequest(volume_id="v")
        ceph_csi_params(
            request, "v",
            {"pool": "rbd", "monitors": "1.2.3.4:6789", "userid": "user1"},
            {"user1": "SECRETKEY"},
            "/var/lib/kubelet/plugins/csi-vol-1234/globalmount",
        )
        assert request.WhichOneof("params") == "ceph"
        assert request.ceph.pool == "rbd"
        assert request.ceph.monitors == "1.2.3.4:6789"
        assert request.ceph.user_id == "user1"
        assert request.ceph.secret == "SECRETKEY"
        assert request.ceph.image == "csi-vol-1234"

    def test_missing_params_rejected(self):
        request = spec.MapVolumeRequest(volume_id="v")
        with pytest.raises(ValueError):
            ceph_csi_params(request, "v", {}, {}, "/x/globalmount")

    def test_mapper_registry(self):
        from oim_amd.csidriver.remote import malloc_params

        assert make_params_mapper("") is malloc_params
        assert make_params_mapper("ceph-csi") is ceph_csi_params
        with pytest.raises(ValueError):
            make_params_mapper("bogus")


class TestMounter:
    def test_format_and_mount_unformatted(self):
        fake = FakeExec()
        mounter = Mounter(fake)
        mounter.format_and_mount("/dev/x", "/mnt/y", "ext4")
        assert fake.calls[0][0] == "blkid"
        assert fake.calls[1][:2] == ["mkfs.ext4", "-F"]
        assert fake.calls[2][0] == "mount"

    def test_format_and_mount_existing_fs(self):
        fake = FakeExec()
        fake.outputs["blkid"] = "TYPE=ext4\n"
        mounter = Mounter(fake)
        mounter.format_and_mount("/dev/x", "/mnt/y", "ext4")
        assert [c[0] for c in fake.calls] == ["blkid", "mount"]

    def test_wrong_fs_rejected(self):
        fake = FakeExec()
        fake.outputs["blkid"] = "TYPE=xfs\n"
        with pytest.raises(RuntimeError):
            Mounter(fake).format_and_mount("/dev/x", "/mnt/y", "ext4")


class TestLocalBackendVolumes:
    def test_volume_lifecycle(self, hipstored):  # noqa: F811
        backend = LocalBackend(hipstored.socket_path)
        volume_id, _ = backend.create_volume("pvc-local", 1 << 20)
        assert volume_id == "pvc-local"
        assert backend.check_volume_exists("pvc-local")
        # idempotent create with same size
        backend.create_volume("pvc-local", 1 << 20)
        # size below minimum rounds up to 1 MiB (local.go:50-57)
        backend.create_volume("tiny", 100)
        # different size on existing volume fails
        with pytest.raises(ValueError):
            backend.create_volume("pvc-local", 2 << 20)
        backend.delete_volume("pvc-local")
        backend.delete_volume("pvc-local")  # idempotent
        assert not backend.check_volume_exists("pvc-local")
        backend.delete_volume("tiny")


@pytest.fixture
def control_plane(hipstored, tmp_path):  # noqa: F811
    """registry + controller(hipstored) + CSI driver in remote mode with
    fake sysfs + FakeExec mounts (reference oim-driver_test.go:148-226)."""
    registry = Registry(db=MemRegistryDB())
    reg_server = RegistryServer(f"unix://{tmp_path}/reg.sock", registry)
    reg_server.start()
    controller = Controller(
        controller_id="host-0",
        hipstored_socket=hipstored.socket_path,
        vm_vhost_device="0000:00:15.0",
    )
    ctrl_server = ControllerServer(f"unix://{tmp_path}/ctrl.sock", controller)
    ctrl_server.start()
    registry.db.store(["host-0", "address"], f"unix://{tmp_path}/ctrl.sock")
    registry.db.store(["host-0", "pci"], "0000:00:15.0")
    fake_exec = FakeExec()
    backend = RemoteBackend(
        registry_address=reg_server.addr(),
        controller_id="host-0",
        sysfs_block_dir=str(tmp_path / "block"),
        dev_dir=str(tmp_path / "dev"),
        device_timeout=5.0,
    )
    driver = OIMDriver(
        driver_name="oim-malloc", node_id="node-1",
        endpoint=f"unix://{tmp_path}/csi.sock",
        backend=backend, mounter=Mounter(fake_exec))
    driver.start()
    yield {
        "tmp": tmp_path,
        "driver": driver,
        "fake_exec": fake_exec,
        "registry": registry,
        "csi_endpoint": f"unix://{tmp_path}/csi.sock",
    }
    driver.stop()
    ctrl_server.stop()
    reg_server.stop()


def csi_channel(env):
    return grpc.insecure_channel(grpc_target(env["csi_endpoint"]))


def single_writer_cap():
    cap = csi.VolumeCapability()
    cap.mount.fs_type = "ext4"
    cap.access_mode.mode = csi.ACCESS_MODE_SINGLE_NODE_WRITER
    return cap


class TestCSIEndToEnd:
    def test_identity(self, control_plane):
        with csi_channel(control_plane) as channel:
            stub = CSIIdentityStub(channel)
            info = stub.GetPluginInfo(csi.GetPluginInfoRequest(), timeout=5)
            assert info.name == "oim-malloc"
            probe = stub.Probe(csi.ProbeRequest(), timeout=5)
            assert probe.ready.value
            caps = stub.GetPluginCapabilities(
                csi.GetPluginCapabilitiesRequest(), timeout=5)
            assert caps.capabilities[0].service.type == \
                csi.PLUGIN_CAPABILITY_CONTROLLER_SERVICE

    def test_create_stage_publish_cycle(self, control_plane):
        env = control_plane
        with csi_channel(env) as channel:
            ctrl = CSIControllerStub(channel)
            node = CSINodeStub(channel)
            # CreateVolume -> ProvisionMallocBDev via proxy
            request = csi.CreateVolumeRequest(name="pvc-e2e")
            request.capacity_range.required_bytes = 1 << 20
            request.volume_capabilities.add().CopyFrom(single_writer_cap())
            reply = ctrl.CreateVolume(request, timeout=10)
            assert reply.volume.volume_id == "pvc-e2e"
            # validate caps
            v = csi.ValidateVolumeCapabilitiesRequest(volume_id="pvc-e2e")
            v.volume_capabilities.add().CopyFrom(single_writer_cap())
            validated = ctrl.ValidateVolumeCapabilities(v, timeout=10)
            assert validated.confirmed.volume_capabilities
            # stage: MapVolume through the proxy; the "kernel" (fake
            # sysfs) shows the device shortly after
            staging = str(env["tmp"] / "staging")

            def hotplug():
                time.sleep(0.3)
                add_sysfs_device(env["tmp"] / "block", "8:32", 0, 0, "sda")

            thread = threading.Thread(target=hotplug)
            thread.start()
            stage = csi.NodeStageVolumeRequest(
                volume_id="pvc-e2e", staging_target_path=staging)
            stage.volume_capability.CopyFrom(single_writer_cap())
            node.NodeStageVolume(stage, timeout=15)
            thread.join()
            # mkfs + mount were invoked on the mknod'ed device
            calls = env["fake_exec"].calls
            assert any(c[0] == "mkfs.ext4" for c in calls)
            mount_call = next(c for c in calls if c[0] == "mount")
            assert mount_call[-2].endswith("/sda")
            # publish (bind mount)
            target = str(env["tmp"] / "published")
            publish = csi.NodePublishVolumeRequest(
                volume_id="pvc-e2e", staging_target_path=staging,
                target_path=target)
            publish.volume_capability.CopyFrom(single_writer_cap())
            node.NodePublishVolume(publish, timeout=10)
            assert any("bind" in c for call in env["fake_exec"].calls[-1:]
                       for c in call)
            # unpublish + unstage + delete
            node.NodeUnpublishVolume(
                csi.NodeUnpublishVolumeRequest(volume_id="pvc-e2e",
                                               target_path=target), timeout=10)
            node.NodeUnstageVolume(
                csi.NodeUnstageVolumeRequest(volume_id="pvc-e2e",
                                             staging_target_path=staging),
                timeout=10)
            ctrl.DeleteVolume(csi.DeleteVolumeRequest(volume_id="pvc-e2e"),
                              timeout=10)
            # volume gone
            with pytest.raises(grpc.RpcError) as excinfo:
                ctrl.ValidateVolumeCapabilities(v, timeout=10)
            assert excinfo.value.code() == grpc.StatusCode.NOT_FOUND

    def test_stage_times_out_without_device(self, control_plane):
        """DeadlineExceeded when the device never appears
        (reference oim-driver_test.go:148-226)."""
        env = control_plane
        env["driver"].node.backend.device_timeout = 0.5
        with csi_channel(env) as channel:
            ctrl = CSIControllerStub(channel)
            node = CSINodeStub(channel)
            request = csi.CreateVolumeRequest(name="pvc-timeout")
            request.capacity_range.required_bytes = 1 << 20
            request.volume_capabilities.add().CopyFrom(single_writer_cap())
            ctrl.CreateVolume(request, timeout=10)
            stage = csi.NodeStageVolumeRequest(
                volume_id="pvc-timeout",
                staging_target_path=str(env["tmp"] / "st2"))
            stage.volume_capability.CopyFrom(single_writer_cap())
            with pytest.raises(grpc.RpcError) as excinfo:
                node.NodeStageVolume(stage, timeout=15)
            assert excinfo.value.code() == grpc.StatusCode.DEADLINE_EXCEEDED

    def test_block_volume_rejected(self, control_plane):
        with csi_channel(control_plane) as channel:
            ctrl = CSIControllerStub(channel)
            request = csi.CreateVolumeRequest(name="pvc-block")
            request.capacity_range.required_bytes = 1 << 20
            cap = request.volume_capabilities.add()
            cap.block.SetInParent()
            cap.access_mode.mode = csi.ACCESS_MODE_SINGLE_NODE_WRITER
            with pytest.raises(grpc.RpcError) as excinfo:
                ctrl.CreateVolume(request, timeout=10)
            assert excinfo.value.code() == grpc.StatusCode.INVALID_ARGUMENT

    def test_unimplemented_methods(self, control_plane):
        with csi_channel(control_plane) as channel:
            method = channel.unary_unary(
                "/csi.v1.Controller/ControllerPublishVolume",
                request_serializer=lambda b: b,
                response_deserializer=lambda b: b)
            with pytest.raises(grpc.RpcError) as excinfo:
                method(b"", timeout=5)
            assert excinfo.value.code() == grpc.StatusCode.UNIMPLEMENTED

    def test_remote_list_volumes(self, control_plane):
        """Remote mode enumerates via the ListMallocBDevs extension."""
        from oim_amd.spec.rpc_csi import CSIControllerStub

        with csi_channel(control_plane) as channel:
            ctrl = CSIControllerStub(channel)
            create = csi.CreateVolumeRequest(name="lv-remote")
            create.capacity_range.required_bytes = 1 << 20
            create.volume_capabilities.add().CopyFrom(single_writer_cap())
            ctrl.CreateVolume(create, timeout=10)
            reply = ctrl.ListVolumes(csi.ListVolumesRequest(), timeout=10)
            assert "lv-remote" in [e.volume.volume_id for e in reply.entries]
            ctrl.DeleteVolume(csi.DeleteVolumeRequest(volume_id="lv-remote"),
                              timeout=10)
