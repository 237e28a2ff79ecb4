"""CSI v0.3 twin personality (reference identityserver0.go /
controllerserver0.go / nodeserver0.go, wired by oim-driver.go:281-284):
the legacy surface pre-1.0 kubelets and the ceph-csi v0.3 emulation
speak. Served by the same servicers through the driver03 adapters;
these tests talk real csi.v0 wire messages over gRPC."""

import threading
import time

import grpc
import pytest

from oim_amd.common.server import grpc_target
from oim_amd.controller import Controller, ControllerServer
from oim_amd.csidriver import (
    FakeExec,
    Mounter,
    OIMDriver,
    RemoteBackend,
    make_params_mapper,
)
from oim_amd.registry import MemRegistryDB, Registry, RegistryServer
from oim_amd.spec import csi_v0 as csi0
from oim_amd.spec.rpc_csi0 import (
    CSI0ControllerStub,
    CSI0IdentityStub,
    CSI0NodeStub,
)

from fixtures import hipstored  # noqa: F401
from test_csidriver import add_sysfs_device  # same fake-sysfs helper


@pytest.fixture
def control_plane03(hipstored, tmp_path):  # noqa: F811
    """Full remote-mode stack serving the 0.3 personality with the
    ceph-csi emulation hook active (the deployment the twins exist
    for: deploy/kubernetes/ceph-csi oim-node.yaml --emulate=ceph-csi
    --csiversion=0.3)."""
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
        params_mapper=make_params_mapper("ceph-csi"),
    )
    driver = OIMDriver(
        driver_name="oim-rbd", node_id="node-1",
        endpoint=f"unix://{tmp_path}/csi.sock",
        backend=backend, mounter=Mounter(fake_exec),
        csi_version="0.3")
    driver.start()
    yield {
        "tmp": tmp_path,
        "fake_exec": fake_exec,
        "registry": registry,
        "hipstored": hipstored,
        "csi_endpoint": f"unix://{tmp_path}/csi.sock",
    }
    driver.stop()
    ctrl_server.stop()
    reg_server.stop()


def channel(env):
    return grpc.insecure_channel(grpc_target(env["csi_endpoint"]))


def writer_cap():
    cap = csi0.VolumeCapability()
    cap.mount.fs_type = "ext4"
    cap.access_mode.mode = csi0.ACCESS_MODE_SINGLE_NODE_WRITER
    return cap


class TestIdentity03:
    def test_plugin_info_probe_caps(self, control_plane03):
        with channel(control_plane03) as ch:
            stub = CSI0IdentityStub(ch)
            info = stub.GetPluginInfo(csi0.GetPluginInfoRequest(), timeout=5)
            assert info.name == "oim-rbd"
            assert info.vendor_version == "0.3.0"
            assert stub.Probe(csi0.ProbeRequest(), timeout=5).ready.value
            caps = stub.GetPluginCapabilities(
                csi0.GetPluginCapabilitiesRequest(), timeout=5)
            assert caps.capabilities[0].service.type == \
                csi0.PLUGIN_CAPABILITY_CONTROLLER_SERVICE


class TestController03:
    def test_create_validate_delete(self, control_plane03):
        with channel(control_plane03) as ch:
            ctrl = CSI0ControllerStub(ch)
            request = csi0.CreateVolumeRequest(name="pvc03")
            request.capacity_range.required_bytes = 1 << 20
            request.volume_capabilities.add().CopyFrom(writer_cap())
            reply = ctrl.CreateVolume(request, timeout=10)
            # v0 Volume: the unique name doubles as the id
            assert reply.volume.id == "pvc03"
            assert reply.volume.capacity_bytes >= 1 << 20
            # idempotent re-create
            assert ctrl.CreateVolume(request, timeout=10).volume.id == "pvc03"
            v = csi0.ValidateVolumeCapabilitiesRequest(volume_id="pvc03")
            v.volume_capabilities.add().CopyFrom(writer_cap())
            validated = ctrl.ValidateVolumeCapabilities(v, timeout=10)
            assert validated.supported
            # multi-writer is not supported -> supported=False (v0
            # boolean semantics, controllerserver0.go:125-130)
            bad = csi0.ValidateVolumeCapabilitiesRequest(volume_id="pvc03")
            cap = bad.volume_capabilities.add()
            cap.mount.fs_type = "ext4"
            cap.access_mode.mode = csi0.ACCESS_MODE_MULTI_NODE_MULTI_WRITER
            assert not ctrl.ValidateVolumeCapabilities(
                bad, timeout=10).supported
            ctrl.DeleteVolume(
                csi0.DeleteVolumeRequest(volume_id="pvc03"), timeout=10)

    def test_unimplemented_rpcs_report_unimplemented(self, control_plane03):
        with channel(control_plane03) as ch:
            stub = ch.unary_unary(
                "/csi.v0.Controller/ListVolumes",
                request_serializer=lambda b: b,
                response_deserializer=lambda b: b)
            with pytest.raises(grpc.RpcError) as excinfo:
                stub(b"", timeout=5)
            assert excinfo.value.code() == grpc.StatusCode.UNIMPLEMENTED

    def test_missing_name_rejected(self, control_plane03):
        with channel(control_plane03) as ch:
            ctrl = CSI0ControllerStub(ch)
            request = csi0.CreateVolumeRequest()
            request.volume_capabilities.add().CopyFrom(writer_cap())
            with pytest.raises(grpc.RpcError) as excinfo:
                ctrl.CreateVolume(request, timeout=10)
            assert excinfo.value.code() == grpc.StatusCode.INVALID_ARGUMENT


class TestNode03:
    def test_node_get_id_and_caps(self, control_plane03):
        with channel(control_plane03) as ch:
            node = CSI0NodeStub(ch)
            # NodeGetId is the v0-only RPC (dropped in CSI 1.0)
            assert node.NodeGetId(csi0.NodeGetIdRequest(),
                                  timeout=5).node_id == "node-1"
            assert node.NodeGetInfo(csi0.NodeGetInfoRequest(),
                                    timeout=5).node_id == "node-1"
            caps = node.NodeGetCapabilities(
                csi0.NodeGetCapabilitiesRequest(), timeout=5)
            assert caps.capabilities[0].rpc.type == \
                csi0.NODE_CAP_STAGE_UNSTAGE_VOLUME

    def test_ceph_csi_stage_flow(self, control_plane03):
        """The twins' raison d'etre: a ceph-csi v0.3 NodeStageVolume —
        volume_attributes + node_stage_secrets repacked into CephParams
        (ceph-csi.go:50-157) — maps the volume and mounts the device."""
        env = control_plane03
        # Round-2 semantics: monitors present => real RADOS wire path,
        # so stand up the loopback cluster in the daemon first.
        from oim_amd import hipstore
        with hipstore.Client(env["hipstored"].socket_path) as client:
            info = client.invoke("rados_cluster_start",
                                 {"arena_mb": 16, "object_mb": 1})
        with channel(env) as ch:
            node = CSI0NodeStub(ch)
            staging = str(env["tmp"] / "staging" / "pvc-ceph" /
                          "globalmount")

            def hotplug():
                time.sleep(0.3)
                add_sysfs_device(env["tmp"] / "block", "8:32", 0, 0, "sda")

            thread = threading.Thread(target=hotplug)
            thread.start()
            stage = csi0.NodeStageVolumeRequest(
                volume_id="pvc-ceph", staging_target_path=staging)
            stage.volume_capability.CopyFrom(writer_cap())
            stage.volume_attributes["pool"] = "rbd"
            stage.volume_attributes["monitors"] = info["mon_host"]
            stage.volume_attributes["adminid"] = "admin"
            stage.node_stage_secrets["admin"] = "sekrit"
            node.NodeStageVolume(stage, timeout=15)
            thread.join()
            calls = env["fake_exec"].calls
            assert any(c[0] == "mkfs.ext4" for c in calls)
            assert any(c[0] == "mount" for c in calls)
            # the controller received CephParams, not MallocParams:
            # the bdev exists as an RBD wire-path disk in the daemon
            with hipstore.Client(env["hipstored"].socket_path) as client:
                bdev = hipstore.get_bdevs(client, "pvc-ceph")[0]
                assert bdev.product_name == "Ceph Rbd Disk"
            node.NodeUnstageVolume(
                csi0.NodeUnstageVolumeRequest(
                    volume_id="pvc-ceph", staging_target_path=staging),
                timeout=15)
