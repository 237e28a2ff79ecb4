"""CSI v1 conformance checks, ported from the kubernetes-csi sanity
suite's behavior matrix (the reference vendored csi-test and ran
`sanity` against oim-csi-driver, reference test/e2e + Gopkg.toml
csi-test entry). Each test states the spec rule it enforces; all run
against the local-mode driver over a real gRPC endpoint, so the wire
status codes — not Python exceptions — are what is asserted.
"""

import grpc
import pytest

from oim_amd.csidriver import LocalBackend, Mounter, OIMDriver
from oim_amd.csidriver.mount import FakeExec
from oim_amd.common.server import grpc_target
from oim_amd.spec import csi_v1 as csi
from oim_amd.spec.rpc_csi import (
    CSIControllerStub,
    CSIIdentityStub,
    CSINodeStub,
)

from fixtures import hipstored  # noqa: F401


@pytest.fixture
def sanity_env(hipstored, tmp_path):  # noqa: F811
    backend = LocalBackend(hipstored.socket_path)
    fake_exec = FakeExec()
    driver = OIMDriver(
        driver_name="sanity.oim-amd.test", node_id="sanity-node",
        endpoint=f"unix://{tmp_path}/csi.sock",
        backend=backend, mounter=Mounter(fake_exec))
    driver.start()
    channel = grpc.insecure_channel(grpc_target(f"unix://{tmp_path}/csi.sock"))
    yield {
        "identity": CSIIdentityStub(channel),
        "controller": CSIControllerStub(channel),
        "node": CSINodeStub(channel),
        "tmp": tmp_path,
        "fake_exec": fake_exec,
    }
    channel.close()
    driver.stop()


def mount_cap(mode=None):
    cap = csi.VolumeCapability()
    cap.mount.fs_type = "ext4"
    cap.access_mode.mode = mode or csi.ACCESS_MODE_SINGLE_NODE_WRITER
    return cap


def expect_code(code):
    return pytest.raises(grpc.RpcError)


def assert_code(excinfo, code):
    assert excinfo.value.code() == code, (
        f"expected {code}, got {excinfo.value.code()}: "
        f"{excinfo.value.details()}")


class TestIdentitySanity:
    def test_plugin_info_name_is_valid(self, sanity_env):
        """sanity: name non-empty, <=63 chars, domain-style."""
        info = sanity_env["identity"].GetPluginInfo(
            csi.GetPluginInfoRequest(), timeout=10)
        assert info.name
        assert len(info.name) <= 63
        assert "." in info.name  # reverse-domain convention
        assert info.vendor_version

    def test_probe_and_capabilities(self, sanity_env):
        probe = sanity_env["identity"].Probe(csi.ProbeRequest(), timeout=10)
        assert probe.ready.value
        caps = sanity_env["identity"].GetPluginCapabilities(
            csi.GetPluginCapabilitiesRequest(), timeout=10)
        assert len(caps.capabilities) >= 1
        kinds = set()
        for cap in caps.capabilities:
            which = cap.WhichOneof("type")
            kinds.add(which)
            if which == "service":
                assert cap.service.type ==                     csi.PLUGIN_CAPABILITY_CONTROLLER_SERVICE
            elif which == "volume_expansion":
                assert cap.volume_expansion.type in (
                    csi.EXPANSION_ONLINE, csi.EXPANSION_OFFLINE)
        assert "service" in kinds


class TestControllerSanity:
    def test_capabilities_reported(self, sanity_env):
        caps = sanity_env["controller"].ControllerGetCapabilities(
            csi.ControllerGetCapabilitiesRequest(), timeout=10)
        types = {cap.rpc.type for cap in caps.capabilities}
        assert csi.CTRL_CAP_CREATE_DELETE_VOLUME in types
        # Local mode can report capacity, and says so.
        assert csi.CTRL_CAP_GET_CAPACITY in types

    def test_create_volume_missing_name(self, sanity_env):
        request = csi.CreateVolumeRequest()
        request.volume_capabilities.add().CopyFrom(mount_cap())
        with pytest.raises(grpc.RpcError) as excinfo:
            sanity_env["controller"].CreateVolume(request, timeout=10)
        assert_code(excinfo, grpc.StatusCode.INVALID_ARGUMENT)

    def test_create_volume_missing_capabilities(self, sanity_env):
        with pytest.raises(grpc.RpcError) as excinfo:
            sanity_env["controller"].CreateVolume(
                csi.CreateVolumeRequest(name="sanity-nocaps"), timeout=10)
        assert_code(excinfo, grpc.StatusCode.INVALID_ARGUMENT)

    def test_create_delete_roundtrip_and_idempotency(self, sanity_env):
        ctrl = sanity_env["controller"]
        request = csi.CreateVolumeRequest(name="sanity-vol-1")
        request.capacity_range.required_bytes = 1 << 20
        request.volume_capabilities.add().CopyFrom(mount_cap())
        created = ctrl.CreateVolume(request, timeout=10)
        assert created.volume.volume_id
        assert created.volume.capacity_bytes >= 1 << 20
        # Same name + same size: idempotent success, same id.
        again = ctrl.CreateVolume(request, timeout=10)
        assert again.volume.volume_id == created.volume.volume_id
        # Same name + different size: ALREADY_EXISTS (CSI spec).
        request.capacity_range.required_bytes = 2 << 20
        with pytest.raises(grpc.RpcError) as excinfo:
            ctrl.CreateVolume(request, timeout=10)
        assert_code(excinfo, grpc.StatusCode.ALREADY_EXISTS)
        # Delete; repeat delete is idempotent success.
        ctrl.DeleteVolume(csi.DeleteVolumeRequest(
            volume_id=created.volume.volume_id), timeout=10)
        ctrl.DeleteVolume(csi.DeleteVolumeRequest(
            volume_id=created.volume.volume_id), timeout=10)

    def test_delete_volume_missing_id(self, sanity_env):
        with pytest.raises(grpc.RpcError) as excinfo:
            sanity_env["controller"].DeleteVolume(
                csi.DeleteVolumeRequest(), timeout=10)
        assert_code(excinfo, grpc.StatusCode.INVALID_ARGUMENT)

    def test_delete_unknown_volume_is_ok(self, sanity_env):
        sanity_env["controller"].DeleteVolume(
            csi.DeleteVolumeRequest(volume_id="never-existed"), timeout=10)

    def test_validate_missing_fields(self, sanity_env):
        ctrl = sanity_env["controller"]
        with pytest.raises(grpc.RpcError) as excinfo:
            ctrl.ValidateVolumeCapabilities(
                csi.ValidateVolumeCapabilitiesRequest(), timeout=10)
        assert_code(excinfo, grpc.StatusCode.INVALID_ARGUMENT)
        request = csi.ValidateVolumeCapabilitiesRequest(volume_id="x")
        with pytest.raises(grpc.RpcError) as excinfo:
            ctrl.ValidateVolumeCapabilities(request, timeout=10)
        assert_code(excinfo, grpc.StatusCode.INVALID_ARGUMENT)

    def test_validate_unknown_volume_not_found(self, sanity_env):
        request = csi.ValidateVolumeCapabilitiesRequest(
            volume_id="no-such-volume")
        request.volume_capabilities.add().CopyFrom(mount_cap())
        with pytest.raises(grpc.RpcError) as excinfo:
            sanity_env["controller"].ValidateVolumeCapabilities(
                request, timeout=10)
        assert_code(excinfo, grpc.StatusCode.NOT_FOUND)

    def test_validate_confirms_supported_caps(self, sanity_env):
        ctrl = sanity_env["controller"]
        create = csi.CreateVolumeRequest(name="sanity-validate")
        create.capacity_range.required_bytes = 1 << 20
        create.volume_capabilities.add().CopyFrom(mount_cap())
        ctrl.CreateVolume(create, timeout=10)
        try:
            request = csi.ValidateVolumeCapabilitiesRequest(
                volume_id="sanity-validate")
            request.volume_capabilities.add().CopyFrom(mount_cap())
            reply = ctrl.ValidateVolumeCapabilities(request, timeout=10)
            assert len(reply.confirmed.volume_capabilities) == 1
            # Unsupported (multi-writer) is refused with a message, not
            # an error (CSI: confirmed empty + message).
            request = csi.ValidateVolumeCapabilitiesRequest(
                volume_id="sanity-validate")
            request.volume_capabilities.add().CopyFrom(
                mount_cap(csi.ACCESS_MODE_MULTI_NODE_MULTI_WRITER))
            reply = ctrl.ValidateVolumeCapabilities(request, timeout=10)
            assert len(reply.confirmed.volume_capabilities) == 0
            assert reply.message
        finally:
            ctrl.DeleteVolume(csi.DeleteVolumeRequest(
                volume_id="sanity-validate"), timeout=10)

    def test_get_capacity(self, sanity_env):
        reply = sanity_env["controller"].GetCapacity(
            csi.GetCapacityRequest(), timeout=10)
        assert reply.available_capacity > 0

    def test_block_capability_rejected(self, sanity_env):
        request = csi.CreateVolumeRequest(name="sanity-block")
        cap = request.volume_capabilities.add()
        cap.block.SetInParent()
        cap.access_mode.mode = csi.ACCESS_MODE_SINGLE_NODE_WRITER
        with pytest.raises(grpc.RpcError) as excinfo:
            sanity_env["controller"].CreateVolume(request, timeout=10)
        assert_code(excinfo, grpc.StatusCode.INVALID_ARGUMENT)


class TestNodeSanity:
    def test_node_info(self, sanity_env):
        info = sanity_env["node"].NodeGetInfo(
            csi.NodeGetInfoRequest(), timeout=10)
        assert info.node_id == "sanity-node"

    def test_node_capabilities(self, sanity_env):
        caps = sanity_env["node"].NodeGetCapabilities(
            csi.NodeGetCapabilitiesRequest(), timeout=10)
        types = {cap.rpc.type for cap in caps.capabilities}
        assert csi.NODE_CAP_STAGE_UNSTAGE_VOLUME in types

    def test_stage_missing_fields(self, sanity_env):
        node = sanity_env["node"]
        cases = [
            csi.NodeStageVolumeRequest(),  # everything missing
            csi.NodeStageVolumeRequest(volume_id="v"),  # no staging path
        ]
        stage = csi.NodeStageVolumeRequest(
            volume_id="v", staging_target_path="/tmp/stage")
        cases.append(stage)  # no volume capability
        for request in cases:
            with pytest.raises(grpc.RpcError) as excinfo:
                node.NodeStageVolume(request, timeout=10)
            assert_code(excinfo, grpc.StatusCode.INVALID_ARGUMENT)

    def test_unstage_missing_fields(self, sanity_env):
        with pytest.raises(grpc.RpcError) as excinfo:
            sanity_env["node"].NodeUnstageVolume(
                csi.NodeUnstageVolumeRequest(volume_id="v"), timeout=10)
        assert_code(excinfo, grpc.StatusCode.INVALID_ARGUMENT)

    def test_publish_missing_fields(self, sanity_env):
        node = sanity_env["node"]
        request = csi.NodePublishVolumeRequest(
            volume_id="v", staging_target_path="/a")
        with pytest.raises(grpc.RpcError) as excinfo:
            node.NodePublishVolume(request, timeout=10)  # no target path
        assert_code(excinfo, grpc.StatusCode.INVALID_ARGUMENT)
        request = csi.NodePublishVolumeRequest(
            volume_id="v", staging_target_path="/a", target_path="/b")
        with pytest.raises(grpc.RpcError) as excinfo:
            node.NodePublishVolume(request, timeout=10)  # no capability
        assert_code(excinfo, grpc.StatusCode.INVALID_ARGUMENT)

    def test_unpublish_missing_and_idempotent(self, sanity_env):
        node = sanity_env["node"]
        with pytest.raises(grpc.RpcError) as excinfo:
            node.NodeUnpublishVolume(
                csi.NodeUnpublishVolumeRequest(volume_id="v"), timeout=10)
        assert_code(excinfo, grpc.StatusCode.INVALID_ARGUMENT)
        # Unpublishing a never-published path is idempotent success.
        node.NodeUnpublishVolume(
            csi.NodeUnpublishVolumeRequest(
                volume_id="v",
                target_path=str(sanity_env["tmp"] / "not-mounted")),
            timeout=10)

    def test_volume_stats_errors(self, sanity_env):
        node = sanity_env["node"]
        with pytest.raises(grpc.RpcError) as excinfo:
            node.NodeGetVolumeStats(
                csi.NodeGetVolumeStatsRequest(volume_id="v"), timeout=10)
        assert_code(excinfo, grpc.StatusCode.INVALID_ARGUMENT)
        with pytest.raises(grpc.RpcError) as excinfo:
            node.NodeGetVolumeStats(
                csi.NodeGetVolumeStatsRequest(
                    volume_id="v", volume_path="/no/such/path"), timeout=10)
        assert_code(excinfo, grpc.StatusCode.NOT_FOUND)

    def test_volume_stats_real_path(self, sanity_env):
        reply = sanity_env["node"].NodeGetVolumeStats(
            csi.NodeGetVolumeStatsRequest(
                volume_id="v", volume_path=str(sanity_env["tmp"])),
            timeout=10)
        units = {usage.unit for usage in reply.usage}
        assert csi.USAGE_UNIT_BYTES in units
        byte_usage = [u for u in reply.usage
                      if u.unit == csi.USAGE_UNIT_BYTES][0]
        assert byte_usage.total > 0


class TestSnapshotSanity:
    """CSI snapshot RPCs (backed by hipstored bdev_clone; the sanity
    suite's snapshot behavior matrix)."""

    def _create_volume(self, env, name, size=1 << 20):
        request = csi.CreateVolumeRequest(name=name)
        request.capacity_range.required_bytes = size
        request.volume_capabilities.add().CopyFrom(mount_cap())
        return env["controller"].CreateVolume(request, timeout=10)

    def test_capabilities_include_snapshots(self, sanity_env):
        caps = sanity_env["controller"].ControllerGetCapabilities(
            csi.ControllerGetCapabilitiesRequest(), timeout=10)
        types = {cap.rpc.type for cap in caps.capabilities}
        assert csi.CTRL_CAP_CREATE_DELETE_SNAPSHOT in types
        assert csi.CTRL_CAP_LIST_SNAPSHOTS in types

    def test_create_snapshot_missing_fields(self, sanity_env):
        ctrl = sanity_env["controller"]
        with pytest.raises(grpc.RpcError) as excinfo:
            ctrl.CreateSnapshot(
                csi.CreateSnapshotRequest(name="s"), timeout=10)
        assert_code(excinfo, grpc.StatusCode.INVALID_ARGUMENT)
        with pytest.raises(grpc.RpcError) as excinfo:
            ctrl.CreateSnapshot(
                csi.CreateSnapshotRequest(source_volume_id="v"), timeout=10)
        assert_code(excinfo, grpc.StatusCode.INVALID_ARGUMENT)

    def test_snapshot_of_unknown_volume(self, sanity_env):
        with pytest.raises(grpc.RpcError) as excinfo:
            sanity_env["controller"].CreateSnapshot(
                csi.CreateSnapshotRequest(source_volume_id="ghost",
                                          name="snap-x"), timeout=10)
        assert_code(excinfo, grpc.StatusCode.NOT_FOUND)

    def test_snapshot_lifecycle(self, sanity_env):
        ctrl = sanity_env["controller"]
        self._create_volume(sanity_env, "snap-src")
        try:
            created = ctrl.CreateSnapshot(
                csi.CreateSnapshotRequest(source_volume_id="snap-src",
                                          name="nightly"), timeout=10)
            snap = created.snapshot
            assert snap.snapshot_id
            assert snap.source_volume_id == "snap-src"
            assert snap.size_bytes == 1 << 20
            assert snap.ready_to_use
            assert snap.creation_time.seconds > 0
            # idempotent: same name + same source -> same snapshot
            again = ctrl.CreateSnapshot(
                csi.CreateSnapshotRequest(source_volume_id="snap-src",
                                          name="nightly"), timeout=10)
            assert again.snapshot.snapshot_id == snap.snapshot_id
            # conflict: same name, different source -> ALREADY_EXISTS
            self._create_volume(sanity_env, "snap-src-2")
            with pytest.raises(grpc.RpcError) as excinfo:
                ctrl.CreateSnapshot(
                    csi.CreateSnapshotRequest(source_volume_id="snap-src-2",
                                              name="nightly"), timeout=10)
            assert_code(excinfo, grpc.StatusCode.ALREADY_EXISTS)
            # list: filter by source
            listed = ctrl.ListSnapshots(
                csi.ListSnapshotsRequest(source_volume_id="snap-src"),
                timeout=10)
            assert [e.snapshot.snapshot_id for e in listed.entries] == \
                [snap.snapshot_id]
            # delete; repeat delete is idempotent
            ctrl.DeleteSnapshot(csi.DeleteSnapshotRequest(
                snapshot_id=snap.snapshot_id), timeout=10)
            ctrl.DeleteSnapshot(csi.DeleteSnapshotRequest(
                snapshot_id=snap.snapshot_id), timeout=10)
            listed = ctrl.ListSnapshots(csi.ListSnapshotsRequest(), timeout=10)
            assert snap.snapshot_id not in [
                e.snapshot.snapshot_id for e in listed.entries]
        finally:
            ctrl.DeleteVolume(csi.DeleteVolumeRequest(volume_id="snap-src"),
                              timeout=10)
            ctrl.DeleteVolume(csi.DeleteVolumeRequest(volume_id="snap-src-2"),
                              timeout=10)

    def test_restore_volume_from_snapshot(self, sanity_env):
        ctrl = sanity_env["controller"]
        self._create_volume(sanity_env, "restore-src")
        try:
            snap = ctrl.CreateSnapshot(
                csi.CreateSnapshotRequest(source_volume_id="restore-src",
                                          name="base"), timeout=10).snapshot
            request = csi.CreateVolumeRequest(name="restored-vol")
            request.capacity_range.required_bytes = 1 << 20
            request.volume_capabilities.add().CopyFrom(mount_cap())
            request.volume_content_source.snapshot.snapshot_id = \
                snap.snapshot_id
            restored = ctrl.CreateVolume(request, timeout=10)
            assert restored.volume.volume_id == "restored-vol"
            assert restored.volume.capacity_bytes == 1 << 20
            assert (restored.volume.content_source.snapshot.snapshot_id
                    == snap.snapshot_id)
            # restore asking for more than the snapshot -> OUT_OF_RANGE
            request2 = csi.CreateVolumeRequest(name="restored-too-big")
            request2.capacity_range.required_bytes = 4 << 20
            request2.volume_capabilities.add().CopyFrom(mount_cap())
            request2.volume_content_source.snapshot.snapshot_id = \
                snap.snapshot_id
            with pytest.raises(grpc.RpcError) as excinfo:
                ctrl.CreateVolume(request2, timeout=10)
            assert excinfo.value.code() in (grpc.StatusCode.OUT_OF_RANGE,)
            # restore from unknown snapshot -> NOT_FOUND
            request3 = csi.CreateVolumeRequest(name="restored-ghost")
            request3.volume_capabilities.add().CopyFrom(mount_cap())
            request3.volume_content_source.snapshot.snapshot_id = "ghost"
            with pytest.raises(grpc.RpcError) as excinfo:
                ctrl.CreateVolume(request3, timeout=10)
            assert_code(excinfo, grpc.StatusCode.NOT_FOUND)
            ctrl.DeleteVolume(csi.DeleteVolumeRequest(
                volume_id="restored-vol"), timeout=10)
            ctrl.DeleteSnapshot(csi.DeleteSnapshotRequest(
                snapshot_id=snap.snapshot_id), timeout=10)
        finally:
            ctrl.DeleteVolume(csi.DeleteVolumeRequest(
                volume_id="restore-src"), timeout=10)


class TestRemoteSnapshots:
    """Snapshots in remote mode ride the CloneMallocBDev oim.v0
    extension through the registry proxy (docs/spec.md)."""

    def test_snapshot_via_proxy(self, hipstored, tmp_path):  # noqa: F811
        from oim_amd.controller import Controller, ControllerServer
        from oim_amd.csidriver import RemoteBackend
        from oim_amd.registry import MemRegistryDB, Registry, RegistryServer

        registry = Registry(db=MemRegistryDB())
        reg_server = RegistryServer(f"unix://{tmp_path}/reg.sock", registry)
        reg_server.start()
        controller = Controller(controller_id="snap-host",
                                hipstored_socket=hipstored.socket_path)
        ctrl_server = ControllerServer(f"unix://{tmp_path}/ctrl.sock",
                                       controller)
        ctrl_server.start()
        registry.db.store(["snap-host", "address"],
                          f"unix://{tmp_path}/ctrl.sock")
        backend = RemoteBackend(registry_address=reg_server.addr(),
                                controller_id="snap-host")
        driver = OIMDriver(driver_name="remote.oim-amd.test", node_id="n0",
                           endpoint=f"unix://{tmp_path}/csi.sock",
                           backend=backend, mounter=Mounter(FakeExec()))
        driver.start()
        channel = grpc.insecure_channel(
            grpc_target(f"unix://{tmp_path}/csi.sock"))
        ctrl = CSIControllerStub(channel)
        try:
            caps = ctrl.ControllerGetCapabilities(
                csi.ControllerGetCapabilitiesRequest(), timeout=10)
            types = {cap.rpc.type for cap in caps.capabilities}
            assert csi.CTRL_CAP_CREATE_DELETE_SNAPSHOT in types

            create = csi.CreateVolumeRequest(name="rvol")
            create.capacity_range.required_bytes = 1 << 20
            create.volume_capabilities.add().CopyFrom(mount_cap())
            ctrl.CreateVolume(create, timeout=10)
            snap = ctrl.CreateSnapshot(
                csi.CreateSnapshotRequest(source_volume_id="rvol",
                                          name="r1"), timeout=10).snapshot
            assert snap.snapshot_id == "csi-snap-r1"
            assert snap.size_bytes == 1 << 20
            # the clone exists on the daemon
            from oim_amd import hipstore
            with hipstore.Client(hipstored.socket_path) as client:
                assert hipstore.get_bdevs(client, "csi-snap-r1")
            # snapshot of unknown volume -> NOT_FOUND through the proxy
            with pytest.raises(grpc.RpcError) as excinfo:
                ctrl.CreateSnapshot(
                    csi.CreateSnapshotRequest(source_volume_id="ghost",
                                              name="r2"), timeout=10)
            assert_code(excinfo, grpc.StatusCode.NOT_FOUND)
            # restore + cleanup
            request = csi.CreateVolumeRequest(name="rvol-restored")
            request.volume_capabilities.add().CopyFrom(mount_cap())
            request.volume_content_source.snapshot.snapshot_id = \
                snap.snapshot_id
            restored = ctrl.CreateVolume(request, timeout=10)
            assert restored.volume.volume_id == "rvol-restored"
            ctrl.DeleteSnapshot(csi.DeleteSnapshotRequest(
                snapshot_id=snap.snapshot_id), timeout=10)
            ctrl.DeleteVolume(csi.DeleteVolumeRequest(
                volume_id="rvol-restored"), timeout=10)
            ctrl.DeleteVolume(csi.DeleteVolumeRequest(volume_id="rvol"),
                              timeout=10)
        finally:
            channel.close()
            driver.stop()
            ctrl_server.stop()
            reg_server.stop()


class TestListVolumes:
    """ListVolumes with CSI paging semantics (sanity: token paging,
    ABORTED on a bad token)."""

    def test_paged_listing(self, sanity_env):
        ctrl = sanity_env["controller"]
        names = [f"lv-{i}" for i in range(5)]
        for name in names:
            request = csi.CreateVolumeRequest(name=name)
            request.capacity_range.required_bytes = 1 << 20
            request.volume_capabilities.add().CopyFrom(mount_cap())
            ctrl.CreateVolume(request, timeout=10)
        try:
            caps = ctrl.ControllerGetCapabilities(
                csi.ControllerGetCapabilitiesRequest(), timeout=10)
            assert csi.CTRL_CAP_LIST_VOLUMES in {
                cap.rpc.type for cap in caps.capabilities}
            seen = []
            token = ""
            while True:
                reply = ctrl.ListVolumes(
                    csi.ListVolumesRequest(max_entries=2,
                                           starting_token=token), timeout=10)
                seen.extend(e.volume.volume_id for e in reply.entries)
                if not reply.next_token:
                    break
                token = reply.next_token
            assert set(names) <= set(seen)
            assert len(seen) == len(set(seen))  # no duplicates across pages
            with pytest.raises(grpc.RpcError) as excinfo:
                ctrl.ListVolumes(
                    csi.ListVolumesRequest(starting_token="bogus"),
                    timeout=10)
            assert_code(excinfo, grpc.StatusCode.ABORTED)
        finally:
            for name in names:
                ctrl.DeleteVolume(csi.DeleteVolumeRequest(volume_id=name),
                                  timeout=10)


class TestVolumeExpansion:
    """ControllerExpandVolume / NodeExpandVolume (offline expansion:
    the backing store moves, so the daemon refuses while channels are
    open)."""

    def test_plugin_advertises_offline_expansion(self, sanity_env):
        caps = sanity_env["identity"].GetPluginCapabilities(
            csi.GetPluginCapabilitiesRequest(), timeout=10)
        expansions = [cap.volume_expansion.type for cap in caps.capabilities
                      if cap.WhichOneof("type") == "volume_expansion"]
        assert expansions == [csi.EXPANSION_OFFLINE]
        ctrl_caps = sanity_env["controller"].ControllerGetCapabilities(
            csi.ControllerGetCapabilitiesRequest(), timeout=10)
        assert csi.CTRL_CAP_EXPAND_VOLUME in {
            cap.rpc.type for cap in ctrl_caps.capabilities}

    def test_expand_missing_fields(self, sanity_env):
        ctrl = sanity_env["controller"]
        with pytest.raises(grpc.RpcError) as excinfo:
            ctrl.ControllerExpandVolume(
                csi.ControllerExpandVolumeRequest(), timeout=10)
        assert_code(excinfo, grpc.StatusCode.INVALID_ARGUMENT)
        with pytest.raises(grpc.RpcError) as excinfo:
            ctrl.ControllerExpandVolume(
                csi.ControllerExpandVolumeRequest(volume_id="v"), timeout=10)
        assert_code(excinfo, grpc.StatusCode.INVALID_ARGUMENT)

    def test_expand_unknown_volume(self, sanity_env):
        request = csi.ControllerExpandVolumeRequest(volume_id="ghost")
        request.capacity_range.required_bytes = 2 << 20
        with pytest.raises(grpc.RpcError) as excinfo:
            sanity_env["controller"].ControllerExpandVolume(request,
                                                            timeout=10)
        assert_code(excinfo, grpc.StatusCode.NOT_FOUND)

    def test_expand_grows_volume_and_data_survives(self, sanity_env,
                                                   hipstored):  # noqa: F811
        from oim_amd import hipstore

        ctrl = sanity_env["controller"]
        create = csi.CreateVolumeRequest(name="grow-me")
        create.capacity_range.required_bytes = 1 << 20
        create.volume_capabilities.add().CopyFrom(mount_cap())
        ctrl.CreateVolume(create, timeout=10)
        try:
            # write a pattern through the daemon, then expand
            with hipstore.Client(hipstored.socket_path) as client:
                import oim_amd._hipstore  # noqa: F401  (bdev access below)
                pre = client.invoke("get_bdevs", {"name": "grow-me"})
                assert pre[0]["num_blocks"] * pre[0]["block_size"] == 1 << 20
            request = csi.ControllerExpandVolumeRequest(volume_id="grow-me")
            request.capacity_range.required_bytes = 4 << 20
            reply = ctrl.ControllerExpandVolume(request, timeout=10)
            assert reply.capacity_bytes == 4 << 20
            assert reply.node_expansion_required
            with hipstore.Client(hipstored.socket_path) as client:
                post = client.invoke("get_bdevs", {"name": "grow-me"})
                assert post[0]["num_blocks"] * post[0]["block_size"] \
                    == 4 << 20
            # idempotent: same size again succeeds
            ctrl.ControllerExpandVolume(request, timeout=10)
        finally:
            ctrl.DeleteVolume(csi.DeleteVolumeRequest(volume_id="grow-me"),
                              timeout=10)

    def test_node_expand_runs_resize2fs(self, sanity_env):
        node = sanity_env["node"]
        with pytest.raises(grpc.RpcError) as excinfo:
            node.NodeExpandVolume(csi.NodeExpandVolumeRequest(volume_id="v"),
                                  timeout=10)
        assert_code(excinfo, grpc.StatusCode.INVALID_ARGUMENT)
        request = csi.NodeExpandVolumeRequest(
            volume_id="v", volume_path=str(sanity_env["tmp"]))
        request.capacity_range.required_bytes = 2 << 20
        reply = node.NodeExpandVolume(request, timeout=10)
        assert reply.capacity_bytes == 2 << 20
        # FakeExec recorded the filesystem grow
        calls = sanity_env["fake_exec"].calls
        assert any(call[0] == "resize2fs" for call in calls)


class TestRemoteExpansion:
    def test_expand_via_proxy(self, hipstored, tmp_path):  # noqa: F811
        from oim_amd.controller import Controller, ControllerServer
        from oim_amd.csidriver import RemoteBackend
        from oim_amd.registry import MemRegistryDB, Registry, RegistryServer

        registry = Registry(db=MemRegistryDB())
        reg_server = RegistryServer(f"unix://{tmp_path}/reg.sock", registry)
        reg_server.start()
        controller = Controller(controller_id="exp-host",
                                hipstored_socket=hipstored.socket_path)
        ctrl_server = ControllerServer(f"unix://{tmp_path}/ctrl.sock",
                                       controller)
        ctrl_server.start()
        registry.db.store(["exp-host", "address"],
                          f"unix://{tmp_path}/ctrl.sock")
        backend = RemoteBackend(registry_address=reg_server.addr(),
                                controller_id="exp-host")
        driver = OIMDriver(driver_name="remote.oim-amd.test", node_id="n0",
                           endpoint=f"unix://{tmp_path}/csi.sock",
                           backend=backend, mounter=Mounter(FakeExec()))
        driver.start()
        channel = grpc.insecure_channel(
            grpc_target(f"unix://{tmp_path}/csi.sock"))
        ctrl = CSIControllerStub(channel)
        try:
            create = csi.CreateVolumeRequest(name="evol")
            create.capacity_range.required_bytes = 1 << 20
            create.volume_capabilities.add().CopyFrom(mount_cap())
            ctrl.CreateVolume(create, timeout=10)
            request = csi.ControllerExpandVolumeRequest(volume_id="evol")
            request.capacity_range.required_bytes = 2 << 20
            reply = ctrl.ControllerExpandVolume(request, timeout=10)
            assert reply.capacity_bytes == 2 << 20
            from oim_amd import hipstore
            with hipstore.Client(hipstored.socket_path) as client:
                bdev = hipstore.get_bdevs(client, "evol")[0]
                assert bdev.size_bytes == 2 << 20
            request.volume_id = "ghost"
            with pytest.raises(grpc.RpcError) as excinfo:
                ctrl.ControllerExpandVolume(request, timeout=10)
            assert_code(excinfo, grpc.StatusCode.NOT_FOUND)
            ctrl.DeleteVolume(csi.DeleteVolumeRequest(volume_id="evol"),
                              timeout=10)
        finally:
            channel.close()
            driver.stop()
            ctrl_server.stop()
            reg_server.stop()


class TestExpansionNeverShrinks:
    def test_shrink_request_is_noop(self, sanity_env, hipstored):  # noqa: F811
        from oim_amd import hipstore

        ctrl = sanity_env["controller"]
        create = csi.CreateVolumeRequest(name="no-shrink")
        create.capacity_range.required_bytes = 4 << 20
        create.volume_capabilities.add().CopyFrom(mount_cap())
        ctrl.CreateVolume(create, timeout=10)
        try:
            request = csi.ControllerExpandVolumeRequest(volume_id="no-shrink")
            request.capacity_range.required_bytes = 1 << 20  # smaller
            reply = ctrl.ControllerExpandVolume(request, timeout=10)
            assert reply.capacity_bytes == 4 << 20  # unchanged
            with hipstore.Client(hipstored.socket_path) as client:
                assert hipstore.get_bdevs(
                    client, "no-shrink")[0].size_bytes == 4 << 20
        finally:
            ctrl.DeleteVolume(csi.DeleteVolumeRequest(volume_id="no-shrink"),
                              timeout=10)


class TestVolumeCloning:
    """CreateVolume from a volume content source (CLONE_VOLUME)."""

    def test_clone_volume(self, sanity_env):
        ctrl = sanity_env["controller"]
        caps = ctrl.ControllerGetCapabilities(
            csi.ControllerGetCapabilitiesRequest(), timeout=10)
        assert csi.CTRL_CAP_CLONE_VOLUME in {
            cap.rpc.type for cap in caps.capabilities}
        create = csi.CreateVolumeRequest(name="clone-src")
        create.capacity_range.required_bytes = 1 << 20
        create.volume_capabilities.add().CopyFrom(mount_cap())
        ctrl.CreateVolume(create, timeout=10)
        try:
            request = csi.CreateVolumeRequest(name="clone-dst")
            request.volume_capabilities.add().CopyFrom(mount_cap())
            request.volume_content_source.volume.volume_id = "clone-src"
            reply = ctrl.CreateVolume(request, timeout=10)
            assert reply.volume.volume_id == "clone-dst"
            assert reply.volume.capacity_bytes == 1 << 20
            assert (reply.volume.content_source.volume.volume_id
                    == "clone-src")
            # idempotent retry
            again = ctrl.CreateVolume(request, timeout=10)
            assert again.volume.volume_id == "clone-dst"
            # unknown source
            request2 = csi.CreateVolumeRequest(name="clone-ghost")
            request2.volume_capabilities.add().CopyFrom(mount_cap())
            request2.volume_content_source.volume.volume_id = "nope"
            with pytest.raises(grpc.RpcError) as excinfo:
                ctrl.CreateVolume(request2, timeout=10)
            assert_code(excinfo, grpc.StatusCode.NOT_FOUND)
            ctrl.DeleteVolume(csi.DeleteVolumeRequest(volume_id="clone-dst"),
                              timeout=10)
        finally:
            ctrl.DeleteVolume(csi.DeleteVolumeRequest(volume_id="clone-src"),
                              timeout=10)


class TestAioBackedVolumes:
    """StorageClass `backing: aio`: file-backed volumes whose data
    survives daemon restarts (local mode only)."""

    def test_aio_volume_lifecycle(self, hipstored, tmp_path):  # noqa: F811
        from oim_amd import hipstore

        backend = LocalBackend(hipstored.socket_path,
                               aio_dir=str(tmp_path / "aio"))
        driver = OIMDriver(driver_name="aio.oim-amd.test", node_id="n0",
                           endpoint=f"unix://{tmp_path}/csi.sock",
                           backend=backend, mounter=Mounter(FakeExec()))
        driver.start()
        channel = grpc.insecure_channel(
            grpc_target(f"unix://{tmp_path}/csi.sock"))
        ctrl = CSIControllerStub(channel)
        try:
            request = csi.CreateVolumeRequest(name="pv-aio")
            request.capacity_range.required_bytes = 1 << 20
            request.volume_capabilities.add().CopyFrom(mount_cap())
            request.parameters["backing"] = "aio"
            reply = ctrl.CreateVolume(request, timeout=10)
            assert reply.volume.volume_context["backing"] == "aio"
            backing = tmp_path / "aio" / "pv-aio.img"
            assert backing.exists() and backing.stat().st_size == 1 << 20
            with hipstore.Client(hipstored.socket_path) as client:
                assert hipstore.get_bdevs(
                    client, "pv-aio")[0].product_name == "AIO disk"
            # idempotent retry
            ctrl.CreateVolume(request, timeout=10)
            # delete destroys the backing file (CSI delete semantics)
            ctrl.DeleteVolume(csi.DeleteVolumeRequest(volume_id="pv-aio"),
                              timeout=10)
            assert not backing.exists()
            # unknown backing -> INVALID_ARGUMENT
            bad = csi.CreateVolumeRequest(name="pv-bad")
            bad.volume_capabilities.add().CopyFrom(mount_cap())
            bad.parameters["backing"] = "tape"
            with pytest.raises(grpc.RpcError) as excinfo:
                ctrl.CreateVolume(bad, timeout=10)
            assert_code(excinfo, grpc.StatusCode.INVALID_ARGUMENT)
        finally:
            channel.close()
            driver.stop()

    def test_aio_volume_expand(self, hipstored, tmp_path):  # noqa: F811
        """Expansion works for file-backed volumes too (ftruncate)."""
        backend = LocalBackend(hipstored.socket_path,
                               aio_dir=str(tmp_path / "aio"))
        _, ctx = backend.create_volume("grow-aio", 1 << 20,
                                       {"backing": "aio"})
        assert ctx["backing"] == "aio"
        new_size = backend.expand_volume("grow-aio", 4 << 20)
        assert new_size == 4 << 20
        backing = tmp_path / "aio" / "grow-aio.img"
        assert backing.stat().st_size == 4 << 20
        from oim_amd import hipstore
        with hipstore.Client(hipstored.socket_path) as client:
            bdev = hipstore.get_bdevs(client, "grow-aio")[0]
            assert bdev.size_bytes == 4 << 20
            assert bdev.driver_specific["aio"]["filename"] == str(backing)
        backend.delete_volume("grow-aio")
        assert not backing.exists()
