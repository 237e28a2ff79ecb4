"""Failure-recovery properties (reference SURVEY.md section 5):
per-operation dialing means component restarts never strand clients;
CSI edge cases around idempotency."""

import grpc
import pytest

from oim_amd import spec
from oim_amd.common.server import grpc_target
from oim_amd.controller import Controller, ControllerServer
from oim_amd.csidriver import LocalBackend, Mounter, FakeExec, OIMDriver
from oim_amd.registry import MemRegistryDB, Registry, RegistryServer
from oim_amd.spec import csi_v1 as csi
from oim_amd.spec.rpc_csi import CSIControllerStub

from fixtures import hipstored  # noqa: F401


class TestControllerRestart:
    def test_proxy_survives_controller_restart(self, hipstored, tmp_path):  # noqa: F811
        """The registry dials the controller per request
        (reference registry.go:196-210, README.md:49): kill and restart
        the controller and the same client keeps working."""
        registry = Registry(db=MemRegistryDB())
        reg_server = RegistryServer(f"unix://{tmp_path}/reg.sock", registry)
        reg_server.start()
        endpoint = f"unix://{tmp_path}/ctrl.sock"

        def start_controller():
            controller = Controller(controller_id="c0",
                                    hipstored_socket=hipstored.socket_path)
            server = ControllerServer(endpoint, controller)
            server.start()
            return server

        ctrl = start_controller()
        registry.db.store(["c0", "address"], endpoint)
        metadata = ((spec.CONTROLLER_ID_KEY, "c0"),)
        try:
            with grpc.insecure_channel(grpc_target(reg_server.addr())) as ch:
                stub = spec.ControllerStub(ch)
                stub.ProvisionMallocBDev(
                    spec.ProvisionMallocBDevRequest(bdev_name="rv", size=1 << 20),
                    metadata=metadata, timeout=30)
                # restart the controller process-equivalent
                ctrl.stop()
                with pytest.raises(grpc.RpcError):
                    stub.CheckMallocBDev(
                        spec.CheckMallocBDevRequest(bdev_name="rv"),
                        metadata=metadata, timeout=5)
                ctrl = start_controller()
                # same channel, no re-dial needed by the client: the
                # proxy's per-request dial picks up the new controller
                stub.CheckMallocBDev(
                    spec.CheckMallocBDevRequest(bdev_name="rv"),
                    metadata=metadata, timeout=30)
                stub.ProvisionMallocBDev(
                    spec.ProvisionMallocBDevRequest(bdev_name="rv", size=0),
                    metadata=metadata, timeout=30)
        finally:
            ctrl.stop()
            reg_server.stop()

    def test_daemon_data_survives_controller_restart(self, hipstored, tmp_path):  # noqa: F811
        """Bdev state lives in hipstored, not the controller: a volume
        provisioned before a controller restart is still mapped after."""
        controller = Controller(controller_id="c1",
                                hipstored_socket=hipstored.socket_path)
        server = ControllerServer(f"unix://{tmp_path}/c1.sock", controller)
        server.start()

        class Ctx:
            def abort(self, code, details):
                raise AssertionError(f"{code}: {details}")

        controller.ProvisionMallocBDev(
            spec.ProvisionMallocBDevRequest(bdev_name="persist", size=1 << 20),
            Ctx())
        reply1 = controller.MapVolume(
            spec.MapVolumeRequest(volume_id="persist",
                                  malloc=spec.MallocParams()), Ctx())
        server.stop()
        # "restarted" controller sees the existing mapping (idempotent
        # MapVolume against daemon state)
        controller2 = Controller(controller_id="c1",
                                 hipstored_socket=hipstored.socket_path)
        server2 = ControllerServer(f"unix://{tmp_path}/c1b.sock", controller2)
        server2.start()
        try:
            reply2 = controller2.MapVolume(
                spec.MapVolumeRequest(volume_id="persist",
                                      malloc=spec.MallocParams()), Ctx())
            assert reply1.scsi_disk.target == reply2.scsi_disk.target
            controller2.UnmapVolume(
                spec.UnmapVolumeRequest(volume_id="persist"), Ctx())
            controller2.ProvisionMallocBDev(
                spec.ProvisionMallocBDevRequest(bdev_name="persist", size=0),
                Ctx())
        finally:
            server2.stop()


class TestCSISanity:
    """csi-test-sanity-style edge cases (reference oim-driver_test.go
    used the upstream suite; these are the load-bearing subset)."""

    @pytest.fixture
    def driver(self, hipstored, tmp_path):  # noqa: F811
        backend = LocalBackend(hipstored.socket_path)
        driver = OIMDriver(driver_name="oim-local", node_id="n0",
                           endpoint=f"unix://{tmp_path}/csi.sock",
                           backend=backend, mounter=Mounter(FakeExec()))
        driver.start()
        yield driver
        driver.stop()

    def _stub(self, driver):
        channel = grpc.insecure_channel(grpc_target(driver.addr()))
        return CSIControllerStub(channel), channel

    def _create(self, stub, name, size=1 << 20):
        request = csi.CreateVolumeRequest(name=name)
        request.capacity_range.required_bytes = size
        cap = request.volume_capabilities.add()
        cap.mount.fs_type = "ext4"
        cap.access_mode.mode = csi.ACCESS_MODE_SINGLE_NODE_WRITER
        return stub.CreateVolume(request, timeout=30)

    def test_create_idempotent_same_size(self, driver):
        stub, channel = self._stub(driver)
        with channel:
            a = self._create(stub, "sanity-1")
            b = self._create(stub, "sanity-1")
            assert a.volume.volume_id == b.volume.volume_id
            stub.DeleteVolume(csi.DeleteVolumeRequest(
                volume_id=a.volume.volume_id), timeout=30)

    def test_create_conflicting_size_rejected(self, driver):
        stub, channel = self._stub(driver)
        with channel:
            self._create(stub, "sanity-2", 1 << 20)
            with pytest.raises(grpc.RpcError) as excinfo:
                self._create(stub, "sanity-2", 2 << 20)
            assert excinfo.value.code() == grpc.StatusCode.ALREADY_EXISTS
            stub.DeleteVolume(csi.DeleteVolumeRequest(volume_id="sanity-2"),
                              timeout=30)

    def test_delete_nonexistent_ok(self, driver):
        stub, channel = self._stub(driver)
        with channel:
            stub.DeleteVolume(csi.DeleteVolumeRequest(volume_id="ghost"),
                              timeout=30)

    def test_create_missing_capabilities(self, driver):
        stub, channel = self._stub(driver)
        with channel:
            request = csi.CreateVolumeRequest(name="nocaps")
            request.capacity_range.required_bytes = 1 << 20
            with pytest.raises(grpc.RpcError) as excinfo:
                stub.CreateVolume(request, timeout=30)
            assert excinfo.value.code() == grpc.StatusCode.INVALID_ARGUMENT

    def test_validate_missing_volume(self, driver):
        stub, channel = self._stub(driver)
        with channel:
            request = csi.ValidateVolumeCapabilitiesRequest(volume_id="ghost")
            cap = request.volume_capabilities.add()
            cap.mount.fs_type = "ext4"
            cap.access_mode.mode = csi.ACCESS_MODE_SINGLE_NODE_WRITER
            with pytest.raises(grpc.RpcError) as excinfo:
                stub.ValidateVolumeCapabilities(request, timeout=30)
            assert excinfo.value.code() == grpc.StatusCode.NOT_FOUND


class TestNodeGetVolumeStats:
    def test_stats_of_mounted_path(self, hipstored, tmp_path):  # noqa: F811
        from oim_amd.spec.rpc_csi import CSINodeStub

        backend = LocalBackend(hipstored.socket_path)
        driver = OIMDriver(driver_name="oim-local", node_id="n0",
                           endpoint=f"unix://{tmp_path}/csi2.sock",
                           backend=backend, mounter=Mounter(FakeExec()))
        driver.start()
        try:
            channel = grpc.insecure_channel(grpc_target(driver.addr()))
            with channel:
                stub = CSINodeStub(channel)
                response = stub.NodeGetVolumeStats(
                    csi.NodeGetVolumeStatsRequest(
                        volume_id="v", volume_path=str(tmp_path)),
                    timeout=30)
                by_unit = {u.unit: u for u in response.usage}
                assert csi.USAGE_UNIT_BYTES in by_unit
                assert by_unit[csi.USAGE_UNIT_BYTES].total > 0
                assert csi.USAGE_UNIT_INODES in by_unit
                with pytest.raises(grpc.RpcError) as excinfo:
                    stub.NodeGetVolumeStats(
                        csi.NodeGetVolumeStatsRequest(
                            volume_id="v", volume_path="/no/such/path"),
                        timeout=30)
                assert excinfo.value.code() == grpc.StatusCode.NOT_FOUND
        finally:
            driver.stop()


class TestGetCapacity:
    def test_local_mode_cpu_host_memory(self, hipstored, tmp_path):  # noqa: F811
        """CPU-mode daemon: malloc bdevs consume host RAM, so capacity
        is host MemAvailable (on a GPU box: free HBM bytes)."""
        backend = LocalBackend(hipstored.socket_path)
        capacity = backend.get_capacity()
        assert capacity is not None and capacity > 0
        driver = OIMDriver(driver_name="oim-local", node_id="n0",
                           endpoint=f"unix://{tmp_path}/csi3.sock",
                           backend=backend, mounter=Mounter(FakeExec()))
        driver.start()
        try:
            with grpc.insecure_channel(grpc_target(driver.addr())) as ch:
                stub = CSIControllerStub(ch)
                reply = stub.GetCapacity(csi.GetCapacityRequest(), timeout=30)
                # MemAvailable moves between samples; same ballpark.
                assert reply.available_capacity > 0
                assert abs(reply.available_capacity - capacity) < capacity
        finally:
            driver.stop()

    def test_hbm_info_rpc(self, hipstored):  # noqa: F811
        from oim_amd import hipstore as hsclient

        with hsclient.Client(hipstored.socket_path) as client:
            info = client.invoke("get_hbm_info")
            assert "total_bytes" in info and "free_bytes" in info


class TestKitchenSink:
    """Every subsystem at once in one daemon — malloc + aio + striped +
    replicated + clone + vhost scsi/blk + NVMe-oF — snapshotted,
    restarted with -c, and verified serving."""

    def test_full_topology_replay(self, tmp_path):
        import json as jsonmod
        import subprocess
        import time as time_mod

        import os
        import fixtures
        from oim_amd import hipstore
        from vhost_client import VhostUserMaster

        backing = tmp_path / "aio.img"
        backing.write_bytes(bytes(1 << 20))
        daemon = fixtures.launch_hipstored(tmp_path, cpu=True)
        with hipstore.Client(daemon.socket_path) as client:
            hipstore.construct_malloc_bdev(client, num_blocks=2048,
                                           block_size=512, name="ks-m")
            hipstore.construct_aio_bdev(client, "ks-aio", str(backing))
            hipstore.construct_striped_malloc_bdev(
                client, "ks-st", num_blocks=512, block_size=512, count=2)
            hipstore.construct_replicated_malloc_bdev(
                client, "ks-re", num_blocks=512, block_size=512, count=2)
            hipstore.bdev_clone(client, "ks-m", "ks-clone")
            client.invoke("construct_vhost_scsi_controller",
                          {"ctrlr": "ks-vs"})
            client.invoke("add_vhost_scsi_lun",
                          {"ctrlr": "ks-vs", "scsi_target_num": 0,
                           "bdev_name": "ks-m"})
            client.invoke("construct_vhost_blk_controller",
                          {"ctrlr": "ks-vb", "dev_name": "ks-clone"})
            hipstore.nvmf_create_target(client, subnqn="nqn.ks",
                                        bdevs=["ks-st"])
            config = hipstore.save_config(client)
        daemon.stop()

        config_path = tmp_path / "ks.json"
        config_path.write_text(jsonmod.dumps(config))
        sock2 = str(tmp_path / "ks2.sock")
        proc = subprocess.Popen(
            [fixtures.DEFAULT_BINARY, "-S", sock2, "-C",
             "-c", str(config_path)], stderr=subprocess.PIPE)
        deadline = time_mod.time() + 30
        import os as osmod
        while not osmod.path.exists(sock2):
            assert proc.poll() is None, proc.stderr.read().decode()
            assert time_mod.time() < deadline
            time_mod.sleep(0.05)
        try:
            with hipstore.Client(sock2) as client:
                names = {b.name for b in hipstore.get_bdevs(client)}
                assert {"ks-m", "ks-aio", "ks-st", "ks-re",
                        "ks-clone"} <= names
                controllers = {c.controller: c
                               for c in hipstore.get_vhost_controllers(
                                   client)}
                assert {"ks-vs", "ks-vb"} <= set(controllers)
                subsystems = client.invoke("nvmf_get_subsystems")
                assert any(s["nqn"] == "nqn.ks" for s in subsystems)
                # the replayed vhost-scsi target actually serves I/O
                master = VhostUserMaster(str(tmp_path / "ks-vs"))
                master.negotiate()
                try:
                    payload = os.urandom(512)
                    assert master.write10(0, 1, payload, 512).status == 0
                    assert master.read10(0, 1, 1, 512).data == payload
                finally:
                    master.close()
        finally:
            proc.terminate()
            proc.wait(timeout=10)


class TestDaemonHardKill:
    """SIGKILL the data daemon with a client mid-conversation: the
    client must fail with an error promptly (bounded-everything
    discipline — never a hang), and the socket path must be reusable
    by a fresh daemon that then replays the saved config (the SPDK
    crash-recovery model: state lives in the config snapshot, not the
    process)."""

    def test_kill_mid_session_fails_fast_and_socket_reusable(self, tmp_path):
        import os
        import signal
        import time

        import fixtures as fx
        from oim_amd import hipstore

        daemon = fx.launch_hipstored(tmp_path)
        try:
            with hipstore.Client(daemon.socket_path) as client:
                hipstore.construct_malloc_bdev(
                    client, num_blocks=4096, block_size=512, name="hk")
                config = client.invoke("save_config")
                daemon.process.send_signal(signal.SIGKILL)
                daemon.process.wait(timeout=10)
                t0 = time.monotonic()
                with pytest.raises((hipstore.RpcError, OSError)):
                    for _ in range(3):  # first call may see buffered EOF
                        client.invoke("get_bdevs")
                assert time.monotonic() - t0 < 5.0  # failed, not hung
        finally:
            daemon.stop()
        # Same socket path, fresh daemon, replayed topology.
        os.unlink(daemon.socket_path)
        daemon2 = fx.launch_hipstored(tmp_path)
        try:
            with hipstore.Client(daemon2.socket_path) as client:
                client.invoke("load_config", config)
                assert hipstore.get_bdevs(client, "hk")[0].num_blocks == 4096
        finally:
            daemon2.stop()
