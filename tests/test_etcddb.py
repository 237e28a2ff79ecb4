"""etcd-backed RegistryDB against an in-process fake etcd KV server,
plus a multi-registry scenario (BASELINE config 5: several registries
sharing one etcd)."""

from concurrent import futures

import grpc
import pytest

from oim_amd.registry import Registry, RegistryServer
from oim_amd.registry.etcddb import EtcdRegistryDB, _range_end
from oim_amd.registry.etcdpb import FakeEtcdServicer, add_fake_etcd_to_server


@pytest.fixture
def fake_etcd():
    servicer = FakeEtcdServicer()
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=4))
    add_fake_etcd_to_server(servicer, server)
    port = server.add_insecure_port("127.0.0.1:0")
    server.start()
    yield f"127.0.0.1:{port}", servicer
    server.stop(0).wait()


class TestRangeEnd:
    def test_simple(self):
        assert _range_end(b"/oim/") == b"/oim0"
        assert _range_end(b"a\xff") == b"b"
        assert _range_end(b"\xff") == b"\0"


class TestEtcdDB:
    def test_store_lookup_delete(self, fake_etcd):
        endpoint, _ = fake_etcd
        db = EtcdRegistryDB([endpoint])
        db.store(["host-0", "address"], "tcp://a:1")
        assert db.lookup(["host-0", "address"]) == "tcp://a:1"
        db.store(["host-0", "address"], "")
        assert db.lookup(["host-0", "address"]) is None
        db.close()

    def test_list_prefix(self, fake_etcd):
        endpoint, _ = fake_etcd
        db = EtcdRegistryDB([endpoint])
        db.store(["gpu-0", "address"], "a")
        db.store(["gpu-0", "pci"], "0000:c1:00.0")
        db.store(["gpu-1", "address"], "b")
        assert len(db.list([])) == 3
        assert [e for e, _ in db.list(["gpu-0"])] == [
            ["gpu-0", "address"], ["gpu-0", "pci"]]
        db.close()

    def test_failover_to_second_endpoint(self, fake_etcd):
        endpoint, _ = fake_etcd
        # first endpoint is dead; client fails over
        db = EtcdRegistryDB(["127.0.0.1:1", endpoint], timeout=2.0)
        db.store(["x", "y"], "v")
        assert db.lookup(["x", "y"]) == "v"
        db.close()

    def test_shared_by_two_registries(self, fake_etcd):
        """Two registry instances on one etcd see each other's writes —
        the HA/multi-registry layout of config 5."""
        endpoint, _ = fake_etcd
        db_a = EtcdRegistryDB([endpoint])
        db_b = EtcdRegistryDB([endpoint])
        registry_a = Registry(db=db_a)
        registry_b = Registry(db=db_b)
        registry_a.db.store(["gpu-3", "address"], "tcp://card3:8999")
        assert registry_b.db.lookup(["gpu-3", "address"]) == "tcp://card3:8999"
        db_a.close()
        db_b.close()


class TestHaControlPlane:
    def test_two_registries_one_etcd_full_path(self, fake_etcd, tmp_path):
        """Config-5 control plane: a controller registers through
        registry A; a client provisions through registry B's proxy
        (both share one etcd)."""
        import grpc

        from oim_amd import spec
        from oim_amd.common.server import grpc_target
        from oim_amd.controller import Controller, ControllerServer
        from fixtures import launch_hipstored

        endpoint, _ = fake_etcd
        reg_a = Registry(db=EtcdRegistryDB([endpoint]))
        server_a = RegistryServer(f"unix://{tmp_path}/rega.sock", reg_a)
        server_a.start()
        reg_b = Registry(db=EtcdRegistryDB([endpoint]))
        server_b = RegistryServer(f"unix://{tmp_path}/regb.sock", reg_b)
        server_b.start()
        daemon = launch_hipstored(tmp_path, cpu=True)
        controller = Controller(
            controller_id="gpu-7",
            hipstored_socket=daemon.socket_path,
            controller_address=f"unix://{tmp_path}/ctrl.sock",
            registry_address=server_a.addr(),  # registers via A
            registry_delay=3600.0,
        )
        ctrl_server = ControllerServer(f"unix://{tmp_path}/ctrl.sock",
                                       controller)
        ctrl_server.start()
        try:
            controller.register()
            # B sees the registration through etcd and proxies to it.
            with grpc.insecure_channel(grpc_target(server_b.addr())) as ch:
                stub = spec.ControllerStub(ch)
                metadata = ((spec.CONTROLLER_ID_KEY, "gpu-7"),)
                stub.ProvisionMallocBDev(
                    spec.ProvisionMallocBDevRequest(bdev_name="ha-vol",
                                                    size=1 << 20),
                    metadata=metadata, timeout=30)
                stub.CheckMallocBDev(
                    spec.CheckMallocBDevRequest(bdev_name="ha-vol"),
                    metadata=metadata, timeout=30)
                stub.ProvisionMallocBDev(
                    spec.ProvisionMallocBDevRequest(bdev_name="ha-vol",
                                                    size=0),
                    metadata=metadata, timeout=30)
        finally:
            ctrl_server.stop()
            daemon.stop()
            server_b.stop()
            server_a.stop()
