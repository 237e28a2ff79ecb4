"""etcd-backed RegistryDB against an in-process fake etcd KV server,
plus a multi-registry scenario (BASELINE config 5: several registries
sharing one etcd)."""

from concurrent import futures

import grpc
import pytest

from oim_amd.registry import MemRegistryDB, Registry, RegistryServer
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
