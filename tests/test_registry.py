"""Registry tests: DB semantics, service, proxy, TLS matrix.

Counterpart of the reference's pkg/oim-registry/registry_test.go
(DB semantics :55-120, proxy :150-250, mutual-TLS matrix :253-392).
"""

import os

import grpc
import pytest

from oim_amd import spec
from oim_amd.common.server import grpc_target
from oim_amd.common.tlsutil import channel_options_for_peer, load_tls_channel_credentials
from oim_amd.controller import Controller, ControllerServer
from oim_amd.registry import (
    FileRegistryDB,
    MemRegistryDB,
    Registry,
    RegistryServer,
)

import ca as ca_util


class TestMemDB:
    def test_store_lookup_delete(self):
        db = MemRegistryDB()
        db.store(["host-0", "address"], "tcp://a:1")
        assert db.lookup(["host-0", "address"]) == "tcp://a:1"
        db.store(["host-0", "address"], "")
        assert db.lookup(["host-0", "address"]) is None

    def test_list_prefix(self):
        db = MemRegistryDB()
        db.store(["a", "x"], "1")
        db.store(["a", "y"], "2")
        db.store(["b", "x"], "3")
        assert len(db.list([])) == 3
        assert [(e, v) for e, v in db.list(["a"])] == [
            (["a", "x"], "1"),
            (["a", "y"], "2"),
        ]
        # element-wise prefix: "a" does not match "ab"
        db.store(["ab", "z"], "4")
        assert len(db.list(["a"])) == 2


class TestFileDB:
    def test_persistence(self, tmp_path):
        path = str(tmp_path / "reg.json")
        db = FileRegistryDB(path)
        db.store(["c1", "address"], "tcp://x:9")
        db2 = FileRegistryDB(path)
        assert db2.lookup(["c1", "address"]) == "tcp://x:9"
        db2.store(["c1", "address"], "")
        db3 = FileRegistryDB(path)
        assert db3.lookup(["c1", "address"]) is None

    def test_corrupt_file_refused_with_path_in_error(self, tmp_path):
        """A corrupt DB must fail startup loudly (state is durable;
        silently discarding it would un-register every controller),
        and the error must say which file."""
        path = str(tmp_path / "reg.json")
        with open(path, "w") as f:
            f.write("{truncated")
        with pytest.raises(RuntimeError, match="reg.json"):
            FileRegistryDB(path)
        with open(path, "w") as f:
            f.write('["a", "list", "not", "a", "map"]')
        with pytest.raises(RuntimeError, match="string-to-string"):
            FileRegistryDB(path)
        # a stale .tmp from a torn flush is harmless: ignored, and the
        # next write replaces it
        with open(path, "w") as f:
            f.write('{"c1/address": "tcp://x:9"}')
        with open(path + ".tmp", "w") as f:
            f.write("{garbage")
        db = FileRegistryDB(path)
        assert db.lookup(["c1", "address"]) == "tcp://x:9"
        db.store(["c2", "address"], "tcp://y:1")
        assert FileRegistryDB(path).lookup(["c2", "address"]) == "tcp://y:1"


class MockController(spec.ControllerServicer):
    """Records requests (reference registry_test.go MockController)."""

    def __init__(self):
        self.map_requests = []

    def MapVolume(self, request, context):
        self.map_requests.append(request.volume_id)
        return spec.MapVolumeReply(
            scsi_disk=spec.SCSIDisk(target=1, lun=0))


@pytest.fixture
def plaintext_registry(tmp_sock):
    registry = Registry(db=MemRegistryDB())
    server = RegistryServer(f"unix://{tmp_sock}", registry)
    server.start()
    yield registry, server
    server.stop()


def registry_channel(server):
    return grpc.insecure_channel(grpc_target(server.addr()))


class TestRegistryService:
    def test_set_get(self, plaintext_registry):
        registry, server = plaintext_registry
        with registry_channel(server) as channel:
            stub = spec.RegistryStub(channel)
            stub.SetValue(spec.SetValueRequest(
                value=spec.Value(path="host-0/address", value="tcp://c:1")),
                timeout=5)
            reply = stub.GetValues(spec.GetValuesRequest(path="host-0"), timeout=5)
            assert [(v.path, v.value) for v in reply.values] == [
                ("host-0/address", "tcp://c:1")]

    def test_invalid_path(self, plaintext_registry):
        registry, server = plaintext_registry
        with registry_channel(server) as channel:
            stub = spec.RegistryStub(channel)
            with pytest.raises(grpc.RpcError) as excinfo:
                stub.SetValue(spec.SetValueRequest(
                    value=spec.Value(path="../escape", value="x")), timeout=5)
            assert excinfo.value.code() == grpc.StatusCode.INVALID_ARGUMENT

    def test_delete_via_empty_value(self, plaintext_registry):
        registry, server = plaintext_registry
        with registry_channel(server) as channel:
            stub = spec.RegistryStub(channel)
            stub.SetValue(spec.SetValueRequest(
                value=spec.Value(path="c/address", value="v")), timeout=5)
            stub.SetValue(spec.SetValueRequest(
                value=spec.Value(path="c/address", value="")), timeout=5)
            reply = stub.GetValues(spec.GetValuesRequest(), timeout=5)
            assert len(reply.values) == 0


class TestProxy:
    @pytest.fixture
    def controller_backend(self, tmp_path):
        mock = MockController()
        from oim_amd.common.server import NonBlockingGRPCServer

        server = NonBlockingGRPCServer(f"unix://{tmp_path}/ctrl.sock")
        server.start(lambda s: spec.add_controller_to_server(mock, s))
        yield mock, f"unix://{tmp_path}/ctrl.sock"
        server.stop()

    def test_proxy_roundtrip(self, plaintext_registry, controller_backend):
        registry, server = plaintext_registry
        mock, ctrl_endpoint = controller_backend
        registry.db.store(["ctrl-a", "address"], ctrl_endpoint)
        with registry_channel(server) as channel:
            stub = spec.ControllerStub(channel)
            reply = stub.MapVolume(
                spec.MapVolumeRequest(volume_id="vol-1",
                                      malloc=spec.MallocParams()),
                metadata=((spec.CONTROLLER_ID_KEY, "ctrl-a"),),
                timeout=10,
            )
            assert reply.scsi_disk.target == 1
        assert mock.map_requests == ["vol-1"]

    def test_proxy_missing_controllerid(self, plaintext_registry):
        registry, server = plaintext_registry
        with registry_channel(server) as channel:
            stub = spec.ControllerStub(channel)
            with pytest.raises(grpc.RpcError) as excinfo:
                stub.MapVolume(
                    spec.MapVolumeRequest(volume_id="v"), timeout=5)
            assert excinfo.value.code() == grpc.StatusCode.UNIMPLEMENTED

    def test_proxy_unregistered_controller(self, plaintext_registry):
        registry, server = plaintext_registry
        with registry_channel(server) as channel:
            stub = spec.ControllerStub(channel)
            with pytest.raises(grpc.RpcError) as excinfo:
                stub.MapVolume(
                    spec.MapVolumeRequest(volume_id="v"),
                    metadata=((spec.CONTROLLER_ID_KEY, "ghost"),),
                    timeout=5,
                )
            assert excinfo.value.code() == grpc.StatusCode.UNAVAILABLE

    def test_unknown_registry_method_not_proxied(self, plaintext_registry):
        registry, server = plaintext_registry
        with registry_channel(server) as channel:
            method = channel.unary_unary(
                f"/{spec.REGISTRY_SERVICE}/Bogus",
                request_serializer=lambda b: b,
                response_deserializer=lambda b: b,
            )
            with pytest.raises(grpc.RpcError) as excinfo:
                method(b"", metadata=((spec.CONTROLLER_ID_KEY, "x"),), timeout=5)
            assert excinfo.value.code() == grpc.StatusCode.UNIMPLEMENTED


@pytest.fixture(scope="module")
def cas(tmp_path_factory):
    """Trusted and evil CA cert depots (reference _work/ca + evil-ca)."""
    base = tmp_path_factory.mktemp("cas")
    trusted = ca_util.make_ca(str(base / "ca"))
    evil = ca_util.make_ca(str(base / "evil-ca"), ca_name="Evil CA")
    return trusted, evil


class TestTLS:
    """Mutual-TLS matrix (reference registry_test.go:253-392)."""

    @pytest.fixture
    def tls_registry(self, cas, tmp_path):
        trusted, _ = cas
        registry = Registry(
            db=MemRegistryDB(),
            tls=ca_util.tls_config(trusted, "component.registry"),
        )
        server = RegistryServer(f"unix://{tmp_path}/reg.sock", registry)
        server.start()
        yield registry, server, trusted
        server.stop()

    def _stub(self, server, ca_dir, name):
        creds = load_tls_channel_credentials(
            ca_util.tls_config(ca_dir, name))
        channel = grpc.secure_channel(
            grpc_target(server.addr()), creds,
            options=channel_options_for_peer("component.registry"))
        return spec.RegistryStub(channel), channel

    def test_admin_can_set(self, tls_registry):
        registry, server, trusted = tls_registry
        stub, channel = self._stub(server, trusted, "user.admin")
        with channel:
            stub.SetValue(spec.SetValueRequest(
                value=spec.Value(path="host-0/pci", value="0000:00:15.0")),
                timeout=5)
            reply = stub.GetValues(spec.GetValuesRequest(), timeout=5)
            assert len(reply.values) == 1

    def test_controller_can_set_own_address_only(self, tls_registry):
        registry, server, trusted = tls_registry
        stub, channel = self._stub(server, trusted, "controller.host-0")
        with channel:
            stub.SetValue(spec.SetValueRequest(
                value=spec.Value(path="host-0/address", value="tcp://x:1")),
                timeout=5)
            # wrong id
            with pytest.raises(grpc.RpcError) as excinfo:
                stub.SetValue(spec.SetValueRequest(
                    value=spec.Value(path="other/address", value="x")),
                    timeout=5)
            assert excinfo.value.code() == grpc.StatusCode.PERMISSION_DENIED
            # wrong key
            with pytest.raises(grpc.RpcError) as excinfo:
                stub.SetValue(spec.SetValueRequest(
                    value=spec.Value(path="host-0/pci", value="x")),
                    timeout=5)
            assert excinfo.value.code() == grpc.StatusCode.PERMISSION_DENIED

    def test_host_cannot_set(self, tls_registry):
        registry, server, trusted = tls_registry
        stub, channel = self._stub(server, trusted, "host.host-0")
        with channel:
            with pytest.raises(grpc.RpcError) as excinfo:
                stub.SetValue(spec.SetValueRequest(
                    value=spec.Value(path="host-0/address", value="x")),
                    timeout=5)
            assert excinfo.value.code() == grpc.StatusCode.PERMISSION_DENIED

    def test_evil_ca_client_rejected(self, tls_registry, cas):
        registry, server, trusted = tls_registry
        _, evil = cas
        # evil client presents a cert from the wrong CA but trusts the
        # real server CA (so the failure is the server rejecting us).
        creds = grpc.ssl_channel_credentials(
            root_certificates=open(os.path.join(trusted, "ca.crt"), "rb").read(),
            private_key=open(os.path.join(evil, "user.admin.key"), "rb").read(),
            certificate_chain=open(os.path.join(evil, "user.admin.crt"), "rb").read(),
        )
        channel = grpc.secure_channel(
            grpc_target(server.addr()), creds,
            options=channel_options_for_peer("component.registry"))
        with channel:
            stub = spec.RegistryStub(channel)
            with pytest.raises(grpc.RpcError) as excinfo:
                stub.GetValues(spec.GetValuesRequest(), timeout=5)
            assert excinfo.value.code() == grpc.StatusCode.UNAVAILABLE

    def test_wrong_server_name_rejected_by_client(self, tls_registry):
        registry, server, trusted = tls_registry
        creds = load_tls_channel_credentials(
            ca_util.tls_config(trusted, "user.admin"))
        channel = grpc.secure_channel(
            grpc_target(server.addr()), creds,
            options=channel_options_for_peer("controller.host-0"))
        with channel:
            stub = spec.RegistryStub(channel)
            with pytest.raises(grpc.RpcError) as excinfo:
                stub.GetValues(spec.GetValuesRequest(), timeout=5)
            assert excinfo.value.code() == grpc.StatusCode.UNAVAILABLE

    def test_proxy_authz_host_mismatch(self, tls_registry, tmp_path):
        """host.host-0 may proxy only to controller host-0."""
        registry, server, trusted = tls_registry
        registry.db.store(["other", "address"], "unix:///nonexistent.sock")
        creds = load_tls_channel_credentials(
            ca_util.tls_config(trusted, "host.host-0"))
        channel = grpc.secure_channel(
            grpc_target(server.addr()), creds,
            options=channel_options_for_peer("component.registry"))
        with channel:
            stub = spec.ControllerStub(channel)
            with pytest.raises(grpc.RpcError) as excinfo:
                stub.MapVolume(
                    spec.MapVolumeRequest(volume_id="v"),
                    metadata=((spec.CONTROLLER_ID_KEY, "other"),),
                    timeout=5,
                )
            assert excinfo.value.code() == grpc.StatusCode.PERMISSION_DENIED


class TestProxyDeadlines:
    def test_client_deadline_propagates_through_proxy(self, tmp_path):
        """A short client deadline on a proxied call to a stalled
        controller surfaces as DEADLINE_EXCEEDED at the client and
        does not wedge the proxy for later calls."""
        import time as time_mod

        from oim_amd import spec
        from oim_amd.common.server import NonBlockingGRPCServer, grpc_target
        from oim_amd.spec.rpc import (ControllerServicer,
                                      add_controller_to_server)

        class StallServicer(ControllerServicer):
            def __init__(self):
                self.calls = 0

            def CheckMallocBDev(self, request, context):
                self.calls += 1
                if self.calls == 1:
                    time_mod.sleep(3)  # beyond the client's deadline
                return spec.CheckMallocBDevReply()

        stalled = StallServicer()
        ctrl_server = NonBlockingGRPCServer(
            endpoint=f"unix://{tmp_path}/slow.sock")
        ctrl_server.start(lambda s: add_controller_to_server(stalled, s))
        registry = Registry(db=MemRegistryDB())
        reg_server = RegistryServer(f"unix://{tmp_path}/reg.sock", registry)
        reg_server.start()
        registry.db.store(["slow", "address"], f"unix://{tmp_path}/slow.sock")
        try:
            with grpc.insecure_channel(grpc_target(reg_server.addr())) as ch:
                stub = spec.ControllerStub(ch)
                md = ((spec.CONTROLLER_ID_KEY, "slow"),)
                started = time_mod.monotonic()
                with pytest.raises(grpc.RpcError) as excinfo:
                    stub.CheckMallocBDev(
                        spec.CheckMallocBDevRequest(bdev_name="x"),
                        metadata=md, timeout=0.5)
                assert excinfo.value.code() == \
                    grpc.StatusCode.DEADLINE_EXCEEDED
                assert time_mod.monotonic() - started < 2.5
                # proxy healthy for the next (fast) call
                stub.CheckMallocBDev(
                    spec.CheckMallocBDevRequest(bdev_name="x"),
                    metadata=md, timeout=10)
        finally:
            reg_server.stop()
            ctrl_server.stop()


class TestProxyGarbagePayload:
    def test_malformed_protobuf_through_proxy(self, tmp_path):
        """The proxy splices raw bytes; a garbage payload must surface
        as a gRPC error from the backend's deserializer, and both
        proxy and controller stay healthy."""
        import fixtures
        from oim_amd.common.server import grpc_target
        from oim_amd.controller import Controller, ControllerServer

        daemon = fixtures.launch_hipstored(tmp_path, cpu=True)
        registry = Registry(db=MemRegistryDB())
        reg_server = RegistryServer(f"unix://{tmp_path}/r.sock", registry)
        reg_server.start()
        controller = Controller(controller_id="fz",
                                hipstored_socket=daemon.socket_path)
        ctrl_server = ControllerServer(f"unix://{tmp_path}/c.sock", controller)
        ctrl_server.start()
        registry.db.store(["fz", "address"], f"unix://{tmp_path}/c.sock")
        try:
            with grpc.insecure_channel(grpc_target(reg_server.addr())) as ch:
                raw = ch.unary_unary(
                    "/oim.v0.Controller/ProvisionMallocBDev",
                    request_serializer=lambda b: b,
                    response_deserializer=lambda b: b)
                md = ((spec.CONTROLLER_ID_KEY, "fz"),)
                for blob in (b"\xff" * 64, b"\x0a", os.urandom(120)):
                    try:
                        raw(blob, metadata=md, timeout=10)
                    except grpc.RpcError:
                        pass  # any clean gRPC error is acceptable
                # healthy afterwards
                stub = spec.ControllerStub(ch)
                stub.ProvisionMallocBDev(
                    spec.ProvisionMallocBDevRequest(bdev_name="ok",
                                                    size=1 << 20),
                    metadata=md, timeout=30)
                stub.ProvisionMallocBDev(
                    spec.ProvisionMallocBDevRequest(bdev_name="ok", size=0),
                    metadata=md, timeout=30)
        finally:
            ctrl_server.stop()
            reg_server.stop()
            daemon.stop()


class TestPathPropertyFuzz:
    def test_setvalue_path_properties(self, tmp_path):
        """Any path either stores + round-trips through GetValues or is
        rejected INVALID_ARGUMENT; the registry survives all of them."""
        from hypothesis import HealthCheck, given, settings
        from hypothesis import strategies as st

        from oim_amd.common.server import grpc_target

        registry = Registry(db=MemRegistryDB())
        server = RegistryServer(f"unix://{tmp_path}/pf.sock", registry)
        server.start()
        channel = grpc.insecure_channel(grpc_target(server.addr()))
        stub = spec.RegistryStub(channel)
        try:
            from oim_amd.common import split_registry_path

            @settings(max_examples=150, deadline=None,
                      suppress_health_check=[
                          HealthCheck.function_scoped_fixture])
            @given(st.text(max_size=40))
            def fuzz(path):
                try:
                    stub.SetValue(spec.SetValueRequest(
                        value=spec.Value(path=path, value="v")), timeout=10)
                except grpc.RpcError as exc:
                    assert exc.code() == grpc.StatusCode.INVALID_ARGUMENT
                    return
                # Accepted paths round-trip in CANONICAL form (empty
                # elements normalized away, like the reference).
                canonical = "/".join(split_registry_path(path))
                reply = stub.GetValues(spec.GetValuesRequest(path=path),
                                       timeout=10)
                assert any(v.path == canonical and v.value == "v"
                           for v in reply.values), (path, canonical)
                stub.SetValue(spec.SetValueRequest(
                    value=spec.Value(path=path, value="")), timeout=10)

            fuzz()
        finally:
            channel.close()
            server.stop()
