"""Negative-path protocol tests: the daemon and NVMe/TCP target must
survive malformed clients (parse errors answered or connection dropped,
process stays healthy)."""

import socket

import pytest

from oim_amd import _hipstore as hs
from oim_amd import hipstore

from fixtures import hipstored  # noqa: F401


class TestJsonRpcRobustness:
    def test_malformed_json_then_healthy(self, hipstored):  # noqa: F811
        with socket.socket(socket.AF_UNIX, socket.SOCK_STREAM) as raw:
            raw.connect(hipstored.socket_path)
            raw.sendall(b'{"jsonrpc": "2.0", "method": [BROKEN')
            raw.settimeout(10)
            reply = raw.recv(65536)
            assert b"-32700" in reply or reply == b""  # parse error or drop
        # daemon still serves new clients
        with hipstore.Client(hipstored.socket_path) as client:
            assert "get_bdevs" in client.invoke("get_rpc_methods")

    def test_non_object_request(self, hipstored):  # noqa: F811
        with socket.socket(socket.AF_UNIX, socket.SOCK_STREAM) as raw:
            raw.connect(hipstored.socket_path)
            raw.sendall(b"[1,2,3]")
            raw.settimeout(10)
            reply = raw.recv(65536)
            assert b"-32600" in reply  # invalid request
        with hipstore.Client(hipstored.socket_path) as client:
            assert client.invoke("get_bdevs") == []

    def test_missing_params_defaults(self, hipstored):  # noqa: F811
        with hipstore.Client(hipstored.socket_path) as client:
            with pytest.raises(hipstore.RpcError):
                client.invoke("construct_malloc_bdev")  # no params at all

    def test_huge_method_name(self, hipstored):  # noqa: F811
        with hipstore.Client(hipstored.socket_path) as client:
            with pytest.raises(hipstore.RpcError) as excinfo:
                client.invoke("x" * 100000)
            assert excinfo.value.code == hipstore.client.ERROR_METHOD_NOT_FOUND


class TestNvmfRobustness:
    def test_garbage_after_handshake_does_not_kill_target(self):
        backing = hs.create_malloc_bdev("robust-ns", 512, 2048)
        target = hs.start_nvmf_tcp_target("", 0, "nqn.robust", True)
        target.add_namespace(backing)
        try:
            with socket.create_connection(("127.0.0.1", target.port),
                                          timeout=10) as raw:
                raw.sendall(b"\x00" * 128)  # ICReq-sized zeros: bad type
            with socket.create_connection(("127.0.0.1", target.port),
                                          timeout=10) as raw:
                raw.sendall(b"GET / HTTP/1.0\r\n\r\n")  # not NVMe/TCP at all
            # target still serves a real initiator
            bdev = hs.create_nvmf_tcp_bdev("robust-init", "127.0.0.1",
                                           target.port, "nqn.robust")
            bdev.write(0, b"\xaa" * 512)
            assert bdev.read(0, 512) == b"\xaa" * 512
        finally:
            target.stop()

    def test_oversized_plen_drops_connection(self):
        backing = hs.create_malloc_bdev("robust-ns2", 512, 2048)
        target = hs.start_nvmf_tcp_target("", 0, "nqn.robust2", False)
        target.add_namespace(backing)
        try:
            import struct

            with socket.create_connection(("127.0.0.1", target.port),
                                          timeout=10) as raw:
                # valid ICReq first
                icreq = struct.pack("<BBBBI", 0x00, 0, 128, 0, 128)
                icreq += struct.pack("<HBBI", 0, 0, 0, 15) + b"\x00" * 112
                raw.sendall(icreq)
                icresp = raw.recv(128)
                assert len(icresp) == 128
                # then a PDU claiming a 1 GiB payload: must be rejected
                raw.sendall(struct.pack("<BBBBI", 0x04, 0, 72, 0, 1 << 30))
                raw.settimeout(10)
                assert raw.recv(16) == b""  # connection dropped
            bdev = hs.create_nvmf_tcp_bdev("robust-init2", "127.0.0.1",
                                           target.port, "nqn.robust2", 1,
                                           False)
            assert bdev.num_blocks == 2048
        finally:
            target.stop()


class TestJsonRpcFuzz:
    """Property-based fuzz of the daemon RPC socket: any byte stream
    must produce an error reply or a dropped connection — never a
    daemon crash, and the daemon must stay healthy for the next
    well-formed client."""

    def _send_raw(self, socket_path, payload: bytes) -> bytes:
        import socket as socketmod
        sock = socketmod.socket(socketmod.AF_UNIX, socketmod.SOCK_STREAM)
        sock.settimeout(5)
        sock.connect(socket_path)
        try:
            sock.sendall(payload)
            sock.shutdown(socketmod.SHUT_WR)
            out = b""
            while True:
                chunk = sock.recv(4096)
                if not chunk:
                    return out
                out += chunk
        except (TimeoutError, ConnectionError, OSError):
            return b""
        finally:
            sock.close()

    def test_random_bytes_never_kill_daemon(self, hipstored):  # noqa: F811
        from hypothesis import HealthCheck, given, settings
        from hypothesis import strategies as st

        @settings(max_examples=120, deadline=None,
                  suppress_health_check=[HealthCheck.function_scoped_fixture])
        @given(st.binary(min_size=0, max_size=512))
        def fuzz(blob):
            self._send_raw(hipstored.socket_path, blob)

        fuzz()
        # daemon is still serving well-formed requests
        with hipstore.Client(hipstored.socket_path) as client:
            assert isinstance(client.invoke("get_rpc_methods"), list)
        assert hipstored.process is None or hipstored.process.poll() is None

    def test_structured_fuzz_params(self, hipstored):  # noqa: F811
        """Valid JSON-RPC envelopes with adversarial params for every
        registered method: errors allowed, crashes not."""
        import json as jsonmod

        from hypothesis import HealthCheck, given, settings
        from hypothesis import strategies as st

        scalar = st.one_of(
            st.none(), st.booleans(),
            st.integers(min_value=-2**63, max_value=2**63 - 1),
            st.text(max_size=40), st.floats(allow_nan=False))
        params = st.dictionaries(
            st.sampled_from(["name", "size", "num_blocks", "block_size",
                             "src", "dest", "ctrlr", "bdev_name", "subnqn",
                             "scsi_target_num", "nbd_device", "devices",
                             "count", "stripe_size_kb", "io_size",
                             "queue_depth", "seconds"]),
            st.one_of(scalar, st.lists(scalar, max_size=3)), max_size=5)

        with hipstore.Client(hipstored.socket_path) as probe:
            methods = probe.invoke("get_rpc_methods")

        @settings(max_examples=150, deadline=None,
                  suppress_health_check=[HealthCheck.function_scoped_fixture])
        @given(method=st.sampled_from(sorted(methods)), p=params)
        def fuzz(method, p):
            if method.startswith("perf_") or method == "perf_run":
                return  # long-running by design; fuzzed elsewhere
            request = jsonmod.dumps({"jsonrpc": "2.0", "id": 1,
                                     "method": method, "params": p})
            self._send_raw(hipstored.socket_path, request.encode())

        fuzz()
        with hipstore.Client(hipstored.socket_path) as client:
            assert isinstance(client.invoke("get_rpc_methods"), list)


class TestNvmfFuzz:
    """Property-based fuzz of the NVMe/TCP target: arbitrary byte
    streams (raw, and framed as plausible PDUs after a real ICReq)
    must never kill the target; a real initiator still works after."""

    def test_pdu_fuzz(self):
        import struct as structmod

        from hypothesis import HealthCheck, given, settings
        from hypothesis import strategies as st

        backing = hs.create_malloc_bdev("fuzz-ns", 512, 2048)
        target = hs.start_nvmf_tcp_target("", 0, "nqn.fuzz", True)
        target.add_namespace(backing)
        try:
            @settings(max_examples=60, deadline=None,
                      suppress_health_check=[
                          HealthCheck.function_scoped_fixture])
            @given(st.binary(min_size=0, max_size=256))
            def raw_fuzz(blob):
                try:
                    with socket.create_connection(
                            ("127.0.0.1", target.port), timeout=3) as raw:
                        raw.sendall(blob)
                        raw.shutdown(socket.SHUT_WR)
                        while raw.recv(4096):
                            pass
                except OSError:
                    pass

            raw_fuzz()

            @settings(max_examples=60, deadline=None,
                      suppress_health_check=[
                          HealthCheck.function_scoped_fixture])
            @given(st.integers(min_value=0, max_value=16),
                   st.binary(max_size=200))
            def framed_fuzz(pdu_type, payload):
                # common header: type, flags, hlen, pdo, plen
                header = structmod.pack("<BBBBI", pdu_type, 0,
                                        8, 0, 8 + len(payload))
                try:
                    with socket.create_connection(
                            ("127.0.0.1", target.port), timeout=3) as raw:
                        raw.sendall(header + payload)
                        raw.shutdown(socket.SHUT_WR)
                        while raw.recv(4096):
                            pass
                except OSError:
                    pass

            framed_fuzz()
            # target still healthy
            bdev = hs.create_nvmf_tcp_bdev("fuzz-init", "127.0.0.1",
                                           target.port, "nqn.fuzz")
            bdev.write(512, b"\xbb" * 512)
            assert bdev.read(512, 512) == b"\xbb" * 512
        finally:
            target.stop()


class TestLoadConfigFuzz:
    def test_nested_config_fuzz(self, hipstored):  # noqa: F811
        """Arbitrary nested snapshot structures either apply or fail
        with a clean error; the daemon survives all of them."""
        from hypothesis import HealthCheck, given, settings
        from hypothesis import strategies as st

        scalar = st.one_of(st.none(), st.booleans(), st.integers(
            min_value=-2**40, max_value=2**40), st.text(max_size=20))
        entry = st.fixed_dictionaries(
            {}, optional={
                "method": st.one_of(scalar, st.sampled_from(
                    ["construct_malloc_bdev", "no_such", "save_config"])),
                "params": st.one_of(scalar, st.dictionaries(
                    st.sampled_from(["name", "num_blocks", "block_size"]),
                    scalar, max_size=3)),
            })
        subsystem = st.fixed_dictionaries(
            {}, optional={
                "subsystem": scalar,
                "config": st.one_of(scalar, st.lists(entry, max_size=3)),
            })
        config = st.one_of(
            scalar,
            st.fixed_dictionaries({}, optional={
                "subsystems": st.one_of(scalar,
                                        st.lists(subsystem, max_size=3))}))

        with hipstore.Client(hipstored.socket_path) as client:
            @settings(max_examples=100, deadline=None,
                      suppress_health_check=[
                          HealthCheck.function_scoped_fixture])
            @given(config)
            def fuzz(payload):
                try:
                    client.invoke("load_config",
                                  payload if isinstance(payload, dict)
                                  else {"subsystems": payload})
                except hipstore.RpcError:
                    pass

            fuzz()
            assert isinstance(client.invoke("get_rpc_methods"), list)


class TestAstralCharacters:
    """json.dumps encodes astral chars as \\uD8xx\\uDCxx surrogate
    pairs; the daemon must decode pairs (not emit CESU-8) or its
    response stream poisons strict UTF-8 clients. Regression for a
    fuzz-found connection wedge."""

    def test_emoji_name_roundtrip(self, hipstored):  # noqa: F811
        with hipstore.Client(hipstored.socket_path) as client:
            name = "vol-\U0001F999"
            client.invoke("construct_malloc_bdev",
                          {"name": name, "num_blocks": 1024,
                           "block_size": 512})
            assert client.invoke("get_bdevs",
                                 {"name": name})[0]["name"] == name
            # error-echo path: astral chars in the error message
            with pytest.raises(hipstore.RpcError):
                client.invoke("load_config", {"subsystems": [{"config": [
                    {"method": "x-\U0001F999", "params": {}}]}]})
            # the connection survives both
            client.invoke("delete_bdev", {"name": name})
            assert isinstance(client.invoke("get_rpc_methods"), list)

    def test_lone_surrogate_rejected(self, hipstored):  # noqa: F811
        import socket as socketmod
        sock = socketmod.socket(socketmod.AF_UNIX, socketmod.SOCK_STREAM)
        sock.settimeout(5)
        sock.connect(hipstored.socket_path)
        try:
            sock.sendall(b'{"jsonrpc":"2.0","id":1,"method":"\\ud800x",'
                         b'"params":{}}')
            reply = sock.recv(65536)
            assert b"parse error" in reply or reply == b""
        finally:
            sock.close()


class TestNvmfPostHandshakeFuzz:
    """Valid ICReq, then random framed PDUs into the command parser
    (the deepest reachable state without a full Fabrics connect)."""

    def test_post_handshake_pdus(self):
        import struct as structmod

        from hypothesis import HealthCheck, given, settings
        from hypothesis import strategies as st

        backing = hs.create_malloc_bdev("phfz-ns", 512, 2048)
        target = hs.start_nvmf_tcp_target("", 0, "nqn.phfz", True)
        target.add_namespace(backing)

        def icreq():
            ch = structmod.pack("<BBBBI", 0x00, 0, 128, 0, 128)
            return ch + structmod.pack("<HBBI", 0, 0, 0, 4) + bytes(112)

        try:
            @settings(max_examples=80, deadline=None,
                      suppress_health_check=[
                          HealthCheck.function_scoped_fixture])
            @given(st.lists(st.tuples(st.integers(0, 10),
                                      st.binary(max_size=200)),
                            min_size=1, max_size=3))
            def fuzz(pdus):
                with socket.create_connection(("127.0.0.1", target.port),
                                              timeout=3) as sock:
                    sock.sendall(icreq())
                    try:
                        sock.recv(128)
                        for pdu_type, payload in pdus:
                            plen = 8 + len(payload)
                            sock.sendall(structmod.pack(
                                "<BBBBI", pdu_type, 0, 8, 0, plen) + payload)
                        sock.shutdown(socket.SHUT_WR)
                        while sock.recv(4096):
                            pass
                    except OSError:
                        pass

            fuzz()
            bdev = hs.create_nvmf_tcp_bdev("phfz-init", "127.0.0.1",
                                           target.port, "nqn.phfz")
            bdev.write(0, b"\x5a" * 512)
            assert bdev.read(0, 512) == b"\x5a" * 512
        finally:
            target.stop()


class TestSetupWaitBounds:
    """A server that accepts and then goes silent must fail bdev
    construction within the setup timeout — never hang the daemon
    (bounded-everything discipline; timeout via
    HIPSTORE_{RADOS,NVMF}_SETUP_TIMEOUT, default 10 s)."""

    @pytest.fixture
    def silent_server(self):
        import socket as socket_mod

        s = socket_mod.socket()
        s.bind(("127.0.0.1", 0))
        s.listen(4)
        port = s.getsockname()[1]
        yield port
        s.close()

    def test_rados_silent_monitor_times_out(self, silent_server,
                                            monkeypatch):
        import time

        import oim_amd._hipstore as hs

        monkeypatch.setenv("HIPSTORE_RADOS_SETUP_TIMEOUT", "1")
        t0 = time.monotonic()
        with pytest.raises(RuntimeError):
            hs.create_rbd_bdev("rbd-silent", f"127.0.0.1:{silent_server}",
                               "rbd", "x", block_size=512,
                               default_size_bytes=1 << 20)
        assert time.monotonic() - t0 < 8.0

    def test_nvmf_silent_target_times_out(self, silent_server,
                                          monkeypatch):
        import time

        import oim_amd._hipstore as hs

        monkeypatch.setenv("HIPSTORE_NVMF_SETUP_TIMEOUT", "1")
        t0 = time.monotonic()
        with pytest.raises(RuntimeError):
            hs.create_nvmf_tcp_bdev("nvmf-silent", "127.0.0.1",
                                    silent_server, "nqn.silent")
        assert time.monotonic() - t0 < 8.0
