"""Unit tests for oim_amd.common (PCI, paths, endpoints, server)."""


import grpc
import pytest

from oim_amd import spec
from oim_amd.common import (
    NonBlockingGRPCServer,
    PCIAddress,
    RegistryPathError,
    complete_pci_address,
    join_registry_path,
    parse_bdf_string,
    parse_endpoint,
    pretty_pci_address,
    split_registry_path,
)
from oim_amd.common.pci import UNSET
from oim_amd.common.server import grpc_target


class TestPCI:
    def test_parse_full(self):
        a = parse_bdf_string("0000:03:1f.6")
        assert (a.domain, a.bus, a.device, a.function) == (0, 3, 0x1F, 6)

    def test_parse_no_domain(self):
        a = parse_bdf_string("00:15.0")
        assert a.domain == UNSET
        assert (a.bus, a.device, a.function) == (0, 0x15, 0)

    def test_parse_invalid(self):
        for bad in ("", "xx", "00:15", "00.15.0", "12345:00:15.0"):
            with pytest.raises(ValueError):
                parse_bdf_string(bad)

    def test_complete_merges_and_defaults_domain(self):
        partial = PCIAddress(bus=5, device=1, function=0)
        fallback = parse_bdf_string("0002:07:00.1")
        merged = complete_pci_address(partial, fallback)
        assert (merged.domain, merged.bus, merged.device, merged.function) == (2, 5, 1, 0)
        # domain unset in both -> 0 (reference remote.go:186-189)
        merged = complete_pci_address(PCIAddress(bus=1, device=2, function=3), PCIAddress())
        assert merged.domain == 0

    def test_pretty(self):
        assert pretty_pci_address(parse_bdf_string("0000:00:15.0")) == "0000:00:15.0"
        assert pretty_pci_address(PCIAddress(bus=0x15)) == "****:15:**.*"


class TestPaths:
    def test_split(self):
        assert split_registry_path("/a//b/c/") == ["a", "b", "c"]

    def test_reject_dots(self):
        for bad in ("a/../b", ".", "a/."):
            with pytest.raises(RegistryPathError):
                split_registry_path(bad)

    def test_join(self):
        assert join_registry_path(["host-0", "address"]) == "host-0/address"
        with pytest.raises(RegistryPathError):
            join_registry_path(["a/b"])


class TestEndpoint:
    def test_parse(self):
        assert parse_endpoint("unix:///tmp/x.sock") == ("unix", "/tmp/x.sock")
        assert parse_endpoint("tcp://0.0.0.0:8999") == ("tcp", "0.0.0.0:8999")
        with pytest.raises(ValueError):
            parse_endpoint("http://x")
        with pytest.raises(ValueError):
            parse_endpoint("unix://")

    def test_target(self):
        assert grpc_target("unix:///tmp/x.sock") == "unix:/tmp/x.sock"
        assert grpc_target("tcp://127.0.0.1:1234") == "127.0.0.1:1234"


class _EchoRegistry(spec.RegistryServicer):
    def GetValues(self, request, context):
        return spec.GetValuesReply(
            values=[spec.Value(path=request.path, value="ok")]
        )


class TestServer:
    def test_unix_socket_lifecycle(self, tmp_sock):
        server = NonBlockingGRPCServer(endpoint=f"unix://{tmp_sock}")
        server.start(lambda s: spec.add_registry_to_server(_EchoRegistry(), s))
        try:
            with grpc.insecure_channel(f"unix:{tmp_sock}") as channel:
                stub = spec.RegistryStub(channel)
                reply = stub.GetValues(spec.GetValuesRequest(path="x"), timeout=5)
                assert reply.values[0].value == "ok"
        finally:
            server.stop()

    def test_tcp_ephemeral_addr(self):
        server = NonBlockingGRPCServer(endpoint="tcp://127.0.0.1:0")
        server.start(lambda s: spec.add_registry_to_server(_EchoRegistry(), s))
        try:
            addr = server.addr()
            assert not addr.endswith(":0")
            host_port = addr[len("tcp://"):]
            with grpc.insecure_channel(host_port) as channel:
                stub = spec.RegistryStub(channel)
                reply = stub.GetValues(spec.GetValuesRequest(path="y"), timeout=5)
                assert reply.values[0].path == "y"
        finally:
            server.stop()

    def test_stale_unix_socket_removed(self, tmp_sock):
        open(tmp_sock, "w").close()
        server = NonBlockingGRPCServer(endpoint=f"unix://{tmp_sock}")
        server.start(lambda s: spec.add_registry_to_server(_EchoRegistry(), s))
        server.stop()


class TestLog:
    def test_levels_and_fields(self):
        from oim_amd import log

        t = log.TestLogger()
        logger = t.with_fields(component="x")
        logger.info("hello %s", "world", extra=1)
        level, msg, fields = t.records[0]
        assert msg == "hello world"
        assert fields == {"component": "x", "extra": 1}

    def test_formatter(self):
        from oim_amd.log import Formatter, Level

        line = Formatter(show_time=False).format(Level.INFO, "msg", {"b": 1, "a": 2})
        assert line == "INFO  msg a=2 b=1"

    def test_context(self):
        import contextvars

        from oim_amd import log

        t = log.TestLogger()

        def run():
            log.with_logger(t)
            log.from_context().info("inner")

        ctx = contextvars.copy_context()
        ctx.run(run)
        assert t.messages() == ["inner"]
        assert log.from_context() is log.L()


class TestCrc32cHardwarePath:
    """crc32c_sw auto-selects the SSE4.2 instruction; the byte table
    stays the bit-exactness reference (reference lib/util/crc32c.c has
    the same instruction/table split)."""

    def test_hw_matches_table_across_sizes(self):
        import os as _os

        import oim_amd._hipstore as hs

        for n in (0, 1, 7, 8, 9, 63, 512, 4096, 65536, 100001):
            blob = _os.urandom(n)
            assert hs.crc32c(blob, 0) == hs.crc32c_table(blob, 0), n
        # chained init values agree too
        blob = _os.urandom(9000)
        a = hs.crc32c(blob[:1234], 0)
        assert hs.crc32c(blob[1234:], a) == hs.crc32c_table(blob[1234:], a)

    def test_known_answer(self):
        import oim_amd._hipstore as hs

        assert hs.crc32c(b"123456789", 0) == 0xE3069283

    def test_combine_matches_whole_buffer_crc(self):
        """GF(2) combine ladder: crc(a||b) from crc(a), crc(b), len(b).
        This is what folds the GPU per-4KiB block CRCs into one
        NVMe/TCP DDGST / RADOS data CRC, so it must agree with the
        straight-line CRC for arbitrary split points."""
        import os as _os
        import random as _random

        import oim_amd._hipstore as hs

        rng = _random.Random(42)
        blob = _os.urandom(50000)
        whole = hs.crc32c(blob, 0)
        for _ in range(20):
            cut = rng.randrange(0, len(blob) + 1)
            a, b = blob[:cut], blob[cut:]
            got = hs.crc32c_combine(hs.crc32c(a, 0), hs.crc32c(b, 0), len(b))
            assert got == whole, cut
        # zero-length pieces on either side are identities
        assert hs.crc32c_combine(whole, hs.crc32c(b"", 0), 0) == whole
        assert hs.crc32c_combine(hs.crc32c(b"", 0), whole, len(blob)) == whole
        # many-piece fold at 4 KiB granularity (the GPU kernel's shape)
        crc = hs.crc32c(blob[:4096], 0)
        for off in range(4096, len(blob), 4096):
            piece = blob[off:off + 4096]
            crc = hs.crc32c_combine(crc, hs.crc32c(piece, 0), len(piece))
        assert crc == whole
