"""RADOS wire path: msgr-v1 client + loopback fake cluster.

Covers the from-scratch RADOS protocol pair (native/src/rados_client.cpp
/ rados_cluster.cpp) that replaces the reference's librados/librbd RBD
bdev (reference vendor/.../lib/bdev/rbd/bdev_rbd.c): messenger
handshake + framing CRCs, RBD object layout, sparse-read semantics,
image-header geometry, error paths, and the daemon RPC surface
(rados_cluster_start / construct_rbd_bdev with mon_host)."""

import random
import socket
import struct

import pytest

import oim_amd._hipstore as hs
from oim_amd import hipstore

from fixtures import hipstored  # noqa: F401  (fixture)


@pytest.fixture(scope="module")
def cluster():
    c = hs.start_rados_cluster(port=0, arena_mb=96, use_hbm=False,
                               device=0, object_bytes=1 << 20)
    yield c
    c.stop()


def make_image(cluster, image, size_mb=16, name=None):
    return hs.create_rbd_bdev(name or f"rbd-{image}",
                              f"127.0.0.1:{cluster.port()}", "rbd", image,
                              block_size=512,
                              default_size_bytes=size_mb << 20,
                              object_bytes=1 << 20)


class TestRbdWire:
    def test_roundtrip_within_object(self, cluster):
        b = make_image(cluster, "img-rt")
        blob = bytes(random.Random(7).getrandbits(8) for _ in range(8192))
        b.write(4096, blob)
        assert b.read(4096, len(blob)) == blob

    def test_object_boundary_crossing(self, cluster):
        b = make_image(cluster, "img-xo")
        blob = bytes(random.Random(8).getrandbits(8) for _ in range(300032))
        off = (1 << 20) - 4096  # spans data objects 0 and 1
        b.write(off, blob)
        assert b.read(off, len(blob)) == blob

    def test_sparse_reads_are_zero(self, cluster):
        b = make_image(cluster, "img-sparse")
        assert b.read(5 << 20, 4096) == b"\0" * 4096
        # partially written object: the unwritten tail reads back zero
        b.write(2 << 20, b"\xaa" * 512)
        assert b.read(2 << 20, 4096) == b"\xaa" * 512 + b"\0" * 3584

    def test_zero_fill(self, cluster):
        b = make_image(cluster, "img-fill")
        b.write(0, b"\xff" * 8192)
        b.fill(0, 0, 4096)
        assert b.read(0, 8192) == b"\0" * 4096 + b"\xff" * 4096

    def test_geometry_from_header_object(self, cluster):
        make_image(cluster, "img-geo", size_mb=24)
        # Re-open with a DIFFERENT default: the stored header wins.
        again = make_image(cluster, "img-geo", size_mb=4, name="rbd-geo2")
        assert again.size_bytes == 24 << 20

    def test_data_survives_reopen(self, cluster):
        b = make_image(cluster, "img-persist")
        blob = bytes(random.Random(9).getrandbits(8) for _ in range(4096))
        b.write(1 << 20, blob)
        del b
        again = make_image(cluster, "img-persist", name="rbd-persist2")
        assert again.read(1 << 20, 4096) == blob

    def test_concurrent_queues(self, cluster):
        b = make_image(cluster, "img-perf", size_mb=8)
        r = hs.run_bdevperf(b, "randrw", 4096, 8, 2, 1.0, max_ios=4000)
        assert r["io_count"] >= 4000

    def test_out_of_bounds_rejected(self, cluster):
        b = make_image(cluster, "img-oob", size_mb=4)
        with pytest.raises(RuntimeError):
            b.write((4 << 20) - 512, b"\0" * 4096)

    def test_arena_exhaustion_is_an_error(self):
        # 4 one-MiB slots; a 16 MiB image cannot fully materialize.
        c = hs.start_rados_cluster(port=0, arena_mb=4, use_hbm=False,
                                   device=0, object_bytes=1 << 20)
        try:
            b = hs.create_rbd_bdev("rbd-full", f"127.0.0.1:{c.port()}",
                                   "rbd", "img-full", block_size=512,
                                   default_size_bytes=16 << 20,
                                   object_bytes=1 << 20)
            with pytest.raises(RuntimeError):
                for i in range(16):  # > 3 data slots
                    b.write(i << 20, b"\x11" * 4096)
        finally:
            c.stop()

    def test_connection_refused(self):
        with pytest.raises(RuntimeError):
            hs.create_rbd_bdev("rbd-nc", "127.0.0.1:1", "rbd", "x",
                               block_size=512,
                               default_size_bytes=1 << 20)


class TestMsgrProtocol:
    """Wire-level checks against the cluster's messenger endpoint."""

    def test_banner_and_addr_exchange(self, cluster):
        s = socket.create_connection(("127.0.0.1", cluster.port()), 5)
        try:
            banner = s.recv(9)
            assert banner == b"ceph v027"
            # server entity_addr (136 B) + peer-as-seen (136 B)
            rest = b""
            while len(rest) < 272:
                chunk = s.recv(272 - len(rest))
                assert chunk
                rest += chunk
            family_be = struct.unpack(">H", rest[8:10])[0]
            assert family_be == socket.AF_INET
        finally:
            s.close()

    def test_bad_banner_drops_connection(self, cluster):
        s = socket.create_connection(("127.0.0.1", cluster.port()), 5)
        try:
            got = b""
            while len(got) < 9 + 272:  # banner + both entity addrs
                chunk = s.recv(9 + 272 - len(got))
                assert chunk
                got += chunk
            s.sendall(b"not ceph!" + b"\0" * 300)
            s.settimeout(5)
            # Server must drop the session: FIN (empty read) or RST
            # (it closed with our surplus bytes still unread).
            try:
                assert s.recv(4096) == b""
            except ConnectionResetError:
                pass
        finally:
            s.close()

    def test_corrupt_header_crc_rejected(self, cluster):
        """A frame whose header CRC is wrong must not be executed."""
        b = make_image(cluster, "img-crc")
        b.write(0, b"\x77" * 4096)
        before = cluster.object_count()
        # Handshake as a client, then send a TAG_MSG with a bad CRC.
        s = socket.create_connection(("127.0.0.1", cluster.port()), 5)
        try:
            s.recv(9)
            buf = b""
            while len(buf) < 272:
                buf += s.recv(272 - len(buf))
            s.sendall(b"ceph v027" + b"\0" * 136)
            connect = struct.pack("<QIIIIIIB", 0, 8, 1, 1, 24, 0, 0, 0)
            s.sendall(connect)
            reply = b""
            while len(reply) < 26:
                chunk = s.recv(26 - len(reply))
                assert chunk
                reply += chunk
            assert reply[0] == 1  # TAG_READY
            header = bytearray(53)
            header[16] = 42  # MSG_OSD_OP, everything else garbage/zero
            struct.pack_into("<I", header, 49, 0xDEADBEEF)  # wrong crc
            s.sendall(b"\x07" + bytes(header))
            s.settimeout(5)
            assert s.recv(4096) == b""  # server dropped the session
        finally:
            s.close()
        assert cluster.object_count() == before


class TestRbdDaemonRpc:
    """construct_rbd_bdev over JSON-RPC with mon_host → wire path."""

    def test_csi_shaped_flow(self, hipstored):  # noqa: F811
        with hipstore.Client(hipstored.socket_path) as client:
            info = client.invoke("rados_cluster_start",
                                 {"arena_mb": 32, "object_mb": 1})
            try:
                name = hipstore.construct_rbd_bdev(
                    client, pool_name="rbd", rbd_name="vol1",
                    block_size=512, name="ceph-vol1", user_id="admin",
                    config={"mon_host": info["mon_host"],
                            "key": "secret", "emu_size_mb": 8,
                            "object_mb": 1})
                assert name == "ceph-vol1"
                bdev = hipstore.get_bdevs(client, "ceph-vol1")[0]
                assert bdev.product_name == "Ceph Rbd Disk"
                assert bdev.num_blocks * bdev.block_size == 8 << 20
                # real data through daemon channels over the wire
                r = hipstore.perf_run(client, "ceph-vol1", io_size=4096,
                                      queue_depth=4, num_queues=1,
                                      seconds=0.2, workload="randwrite")
                assert r["io_count"] > 0
                hipstore.delete_bdev(client, "ceph-vol1")
            finally:
                client.invoke("rados_cluster_stop",
                              {"port": info["port"]})

    def test_bad_mon_host_is_invalid_params(self, hipstored):  # noqa: F811
        with hipstore.Client(hipstored.socket_path) as client:
            with pytest.raises(hipstore.RpcError):
                hipstore.construct_rbd_bdev(
                    client, pool_name="rbd", rbd_name="x", block_size=512,
                    config={"mon_host": "127.0.0.1:1"})


@pytest.mark.gpu
class TestRadosGpu:
    """HBM arena + GPU CRC path of the fake cluster."""

    def test_hbm_arena_roundtrip_and_gpu_crc(self):
        if not hs.gpu_available():
            pytest.skip("no GPU")
        c = hs.start_rados_cluster(port=0, arena_mb=64, use_hbm=True,
                                   device=0, object_bytes=1 << 20)
        try:
            b = hs.create_rbd_bdev("rbd-hbm", f"127.0.0.1:{c.port()}",
                                   "rbd", "img-hbm", block_size=512,
                                   default_size_bytes=16 << 20,
                                   object_bytes=1 << 20)
            blob = bytes(random.Random(3).getrandbits(8)
                         for _ in range(64 * 1024))
            off = (1 << 20) - 8192
            b.write(off, blob)       # small op: SSE4.2 verify on readback
            assert b.read(off, len(blob)) == blob
            # Whole-object aligned op (>= 1 MiB): the GPU per-4KiB CRC
            # kernel verifies the landed extent / hashes the reply.
            big = bytes(random.Random(4).getrandbits(8)
                        for _ in range(1 << 20))
            b.write(4 << 20, big)
            assert b.read(4 << 20, len(big)) == big
            r = hs.run_bdevperf(b, "randrw", 4096, 8, 2, 1.0,
                                max_ios=2000)
            assert r["io_count"] >= 2000
        finally:
            c.stop()


class TestRadosConfigReplay:
    def test_cluster_and_rbd_survive_snapshot_replay(self, tmp_path):
        """save_config captures the loopback cluster (with its bound
        port) BEFORE the bdev subsystem, so an rbd bdev whose mon_host
        points at it reconnects on `load_config` in a fresh daemon."""
        import pathlib

        import fixtures as fx

        dir_a = pathlib.Path(tmp_path) / "a"
        dir_a.mkdir()
        daemon = fx.launch_hipstored(dir_a)
        try:
            with hipstore.Client(daemon.socket_path) as client:
                info = client.invoke("rados_cluster_start",
                                     {"arena_mb": 16, "object_mb": 1})
                hipstore.construct_rbd_bdev(
                    client, pool_name="rbd", rbd_name="imgr",
                    block_size=512, name="rbd-replay",
                    config={"mon_host": info["mon_host"],
                            "emu_size_mb": 4, "object_mb": 1})
                config = client.invoke("save_config")
        finally:
            daemon.stop()
        # The cluster died with the daemon; replaying in a new daemon
        # restarts it on the SAME port and reconnects the bdev.
        dir_b = pathlib.Path(tmp_path) / "b"
        dir_b.mkdir()
        daemon2 = fx.launch_hipstored(dir_b)
        try:
            with hipstore.Client(daemon2.socket_path) as client:
                client.invoke("load_config", config)
                bdev = hipstore.get_bdevs(client, "rbd-replay")[0]
                assert bdev.product_name == "Ceph Rbd Disk"
                assert bdev.num_blocks * bdev.block_size == 4 << 20
                r = hipstore.perf_run(client, "rbd-replay", io_size=4096,
                                      queue_depth=2, num_queues=1,
                                      seconds=0.2, workload="randwrite")
                assert r["io_count"] > 0
        finally:
            daemon2.stop()


class TestClusterDeathMidWorkload:
    def test_run_fails_fast_when_cluster_stops(self, monkeypatch):
        """Stopping the cluster under a running workload must surface
        an I/O failure within the bounded-wait window — dead
        connections fail inflight ops (fail_all), they never hang the
        perf loop."""
        import threading
        import time

        monkeypatch.setenv("HIPSTORE_SYNC_TIMEOUT_S", "5")
        c = hs.start_rados_cluster(port=0, arena_mb=32, use_hbm=False,
                                   device=0, object_bytes=1 << 20)
        b = make_image(c, "img-die", size_mb=8, name="rbd-die")
        b.write(0, b"\x5a" * 4096)
        killer = threading.Timer(0.3, c.stop)
        killer.start()
        t0 = time.monotonic()
        try:
            with pytest.raises(RuntimeError):
                hs.run_bdevperf(b, "randread", 4096, 8, 2, 30.0)
        finally:
            killer.join()
        assert time.monotonic() - t0 < 25.0  # failed fast, not 30 s
