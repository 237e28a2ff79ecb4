"""Striped / replicated composite bdevs (CPU children; the xGMI
peer-copy path is covered by gpu-marked tests in test_gpu.py)."""

import random

import pytest

from oim_amd import _hipstore as hs
from oim_amd import hipstore

from fixtures import hipstored  # noqa: F401

BLOCK = 512
STRIPE = 4096  # small stripe so tests cross boundaries quickly


def make_children(n, blocks=256):
    return [hs.create_malloc_bdev(f"child-{random.random()}", BLOCK, blocks)
            for _ in range(n)]


class TestStriped:
    def test_size_and_geometry(self):
        children = make_children(4)
        bdev = hs.create_striped_bdev("s0", children, STRIPE)
        assert bdev.size_bytes == 4 * 256 * BLOCK
        assert bdev.product_name == "Striped Malloc disk"

    def test_stripe_mapping(self):
        """A write spanning several stripe units lands on the right
        children at the right child offsets."""
        children = make_children(2)
        bdev = hs.create_striped_bdev("s1", children, STRIPE)
        rng = random.Random(3)
        data = bytes(rng.getrandbits(8) for _ in range(4 * STRIPE))
        bdev.write(0, data)
        # unit u goes to child u%2 at child offset (u//2)*STRIPE
        for unit in range(4):
            child = children[unit % 2]
            child_off = (unit // 2) * STRIPE
            expect = data[unit * STRIPE:(unit + 1) * STRIPE]
            assert child.read(child_off, STRIPE) == expect, f"unit {unit}"
        assert bdev.read(0, len(data)) == data

    def test_unaligned_span(self):
        children = make_children(3)
        bdev = hs.create_striped_bdev("s2", children, STRIPE)
        rng = random.Random(5)
        data = bytes(rng.getrandbits(8) for _ in range(7 * BLOCK))
        offset = STRIPE - 2 * BLOCK  # crosses a stripe boundary mid-write
        bdev.write(offset, data)
        assert bdev.read(offset, len(data)) == data

    def test_fill_and_bounds(self):
        children = make_children(2)
        bdev = hs.create_striped_bdev("s3", children, STRIPE)
        bdev.fill(0, 0x77, bdev.size_bytes)
        assert bdev.read(STRIPE, BLOCK) == b"\x77" * BLOCK
        with pytest.raises(RuntimeError):
            bdev.read(bdev.size_bytes, BLOCK)

    def test_stripe_size_validation(self):
        children = make_children(2)
        with pytest.raises(RuntimeError):
            hs.create_striped_bdev("bad", children, 100)  # not block multiple


class TestReplicated:
    def test_mirrored_writes(self):
        children = make_children(3)
        bdev = hs.create_replicated_bdev("r0", children)
        assert bdev.size_bytes == 256 * BLOCK
        rng = random.Random(7)
        data = bytes(rng.getrandbits(8) for _ in range(8 * BLOCK))
        bdev.write(2 * BLOCK, data)
        for child in children:
            assert child.read(2 * BLOCK, len(data)) == data
        assert bdev.read(2 * BLOCK, len(data)) == data

    def test_fill_mirrored(self):
        children = make_children(2)
        bdev = hs.create_replicated_bdev("r1", children)
        bdev.fill(0, 0xEE, 16 * BLOCK)
        for child in children:
            assert child.read(0, BLOCK) == b"\xee" * BLOCK


class TestCompositeRpc:
    def test_striped_via_daemon(self, hipstored):  # noqa: F811
        with hipstore.Client(hipstored.socket_path) as client:
            name = client.invoke("construct_striped_malloc_bdev", {
                "name": "stripe0", "num_blocks": 2048, "block_size": 512,
                "stripe_size_kb": 64, "count": 4})
            assert name == "stripe0"
            bdevs = hipstore.get_bdevs(client, "stripe0")
            assert bdevs[0].product_name == "Striped Malloc disk"
            assert bdevs[0].num_blocks == 4 * 2048
            result = hipstore.perf_run(client, "stripe0", io_size=4096,
                                       queue_depth=8, num_queues=1,
                                       seconds=0.2)
            assert result["io_count"] > 0
            hipstore.delete_bdev(client, "stripe0")

    def test_replicated_via_daemon(self, hipstored):  # noqa: F811
        with hipstore.Client(hipstored.socket_path) as client:
            client.invoke("construct_replicated_malloc_bdev", {
                "name": "repl0", "num_blocks": 2048, "block_size": 512,
                "count": 2})
            bdevs = hipstore.get_bdevs(client, "repl0")
            assert bdevs[0].product_name == "Replicated Malloc disk"
            assert bdevs[0].num_blocks == 2048
            hipstore.delete_bdev(client, "repl0")


class TestPerfEdgeCases:
    def test_session_zero_and_tiny_steps(self):
        b = hs.create_malloc_bdev(f"edge-{random.random()}", 512, 8192)
        s = hs.PerfSession(b, "randread", 4096, 8, 4)
        r = s.step(0)
        assert r["io_count"] == 0
        r = s.step(3)  # fewer IOs than queues
        assert 0 < r["io_count"] <= 4 * 3
        r = s.step(1000)
        assert r["io_count"] >= 1000

    def test_bdevperf_max_ios_cap(self):
        b = hs.create_malloc_bdev(f"edge2-{random.random()}", 512, 8192)
        r = hs.run_bdevperf(b, "randread", 4096, 8, 2, 60.0, max_ios=500)
        assert 500 <= r["io_count"] <= 600  # cap honored, not the 60s
        assert r["seconds"] < 10


class TestCloneOfComposite:
    def test_clone_striped_into_malloc(self, hipstored):  # noqa: F811
        """bdev_clone reads any Bdev (striped included) and produces a
        plain malloc clone with identical content."""
        with hipstore.Client(hipstored.socket_path) as client:
            client.invoke("construct_striped_malloc_bdev", {
                "name": "clsrc-st", "num_blocks": 1024, "block_size": 512,
                "stripe_size_kb": 64, "count": 2})
            result = hipstore.perf_run(client, "clsrc-st", io_size=4096,
                                       queue_depth=4, num_queues=1,
                                       seconds=0.2, workload="randwrite")
            assert result["io_count"] > 0
            client.invoke("bdev_clone", {"src": "clsrc-st",
                                         "name": "clst-copy"})
            copy = hipstore.get_bdevs(client, "clst-copy")[0]
            src = hipstore.get_bdevs(client, "clsrc-st")[0]
            assert copy.num_blocks == src.num_blocks
            assert copy.product_name == "Striped Malloc disk"  # re-badged
            hipstore.delete_bdev(client, "clst-copy")
            hipstore.delete_bdev(client, "clsrc-st")
