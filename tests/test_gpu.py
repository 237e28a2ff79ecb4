"""MI355X GPU tests: HBM bdev numerics + kernels vs CPU references.

All marked `gpu`; run via `pytest -m gpu` on a GPU box. Every data-path
result is compared against plain host-side byte operations (the CPU
reference), and CRC32C against the bit-exact software table.
"""

import os
import random

import pytest

from oim_amd import _hipstore as hs
from oim_amd import hipstore

from fixtures import launch_hipstored

pytestmark = pytest.mark.gpu

needs_gpu = pytest.mark.skipif(not hs.gpu_available(), reason="no HIP device")


@needs_gpu
class TestHbmBdev:
    @pytest.fixture(scope="class")
    def bdev(self):
        # 64 MiB HBM bdev
        return hs.create_hbm_bdev("gpu-test", 512, 131072, device=0)

    def test_roundtrip_single_block(self, bdev):
        payload = bytes(random.getrandbits(8) for _ in range(512))
        bdev.write(0, payload)
        assert bdev.read(0, 512) == payload

    def test_roundtrip_large_offsets(self, bdev):
        rng = random.Random(7)
        blob = bytes(rng.getrandbits(8) for _ in range(64 * 1024))
        offset = 32 * 1024 * 1024 + 4096
        bdev.write(offset, blob)
        assert bdev.read(offset, len(blob)) == blob
        # Neighbors untouched (bdev zero-initialized)
        assert bdev.read(offset - 512, 512) == b"\x00" * 512
        assert bdev.read(offset + len(blob), 512) == b"\x00" * 512

    def test_fill(self, bdev):
        bdev.fill(1024 * 1024, 0xCD, 8192)
        assert bdev.read(1024 * 1024, 8192) == b"\xcd" * 8192

    def test_unaligned_tail(self, bdev):
        # 512-byte granularity I/O smaller than one 4 KiB tile
        payload = bytes(random.getrandbits(8) for _ in range(1536))
        bdev.write(2048, payload)
        assert bdev.read(2048, 1536) == payload

    def test_out_of_bounds(self, bdev):
        with pytest.raises(RuntimeError):
            bdev.read(bdev.size_bytes, 512)
        with pytest.raises(RuntimeError):
            bdev.write(bdev.size_bytes - 512, b"x" * 1024)

    def test_persistence_across_channels(self, bdev):
        # Data survives independent I/O sessions (the Malloc bdev
        # contract: survives Map/Unmap, reference spec.md:116-119).
        bdev.write(4096, b"\xee" * 512)
        assert bdev.read(4096, 512) == b"\xee" * 512


@needs_gpu
class TestPersistentEngine:
    """The on-GPU polling service kernel: same numerics contract as the
    batched engine, verified against host byte references."""

    @pytest.fixture(scope="class")
    def bdev(self):
        return hs.create_hbm_bdev("gpu-pers", 512, 131072, device=0,
                                  persistent=True)

    def test_roundtrip(self, bdev):
        rng = random.Random(41)
        blob = bytes(rng.getrandbits(8) for _ in range(64 * 1024))
        bdev.write(8192, blob)
        assert bdev.read(8192, len(blob)) == blob

    def test_fill_and_small_blocks(self, bdev):
        bdev.fill(0, 0x3C, 4096)
        assert bdev.read(0, 512) == b"\x3c" * 512
        payload = bytes(random.getrandbits(8) for _ in range(512))
        bdev.write(512, payload)
        assert bdev.read(512, 512) == payload

    def test_idle_exit_and_relaunch(self, bdev):
        """The service kernel self-exits when idle (~1s) and must
        transparently relaunch on the next submission."""
        import time
        bdev.write(0, b"\x11" * 512)
        time.sleep(2.5)  # beyond the idle timeout
        bdev.write(512, b"\x22" * 512)
        assert bdev.read(0, 512) == b"\x11" * 512
        assert bdev.read(512, 512) == b"\x22" * 512

    def test_perf_sanity(self):
        bdev = hs.create_hbm_bdev("gpu-pers-perf", 4096, 262144, device=0,
                                  persistent=True)
        bdev.fill(0, 0x5A, bdev.size_bytes)
        r = hs.run_bdevperf(bdev, "randread", 4096, 32, 4, 1.0)
        assert r["iops"] > 100_000, r


@needs_gpu
class TestHbmCapacity:
    def test_quarter_terabyte_bdev(self):
        """Size for the hardware: a 256 GiB HBM bdev (of 288 GB) with
        I/O at the far end of the address space."""
        try:
            bdev = hs.create_hbm_bdev("huge", 4096, (256 << 30) // 4096,
                                      device=0)
        except RuntimeError as exc:
            pytest.skip(f"insufficient free HBM: {exc}")
        payload = bytes(random.getrandbits(8) for _ in range(4096))
        far = bdev.size_bytes - 4096
        bdev.write(far, payload)
        bdev.write(0, payload[::-1])
        assert bdev.read(far, 4096) == payload
        assert bdev.read(0, 4096) == payload[::-1]
        del bdev  # free 256 GiB before later tests


@needs_gpu
class TestCrc32c:
    def test_gpu_matches_software(self):
        bdev = hs.create_hbm_bdev("crc-test", 4096, 256, device=0)
        rng = random.Random(13)
        blocks = [bytes(rng.getrandbits(8) for _ in range(4096)) for _ in range(16)]
        for i, blk in enumerate(blocks):
            bdev.write(i * 4096, blk)
        gpu = hs.crc32c_gpu_blocks(bdev, 0, 4096, 16)
        sw = [hs.crc32c(blk) for blk in blocks]
        assert list(gpu) == sw

    def test_known_answer(self):
        assert hs.crc32c(b"123456789") == 0xE3069283


@needs_gpu
class TestPerf:
    def test_randread_sanity(self):
        bdev = hs.create_hbm_bdev("perf-gpu", 4096, 262144, device=0)  # 1 GiB
        bdev.fill(0, 0x5A, bdev.size_bytes)
        # An MI355X must beat 100k IOPS trivially; guards against a
        # silently serialized path. Warm up and allow one retry: some
        # pool boxes sit in a low-power state and the batched engine's
        # bursty launches eat multi-second clock-ramp stalls on the
        # first run (observed 6 s outliers; the always-busy persistent
        # engine is immune, and the same test is clean immediately
        # after — this is a sanity guard, not a benchmark).
        hs.run_bdevperf(bdev, "randread", 4096, 32, 4, 0.5)  # warm
        last = None
        for _ in range(2):
            last = hs.run_bdevperf(bdev, "randread", 4096, 32, 4, 1.0)
            if last["iops"] > 100_000 and last["lat_p99_us"] < 100_000:
                break
        assert last["iops"] > 100_000, last
        assert last["lat_p99_us"] < 100_000, last

    def test_randwrite_correct_and_fast(self):
        bdev = hs.create_hbm_bdev("perfw-gpu", 4096, 65536, device=0)
        hs.run_bdevperf(bdev, "randwrite", 4096, 32, 2, 0.3)  # warm
        last = None
        for _ in range(2):
            last = hs.run_bdevperf(bdev, "randwrite", 4096, 32, 2, 0.5)
            if last["iops"] > 50_000:
                break
        assert last["iops"] > 50_000, last


@needs_gpu
class TestCompositeGpu:
    def test_striped_hbm_correctness(self):
        # Two HBM children (same device on a 1-GPU box; stripe logic is
        # identical across devices).
        children = [hs.create_hbm_bdev(f"sg-{i}", 512, 65536, device=0)
                    for i in range(2)]
        bdev = hs.create_striped_bdev("sg", children, 65536)
        rng = random.Random(11)
        data = bytes(rng.getrandbits(8) for _ in range(4 * 65536))
        bdev.write(0, data)
        assert bdev.read(0, len(data)) == data
        # stripe unit 1 lives on child 1 at offset 0
        assert children[1].read(0, 65536) == data[65536:2 * 65536]
        r = hs.run_bdevperf(bdev, "randread", 4096, 32, 2, 0.5)
        assert r["iops"] > 50_000, r

    @pytest.mark.skipif(hs.gpu_device_count() < 2,
                        reason="needs >=2 GPUs for xGMI replication")
    def test_replicated_xgmi_fanout(self):
        children = [hs.create_hbm_bdev(f"rg-{i}", 512, 65536, device=i)
                    for i in range(2)]
        bdev = hs.create_replicated_bdev("rg", children)
        rng = random.Random(13)
        data = bytes(rng.getrandbits(8) for _ in range(16 * 4096))
        bdev.write(8192, data)
        # Every replica holds the data (replica 1 was filled over xGMI
        # peer copy from device 0).
        for child in children:
            assert child.read(8192, len(data)) == data

    @pytest.mark.skipif(hs.gpu_device_count() < 3,
                        reason="needs >=3 GPUs for RCCL broadcast fan-out")
    def test_replicated_rccl_broadcast(self):
        """>=3 replicas use ncclBroadcast over xGMI for the fan-out."""
        n = min(hs.gpu_device_count(), 4)
        children = [hs.create_hbm_bdev(f"rc-{i}", 512, 65536, device=i)
                    for i in range(n)]
        bdev = hs.create_replicated_bdev("rc", children)
        rng = random.Random(17)
        data = bytes(rng.getrandbits(8) for _ in range(64 * 4096))
        bdev.write(4096, data)
        for child in children:
            assert child.read(4096, len(data)) == data


@needs_gpu
class TestHbmCopy:
    def test_same_device_copy(self):
        a = hs.create_hbm_bdev("cp-a", 4096, 16384, device=0)  # 64 MiB
        b = hs.create_hbm_bdev("cp-b", 4096, 16384, device=0)
        rng = random.Random(51)
        blob = bytes(rng.getrandbits(8) for _ in range(1 << 20))
        a.write(4096, blob)
        hs.hbm_copy(a, 4096, b, 8192, len(blob))
        assert b.read(8192, len(blob)) == blob
        # sub-tile granularity (16-byte aligned lengths); read back a
        # whole block (bdev reads are block-granular)
        hs.hbm_copy(a, 4096, b, 0, 512)
        assert b.read(0, 4096)[:512] == blob[:512]

    def test_rejects_cpu_bdev(self):
        a = hs.create_hbm_bdev("cp-c", 4096, 1024, device=0)
        c = hs.create_malloc_bdev("cp-d", 4096, 1024)
        with pytest.raises(RuntimeError):
            hs.hbm_copy(a, 0, c, 0, 4096)

    @pytest.mark.skipif(hs.gpu_device_count() < 2,
                        reason="needs >=2 GPUs for xGMI copy")
    def test_cross_device_copy(self):
        a = hs.create_hbm_bdev("cp-x0", 4096, 16384, device=0)
        b = hs.create_hbm_bdev("cp-x1", 4096, 16384, device=1)
        rng = random.Random(53)
        blob = bytes(rng.getrandbits(8) for _ in range(1 << 20))
        a.write(0, blob)
        hs.hbm_copy(a, 0, b, 0, len(blob))
        assert b.read(0, len(blob)) == blob


@needs_gpu
class TestNvmfGpu:
    def test_hbm_namespace_gpu_digest_roundtrip(self):
        """NVMe/TCP loopback with an HBM-resident namespace: C2HData
        digests for aligned >=16 KiB reads are produced by the GPU
        CRC32C kernel + host combine; the initiator verifies them in
        software, so a mismatch anywhere fails the read."""
        backing = hs.create_hbm_bdev("nvmf-hbm", 4096, 16384, device=0)
        target = hs.start_nvmf_tcp_target(
            "", 0, "nqn.2026-01.com.amd:gpu-ns", True)
        target.add_namespace(backing)
        try:
            bdev = hs.create_nvmf_tcp_bdev(
                "nvmf-gpu", "127.0.0.1", target.port,
                "nqn.2026-01.com.amd:gpu-ns")
            rng = random.Random(31)
            data = bytes(rng.getrandbits(8) for _ in range(128 * 1024))
            bdev.write(0, data)
            # 128 KiB aligned read -> GPU-digested C2HData
            assert bdev.read(0, len(data)) == data
            assert backing.read(0, len(data)) == data
            r = hs.run_bdevperf(bdev, "randread", 65536, 8, 2, 0.5)
            assert r["iops"] > 100, r
        finally:
            target.stop()


@needs_gpu
class TestDaemonHbm:
    def test_daemon_gpu_mode(self, tmp_path):
        fixture = launch_hipstored(tmp_path, cpu=False)
        try:
            with hipstore.Client(fixture.socket_path) as client:
                name = hipstore.construct_malloc_bdev(
                    client, num_blocks=262144, block_size=4096, name="hbm0")
                bdevs = hipstore.get_bdevs(client, name)
                assert bdevs[0].product_name == "Malloc disk"
                assert "hbm" in bdevs[0].driver_specific
                pci = bdevs[0].driver_specific["hbm"]["pci_address"]
                assert pci.count(":") == 2
                result = hipstore.perf_run(client, "hbm0", io_size=4096,
                                           queue_depth=32, num_queues=4,
                                           seconds=1.0)
                assert result["iops"] > 100_000, result
                hipstore.delete_bdev(client, name)
        finally:
            fixture.stop()


@needs_gpu
class TestCapacityReporting:
    def test_hbm_info(self):
        total, free = hs.hbm_info(0)
        assert total > 200 << 30  # 288 GB class
        assert 0 < free <= total

    def test_csi_get_capacity_local_mode(self, tmp_path):
        import grpc

        from oim_amd.common.server import grpc_target
        from oim_amd.csidriver import FakeExec, LocalBackend, Mounter, OIMDriver
        from oim_amd.spec import csi_v1 as csi
        from oim_amd.spec.rpc_csi import CSIControllerStub

        fixture = launch_hipstored(tmp_path, cpu=False)
        driver = OIMDriver(driver_name="oim-local", node_id="n0",
                           endpoint=f"unix://{tmp_path}/csi.sock",
                           backend=LocalBackend(fixture.socket_path),
                           mounter=Mounter(FakeExec()))
        driver.start()
        try:
            with grpc.insecure_channel(grpc_target(driver.addr())) as ch:
                stub = CSIControllerStub(ch)
                response = stub.GetCapacity(csi.GetCapacityRequest(), timeout=30)
                assert response.available_capacity > 100 << 30
                caps = stub.ControllerGetCapabilities(
                    csi.ControllerGetCapabilitiesRequest(), timeout=30)
                types = {c.rpc.type for c in caps.capabilities}
                assert csi.CTRL_CAP_GET_CAPACITY in types
        finally:
            driver.stop()
            fixture.stop()


@needs_gpu
class TestInvalidWorkload:
    def test_unaligned_io_size_fails_cleanly(self):
        """io_size below the block size must error out, not hang
        (regression: failed resubmissions mutated the immediate list
        mid-iteration)."""
        bdev = hs.create_hbm_bdev("inv-gpu", 4096, 65536, device=0,
                                  persistent=True)
        with pytest.raises(RuntimeError):
            hs.run_bdevperf(bdev, "randread", 512, 8, 2, 0.5)


@needs_gpu
class TestChannelAutoFallback:
    """Past the per-device HW-queue cap (18 per-queue service kernels),
    new channels transparently fall back to the batched engine; IO on
    a mixed per-queue + batched channel set stays correct.

    Round-1's "mixed-engine wedge" was a HOST bug, not a GPU mystery:
    HbmBdev::poll() dispatched on bdev-level flags, so fallback
    channels were polled through the wrong class (batched-on-
    persistent read garbage and stalled; shared fallbacks segfaulted).
    poll() now dispatches on the channel's kind like submit() always
    did, and these run by default again. Every wait is bounded, so a
    regression costs seconds, not the suite budget.
    """

    @pytest.mark.timeout(120)
    def test_low_cap_mixed_engines(self, monkeypatch):
        """4 per-queue + 2 batched via HIPSTORE_PERQ_CAP=4 — exercises
        the fallback decision and mixed-kind dispatch with small
        resident-kernel counts."""
        import subprocess, sys, os
        script = os.path.join(os.path.dirname(__file__), "_qsweep_debug.py")
        env = dict(os.environ, HIPSTORE_PERQ_CAP="4")
        proc = subprocess.run([sys.executable, script, "6"], env=env,
                              capture_output=True, text=True, timeout=100)
        assert proc.returncode == 0, proc.stdout + proc.stderr
        assert "ios=" in proc.stdout

    @pytest.mark.timeout(180)
    def test_22_queues_mixed_engines(self):
        bdev = hs.create_hbm_bdev("fallback-0", 4096, 262144, device=0,
                                  persistent=True)
        result = hs.run_bdevperf(bdev, "randread", 4096, 8, 22, 20.0,
                                 max_ios=200000)
        assert result["io_count"] >= 200000
        del bdev, result
        # shared-fallback flavour at a tiny cap (the round-1
        # segfaulting config), isolated in its own process
        import subprocess
        import sys as sys_mod
        script = os.path.join(os.path.dirname(__file__),
                              "_qsweep_debug.py")
        env = dict(os.environ, HIPSTORE_PERQ_CAP="4",
                   HIPSTORE_FALLBACK="shared")
        proc = subprocess.run([sys_mod.executable, script, "6"],
                              env=env, capture_output=True, text=True,
                              timeout=100)
        assert proc.returncode == 0, proc.stdout + proc.stderr
        bdev = hs.create_hbm_bdev("fallback-1", 4096, 262144, device=0,
                                  persistent=True)
        result = hs.run_bdevperf(bdev, "randread", 4096, 8, 2, 5.0,
                                 max_ios=20000)
        assert result["io_count"] >= 20000
        # correctness across the cap boundary
        import random
        rng = random.Random(11)
        data = bytes(rng.getrandbits(8) for _ in range(8 * 4096))
        bdev.write(64 * 4096, data)
        assert bdev.read(64 * 4096, len(data)) == data


@needs_gpu
class TestHbmResize:
    """Offline HBM bdev expansion: data survives the backing-store
    move; live channels refuse the resize (kIoFailed)."""

    def test_grow_preserves_data(self):
        import os as _os
        bdev = hs.create_hbm_bdev("resize-0", 512, 2048, device=0,
                                  persistent=True)
        data = _os.urandom(4096)
        bdev.write(0, data)
        session = hs.PerfSession(bdev, "randread", 512, 1, 1)
        session.step(10)
        assert bdev.resize(65536) == -5  # busy: channels hold the base
        del session
        assert bdev.resize(65536) == 0
        assert bdev.num_blocks == 65536
        assert bdev.read(0, 4096) == data  # moved with the store
        tail = _os.urandom(512)
        bdev.write(65535 * 512, tail)  # new extent is addressable
        assert bdev.read(65535 * 512, 512) == tail
        # grown region arrived zeroed
        assert bdev.read(4096, 512) == b"\0" * 512


@pytest.mark.gpu
class TestEngineContractProbes:
    """Hardware-contract probes (round-2 diagnosis tooling, kept as
    regression guards): every leg of the host<->GPU polling contract,
    and the real service kernel serving a descriptor from a minimal
    standalone harness. A red here pinpoints WHICH leg broke (e.g. the
    firmware __threadfence_system wave-hang this caught)."""

    def test_polling_contract_legs(self):
        r = hs.persistent_probe(0, 1)
        assert r["launch_err"] == 0
        assert r["hb_host_early"] > 0, "GPU->host pinned heartbeat"
        assert r["hb_dev_early"] > 0, "kernel runs at all"
        assert r["cq0_ms"] >= 0, "host tail -> leader -> CQ publish"
        assert r["cq1_ms"] >= 0, "leader -> worker agent relay"
        assert r["stream_drained"] == 1

    @pytest.mark.parametrize("variant", [0, 1, 2])
    def test_service_kernel_serves(self, variant):
        r = hs.persistent_kernel_probe(0, variant)
        assert r["launch_err"] == 0
        assert r["cq_ms"] >= 0, "service kernel must publish its CQ"
        if variant != 1:
            assert r["data_ok"] == 1
        assert r["dev_known"] == 1
        assert r["stream_drained"] == 1
