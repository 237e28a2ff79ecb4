"""ublk host attach: a bdev as a real kernel block device.

Reference role: lib/nbd/nbd.c + local.go's createDevice — the pool's
kernels ship ublk_drv (no nbd module), so ublk is the path that makes
NodeStage yield a real /dev node on a GPU box. CPU CI has no
ublk-control either, so the wire surface is tested for its error
contract here and the data path under ``-m gpu``."""

import os

import pytest

from oim_amd import hipstore

from fixtures import hipstored, launch_hipstored  # noqa: F401


def _ublk_available():
    if os.path.exists("/dev/ublk-control"):
        return True
    try:
        with open("/proc/misc") as f:
            return "ublk-control" in f.read()
    except OSError:
        return False


def _ublk_blocked_reason():
    """None when /dev/ublk-control is usable; else why not (the node
    is mknod'd from /proc/misc first — no udev in the containers)."""
    import stat as stat_mod
    path = "/dev/ublk-control"
    if not os.path.exists(path):
        try:
            with open("/proc/misc") as f:
                minor = next((int(line.split()[0]) for line in f
                              if "ublk-control" in line), None)
        except OSError as e:
            return f"/proc/misc: {e}"
        if minor is None:
            return "ublk driver absent"
        try:
            os.mknod(path, 0o600 | stat_mod.S_IFCHR, os.makedev(10, minor))
        except OSError as e:
            return f"mknod: {e}"
    try:
        os.close(os.open(path, os.O_RDWR))
    except OSError as e:
        return f"open: {e}"
    return None


class TestUblkSurface:
    def test_unavailable_is_clean_error(self, hipstored):  # noqa: F811
        if _ublk_available():
            pytest.skip("ublk present; covered by the gpu e2e test")
        with hipstore.Client(hipstored.socket_path) as client:
            hipstore.construct_malloc_bdev(client, 2048, 512, name="ub0")
            with pytest.raises(hipstore.RpcError):
                client.invoke("ublk_start_disk", {"bdev_name": "ub0"})
            assert client.invoke("ublk_get_disks") == []
            # stopping a never-started device is INVALID_PARAMS
            with pytest.raises(hipstore.RpcError):
                client.invoke("ublk_stop_disk", {"dev_id": 0})

    def test_missing_bdev(self, hipstored):  # noqa: F811
        with hipstore.Client(hipstored.socket_path) as client:
            with pytest.raises(hipstore.RpcError) as excinfo:
                client.invoke("ublk_start_disk", {"bdev_name": "ghost"})
            assert "does not exist" in str(excinfo.value)


@pytest.mark.gpu
class TestUblkE2E:
    """Real /dev/ublkbN backed by HBM: the round-2 host-attach goal
    ('NodeStage on a GPU box yields a real /dev node')."""

    def test_block_device_roundtrip(self, tmp_path):
        if not _ublk_available():
            pytest.skip("kernel has no ublk support")
        reason = _ublk_blocked_reason()
        if reason:
            pytest.skip(f"ublk blocked in this environment: {reason}")
        daemon = launch_hipstored(tmp_path, cpu=False)
        try:
            with hipstore.Client(daemon.socket_path) as client:
                hipstore.construct_malloc_bdev(
                    client, num_blocks=32768, block_size=512, name="ubg0")
                disk = client.invoke("ublk_start_disk",
                                     {"bdev_name": "ubg0"})
                dev = disk["device"]
                assert os.path.exists(dev), f"{dev} missing"
                # bdev is claimed while exported
                assert hipstore.get_bdevs(client, "ubg0")[0].claimed
                payload = os.urandom(64 * 1024)
                fd = os.open(dev, os.O_RDWR | os.O_DIRECT)
                try:
                    import mmap as mmap_mod
                    buf = mmap_mod.mmap(-1, len(payload))
                    buf[:] = payload
                    os.lseek(fd, 4096, os.SEEK_SET)
                    assert os.write(fd, buf) == len(payload)
                    os.fsync(fd)
                    os.lseek(fd, 4096, os.SEEK_SET)
                    got = os.read(fd, len(payload))
                    assert got == payload
                finally:
                    os.close(fd)
                listed = client.invoke("ublk_get_disks")
                assert [d["device"] for d in listed] == [dev]
                client.invoke("ublk_stop_disk", {"dev_id": disk["dev_id"]})
                assert client.invoke("ublk_get_disks") == []
                assert not hipstore.get_bdevs(client, "ubg0")[0].claimed
                hipstore.delete_bdev(client, "ubg0")
        finally:
            daemon.stop()

    def test_local_backend_create_device_uses_ublk(self, tmp_path):
        if not _ublk_available():
            pytest.skip("kernel has no ublk support")
        reason = _ublk_blocked_reason()
        if reason:
            pytest.skip(f"ublk blocked in this environment: {reason}")
        from oim_amd.csidriver.local import LocalBackend

        daemon = launch_hipstored(tmp_path, cpu=False)
        try:
            backend = LocalBackend(daemon.socket_path)
            backend.create_volume("ublk-vol", 16 << 20)
            dev = backend.create_device("ublk-vol", {})
            assert dev.startswith("/dev/ublkb") and os.path.exists(dev)
            # idempotent: same device on re-create
            assert backend.create_device("ublk-vol", {}) == dev
            backend.delete_device("ublk-vol")
            backend.delete_volume("ublk-vol")
        finally:
            daemon.stop()

    def test_nodestage_yields_mounted_ext4(self, tmp_path):
        """The reference proves its device path by a guest seeing sda
        (controller_test.go:306-340). The achievable equivalent here:
        CSI NodeStageVolume on a GPU box produces a REAL mounted ext4
        on /dev/ublkbN backed by HBM, with file I/O through the page
        cache hitting the engine."""
        if not _ublk_available():
            pytest.skip("kernel has no ublk support")
        reason = _ublk_blocked_reason()
        if reason:
            pytest.skip(f"ublk blocked in this environment: {reason}")
        if os.geteuid() != 0:
            pytest.skip("needs root for mkfs/mount")
        import grpc

        from oim_amd.common.server import grpc_target
        from oim_amd.csidriver import LocalBackend, OIMDriver
        from oim_amd.spec import csi_v1 as csi
        from oim_amd.spec.rpc_csi import CSIControllerStub, CSINodeStub

        daemon = launch_hipstored(tmp_path, cpu=False)
        driver = None
        staging = str(tmp_path / "staging")
        try:
            backend = LocalBackend(daemon.socket_path)
            driver = OIMDriver(driver_name="oim-malloc", node_id="gpu-e2e",
                               endpoint=f"unix://{tmp_path}/csi.sock",
                               backend=backend)
            driver.start()
            with grpc.insecure_channel(
                    grpc_target(f"unix://{tmp_path}/csi.sock")) as ch:
                ctrl = CSIControllerStub(ch)
                node = CSINodeStub(ch)
                req = csi.CreateVolumeRequest(name="pvc-ublk")
                req.capacity_range.required_bytes = 64 << 20
                cap = req.volume_capabilities.add()
                cap.mount.fs_type = "ext4"
                cap.access_mode.mode = csi.ACCESS_MODE_SINGLE_NODE_WRITER
                ctrl.CreateVolume(req, timeout=60)
                stage = csi.NodeStageVolumeRequest(
                    volume_id="pvc-ublk", staging_target_path=staging)
                stage.volume_capability.mount.fs_type = "ext4"
                stage.volume_capability.access_mode.mode = \
                    csi.ACCESS_MODE_SINGLE_NODE_WRITER
                node.NodeStageVolume(stage, timeout=120)
                assert os.path.ismount(staging)
                probe = os.path.join(staging, "hello.bin")
                payload = os.urandom(1 << 20)
                with open(probe, "wb") as f:
                    f.write(payload)
                    f.flush()
                    os.fsync(f.fileno())
                with open(probe, "rb") as f:
                    assert f.read() == payload
                node.NodeUnstageVolume(
                    csi.NodeUnstageVolumeRequest(
                        volume_id="pvc-ublk",
                        staging_target_path=staging), timeout=60)
                assert not os.path.ismount(staging)
                ctrl.DeleteVolume(
                    csi.DeleteVolumeRequest(volume_id="pvc-ublk"),
                    timeout=30)
        finally:
            if driver is not None:
                driver.stop()
            daemon.stop()
