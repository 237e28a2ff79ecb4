"""Kernel NBD end-to-end: HBM bdev -> /dev/nbdX -> mkfs.ext4 -> mount
-> file I/O (BASELINE config 2's host-attach story).

Marked gpu (runs on the MI355X box as root); skips when the kernel has
no nbd module (containers usually lack it)."""

import hashlib
import os
import subprocess

import pytest

from oim_amd.common.util import get_blk_size64
from oim_amd.csidriver import LocalBackend, Mounter

from fixtures import launch_hipstored

pytestmark = pytest.mark.gpu


def nbd_available():
    if not os.path.exists("/dev/nbd0"):
        subprocess.run(["modprobe", "nbd", "max_part=4"],
                       capture_output=True, timeout=60)
    return os.path.exists("/dev/nbd0")


@pytest.mark.skipif(not nbd_available(), reason="no kernel nbd module")
class TestNbdEndToEnd:
    def test_mkfs_mount_file_io(self, tmp_path):
        fixture = launch_hipstored(tmp_path, cpu=False)
        mounted = False
        mountpoint = str(tmp_path / "mnt")
        os.makedirs(mountpoint)
        try:
            backend = LocalBackend(fixture.socket_path)
            backend.create_volume("nbd-vol", 256 << 20)
            device = backend.create_device("nbd-vol", {})
            assert device.startswith("/dev/nbd")
            assert get_blk_size64(device) == 256 << 20
            mounter = Mounter()  # real mount/mkfs/blkid (we are root)
            mounter.format_and_mount(device, mountpoint, "ext4")
            mounted = True
            payload = os.urandom(8 << 20)
            path = os.path.join(mountpoint, "blob.bin")
            with open(path, "wb") as f:
                f.write(payload)
                f.flush()
                os.fsync(f.fileno())
            subprocess.run(["sync"], timeout=60)
            with open(path, "rb") as f:
                back = f.read()
            assert hashlib.sha256(back).hexdigest() == \
                hashlib.sha256(payload).hexdigest()
            mounter.unmount(mountpoint)
            mounted = False
            # Data survives a remount (it lives in HBM).
            mounter.mount(device, mountpoint, "ext4")
            mounted = True
            with open(path, "rb") as f:
                assert f.read(1 << 20) == payload[:1 << 20]
            mounter.unmount(mountpoint)
            mounted = False
            backend.delete_device("nbd-vol")
            backend.delete_volume("nbd-vol")
        finally:
            if mounted:
                subprocess.run(["umount", mountpoint], capture_output=True)
            fixture.stop()
