"""Mount/format helpers (counterpart of the reference's pkg/mount fork
of k8s mount-utils: SafeFormatAndMount, mount_linux.go:432-517).

Commands run through an injectable executor so unit tests use FakeExec
(the reference's exec.go pattern) and a non-root CI never needs sudo.
"""

from __future__ import annotations

import os
import subprocess
from typing import List, Optional, Sequence


class ExecError(RuntimeError):
    def __init__(self, cmd: Sequence[str], returncode: int, output: str):
        super().__init__(f"{' '.join(cmd)} failed ({returncode}): {output}")
        self.cmd = list(cmd)
        self.returncode = returncode
        self.output = output


class OsExec:
    """Runs commands for real."""

    def run(self, *cmd: str) -> str:
        proc = subprocess.run(cmd, capture_output=True, text=True)
        if proc.returncode != 0:
            raise ExecError(cmd, proc.returncode, proc.stderr or proc.stdout)
        return proc.stdout


class FakeExec:
    """Records commands; scripted outputs (reference exec.go FakeExec)."""

    def __init__(self):
        self.calls: List[List[str]] = []
        self.outputs = {}  # cmd[0] -> str or Exception

    def run(self, *cmd: str) -> str:
        self.calls.append(list(cmd))
        result = self.outputs.get(cmd[0], "")
        if isinstance(result, Exception):
            raise result
        return result


class Mounter:
    def __init__(self, execer: Optional[OsExec] = None):
        self.exec = execer or OsExec()

    def is_mount_point(self, path: str) -> bool:
        """Reference IsLikelyNotMountPoint (mount_linux.go:240): a mount
        point has a different device than its parent."""
        try:
            st = os.lstat(path)
            parent = os.lstat(os.path.dirname(path.rstrip("/")) or "/")
        except OSError:
            return False
        return st.st_dev != parent.st_dev

    def mount(self, source: str, target: str, fstype: str = "",
              options: Sequence[str] = ()) -> None:
        cmd = ["mount"]
        if fstype:
            cmd += ["-t", fstype]
        if options:
            cmd += ["-o", ",".join(options)]
        cmd += [source, target]
        self.exec.run(*cmd)

    def bind_mount(self, source: str, target: str, readonly: bool = False) -> None:
        options = ["bind"]
        self.exec.run("mount", "-o", ",".join(options), source, target)
        if readonly:
            # bind mounts need a remount to become ro (mount_linux.go).
            self.exec.run("mount", "-o", "bind,remount,ro", source, target)

    def unmount(self, target: str) -> None:
        self.exec.run("umount", target)

    def get_fs_type(self, device: str) -> str:
        """blkid probe; "" for an unformatted device (mount_linux.go:432+)."""
        try:
            out = self.exec.run(
                "blkid", "-p", "-s", "TYPE", "-s", "PTTYPE", "-o", "export",
                device)
        except ExecError as err:
            if err.returncode == 2:  # blkid: nothing found
                return ""
            raise
        for line in out.splitlines():
            if line.startswith("TYPE="):
                return line.split("=", 1)[1].strip()
        return ""

    def resize_fs(self, device: str, fstype: str = "ext4") -> None:
        """Grow the filesystem to the (already grown) device
        (NodeExpandVolume)."""
        if fstype.startswith("ext"):
            self.exec.run("resize2fs", device)
        elif fstype == "xfs":
            self.exec.run("xfs_growfs", device)
        else:
            raise ValueError(f"cannot grow filesystem type {fstype!r}")

    def format_and_mount(self, device: str, target: str, fstype: str = "ext4",
                         options: Sequence[str] = ()) -> None:
        """SafeFormatAndMount: probe, mkfs when unformatted, mount."""
        current = self.get_fs_type(device)
        if current == "":
            mkfs = [f"mkfs.{fstype}"]
            if fstype.startswith("ext"):
                # -F: don't prompt when the target is a whole device.
                mkfs += ["-F", "-m0"]
            mkfs.append(device)
            self.exec.run(*mkfs)
        elif current != fstype and fstype:
            raise RuntimeError(
                f"device {device} already formatted {current}, want {fstype}")
        self.mount(device, target, fstype, options)
