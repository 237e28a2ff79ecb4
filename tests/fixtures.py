"""Test fixtures: launch a hipstored daemon (CPU mode by default).

Counterpart of the reference's test/pkg/spdk fixture (spdk.go:84-226),
without the sudo/hugepage machinery SPDK needed: hipstored runs as a
plain process. Env overrides:
  TEST_HIPSTORED_BINARY  path to the daemon (default: <repo>/bin/hipstored)
  TEST_HIPSTORED_SOCKET  attach to an already-running daemon instead
"""

from __future__ import annotations

import os
import subprocess
import time
from typing import Optional

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
DEFAULT_BINARY = os.path.join(REPO_ROOT, "bin", "hipstored")


class HipstoredFixture:
    def __init__(self, socket_path: str, process: Optional[subprocess.Popen]):
        self.socket_path = socket_path
        self.process = process

    def stop(self):
        if self.process is not None:
            self.process.terminate()
            try:
                self.process.wait(timeout=10)
            except subprocess.TimeoutExpired:
                self.process.kill()
                self.process.wait()
            self.process = None


def launch_hipstored(tmp_path, cpu: bool = True, device: int = 0) -> HipstoredFixture:
    existing = os.environ.get("TEST_HIPSTORED_SOCKET")
    if existing:
        return HipstoredFixture(existing, None)
    binary = os.environ.get("TEST_HIPSTORED_BINARY", DEFAULT_BINARY)
    if not os.path.exists(binary):
        pytest.skip(f"hipstored binary not built: {binary} (run make)")
    socket_path = str(tmp_path / "hipstored.sock")
    cmd = [binary, "-S", socket_path, "-d", str(device)]
    if cpu:
        cmd.append("-C")
    env = dict(os.environ)
    env.setdefault("GPU_MAX_HW_QUEUES", "24")
    # stderr goes to a file, not a PIPE: an undrained pipe back-
    # pressures the daemon after 64 KiB of diagnostics and wedges it.
    stderr_path = str(tmp_path / "hipstored.stderr")
    stderr_file = open(stderr_path, "w")  # noqa: SIM115 (daemon lifetime)
    process = subprocess.Popen(cmd, stderr=stderr_file, env=env)
    deadline = time.time() + 30
    while not os.path.exists(socket_path):
        if process.poll() is not None:
            err = open(stderr_path).read()
            pytest.fail(f"hipstored exited early: {err}")
        if time.time() > deadline:
            process.kill()
            pytest.fail("hipstored did not create its socket in 30s")
        time.sleep(0.05)
    return HipstoredFixture(socket_path, process)


@pytest.fixture
def hipstored(tmp_path):
    fixture = launch_hipstored(tmp_path, cpu=True)
    yield fixture
    fixture.stop()
