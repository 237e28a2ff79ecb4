"""CLI entry-point tests: oimctl against a live registry; flag parsing."""

import subprocess
import sys

import pytest

from oim_amd.cmd import oimctl
from oim_amd.registry import MemRegistryDB, Registry, RegistryServer


@pytest.fixture
def live_registry(tmp_path):
    registry = Registry(db=MemRegistryDB())
    server = RegistryServer(f"unix://{tmp_path}/reg.sock", registry)
    server.start()
    yield registry, server
    server.stop()


class TestOimctl:
    def test_set_get_delete(self, live_registry, capsys):
        registry, server = live_registry
        endpoint = server.addr()
        assert oimctl.main(["--registry", endpoint, "set",
                            "host-0/pci", "0000:00:15.0"]) == 0
        assert oimctl.main(["--registry", endpoint, "get"]) == 0
        out = capsys.readouterr().out
        assert "host-0/pci: 0000:00:15.0" in out
        assert oimctl.main(["--registry", endpoint, "delete", "host-0/pci"]) == 0
        assert oimctl.main(["--registry", endpoint, "get"]) == 0
        assert "host-0" not in capsys.readouterr().out


class TestMainModules:
    def test_help_screens(self):
        """Every binary parses --help (flag wiring sanity)."""
        for module in ("oim_amd.cmd.oim_registry", "oim_amd.cmd.oim_controller",
                       "oim_amd.cmd.oim_csi_driver", "oim_amd.cmd.oimctl"):
            proc = subprocess.run(
                [sys.executable, "-m", module, "--help"],
                capture_output=True, text=True, timeout=60)
            assert proc.returncode == 0, proc.stderr
            assert "usage" in proc.stdout.lower()

    def test_csi_driver_mode_validation(self):
        from oim_amd.cmd import oim_csi_driver

        with pytest.raises(SystemExit):
            oim_csi_driver.main(["--nodeid", "n1"])  # no mode selected
        with pytest.raises(SystemExit):
            oim_csi_driver.main([
                "--nodeid", "n1", "--hipstored-socket", "/x",
                "--oim-registry-address", "tcp://y:1"])  # both modes
        with pytest.raises(SystemExit):
            oim_csi_driver.main([
                "--nodeid", "n1", "--hipstored-socket", "/x",
                "--csiversion", "0.9"])  # unknown personality

    def test_csi_driver_serves_03_personality(self, tmp_path):
        """--csiversion 0.3 (reference main.go flag) serves the legacy
        twins: a csi.v0 GetPluginInfo round-trips over the wire."""
        import os as os_mod
        import subprocess
        import time as time_mod

        import fixtures
        import grpc
        from oim_amd.common.server import grpc_target
        from oim_amd.spec import csi_v0 as csi0
        from oim_amd.spec.rpc_csi0 import CSI0IdentityStub, CSI0NodeStub

        daemon = fixtures.launch_hipstored(tmp_path, cpu=True)
        endpoint = f"unix://{tmp_path}/csi03.sock"
        proc = subprocess.Popen(
            [sys.executable, "-m", "oim_amd.cmd.oim_csi_driver",
             "--nodeid", "n03", "--drivername", "oim-03",
             "--hipstored-socket", daemon.socket_path,
             "--endpoint", endpoint, "--csiversion", "0.3"],
            stderr=subprocess.DEVNULL)
        try:
            deadline = time_mod.time() + 15
            sock_path = endpoint[len("unix://"):]
            while not os_mod.path.exists(sock_path):
                assert proc.poll() is None, "driver exited early"
                assert time_mod.time() < deadline, "driver never listened"
                time_mod.sleep(0.05)
            with grpc.insecure_channel(grpc_target(endpoint)) as ch:
                info = CSI0IdentityStub(ch).GetPluginInfo(
                    csi0.GetPluginInfoRequest(), timeout=5)
                assert info.name == "oim-03"
                node_id = CSI0NodeStub(ch).NodeGetId(
                    csi0.NodeGetIdRequest(), timeout=5).node_id
                assert node_id == "n03"
        finally:
            proc.terminate()
            proc.wait(timeout=10)
            daemon.stop()


class TestOimctlVolumes:
    def test_provision_map_unmap_via_proxy(self, tmp_path, capsys):
        import fixtures
        from oim_amd.controller import Controller, ControllerServer

        daemon = fixtures.launch_hipstored(tmp_path, cpu=True)
        registry = Registry(db=MemRegistryDB())
        reg_server = RegistryServer(f"unix://{tmp_path}/r.sock", registry)
        reg_server.start()
        controller = Controller(controller_id="c9",
                                hipstored_socket=daemon.socket_path)
        ctrl_server = ControllerServer(f"unix://{tmp_path}/c.sock", controller)
        ctrl_server.start()
        registry.db.store(["c9", "address"], f"unix://{tmp_path}/c.sock")
        endpoint = reg_server.addr()
        try:
            assert oimctl.main(["--registry", endpoint, "provision",
                                "--controller", "c9", "volx", "64MiB"]) == 0
            assert oimctl.main(["--registry", endpoint, "check",
                                "--controller", "c9", "volx"]) == 0
            assert oimctl.main(["--registry", endpoint, "map",
                                "--controller", "c9", "volx"]) == 0
            out = capsys.readouterr().out
            assert "target 0 lun 0" in out
            assert oimctl.main(["--registry", endpoint, "unmap",
                                "--controller", "c9", "volx"]) == 0
            assert oimctl.main(["--registry", endpoint, "clone",
                                "--controller", "c9", "volx", "volx-c"]) == 0
            assert oimctl.main(["--registry", endpoint, "list",
                                "--controller", "c9"]) == 0
            out = capsys.readouterr().out
            assert "volx-c" in out
            assert oimctl.main(["--registry", endpoint, "stats",
                                "--controller", "c9"]) == 0
            out = capsys.readouterr().out
            assert "volx" in out and "reads=" in out
            assert oimctl.main(["--registry", endpoint, "resize",
                                "--controller", "c9", "volx", "128MiB"]) == 0
            assert oimctl.main(["--registry", endpoint, "provision",
                                "--controller", "c9", "volx-c", "0"]) == 0
            assert oimctl.main(["--registry", endpoint, "provision",
                                "--controller", "c9", "volx", "0"]) == 0
        finally:
            ctrl_server.stop()
            reg_server.stop()
            daemon.stop()

    def test_parse_size(self):
        assert oimctl.parse_size("4096") == 4096
        assert oimctl.parse_size("64MiB") == 64 << 20
        assert oimctl.parse_size("1G") == 1 << 30
        assert oimctl.parse_size("2k") == 2048


class TestOimctlTLS:
    def test_set_get_over_mtls(self, tmp_path, capsys):
        import ca as ca_util
        from oim_amd.registry import MemRegistryDB, Registry, RegistryServer

        trusted = ca_util.make_ca(str(tmp_path / "ca"))
        registry = Registry(
            db=MemRegistryDB(),
            tls=ca_util.tls_config(trusted, "component.registry"))
        server = RegistryServer(f"unix://{tmp_path}/reg.sock", registry)
        server.start()
        try:
            admin = ca_util.tls_config(trusted, "user.admin")
            argv_base = ["--registry", server.addr(),
                         "--ca", admin.ca, "--key", admin.key]
            assert oimctl.main(argv_base + ["set", "sec-0/address",
                                            "tcp://x:1"]) == 0
            assert oimctl.main(argv_base + ["get"]) == 0
            assert "sec-0/address: tcp://x:1" in capsys.readouterr().out
        finally:
            server.stop()
