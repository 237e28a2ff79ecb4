"""bench.py contract tests (CPU): single-process and 2-rank gloo runs
must emit the JSON line the driver parses."""

import json
import os
import subprocess
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BENCH = os.path.join(REPO_ROOT, "bench.py")

REQUIRED_KEYS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


def last_json_line(output: str) -> dict:
    for line in reversed(output.strip().splitlines()):
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{output}")


@pytest.mark.skipif(not os.path.exists(os.path.join(REPO_ROOT, "bin", "hipstored")),
                    reason="hipstored not built")
class TestBenchContract:
    def test_single_process(self):
        proc = subprocess.run(
            [sys.executable, BENCH, "--steps", "2", "--warmup", "1"],
            capture_output=True, text=True, timeout=600, cwd=REPO_ROOT)
        assert proc.returncode == 0, proc.stderr[-2000:]
        result = last_json_line(proc.stdout)
        assert REQUIRED_KEYS.issubset(result.keys())
        assert result["metric"] == "4KiB_randread_IOPS"
        assert result["value"] > 0
        assert result["n_gpus"] == 1
        assert result["steps"] == 2
        assert result["config"]["p99_us"] > 0
        assert result["scaling"] == "weak"

    def test_two_rank_gloo(self):
        env = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT="29517")
        proc = subprocess.run(
            [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
             "--master-port", "29517", BENCH,
             "--gpus", "2", "--steps", "2", "--warmup", "0"],
            capture_output=True, text=True, timeout=600, cwd=REPO_ROOT,
            env=env)
        assert proc.returncode == 0, proc.stderr[-2000:]
        result = last_json_line(proc.stdout)
        assert result["n_gpus"] == 2
        # The reported value must be the WHOLE-JOB aggregate:
        # value [IOPS] x elapsed [s] == 2 ranks x steps x STEP_IOS.
        expected_ios = 2 * 2 * result["config"]["step_ios_per_gpu"]
        measured_ios = result["value"] * result["ms_per_step"] \
            * result["steps"] / 1000
        assert abs(measured_ios - expected_ios) / expected_ios < 0.01

    def test_vhost_frontend_single_process(self):
        proc = subprocess.run(
            [sys.executable, BENCH, "--steps", "1", "--warmup", "0",
             "--frontend", "vhost", "--queue-depth", "8",
             "--vhost-numjobs", "2", "--bdev-gb", "0.25"],
            capture_output=True, text=True, timeout=600, cwd=REPO_ROOT)
        assert proc.returncode == 0, proc.stderr[-2000:]
        result = last_json_line(proc.stdout)
        assert result["config"]["frontend"].startswith("vhost-user-scsi")
        assert result["value"] > 0
        assert result["config"]["p99_us"] > 0


class TestVhostHarness:
    def test_cli_run(self, tmp_path):
        import fixtures
        from oim_amd import hipstore
        from oim_amd.bench import vhost_harness

        daemon = fixtures.launch_hipstored(tmp_path, cpu=True)
        try:
            with hipstore.Client(daemon.socket_path) as client:
                hipstore.construct_malloc_bdev(
                    client, num_blocks=16384, block_size=512, name="vhb")
            import contextlib, io
            out = io.StringIO()
            with contextlib.redirect_stdout(out):
                rc = vhost_harness.main([
                    "--socket", daemon.socket_path, "--bdev", "vhb",
                    "--rw", "randrw", "--bs", "4096", "--iodepth", "8",
                    "--runtime", "0.3", "--perfdash"])
            assert rc == 0
            text = out.getvalue()
            assert "IOPS=" in text
            assert "[Finished:Performance]" in text
            # controller cleaned up: socket name free for a rerun
            with hipstore.Client(daemon.socket_path) as client:
                assert not [c for c in hipstore.get_vhost_controllers(client)
                            if c.controller == "vhost-bench"]
            # blk personality path
            out = io.StringIO()
            with contextlib.redirect_stdout(out):
                rc = vhost_harness.main([
                    "--socket", daemon.socket_path, "--bdev", "vhb",
                    "--personality", "blk", "--bs", "4096",
                    "--iodepth", "8", "--runtime", "0.3"])
            assert rc == 0 and "IOPS=" in out.getvalue()
        finally:
            daemon.stop()

    def test_pipelined_daemon_qd32(self, tmp_path, monkeypatch):
        """Deep-queue run against the pipelined ring worker."""
        import contextlib
        import io

        import fixtures
        from oim_amd import hipstore
        from oim_amd.bench import vhost_harness

        monkeypatch.setenv("HIPSTORE_VHOST_PIPELINE", "1")
        daemon = fixtures.launch_hipstored(tmp_path, cpu=True)
        try:
            with hipstore.Client(daemon.socket_path) as client:
                hipstore.construct_malloc_bdev(
                    client, num_blocks=32768, block_size=512, name="vhp")
            out = io.StringIO()
            with contextlib.redirect_stdout(out):
                rc = vhost_harness.main([
                    "--socket", daemon.socket_path, "--bdev", "vhp",
                    "--rw", "randrw", "--bs", "4096", "--iodepth", "32",
                    "--numjobs", "3", "--runtime", "0.5"])
            assert rc == 0 and "IOPS=" in out.getvalue()
        finally:
            daemon.stop()
