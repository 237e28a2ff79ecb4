"""Aux subsystem tests: CmdMonitor, LogWriter, perftype emitter,
fio harness CLI."""

import io
import json
import subprocess
import sys
import threading


from oim_amd import log
from oim_amd.bench.perftype import (
    DataItem,
    PerfData,
    emit_perf_data,
    perf_result_to_data_item,
)
from oim_amd.common.util import CmdMonitor, LogWriter

from fixtures import hipstored  # noqa: F401


class TestCmdMonitor:
    def test_detects_exit(self):
        exited = threading.Event()
        codes = []

        def on_exit(code):
            codes.append(code)
            exited.set()

        process = subprocess.Popen([sys.executable, "-c", "import sys; sys.exit(3)"])
        CmdMonitor(process, on_exit)
        assert exited.wait(timeout=30)
        assert codes == [3]


class TestLogWriter:
    def test_line_buffering(self):
        t = log.TestLogger()
        writer = LogWriter(logger=t, prefix="child: ")
        writer.write("hello ")
        writer.write("world\npartial")
        assert t.messages() == ["child: hello world"]
        writer.flush()
        assert t.messages() == ["child: hello world", "child: partial"]


class TestPerftype:
    def test_emit_shape(self):
        out = io.StringIO()
        perf = PerfData(
            version="v1",
            data_items=[DataItem(data={"iops": 100.0}, unit="mixed",
                                 labels={"bs": "4096"})],
            labels={"suite": "x"},
        )
        emit_perf_data(perf, out)
        line = out.getvalue()
        assert line.startswith("[Finished:Performance] ")
        payload = json.loads(line.split(" ", 1)[1])
        assert payload["dataItems"][0]["data"]["iops"] == 100.0
        assert payload["labels"] == {"suite": "x"}

    def test_result_mapping(self):
        result = {"iops": 1, "throughput_mbps": 2, "lat_avg_us": 3,
                  "lat_p50_us": 4, "lat_p99_us": 5, "lat_p999_us": 6}
        item = perf_result_to_data_item(result, {"a": "b"})
        assert item.data["lat_p99_us"] == 5
        assert item.labels == {"a": "b"}


class TestFioHarness:
    def test_cli_run(self, hipstored, capsys):  # noqa: F811
        from oim_amd import hipstore
        from oim_amd.bench import fio_harness

        with hipstore.Client(hipstored.socket_path) as client:
            hipstore.construct_malloc_bdev(client, 8192, 512, name="fio0")
        rc = fio_harness.main([
            "--socket", hipstored.socket_path, "--bdev", "fio0",
            "--rw", "randread", "--bs", "4096", "--iodepth", "4",
            "--numjobs", "1", "--runtime", "0.2", "--perfdash"])
        assert rc == 0
        out = capsys.readouterr().out
        assert "IOPS=" in out
        assert "[Finished:Performance]" in out


class TestPayloadFormatters:
    """gRPC payload logging (reference pkg/oim-common/tracing.go +
    protosanitizer StripSecrets): secrets must never reach logs, and
    the interceptors must pass calls through untouched."""

    def _msg(self):
        from oim_amd.spec import csi_v1 as csi

        return csi.NodeStageVolumeRequest(
            volume_id="vol-1",
            staging_target_path="/mnt/stage",
            secrets={"admin": "hunter2"})

    def test_strip_secrets_redacts(self):
        from oim_amd.common.tracing import strip_secrets_formatter

        out = strip_secrets_formatter(self._msg())
        assert "hunter2" not in out
        assert "secrets=<redacted>" in out
        assert "vol-1" in out  # non-secret fields still logged

    def test_complete_formatter_includes_everything(self):
        from oim_amd.common.tracing import complete_formatter

        out = complete_formatter(self._msg())
        assert "hunter2" in out and "vol-1" in out

    def test_null_formatter_and_non_proto(self):
        from oim_amd.common.tracing import (null_formatter,
                                            strip_secrets_formatter)

        assert null_formatter(self._msg()) == "<omitted>"
        assert strip_secrets_formatter("plain") == "'plain'"

    def test_server_interceptor_passthrough(self, tmp_path):
        """A servicer behind LogServerInterceptor(strip_secrets) must
        behave identically; the log line carries the redacted form."""
        import grpc

        from oim_amd.common.server import NonBlockingGRPCServer
        from oim_amd.common.tracing import (LogServerInterceptor,
                                            strip_secrets_formatter)
        from oim_amd.spec import csi_v1 as csi
        from oim_amd.spec.rpc_csi import (CSIIdentityServicer,
                                          CSIIdentityStub,
                                          add_csi_identity_to_server)

        class Identity(CSIIdentityServicer):
            def GetPluginInfo(self, request, context):
                return csi.GetPluginInfoResponse(name="t", vendor_version="1")

        endpoint = f"unix://{tmp_path}/trace.sock"
        server = NonBlockingGRPCServer(
            endpoint,
            interceptors=[LogServerInterceptor(strip_secrets_formatter)])
        server.start(lambda srv: add_csi_identity_to_server(Identity(),
                                                             srv))
        try:
            with grpc.insecure_channel(
                    f"unix:{tmp_path}/trace.sock") as channel:
                stub = CSIIdentityStub(channel)
                resp = stub.GetPluginInfo(csi.GetPluginInfoRequest())
            assert resp.name == "t"
        finally:
            server.stop()
