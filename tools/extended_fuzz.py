"""Extended fuzz sweeps (longer-running than the in-suite versions in
tests/test_protocol_robustness.py / tests/test_vhost.py; same shapes).
The in-suite 100-example config fuzz found a real bug (JSON surrogate
pair decoding wedging client streams) — rerun these periodically with
fresh seeds.

    python tools/extended_fuzz.py [config|vhost|nvmf|rados|all] [examples]

Each surface runs against its own fresh CPU-mode daemon/target and
asserts the server stays healthy afterwards.
"""

import json as jsonmod
import os
import pathlib
import socket as socketmod
import struct
import sys
import tempfile

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tests"))

from hypothesis import HealthCheck, given, settings  # noqa: E402
from hypothesis import strategies as st  # noqa: E402

SUPPRESS = [HealthCheck.function_scoped_fixture]


def fuzz_config(examples: int) -> None:
    import fixtures
    from oim_amd import hipstore

    scalar = st.one_of(
        st.none(), st.booleans(),
        st.integers(min_value=-2**62, max_value=2**62),
        st.text(max_size=40),
        st.floats(allow_nan=False, allow_infinity=False))
    entry = st.fixed_dictionaries({}, optional={
        "method": st.one_of(scalar, st.sampled_from(
            ["construct_malloc_bdev", "construct_aio_bdev", "bdev_clone",
             "bdev_copy", "resize_malloc_bdev", "save_config",
             "load_config", "construct_striped_malloc_bdev", "no_such"])),
        "params": st.one_of(scalar, st.dictionaries(
            st.sampled_from(["name", "num_blocks", "block_size", "src",
                             "dst", "size", "filename", "count",
                             "stripe_size_kb", "subsystems"]),
            scalar, max_size=5)),
    })
    subsystem = st.fixed_dictionaries({}, optional={
        "subsystem": scalar,
        "config": st.one_of(scalar, st.lists(entry, max_size=4)),
    })
    config = st.one_of(scalar, st.fixed_dictionaries({}, optional={
        "subsystems": st.one_of(scalar, st.lists(subsystem, max_size=4))}))

    with tempfile.TemporaryDirectory() as d:
        daemon = fixtures.launch_hipstored(pathlib.Path(d), cpu=True)
        client = hipstore.Client(daemon.socket_path, timeout=10)

        @settings(max_examples=examples, deadline=None,
                  suppress_health_check=SUPPRESS)
        @given(config)
        def run(payload):
            try:
                client.invoke("load_config",
                              payload if isinstance(payload, dict)
                              else {"subsystems": payload})
            except hipstore.RpcError:
                pass

        try:
            run()
            assert isinstance(client.invoke("get_rpc_methods"), list)
            print(f"config fuzz: {examples} examples clean")
        finally:
            client.close()
            daemon.stop()


def fuzz_vhost(examples: int) -> None:
    import fixtures
    from oim_amd import hipstore
    from vhost_client import VhostUserMaster

    with tempfile.TemporaryDirectory() as d:
        daemon = fixtures.launch_hipstored(pathlib.Path(d), cpu=True)
        client = hipstore.Client(daemon.socket_path)
        client.invoke("construct_vhost_scsi_controller", {"ctrlr": "xfz"})
        path = os.path.join(os.path.dirname(daemon.socket_path), "xfz")

        @settings(max_examples=examples, deadline=None,
                  suppress_health_check=SUPPRESS)
        @given(st.lists(st.tuples(st.integers(0, 40),
                                  st.binary(max_size=280)), max_size=4))
        def run(messages):
            sock = socketmod.socket(socketmod.AF_UNIX,
                                    socketmod.SOCK_STREAM)
            sock.settimeout(3)
            try:
                sock.connect(path)
                for request, payload in messages:
                    sock.sendall(struct.pack("<III", request, 0x1,
                                             len(payload)) + payload)
                sock.shutdown(socketmod.SHUT_WR)
                while sock.recv(4096):
                    pass
            except OSError:
                pass
            finally:
                sock.close()

        try:
            run()
            master = VhostUserMaster(path)
            master.negotiate()
            master.close()
            print(f"vhost fuzz: {examples} streams clean")
        finally:
            client.invoke("remove_vhost_controller", {"ctrlr": "xfz"})
            client.close()
            daemon.stop()


def fuzz_nvmf(examples: int) -> None:
    import oim_amd._hipstore as hs

    backing = hs.create_malloc_bdev("xnfz", 512, 2048)
    target = hs.start_nvmf_tcp_target("", 0, "nqn.xnfz", True)
    target.add_namespace(backing)

    def icreq():
        ch = struct.pack("<BBBBI", 0x00, 0, 128, 0, 128)
        return ch + struct.pack("<HBBI", 0, 0, 0, 4) + bytes(112)

    @settings(max_examples=examples, deadline=None,
              suppress_health_check=SUPPRESS)
    @given(st.lists(st.tuples(st.integers(0, 10), st.binary(max_size=200)),
                    min_size=1, max_size=3))
    def run(pdus):
        with socketmod.create_connection(("127.0.0.1", target.port),
                                         timeout=3) as sock:
            sock.sendall(icreq())
            try:
                sock.recv(128)
                for pdu_type, payload in pdus:
                    plen = 8 + len(payload)
                    sock.sendall(struct.pack("<BBBBI", pdu_type, 0, 8, 0,
                                             plen) + payload)
                sock.shutdown(socketmod.SHUT_WR)
                while sock.recv(4096):
                    pass
            except OSError:
                pass

    try:
        run()
        bdev = hs.create_nvmf_tcp_bdev("xnfz-init", "127.0.0.1",
                                       target.port, "nqn.xnfz")
        bdev.write(0, b"\x5a" * 512)
        assert bdev.read(0, 512) == b"\x5a" * 512
        print(f"nvmf fuzz: {examples} sessions clean")
    finally:
        target.stop()


def fuzz_rados(examples: int) -> None:
    """Hostile msgr-v1 sessions against the loopback cluster: random
    handshake garbage, then (on a well-handshaken session) random
    tagged frames with corrupt headers/fronts/CRCs. The cluster must
    drop bad sessions and keep serving good ones."""
    import oim_amd._hipstore as hs

    cluster = hs.start_rados_cluster(port=0, arena_mb=16, use_hbm=False,
                                     device=0, object_bytes=1 << 20)

    def handshake(sock):
        got = b""
        while len(got) < 9 + 272:
            chunk = sock.recv(9 + 272 - len(got))
            if not chunk:
                raise OSError("early close")
            got += chunk
        sock.sendall(b"ceph v027" + bytes(136))
        sock.sendall(struct.pack("<QIIIIIIB", 0, 8, 1, 1, 24, 0, 0, 0))
        reply = b""
        while len(reply) < 26:
            chunk = sock.recv(26 - len(reply))
            if not chunk:
                raise OSError("early close")
            reply += chunk

    @settings(max_examples=examples, deadline=None,
              suppress_health_check=SUPPRESS)
    @given(st.booleans(),
           st.lists(st.tuples(st.integers(0, 20), st.binary(max_size=300)),
                    min_size=1, max_size=3))
    def run(do_handshake, frames):
        try:
            with socketmod.create_connection(
                    ("127.0.0.1", cluster.port()), timeout=3) as sock:
                if do_handshake:
                    handshake(sock)
                for tag, payload in frames:
                    sock.sendall(bytes([tag]) + payload)
                sock.shutdown(socketmod.SHUT_WR)
                while sock.recv(4096):
                    pass
        except OSError:
            pass

    try:
        run()
        # the cluster must still serve a well-formed client
        bdev = hs.create_rbd_bdev("xrfz", f"127.0.0.1:{cluster.port()}",
                                  "rbd", "xrfz-img", block_size=512,
                                  default_size_bytes=4 << 20,
                                  object_bytes=1 << 20)
        bdev.write(0, b"\xa5" * 4096)
        assert bdev.read(0, 4096) == b"\xa5" * 4096
        print(f"rados fuzz: {examples} sessions clean")
    finally:
        cluster.stop()


def main() -> int:
    surface = sys.argv[1] if len(sys.argv) > 1 else "all"
    examples = int(sys.argv[2]) if len(sys.argv) > 2 else 500
    if surface in ("config", "all"):
        fuzz_config(examples)
    if surface in ("vhost", "all"):
        fuzz_vhost(examples)
    if surface in ("nvmf", "all"):
        fuzz_nvmf(examples)
    if surface in ("rados", "all"):
        fuzz_rados(examples)
    return 0


if __name__ == "__main__":
    sys.exit(main())
