"""Localize the vhost multi-ring plateau (~2.2M IOPS): drive N
controllers x R rings concurrently. If 2x(4-ring controllers) beats
1x(8-ring), the serialization is per-controller (slave-side shared
state); if totals match, it is global (engine / runtime / host)."""
import os
import pathlib
import sys
import tempfile
import threading

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tests"))

import fixtures  # noqa: E402
import oim_amd._hipstore as hs  # noqa: E402
from oim_amd import hipstore  # noqa: E402


def run_case(sock_dir, client, n_ctrl, rings, qd, total_per_ctrl):
    socks = []
    for i in range(n_ctrl):
        name = f"sp{i}"
        ctrl = f"vsp{i}"
        try:
            hipstore.get_bdevs(client, name)
        except hipstore.RpcError:
            hipstore.construct_malloc_bdev(
                client, num_blocks=1 << 20, block_size=512, name=name)
        client.invoke("construct_vhost_blk_controller",
                      {"ctrlr": ctrl, "dev_name": name})
        socks.append(os.path.join(sock_dir, ctrl))
    sessions = [hs.VhostMasterSession(s, "blk", rings, qd, 4096, 512,
                                      (1 << 20) * 512) for s in socks]
    for s in sessions:
        s.run(20000, "randread")  # warm
    results = [None] * n_ctrl
    import time

    def drive(i):
        results[i] = sessions[i].run(total_per_ctrl, "randread")

    t0 = time.perf_counter()
    threads = [threading.Thread(target=drive, args=(i,))
               for i in range(n_ctrl)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    dt = time.perf_counter() - t0
    total = sum(r["io_count"] for r in results)
    p99 = max(r["lat_p99_us"] for r in results)
    print("ctrl=%d rings=%d qd=%d: total %8.0f IOPS p99=%5.0fus"
          % (n_ctrl, rings, qd, total / dt, p99), flush=True)
    del sessions
    for i in range(n_ctrl):
        client.invoke("remove_vhost_controller", {"ctrlr": f"vsp{i}"})


def main():
    tmp = pathlib.Path(tempfile.mkdtemp())
    daemon = fixtures.launch_hipstored(tmp, cpu=not hs.gpu_available())
    try:
        with hipstore.Client(daemon.socket_path) as client:
            sock_dir = os.path.dirname(daemon.socket_path)
            run_case(sock_dir, client, 1, 4, 32, 400000)
            run_case(sock_dir, client, 1, 8, 32, 400000)
            run_case(sock_dir, client, 2, 4, 32, 250000)
            run_case(sock_dir, client, 4, 2, 32, 150000)
            run_case(sock_dir, client, 2, 2, 64, 250000)
    finally:
        daemon.stop()


if __name__ == "__main__":
    main()
