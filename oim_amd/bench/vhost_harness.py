"""fio-shaped benchmark over the vhost-user-scsi data path.

Measures the VM-guest attach path end to end: virtio descriptor →
hipstored ring worker → engine channel → HBM/RAM → used ring, with
multiple chains outstanding per ring (pair with
HIPSTORE_VHOST_PIPELINE=1 on the daemon to measure the pipelined
worker). Counterpart of fio_harness (which measures in-daemon IOPS)
for the vhost front-end; emits the same perfdash PerfData shape.

  python -m oim_amd.bench.vhost_harness --socket /var/tmp/hipstored.sock \\
      --bdev Malloc0 --rw randread --bs 4096 --iodepth 16 --runtime 5
"""

from __future__ import annotations

import argparse
import json
import os
import random
import statistics
import struct
import sys
import time

from .. import hipstore
from .vhost_client import (
    DESC_NEXT,
    DESC_WRITE,
    GPA_BASE,
    VhostUserMaster,
)


class OutstandingRing:
    """Drives up to `iodepth` 3-descriptor READ/WRITE chains through
    one VhostRing, a fixed slot layout per in-flight tag. personality
    "scsi" (virtio-scsi CDBs) or "blk" (virtio-blk outhdr + status).
    req_base/data_base are guest offsets owned by this ring."""

    def __init__(self, ring, req_base: int, data_base: int,
                 block_size: int, io_size: int, num_blocks: int,
                 iodepth: int, rw: str, personality: str = "scsi",
                 seed: int = 0x5EED):
        self.ring = ring
        self.mem = ring.m.mem
        self.req_base = req_base
        self.data_base = data_base
        self.personality = personality
        self.block = block_size
        self.io_size = io_size
        self.blocks_per_io = io_size // block_size
        self.units = num_blocks // self.blocks_per_io
        self.iodepth = iodepth
        self.rw = rw
        self.rng = random.Random(seed)
        self.submit_ts = {}
        self.lat_us = []
        self.completed = 0
        if iodepth * 3 > ring.qsize:
            raise ValueError("iodepth needs a bigger ring (qsize)")

    def _chain(self, tag: int):
        lba = (self.rng.randrange(self.units)) * self.blocks_per_io
        write = self.rw == "randwrite" or (
            self.rw == "randrw" and self.rng.random() < 0.5)
        req_gpa = self.req_base + 0x100 * tag       # request header
        resp_gpa = req_gpa + 0x80                   # response / status
        data_gpa = self.data_base + self.io_size * tag
        base = tag * 3
        if self.personality == "blk":
            sector = lba * self.block // 512
            header = struct.pack("<IIQ", 1 if write else 0, 0, sector)
            self.mem[req_gpa:req_gpa + 16] = header
            self.ring.write_desc(base, GPA_BASE + req_gpa, 16,
                                 DESC_NEXT, base + 1)
            flags = DESC_NEXT if write else (DESC_WRITE | DESC_NEXT)
            self.ring.write_desc(base + 1, GPA_BASE + data_gpa,
                                 self.io_size, flags, base + 2)
            self.ring.write_desc(base + 2, GPA_BASE + resp_gpa, 1,
                                 DESC_WRITE)
        else:
            cdb = bytearray(10)
            cdb[0] = 0x2A if write else 0x28
            cdb[2:6] = struct.pack(">I", lba)
            cdb[7:9] = struct.pack(">H", self.blocks_per_io)
            req = (bytes([1, 0, 0x40, 0, 0, 0, 0, 0])
                   + struct.pack("<Q", tag + 1) + bytes(3)
                   + bytes(cdb).ljust(32, b"\0"))
            self.mem[req_gpa:req_gpa + 51] = req
            self.ring.write_desc(base, GPA_BASE + req_gpa, 51,
                                 DESC_NEXT, base + 1)
            if write:
                self.ring.write_desc(base + 1, GPA_BASE + data_gpa,
                                     self.io_size, DESC_NEXT, base + 2)
                self.ring.write_desc(base + 2, GPA_BASE + resp_gpa, 108,
                                     DESC_WRITE)
            else:
                self.ring.write_desc(base + 1, GPA_BASE + resp_gpa, 108,
                                     DESC_WRITE | DESC_NEXT, base + 2)
                self.ring.write_desc(base + 2, GPA_BASE + data_gpa,
                                     self.io_size, DESC_WRITE)
        self.submit_ts[base] = time.perf_counter()
        self.ring.submit(base)

    def run(self, seconds: float) -> None:
        deadline = time.perf_counter() + seconds
        for tag in range(self.iodepth):
            self._chain(tag)
        stopping = False
        inflight = self.iodepth
        while inflight:
            self.ring.wait_used()
            head = self.ring.last_used_head
            now = time.perf_counter()
            started = self.submit_ts.pop(head)
            self.lat_us.append((now - started) * 1e6)
            self.completed += 1
            inflight -= 1
            if not stopping and now >= deadline:
                stopping = True
            if not stopping:
                self._chain(head // 3)
                inflight += 1

    def run_count(self, total_ios: int) -> None:
        """Complete exactly `total_ios` I/Os (step-shaped runs for
        bench.py's fixed-work steps), keeping `iodepth` outstanding."""
        initial = min(self.iodepth, total_ios)
        for tag in range(initial):
            self._chain(tag)
        submitted = initial
        inflight = initial
        while inflight:
            self.ring.wait_used()
            head = self.ring.last_used_head
            now = time.perf_counter()
            started = self.submit_ts.pop(head)
            self.lat_us.append((now - started) * 1e6)
            self.completed += 1
            inflight -= 1
            if submitted < total_ios:
                self._chain(head // 3)
                submitted += 1
                inflight += 1


class VhostAttachment:
    """A vhost-user controller on `socket`'s daemon with `numjobs`
    request rings and an OutstandingRing job per ring — the reusable
    core shared by the time-driven CLI below and bench.py's
    step-driven `--frontend vhost` mode."""

    def __init__(self, client, socket_path: str, bdev_name: str,
                 ctrlr: str, personality: str, numjobs: int,
                 iodepth: int, bs: int, rw: str, create: bool = True):
        """`create=False` attaches to an ALREADY-constructed controller
        whose LUN 0 / target 0 is the bdev (e.g. the `vhost.0`
        controller MapVolume set up) instead of building its own."""
        self.client = client
        self.ctrlr = ctrlr
        self.created = create
        bdev = hipstore.get_bdevs(client, bdev_name)[0]
        if create:
            if personality == "blk":
                client.invoke("construct_vhost_blk_controller",
                              {"ctrlr": ctrlr, "dev_name": bdev_name})
            else:
                client.invoke("construct_vhost_scsi_controller",
                              {"ctrlr": ctrlr})
                client.invoke("add_vhost_scsi_lun",
                              {"ctrlr": ctrlr, "scsi_target_num": 0,
                               "bdev_name": bdev_name})
        vhost_path = os.path.join(os.path.dirname(socket_path), ctrlr)
        max_jobs = 8 if personality == "blk" else 6
        if not 1 <= numjobs <= max_jobs:
            raise ValueError(f"numjobs must be 1..{max_jobs} "
                             f"for {personality}")
        first_queue = 0 if personality == "blk" else 2
        qsize = 16
        while qsize < iodepth * 3 + 1:
            qsize *= 2
        # per-ring slab: ring header + 0x100/request slot + data buffers
        header_guess = ((16 + 2) * qsize + 0x2000 + 8 * qsize + 0xFFF) \
            & ~0xFFF
        slab = header_guess + iodepth * 0x100 + iodepth * bs
        slab = (slab + 0xFFFF) & ~0xFFFF
        mem = 0x10000 + numjobs * slab
        self.master = VhostUserMaster(
            vhost_path, mem_size=max(mem, 8 << 20), qsize=qsize,
            queue=first_queue)
        self.master.handshake()
        self.rings = []
        self.jobs = []
        for i in range(numjobs):
            base = 0x10000 + i * slab
            ring = self.master.add_ring(first_queue + i, base, qsize)
            req_base = base + ((ring.header_bytes + 0xFFF) & ~0xFFF)
            data_base = req_base + iodepth * 0x100
            self.rings.append(ring)
            self.jobs.append(OutstandingRing(
                ring, req_base, data_base, bdev.block_size, bs,
                bdev.num_blocks, iodepth, rw,
                personality=personality, seed=0x5EED + i))

    def close(self) -> None:
        for ring in self.rings:
            ring.close()
        self.master.close()
        if self.created:
            self.client.invoke("remove_vhost_controller",
                               {"ctrlr": self.ctrlr})


def _native_main(args) -> int:
    """Drive the controller with the C++ vhost-user master
    (native/src/vhost_master.cpp): one standing session, repeated
    fixed-count runs until --runtime elapses. This is the same driver
    bench.py --frontend vhost uses."""
    from oim_amd import _hipstore as hs

    with hipstore.Client(args.socket) as client:
        bdev = hipstore.get_bdevs(client, args.bdev)[0]
        if args.personality == "blk":
            client.invoke("construct_vhost_blk_controller",
                          {"ctrlr": args.ctrlr, "dev_name": args.bdev})
        else:
            client.invoke("construct_vhost_scsi_controller",
                          {"ctrlr": args.ctrlr})
            client.invoke("add_vhost_scsi_lun",
                          {"ctrlr": args.ctrlr, "scsi_target_num": 0,
                           "bdev_name": args.bdev})
        vhost_path = os.path.join(os.path.dirname(args.socket),
                                  args.ctrlr)
        try:
            session = hs.VhostMasterSession(
                vhost_path, args.personality, args.numjobs,
                args.iodepth, args.bs, bdev.block_size,
                bdev.num_blocks * bdev.block_size)
            session.run(max(args.numjobs * args.iodepth * 16, 4096),
                        args.rw)  # warm: rings built, memory registered
            chunk = max(args.numjobs * args.iodepth * 256, 20000)
            total = 0
            start = time.perf_counter()
            while time.perf_counter() - start < args.runtime:
                r = session.run(chunk, args.rw)
                total += r["io_count"]
            elapsed = time.perf_counter() - start
            del session
        finally:
            client.invoke("remove_vhost_controller",
                          {"ctrlr": args.ctrlr})
    iops = total / elapsed
    print(f"{args.bdev} via vhost ({args.personality}, native master): "
          f"rw={args.rw}, bs={args.bs}, iodepth={args.iodepth}, "
          f"numjobs={args.numjobs}")
    print(f"  IOPS={iops:,.0f}, BW={iops * args.bs / 1e6:,.1f} MB/s "
          f"({total} ios in {elapsed:.2f}s)")
    print(f"  lat (usec, last chunk): avg={r['lat_avg_us']:.1f}, "
          f"p50={r['lat_p50_us']:.0f}, p99={r['lat_p99_us']:.0f}")
    if args.perfdash:
        payload = {
            "version": "v1",
            "dataItems": [{
                "data": {
                    "iops": iops,
                    "throughput_mbps": iops * args.bs / 1e6,
                    "lat_p50_us": r["lat_p50_us"],
                    "lat_p99_us": r["lat_p99_us"],
                },
                "unit": "mixed",
                "labels": {"bdev": args.bdev, "rw": args.rw,
                           "bs": str(args.bs),
                           "iodepth": str(args.iodepth),
                           "numjobs": str(args.numjobs),
                           "master": "native",
                           "path": f"vhost-user-{args.personality}"},
            }],
            "labels": {"suite": "hipstored-vhost"},
        }
        print("[Finished:Performance] " + json.dumps(payload))
    return 0


def main(argv=None) -> int:
    parser = argparse.ArgumentParser(
        description="vhost-user-scsi data-path benchmark")
    parser.add_argument("--socket", required=True,
                        help="hipstored JSON-RPC socket")
    parser.add_argument("--bdev", required=True)
    parser.add_argument("--rw", default="randread",
                        choices=["randread", "randwrite", "randrw"])
    parser.add_argument("--bs", type=int, default=4096)
    parser.add_argument("--iodepth", type=int, default=16,
                        help="outstanding chains PER ring")
    parser.add_argument("--numjobs", type=int, default=1,
                        help="request rings driven in parallel "
                             "(scsi: up to 6, blk: up to 8)")
    parser.add_argument("--runtime", type=float, default=5.0)
    parser.add_argument("--ctrlr", default="vhost-bench")
    parser.add_argument("--personality", default="scsi",
                        choices=["scsi", "blk"])
    parser.add_argument("--master", default="python",
                        choices=["python", "native"],
                        help="ring driver: the Python conformance "
                             "master (interpreter-bound, ~50-200k "
                             "IOPS) or the C++ native master, which "
                             "measures the daemon's data path")
    parser.add_argument("--perfdash", action="store_true")
    args = parser.parse_args(argv)

    if args.master == "native":
        return _native_main(args)

    with hipstore.Client(args.socket) as client:
        try:
            attachment = VhostAttachment(
                client, args.socket, args.bdev, args.ctrlr,
                args.personality, args.numjobs, args.iodepth, args.bs,
                args.rw)
        except ValueError as e:
            raise SystemExit(str(e))
        jobs = attachment.jobs
        try:
            import threading
            threads = [threading.Thread(target=j.run, args=(args.runtime,))
                       for j in jobs]
            start = time.perf_counter()
            for t in threads:
                t.start()
            for t in threads:
                t.join()
            elapsed = time.perf_counter() - start
        finally:
            attachment.close()

    completed = sum(j.completed for j in jobs)
    iops = completed / elapsed
    lat = sorted(x for j in jobs for x in j.lat_us)
    pct = lambda p: lat[min(len(lat) - 1, int(len(lat) * p))] if lat else 0
    print(f"{args.bdev} via vhost ({args.personality}): rw={args.rw}, "
          f"bs={args.bs}, iodepth={args.iodepth}, numjobs={args.numjobs}")
    print(f"  IOPS={iops:,.0f}, BW={iops * args.bs / 1e6:,.1f} MB/s "
          f"({completed} ios in {elapsed:.2f}s)")
    if lat:
        print(f"  lat (usec): avg={statistics.fmean(lat):.1f}, "
              f"p50={pct(0.5):.0f}, p99={pct(0.99):.0f}, "
              f"max={lat[-1]:.0f}")
    if args.perfdash:
        payload = {
            "version": "v1",
            "dataItems": [{
                "data": {
                    "iops": iops,
                    "throughput_mbps": iops * args.bs / 1e6,
                    "lat_p50_us": pct(0.5),
                    "lat_p99_us": pct(0.99),
                },
                "unit": "mixed",
                "labels": {"bdev": args.bdev, "rw": args.rw,
                           "bs": str(args.bs),
                           "iodepth": str(args.iodepth),
                           "numjobs": str(args.numjobs),
                           "path": f"vhost-user-{args.personality}"},
            }],
            "labels": {"suite": "hipstored-vhost"},
        }
        print("[Finished:Performance] " + json.dumps(payload))
    return 0


if __name__ == "__main__":
    sys.exit(main())
