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
    DATA_IN_OFF,
    DESC_NEXT,
    DESC_WRITE,
    GPA_BASE,
    REQ_OFF,
    RESP_OFF,
    VhostUserMaster,
)


class OutstandingScsiRing:
    """Drives up to `iodepth` 3-descriptor READ/WRITE chains through
    one vhost ring, reusing a fixed slot layout per in-flight tag.
    personality "scsi" (virtio-scsi CDBs) or "blk" (virtio-blk
    outhdr + status byte)."""

    def __init__(self, master: VhostUserMaster, block_size: int,
                 io_size: int, num_blocks: int, iodepth: int, rw: str,
                 personality: str = "scsi"):
        self.personality = personality
        self.m = master
        self.block = block_size
        self.io_size = io_size
        self.blocks_per_io = io_size // block_size
        self.units = num_blocks // self.blocks_per_io
        self.iodepth = iodepth
        self.rw = rw
        self.rng = random.Random(0x5EED)
        self.submit_ts = {}
        self.lat_us = []
        self.completed = 0
        if iodepth * 3 > master.qsize:
            raise ValueError("iodepth needs a bigger ring (qsize)")

    def _chain(self, tag: int):
        lba = (self.rng.randrange(self.units)) * self.blocks_per_io
        write = self.rw == "randwrite" or (
            self.rw == "randrw" and self.rng.random() < 0.5)
        req_gpa = REQ_OFF + 0x100 * tag
        resp_gpa = RESP_OFF + 0x100 * tag
        data_gpa = DATA_IN_OFF + self.io_size * tag
        base = tag * 3
        if self.personality == "blk":
            sector = lba * self.block // 512
            header = struct.pack("<IIQ", 1 if write else 0, 0, sector)
            self.m.mem[req_gpa:req_gpa + 16] = header
            self.m._write_desc(base, GPA_BASE + req_gpa, 16,
                               DESC_NEXT, base + 1)
            if write:
                self.m._write_desc(base + 1, GPA_BASE + data_gpa,
                                   self.io_size, DESC_NEXT, base + 2)
            else:
                self.m._write_desc(base + 1, GPA_BASE + data_gpa,
                                   self.io_size, DESC_WRITE | DESC_NEXT,
                                   base + 2)
            self.m._write_desc(base + 2, GPA_BASE + resp_gpa, 1,
                               DESC_WRITE)
            self.submit_ts[base] = time.perf_counter()
            self.m._submit(base)
            return
        cdb = bytearray(10)
        cdb[0] = 0x2A if write else 0x28
        cdb[2:6] = struct.pack(">I", lba)
        cdb[7:9] = struct.pack(">H", self.blocks_per_io)
        req = (bytes([1, 0, 0x40, 0, 0, 0, 0, 0])
               + struct.pack("<Q", tag + 1) + bytes(3)
               + bytes(cdb).ljust(32, b"\0"))
        self.m.mem[req_gpa:req_gpa + 51] = req
        if write:
            self.m._write_desc(base, GPA_BASE + req_gpa, 51,
                               DESC_NEXT, base + 1)
            self.m._write_desc(base + 1, GPA_BASE + data_gpa, self.io_size,
                               DESC_NEXT, base + 2)
            self.m._write_desc(base + 2, GPA_BASE + resp_gpa, 108,
                               DESC_WRITE)
        else:
            self.m._write_desc(base, GPA_BASE + req_gpa, 51,
                               DESC_NEXT, base + 1)
            self.m._write_desc(base + 1, GPA_BASE + resp_gpa, 108,
                               DESC_WRITE | DESC_NEXT, base + 2)
            self.m._write_desc(base + 2, GPA_BASE + data_gpa, self.io_size,
                               DESC_WRITE)
        self.submit_ts[base] = time.perf_counter()
        self.m._submit(base)

    def run(self, seconds: float) -> None:
        deadline = time.perf_counter() + seconds
        for tag in range(self.iodepth):
            self._chain(tag)
        stopping = False
        inflight = self.iodepth
        while inflight:
            self.m._wait_used()
            head = self.m.last_used_head
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


def main(argv=None) -> int:
    parser = argparse.ArgumentParser(
        description="vhost-user-scsi data-path benchmark")
    parser.add_argument("--socket", required=True,
                        help="hipstored JSON-RPC socket")
    parser.add_argument("--bdev", required=True)
    parser.add_argument("--rw", default="randread",
                        choices=["randread", "randwrite", "randrw"])
    parser.add_argument("--bs", type=int, default=4096)
    parser.add_argument("--iodepth", type=int, default=16)
    parser.add_argument("--runtime", type=float, default=5.0)
    parser.add_argument("--ctrlr", default="vhost-bench")
    parser.add_argument("--personality", default="scsi",
                        choices=["scsi", "blk"])
    parser.add_argument("--perfdash", action="store_true")
    args = parser.parse_args(argv)

    with hipstore.Client(args.socket) as client:
        bdevs = hipstore.get_bdevs(client, args.bdev)
        bdev = bdevs[0]
        if args.personality == "blk":
            client.invoke("construct_vhost_blk_controller",
                          {"ctrlr": args.ctrlr, "dev_name": args.bdev})
        else:
            client.invoke("construct_vhost_scsi_controller",
                          {"ctrlr": args.ctrlr})
            client.invoke("add_vhost_scsi_lun",
                          {"ctrlr": args.ctrlr, "scsi_target_num": 0,
                           "bdev_name": args.bdev})
        vhost_path = os.path.join(os.path.dirname(args.socket), args.ctrlr)
        qsize = 1
        while qsize < args.iodepth * 3 + 1:
            qsize *= 2
        mem = (1 << 20) + DATA_IN_OFF + args.iodepth * args.bs
        master = VhostUserMaster(
            vhost_path, mem_size=max(mem, 8 << 20), qsize=max(qsize, 16),
            queue=0 if args.personality == "blk" else 2)
        master.negotiate()
        try:
            ring = OutstandingScsiRing(
                master, bdev.block_size, args.bs, bdev.num_blocks,
                args.iodepth, args.rw, personality=args.personality)
            start = time.perf_counter()
            ring.run(args.runtime)
            elapsed = time.perf_counter() - start
        finally:
            master.close()
            client.invoke("remove_vhost_controller", {"ctrlr": args.ctrlr})

    iops = ring.completed / elapsed
    lat = sorted(ring.lat_us)
    pct = lambda p: lat[min(len(lat) - 1, int(len(lat) * p))] if lat else 0
    print(f"{args.bdev} via vhost: rw={args.rw}, bs={args.bs}, "
          f"iodepth={args.iodepth}")
    print(f"  IOPS={iops:,.0f}, BW={iops * args.bs / 1e6:,.1f} MB/s "
          f"({ring.completed} ios in {elapsed:.2f}s)")
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
                           "path": f"vhost-user-{args.personality}"},
            }],
            "labels": {"suite": "hipstored-vhost"},
        }
        print("[Finished:Performance] " + json.dumps(payload))
    return 0


if __name__ == "__main__":
    sys.exit(main())
