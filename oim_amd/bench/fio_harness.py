"""fio-shaped benchmark CLI against a hipstored daemon.

  python -m oim_amd.bench.fio_harness --socket /var/tmp/hipstored.sock \
      --bdev Malloc0 --rw randread --bs 4096 --iodepth 32 --numjobs 8 \
      --runtime 10 [--perfdash]

The workload loop runs in-daemon (perf_run RPC) next to the data path —
fio's I/O-engine role — and this CLI formats the result like an fio
summary plus, with --perfdash, the perfdash JSON block."""

from __future__ import annotations

import argparse
import sys

from .. import hipstore, log
from .perftype import PerfData, emit_perf_data, perf_result_to_data_item


def main(argv=None) -> int:
    parser = argparse.ArgumentParser(description="hipstored fio-shaped bench")
    parser.add_argument("--socket", required=True)
    parser.add_argument("--bdev", required=True)
    parser.add_argument("--rw", default="randread",
                        choices=["randread", "randwrite", "randrw"])
    parser.add_argument("--bs", type=int, default=4096)
    parser.add_argument("--iodepth", type=int, default=32)
    parser.add_argument("--numjobs", type=int, default=1)
    parser.add_argument("--runtime", type=float, default=5.0)
    parser.add_argument("--perfdash", action="store_true",
                        help="also emit the perfdash JSON block")
    log.add_flags(parser)
    args = parser.parse_args(argv)
    log.init_from_args(args)

    with hipstore.Client(args.socket, timeout=args.runtime + 120) as client:
        result = hipstore.perf_run(
            client, args.bdev, workload=args.rw, io_size=args.bs,
            queue_depth=args.iodepth, num_queues=args.numjobs,
            seconds=args.runtime)

    print(f"{args.bdev}: ({'g=0'}): rw={args.rw}, bs={args.bs}, "
          f"iodepth={args.iodepth}, numjobs={args.numjobs}")
    print(f"  IOPS={result['iops']:,.0f}, "
          f"BW={result['throughput_mbps']:,.1f} MB/s "
          f"({result['io_count']:,} ios in {result['seconds']:.2f}s)")
    print(f"  lat (usec): avg={result['lat_avg_us']:.1f}, "
          f"p50={result['lat_p50_us']:.0f}, p90={result['lat_p90_us']:.0f}, "
          f"p99={result['lat_p99_us']:.0f}, p99.9={result['lat_p999_us']:.0f}, "
          f"max={result['lat_max_us']:.0f}")
    if args.perfdash:
        perf = PerfData(
            version="v1",
            data_items=[perf_result_to_data_item(result, {
                "bdev": args.bdev, "rw": args.rw, "bs": str(args.bs),
                "iodepth": str(args.iodepth), "numjobs": str(args.numjobs),
            })],
            labels={"suite": "hipstored-fio"},
        )
        emit_perf_data(perf, sys.stdout)
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
