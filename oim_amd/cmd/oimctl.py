"""oimctl: admin CLI for the registry (reference cmd/oimctl/main.go,
extended with proxied volume operations).

  oimctl --registry tcp://reg:8999 --ca ca.crt --key user.admin.key \
      set host-0/pci 0000:c1:00.0
  oimctl ... get [prefix]
  oimctl ... provision --controller gpu-0 vol1 64MiB
  oimctl ... map --controller gpu-0 vol1
  oimctl ... unmap --controller gpu-0 vol1
  oimctl ... check --controller gpu-0 vol1
"""

import argparse

import grpc

from .. import log, spec
from ..common import TLSConfig
from ..common.server import grpc_target
from ..common.tlsutil import channel_options_for_peer, load_tls_channel_credentials


def main(argv=None) -> int:
    parser = argparse.ArgumentParser(description="OIM admin CLI")
    parser.add_argument("--registry", required=True, help="registry endpoint")
    parser.add_argument("--ca", default="")
    parser.add_argument("--key", default="", help="user.admin key file")
    log.add_flags(parser)
    sub = parser.add_subparsers(dest="command", required=True)
    set_cmd = sub.add_parser("set", help="set a registry value")
    set_cmd.add_argument("path")
    set_cmd.add_argument("value")
    get_cmd = sub.add_parser("get", help="list registry values")
    get_cmd.add_argument("prefix", nargs="?", default="")
    delete_cmd = sub.add_parser("delete", help="delete a registry value")
    delete_cmd.add_argument("path")
    for name, help_text in (
        ("provision", "create (or with size 0 delete) a malloc bdev"),
        ("map", "map a provisioned volume to a SCSI target"),
        ("unmap", "unmap a volume"),
        ("check", "check that a malloc bdev exists"),
        ("clone", "clone a malloc bdev (device-side HBM-rate copy)"),
        ("resize", "grow a malloc bdev (offline)"),
        ("list", "list malloc bdevs"),
        ("stats", "per-bdev I/O counters"),
    ):
        cmd = sub.add_parser(name, help=help_text + " via the registry proxy")
        cmd.add_argument("--controller", required=True,
                         help="controller ID to proxy to")
        if name in ("list", "stats"):
            cmd.add_argument("volume", nargs="?", default="",
                             help="name filter" if name == "stats"
                             else "name prefix filter")
        else:
            cmd.add_argument("volume")
        if name == "provision":
            cmd.add_argument("size", help='bytes, or "64MiB"-style; 0 deletes')
        if name == "clone":
            cmd.add_argument("dest", help="name of the clone to create")
        if name == "resize":
            cmd.add_argument("size", help='new size: bytes or "8GiB"-style')
    args = parser.parse_args(argv)
    log.init_from_args(args)

    target = grpc_target(args.registry)
    if args.ca and args.key:
        creds = load_tls_channel_credentials(TLSConfig(ca=args.ca, key=args.key))
        channel = grpc.secure_channel(
            target, creds, options=channel_options_for_peer("component.registry"))
    else:
        channel = grpc.insecure_channel(target)
    with channel:
        stub = spec.RegistryStub(channel)
        if args.command == "set":
            stub.SetValue(spec.SetValueRequest(
                value=spec.Value(path=args.path, value=args.value)), timeout=30)
        elif args.command == "delete":
            stub.SetValue(spec.SetValueRequest(
                value=spec.Value(path=args.path, value="")), timeout=30)
        elif args.command == "get":
            reply = stub.GetValues(
                spec.GetValuesRequest(path=args.prefix), timeout=30)
            for value in reply.values:
                print(f"{value.path}: {value.value}")
        else:
            controller = spec.ControllerStub(channel)
            metadata = ((spec.CONTROLLER_ID_KEY, args.controller),)
            if args.command == "provision":
                stub_size = parse_size(args.size)
                controller.ProvisionMallocBDev(
                    spec.ProvisionMallocBDevRequest(
                        bdev_name=args.volume, size=stub_size),
                    metadata=metadata, timeout=60)
                print(f"provisioned {args.volume} ({stub_size} bytes)"
                      if stub_size else f"deleted {args.volume}")
            elif args.command == "map":
                reply = controller.MapVolume(
                    spec.MapVolumeRequest(volume_id=args.volume,
                                          malloc=spec.MallocParams()),
                    metadata=metadata, timeout=60)
                pci = reply.pci_address
                print(f"mapped {args.volume} at "
                      f"{pci.domain:04x}:{pci.bus:02x}:{pci.device:02x}."
                      f"{pci.function} target {reply.scsi_disk.target} "
                      f"lun {reply.scsi_disk.lun}")
            elif args.command == "unmap":
                controller.UnmapVolume(
                    spec.UnmapVolumeRequest(volume_id=args.volume),
                    metadata=metadata, timeout=60)
                print(f"unmapped {args.volume}")
            elif args.command == "check":
                controller.CheckMallocBDev(
                    spec.CheckMallocBDevRequest(bdev_name=args.volume),
                    metadata=metadata, timeout=60)
                print(f"{args.volume} exists")
            elif args.command == "clone":
                controller.CloneMallocBDev(
                    spec.CloneMallocBDevRequest(source=args.volume,
                                                dest=args.dest),
                    metadata=metadata, timeout=120)
                print(f"cloned {args.volume} -> {args.dest}")
            elif args.command == "list":
                reply = controller.ListMallocBDevs(
                    spec.ListMallocBDevsRequest(prefix=args.volume),
                    metadata=metadata, timeout=60)
                for info in reply.bdevs:
                    print(f"{info.name}  {info.size}  bs={info.block_size}  "
                          f"{info.product_name}")
            elif args.command == "stats":
                reply = controller.GetIOStats(
                    spec.GetIOStatsRequest(bdev_name=args.volume),
                    metadata=metadata, timeout=60)
                for st in reply.bdevs:
                    print(f"{st.name}  reads={st.num_read_ops} "
                          f"writes={st.num_write_ops} "
                          f"unmaps={st.num_unmap_ops} "
                          f"rB={st.bytes_read} wB={st.bytes_written}")
            elif args.command == "resize":
                new_size = parse_size(args.size)
                controller.ResizeMallocBDev(
                    spec.ResizeMallocBDevRequest(bdev_name=args.volume,
                                                 size=new_size),
                    metadata=metadata, timeout=120)
                print(f"resized {args.volume} to {new_size} bytes")
    return 0


_UNITS = {"": 1, "k": 1 << 10, "m": 1 << 20, "g": 1 << 30, "t": 1 << 40}


def parse_size(text: str) -> int:
    """"64MiB" / "1G" / "4096" -> bytes."""
    t = text.strip().lower().removesuffix("ib").removesuffix("b")
    for suffix, mult in _UNITS.items():
        if suffix and t.endswith(suffix):
            return int(float(t[: -len(suffix)]) * mult)
    return int(t)


if __name__ == "__main__":
    raise SystemExit(main())
