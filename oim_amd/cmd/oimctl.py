"""oimctl: admin CLI for the registry (reference cmd/oimctl/main.go).

  oimctl --registry tcp://reg:8999 --ca ca.crt --key user.admin.key \
      set host-0/pci 0000:c1:00.0
  oimctl ... get [prefix]
"""

import argparse
import sys

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
        else:
            reply = stub.GetValues(
                spec.GetValuesRequest(path=args.prefix), timeout=30)
            for value in reply.values:
                print(f"{value.path}: {value.value}")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
