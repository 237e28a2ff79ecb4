"""oim-csi-driver main (reference cmd/oim-csi-driver/main.go)."""

import argparse

from .. import log
from ..common import TLSConfig
from ..csidriver import LocalBackend, OIMDriver, RemoteBackend, make_params_mapper


def main(argv=None) -> int:
    parser = argparse.ArgumentParser(description="OIM CSI driver")
    parser.add_argument("--endpoint", default="unix:///tmp/oim-csi.sock",
                        help="CSI gRPC endpoint (kubelet side)")
    parser.add_argument("--drivername", default="oim-malloc")
    parser.add_argument("--nodeid", required=True)
    # local mode
    parser.add_argument("--aio-dir", default="/var/lib/oim-aio",
                        help="backing dir for `backing: aio` volumes")
    parser.add_argument("--hipstored-socket", default="",
                        help="local mode: hipstored JSON-RPC socket")
    # remote mode
    parser.add_argument("--oim-registry-address", default="",
                        help="remote mode: registry endpoint")
    parser.add_argument("--controller-id", default="",
                        help="remote mode: controller to talk to")
    parser.add_argument("--emulate", default="",
                        help="volume-parameter emulation (ceph-csi)")
    parser.add_argument("--csiversion", default="1.0",
                        choices=["1.0", "0.3"],
                        help="CSI personality to serve (reference "
                             "main.go --csiversion; 0.3 is the legacy "
                             "twin for pre-1.0 kubelets / ceph-csi "
                             "v0.3 emulation)")
    parser.add_argument("--ca", default="")
    parser.add_argument("--key", default="")
    log.add_flags(parser)
    args = parser.parse_args(argv)
    log.init_from_args(args)

    local = bool(args.hipstored_socket)
    remote = bool(args.oim_registry_address)
    if local == remote:
        parser.error("exactly one of --hipstored-socket (local mode) and "
                     "--oim-registry-address (remote mode) is required")
    if local:
        if args.emulate:
            parser.error("--emulate requires remote mode")
        backend = LocalBackend(args.hipstored_socket,
                               aio_dir=args.aio_dir)
    else:
        if not args.controller_id:
            parser.error("remote mode requires --controller-id")
        tls = None
        if args.ca and args.key:
            tls = TLSConfig(ca=args.ca, key=args.key)
        backend = RemoteBackend(
            registry_address=args.oim_registry_address,
            controller_id=args.controller_id,
            tls=tls,
            params_mapper=make_params_mapper(args.emulate),
        )
    driver = OIMDriver(
        driver_name=args.drivername, node_id=args.nodeid,
        endpoint=args.endpoint, backend=backend,
        csi_version=args.csiversion)
    driver.run()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
