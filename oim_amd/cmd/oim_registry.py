"""oim-registry main (reference cmd/oim-registry/main.go)."""

import argparse

from .. import log
from ..common import TLSConfig
from ..registry import FileRegistryDB, MemRegistryDB, Registry, RegistryServer


def main(argv=None) -> int:
    parser = argparse.ArgumentParser(description="OIM registry")
    parser.add_argument("--endpoint", default="tcp://:8999",
                        help="gRPC listen endpoint")
    parser.add_argument("--ca", default="", help="CA certificate (enables mTLS)")
    parser.add_argument("--key", default="",
                        help="component.registry key file (cert derived as .crt)")
    parser.add_argument("--db-file", default="",
                        help="persist registry entries to this JSON file")
    parser.add_argument("--etcd-endpoints", default="",
                        help="comma-separated etcd gRPC endpoints (HA backend)")
    log.add_flags(parser)
    args = parser.parse_args(argv)
    log.init_from_args(args)

    tls = None
    if args.ca and args.key:
        # Empty peer name: accept any CA-signed client; authorization is
        # CN-based per call (reference cmd/oim-registry/main.go:53-56).
        tls = TLSConfig(ca=args.ca, key=args.key)
    if args.etcd_endpoints:
        from ..registry.etcddb import EtcdRegistryDB

        db = EtcdRegistryDB(args.etcd_endpoints.split(","))
    elif args.db_file:
        db = FileRegistryDB(args.db_file)
    else:
        db = MemRegistryDB()
    registry = Registry(db=db, tls=tls)
    server = RegistryServer(args.endpoint, registry)
    server.run()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
