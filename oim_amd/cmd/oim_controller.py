"""oim-controller main (reference cmd/oim-controller/main.go).

On an 8-GPU MI355X node, run one instance per GPU:
  oim-controller --controller-id host-0-gpu3 --hipstored-socket \
      /var/tmp/hipstored-3.sock --gpu 3 ...
--gpu N auto-fills the PCI address from the GPU's own BDF.
"""

import argparse

from .. import log
from ..common import TLSConfig
from ..controller import Controller, ControllerServer


def main(argv=None) -> int:
    parser = argparse.ArgumentParser(description="OIM controller (per-GPU card agent)")
    parser.add_argument("--endpoint", default="tcp://:8999")
    parser.add_argument("--hipstored-socket", default="/var/tmp/hipstored.sock",
                        help="JSON-RPC socket of the data-path daemon")
    parser.add_argument("--vhost-scsi-controller", default="vhost.0")
    parser.add_argument("--vm-vhost-device", default="",
                        help="PCI BDF reported to hosts for mapped volumes")
    parser.add_argument("--controllerid", required=True)
    parser.add_argument("--controller-address", default="",
                        help="external address registered with the registry")
    parser.add_argument("--registry", default="",
                        help="registry endpoint for self-registration")
    parser.add_argument("--registry-delay", type=float, default=60.0)
    parser.add_argument("--gpu", type=int, default=-1,
                        help="HIP device index; fills the PCI address from "
                             "the GPU BDF")
    parser.add_argument("--pci-address", default="",
                        help="explicit PCI BDF for <id>/pci registration")
    parser.add_argument("--ca", default="")
    parser.add_argument("--key", default="")
    log.add_flags(parser)
    args = parser.parse_args(argv)
    log.init_from_args(args)

    tls = None
    if args.ca and args.key:
        tls = TLSConfig(ca=args.ca, key=args.key)
    pci = args.pci_address
    if not pci and args.gpu >= 0:
        from oim_amd import _hipstore

        pci = _hipstore.gpu_pci_address(args.gpu)
    controller = Controller(
        controller_id=args.controllerid,
        hipstored_socket=args.hipstored_socket,
        vhost_controller=args.vhost_scsi_controller,
        vm_vhost_device=args.vm_vhost_device,
        controller_address=args.controller_address,
        registry_address=args.registry,
        registry_delay=args.registry_delay,
        tls=tls,
        pci_address=pci,
    )
    server = ControllerServer(args.endpoint, controller)
    controller.start()
    server.run()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
