"""ceph-csi emulation: repack ceph-csi volume parameters into CephParams
(reference pkg/oim-csi-driver/ceph-csi.go:50-157).

When the driver runs with --emulate=ceph-csi it accepts the volume
attributes/secrets that the ceph-csi rbd plugin would receive and maps
them onto the oim.v0 CephParams oneof, so accelerated nodes mount Ceph
volumes through the OIM controller while other nodes keep running the
stock rbdplugin (reference deploy/kubernetes/ceph-csi)."""

from __future__ import annotations

from typing import Dict

from .. import spec

EMULATE_CEPH_CSI = "ceph-csi"


def ceph_csi_params(request: spec.MapVolumeRequest, volume_id: str,
                    volume_context: Dict[str, str],
                    secrets: Dict[str, str], staging_path: str) -> None:
    """Extract CephParams the way ceph-csi lays them out:
      - pool / monitors / adminid|userid from volume attributes
      - keys from the NodeStage secrets (admin/user keyring entries)
      - image name from the staging path suffix (ceph-csi derives the
        rbd image name csi-vol-<uuid> from the volume handle; the
        reference recovers it from the '<image>/globalmount' path tail,
        ceph-csi.go:120-141)
    """
    pool = volume_context.get("pool", "")
    monitors = volume_context.get("monitors", "")
    user = volume_context.get("userid") or volume_context.get("adminid") or "admin"
    secret = (secrets.get(user) or secrets.get("key")
              or secrets.get("userKey") or secrets.get("adminKey") or "")
    image = volume_context.get("imageName", "")
    if not image and staging_path:
        parts = [p for p in staging_path.split("/") if p]
        if parts and parts[-1] == "globalmount" and len(parts) >= 2:
            image = parts[-2]
        elif parts:
            image = parts[-1]
    if not pool or not monitors:
        raise ValueError(
            "ceph-csi emulation needs 'pool' and 'monitors' volume attributes")
    if not image:
        raise ValueError("ceph-csi emulation could not derive the image name")
    request.ceph.CopyFrom(spec.CephParams(
        user_id=user, secret=secret, monitors=monitors, pool=pool,
        image=image))


# Emulation registry (reference supportedCSI0Drivers oim-driver.go:81-99).
EMULATIONS = {
    EMULATE_CEPH_CSI: ceph_csi_params,
}
