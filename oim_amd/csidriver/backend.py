"""The OIMBackend seam (reference oim-driver.go:71-78): what differs
between local mode (drive hipstored directly, device via kernel NBD)
and remote mode (drive a controller through the registry proxy, device
via PCI/SCSI discovery)."""

from __future__ import annotations

from typing import Dict, Tuple


class VolumeExistsError(ValueError):
    """A volume with this name exists with incompatible parameters
    (CSI CreateVolume must answer ALREADY_EXISTS, spec.md csi v1)."""


class OIMBackend:
    def create_volume(self, name: str, size: int,
                      parameters: Dict[str, str] = None
                      ) -> Tuple[str, Dict[str, str]]:
        """Returns (volume_id, volume_context)."""
        raise NotImplementedError

    def delete_volume(self, volume_id: str) -> None:
        raise NotImplementedError

    def check_volume_exists(self, volume_id: str) -> bool:
        raise NotImplementedError

    def create_device(self, volume_id: str,
                      volume_context: Dict[str, str]) -> str:
        """Makes the volume appear as a host block device; returns its
        /dev path. Idempotent."""
        raise NotImplementedError

    def delete_device(self, volume_id: str) -> None:
        raise NotImplementedError

    def get_capacity(self):
        """Available bytes for new volumes, or None when the backend
        cannot tell (remote mode: the oim.v0 API has no capacity RPC)."""
        return None

    def expand_volume(self, volume_id, size):
        """Grow the volume to size bytes; returns the new size.
        Raises LookupError (unknown volume) or RuntimeError (busy /
        unsupported)."""
        raise NotImplementedError

    def supports_expansion(self) -> bool:
        return False

    def list_volumes(self):
        """[(volume_id, size_bytes), ...] or None when the backend
        cannot enumerate (remote mode: no list RPC in oim.v0)."""
        return None

    # --- snapshots (optional; local mode only — backed by hipstored's
    # bdev_clone, an HBM-rate device-side copy) ---------------------------

    def supports_snapshots(self) -> bool:
        return False

    def create_snapshot(self, name, source_volume_id):
        """Returns (snapshot_id, size_bytes, creation_unix_seconds).
        Raises VolumeExistsError on a name conflict with a different
        source, LookupError if the source volume does not exist."""
        raise NotImplementedError

    def delete_snapshot(self, snapshot_id) -> None:
        raise NotImplementedError

    def list_snapshots(self):
        """[(snapshot_id, source_volume_id, size_bytes, ctime), ...]"""
        raise NotImplementedError

    def restore_snapshot(self, snapshot_id, volume_name):
        """Clone a snapshot into a new volume; returns (volume_id,
        size_bytes). Raises LookupError for unknown snapshots."""
        raise NotImplementedError

    def clone_volume(self, source_volume_id, volume_name):
        """CSI volume cloning (CreateVolume from a volume source);
        same contract as restore_snapshot."""
        raise NotImplementedError
