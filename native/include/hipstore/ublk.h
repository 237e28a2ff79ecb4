// Kernel ublk export: exposes a bdev as /dev/ublkbN (reference role:
// lib/nbd/nbd.c — the local-mode host attach). The pool's kernels
// ship ublk_drv built in but no nbd module, so ublk is the path that
// produces a REAL kernel block device on a GPU box: the ublk server
// (io_uring URING_CMD fetch/commit loop) moves request payloads
// through pinned per-tag buffers that the HBM engine DMAs directly.
// Requires /dev/ublk-control (mknod'd from /proc/misc when absent —
// the containers have no udev).

#pragma once

#include <cstdint>
#include <string>
#include <vector>

namespace hipstore {

bool ublk_available();

struct UblkDisk {
  int dev_id;
  std::string bdev_name;
  std::string block_path;  // /dev/ublkb<id>
};

// Attach bdev to a new ublk device; returns the created disk (block
// node mknod'd if udev didn't). Throws with errno detail on failure.
UblkDisk ublk_start(const std::string& bdev_name, int queue_depth = 32);
void ublk_stop(int dev_id);
std::vector<UblkDisk> ublk_list();
void ublk_stop_all();

}  // namespace hipstore
