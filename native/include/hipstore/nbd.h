// Kernel NBD export: exposes a bdev as /dev/nbdX (reference
// lib/nbd/nbd.c). The data pump copies NBD socket payloads through
// pinned bounce buffers into HBM via the bdev's async channel.
// Requires the nbd kernel module; nbd_available() gates tests.

#pragma once

#include <string>
#include <utility>
#include <vector>

namespace hipstore {

bool nbd_available();

// Attach bdev to an NBD device node ("/dev/nbd0"); throws on failure.
void nbd_start(const std::string& bdev_name, const std::string& device);
void nbd_stop(const std::string& device);
// (bdev_name, nbd_device) pairs currently exported.
std::vector<std::pair<std::string, std::string>> nbd_list();
void nbd_stop_all();

}  // namespace hipstore
