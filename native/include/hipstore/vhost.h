// vhost-user-scsi target: the VM-guest attach path.
//
// MI355X-native replacement for the reference's SPDK vhost-scsi device
// (reference vendor/github.com/spdk/spdk/lib/vhost/vhost_scsi.c and the
// rte_vhost user-space protocol handling): a from-scratch vhost-user
// slave speaking the standard protocol over a unix socket —
// SET_MEM_TABLE guest memory mapped via the passed fds, virtio split
// rings with eventfd kick/call, one worker thread per request queue —
// executing a virtio-scsi / SPC command set (INQUIRY, READ CAPACITY
// 10/16, READ/WRITE 6/10/16, REPORT LUNS, MODE SENSE, SYNCHRONIZE
// CACHE) against hipstored bdevs. HBM bdevs are served through the
// same per-queue engine channels as every other front-end, so guest
// I/O lands in MI355X HBM3E via the LDS-staged copy kernels.
//
// The RPC plane (construct_vhost_scsi_controller / add_vhost_scsi_lun,
// methods.cpp) creates one listening socket per controller at
// <vhost_socket_dir>/<ctrlr>, exactly where SPDK would put it, so a
// vhost-user master (QEMU, or the in-process test client in
// tests/test_vhost.py) can attach without caring which daemon serves.

#pragma once

#include <functional>
#include <memory>
#include <string>

#include "hipstore/bdev.h"

namespace hipstore {

class VhostUserScsiDev;
using VhostDevPtr = std::shared_ptr<VhostUserScsiDev>;

// Directory the per-controller unix sockets are created in (SPDK's
// vhost -S dir). Defaults to the directory of the JSON-RPC socket.
void vhost_set_socket_dir(const std::string& dir);
std::string vhost_socket_path(const std::string& ctrlr_name);

// Start serving <dir>/<name>. resolver(target_num) returns the bdev
// behind LUN 0 of that SCSI target (or nullptr), called per command so
// hot-add/remove through the RPC plane is visible immediately.
// Throws std::runtime_error if the socket cannot be bound.
VhostDevPtr vhost_start(const std::string& name,
                        std::function<BdevPtr(int target)> resolver);

// virtio-blk personality (SPDK's vhost-user-blk twin): one bdev, all
// rings are request queues, 16-byte virtio_blk_outhdr + trailing
// status byte. resolver(0) supplies the bdev.
VhostDevPtr vhost_start_blk(const std::string& name,
                            std::function<BdevPtr(int)> resolver,
                            bool readonly);

// Stop the accept loop and all ring workers, unlink the socket. Safe
// to call twice. Must NOT be called while holding locks the resolver
// takes (workers may be blocked in it).
void vhost_stop(const VhostDevPtr& dev);

// Daemon shutdown: stop every live device.
void vhost_stop_all();

}  // namespace hipstore
