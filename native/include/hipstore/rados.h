// RADOS-protocol RBD path: wire client + loopback fake cluster.
//
// Replaces the reference's librados/librbd-backed RBD bdev
// (vendor/github.com/spdk/spdk/lib/bdev/rbd/bdev_rbd.c) with a
// from-scratch msgr-v1 client speaking to an in-repo fake cluster
// over TCP — the NVMe/TCP loopback pattern (nvmf_target.cpp) applied
// to RADOS, since this environment has no external network. The fake
// cluster's object store lives in MI355X HBM when a GPU is present,
// and message data CRC32Cs are computed/verified by the GPU kernel
// against the HBM-resident bytes.

#pragma once

#include <cstdint>
#include <memory>
#include <string>

#include "hipstore/bdev.h"

namespace hipstore {

// A running loopback cluster (single endpoint playing mon+osd; every
// placement group maps to OSD 0 at this address). Objects live in a
// `arena_mb` backing arena: HBM bdev on `device` when use_hbm, host
// RAM otherwise. Granularity: one `object_bytes`-sized slot per
// object (RBD's default 4 MiB object size).
class RadosCluster {
 public:
  virtual ~RadosCluster() = default;
  virtual uint16_t port() const = 0;
  virtual uint64_t object_count() const = 0;
  virtual void stop() = 0;
};

std::shared_ptr<RadosCluster> start_rados_cluster(
    uint16_t port, uint64_t arena_mb, bool use_hbm, int device,
    uint64_t object_bytes = 4ull << 20);

// RBD image bdev over the wire client. `mon_host` is "host:port" of
// the cluster. The image's size lives in the first 8 bytes of its
// rbd_header.<image> object (reduced header; real RBD v2 keeps it in
// omap). If the header object does not exist it is created with
// size = default_size_bytes (the ceph-csi provisioning flow).
// Data objects are rbd_data.<image>.<16-hex block index> of
// `object_bytes` each, the standard RBD layout.
BdevPtr create_rbd_bdev(const std::string& name, const std::string& mon_host,
                        const std::string& pool, const std::string& image,
                        uint64_t block_size, uint64_t default_size_bytes,
                        uint64_t object_bytes = 4ull << 20);

}  // namespace hipstore
