// NVMe-oF/TCP initiator bdev + target (BASELINE config 3).
//
// MI355X-native replacement for the reference's SPDK NVMe/TCP path
// (reference lib/nvme/nvme_tcp.c, lib/bdev/nvme): a from-scratch
// NVMe/TCP PDU state machine. The initiator exposes a remote namespace
// as a local bdev (one TCP connection per I/O channel, polled — no
// threads in the data path); the target exports hipstored bdevs as
// namespaces so the pair is loopback-testable in this image (no
// external storage network exists here).
//
// Data digests (DDGST, CRC32C) are negotiated on both sides. For
// HBM-resident target namespaces the digest of each C2HData payload is
// produced by the GPU: k_crc32c_blocks computes per-4 KiB-block CRCs in
// parallel and the host folds them with crc32c_combine.

#pragma once

#include <cstdint>
#include <memory>
#include <string>
#include <vector>

#include "hipstore/bdev.h"

namespace hipstore {

class NvmfTcpTarget {
 public:
  virtual ~NvmfTcpTarget() = default;
  virtual uint16_t port() const = 0;  // bound port (0 requested -> actual)
  virtual void stop() = 0;
  // nsid = index into the namespace list + 1.
  virtual void add_namespace(BdevPtr bdev) = 0;
};

std::shared_ptr<NvmfTcpTarget> start_nvmf_tcp_target(
    const std::string& listen_addr, uint16_t port, const std::string& subnqn,
    bool enable_digests = true);

// Connects (admin queue: Connect + enable controller + Identify
// namespace) and returns a bdev sized from the remote namespace.
// Throws on connection/protocol errors.
BdevPtr create_nvmf_tcp_bdev(const std::string& name,
                             const std::string& traddr, uint16_t trsvcid,
                             const std::string& subnqn, uint32_t nsid = 1,
                             bool enable_digests = true);

}  // namespace hipstore
