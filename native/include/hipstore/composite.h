// Striped / replicated composite bdevs (BASELINE config 5).
//
// On an 8-GPU MI355X node, a striped malloc bdev shards its block
// address space across per-GPU HBM children (stripe units round-robin
// over the children), and a replicated bdev mirrors writes to every
// child. Replica fan-out between HBM children uses direct xGMI
// point-to-point copies (hipMemcpyPeerAsync) from the primary's
// backing store — the per-link-direct topology SURVEY.md section 2.4
// calls for (7 x ~153 GB/s links per GPU; a host-side 8x PCIe write
// would bottleneck at ~63 GB/s) — overlapped on a dedicated stream.
// CPU children (tests) fall back to host-side mirrored writes.

#pragma once

#include <vector>

#include "hipstore/bdev.h"

namespace hipstore {

// `stripe_size` bytes per stripe unit; must be a multiple of the
// children's block size. All children must share block size and size.
BdevPtr create_striped_bdev(const std::string& name,
                            std::vector<BdevPtr> children,
                            uint64_t stripe_size);

// Mirror: every child holds a full copy. Reads round-robin across
// children; writes land on the primary and fan out over xGMI.
BdevPtr create_replicated_bdev(const std::string& name,
                               std::vector<BdevPtr> children);

}  // namespace hipstore
