// Native vhost-user master benchmark (vhost_master.cpp): drives the
// daemon's vhost-user slave at native speed — the Python master
// (oim_amd/bench/vhost_client.py) stays the conformance tool, this
// measures the host-attach data path's real ceiling.

#pragma once

#include <cstdint>
#include <string>

#include "hipstore/engine.h"

namespace hipstore {

// Complete `total_ios` I/Os of `io_size` across `num_rings` request
// rings (each keeping `iodepth` chains outstanding) against the
// vhost-user socket at `socket_path`. `personality` is "scsi" or
// "blk"; `workload` randread / randwrite / randrw. `capacity_bytes` /
// `block_size` describe the exported LUN (offsets are generated below
// capacity).
PerfResult vhost_master_bench(const std::string& socket_path,
                              const std::string& personality,
                              int num_rings, int iodepth, uint32_t io_size,
                              const std::string& workload,
                              uint64_t total_ios, uint32_t block_size,
                              uint64_t capacity_bytes);

}  // namespace hipstore
