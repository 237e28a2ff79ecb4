// Native vhost-user master benchmark (vhost_master.cpp): drives the
// daemon's vhost-user slave at native speed — the Python master
// (oim_amd/bench/vhost_client.py) stays the conformance tool, this
// measures the host-attach data path's real ceiling.

#pragma once

#include <cstdint>
#include <string>
#include <vector>

#include "hipstore/engine.h"

namespace hipstore {

// One request ring's master-side state (POD; internals of
// VhostMasterSession, public only for the vector member).
struct VhostMasterRing {
  uint8_t* mem = nullptr;
  uint64_t base = 0;
  uint32_t qsize = 0;
  int kick = -1;
  int call = -1;
  uint64_t desc_off = 0, avail_off = 0, used_off = 0, req_off = 0,
           data_off = 0;
  uint16_t avail_idx = 0;
  uint16_t used_idx = 0;
};

// A live vhost-user master session: the handshake, guest-memory
// SET_MEM_TABLE (which the slave hipHostRegisters — expensive) and
// ring setup happen ONCE in the constructor; each run() drives
// `total_ios` I/Os through the standing rings. Used by bench.py's
// --frontend vhost so per-step numbers measure the data path, not
// session bring-up.
class VhostMasterSession {
 public:
  VhostMasterSession(const std::string& socket_path,
                     const std::string& personality, int num_rings,
                     int iodepth, uint32_t io_size, uint32_t block_size,
                     uint64_t capacity_bytes);
  ~VhostMasterSession();
  VhostMasterSession(const VhostMasterSession&) = delete;
  VhostMasterSession& operator=(const VhostMasterSession&) = delete;

  PerfResult run(uint64_t total_ios, const std::string& workload);

 private:
  std::string personality_;
  int num_rings_;
  int iodepth_;
  uint32_t io_size_;
  uint32_t block_size_;
  uint64_t capacity_bytes_;
  int sock_ = -1;
  int memfd_ = -1;
  uint8_t* mem_ = nullptr;
  uint64_t mem_size_ = 0;
  uint64_t run_seq_ = 0;
  std::vector<VhostMasterRing> rings_;
};

// One-shot convenience wrapper (tests): one session, one run.
PerfResult vhost_master_bench(const std::string& socket_path,
                              const std::string& personality,
                              int num_rings, int iodepth, uint32_t io_size,
                              const std::string& workload,
                              uint64_t total_ios, uint32_t block_size,
                              uint64_t capacity_bytes);

}  // namespace hipstore
