// GPU data-path engine API (implemented in gpu.hip).
//
// MI355X-native replacement for the reference's SPDK copy engine +
// malloc bdev pair (reference lib/copy/copy_engine.c,
// lib/bdev/malloc/bdev_malloc.c): block I/O executes as batched HIP
// kernel launches on per-queue streams, 4 KiB tiles staged through LDS
// (visible in rocprof as hipstore::k_copy_blocks), with pinned-host
// descriptor rings read by the GPU directly.

#pragma once

#include <cstdint>
#include <map>
#include <string>
#include <vector>

#include "hipstore/bdev.h"

namespace hipstore {

// True when at least one HIP device is present (cached).
bool gpu_available();
int gpu_device_count();
// PCI BDF string ("0000:c1:00.0") of a device, for registry <id>/pci.
std::string gpu_pci_address(int device);

// Create an HBM-resident malloc bdev on `device`. Throws std::runtime_error
// when no GPU or allocation fails. product_name stays "Malloc disk": the
// HBM bdev IS the malloc bdev on this platform, and the reference
// controller keys its keep-on-unmap logic on that string
// (reference controller.go:205).
// `persistent` selects the on-GPU polling service kernel (lowest
// latency; waves poll the submission ring and self-exit when idle)
// instead of batched per-poll launches.
BdevPtr create_hbm_bdev(const std::string& name, uint64_t block_size,
                        uint64_t num_blocks, int device,
                        bool persistent = false);

// Persistent-engine debug counters: 0=launches, 1=relaunches,
// 2=stall queries.
uint64_t persistent_stat(int which);

// Hardware liveness probe for the persistent-service idioms: launches
// a minimal leader+worker kernel pair and reports which legs of the
// host<->GPU polling contract work on this box (heartbeats, host-tail
// visibility, CQ publish, cross-workgroup relay). Keys: launch_err,
// hb_host_early/final, hb_dev_early/final, relay_final, cq0_ms,
// cq1_ms (-1 = never seen), stream_drained.
// flags: 1 atomics, 2 pre-launch pageable H2D memcpyAsync on the
// kernel's stream, 4 worker agent fetch_adds, 8 service-sized grid.
std::map<std::string, long long> persistent_probe(int device, int flags);

// Launches the real persistent-copy service kernel with a minimal
// standalone setup and serves one descriptor (0 = pinned->pinned,
// 1 = pinned->HBM, 2 = HBM->pinned).
std::map<std::string, long long> persistent_kernel_probe(int device,
                                                         int variant);

// HBM capacity of `device`: (total_bytes, free_bytes) via
// hipMemGetInfo; (0, 0) without a GPU.
std::pair<uint64_t, uint64_t> hbm_info(int device);

// Pinned-host buffer helpers (fall back to plain malloc without a GPU).
void* alloc_pinned(size_t bytes);
void free_pinned(void* ptr);

// Device-side range copy between HBM bdevs (volume clone / stripe
// rebuild): same-device copies run the LDS-staged block kernel at HBM
// rates; cross-device copies go over xGMI peer access. Returns
// IoStatus; kIoInvalid when either bdev is not HBM-resident or ranges
// are out of bounds/misaligned.
int hbm_copy_sync(Bdev* src, uint64_t src_offset, Bdev* dst,
                  uint64_t dst_offset, uint64_t length);

// Synchronous convenience wrappers (tests, NBD pump): create a
// throwaway channel, submit, poll to completion. Returns IoStatus.
int bdev_read_sync(Bdev* bdev, uint64_t offset, void* buf, uint64_t len);
int bdev_write_sync(Bdev* bdev, uint64_t offset, const void* buf, uint64_t len);
int bdev_fill_sync(Bdev* bdev, uint64_t offset, uint8_t value, uint64_t len);

// fio-shaped benchmark harness (SPDK bdevperf analog): `workload` is
// randread / randwrite / randrw; runs num_queues submitter threads each
// holding queue_depth I/Os in flight for `seconds`, then reports
// latency percentiles measured per-I/O.
struct PerfResult {
  double seconds = 0;
  uint64_t io_count = 0;
  double iops = 0;
  double throughput_mbps = 0;
  double lat_avg_us = 0;
  double lat_p50_us = 0;
  double lat_p90_us = 0;
  double lat_p99_us = 0;
  double lat_p999_us = 0;
  double lat_max_us = 0;
};

PerfResult run_bdevperf(Bdev* bdev, const std::string& workload,
                        uint32_t io_size, uint32_t queue_depth,
                        int num_queues, double seconds,
                        uint64_t max_ios = 0);

// Persistent-queue variant for stepped benchmarking (bench.py): queue
// threads, their I/O channels (HIP streams + pinned rings) and buffers
// live across step() calls, so a step measures steady-state IOPS
// without per-call setup.
class PerfSession {
 public:
  PerfSession(BdevPtr bdev, std::string workload, uint32_t io_size,
              uint32_t queue_depth, int num_queues);
  ~PerfSession();

  // Run until `total_ios` I/Os complete (spread over the queues);
  // returns the step's aggregate stats.
  PerfResult step(uint64_t total_ios);

 private:
  struct Impl;
  std::unique_ptr<Impl> impl_;
};

}  // namespace hipstore
