// Block-device core of hipstored.
//
// MI355X-native replacement for the bdev subset of the reference's SPDK
// data path (reference vendor/github.com/spdk/spdk/lib/bdev): a named
// block device with async read/write/fill I/O driven through per-queue
// channels. Concrete bdevs:
//   - MallocBdev (cpu_bdev.cpp): host-RAM backing — config 1 / CI fake
//     (reference lib/bdev/malloc/bdev_malloc.c semantics)
//   - HbmBdev (gpu.hip): hipMalloc backing in MI355X HBM3E; I/O runs as
//     LDS-staged HIP kernels on per-queue streams
//   - RbdBdev / NvmfBdev: network paths (later milestones)

#pragma once

#include <atomic>
#include <cstdint>
#include <functional>
#include <map>
#include <memory>
#include <mutex>
#include <string>
#include <vector>

namespace hipstore {

enum class IoOp : uint8_t {
  kRead = 0,    // bdev -> host buffer
  kWrite = 1,   // host buffer -> bdev
  kFill = 2,    // pattern fill (write_zeroes / unmap)
  kFlush = 3,   // no-op barrier for RAM/HBM bdevs
};

enum IoStatus : int {
  kIoOk = 0,
  kIoInvalid = -22,   // out-of-range / misaligned
  kIoFailed = -5,
};

using IoCompletion = std::function<void(int status)>;

struct IoRequest {
  IoOp op = IoOp::kRead;
  uint64_t offset = 0;  // bytes; must be block-aligned
  uint64_t length = 0;  // bytes; must be a multiple of block_size
  void* buffer = nullptr;  // host buffer for read/write
  uint8_t fill = 0;
  IoCompletion on_complete;  // invoked from poll()
};

// One submission context (SPDK io_channel analog). Channels are not
// thread-safe; each submitting thread owns its own.
class IoChannel {
 public:
  virtual ~IoChannel() = default;
};

class Bdev {
 public:
  Bdev(std::string name, std::string product, uint64_t block_size,
       uint64_t num_blocks)
      : name_(std::move(name)),
        product_(std::move(product)),
        uuid_(make_uuid()),
        block_size_(block_size),
        num_blocks_(num_blocks) {}
  virtual ~Bdev() = default;

  const std::string& name() const { return name_; }
  const std::string& product_name() const { return product_; }
  // Factories re-badge wrapped bdevs (e.g. the RBD emulation reuses the
  // HBM backing but must not report "Malloc disk" — the controller's
  // unmap logic keys on that string).
  void set_product(const std::string& product) { product_ = product; }
  const std::string& uuid() const { return uuid_; }
  uint64_t block_size() const { return block_size_; }
  uint64_t num_blocks() const { return num_blocks_; }
  uint64_t size_bytes() const { return block_size_ * num_blocks_; }

  bool claimed() const { return claimed_.load(); }
  bool claim() {
    bool expected = false;
    return claimed_.compare_exchange_strong(expected, true);
  }
  void release() { claimed_.store(false); }

  // HBM bdevs expose their device-resident backing store for on-GPU
  // compute (CRC32C, verify, striping); others return nullptr/-1.
  virtual void* device_base() { return nullptr; }
  virtual int gpu_device() const { return -1; }
  // File-backed bdevs report their backing path (get_bdevs
  // driver_specific, SPDK aio shape); others return "".
  virtual std::string backing_path() const { return {}; }

  virtual std::shared_ptr<IoChannel> get_channel() = 0;
  // Grow (or shrink) the device to new_num_blocks. Returns kIoOk, or
  // kIoInvalid when the bdev type cannot resize, or kIoFailed when it
  // temporarily cannot (e.g. live channels hold the old backing).
  virtual int resize(uint64_t new_num_blocks) {
    (void)new_num_blocks;
    return kIoInvalid;
  }
  // Enqueue asynchronously; completion fires from poll() on the
  // submitting thread (the SPDK poller discipline, which is what makes
  // completions race-free without locks).
  virtual void submit(IoChannel* ch, IoRequest req) = 0;
  // Drive completions; returns number completed.
  virtual int poll(IoChannel* ch) = 0;

  // Cumulative I/O statistics (updated at submit; SPDK's
  // get_bdevs_iostat shape).
  struct IoStat {
    std::atomic<uint64_t> num_read_ops{0};
    std::atomic<uint64_t> num_write_ops{0};
    std::atomic<uint64_t> num_unmap_ops{0};
    std::atomic<uint64_t> bytes_read{0};
    std::atomic<uint64_t> bytes_written{0};
  };
  IoStat& iostat() { return iostat_; }
  void account(const IoRequest& req) {
    switch (req.op) {
      case IoOp::kRead:
        iostat_.num_read_ops.fetch_add(1, std::memory_order_relaxed);
        iostat_.bytes_read.fetch_add(req.length, std::memory_order_relaxed);
        break;
      case IoOp::kWrite:
        iostat_.num_write_ops.fetch_add(1, std::memory_order_relaxed);
        iostat_.bytes_written.fetch_add(req.length, std::memory_order_relaxed);
        break;
      case IoOp::kFill:
        iostat_.num_unmap_ops.fetch_add(1, std::memory_order_relaxed);
        break;
      case IoOp::kFlush:
        break;
    }
  }

  bool check_bounds(const IoRequest& req) const {
    return req.length > 0 && req.offset % block_size_ == 0 &&
           req.length % block_size_ == 0 &&
           req.offset + req.length <= size_bytes();
  }

 protected:
  // For resize implementations only; callers must have moved/zeroed
  // the backing store to match first.
  void set_num_blocks(uint64_t n) { num_blocks_ = n; }

 private:
  static std::string make_uuid();

  std::string name_;
  std::string product_;
  std::string uuid_;
  uint64_t block_size_;
  uint64_t num_blocks_;
  std::atomic<bool> claimed_{false};
  IoStat iostat_;
};

using BdevPtr = std::shared_ptr<Bdev>;

// Name -> bdev registry (SPDK's global bdev list).
class BdevManager {
 public:
  // Returns false if the name exists.
  bool add(BdevPtr bdev);
  // Returns false if absent or claimed.
  bool remove(const std::string& name);
  BdevPtr find(const std::string& name) const;
  std::vector<BdevPtr> list() const;

  static BdevManager& instance();

 private:
  mutable std::mutex mutex_;
  std::map<std::string, BdevPtr> bdevs_;
};

// Host-RAM malloc bdev (reference bdev_malloc.c contract: data persists
// until the bdev is deleted; reads/writes are copies through the copy
// engine — here a synchronous memcpy completed on next poll()).
BdevPtr create_malloc_bdev(const std::string& name, uint64_t block_size,
                           uint64_t num_blocks);

// File-backed bdev (SPDK aio bdev): geometry from the file's current
// size; data persists across daemon restarts.
BdevPtr create_file_bdev(const std::string& name, const std::string& path,
                         uint64_t block_size);

}  // namespace hipstore
