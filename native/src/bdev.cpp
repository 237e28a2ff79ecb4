#include "hipstore/bdev.h"

#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

#include <cstring>
#include <deque>
#include <new>
#include <random>

namespace hipstore {

std::string Bdev::make_uuid() {
  // Random v4-format UUID; no libuuid dependency.
  static std::mutex mutex;
  static std::mt19937_64 rng{std::random_device{}()};
  std::lock_guard<std::mutex> lock(mutex);
  uint64_t a = rng(), b = rng();
  char buf[40];
  snprintf(buf, sizeof(buf), "%08x-%04x-4%03x-%04x-%012llx",
           static_cast<uint32_t>(a >> 32), static_cast<uint16_t>(a >> 16),
           static_cast<uint16_t>(a & 0xFFF),
           static_cast<uint16_t>(0x8000 | ((b >> 48) & 0x3FFF)),
           static_cast<unsigned long long>(b & 0xFFFFFFFFFFFFULL));
  return buf;
}

BdevManager& BdevManager::instance() {
  static BdevManager manager;
  return manager;
}

bool BdevManager::add(BdevPtr bdev) {
  std::lock_guard<std::mutex> lock(mutex_);
  return bdevs_.emplace(bdev->name(), std::move(bdev)).second;
}

bool BdevManager::remove(const std::string& name) {
  std::lock_guard<std::mutex> lock(mutex_);
  auto it = bdevs_.find(name);
  if (it == bdevs_.end()) return false;
  bdevs_.erase(it);
  return true;
}

BdevPtr BdevManager::find(const std::string& name) const {
  std::lock_guard<std::mutex> lock(mutex_);
  auto it = bdevs_.find(name);
  return it == bdevs_.end() ? nullptr : it->second;
}

std::vector<BdevPtr> BdevManager::list() const {
  std::lock_guard<std::mutex> lock(mutex_);
  std::vector<BdevPtr> out;
  out.reserve(bdevs_.size());
  for (const auto& [_, bdev] : bdevs_) out.push_back(bdev);
  return out;
}

namespace {

// Host-RAM bdev: the CI stand-in for the HBM bdev and the "SPDK Malloc
// bdev on CPU" of BASELINE config 1. I/O executes at submit time (a
// memcpy, like SPDK's default copy engine, reference
// lib/copy/copy_engine.c mem_copy_submit) but completes at poll time to
// preserve the async contract.
class MallocChannel : public IoChannel {
 public:
  std::deque<std::pair<IoCompletion, int>> done;
};

class MallocBdev : public Bdev {
 public:
  MallocBdev(const std::string& name, uint64_t block_size, uint64_t num_blocks)
      : Bdev(name, "Malloc disk", block_size, num_blocks) {
    // Anonymous mmap: zero pages are faulted lazily, so creating a
    // large bdev is O(1) instead of an eager multi-GiB memset (the
    // build container faults large memory at ~100 MB/s).
    void* p = mmap(nullptr, size_bytes(), PROT_READ | PROT_WRITE,
                   MAP_PRIVATE | MAP_ANONYMOUS, -1, 0);
    if (p == MAP_FAILED) throw std::bad_alloc();
    data_ = static_cast<uint8_t*>(p);
  }

  ~MallocBdev() override { munmap(data_, size_bytes()); }

  std::shared_ptr<IoChannel> get_channel() override {
    return std::make_shared<MallocChannel>();
  }

  void submit(IoChannel* ch, IoRequest req) override {
    auto* channel = static_cast<MallocChannel*>(ch);
    int status = kIoOk;
    if (!check_bounds(req)) {
      status = kIoInvalid;
    } else {
      account(req);
      std::lock_guard<std::mutex> lock(mutex_);
      switch (req.op) {
        case IoOp::kRead:
          memcpy(req.buffer, data_ + req.offset, req.length);
          break;
        case IoOp::kWrite:
          memcpy(data_ + req.offset, req.buffer, req.length);
          break;
        case IoOp::kFill:
          memset(data_ + req.offset, req.fill, req.length);
          break;
        case IoOp::kFlush:
          break;
      }
    }
    channel->done.emplace_back(std::move(req.on_complete), status);
  }

  int resize(uint64_t new_num_blocks) override {
    // Channels never hold the data pointer (all access goes through
    // data_ under mutex_ at submit time), so an online swap is safe.
    std::lock_guard<std::mutex> lock(mutex_);
    const uint64_t old_bytes = size_bytes();
    const uint64_t new_bytes = new_num_blocks * block_size();
    if (new_bytes == old_bytes) return kIoOk;
    void* p = mremap(data_, old_bytes, new_bytes, MREMAP_MAYMOVE);
    if (p == MAP_FAILED) return kIoFailed;
    data_ = static_cast<uint8_t*>(p);  // growth pages arrive zeroed
    set_num_blocks(new_num_blocks);
    return kIoOk;
  }

  int poll(IoChannel* ch) override {
    auto* channel = static_cast<MallocChannel*>(ch);
    // Only drain what was queued at entry: completion callbacks may
    // resubmit, which appends to the deque again — draining to empty
    // would never return to the caller.
    size_t n = channel->done.size();
    for (size_t i = 0; i < n; ++i) {
      auto [cb, status] = std::move(channel->done.front());
      channel->done.pop_front();
      if (cb) cb(status);
    }
    return static_cast<int>(n);
  }

 private:
  std::mutex mutex_;  // serializes overlapping I/O from many channels
  uint8_t* data_ = nullptr;
};

// File-backed bdev (SPDK's aio bdev): pread/pwrite against a regular
// file or block device. The only bdev type whose DATA survives daemon
// restarts — malloc bdevs are RAM/HBM by contract.
class FileBdev : public Bdev {
 public:
  FileBdev(const std::string& name, const std::string& path,
           uint64_t block_size, uint64_t num_blocks, int fd)
      : Bdev(name, "AIO disk", block_size, num_blocks),
        path_(path),
        fd_(fd) {}

  ~FileBdev() override { ::close(fd_); }

  const std::string& path() const { return path_; }
  std::string backing_path() const override { return path_; }

  std::shared_ptr<IoChannel> get_channel() override {
    return std::make_shared<MallocChannel>();
  }

  void submit(IoChannel* ch, IoRequest req) override {
    auto* channel = static_cast<MallocChannel*>(ch);
    int status = kIoOk;
    if (!check_bounds(req)) {
      status = kIoInvalid;
    } else {
      account(req);
      switch (req.op) {
        case IoOp::kRead:
          if (full_pread(req.buffer, req.length, req.offset) != 0) {
            status = kIoFailed;
          }
          break;
        case IoOp::kWrite:
          if (full_pwrite(req.buffer, req.length, req.offset) != 0) {
            status = kIoFailed;
          }
          break;
        case IoOp::kFill: {
          if (req.fill == 0 &&
              fallocate(fd_, FALLOC_FL_PUNCH_HOLE | FALLOC_FL_KEEP_SIZE,
                        static_cast<off_t>(req.offset),
                        static_cast<off_t>(req.length)) == 0) {
            break;  // real discard: the extent becomes a hole
          }
          std::vector<uint8_t> zeros(
              std::min<uint64_t>(req.length, 1 << 20), req.fill);
          uint64_t done = 0;
          while (done < req.length && status == kIoOk) {
            uint64_t n = std::min<uint64_t>(zeros.size(), req.length - done);
            if (full_pwrite(zeros.data(), n, req.offset + done) != 0) {
              status = kIoFailed;
            }
            done += n;
          }
          break;
        }
        case IoOp::kFlush:
          if (fdatasync(fd_) != 0) status = kIoFailed;
          break;
      }
    }
    channel->done.emplace_back(std::move(req.on_complete), status);
  }

  int poll(IoChannel* ch) override {
    auto* channel = static_cast<MallocChannel*>(ch);
    size_t n = channel->done.size();
    for (size_t i = 0; i < n; ++i) {
      auto [cb, status] = std::move(channel->done.front());
      channel->done.pop_front();
      if (cb) cb(status);
    }
    return static_cast<int>(n);
  }

  int resize(uint64_t new_num_blocks) override {
    const uint64_t new_bytes = new_num_blocks * block_size();
    if (ftruncate(fd_, static_cast<off_t>(new_bytes)) != 0) return kIoFailed;
    set_num_blocks(new_num_blocks);
    return kIoOk;
  }

 private:
  int full_pread(void* buf, uint64_t len, uint64_t off) {
    uint8_t* p = static_cast<uint8_t*>(buf);
    while (len > 0) {
      ssize_t n = pread(fd_, p, len, static_cast<off_t>(off));
      if (n < 0) return -1;
      if (n == 0) {  // short file (sparse tail): reads as zeros
        memset(p, 0, len);
        return 0;
      }
      p += n;
      off += n;
      len -= n;
    }
    return 0;
  }

  int full_pwrite(const void* buf, uint64_t len, uint64_t off) {
    const uint8_t* p = static_cast<const uint8_t*>(buf);
    while (len > 0) {
      ssize_t n = pwrite(fd_, p, len, static_cast<off_t>(off));
      if (n <= 0) return -1;
      p += n;
      off += n;
      len -= n;
    }
    return 0;
  }

  const std::string path_;
  const int fd_;
};

}  // namespace

BdevPtr create_malloc_bdev(const std::string& name, uint64_t block_size,
                           uint64_t num_blocks) {
  return std::make_shared<MallocBdev>(name, block_size, num_blocks);
}

BdevPtr create_file_bdev(const std::string& name, const std::string& path,
                         uint64_t block_size) {
  // The file must already exist (callers size it first); O_CREAT here
  // would leave stray empty files behind on validation failures.
  int fd = ::open(path.c_str(), O_RDWR | O_CLOEXEC);
  if (fd < 0) {
    throw std::runtime_error("aio bdev: cannot open " + path);
  }
  struct stat st {};
  if (fstat(fd, &st) != 0) {
    ::close(fd);
    throw std::runtime_error("aio bdev: cannot stat " + path);
  }
  uint64_t bytes = static_cast<uint64_t>(st.st_size);
  if (S_ISBLK(st.st_mode)) {
    ::close(fd);
    throw std::runtime_error(
        "aio bdev: raw block devices are not supported here; use a file");
  }
  if (bytes == 0 || bytes % block_size != 0) {
    ::close(fd);
    throw std::runtime_error(
        "aio bdev: file size must be a non-zero multiple of block_size "
        "(create/truncate the file first)");
  }
  return std::make_shared<FileBdev>(name, path, block_size,
                                    bytes / block_size, fd);
}

}  // namespace hipstore
