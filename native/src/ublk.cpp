// ublk server: exposes a bdev as a real kernel block device
// (/dev/ublkbN) on boxes whose kernels ship ublk_drv (the GPU pool's
// do; their nbd module is absent). Reference role: lib/nbd/nbd.c —
// the local-mode host attach behind start_nbd_disk.
//
// From-scratch io_uring + ublk UAPI (no liburing in the image):
//   - /dev/ublk-control: SQE128 io_uring URING_CMDs (ADD_DEV /
//     SET_PARAMS / START_DEV / STOP_DEV / DEL_DEV, ioctl-encoded).
//   - /dev/ublkcN: per-queue URING_CMD FETCH_REQ /
//     COMMIT_AND_FETCH_REQ loop over the mmap'd descriptor area.
//   - Request payloads move through per-tag PINNED buffers
//     (alloc_pinned), so the HBM engine DMAs guest I/O directly —
//     kernel <-> pinned bounce <-> HBM, one hop each way.
// The containers run no udev: missing /dev nodes (ublk-control from
// /proc/misc, ublkcN/ublkbN from GET_PARAMS' devt majors) are
// mknod'd here.

#include <fcntl.h>
#include <linux/ioctl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <sys/sysmacros.h>
#include <sys/syscall.h>
#include <unistd.h>

#include <atomic>
#include <chrono>
#include <cerrno>
#include <cstdio>
#include <cstring>
#include <map>
#include <mutex>
#include <stdexcept>
#include <string>
#include <thread>
#include <vector>

#include "hipstore/bdev.h"
#include "hipstore/engine.h"
#include "hipstore/ublk.h"

namespace hipstore {

namespace {

// ---------------------------------------------------------------------------
// io_uring (raw syscalls; SQE128 rings as ublk requires)
// ---------------------------------------------------------------------------

struct IoSqringOffsets {
  uint32_t head, tail, ring_mask, ring_entries, flags, dropped, array, resv1;
  uint64_t user_addr;
};

struct IoCqringOffsets {
  uint32_t head, tail, ring_mask, ring_entries, overflow, cqes, flags, resv1;
  uint64_t user_addr;
};

struct IoUringParams {
  uint32_t sq_entries, cq_entries, flags, sq_thread_cpu, sq_thread_idle;
  uint32_t features, wq_fd, resv[3];
  IoSqringOffsets sq_off;
  IoCqringOffsets cq_off;
};

struct IoUringCqe {
  uint64_t user_data;
  int32_t res;
  uint32_t flags;
};

constexpr uint32_t kSetupSqe128 = 1u << 10;
constexpr uint32_t kEnterGetevents = 1u;
constexpr uint64_t kOffSqRing = 0;
constexpr uint64_t kOffCqRing = 0x8000000ull;
constexpr uint64_t kOffSqes = 0x10000000ull;
constexpr uint32_t kFeatSingleMmap = 1u;
constexpr uint8_t kOpUringCmd = 46;

int sys_io_uring_setup(unsigned entries, IoUringParams* p) {
  return static_cast<int>(syscall(425, entries, p));
}

int sys_io_uring_enter(int fd, unsigned to_submit, unsigned min_complete,
                       unsigned flags) {
  return static_cast<int>(
      syscall(426, fd, to_submit, min_complete, flags, nullptr, 0));
}

// A SQE128 ring. Single-threaded use per instance.
class UringSqe128 {
 public:
  void init(unsigned entries) {
    memset(&params_, 0, sizeof(params_));
    params_.flags = kSetupSqe128;
    fd_ = sys_io_uring_setup(entries, &params_);
    if (fd_ < 0) {
      throw std::runtime_error(std::string("io_uring_setup: ") +
                               strerror(errno));
    }
    const size_t sq_sz = params_.sq_off.array + params_.sq_entries * 4;
    const size_t cq_sz =
        params_.cq_off.cqes + params_.cq_entries * sizeof(IoUringCqe);
    sq_ring_sz_ = std::max(sq_sz, cq_sz);
    sq_ring_ = mmap(nullptr, sq_ring_sz_, PROT_READ | PROT_WRITE,
                    MAP_SHARED | MAP_POPULATE, fd_, kOffSqRing);
    if (sq_ring_ == MAP_FAILED) {
      throw std::runtime_error("io_uring: sq ring mmap failed");
    }
    if (params_.features & kFeatSingleMmap) {
      cq_ring_ = sq_ring_;
    } else {
      cq_ring_ = mmap(nullptr, cq_sz, PROT_READ | PROT_WRITE,
                      MAP_SHARED | MAP_POPULATE, fd_, kOffCqRing);
      if (cq_ring_ == MAP_FAILED) {
        throw std::runtime_error("io_uring: cq ring mmap failed");
      }
    }
    sqes_sz_ = params_.sq_entries * 128;
    sqes_ = mmap(nullptr, sqes_sz_, PROT_READ | PROT_WRITE,
                 MAP_SHARED | MAP_POPULATE, fd_, kOffSqes);
    if (sqes_ == MAP_FAILED) {
      throw std::runtime_error("io_uring: sqes mmap failed");
    }
    sq_head_ = ring_u32(sq_ring_, params_.sq_off.head);
    sq_tail_ = ring_u32(sq_ring_, params_.sq_off.tail);
    sq_mask_ = *ring_u32(sq_ring_, params_.sq_off.ring_mask);
    sq_array_ = ring_u32(sq_ring_, params_.sq_off.array);
    cq_head_ = ring_u32(cq_ring_, params_.cq_off.head);
    cq_tail_ = ring_u32(cq_ring_, params_.cq_off.tail);
    cq_mask_ = *ring_u32(cq_ring_, params_.cq_off.ring_mask);
    cqes_ = reinterpret_cast<IoUringCqe*>(
        static_cast<uint8_t*>(cq_ring_) + params_.cq_off.cqes);
  }

  ~UringSqe128() {
    if (sqes_ != nullptr && sqes_ != MAP_FAILED) munmap(sqes_, sqes_sz_);
    if (cq_ring_ != nullptr && cq_ring_ != sq_ring_ &&
        cq_ring_ != MAP_FAILED) {
      munmap(cq_ring_, sq_ring_sz_);
    }
    if (sq_ring_ != nullptr && sq_ring_ != MAP_FAILED) {
      munmap(sq_ring_, sq_ring_sz_);
    }
    if (fd_ >= 0) close(fd_);
  }

  // Queue one URING_CMD sqe (not yet submitted to the kernel).
  void push_cmd(int fd, uint32_t cmd_op, const void* cmd, size_t cmd_len,
                uint64_t user_data) {
    const uint32_t tail = *sq_tail_;
    const uint32_t idx = tail & sq_mask_;
    uint8_t* sqe = static_cast<uint8_t*>(sqes_) + idx * 128;
    memset(sqe, 0, 128);
    sqe[0] = kOpUringCmd;                       // opcode
    memcpy(sqe + 4, &fd, 4);                    // fd
    memcpy(sqe + 8, &cmd_op, 4);                // cmd_op (union w/ off)
    memcpy(sqe + 32, &user_data, 8);            // user_data
    memcpy(sqe + 48, cmd, cmd_len);             // cmd payload (SQE128)
    sq_array_[idx] = idx;
    __atomic_store_n(sq_tail_, tail + 1, __ATOMIC_RELEASE);
    ++pending_;
  }

  // Submit queued sqes; wait for at least `min_complete` completions.
  int submit(unsigned min_complete) {
    const unsigned to_submit = pending_;
    pending_ = 0;
    int r = sys_io_uring_enter(fd_, to_submit, min_complete,
                               min_complete ? kEnterGetevents : 0);
    if (r < 0 && errno == EINTR) r = 0;
    return r;
  }

  // Pop one completion if available.
  bool pop(IoUringCqe* out) {
    const uint32_t head = *cq_head_;
    if (head == __atomic_load_n(cq_tail_, __ATOMIC_ACQUIRE)) return false;
    *out = cqes_[head & cq_mask_];
    __atomic_store_n(cq_head_, head + 1, __ATOMIC_RELEASE);
    return true;
  }

  int fd() const { return fd_; }

 private:
  static uint32_t* ring_u32(void* base, uint32_t off) {
    return reinterpret_cast<uint32_t*>(static_cast<uint8_t*>(base) + off);
  }

  int fd_ = -1;
  IoUringParams params_{};
  void* sq_ring_ = nullptr;
  void* cq_ring_ = nullptr;
  void* sqes_ = nullptr;
  size_t sq_ring_sz_ = 0;
  size_t sqes_sz_ = 0;
  uint32_t* sq_head_ = nullptr;
  uint32_t* sq_tail_ = nullptr;
  uint32_t sq_mask_ = 0;
  uint32_t* sq_array_ = nullptr;
  uint32_t* cq_head_ = nullptr;
  uint32_t* cq_tail_ = nullptr;
  uint32_t cq_mask_ = 0;
  IoUringCqe* cqes_ = nullptr;
  unsigned pending_ = 0;
};

// ---------------------------------------------------------------------------
// ublk UAPI (linux/ublk_cmd.h layouts; ioctl-encoded command opcodes)
// ---------------------------------------------------------------------------

#pragma pack(push, 1)
struct UblkCtrlCmd {
  uint32_t dev_id;
  uint16_t queue_id;
  uint16_t len;
  uint64_t addr;
  uint64_t data0;
  uint16_t dev_path_len;
  uint16_t pad;
  uint32_t reserved;
};
static_assert(sizeof(UblkCtrlCmd) == 32, "ublksrv_ctrl_cmd");

struct UblkCtrlDevInfo {
  uint16_t nr_hw_queues;
  uint16_t queue_depth;
  uint16_t state;
  uint16_t pad0;
  uint32_t max_io_buf_bytes;
  uint32_t dev_id;
  int32_t ublksrv_pid;
  uint32_t pad1;
  uint64_t flags;
  uint64_t ublksrv_flags;
  uint32_t owner_uid;
  uint32_t owner_gid;
  uint64_t reserved1;
  uint64_t reserved2;
};
static_assert(sizeof(UblkCtrlDevInfo) == 64, "ublksrv_ctrl_dev_info");

struct UblkIoDesc {
  uint32_t op_flags;
  uint32_t nr_sectors;
  uint64_t start_sector;
  uint64_t addr;
};
static_assert(sizeof(UblkIoDesc) == 24, "ublksrv_io_desc");

struct UblkIoCmd {
  uint16_t q_id;
  uint16_t tag;
  int32_t result;
  uint64_t addr;
};
static_assert(sizeof(UblkIoCmd) == 16, "ublksrv_io_cmd");

struct UblkParamBasic {
  uint32_t attrs;
  uint8_t logical_bs_shift;
  uint8_t physical_bs_shift;
  uint8_t io_opt_shift;
  uint8_t io_min_shift;
  uint32_t max_sectors;
  uint32_t chunk_sectors;
  uint64_t dev_sectors;
  uint64_t virt_boundary_mask;
};
static_assert(sizeof(UblkParamBasic) == 32, "ublk_param_basic");

struct UblkParamDiscard {
  uint32_t discard_alignment;
  uint32_t discard_granularity;
  uint32_t max_discard_sectors;
  uint32_t max_write_zeroes_sectors;
  uint16_t max_discard_segments;
  uint16_t reserved0;
};
static_assert(sizeof(UblkParamDiscard) == 20, "ublk_param_discard");

struct UblkParamDevt {
  uint32_t char_major;
  uint32_t char_minor;
  uint32_t disk_major;
  uint32_t disk_minor;
};

struct UblkParamZoned {
  uint32_t max_open_zones;
  uint32_t max_active_zones;
  uint32_t max_zone_append_sectors;
  uint8_t reserved[20];
};

struct UblkParams {
  uint32_t len;
  uint32_t types;
  UblkParamBasic basic;
  UblkParamDiscard discard;
  UblkParamDevt devt;
  UblkParamZoned zoned;
};
#pragma pack(pop)

constexpr uint32_t kParamTypeBasic = 1;
constexpr uint32_t kParamTypeDiscard = 2;
constexpr uint32_t kParamTypeDevt = 4;

constexpr uint64_t kFlagCmdIoctlEncode = 1ull << 6;

// _IOR/_IOWR('u', nr, struct ublksrv_ctrl_cmd) — the ioctl encoding
// modern kernels require for ublk URING_CMDs.
constexpr uint32_t ublk_ctrl_ior(uint32_t nr) {
  return (2u << 30) | (sizeof(UblkCtrlCmd) << 16) | ('u' << 8) | nr;
}
constexpr uint32_t ublk_ctrl_iowr(uint32_t nr) {
  return (3u << 30) | (sizeof(UblkCtrlCmd) << 16) | ('u' << 8) | nr;
}
constexpr uint32_t ublk_io_iowr(uint32_t nr) {
  return (3u << 30) | (sizeof(UblkIoCmd) << 16) | ('u' << 8) | nr;
}

[[maybe_unused]] constexpr uint32_t kCmdGetDevInfo = ublk_ctrl_ior(0x02);
constexpr uint32_t kCmdAddDev = ublk_ctrl_iowr(0x04);
constexpr uint32_t kCmdDelDev = ublk_ctrl_iowr(0x05);
constexpr uint32_t kCmdStartDev = ublk_ctrl_iowr(0x06);
constexpr uint32_t kCmdStopDev = ublk_ctrl_iowr(0x07);
constexpr uint32_t kCmdSetParams = ublk_ctrl_iowr(0x08);
constexpr uint32_t kCmdGetParams = ublk_ctrl_ior(0x09);
[[maybe_unused]] constexpr uint32_t kCmdGetFeatures = ublk_ctrl_ior(0x13);
constexpr uint32_t kIoFetchReq = ublk_io_iowr(0x20);
constexpr uint32_t kIoCommitAndFetchReq = ublk_io_iowr(0x21);

// ublksrv_io_desc op codes (op_flags & 0xff)
constexpr uint32_t kUblkOpRead = 0;
constexpr uint32_t kUblkOpWrite = 1;
constexpr uint32_t kUblkOpFlush = 2;
constexpr uint32_t kUblkOpDiscard = 3;
[[maybe_unused]] constexpr uint32_t kUblkOpWriteSame = 4;
constexpr uint32_t kUblkOpWriteZeroes = 5;

constexpr uint32_t kMaxQueueDepthUapi = 4096;  // UBLK_MAX_QUEUE_DEPTH

// ---------------------------------------------------------------------------
// device node helpers (no udev in the containers)
// ---------------------------------------------------------------------------

void ensure_node(const std::string& path, mode_t type, uint32_t major,
                 uint32_t minor) {
  struct stat st{};
  if (stat(path.c_str(), &st) == 0) return;
  if (mknod(path.c_str(), type | 0600,
            makedev(major, minor)) != 0 && errno != EEXIST) {
    throw std::runtime_error("mknod " + path + ": " + strerror(errno));
  }
}

// Opens /dev/ublk-control, mknod'ing it from /proc/misc when udev
// never created it. On failure *reason says which step failed.
int open_control(std::string* reason = nullptr) {
  const char* path = "/dev/ublk-control";
  int fd = open(path, O_RDWR);
  if (fd >= 0) return fd;
  FILE* f = fopen("/proc/misc", "r");
  if (f == nullptr) {
    if (reason) *reason = "/proc/misc unreadable";
    return -1;
  }
  char name[64];
  int minor = -1, m;
  while (fscanf(f, "%d %63s", &m, name) == 2) {
    if (strcmp(name, "ublk-control") == 0) {
      minor = m;
      break;
    }
  }
  fclose(f);
  if (minor < 0) {
    if (reason) *reason = "no ublk-control in /proc/misc (driver absent)";
    return -1;
  }
  if (mknod(path, S_IFCHR | 0600, makedev(10, minor)) != 0 &&
      errno != EEXIST) {
    if (reason) {
      *reason = std::string("mknod /dev/ublk-control: ") + strerror(errno);
    }
    return -1;
  }
  fd = open(path, O_RDWR);
  if (fd < 0 && reason) {
    *reason = std::string("open /dev/ublk-control: ") + strerror(errno);
  }
  return fd;
}

// One blocking control command round trip on its own tiny ring.
int ctrl_cmd(UringSqe128* ring, int ctrl_fd, uint32_t cmd_op,
             uint32_t dev_id, void* payload, uint16_t payload_len) {
  UblkCtrlCmd cmd{};
  cmd.dev_id = dev_id;
  cmd.addr = reinterpret_cast<uint64_t>(payload);
  cmd.len = payload_len;
  ring->push_cmd(ctrl_fd, cmd_op, &cmd, sizeof(cmd), cmd_op);
  if (ring->submit(1) < 0) return -errno;
  IoUringCqe cqe{};
  while (!ring->pop(&cqe)) {
    if (sys_io_uring_enter(ring->fd(), 0, 1, kEnterGetevents) < 0 &&
        errno != EINTR) {
      return -errno;
    }
  }
  return cqe.res;
}

// ---------------------------------------------------------------------------
// the server
// ---------------------------------------------------------------------------

class UblkServer {
 public:
  UblkServer(BdevPtr bdev, int queue_depth)
      : bdev_(std::move(bdev)),
        queue_depth_(std::min(std::max(queue_depth, 1), 512)) {}

  ~UblkServer() { stop(); }

  UblkDisk start() {
    std::string reason;
    ctrl_fd_ = open_control(&reason);
    if (ctrl_fd_ < 0) {
      throw std::runtime_error("ublk: " + reason);
    }
    ctrl_ring_.init(4);

    UblkCtrlDevInfo info{};
    info.nr_hw_queues = 1;
    info.queue_depth = static_cast<uint16_t>(queue_depth_);
    info.max_io_buf_bytes = kMaxIoBytes;
    info.dev_id = static_cast<uint32_t>(-1);
    info.ublksrv_pid = static_cast<int32_t>(getpid());
    info.flags = kFlagCmdIoctlEncode;
    int r = ctrl_cmd(&ctrl_ring_, ctrl_fd_, kCmdAddDev, info.dev_id, &info,
                     sizeof(info));
    if (r < 0) {
      throw std::runtime_error(std::string("ublk ADD_DEV: ") +
                               strerror(-r));
    }
    dev_id_ = static_cast<int>(info.dev_id);
    added_ = true;

    UblkParams params{};
    params.len = sizeof(params);
    params.types = kParamTypeBasic | kParamTypeDiscard;
    const uint64_t bs = bdev_->block_size();
    uint8_t bs_shift = 9;
    while ((1u << bs_shift) < bs && bs_shift < 12) ++bs_shift;
    params.basic.logical_bs_shift = bs_shift;
    params.basic.physical_bs_shift = bs_shift;
    params.basic.io_opt_shift = 12;
    params.basic.io_min_shift = bs_shift;
    params.basic.max_sectors = kMaxIoBytes / 512;
    params.basic.dev_sectors = bdev_->size_bytes() / 512;
    params.discard.discard_granularity = static_cast<uint32_t>(bs);
    params.discard.max_discard_sectors = 1u << 16;
    params.discard.max_write_zeroes_sectors = 1u << 16;
    params.discard.max_discard_segments = 1;
    r = ctrl_cmd(&ctrl_ring_, ctrl_fd_, kCmdSetParams, info.dev_id, &params,
                 sizeof(params));
    if (r < 0) {
      throw std::runtime_error(std::string("ublk SET_PARAMS: ") +
                               strerror(-r));
    }

    // Char node (devt params are valid as soon as the device exists).
    UblkParams got{};
    got.len = sizeof(got);
    r = ctrl_cmd(&ctrl_ring_, ctrl_fd_, kCmdGetParams, info.dev_id, &got,
                 sizeof(got));
    if (r < 0 || !(got.types & kParamTypeDevt)) {
      throw std::runtime_error("ublk GET_PARAMS(devt) failed");
    }
    const std::string char_path = "/dev/ublkc" + std::to_string(dev_id_);
    ensure_node(char_path, S_IFCHR, got.devt.char_major,
                got.devt.char_minor);
    char_fd_ = open(char_path.c_str(), O_RDWR);
    if (char_fd_ < 0) {
      throw std::runtime_error("ublk: open " + char_path + ": " +
                               strerror(errno));
    }
    // Descriptor area for queue 0.
    const size_t cmd_buf_sz =
        ((kMaxQueueDepthUapi * sizeof(UblkIoDesc)) + 4095) & ~size_t{4095};
    descs_ = static_cast<UblkIoDesc*>(
        mmap(nullptr, cmd_buf_sz, PROT_READ, MAP_SHARED | MAP_POPULATE,
             char_fd_, 0));
    if (descs_ == MAP_FAILED) {
      throw std::runtime_error("ublk: desc mmap failed");
    }
    descs_len_ = cmd_buf_sz;

    queue_thread_ = std::thread([this] { queue_loop(); });
    // Wait for all FETCH submissions before START_DEV.
    while (!queue_ready_.load(std::memory_order_acquire)) {
      if (queue_failed_.load()) {
        throw std::runtime_error("ublk: queue thread failed at startup");
      }
      std::this_thread::sleep_for(std::chrono::milliseconds(1));
    }
    r = ctrl_cmd(&ctrl_ring_, ctrl_fd_, kCmdStartDev, info.dev_id, nullptr,
                 0);
    if (r < 0) {
      throw std::runtime_error(std::string("ublk START_DEV: ") +
                               strerror(-r));
    }
    started_ = true;
    // Block node (disk devt valid once started).
    memset(&got, 0, sizeof(got));
    got.len = sizeof(got);
    r = ctrl_cmd(&ctrl_ring_, ctrl_fd_, kCmdGetParams, info.dev_id, &got,
                 sizeof(got));
    const std::string block_path = "/dev/ublkb" + std::to_string(dev_id_);
    if (r >= 0 && (got.types & kParamTypeDevt) && got.devt.disk_major) {
      ensure_node(block_path, S_IFBLK, got.devt.disk_major,
                  got.devt.disk_minor);
    }
    return UblkDisk{dev_id_, bdev_->name(), block_path};
  }

  void stop() {
    if (stopped_) return;
    stopped_ = true;
    if (started_) {
      (void)ctrl_cmd(&ctrl_ring_, ctrl_fd_, kCmdStopDev,
                     static_cast<uint32_t>(dev_id_), nullptr, 0);
    }
    stopping_.store(true, std::memory_order_release);
    if (queue_thread_.joinable()) queue_thread_.join();
    if (added_) {
      (void)ctrl_cmd(&ctrl_ring_, ctrl_fd_, kCmdDelDev,
                     static_cast<uint32_t>(dev_id_), nullptr, 0);
    }
    if (descs_ != nullptr && descs_ != MAP_FAILED) {
      munmap(const_cast<UblkIoDesc*>(descs_), descs_len_);
    }
    if (char_fd_ >= 0) close(char_fd_);
    if (ctrl_fd_ >= 0) close(ctrl_fd_);
  }

  int dev_id() const { return dev_id_; }
  const std::string& bdev_name() const { return bdev_->name(); }

 private:
  static constexpr uint32_t kMaxIoBytes = 256 << 10;

  void queue_loop() {
    try {
      UringSqe128 ring;
      ring.init(static_cast<unsigned>(queue_depth_) * 2);
      auto channel = bdev_->get_channel();
      std::vector<uint8_t*> bufs(queue_depth_);
      for (int t = 0; t < queue_depth_; ++t) {
        bufs[t] = static_cast<uint8_t*>(alloc_pinned(kMaxIoBytes));
      }
      // Prime: FETCH for every tag.
      for (int t = 0; t < queue_depth_; ++t) {
        UblkIoCmd cmd{};
        cmd.q_id = 0;
        cmd.tag = static_cast<uint16_t>(t);
        cmd.addr = reinterpret_cast<uint64_t>(bufs[t]);
        ring.push_cmd(char_fd_, kIoFetchReq, &cmd, sizeof(cmd),
                      static_cast<uint64_t>(t));
      }
      if (ring.submit(0) < 0) {
        throw std::runtime_error("ublk: FETCH submit failed");
      }
      queue_ready_.store(true, std::memory_order_release);

      int inflight_bdev = 0;
      std::vector<std::pair<int, int>> done;  // (tag, result-bytes-or-neg)
      std::mutex done_mutex;  // completions fire on this thread (poll)
      bool abort_seen = false;
      std::chrono::steady_clock::time_point stop_observed_{};
      while (true) {
        // 1) kernel completions -> bdev submissions
        IoUringCqe cqe{};
        bool any = false;
        while (ring.pop(&cqe)) {
          any = true;
          const int tag = static_cast<int>(cqe.user_data);
          if (cqe.res < 0) {  // UBLK_IO_RES_ABORT et al: tag retired
            abort_seen = true;
            continue;
          }
          // cqe.res == UBLK_IO_RES_OK: a request is ready in descs_[tag].
          const UblkIoDesc d = descs_[tag];
          const uint32_t op = d.op_flags & 0xFF;
          IoRequest req;
          req.offset = d.start_sector * 512;
          req.length = static_cast<uint64_t>(d.nr_sectors) * 512;
          req.buffer = bufs[tag];
          bool ok = true;
          switch (op) {
            case kUblkOpRead: req.op = IoOp::kRead; break;
            case kUblkOpWrite: req.op = IoOp::kWrite; break;
            case kUblkOpFlush: req.op = IoOp::kFlush; req.length = 0; break;
            case kUblkOpDiscard:
            case kUblkOpWriteZeroes:
              req.op = IoOp::kFill;
              req.fill = 0;
              req.buffer = nullptr;
              break;
            default: ok = false; break;
          }
          if (!ok || req.length > kMaxIoBytes) {
            std::lock_guard<std::mutex> lock(done_mutex);
            done.emplace_back(tag, -EINVAL);
            continue;
          }
          const uint32_t bytes = static_cast<uint32_t>(req.length);
          req.on_complete = [tag, bytes, &done, &done_mutex,
                             &inflight_bdev](int status) {
            std::lock_guard<std::mutex> lock(done_mutex);
            done.emplace_back(tag,
                              status == kIoOk ? static_cast<int>(bytes)
                                              : -EIO);
            --inflight_bdev;
          };
          ++inflight_bdev;
          bdev_->submit(channel.get(), std::move(req));
        }
        // 2) drive the engine
        if (bdev_->poll(channel.get()) > 0) any = true;
        // 3) bdev completions -> COMMIT_AND_FETCH
        {
          std::lock_guard<std::mutex> lock(done_mutex);
          for (auto& [tag, result] : done) {
            UblkIoCmd cmd{};
            cmd.q_id = 0;
            cmd.tag = static_cast<uint16_t>(tag);
            cmd.result = result;
            cmd.addr = reinterpret_cast<uint64_t>(bufs[tag]);
            ring.push_cmd(char_fd_, kIoCommitAndFetchReq, &cmd, sizeof(cmd),
                          static_cast<uint64_t>(tag));
            any = true;
          }
          done.clear();
        }
        if (ring.submit(0) < 0 && errno != EBUSY) break;
        if (stopping_.load(std::memory_order_acquire)) {
          // Normal teardown: STOP_DEV retires every tag with an abort
          // CQE. Bound the wait anyway — a kernel that never delivers
          // them (or a wedged bdev) must not hang stop()'s join.
          using clock = std::chrono::steady_clock;
          if (stop_observed_ == clock::time_point{}) {
            stop_observed_ = clock::now();
          }
          if ((abort_seen && inflight_bdev == 0) ||
              clock::now() - stop_observed_ > std::chrono::seconds(5)) {
            break;
          }
        }
        if (!any) {
          std::this_thread::sleep_for(std::chrono::microseconds(50));
        }
      }
      for (uint8_t* buf : bufs) free_pinned(buf);
    } catch (const std::exception& e) {
      fprintf(stderr, "[ublk] queue thread: %s\n", e.what());
      queue_failed_.store(true);
      queue_ready_.store(true);  // unblock start()
    }
  }

  BdevPtr bdev_;
  int queue_depth_;
  int ctrl_fd_ = -1;
  int char_fd_ = -1;
  int dev_id_ = -1;
  bool added_ = false;
  bool started_ = false;
  bool stopped_ = false;
  UringSqe128 ctrl_ring_;
  const UblkIoDesc* descs_ = nullptr;
  size_t descs_len_ = 0;
  std::thread queue_thread_;
  std::atomic<bool> queue_ready_{false};
  std::atomic<bool> queue_failed_{false};
  std::atomic<bool> stopping_{false};
};

struct UblkState {
  std::mutex mutex;
  std::map<int, std::unique_ptr<UblkServer>> servers;
  std::map<int, UblkDisk> disks;
};

UblkState& ublk_state() {
  static UblkState state;
  return state;
}

}  // namespace

bool ublk_available() {
  int fd = open_control();
  if (fd < 0) return false;
  close(fd);
  return true;
}

UblkDisk ublk_start(const std::string& bdev_name, int queue_depth) {
  BdevPtr bdev = BdevManager::instance().find(bdev_name);
  if (!bdev) throw std::runtime_error("ublk: no bdev " + bdev_name);
  if (!bdev->claim()) {
    throw std::runtime_error("ublk: bdev " + bdev_name + " is claimed");
  }
  auto server = std::make_unique<UblkServer>(bdev, queue_depth);
  UblkDisk disk;
  try {
    disk = server->start();
  } catch (...) {
    bdev->release();
    throw;
  }
  auto& state = ublk_state();
  std::lock_guard<std::mutex> lock(state.mutex);
  state.disks[disk.dev_id] = disk;
  state.servers[disk.dev_id] = std::move(server);
  return disk;
}

void ublk_stop(int dev_id) {
  std::unique_ptr<UblkServer> server;
  {
    auto& state = ublk_state();
    std::lock_guard<std::mutex> lock(state.mutex);
    auto it = state.servers.find(dev_id);
    if (it == state.servers.end()) {
      throw std::runtime_error("ublk: no device " + std::to_string(dev_id));
    }
    server = std::move(it->second);
    state.servers.erase(it);
    state.disks.erase(dev_id);
  }
  const std::string name = server->bdev_name();
  server->stop();
  server.reset();
  BdevPtr bdev = BdevManager::instance().find(name);
  if (bdev) bdev->release();
}

std::vector<UblkDisk> ublk_list() {
  auto& state = ublk_state();
  std::lock_guard<std::mutex> lock(state.mutex);
  std::vector<UblkDisk> out;
  for (const auto& [id, disk] : state.disks) out.push_back(disk);
  return out;
}

void ublk_stop_all() {
  std::vector<int> ids;
  {
    auto& state = ublk_state();
    std::lock_guard<std::mutex> lock(state.mutex);
    for (const auto& [id, server] : state.servers) ids.push_back(id);
  }
  for (int id : ids) {
    try {
      ublk_stop(id);
    } catch (const std::exception&) {
    }
  }
}

}  // namespace hipstore
