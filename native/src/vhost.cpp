// vhost-user-scsi target (see include/hipstore/vhost.h).
//
// Protocol per the public vhost-user specification; command set per
// SPC-4/SBC-3 to the depth a Linux guest's sd driver probes. The
// reference gets all of this from DPDK's rte_vhost + SPDK's scsi lib;
// here it is one self-contained file because only the disk (type 0,
// single LUN) personality is needed.

#include "hipstore/vhost.h"

#include <poll.h>
#include <sys/eventfd.h>
#include <sys/mman.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <pthread.h>
#include <sched.h>
#include <unistd.h>

#include <hip/hip_runtime.h>

#include <cstring>
#include <map>
#include <mutex>
#include <set>
#include <stdexcept>
#include <thread>
#include <vector>

#include "hipstore/engine.h"

namespace hipstore {
namespace {

// ---- vhost-user wire protocol ---------------------------------------

enum : uint32_t {
  kGetFeatures = 1,
  kSetFeatures = 2,
  kSetOwner = 3,
  kResetOwner = 4,
  kSetMemTable = 5,
  kSetLogBase = 6,
  kSetLogFd = 7,
  kSetVringNum = 8,
  kSetVringAddr = 9,
  kSetVringBase = 10,
  kGetVringBase = 11,
  kSetVringKick = 12,
  kSetVringCall = 13,
  kSetVringErr = 14,
  kGetProtocolFeatures = 15,
  kSetProtocolFeatures = 16,
  kGetQueueNum = 17,
  kSetVringEnable = 18,
  kGetConfig = 24,
  kSetConfig = 25,
};

constexpr uint32_t kVersionMask = 0x3;
constexpr uint32_t kVersion1 = 0x1;
constexpr uint32_t kFlagReply = 0x4;
constexpr uint64_t kNoFdMask = 0x100;  // VHOST_USER_VRING_NOFD_MASK

// Device features: virtio 1.0 + indirect descriptors + the
// protocol-features escape hatch.
constexpr uint64_t kFeatVersion1 = 1ull << 32;
constexpr uint64_t kFeatIndirect = 1ull << 28;
constexpr uint64_t kFeatProtocol = 1ull << 30;
constexpr uint64_t kDeviceFeatures = kFeatVersion1 | kFeatIndirect |
                                     kFeatProtocol;
constexpr uint64_t kFeatBlkDiscard = 1ull << 13;
constexpr uint64_t kFeatBlkWriteZeroes = 1ull << 14;
constexpr uint64_t kProtocolFeatureMq = 1ull << 0;
constexpr uint64_t kProtocolFeatureConfig = 1ull << 9;  // GET/SET_CONFIG

constexpr int kMaxVrings = 8;      // controlq + eventq + 6 request queues
constexpr int kMaxRegions = 8;
// Largest legal payloads: SET_MEM_TABLE (8 + 32*regions) and
// GET/SET_CONFIG (12-byte header + up to 256 config bytes).
constexpr size_t kMaxPayload = 12 + 256;

struct VringStateWire {
  uint32_t index;
  uint32_t num;
};

struct VringAddrWire {
  uint32_t index;
  uint32_t flags;
  uint64_t desc_user_addr;
  uint64_t used_user_addr;
  uint64_t avail_user_addr;
  uint64_t log_guest_addr;
};

struct MemRegionWire {
  uint64_t guest_phys_addr;
  uint64_t memory_size;
  uint64_t userspace_addr;
  uint64_t mmap_offset;
};

// ---- virtio split ring ----------------------------------------------

constexpr uint16_t kDescNext = 1;
constexpr uint16_t kDescWrite = 2;
constexpr uint16_t kDescIndirect = 4;

struct VringDesc {
  uint64_t addr;  // guest physical
  uint32_t len;
  uint16_t flags;
  uint16_t next;
};

struct VringAvail {
  uint16_t flags;
  uint16_t idx;
  uint16_t ring[];
};

struct VringUsedElem {
  uint32_t id;
  uint32_t len;
};

struct VringUsed {
  uint16_t flags;
  uint16_t idx;
  VringUsedElem ring[];
};

// ---- virtio-scsi ------------------------------------------------------

constexpr size_t kCdbSize = 32;
constexpr size_t kSenseSize = 96;

struct __attribute__((packed)) ScsiCmdReq {
  uint8_t lun[8];
  uint64_t tag;
  uint8_t task_attr;
  uint8_t prio;
  uint8_t crn;
  uint8_t cdb[kCdbSize];
};
static_assert(sizeof(ScsiCmdReq) == 51, "virtio_scsi_cmd_req layout");

struct __attribute__((packed)) ScsiCmdResp {
  uint32_t sense_len;
  uint32_t resid;
  uint16_t status_qualifier;
  uint8_t status;
  uint8_t response;
  uint8_t sense[kSenseSize];
};
static_assert(sizeof(ScsiCmdResp) == 108, "virtio_scsi_cmd_resp layout");

constexpr uint8_t kRespOk = 0;        // VIRTIO_SCSI_S_OK
constexpr uint8_t kRespBadTarget = 3; // VIRTIO_SCSI_S_BAD_TARGET
constexpr uint8_t kRespFailure = 9;   // VIRTIO_SCSI_S_FAILURE
constexpr uint8_t kStatusGood = 0x00;
constexpr uint8_t kStatusCheckCondition = 0x02;

constexpr uint64_t kMaxIoBytes = 4ull << 20;  // per-command cap

void be32(uint8_t* p, uint32_t v) {
  p[0] = v >> 24; p[1] = v >> 16; p[2] = v >> 8; p[3] = v;
}
void be64(uint8_t* p, uint64_t v) {
  be32(p, v >> 32);
  be32(p + 4, static_cast<uint32_t>(v));
}
uint16_t rbe16(const uint8_t* p) { return uint16_t(p[0]) << 8 | p[1]; }
uint32_t rbe32(const uint8_t* p) {
  return uint32_t(p[0]) << 24 | uint32_t(p[1]) << 16 | uint32_t(p[2]) << 8 |
         p[3];
}
uint64_t rbe64(const uint8_t* p) {
  return uint64_t(rbe32(p)) << 32 | rbe32(p + 4);
}

struct Iov {
  uint8_t* base;
  size_t len;
};

size_t iov_total(const std::vector<Iov>& iovs) {
  size_t n = 0;
  for (const Iov& v : iovs) n += v.len;
  return n;
}

// Copy len bytes from a flat buffer into guest iovs; returns copied.
size_t scatter(const std::vector<Iov>& iovs, const uint8_t* src, size_t len) {
  size_t done = 0;
  for (const Iov& v : iovs) {
    if (done >= len) break;
    size_t n = std::min(v.len, len - done);
    memcpy(v.base, src + done, n);
    done += n;
  }
  return done;
}

size_t gather(uint8_t* dst, const std::vector<Iov>& iovs, size_t len) {
  size_t done = 0;
  for (const Iov& v : iovs) {
    if (done >= len) break;
    size_t n = std::min(v.len, len - done);
    memcpy(dst + done, v.base, n);
    done += n;
  }
  return done;
}

}  // namespace

// ---- device ----------------------------------------------------------

class VhostUserScsiDev {
 public:
  enum class Personality { kScsi, kBlk };

  VhostUserScsiDev(std::string name, std::string socket_path,
                   std::function<BdevPtr(int)> resolver,
                   Personality personality = Personality::kScsi,
                   bool readonly = false)
      : name_(std::move(name)),
        socket_path_(std::move(socket_path)),
        resolver_(std::move(resolver)),
        personality_(personality),
        readonly_(readonly) {}

  ~VhostUserScsiDev() { stop(); }

  void start() {
    listen_fd_ = ::socket(AF_UNIX, SOCK_STREAM | SOCK_CLOEXEC, 0);
    if (listen_fd_ < 0) throw std::runtime_error("vhost: socket() failed");
    sockaddr_un addr{};
    addr.sun_family = AF_UNIX;
    if (socket_path_.size() >= sizeof(addr.sun_path)) {
      ::close(listen_fd_);
      listen_fd_ = -1;
      throw std::runtime_error("vhost: socket path too long: " + socket_path_);
    }
    strncpy(addr.sun_path, socket_path_.c_str(), sizeof(addr.sun_path) - 1);
    ::unlink(socket_path_.c_str());
    if (::bind(listen_fd_, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)) <
            0 ||
        ::listen(listen_fd_, 1) < 0) {
      ::close(listen_fd_);
      listen_fd_ = -1;
      throw std::runtime_error("vhost: cannot bind " + socket_path_);
    }
    if (pipe(stop_pipe_) < 0) throw std::runtime_error("vhost: pipe failed");
    accept_thread_ = std::thread([this] { accept_loop(); });
  }

  void stop() {
    bool expected = false;
    if (!stopping_.compare_exchange_strong(expected, true)) return;
    if (stop_pipe_[1] >= 0) (void)!write(stop_pipe_[1], "x", 1);
    if (accept_thread_.joinable()) accept_thread_.join();
    teardown_session();
    if (listen_fd_ >= 0) ::close(listen_fd_);
    listen_fd_ = -1;
    for (int fd : stop_pipe_)
      if (fd >= 0) ::close(fd);
    stop_pipe_[0] = stop_pipe_[1] = -1;
    ::unlink(socket_path_.c_str());
  }

  const std::string& socket_path() const { return socket_path_; }

 private:
  // One channel per (ring worker, bdev): channels are single-threaded
  // by contract, and per-worker ownership lets the request queues run
  // their I/O in parallel instead of convoying on one mutex. The map
  // lives in the worker thread and dies with it (teardown joins the
  // workers, so no channel outlives the session).
  using ChannelCache =
      std::map<Bdev*, std::pair<BdevPtr, std::shared_ptr<IoChannel>>>;

  // -- connection handling ---------------------------------------------

  void accept_loop() {
    while (true) {
      pollfd fds[2] = {{listen_fd_, POLLIN, 0}, {stop_pipe_[0], POLLIN, 0}};
      if (::poll(fds, 2, -1) < 0) {
        if (errno == EINTR) continue;
        return;
      }
      if (fds[1].revents) return;
      int conn = ::accept(listen_fd_, nullptr, nullptr);
      if (conn < 0) continue;
      serve(conn);
      ::close(conn);
      teardown_session();  // master went away: rings + memory are stale
    }
  }

  struct Msg {
    uint32_t request = 0;
    uint32_t flags = 0;
    std::vector<uint8_t> payload;
    std::vector<int> fds;
  };

  bool recv_msg(int conn, Msg* msg) {
    uint8_t hdr[12];
    alignas(cmsghdr) char cbuf[CMSG_SPACE(sizeof(int) * kMaxRegions)];
    iovec iov{hdr, sizeof(hdr)};
    msghdr mh{};
    mh.msg_iov = &iov;
    mh.msg_iovlen = 1;
    mh.msg_control = cbuf;
    mh.msg_controllen = sizeof(cbuf);
    ssize_t n = ::recvmsg(conn, &mh, MSG_CMSG_CLOEXEC);
    if (n <= 0) return false;
    for (cmsghdr* c = CMSG_FIRSTHDR(&mh); c; c = CMSG_NXTHDR(&mh, c)) {
      if (c->cmsg_level == SOL_SOCKET && c->cmsg_type == SCM_RIGHTS) {
        int nfds = (c->cmsg_len - CMSG_LEN(0)) / sizeof(int);
        const int* fds = reinterpret_cast<const int*>(CMSG_DATA(c));
        msg->fds.assign(fds, fds + nfds);
      }
    }
    // Tolerate a header split across reads (SOCK_STREAM).
    size_t have = n;
    while (have < sizeof(hdr)) {
      ssize_t m = ::recv(conn, hdr + have, sizeof(hdr) - have, 0);
      if (m <= 0) return false;
      have += m;
    }
    memcpy(&msg->request, hdr, 4);
    memcpy(&msg->flags, hdr + 4, 4);
    uint32_t size;
    memcpy(&size, hdr + 8, 4);
    if (size > kMaxPayload) return false;
    msg->payload.resize(size);
    size_t got = 0;
    while (got < size) {
      ssize_t m = ::recv(conn, msg->payload.data() + got, size - got, 0);
      if (m <= 0) return false;
      got += m;
    }
    return (msg->flags & kVersionMask) == kVersion1;
  }

  // virtio_blk_config: capacity in 512-byte sectors at offset 0,
  // blk_size (u32) at offset 20; everything else zero (no geometry /
  // topology hints).
  void fill_virtio_config(uint8_t* config, size_t size) {
    if (personality_ != Personality::kBlk || size < 24) return;
    BdevPtr bdev = resolver_(0);
    if (!bdev) return;
    const uint64_t sectors = bdev->size_bytes() / 512;
    memcpy(config, &sectors, 8);
    const uint32_t blk_size = static_cast<uint32_t>(bdev->block_size());
    memcpy(config + 20, &blk_size, 4);
    const uint16_t num_queues = 8;  // MQ-aware guests read this
    memcpy(config + 34, &num_queues, 2);
    if (size >= 56) {
      // discard/write-zeroes geometry (virtio_blk_config offsets)
      const uint32_t max_sectors = kMaxIoBytes / 512;
      const uint32_t one_seg = 256;
      const uint32_t align = blk_size / 512;
      memcpy(config + 36, &max_sectors, 4);   // max_discard_sectors
      memcpy(config + 40, &one_seg, 4);       // max_discard_seg
      memcpy(config + 44, &align, 4);         // discard_sector_alignment
      memcpy(config + 48, &max_sectors, 4);   // max_write_zeroes_sectors
      memcpy(config + 52, &one_seg, 4);       // max_write_zeroes_seg
    }
  }

  void send_reply_big(int conn, uint32_t request, const void* payload,
                      uint32_t size) {
    uint8_t buf[12 + 12 + 256];
    uint32_t flags = kVersion1 | kFlagReply;
    memcpy(buf, &request, 4);
    memcpy(buf + 4, &flags, 4);
    memcpy(buf + 8, &size, 4);
    memcpy(buf + 12, payload, size);
    (void)!::send(conn, buf, 12 + size, MSG_NOSIGNAL);
  }

  void send_reply(int conn, uint32_t request, const void* payload,
                  uint32_t size) {
    uint8_t buf[12 + 64];
    uint32_t flags = kVersion1 | kFlagReply;
    memcpy(buf, &request, 4);
    memcpy(buf + 4, &flags, 4);
    memcpy(buf + 8, &size, 4);
    memcpy(buf + 12, payload, size);
    (void)!::send(conn, buf, 12 + size, MSG_NOSIGNAL);
  }

  static void close_unconsumed(Msg* msg) {
    for (int& fd : msg->fds) {
      if (fd >= 0) ::close(fd);
      fd = -1;
    }
  }

  void serve(int conn) {
    Msg msg;
    while (!stopping_.load(std::memory_order_relaxed)) {
      pollfd fds[2] = {{conn, POLLIN, 0}, {stop_pipe_[0], POLLIN, 0}};
      if (::poll(fds, 2, -1) < 0 && errno != EINTR) return;
      if (fds[1].revents) return;
      if (!fds[0].revents) continue;
      msg = Msg{};
      bool ok = recv_msg(conn, &msg);
      if (ok) ok = handle(conn, msg);
      // Close fds the handler did not consume — ALSO on failure paths
      // (a malformed message with fds attached must not leak them).
      close_unconsumed(&msg);
      if (!ok) return;
    }
  }

  template <typename T>
  static bool read_payload(const Msg& msg, T* out) {
    if (msg.payload.size() < sizeof(T)) return false;
    memcpy(out, msg.payload.data(), sizeof(T));
    return true;
  }

  bool handle(int conn, Msg& msg) {
    switch (msg.request) {
      case kGetFeatures: {
        uint64_t f = kDeviceFeatures;
        if (personality_ == Personality::kBlk) {
          f |= kFeatBlkDiscard | kFeatBlkWriteZeroes;
        }
        send_reply(conn, msg.request, &f, 8);
        return true;
      }
      case kSetFeatures:
        read_payload(msg, &negotiated_features_);
        return true;
      case kGetProtocolFeatures: {
        // blk devices carry their geometry in virtio config space
        // (QEMU reads capacity via GET_CONFIG); scsi needs only MQ.
        uint64_t f = kProtocolFeatureMq;
        if (personality_ == Personality::kBlk) f |= kProtocolFeatureConfig;
        send_reply(conn, msg.request, &f, 8);
        return true;
      }
      case kSetProtocolFeatures:
        read_payload(msg, &protocol_features_);
        return true;
      case kGetQueueNum: {
        uint64_t n = kMaxVrings;
        send_reply(conn, msg.request, &n, 8);
        return true;
      }
      case kSetOwner:
      case kResetOwner:
      case kSetLogBase:
      case kSetLogFd:
      case kSetVringErr:
        return true;  // accepted, nothing to do
      case kSetMemTable:
        return set_mem_table(msg);
      case kSetVringNum: {
        VringStateWire s;
        if (!read_payload(msg, &s) || s.index >= kMaxVrings) return false;
        rings_[s.index].num = s.num;
        return true;
      }
      case kSetVringBase: {
        VringStateWire s;
        if (!read_payload(msg, &s) || s.index >= kMaxVrings) return false;
        rings_[s.index].last_avail = static_cast<uint16_t>(s.num);
        return true;
      }
      case kSetVringAddr:
        return set_vring_addr(msg);
      case kSetVringKick:
        return set_vring_fd(msg, /*is_kick=*/true);
      case kSetVringCall:
        return set_vring_fd(msg, /*is_kick=*/false);
      case kSetVringEnable: {
        VringStateWire s;
        if (!read_payload(msg, &s) || s.index >= kMaxVrings) return false;
        rings_[s.index].enabled.store(s.num != 0, std::memory_order_release);
        return true;
      }
      case kGetConfig: {
        // VhostUserConfig {u32 offset; u32 size; u32 flags; u8 region[]}
        uint32_t offset = 0, size = 0, flags = 0;
        if (msg.payload.size() < 12) return false;
        memcpy(&offset, msg.payload.data(), 4);
        memcpy(&size, msg.payload.data() + 4, 4);
        memcpy(&flags, msg.payload.data() + 8, 4);
        if (size > 256) return false;
        uint8_t config[256] = {0};
        fill_virtio_config(config, sizeof(config));
        uint8_t reply[12 + 256];
        memcpy(reply, &offset, 4);
        memcpy(reply + 4, &size, 4);
        memcpy(reply + 8, &flags, 4);
        for (uint32_t i = 0; i < size; ++i) {
          reply[12 + i] =
              (offset + i < sizeof(config)) ? config[offset + i] : 0;
        }
        send_reply_big(conn, msg.request, reply, 12 + size);
        return true;
      }
      case kSetConfig:
        return true;  // no writable config fields
      case kGetVringBase: {
        VringStateWire s;
        if (!read_payload(msg, &s) || s.index >= kMaxVrings) return false;
        stop_ring(s.index);
        VringStateWire reply{s.index, rings_[s.index].last_avail};
        send_reply(conn, msg.request, &reply, sizeof(reply));
        return true;
      }
      default:
        // Unknown request: per spec a slave may ignore what it does
        // not implement (no reply expected without NEED_REPLY).
        return true;
    }
  }

  // -- guest memory ------------------------------------------------------

  struct Region {
    uint64_t gpa = 0;
    uint64_t size = 0;
    uint64_t uaddr = 0;
    uint8_t* map = nullptr;   // mmap base (includes mmap_offset slack)
    size_t maplen = 0;
    uint8_t* base = nullptr;  // map + mmap_offset
    bool hip_registered = false;
  };

  bool set_mem_table(Msg& msg) {
    unmap_regions();
    uint32_t nregions = 0;
    if (msg.payload.size() < 8) return false;
    memcpy(&nregions, msg.payload.data(), 4);
    if (nregions > kMaxRegions || msg.fds.size() < nregions ||
        msg.payload.size() < 8 + nregions * sizeof(MemRegionWire)) {
      return false;
    }
    for (uint32_t i = 0; i < nregions; ++i) {
      MemRegionWire w;
      memcpy(&w, msg.payload.data() + 8 + i * sizeof(w), sizeof(w));
      size_t maplen = w.memory_size + w.mmap_offset;
      void* map = ::mmap(nullptr, maplen, PROT_READ | PROT_WRITE, MAP_SHARED,
                         msg.fds[i], 0);
      ::close(msg.fds[i]);
      msg.fds[i] = -1;
      if (map == MAP_FAILED) {
        unmap_regions();
        return false;
      }
      Region region;
      region.gpa = w.guest_phys_addr;
      region.size = w.memory_size;
      region.uaddr = w.userspace_addr;
      region.map = static_cast<uint8_t*>(map);
      region.maplen = maplen;
      region.base = region.map + w.mmap_offset;
      // Register guest memory with the GPU so the HBM engine's copy
      // kernels DMA guest<->HBM directly (SPDK requires DMA-able guest
      // memory for vhost the same way). Without this, the engine's
      // hipHostGetDevicePointer on a guest address fails.
      if (gpu_available()) {
        region.hip_registered =
            hipHostRegister(region.base, region.size,
                            hipHostRegisterMapped) == hipSuccess;
      }
      regions_.push_back(region);
    }
    return true;
  }

  void unmap_regions() {
    for (const Region& r : regions_) {
      if (r.hip_registered) (void)hipHostUnregister(r.base);
      ::munmap(r.map, r.maplen);
    }
    regions_.clear();
  }

  // Guest-controlled addr/len: `gpa + len` can wrap UINT64_MAX, so the
  // range check is phrased subtraction-only (no sums that can overflow).
  uint8_t* gpa_to_ptr(uint64_t gpa, uint64_t len) {
    for (const Region& r : regions_) {
      if (gpa >= r.gpa && len <= r.size && gpa - r.gpa <= r.size - len) {
        return r.base + (gpa - r.gpa);
      }
    }
    return nullptr;
  }

  uint8_t* uaddr_to_ptr(uint64_t uaddr, uint64_t len) {
    for (const Region& r : regions_) {
      if (uaddr >= r.uaddr && len <= r.size &&
          uaddr - r.uaddr <= r.size - len) {
        return r.base + (uaddr - r.uaddr);
      }
    }
    return nullptr;
  }

  // -- vrings ------------------------------------------------------------

  struct Vring {
    uint32_t num = 0;
    uint16_t last_avail = 0;
    uint16_t used_idx = 0;
    VringDesc* desc = nullptr;
    VringAvail* avail = nullptr;
    VringUsed* used = nullptr;
    int kick = -1;
    int call = -1;
    std::atomic<bool> enabled{false};
    std::atomic<bool> running{false};
    std::thread worker;
    int stop_pipe[2] = {-1, -1};
  };

  bool set_vring_addr(const Msg& msg) {
    VringAddrWire a;
    if (!read_payload(msg, &a) || a.index >= kMaxVrings) return false;
    Vring& ring = rings_[a.index];
    uint32_t num = ring.num ? ring.num : 1;
    ring.desc = reinterpret_cast<VringDesc*>(
        uaddr_to_ptr(a.desc_user_addr, sizeof(VringDesc) * num));
    ring.avail = reinterpret_cast<VringAvail*>(
        uaddr_to_ptr(a.avail_user_addr, 4 + 2 * num));
    ring.used = reinterpret_cast<VringUsed*>(
        uaddr_to_ptr(a.used_user_addr, 4 + 8 * num));
    return ring.desc && ring.avail && ring.used;
  }

  bool set_vring_fd(Msg& msg, bool is_kick) {
    uint64_t v = 0;
    if (!read_payload(msg, &v)) return false;
    unsigned index = v & 0xff;
    if (index >= kMaxVrings) return false;
    int fd = -1;
    if (!(v & kNoFdMask)) {
      if (msg.fds.empty()) return false;
      fd = msg.fds[0];
      msg.fds[0] = -1;  // consumed
    }
    Vring& ring = rings_[index];
    if (is_kick) {
      stop_ring(index);
      if (ring.kick >= 0) ::close(ring.kick);
      ring.kick = fd;
      // Without protocol features the ring starts enabled as soon as
      // it is kickable (vhost-user spec).
      if (!(negotiated_features_ & kFeatProtocol)) {
        ring.enabled.store(true, std::memory_order_release);
      }
      if (fd >= 0 && ring.desc) start_ring(index);
    } else {
      if (ring.call >= 0) ::close(ring.call);
      ring.call = fd;
    }
    return true;
  }

  void start_ring(unsigned index) {
    Vring& ring = rings_[index];
    if (ring.running.exchange(true)) return;
    if (pipe(ring.stop_pipe) < 0) {
      ring.running.store(false);
      return;
    }
    ring.used_idx = ring.used->idx;
    ring.worker = std::thread([this, index] { ring_worker(index); });
  }

  void stop_ring(unsigned index) {
    Vring& ring = rings_[index];
    if (!ring.running.exchange(false)) return;
    (void)!write(ring.stop_pipe[1], "x", 1);
    if (ring.worker.joinable()) ring.worker.join();
    for (int& fd : ring.stop_pipe) {
      if (fd >= 0) ::close(fd);
      fd = -1;
    }
  }

  void teardown_session() {
    for (int i = 0; i < kMaxVrings; ++i) {
      stop_ring(i);
      Vring& ring = rings_[i];
      if (ring.kick >= 0) ::close(ring.kick);
      if (ring.call >= 0) ::close(ring.call);
      ring.kick = ring.call = -1;
      ring.num = 0;
      ring.last_avail = ring.used_idx = 0;
      ring.desc = nullptr;
      ring.avail = nullptr;
      ring.used = nullptr;
      ring.enabled.store(false);
    }
    unmap_regions();
    negotiated_features_ = 0;
    if (getenv("HIPSTORE_DEBUG") != nullptr) {
      fprintf(stderr, "[vhost %s] session served async=%llu sync=%llu\n",
              name_.c_str(),
              static_cast<unsigned long long>(async_count_.load()),
              static_cast<unsigned long long>(sync_count_.load()));
    }
  }

  static bool pipeline_enabled() {
    // Default ON since round 2's GPU A/B (tools/perf_batch.sh):
    // pipelined +17% scsi / +4% blk randread through the same master.
    // HIPSTORE_VHOST_PIPELINE=0 selects the one-at-a-time worker.
    static const bool on = [] {
      const char* env = getenv("HIPSTORE_VHOST_PIPELINE");
      return env == nullptr || atoi(env) != 0;
    }();
    return on;
  }

  // Optional SPDK-reactor-style pinning for the spinning ring
  // workers (HIPSTORE_VHOST_AFFINITY_BASE=<core>): unpinned hot
  // pollers migrate under CFS and crowd each other once several rings
  // spin concurrently (the vhost scale probe's global plateau).
  static void maybe_pin_ring_worker(unsigned index) {
    static std::atomic<int> next_slot{0};
    const char* env = getenv("HIPSTORE_VHOST_AFFINITY_BASE");
    if (env == nullptr) return;
    const long ncpu = sysconf(_SC_NPROCESSORS_ONLN);
    if (ncpu <= 0) return;
    (void)index;
    const int core =
        (atoi(env) + next_slot.fetch_add(1)) % static_cast<int>(ncpu);
    cpu_set_t set;
    CPU_ZERO(&set);
    CPU_SET(core, &set);
    (void)pthread_setaffinity_np(pthread_self(), sizeof(set), &set);
  }

  void ring_worker(unsigned index) {
    maybe_pin_ring_worker(index);
    ChannelCache channels;  // this worker's engine channels
    if (pipeline_enabled()) {
      ring_worker_pipelined(index, &channels);
      return;
    }
    Vring& ring = rings_[index];
    while (ring.running.load(std::memory_order_relaxed)) {
      pollfd fds[2] = {{ring.kick, POLLIN, 0}, {ring.stop_pipe[0], POLLIN, 0}};
      // Timeout so an enable flag flipped after the last kick is seen.
      if (::poll(fds, 2, 100) < 0 && errno != EINTR) return;
      if (fds[1].revents) return;
      if (fds[0].revents) {
        uint64_t n;
        (void)!read(ring.kick, &n, 8);
      }
      if (!ring.enabled.load(std::memory_order_acquire)) continue;
      drain_ring(index, &channels);
    }
  }

  // -- pipelined worker (HIPSTORE_VHOST_PIPELINE=1) ----------------------
  //
  // Keeps up to kMaxInflight commands outstanding per ring instead of
  // one sync I/O per command: simple single-extent READ/WRITE chains
  // (the entire hot path of a guest) are submitted asynchronously and
  // their used entries published out of order as the engine completes
  // them (virtio permits this — the id field identifies the chain).
  // Everything else falls back to the synchronous per-command path.
  static constexpr uint32_t kMaxInflight = 64;

  struct InflightIo {
    uint16_t head;
    std::vector<Iov> resp_iovs;   // SCSI response location (empty = blk)
    uint8_t* blk_status = nullptr;
    uint32_t data_in_bytes = 0;   // reported in the used entry
  };

  void ring_worker_pipelined(unsigned index, ChannelCache* channels) {
    Vring& ring = rings_[index];
    // shared (not stack) so a completion fired from the channel
    // destructor after an abandoned drain cannot touch a dead frame
    auto inflight_box = std::make_shared<uint32_t>(0);
    uint32_t& inflight = *inflight_box;
    std::chrono::steady_clock::time_point stop_seen{};
    while (ring.running.load(std::memory_order_relaxed) ||
           inflight > 0) {
      if (!ring.running.load(std::memory_order_relaxed)) {
        // Bounded drain: give outstanding I/O 10 s, then abandon (the
        // channel dtor will reap; a dead engine must not wedge join).
        const auto now = std::chrono::steady_clock::now();
        if (stop_seen == std::chrono::steady_clock::time_point{}) {
          stop_seen = now;
        } else if (now - stop_seen > std::chrono::seconds(10)) {
          return;
        }
      }
      // Sleep on the kick only when idle; otherwise poll channels hot.
      // While hot, advertise VRING_USED_F_NO_NOTIFY so a
      // flag-honoring master skips its kick syscalls; before actually
      // sleeping, clear the flag and RE-CHECK the avail ring (classic
      // missed-kick avoidance, virtio 2.6.10).
      int timeout_ms = inflight > 0 ? 0 : 100;
      if (ring.used != nullptr) {
        if (timeout_ms == 0) {
          __atomic_store_n(&ring.used->flags, uint16_t{1},
                           __ATOMIC_RELEASE);
        } else {
          __atomic_store_n(&ring.used->flags, uint16_t{0},
                           __ATOMIC_SEQ_CST);
          if (ring.enabled.load(std::memory_order_acquire) &&
              ring.avail != nullptr &&
              __atomic_load_n(&ring.avail->idx, __ATOMIC_ACQUIRE) !=
                  ring.last_avail) {
            timeout_ms = 0;  // work raced in: stay hot
          }
        }
      }
      pollfd fds[2] = {{ring.kick, POLLIN, 0}, {ring.stop_pipe[0], POLLIN, 0}};
      if (::poll(fds, 2, timeout_ms) < 0 && errno != EINTR) return;
      if (fds[1].revents && inflight == 0) return;
      if (fds[0].revents) {
        uint64_t n;
        (void)!read(ring.kick, &n, 8);
      }
      if (ring.enabled.load(std::memory_order_acquire)) {
        bool published = false;
        while (inflight < kMaxInflight) {
          uint16_t avail_idx =
              __atomic_load_n(&ring.avail->idx, __ATOMIC_ACQUIRE);
          if (ring.last_avail == avail_idx) break;
          uint16_t head = ring.avail->ring[ring.last_avail % ring.num];
          ring.last_avail++;
          if (submit_async(index, head, channels, inflight_box)) {
            async_count_.fetch_add(1, std::memory_order_relaxed);
            continue;
          }
          sync_count_.fetch_add(1, std::memory_order_relaxed);
          // Fallback: synchronous command (probe, trim, errors...)
          uint32_t written = 0;
          try {
            written = process_chain(index, head, channels);
          } catch (const std::exception& e) {
            fprintf(stderr, "vhost %s: command failed: %s\n",
                    name_.c_str(), e.what());
          }
          publish_used(ring, head, written);
          published = true;
        }
        if (published && ring.call >= 0 &&
            !(__atomic_load_n(&ring.avail->flags, __ATOMIC_ACQUIRE) & 1)) {
          uint64_t one = 1;
          (void)!write(ring.call, &one, 8);
        }
      }
      for (auto& [bdev, entry] : *channels) {
        entry.first->poll(entry.second.get());
      }
    }
  }

  void publish_used(Vring& ring, uint16_t head, uint32_t written) {
    ring.used->ring[ring.used_idx % ring.num] = VringUsedElem{head, written};
    __atomic_store_n(&ring.used->idx, ++ring.used_idx, __ATOMIC_RELEASE);
  }

  // Try the async fast path; returns false to use the sync fallback.
  bool submit_async(unsigned ring_index, uint16_t head,
                    ChannelCache* channels,
                    std::shared_ptr<uint32_t> inflight) {
    Vring& ring = rings_[ring_index];
    std::vector<Iov> out, in;
    if (!collect_iovs(ring, head, &out, &in)) return false;
    const bool blk = personality_ == Personality::kBlk;
    if (!blk && ring_index < 2) return false;  // scsi control/event queue

    BdevPtr bdev;
    bool is_write = false;
    uint64_t offset = 0;
    Iov data{nullptr, 0};
    auto done = std::make_shared<InflightIo>();
    done->head = head;

    if (blk) {
      if (out.empty() || in.empty()) return false;
      uint8_t header[16];
      if (gather(header, out, sizeof(header)) < sizeof(header)) return false;
      uint32_t type;
      uint64_t sector;
      memcpy(&type, header, 4);
      memcpy(&sector, header + 8, 8);
      if (type != 0 && type != 1) return false;  // only IN/OUT
      is_write = type == 1;
      if (is_write && readonly_) return false;
      // single data extent + 1-byte status, nothing scattered
      std::vector<Iov> data_io = is_write ? out : in;
      if (is_write) {
        size_t skip = sizeof(header);
        std::vector<Iov> trimmed;
        for (Iov v : data_io) {
          if (skip >= v.len) { skip -= v.len; continue; }
          trimmed.push_back(Iov{v.base + skip, v.len - skip});
          skip = 0;
        }
        data_io.swap(trimmed);
        if (in.size() != 1 || in[0].len != 1) return false;
        done->blk_status = in[0].base;
      } else {
        if (data_io.size() != 2 || data_io[1].len != 1) return false;
        done->blk_status = data_io[1].base;
        data_io.pop_back();
      }
      if (data_io.size() != 1) return false;
      data = data_io[0];
      bdev = resolver_(0);
      offset = sector * 512;
      if (!is_write) done->data_in_bytes = static_cast<uint32_t>(data.len);
    } else {
      ScsiCmdReq req{};
      if (out.empty() || in.empty()) return false;
      if (gather(reinterpret_cast<uint8_t*>(&req), out, sizeof(req)) <
          sizeof(req)) {
        return false;
      }
      const uint8_t opc = req.cdb[0];
      if (opc != 0x28 && opc != 0x2a && opc != 0x88 && opc != 0x8a) {
        return false;  // only READ/WRITE 10/16
      }
      is_write = opc == 0x2a || opc == 0x8a;
      if (req.lun[0] != 1 ||
          (((uint16_t(req.lun[2]) << 8) | req.lun[3]) & 0x3fff) != 0) {
        return false;
      }
      bdev = resolver_(req.lun[1]);
      if (!bdev) return false;
      uint64_t lba, count;
      if (opc == 0x28 || opc == 0x2a) {
        lba = rbe32(req.cdb + 2);
        count = rbe16(req.cdb + 7);
      } else {
        lba = rbe64(req.cdb + 2);
        count = rbe32(req.cdb + 10);
      }
      const uint64_t bytes = count * bdev->block_size();
      if (bytes == 0 || lba + count > bdev->num_blocks() ||
          bytes > kMaxIoBytes) {
        return false;
      }
      offset = lba * bdev->block_size();
      if (is_write) {
        // out = [req][data]; single contiguous extent only
        size_t skip = sizeof(ScsiCmdReq);
        std::vector<Iov> trimmed;
        for (Iov v : out) {
          if (skip >= v.len) { skip -= v.len; continue; }
          trimmed.push_back(Iov{v.base + skip, v.len - skip});
          skip = 0;
        }
        if (trimmed.size() != 1 || trimmed[0].len < bytes) return false;
        data = Iov{trimmed[0].base, bytes};
        done->resp_iovs = in;
      } else {
        // in = [resp][data]
        size_t skip = sizeof(ScsiCmdResp);
        std::vector<Iov> trimmed;
        for (Iov v : in) {
          if (skip >= v.len) { skip -= v.len; continue; }
          trimmed.push_back(Iov{v.base + skip, v.len - skip});
          skip = 0;
        }
        if (trimmed.size() != 1 || trimmed[0].len < bytes) return false;
        data = Iov{trimmed[0].base, bytes};
        done->resp_iovs = in;
        done->data_in_bytes = static_cast<uint32_t>(bytes);
      }
    }
    if (!bdev || data.base == nullptr) return false;
    if (offset % bdev->block_size() != 0 ||
        data.len % bdev->block_size() != 0 ||
        offset + data.len > bdev->size_bytes()) {
      return false;
    }

    auto ch = channel_for(channels, bdev);
    Vring* ring_ptr = &ring;
    IoRequest io;
    io.op = is_write ? IoOp::kWrite : IoOp::kRead;
    io.offset = offset;
    io.length = data.len;
    io.buffer = data.base;
    io.on_complete = [this, ring_ptr, done, inflight, blk](int status) {
      // (inflight shared_ptr keeps the counter alive past an
      // abandoned drain; see ring_worker_pipelined)
      uint32_t written = done->data_in_bytes;
      if (blk) {
        *done->blk_status = status == kIoOk ? 0 : 1;
        written = (status == kIoOk ? written : 0) + 1;
      } else {
        ScsiCmdResp resp{};
        if (status != kIoOk) resp.response = kRespFailure;
        size_t resp_written = scatter(
            done->resp_iovs, reinterpret_cast<const uint8_t*>(&resp),
            sizeof(resp));
        written = static_cast<uint32_t>(resp_written) +
                  (status == kIoOk ? written : 0);
      }
      publish_used(*ring_ptr, done->head, written);
      // VRING_AVAIL_F_NO_INTERRUPT: a polling master suppresses call
      // eventfds (virtio 2.6.7) — saves one syscall per completion.
      if (ring_ptr->call >= 0 &&
          !(__atomic_load_n(&ring_ptr->avail->flags, __ATOMIC_ACQUIRE) &
            1)) {
        uint64_t one = 1;
        (void)!write(ring_ptr->call, &one, 8);
      }
      --*inflight;
    };
    bdev->submit(ch.get(), std::move(io));
    ++*inflight;
    return true;
  }

  void drain_ring(unsigned index, ChannelCache* channels) {
    Vring& ring = rings_[index];
    bool did_work = false;
    while (true) {
      uint16_t avail_idx =
          __atomic_load_n(&ring.avail->idx, __ATOMIC_ACQUIRE);
      if (ring.last_avail == avail_idx) break;
      uint16_t head = ring.avail->ring[ring.last_avail % ring.num];
      ring.last_avail++;
      uint32_t written = 0;
      try {
        written = process_chain(index, head, channels);
      } catch (const std::exception& e) {
        // A bad buffer (e.g. unregisterable guest memory) must fail
        // the one command, never kill the ring worker / daemon.
        fprintf(stderr, "vhost %s: command failed: %s\n", name_.c_str(),
                e.what());
      }
      ring.used->ring[ring.used_idx % ring.num] =
          VringUsedElem{head, written};
      __atomic_store_n(&ring.used->idx, ++ring.used_idx, __ATOMIC_RELEASE);
      did_work = true;
    }
    if (did_work && ring.call >= 0 &&
        !(__atomic_load_n(&ring.avail->flags, __ATOMIC_ACQUIRE) & 1)) {
      uint64_t one = 1;
      (void)!write(ring.call, &one, 8);
    }
  }

  // Walk a descriptor chain into readable/writable iov lists.
  bool collect_iovs(const Vring& ring, uint16_t head, std::vector<Iov>* out,
                    std::vector<Iov>* in) {
    uint16_t idx = head;
    for (int hops = 0; hops < 256; ++hops) {
      if (idx >= ring.num) return false;
      VringDesc d = ring.desc[idx];
      if (d.flags & kDescIndirect) {
        auto* table = reinterpret_cast<VringDesc*>(gpa_to_ptr(d.addr, d.len));
        if (!table || d.len % sizeof(VringDesc)) return false;
        const uint32_t count = d.len / sizeof(VringDesc);
        // Follow each entry's `next` link (spec-legal chains need not
        // be laid out sequentially), bounded like the outer walk.
        uint32_t j = 0;
        bool terminated = false;
        for (uint32_t ihops = 0; ihops < 256; ++ihops) {
          if (j >= count) return false;
          VringDesc ind = table[j];
          if (ind.flags & kDescIndirect) return false;  // no nesting
          uint8_t* p = gpa_to_ptr(ind.addr, ind.len);
          if (!p) return false;
          ((ind.flags & kDescWrite) ? in : out)->push_back(Iov{p, ind.len});
          if (!(ind.flags & kDescNext)) {
            terminated = true;
            break;
          }
          j = ind.next;
        }
        if (!terminated) return false;
      } else {
        uint8_t* p = gpa_to_ptr(d.addr, d.len);
        if (!p) return false;
        ((d.flags & kDescWrite) ? in : out)->push_back(Iov{p, d.len});
      }
      if (!(d.flags & kDescNext)) return true;
      idx = d.next;
    }
    return false;
  }

  // Returns bytes written into device-writable descriptors (the value
  // published in the used ring).
  uint32_t process_chain(unsigned ring_index, uint16_t head,
                         ChannelCache* channels) {
    Vring& ring = rings_[ring_index];
    std::vector<Iov> out, in;
    if (!collect_iovs(ring, head, &out, &in)) return 0;
    if (personality_ == Personality::kBlk) {
      return handle_blk_cmd(out, in, channels);
    }
    if (ring_index >= 2) return handle_scsi_cmd(out, in, channels);
    // Control / event queue: acknowledge TMFs and AN queries with an
    // all-zero response (FUNCTION_COMPLETE / S_OK).
    size_t n = iov_total(in);
    for (Iov& v : in) memset(v.base, 0, v.len);
    return static_cast<uint32_t>(n);
  }

  // -- virtio-blk command execution --------------------------------------

  // Request: virtio_blk_outhdr {u32 type; u32 ioprio; u64 sector} in
  // device-readable descriptors, then data; the LAST device-writable
  // byte is the status (OK=0, IOERR=1, UNSUPP=2). Sectors are always
  // 512-byte units regardless of bdev block size.
  uint32_t handle_blk_cmd(const std::vector<Iov>& out,
                          const std::vector<Iov>& in,
                          ChannelCache* channels) {
    constexpr uint32_t kTypeIn = 0, kTypeOut = 1, kTypeFlush = 4,
                       kTypeGetId = 8, kTypeDiscard = 11,
                       kTypeWriteZeroes = 13;
    constexpr uint8_t kStatusOk = 0, kStatusIoErr = 1, kStatusUnsupp = 2;
    if (out.empty() || in.empty()) return 0;
    uint8_t header[16];
    if (gather(header, out, sizeof(header)) < sizeof(header)) return 0;
    uint32_t type;
    uint64_t sector;
    memcpy(&type, header, 4);
    memcpy(&sector, header + 8, 8);
    size_t id_bytes = 0;

    // status byte = last writable byte; data-in = everything before it.
    // Drop guest-supplied zero-length writable descriptors first: a
    // zero-length final iov would make base-1 an OOB status write and
    // wrap `--len` to SIZE_MAX.
    std::vector<Iov> data_in;
    data_in.reserve(in.size());
    for (const Iov& v : in) {
      if (v.len > 0) data_in.push_back(v);
    }
    if (data_in.empty()) return 0;
    uint8_t* status_ptr;
    {
      Iov& last = data_in.back();
      status_ptr = last.base + last.len - 1;
      if (--last.len == 0) data_in.pop_back();
    }
    // data-out = readable bytes after the header
    std::vector<Iov> data_out;
    {
      size_t skip = sizeof(header);
      for (Iov v : out) {
        if (skip >= v.len) {
          skip -= v.len;
          continue;
        }
        data_out.push_back(Iov{v.base + skip, v.len - skip});
        skip = 0;
      }
    }
    uint8_t status = kStatusOk;
    BdevPtr bdev = resolver_(0);
    if (!bdev) {
      status = kStatusIoErr;
    } else if (type == kTypeFlush) {
      status = kStatusOk;
    } else if (type == kTypeGetId) {
      char id[20] = {0};
      strncpy(id, bdev->name().c_str(), sizeof(id) - 1);
      id_bytes = scatter(
          data_in, reinterpret_cast<uint8_t*>(id),
          std::min<size_t>(sizeof(id), iov_total(data_in)));
      status = kStatusOk;
    } else if (type == kTypeDiscard || type == kTypeWriteZeroes) {
      // data-out = N x 16-byte segments {sector u64; num u32; flags u32}
      const uint64_t seg_bytes = iov_total(data_out);
      status = kStatusOk;
      if (readonly_ || seg_bytes % 16 != 0 || seg_bytes == 0 ||
          seg_bytes / 16 > 256) {
        status = kStatusIoErr;
      } else {
        std::vector<uint8_t> segs(seg_bytes);
        gather(segs.data(), data_out, seg_bytes);
        for (size_t off = 0; off < seg_bytes && status == kStatusOk;
             off += 16) {
          uint64_t seg_sector;
          uint32_t num;
          memcpy(&seg_sector, segs.data() + off, 8);
          memcpy(&num, segs.data() + off + 8, 4);
          if (num == 0) continue;
          const uint64_t z_off = seg_sector * 512;
          const uint64_t z_len = static_cast<uint64_t>(num) * 512;
          if (z_off % bdev->block_size() != 0 ||
              z_len % bdev->block_size() != 0 ||
              z_off + z_len > bdev->size_bytes() || z_len > kMaxIoBytes) {
            status = kStatusIoErr;
            break;
          }
          auto ch = channel_for(channels, bdev);
          int io_status = 1;
          bool done = false;
          IoRequest zero;
          zero.op = IoOp::kFill;
          zero.offset = z_off;
          zero.length = z_len;
          zero.fill = 0;
          zero.on_complete = [&](int st) {
            io_status = st;
            done = true;
          };
          bdev->submit(ch.get(), std::move(zero));
          while (!done) bdev->poll(ch.get());
          if (io_status != kIoOk) status = kStatusIoErr;
        }
      }
    } else if (type == kTypeIn || type == kTypeOut) {
      const bool is_write = type == kTypeOut;
      const std::vector<Iov>& iovs = is_write ? data_out : data_in;
      const uint64_t bytes = iov_total(iovs);
      const uint64_t offset = sector * 512;
      if ((is_write && readonly_) || bytes == 0 ||
          bytes % bdev->block_size() != 0 ||
          offset % bdev->block_size() != 0 ||
          offset + bytes > bdev->size_bytes() || bytes > kMaxIoBytes) {
        status = kStatusIoErr;
      } else if (iovs.size() == 1) {
        if (bdev_io(channels, bdev, is_write ? IoOp::kWrite : IoOp::kRead,
                    offset, iovs[0].base, bytes) != kIoOk) {
          status = kStatusIoErr;
        }
      } else {
        uint8_t* bounce = static_cast<uint8_t*>(alloc_pinned(bytes));
        if (is_write) gather(bounce, iovs, bytes);
        int io_status =
            bdev_io(channels, bdev, is_write ? IoOp::kWrite : IoOp::kRead,
                    offset, bounce, bytes);
        if (io_status == kIoOk && !is_write) scatter(iovs, bounce, bytes);
        free_pinned(bounce);
        if (io_status != kIoOk) status = kStatusIoErr;
      }
    } else {
      status = kStatusUnsupp;
    }
    *status_ptr = status;
    const uint32_t data_written =
        (type == kTypeIn && status == kStatusOk)
            ? static_cast<uint32_t>(iov_total(data_in))
            : (type == kTypeGetId ? static_cast<uint32_t>(id_bytes)
                                  : 0u);
    return data_written + 1;  // +1 for the status byte
  }

  // -- SCSI command execution --------------------------------------------

  static std::shared_ptr<IoChannel> channel_for(ChannelCache* cache,
                                                const BdevPtr& bdev) {
    auto it = cache->find(bdev.get());
    if (it != cache->end()) return it->second.second;
    auto ch = bdev->get_channel();
    (*cache)[bdev.get()] = {bdev, ch};  // pins the bdev alive with it
    return ch;
  }

  static int bdev_io(ChannelCache* cache, const BdevPtr& bdev, IoOp op,
                     uint64_t offset, void* buf, uint64_t len) {
    auto ch = channel_for(cache, bdev);
    int status = 1;
    bool done = false;
    IoRequest req;
    req.op = op;
    req.offset = offset;
    req.length = len;
    req.buffer = buf;
    req.on_complete = [&](int s) {
      status = s;
      done = true;
    };
    bdev->submit(ch.get(), std::move(req));
    while (!done) bdev->poll(ch.get());
    return status;
  }

  static void build_sense(ScsiCmdResp* resp, uint8_t key, uint8_t asc,
                          uint8_t ascq) {
    resp->status = kStatusCheckCondition;
    resp->response = kRespOk;
    memset(resp->sense, 0, kSenseSize);
    resp->sense[0] = 0x70;  // fixed format, current error
    resp->sense[2] = key;
    resp->sense[7] = 10;  // additional length
    resp->sense[12] = asc;
    resp->sense[13] = ascq;
    resp->sense_len = 18;
  }

  uint32_t handle_scsi_cmd(const std::vector<Iov>& out,
                           const std::vector<Iov>& in,
                           ChannelCache* channels) {
    if (out.empty() || in.empty()) return 0;
    ScsiCmdReq req{};
    if (gather(reinterpret_cast<uint8_t*>(&req), out, sizeof(req)) <
        sizeof(req)) {
      return 0;
    }
    // Data-out = readable bytes after the request header.
    std::vector<Iov> data_out;
    {
      size_t skip = sizeof(ScsiCmdReq);
      for (Iov v : out) {
        if (skip >= v.len) {
          skip -= v.len;
          continue;
        }
        data_out.push_back(Iov{v.base + skip, v.len - skip});
        skip = 0;
      }
    }
    // Data-in = writable bytes after the response.
    ScsiCmdResp resp{};
    std::vector<Iov> data_in;
    {
      size_t skip = sizeof(ScsiCmdResp);
      for (Iov v : in) {
        if (skip >= v.len) {
          skip -= v.len;
          continue;
        }
        data_in.push_back(Iov{v.base + skip, v.len - skip});
        skip = 0;
      }
    }

    size_t data_in_written =
        execute(req, data_out, data_in, &resp, channels);

    // Response goes into the first sizeof(resp) writable bytes.
    size_t resp_written =
        scatter(in, reinterpret_cast<const uint8_t*>(&resp), sizeof(resp));
    return static_cast<uint32_t>(resp_written + data_in_written);
  }

  // Returns the number of data-in bytes produced (for the used-ring
  // "written" field).
  size_t execute(const ScsiCmdReq& req, const std::vector<Iov>& data_out,
                 const std::vector<Iov>& data_in, ScsiCmdResp* resp,
                 ChannelCache* channels) {
    // Single-level LUN addressing: lun[0]==1 selects the target by
    // lun[1]; bytes 2-3 carry the LUN (flat-space). Only LUN 0 exists.
    if (req.lun[0] != 1) {
      resp->response = kRespBadTarget;
      return 0;
    }
    int target = req.lun[1];
    uint16_t lun = ((uint16_t(req.lun[2]) << 8) | req.lun[3]) & 0x3fff;
    BdevPtr bdev = resolver_(target);
    if (!bdev) {
      resp->response = kRespBadTarget;
      return 0;
    }
    resp->response = kRespOk;
    resp->status = kStatusGood;
    if (lun != 0) {
      // Exists-but-wrong-LUN: LOGICAL UNIT NOT SUPPORTED.
      build_sense(resp, 0x05, 0x25, 0x00);
      return 0;
    }

    const uint8_t* cdb = req.cdb;
    const uint64_t block = bdev->block_size();
    const uint64_t nblocks = bdev->num_blocks();
    switch (cdb[0]) {
      case 0x00:  // TEST UNIT READY
      case 0x35:  // SYNCHRONIZE CACHE(10)
      case 0x91:  // SYNCHRONIZE CACHE(16)
        return 0;
      case 0x42: {  // UNMAP: zero the listed extents (thin-provision trim)
        const uint32_t param_len = rbe16(cdb + 7);
        if (param_len < 8) return 0;  // empty list: success
        std::vector<uint8_t> list(param_len);
        if (gather(list.data(), data_out, param_len) < param_len) {
          build_sense(resp, 0x05, 0x24, 0x00);
          return 0;
        }
        const uint16_t desc_bytes = rbe16(list.data() + 2);
        for (uint32_t off = 8; off + 16 <= 8u + desc_bytes &&
                               off + 16 <= param_len;
             off += 16) {
          const uint64_t lba = rbe64(list.data() + off);
          const uint32_t nlb = rbe32(list.data() + off + 8);
          if (nlb == 0) continue;
          if (lba + nlb > nblocks ||
              static_cast<uint64_t>(nlb) * block > kMaxIoBytes) {
            build_sense(resp, 0x05, 0x21, 0x00);
            return 0;
          }
          IoRequest zero;
          auto ch = channel_for(channels, bdev);
          int status = 1;
          bool done = false;
          zero.op = IoOp::kFill;
          zero.offset = lba * block;
          zero.length = static_cast<uint64_t>(nlb) * block;
          zero.fill = 0;
          zero.on_complete = [&](int st) {
            status = st;
            done = true;
          };
          bdev->submit(ch.get(), std::move(zero));
          while (!done) bdev->poll(ch.get());
          if (status != kIoOk) {
            resp->response = kRespFailure;
            return 0;
          }
        }
        return 0;
      }
      case 0x12: {  // INQUIRY
        uint8_t buf[96] = {};
        size_t len;
        if (cdb[1] & 0x1) {  // EVPD
          switch (cdb[2]) {
            case 0x00:  // supported VPD pages
              buf[1] = 0x00;
              buf[3] = 4;
              buf[4] = 0x00;
              buf[5] = 0x80;
              buf[6] = 0x83;
              buf[7] = 0xB2;
              len = 8;
              break;
            case 0x83: {  // device identification: T10 vendor-id
              // designator from the bdev uuid (/dev/disk/by-id source)
              const std::string& uuid = bdev->uuid();
              const size_t id_len = std::min<size_t>(uuid.size(), 36);
              buf[1] = 0x83;
              buf[4] = 0x02;  // codeset: ASCII
              buf[5] = 0x01;  // assoc LUN, designator type: T10 vendor id
              buf[7] = static_cast<uint8_t>(8 + id_len);
              memcpy(buf + 8, "HIPSTORE", 8);
              memcpy(buf + 16, uuid.data(), id_len);
              len = 16 + id_len;
              buf[3] = static_cast<uint8_t>(len - 4);
              break;
            }
            case 0xB2:  // logical block provisioning: LBPU (UNMAP)
              buf[1] = 0xB2;
              buf[3] = 4;
              buf[5] = 0x80;  // LBPU
              len = 8;
              break;
            case 0x80: {  // unit serial number
              const std::string& uuid = bdev->uuid();
              size_t n = std::min<size_t>(uuid.size(), 36);
              buf[1] = 0x80;
              buf[3] = n;
              memcpy(buf + 4, uuid.data(), n);
              len = 4 + n;
              break;
            }
            default:
              build_sense(resp, 0x05, 0x24, 0x00);  // INVALID FIELD IN CDB
              return 0;
          }
        } else {
          buf[0] = 0x00;  // direct-access block device, connected
          buf[2] = 0x06;  // SPC-4
          buf[3] = 0x02;  // response data format
          buf[4] = 91;    // additional length (96 - 5)
          memcpy(buf + 8, "HIPSTORE", 8);
          memset(buf + 16, ' ', 16);
          memcpy(buf + 16, bdev->name().data(),
                 std::min<size_t>(bdev->name().size(), 16));
          memcpy(buf + 32, "0001", 4);
          len = 96;
        }
        size_t alloc = rbe16(cdb + 3);
        return scatter(data_in, buf, std::min(len, alloc));
      }
      case 0x25: {  // READ CAPACITY(10)
        uint8_t buf[8];
        uint64_t last = nblocks - 1;
        be32(buf, last > 0xffffffffull ? 0xffffffffu
                                       : static_cast<uint32_t>(last));
        be32(buf + 4, static_cast<uint32_t>(block));
        return scatter(data_in, buf, sizeof(buf));
      }
      case 0x9e: {  // SERVICE ACTION IN(16)
        if ((cdb[1] & 0x1f) != 0x10) {  // READ CAPACITY(16)
          build_sense(resp, 0x05, 0x20, 0x00);
          return 0;
        }
        uint8_t buf[32] = {};
        be64(buf, nblocks - 1);
        be32(buf + 8, static_cast<uint32_t>(block));
        buf[14] = 0x80;  // LBPME: logical block provisioning (UNMAP ok)
        uint32_t alloc = rbe32(cdb + 10);
        return scatter(data_in, buf, std::min<size_t>(sizeof(buf), alloc));
      }
      case 0xa0: {  // REPORT LUNS
        uint8_t buf[16] = {};
        be32(buf, 8);  // one 8-byte LUN entry (LUN 0 = all zeros)
        uint32_t alloc = rbe32(cdb + 6);
        return scatter(data_in, buf, std::min<size_t>(sizeof(buf), alloc));
      }
      case 0x1a: {  // MODE SENSE(6)
        uint8_t buf[4] = {3, 0, 0, 0};  // no pages, no block descriptors
        return scatter(data_in, buf, std::min<size_t>(sizeof(buf), cdb[4]));
      }
      case 0x08:    // READ(6)
      case 0x28:    // READ(10)
      case 0x88:    // READ(16)
      case 0x0a:    // WRITE(6)
      case 0x2a:    // WRITE(10)
      case 0x8a: {  // WRITE(16)
        bool is_write = cdb[0] == 0x0a || cdb[0] == 0x2a || cdb[0] == 0x8a;
        uint64_t lba;
        uint64_t count;
        switch (cdb[0] & 0xe0) {
          case 0x00:  // 6-byte
            lba = (uint64_t(cdb[1] & 0x1f) << 16) | rbe16(cdb + 2);
            count = cdb[4] ? cdb[4] : 256;
            break;
          case 0x20:  // 10-byte
            lba = rbe32(cdb + 2);
            count = rbe16(cdb + 7);
            break;
          default:  // 16-byte
            lba = rbe64(cdb + 2);
            count = rbe32(cdb + 10);
            break;
        }
        uint64_t bytes = count * block;
        if (lba + count > nblocks || bytes > kMaxIoBytes) {
          build_sense(resp, 0x05, 0x21, 0x00);  // LBA OUT OF RANGE
          return 0;
        }
        if (bytes == 0) return 0;
        const std::vector<Iov>& iovs = is_write ? data_out : data_in;
        int status;
        if (iovs.size() == 1 && iovs[0].len >= bytes) {
          status = bdev_io(channels, bdev,
                           is_write ? IoOp::kWrite : IoOp::kRead,
                           lba * block, iovs[0].base, bytes);
        } else {
          if (iov_total(iovs) < bytes) {
            build_sense(resp, 0x05, 0x24, 0x00);
            return 0;
          }
          // Pinned bounce: the HBM engine DMAs from host memory, so a
          // plain vector is not eligible. alloc_pinned falls back to
          // malloc on CPU-only daemons.
          uint8_t* bounce = static_cast<uint8_t*>(alloc_pinned(bytes));
          if (is_write) gather(bounce, iovs, bytes);
          status = bdev_io(channels, bdev,
                           is_write ? IoOp::kWrite : IoOp::kRead,
                           lba * block, bounce, bytes);
          if (!is_write && status == kIoOk) {
            scatter(iovs, bounce, bytes);
          }
          free_pinned(bounce);
        }
        if (status != kIoOk) {
          resp->response = kRespFailure;
          return 0;
        }
        return is_write ? 0 : bytes;
      }
      default:
        build_sense(resp, 0x05, 0x20, 0x00);  // INVALID COMMAND OPCODE
        return 0;
    }
  }

  const std::string name_;
  const std::string socket_path_;
  const std::function<BdevPtr(int)> resolver_;
  const Personality personality_ = Personality::kScsi;
  const bool readonly_ = false;

  int listen_fd_ = -1;
  int stop_pipe_[2] = {-1, -1};
  std::thread accept_thread_;
  std::atomic<bool> stopping_{false};
  // Pipelined-worker dispatch mix (HIPSTORE_DEBUG prints on session
  // reset): fast-path async submissions vs per-command sync fallbacks
  // — the decisive counter when a front-end measures slower than the
  // engine can serve.
  std::atomic<uint64_t> async_count_{0};
  std::atomic<uint64_t> sync_count_{0};

  uint64_t negotiated_features_ = 0;
  uint64_t protocol_features_ = 0;
  std::vector<Region> regions_;
  Vring rings_[kMaxVrings];

};

// ---- module state ------------------------------------------------------

namespace {
std::mutex g_vhost_mutex;
std::string g_socket_dir = "/var/tmp";
std::set<VhostDevPtr> g_devices;
}  // namespace

void vhost_set_socket_dir(const std::string& dir) {
  std::lock_guard<std::mutex> lock(g_vhost_mutex);
  g_socket_dir = dir.empty() ? "." : dir;
}

std::string vhost_socket_path(const std::string& ctrlr_name) {
  std::lock_guard<std::mutex> lock(g_vhost_mutex);
  return g_socket_dir + "/" + ctrlr_name;
}

VhostDevPtr vhost_start(const std::string& name,
                        std::function<BdevPtr(int)> resolver) {
  auto dev = std::make_shared<VhostUserScsiDev>(name, vhost_socket_path(name),
                                                std::move(resolver));
  dev->start();
  std::lock_guard<std::mutex> lock(g_vhost_mutex);
  g_devices.insert(dev);
  return dev;
}

VhostDevPtr vhost_start_blk(const std::string& name,
                            std::function<BdevPtr(int)> resolver,
                            bool readonly) {
  auto dev = std::make_shared<VhostUserScsiDev>(
      name, vhost_socket_path(name), std::move(resolver),
      VhostUserScsiDev::Personality::kBlk, readonly);
  dev->start();
  std::lock_guard<std::mutex> lock(g_vhost_mutex);
  g_devices.insert(dev);
  return dev;
}

void vhost_stop(const VhostDevPtr& dev) {
  if (!dev) return;
  dev->stop();
  std::lock_guard<std::mutex> lock(g_vhost_mutex);
  g_devices.erase(dev);
}

void vhost_stop_all() {
  std::set<VhostDevPtr> devices;
  {
    std::lock_guard<std::mutex> lock(g_vhost_mutex);
    devices.swap(g_devices);
  }
  for (const VhostDevPtr& dev : devices) dev->stop();
}

}  // namespace hipstore
