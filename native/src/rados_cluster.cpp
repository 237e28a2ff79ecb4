// Loopback fake RADOS cluster (mon+osd on one messenger endpoint).
//
// The in-repo peer for the msgr-v1 client (rados_client.cpp): the
// NVMe/TCP loopback-target pattern (nvmf_target.cpp) applied to
// RADOS, replacing the reference's dependency on a real Ceph cluster
// (vendor/github.com/spdk/spdk/lib/bdev/rbd/bdev_rbd.c connects out
// via librados). Single endpoint: every placement group maps to OSD 0
// here, so the client needs no osdmap — documented protocol subset.
//
// MI355X-native part: the object store is an arena in HBM (one
// `object_bytes` slot per object), and message data CRC32Cs are
// computed/verified by the GPU kernel against the HBM-resident bytes:
//   - inbound writes land in HBM first, then the footer's data_crc is
//     checked against a GPU-computed CRC of the landed extent (k_crc32c
//     per-4KiB blocks + host GF(2) combine ladder, crc32c.cpp);
//   - outbound reads carry a data_crc the GPU computed from HBM before
//     the bytes are streamed to the socket.
// Unaligned/partial extents fall back to the software CRC.

#include <arpa/inet.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <algorithm>
#include <atomic>
#include <cerrno>
#include <chrono>
#include <cstdio>
#include <map>
#include <memory>
#include <mutex>
#include <stdexcept>
#include <thread>
#include <vector>

#include "hipstore/crc32c.h"
#include "hipstore/engine.h"
#include "hipstore/rados.h"
#include "rados_common.h"

namespace hipstore {

namespace {

using namespace rados;

constexpr int32_t kEnoent = -2;
constexpr int32_t kEio = -5;
constexpr int32_t kEnospc = -28;
constexpr int32_t kEinval = -22;

// Bounded sync I/O against the arena on a caller-owned channel (the
// engine's run_sync creates a channel per call; connection threads
// keep one hot instead).
int arena_io(Bdev* bdev, IoChannel* ch, IoOp op, uint64_t offset,
             void* buf, uint64_t len) {
  IoRequest req;
  req.op = op;
  req.offset = offset;
  req.length = len;
  req.buffer = buf;
  struct State {
    std::atomic<int> result{kIoFailed};
    std::atomic<bool> done{false};
  };
  auto state = std::make_shared<State>();
  req.on_complete = [state](int status) {
    state->result.store(status, std::memory_order_relaxed);
    state->done.store(true, std::memory_order_release);
  };
  bdev->submit(ch, std::move(req));
  const auto deadline =
      std::chrono::steady_clock::now() + std::chrono::seconds(30);
  while (!state->done.load(std::memory_order_acquire)) {
    bdev->poll(ch);
    if (std::chrono::steady_clock::now() > deadline) return kIoFailed;
  }
  return state->result.load(std::memory_order_relaxed);
}

class RadosClusterImpl : public RadosCluster,
                         public std::enable_shared_from_this<RadosClusterImpl> {
 public:
  RadosClusterImpl(uint16_t port, uint64_t arena_mb, bool use_hbm,
                   int device, uint64_t object_bytes)
      : object_bytes_(object_bytes) {
    const uint64_t arena_bytes = arena_mb << 20;
    slot_count_ = arena_bytes / object_bytes_;
    if (slot_count_ == 0) {
      throw std::runtime_error("rados: arena smaller than one object");
    }
    hbm_ = use_hbm && gpu_available();
    if (hbm_) {
      // Batched engine, deliberately: the cluster executes one OSD op
      // at a time under its mutex with TCP gaps in between, so a
      // persistent service kernel idles out and pays a relaunch per
      // op (measured 9k IOPS p50 3 ms vs 27k batched). Per-op arena
      // round-trip latency, not engine peak, bounds this path.
      arena_ = create_hbm_bdev("rados-arena", 4096,
                               arena_bytes / 4096, device);
    } else {
      arena_ = create_malloc_bdev("rados-arena", 4096, arena_bytes / 4096);
    }
    slot_used_.assign(slot_count_, false);

    listen_fd_ = socket(AF_INET, SOCK_STREAM, 0);
    if (listen_fd_ < 0) throw std::runtime_error("rados: socket failed");
    int one = 1;
    setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in sa{};
    sa.sin_family = AF_INET;
    sa.sin_port = htons(port);
    sa.sin_addr.s_addr = htonl(INADDR_LOOPBACK);
    if (bind(listen_fd_, reinterpret_cast<sockaddr*>(&sa), sizeof(sa)) < 0 ||
        listen(listen_fd_, 16) < 0) {
      close(listen_fd_);
      throw std::runtime_error("rados: bind/listen failed");
    }
    socklen_t slen = sizeof(sa);
    getsockname(listen_fd_, reinterpret_cast<sockaddr*>(&sa), &slen);
    port_ = ntohs(sa.sin_port);
  }

  void start() {
    accept_thread_ = std::thread([self = shared_from_this()] {
      self->accept_loop();
    });
  }

  ~RadosClusterImpl() override { stop(); }

  uint16_t port() const override { return port_; }

  uint64_t object_count() const override {
    std::lock_guard<std::mutex> lock(mutex_);
    return objects_.size();
  }

  void stop() override {
    bool expected = false;
    if (!stopping_.compare_exchange_strong(expected, true)) return;
    shutdown(listen_fd_, SHUT_RDWR);
    close(listen_fd_);
    if (accept_thread_.joinable()) accept_thread_.join();
    std::vector<std::thread> conns;
    {
      std::lock_guard<std::mutex> lock(mutex_);
      conns.swap(conn_threads_);
      for (int fd : conn_fds_) shutdown(fd, SHUT_RDWR);
    }
    for (auto& t : conns) t.join();
  }

 private:
  struct ObjMeta {
    uint64_t slot;
    uint64_t length;  // logical object length (<= object_bytes_)
  };

  void accept_loop() {
    while (!stopping_.load()) {
      int fd = accept(listen_fd_, nullptr, nullptr);
      if (fd < 0) {
        if (stopping_.load()) break;
        if (errno == EINTR) continue;
        break;
      }
      int one = 1;
      setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
      std::lock_guard<std::mutex> lock(mutex_);
      conn_fds_.push_back(fd);
      conn_threads_.emplace_back([self = shared_from_this(), fd] {
        self->serve(fd);
      });
    }
  }

  void serve(int fd) {
    if (!msgr_handshake(fd, /*server=*/true, kEntityOsd)) {
      close(fd);
      return;
    }
    auto channel = arena_->get_channel();
    uint8_t* bounce =
        static_cast<uint8_t*>(alloc_pinned(object_bytes_));
    uint64_t reply_seq = 0;
    MsgrFrame frame;
    // Data CRC is NOT verified at receive time on the HBM path: the
    // payload lands in HBM first and the GPU computes the check CRC.
    while (!stopping_.load() &&
           msgr_recv(fd, &frame, /*verify_data_crc=*/false)) {
      if (frame.header.type == kMsgPing) continue;
      if (frame.header.type != kMsgOsdOp) break;
      OsdOpReply reply;
      std::vector<uint8_t> reply_data;
      uint32_t reply_data_crc = ~0u;  // ~0u => software CRC in msgr_send
      try {
        OsdOpRequest req = decode_osd_op_front(frame.front);
        reply.oid = req.oid;
        reply.result = 0;
        size_t data_off = 0;
        for (CephOsdOp op : req.ops) {
          int32_t r = execute(channel.get(), bounce, req.pool, req.oid,
                              &op, frame, data_off, &reply_data,
                              &reply_data_crc);
          data_off += op.payload_len;
          reply.ops.push_back(op);
          if (r < 0) {
            reply.result = r;
            break;
          }
        }
      } catch (const std::exception& e) {
        fprintf(stderr, "[rados-cluster] bad request: %s\n", e.what());
        reply.result = kEinval;
      }
      if (!msgr_send(fd, ++reply_seq, frame.header.tid, kMsgOsdOpReply,
                     encode_osd_op_reply_front(reply), reply_data.data(),
                     static_cast<uint32_t>(reply_data.size()),
                     reply_data.empty() ? ~0u : reply_data_crc)) {
        break;
      }
    }
    free_pinned(bounce);
    close(fd);
  }

  // Arena extents are 4 KiB-block device I/O; RADOS ops are
  // byte-granular (the 8-byte image header, odd tails), so unaligned
  // edges go through a read-modify-write on the connection's bounce.
  // Callers hold mutex_ — the fake cluster serializes ops
  // cluster-wide (one PG, in effect), which makes RMW race-free.
  int arena_read_any(IoChannel* channel, uint8_t* bounce, uint64_t off,
                     uint8_t* dst, uint64_t len) {
    const uint64_t a0 = off & ~4095ull;
    const uint64_t a1 = (off + len + 4095) & ~4095ull;
    if (arena_io(arena_.get(), channel, IoOp::kRead, a0, bounce,
                 a1 - a0) != kIoOk) {
      return kIoFailed;
    }
    memcpy(dst, bounce + (off - a0), len);
    return kIoOk;
  }

  int arena_write_any(IoChannel* channel, uint8_t* bounce, uint64_t off,
                      const uint8_t* src, uint64_t len) {
    const uint64_t a0 = off & ~4095ull;
    const uint64_t a1 = (off + len + 4095) & ~4095ull;
    if ((off != a0 || off + len != a1) &&
        arena_io(arena_.get(), channel, IoOp::kRead, a0, bounce,
                 a1 - a0) != kIoOk) {
      return kIoFailed;
    }
    memcpy(bounce + (off - a0), src, len);
    return arena_io(arena_.get(), channel, IoOp::kWrite, a0, bounce,
                    a1 - a0);
  }

  // CRC32C of [offset, offset+len) of the arena. GPU kernel (per-4KiB
  // CRCs + GF(2) combine ladder) for LARGE aligned extents on the HBM
  // arena, where in-place HBM-rate hashing amortizes the launch+sync
  // round trip and beats reading the extent back to the host;
  // software (SSE4.2) on a readback otherwise — a per-4KiB-op GPU
  // launch under the cluster mutex measured ~200 us/op and collapsed
  // randwrite to 5k IOPS (p99 28 ms).
  uint32_t arena_crc(IoChannel* channel, uint8_t* bounce, uint64_t offset,
                     uint64_t len) {
    if (hbm_ && offset % 4096 == 0 && len % 4096 == 0 &&
        len >= (1ull << 20)) {
      const uint32_t count = static_cast<uint32_t>(len / 4096);
      std::vector<uint32_t> crcs(count);
      crc32c_hbm_blocks(arena_.get(), offset, 4096, count, crcs.data());
      uint32_t crc = crcs[0];
      for (uint32_t i = 1; i < count; ++i) {
        crc = crc32c_combine(crc, crcs[i], 4096);
      }
      return crc;
    }
    std::vector<uint8_t> tmp(len);
    if (arena_read_any(channel, bounce, offset, tmp.data(), len) != kIoOk) {
      return ~0u;
    }
    return crc32c_sw(0, tmp.data(), len);
  }

  int32_t execute(IoChannel* channel, uint8_t* bounce, uint64_t pool,
                  const std::string& oid, CephOsdOp* op,
                  const MsgrFrame& frame, size_t data_off,
                  std::vector<uint8_t>* reply_data,
                  uint32_t* reply_data_crc) {
    const auto key = std::make_pair(pool, oid);
    const bool is_write =
        (op->op & kOsdOpModeWr) != 0;
    if (op->offset + op->length > object_bytes_ &&
        op->op != kOsdOpDelete && op->op != kOsdOpStat) {
      return kEinval;
    }
    std::unique_lock<std::mutex> lock(mutex_);
    auto it = objects_.find(key);
    if (!is_write && it == objects_.end()) return kEnoent;

    switch (op->op) {
      case kOsdOpRead: {
        const ObjMeta meta = it->second;
        // Short-read semantics past the object's logical length.
        const uint64_t avail =
            op->offset >= meta.length ? 0 : meta.length - op->offset;
        const uint64_t len = std::min<uint64_t>(op->length, avail);
        op->payload_len = static_cast<uint32_t>(len);
        if (len == 0) return 0;
        const uint64_t arena_off = meta.slot * object_bytes_ + op->offset;
        const size_t reply_off = reply_data->size();
        reply_data->resize(reply_off + len);
        if (arena_read_any(channel, bounce, arena_off,
                           reply_data->data() + reply_off, len) != kIoOk) {
          return kEio;
        }
        // Outbound data CRC (whole reply data is this op's payload:
        // the client sends one op/message). Large aligned extents
        // hash in place in HBM on the GPU; otherwise SSE4.2 over the
        // reply bytes just read — never a second arena readback.
        if (reply_off == 0) {
          if (hbm_ && arena_off % 4096 == 0 && len % 4096 == 0 &&
              len >= (1ull << 20)) {
            *reply_data_crc = arena_crc(channel, bounce, arena_off, len);
          } else {
            *reply_data_crc =
                crc32c_sw(0, reply_data->data() + reply_off, len);
          }
        } else {
          *reply_data_crc = ~0u;  // multi-op reply: software CRC
        }
        return 0;
      }
      case kOsdOpStat: {
        const ObjMeta meta = it->second;
        // le64 size + le32 sec + le32 nsec (utime).
        std::vector<uint8_t> payload;
        put_le<uint64_t>(&payload, meta.length);
        put_le<uint32_t>(&payload, 0);
        put_le<uint32_t>(&payload, 0);
        op->payload_len = static_cast<uint32_t>(payload.size());
        reply_data->insert(reply_data->end(), payload.begin(),
                           payload.end());
        *reply_data_crc = ~0u;
        return 0;
      }
      case kOsdOpWrite:
      case kOsdOpWriteFull: {
        if (op->payload_len != op->length ||
            data_off + op->payload_len > frame.data.size()) {
          return kEinval;
        }
        ObjMeta* meta = ensure_object_locked(key);
        if (meta == nullptr) return kEnospc;
        const uint64_t arena_off =
            meta->slot * object_bytes_ + op->offset;
        meta->length =
            op->op == kOsdOpWriteFull
                ? op->offset + op->length
                : std::max<uint64_t>(meta->length,
                                     op->offset + op->length);
        if (arena_write_any(channel, bounce, arena_off,
                            frame.data.data() + data_off,
                            op->length) != kIoOk) {
          return kEio;
        }
        // Data-CRC check against the messenger footer. Whole-object
        // aligned writes verify the LANDED HBM extent with the GPU
        // kernel (end-to-end: wire + DMA); small ops verify the
        // received buffer with SSE4.2, which is what Ceph's
        // messenger itself does — a per-small-op GPU launch under
        // the cluster mutex measured ~200 us and collapsed
        // randwrite to 5k IOPS.
        if (hbm_ && frame.data.size() == op->payload_len &&
            arena_off % 4096 == 0 && op->length % 4096 == 0 &&
            op->length >= (1ull << 20)) {
          const uint32_t crc =
              arena_crc(channel, bounce, arena_off, op->length);
          if (crc != frame.footer_data_crc) {
            fprintf(stderr,
                    "[rados-cluster] data CRC mismatch on %s "
                    "(wire %08x, landed %08x)\n",
                    oid.c_str(), frame.footer_data_crc, crc);
            return kEio;
          }
        } else if (crc32c_sw(0, frame.data.data(), frame.data.size()) !=
                   frame.footer_data_crc) {
          return kEio;
        }
        return 0;
      }
      case kOsdOpZero:
      case kOsdOpTruncate: {
        ObjMeta* meta = ensure_object_locked(key);
        if (meta == nullptr) return kEnospc;
        const uint64_t slot = meta->slot;
        uint64_t zero_len = op->length;
        if (op->op == kOsdOpTruncate) {
          meta->length = op->offset;
          zero_len = 0;
        } else {
          meta->length = std::max<uint64_t>(meta->length,
                                            op->offset + op->length);
        }
        if (zero_len > 0) {
          // Zero-fill through the same RMW path (byte-granular).
          std::vector<uint8_t> zeros(zero_len, 0);
          if (arena_write_any(channel, bounce,
                              slot * object_bytes_ + op->offset,
                              zeros.data(), zero_len) != kIoOk) {
            return kEio;
          }
        }
        return 0;
      }
      case kOsdOpCreate: {
        if (ensure_object_locked(key) == nullptr) return kEnospc;
        return 0;
      }
      case kOsdOpDelete: {
        if (it == objects_.end()) return kEnoent;
        slot_used_[it->second.slot] = false;
        objects_.erase(it);
        return 0;
      }
      default:
        return kEinval;
    }
  }

  // mutex_ held. Returns null when the arena is full.
  ObjMeta* ensure_object_locked(const std::pair<uint64_t, std::string>& key) {
    auto it = objects_.find(key);
    if (it != objects_.end()) return &it->second;
    for (uint64_t s = 0; s < slot_count_; ++s) {
      if (slot_used_[s]) continue;
      slot_used_[s] = true;
      auto [pos, ok] = objects_.emplace(key, ObjMeta{s, 0});
      (void)ok;
      return &pos->second;
    }
    return nullptr;
  }

  uint64_t object_bytes_;
  uint64_t slot_count_ = 0;
  bool hbm_ = false;
  BdevPtr arena_;
  int listen_fd_ = -1;
  uint16_t port_ = 0;
  std::atomic<bool> stopping_{false};
  std::thread accept_thread_;
  mutable std::mutex mutex_;
  std::map<std::pair<uint64_t, std::string>, ObjMeta> objects_;
  std::vector<bool> slot_used_;
  std::vector<std::thread> conn_threads_;
  std::vector<int> conn_fds_;
};

}  // namespace

std::shared_ptr<RadosCluster> start_rados_cluster(
    uint16_t port, uint64_t arena_mb, bool use_hbm, int device,
    uint64_t object_bytes) {
  auto cluster = std::make_shared<RadosClusterImpl>(
      port, arena_mb, use_hbm, device, object_bytes);
  cluster->start();
  return cluster;
}

}  // namespace hipstore
