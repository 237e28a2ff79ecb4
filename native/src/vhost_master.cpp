// Native vhost-user master benchmark: plays QEMU's role against
// hipstored's vhost-user targets at native speed. The Python master
// (oim_amd/bench/vhost_client.py) is the conformance tool; this is
// the throughput tool — the Python synthetic guest tops out around
// 200k IOPS while the daemon's virtqueue workers can go far higher,
// so measuring the HOST-ATTACH data path at its real ceiling needs a
// master that doesn't serialize on an interpreter.
//
// Guest memory is a memfd shared via SET_MEM_TABLE; each request ring
// gets its own slab (descriptor table + split avail/used rings +
// per-tag request/data buffers), its own kick/call eventfds, and its
// own submitter thread keeping `iodepth` 3-descriptor chains
// outstanding — the same shape as the Python harness, minus the GIL.

#include <poll.h>
#include <pthread.h>
#include <sched.h>
#include <sys/eventfd.h>
#include <sys/mman.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>

#include <atomic>
#include <chrono>
#include <cstring>
#include <random>
#include <stdexcept>
#include <string>
#include <thread>
#include <vector>

#include "hipstore/engine.h"  // PerfResult
#include "hipstore/vhost_master.h"

namespace hipstore {

namespace {

// vhost-user message ids (matching native/src/vhost.cpp's slave)
enum : uint32_t {
  kGetFeatures = 1,
  kSetFeatures = 2,
  kSetOwner = 3,
  kSetMemTable = 5,
  kSetVringNum = 8,
  kSetVringAddr = 9,
  kSetVringBase = 10,
  kGetVringBase = 11,
  kSetVringKick = 12,
  kSetVringCall = 13,
  kGetProtocolFeatures = 15,
  kSetProtocolFeatures = 16,
  kSetVringEnable = 18,
};

constexpr uint64_t kFeatVersion1 = 1ull << 32;
constexpr uint64_t kFeatProtocol = 1ull << 30;

constexpr uint16_t kDescNext = 1;
constexpr uint16_t kDescWrite = 2;

constexpr uint64_t kGpaBase = 0x10000000ull;
constexpr uint64_t kUaddrBase = 0x7F0000000000ull;

#pragma pack(push, 1)
struct VringDesc {
  uint64_t addr;
  uint32_t len;
  uint16_t flags;
  uint16_t next;
};
#pragma pack(pop)

struct Msg {
  uint32_t request;
  uint32_t flags;
  uint32_t size;
};

void send_msg(int fd, uint32_t request, const void* payload, uint32_t size,
              int pass_fd = -1) {
  Msg header{request, 0x1, size};
  iovec iov[2] = {{&header, sizeof(header)},
                  {const_cast<void*>(payload), size}};
  msghdr msg{};
  msg.msg_iov = iov;
  msg.msg_iovlen = size ? 2 : 1;
  char cbuf[CMSG_SPACE(sizeof(int))] = {0};
  if (pass_fd >= 0) {
    msg.msg_control = cbuf;
    msg.msg_controllen = sizeof(cbuf);
    cmsghdr* cm = CMSG_FIRSTHDR(&msg);
    cm->cmsg_level = SOL_SOCKET;
    cm->cmsg_type = SCM_RIGHTS;
    cm->cmsg_len = CMSG_LEN(sizeof(int));
    memcpy(CMSG_DATA(cm), &pass_fd, sizeof(int));
  }
  if (sendmsg(fd, &msg, 0) < 0) {
    throw std::runtime_error("vhost master: sendmsg failed");
  }
}

std::vector<uint8_t> recv_reply(int fd) {
  Msg header;
  uint8_t* p = reinterpret_cast<uint8_t*>(&header);
  size_t need = sizeof(header);
  while (need) {
    ssize_t r = read(fd, p, need);
    if (r <= 0) throw std::runtime_error("vhost master: short reply");
    p += r;
    need -= static_cast<size_t>(r);
  }
  std::vector<uint8_t> payload(header.size);
  size_t off = 0;
  while (off < payload.size()) {
    ssize_t r = read(fd, payload.data() + off, payload.size() - off);
    if (r <= 0) throw std::runtime_error("vhost master: short payload");
    off += static_cast<size_t>(r);
  }
  return payload;
}

// Latency histogram, 1 us buckets (same shape as the engine's).
struct Hist {
  std::vector<uint32_t> buckets = std::vector<uint32_t>(65536, 0);
  uint64_t count = 0;
  double sum = 0;
  uint32_t max = 0;
  void record(uint32_t us) {
    ++buckets[std::min<uint32_t>(us, 65535)];
    ++count;
    sum += us;
    if (us > max) max = us;
  }
  double pct(double p) const {
    if (!count) return 0;
    uint64_t target = static_cast<uint64_t>(p * (count - 1)), seen = 0;
    for (uint32_t i = 0; i < 65536; ++i) {
      seen += buckets[i];
      if (seen > target) return i;
    }
    return max;
  }
};

}  // namespace

VhostMasterSession::VhostMasterSession(const std::string& socket_path,
                                       const std::string& personality,
                                       int num_rings, int iodepth,
                                       uint32_t io_size,
                                       uint32_t block_size,
                                       uint64_t capacity_bytes)
    : personality_(personality),
      num_rings_(num_rings),
      iodepth_(iodepth),
      io_size_(io_size),
      block_size_(block_size),
      capacity_bytes_(capacity_bytes) {
  const bool blk = personality == "blk";
  const int first_queue = blk ? 0 : 2;
  const int max_rings = blk ? 8 : 6;
  if (num_rings < 1 || num_rings > max_rings) {
    throw std::runtime_error("vhost master: bad ring count");
  }
  if (iodepth < 1 || iodepth * 3 + 1 > 4096) {
    throw std::runtime_error("vhost master: bad iodepth");
  }
  uint32_t qsize = 16;
  while (qsize < static_cast<uint32_t>(iodepth) * 3 + 1) qsize *= 2;

  // Region layout: per-ring slab of ring header + request slots +
  // data buffers, mirroring oim_amd/bench/vhost_harness.py.
  const uint64_t header_raw = 16ull * qsize + 4 + 2 * qsize;
  const uint64_t used_off_rel = (header_raw + 0xFFF) & ~0xFFFull;
  const uint64_t header_bytes = used_off_rel + 4 + 8ull * qsize;
  uint64_t slab = ((header_bytes + 0xFFF) & ~0xFFFull) +
                  0x100ull * iodepth +
                  static_cast<uint64_t>(io_size) * iodepth;
  slab = (slab + 0xFFFF) & ~0xFFFFull;
  const uint64_t mem_size =
      std::max<uint64_t>(0x10000 + slab * num_rings, 8 << 20);

  int sock = socket(AF_UNIX, SOCK_STREAM, 0);
  if (sock < 0) throw std::runtime_error("vhost master: socket failed");
  sockaddr_un sa{};
  sa.sun_family = AF_UNIX;
  snprintf(sa.sun_path, sizeof(sa.sun_path), "%s", socket_path.c_str());
  if (connect(sock, reinterpret_cast<sockaddr*>(&sa), sizeof(sa)) < 0) {
    close(sock);
    throw std::runtime_error("vhost master: connect " + socket_path);
  }
  int memfd = memfd_create("vhost-native-guest", 0);
  if (memfd < 0 || ftruncate(memfd, mem_size) != 0) {
    close(sock);
    throw std::runtime_error("vhost master: memfd failed");
  }
  uint8_t* mem = static_cast<uint8_t*>(
      mmap(nullptr, mem_size, PROT_READ | PROT_WRITE, MAP_SHARED, memfd, 0));
  if (mem == MAP_FAILED) {
    close(memfd);
    close(sock);
    throw std::runtime_error("vhost master: mmap failed");
  }

  std::vector<VhostMasterRing> rings(num_rings);
  try {
    // Handshake (GET/SET features, owner, memory table).
    send_msg(sock, kGetFeatures, nullptr, 0);
    auto feat_payload = recv_reply(sock);
    uint64_t features = 0;
    memcpy(&features, feat_payload.data(), 8);
    if (!(features & kFeatVersion1)) {
      throw std::runtime_error("vhost master: slave lacks VERSION_1");
    }
    const uint64_t want = kFeatVersion1 | kFeatProtocol;
    send_msg(sock, kSetFeatures, &want, 8);
    send_msg(sock, kGetProtocolFeatures, nullptr, 0);
    auto proto_payload = recv_reply(sock);
    uint64_t proto = 0;
    memcpy(&proto, proto_payload.data(), 8);
    proto &= 0x1;
    send_msg(sock, kSetProtocolFeatures, &proto, 8);
    send_msg(sock, kSetOwner, nullptr, 0);
    struct {
      uint32_t nregions;
      uint32_t pad;
      uint64_t gpa, size, uaddr, mmap_off;
    } table{1, 0, kGpaBase, mem_size, kUaddrBase, 0};
    send_msg(sock, kSetMemTable, &table, sizeof(table), memfd);

    // Ring setup.
    for (int i = 0; i < num_rings; ++i) {
      VhostMasterRing& ring = rings[i];
      ring.mem = mem;
      ring.qsize = qsize;
      ring.base = 0x10000 + slab * i;
      ring.desc_off = ring.base;
      ring.avail_off = ring.base + 16ull * qsize;
      ring.used_off = ring.base + used_off_rel;
      ring.req_off = ring.base + ((header_bytes + 0xFFF) & ~0xFFFull);
      ring.data_off = ring.req_off + 0x100ull * iodepth;
      ring.kick = eventfd(0, 0);
      ring.call = eventfd(0, EFD_NONBLOCK);
      const uint32_t queue = first_queue + i;
      struct {
        uint32_t q, v;
      } u32x2{queue, qsize};
      send_msg(sock, kSetVringNum, &u32x2, 8);
      u32x2 = {queue, 0};
      send_msg(sock, kSetVringBase, &u32x2, 8);
      struct {
        uint32_t index, flags;
        uint64_t desc, used, avail, log;
      } addr{queue, 0, kUaddrBase + ring.desc_off,
             kUaddrBase + ring.used_off, kUaddrBase + ring.avail_off, 0};
      send_msg(sock, kSetVringAddr, &addr, sizeof(addr));
      uint64_t qword = queue;
      send_msg(sock, kSetVringCall, &qword, 8, ring.call);
      send_msg(sock, kSetVringKick, &qword, 8, ring.kick);
      u32x2 = {queue, 1};
      send_msg(sock, kSetVringEnable, &u32x2, 8);
      // We poll used->idx; suppress the slave's call eventfds
      // (VRING_AVAIL_F_NO_INTERRUPT, one syscall per completion saved
      // on each side).
      __atomic_store_n(reinterpret_cast<uint16_t*>(mem + ring.avail_off),
                       uint16_t{1}, __ATOMIC_RELEASE);
    }

    // Rings configured; session stays up across run() calls.
    (void)0;
    sock_ = sock;
    memfd_ = memfd;
    mem_ = mem;
    mem_size_ = mem_size;
    rings_ = std::move(rings);
  } catch (...) {
    for (auto& ring : rings) {
      if (ring.kick >= 0) close(ring.kick);
      if (ring.call >= 0) close(ring.call);
    }
    munmap(mem, mem_size);
    close(memfd);
    close(sock);
    throw;
  }
}

VhostMasterSession::~VhostMasterSession() {
  for (auto& ring : rings_) {
    if (ring.kick >= 0) close(ring.kick);
    if (ring.call >= 0) close(ring.call);
  }
  if (mem_ != nullptr) munmap(mem_, mem_size_);
  if (memfd_ >= 0) close(memfd_);
  if (sock_ >= 0) close(sock_);
}

PerfResult VhostMasterSession::run(uint64_t total_ios,
                                   const std::string& workload) {
  const bool blk = personality_ == "blk";
  const int num_rings = num_rings_;
  const int iodepth = iodepth_;
  const uint32_t io_size = io_size_;
  const uint32_t block_size = block_size_;
  const uint64_t capacity_bytes = capacity_bytes_;
  uint8_t* mem = mem_;
  std::vector<VhostMasterRing>& rings = rings_;
  ++run_seq_;
  PerfResult result;
  {
    // Drive.
    const bool do_write_only = workload == "randwrite";
    const bool do_mix = workload == "randrw";
    const uint64_t units = capacity_bytes / io_size;
    if (units == 0) throw std::runtime_error("vhost master: tiny capacity");
    const uint64_t per_ring =
        (total_ios + num_rings - 1) / num_rings;
    std::vector<Hist> hists(num_rings);
    std::atomic<bool> failed{false};
    std::vector<std::thread> threads;
    const auto t0 = std::chrono::steady_clock::now();
    for (int r = 0; r < num_rings; ++r) {
      threads.emplace_back([&, r] {
        using clock = std::chrono::steady_clock;
        // Busy-polling master threads migrate under CFS like the
        // slave's ring workers; same optional pinning cure
        // (HIPSTORE_MASTER_AFFINITY_BASE=<core>).
        if (const char* env = getenv("HIPSTORE_MASTER_AFFINITY_BASE")) {
          const long ncpu = sysconf(_SC_NPROCESSORS_ONLN);
          if (ncpu > 0) {
            cpu_set_t set;
            CPU_ZERO(&set);
            CPU_SET((atoi(env) + r) % static_cast<int>(ncpu), &set);
            (void)pthread_setaffinity_np(pthread_self(), sizeof(set),
                                         &set);
          }
        }
        VhostMasterRing& ring = rings[r];
        Hist& hist = hists[r];
        std::mt19937_64 rng(0x5EEDull + r + run_seq_ * 1315423911ull);
        std::vector<clock::time_point> t_submit(iodepth);
        auto* descs = reinterpret_cast<VringDesc*>(mem + ring.desc_off);
        auto* avail_idx_p =
            reinterpret_cast<volatile uint16_t*>(mem + ring.avail_off + 2);
        auto* avail_ring =
            reinterpret_cast<uint16_t*>(mem + ring.avail_off + 4);
        auto* used_idx_p =
            reinterpret_cast<volatile uint16_t*>(mem + ring.used_off + 2);
        auto* used_ring =
            reinterpret_cast<uint32_t*>(mem + ring.used_off + 4);

        auto chain = [&](int tag) {
          const uint64_t lba_unit = rng() % units;
          const bool is_write = do_write_only || (do_mix && (rng() & 1));
          const uint64_t req_gpa = kGpaBase + ring.req_off + 0x100ull * tag;
          const uint64_t resp_gpa = req_gpa + 0x80;
          const uint64_t data_gpa =
              kGpaBase + ring.data_off + static_cast<uint64_t>(io_size) * tag;
          uint8_t* req = mem + ring.req_off + 0x100ull * tag;
          const int base = tag * 3;
          if (blk) {
            const uint64_t sector = lba_unit * io_size / 512;
            uint32_t type = is_write ? 1 : 0;
            memcpy(req, &type, 4);
            memset(req + 4, 0, 4);
            memcpy(req + 8, &sector, 8);
            descs[base] = {req_gpa, 16, kDescNext,
                           static_cast<uint16_t>(base + 1)};
            descs[base + 1] = {data_gpa, io_size,
                               static_cast<uint16_t>(
                                   (is_write ? 0 : kDescWrite) | kDescNext),
                               static_cast<uint16_t>(base + 2)};
            descs[base + 2] = {resp_gpa, 1, kDescWrite, 0};
          } else {
            const uint32_t lba32 =
                static_cast<uint32_t>(lba_unit * (io_size / block_size));
            const uint16_t blocks =
                static_cast<uint16_t>(io_size / block_size);
            memset(req, 0, 51);
            req[0] = 1;   // lun addressing: bus 1
            req[1] = 0;   // target 0
            req[2] = 0x40;
            uint64_t id = tag + 1;
            memcpy(req + 8, &id, 8);
            uint8_t* cdb = req + 19;
            cdb[0] = is_write ? 0x2A : 0x28;
            cdb[2] = static_cast<uint8_t>(lba32 >> 24);
            cdb[3] = static_cast<uint8_t>(lba32 >> 16);
            cdb[4] = static_cast<uint8_t>(lba32 >> 8);
            cdb[5] = static_cast<uint8_t>(lba32);
            cdb[7] = static_cast<uint8_t>(blocks >> 8);
            cdb[8] = static_cast<uint8_t>(blocks);
            descs[base] = {req_gpa, 51, kDescNext,
                           static_cast<uint16_t>(base + 1)};
            if (is_write) {
              descs[base + 1] = {data_gpa, io_size, kDescNext,
                                 static_cast<uint16_t>(base + 2)};
              descs[base + 2] = {resp_gpa, 108, kDescWrite, 0};
            } else {
              descs[base + 1] = {resp_gpa, 108,
                                 kDescWrite | kDescNext,
                                 static_cast<uint16_t>(base + 2)};
              descs[base + 2] = {data_gpa, io_size, kDescWrite, 0};
            }
          }
          t_submit[tag] = clock::now();
          avail_ring[ring.avail_idx % ring.qsize] =
              static_cast<uint16_t>(base);
          __atomic_store_n(avail_idx_p,
                           static_cast<uint16_t>(++ring.avail_idx),
                           __ATOMIC_SEQ_CST);
          // VRING_USED_F_NO_NOTIFY: a hot slave suppresses kicks; the
          // seq_cst publish above orders idx before this flag read
          // (virtio 2.6.10 missed-kick protocol).
          if (!(__atomic_load_n(
                    reinterpret_cast<const uint16_t*>(mem + ring.used_off),
                    __ATOMIC_ACQUIRE) &
                1)) {
            uint64_t one = 1;
            (void)!write(ring.kick, &one, 8);
          }
        };

        const uint64_t initial =
            std::min<uint64_t>(iodepth, per_ring);
        for (uint64_t t = 0; t < initial; ++t) chain(static_cast<int>(t));
        uint64_t submitted = initial;
        uint64_t inflight = initial;
        const auto hard_deadline = clock::now() + std::chrono::seconds(120);
        uint32_t empty_spins = 0;
        while (inflight) {
          if (__atomic_load_n(used_idx_p, __ATOMIC_ACQUIRE) ==
              ring.used_idx) {
            // We advertised NO_INTERRUPT, so busy-poll the used index
            // (pause-hygiene spin; deadline check amortized).
            __builtin_ia32_pause();
            if ((++empty_spins & 0xFFFF) == 0 &&
                clock::now() > hard_deadline) {
              failed.store(true);
              break;
            }
            continue;
          }
          empty_spins = 0;
          const uint32_t head =
              used_ring[2 * (ring.used_idx % ring.qsize)];
          ++ring.used_idx;
          const int tag = static_cast<int>(head / 3);
          const auto now = clock::now();
          hist.record(static_cast<uint32_t>(
              std::chrono::duration_cast<std::chrono::microseconds>(
                  now - t_submit[tag]).count()));
          // Completion status: blk = trailing status byte;
          // scsi = virtio_scsi_cmd_resp.status at offset 10
          // (sense_len u32, resid u32, qualifier u16, status, response).
          const uint8_t* resp = mem + ring.req_off + 0x100ull * tag + 0x80;
          if ((blk ? resp[0] : resp[10]) != 0) failed.store(true);
          --inflight;
          if (submitted < per_ring) {
            chain(tag);
            ++submitted;
            ++inflight;
          }
        }
      });
    }
    for (auto& t : threads) t.join();
    const double elapsed =
        std::chrono::duration<double>(std::chrono::steady_clock::now() - t0)
            .count();
    if (failed.load()) {
      throw std::runtime_error(
          "vhost master: I/O failure or completion timeout");
    }
    Hist total;
    for (auto& h : hists) {
      for (uint32_t i = 0; i < 65536; ++i) total.buckets[i] += h.buckets[i];
      total.count += h.count;
      total.sum += h.sum;
      total.max = std::max(total.max, h.max);
    }
    result.seconds = elapsed;
    result.io_count = total.count;
    result.iops = total.count / elapsed;
    result.throughput_mbps =
        total.count * static_cast<double>(io_size) / elapsed / 1e6;
    if (total.count) {
      result.lat_avg_us = total.sum / total.count;
      result.lat_p50_us = total.pct(0.50);
      result.lat_p90_us = total.pct(0.90);
      result.lat_p99_us = total.pct(0.99);
      result.lat_p999_us = total.pct(0.999);
      result.lat_max_us = total.max;
    }
  }
  return result;
}

PerfResult vhost_master_bench(const std::string& socket_path,
                              const std::string& personality,
                              int num_rings, int iodepth, uint32_t io_size,
                              const std::string& workload,
                              uint64_t total_ios, uint32_t block_size,
                              uint64_t capacity_bytes) {
  VhostMasterSession session(socket_path, personality, num_rings, iodepth,
                             io_size, block_size, capacity_bytes);
  return session.run(total_ios, workload);
}

}  // namespace hipstore
