// NVMe-oF/TCP initiator bdev.
//
// One TCP connection per I/O channel (= one NVMe queue pair, NVMe/TCP
// maps queues to connections), polled from Bdev::poll like every other
// hipstored queue — no reactor threads (the SPDK model this replaces
// ran dedicated pollers; here the engine's existing poll discipline
// carries the transport). Digests (HDGST/DDGST, CRC32C) are negotiated
// and verified per PDU.

#include <arpa/inet.h>
#include <fcntl.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <poll.h>
#include <sys/socket.h>
#include <unistd.h>

#include <atomic>
#include <cstring>
#include <map>
#include <stdexcept>
#include <vector>

#include "hipstore/engine.h"
#include "hipstore/nvmf.h"
#include "nvmf_common.h"

namespace hipstore {

namespace {

using namespace nvmf;

constexpr const char* kHostNqn = "nqn.2014-08.org.nvmexpress:uuid:"
                                 "8f6a52f2-0000-4000-8000-oimamdinitiat";

// Setup-path wait bound: a target that accepts and then goes silent
// during ICReq/Fabrics/Identify (or a blackholed address) must fail
// create_nvmf_tcp_bdev, not hang the daemon. The I/O path switches
// the fd to non-blocking polled mode afterwards, where these
// timeouts are inert.
int setup_timeout_s() {
  const char* env = getenv("HIPSTORE_NVMF_SETUP_TIMEOUT");
  const int v = env != nullptr ? atoi(env) : 0;
  return v > 0 ? v : 10;
}

int tcp_connect(const std::string& addr, uint16_t port) {
  int fd = socket(AF_INET, SOCK_STREAM | SOCK_CLOEXEC, 0);
  if (fd < 0) throw std::runtime_error("nvmf: socket failed");
  sockaddr_in sa{};
  sa.sin_family = AF_INET;
  sa.sin_port = htons(port);
  sa.sin_addr.s_addr = addr.empty() ? htonl(INADDR_LOOPBACK)
                                    : inet_addr(addr.c_str());
  const int flags = fcntl(fd, F_GETFL, 0);
  fcntl(fd, F_SETFL, flags | O_NONBLOCK);
  int rc = connect(fd, reinterpret_cast<sockaddr*>(&sa), sizeof(sa));
  if (rc < 0 && errno == EINPROGRESS) {
    pollfd pfd{fd, POLLOUT, 0};
    rc = ::poll(&pfd, 1, setup_timeout_s() * 1000);
    int soerr = 0;
    socklen_t slen = sizeof(soerr);
    if (rc == 1) getsockopt(fd, SOL_SOCKET, SO_ERROR, &soerr, &slen);
    rc = (rc == 1 && soerr == 0) ? 0 : -1;
  }
  if (rc < 0) {
    close(fd);
    throw std::runtime_error("nvmf: connect to " + addr + " failed");
  }
  fcntl(fd, F_SETFL, flags);
  timeval tv{setup_timeout_s(), 0};
  setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
  setsockopt(fd, SOL_SOCKET, SO_SNDTIMEO, &tv, sizeof(tv));
  int one = 1;
  setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
  return fd;
}

// A connected NVMe/TCP queue: handshake + blocking admin round trips.
// I/O channels reuse the fd afterwards in non-blocking polled mode.
struct QueuePair {
  int fd = -1;
  bool hdgst = false;
  bool ddgst = false;
  uint32_t maxh2cdata = kMaxDataPerPdu;
  uint16_t next_cid = 1;

  ~QueuePair() {
    if (fd >= 0) close(fd);
  }

  void handshake(const std::string& addr, uint16_t port, bool want_digests) {
    fd = tcp_connect(addr, port);
    IcReq icreq{};
    icreq.ch = {kIcReq, 0, sizeof(IcReq), 0, sizeof(IcReq)};
    icreq.pfv = 0;
    icreq.hpda = 0;
    icreq.dgst = want_digests ? 0x3 : 0x0;
    icreq.maxr2t = 15;
    if (!write_exact(fd, &icreq, sizeof(icreq))) {
      throw std::runtime_error("nvmf: ICReq send failed");
    }
    IcResp icresp{};
    if (!read_exact(fd, &icresp, sizeof(icresp)) ||
        icresp.ch.type != kIcResp) {
      throw std::runtime_error("nvmf: ICResp receive failed");
    }
    hdgst = icresp.dgst & 0x1;
    ddgst = icresp.dgst & 0x2;
    if (icresp.maxh2cdata) maxh2cdata = icresp.maxh2cdata;
  }

  bool send_capsule(const Sqe& sqe, const void* data, uint32_t dlen) {
    struct {
      CommonHeader ch;
      Sqe sqe;
    } __attribute__((packed)) capsule{};
    const uint8_t pdo =
        dlen ? static_cast<uint8_t>(72 + (hdgst ? 4 : 0)) : 0;
    capsule.ch = {kCapsuleCmd,
                  static_cast<uint8_t>((hdgst ? kFlagHdgst : 0) |
                                       (dlen && ddgst ? kFlagDdgst : 0)),
                  72, pdo,
                  static_cast<uint32_t>(72 + (hdgst ? 4 : 0) + dlen +
                                        (dlen && ddgst ? 4 : 0))};
    capsule.sqe = sqe;
    uint32_t dd = (dlen && ddgst) ? crc32c_sw(0, data, dlen) : 0;
    return send_pdu(fd, &capsule, 72, hdgst, data, dlen, dd, ddgst);
  }

  // Blocking command round trip (admin queue / setup only). Response
  // data (C2HData) is appended to *read_data when non-null.
  Cqe command(Sqe sqe, std::vector<uint8_t>* read_data,
              const void* write_data = nullptr, uint32_t write_len = 0) {
    const uint16_t cid = next_cid++;
    sqe.set_cid(cid);
    if (!send_capsule(sqe, write_data, write_len)) {
      throw std::runtime_error("nvmf: capsule send failed");
    }
    while (true) {
      CommonHeader ch;
      if (!read_exact(fd, &ch, sizeof(ch))) {
        throw std::runtime_error("nvmf: connection closed");
      }
      std::vector<uint8_t> header(ch.hlen);
      memcpy(header.data(), &ch, sizeof(ch));
      if (!read_exact(fd, header.data() + sizeof(ch),
                      ch.hlen - sizeof(ch))) {
        throw std::runtime_error("nvmf: short header");
      }
      if (hdgst) {
        uint32_t hd;
        if (!read_exact(fd, &hd, 4) ||
            hd != crc32c_sw(0, header.data(), ch.hlen)) {
          throw std::runtime_error("nvmf: header digest mismatch");
        }
      }
      const uint32_t pdo = ch.pdo ? ch.pdo : ch.hlen + (hdgst ? 4 : 0);
      uint32_t dlen = 0;
      if (ch.plen > pdo) {
        dlen = ch.plen - pdo - (ddgst ? 4 : 0);
        uint32_t pad = pdo - ch.hlen - (hdgst ? 4 : 0);
        uint8_t padbuf[16];
        if (pad > sizeof(padbuf) || (pad && !read_exact(fd, padbuf, pad))) {
          throw std::runtime_error("nvmf: bad pad");
        }
      }
      std::vector<uint8_t> data(dlen);
      if (dlen && !read_exact(fd, data.data(), dlen)) {
        throw std::runtime_error("nvmf: short data");
      }
      if (dlen && ddgst) {
        uint32_t dd;
        if (!read_exact(fd, &dd, 4) ||
            dd != crc32c_sw(0, data.data(), dlen)) {
          throw std::runtime_error("nvmf: data digest mismatch");
        }
      }
      if (ch.type == kC2HData) {
        if (read_data) {
          read_data->insert(read_data->end(), data.begin(), data.end());
        }
        continue;
      }
      if (ch.type == kCapsuleResp) {
        Cqe cqe;
        memcpy(cqe.bytes, header.data() + 8, 16);
        if (cqe.cid() != cid) {
          throw std::runtime_error("nvmf: unexpected cid");
        }
        return cqe;
      }
      throw std::runtime_error("nvmf: unexpected PDU during setup");
    }
  }

  void fabrics_connect(const std::string& subnqn, uint16_t qid,
                       uint16_t sqsize) {
    Sqe sqe{};
    sqe.set_opc(kOpcFabrics);
    sqe.set_fctype(kFctypeConnect);
    sqe.set_sgl_transport(kConnectDataSize);
    sqe.set_cdw(10, 0);                       // recfmt
    sqe.set_cdw(10, 0);
    uint32_t cdw10 = 0;                       // recfmt 0
    sqe.set_cdw(10, cdw10);
    uint32_t cdw11 = qid | (static_cast<uint32_t>(sqsize) << 16);
    // Connect layout: recfmt(2) qid(2) @40, sqsize(2) cattr(1) @44.
    uint16_t recfmt = 0;
    memcpy(sqe.bytes + 40, &recfmt, 2);
    memcpy(sqe.bytes + 42, &qid, 2);
    memcpy(sqe.bytes + 44, &sqsize, 2);
    (void)cdw11;
    std::vector<uint8_t> data(kConnectDataSize, 0);
    // hostid[16] @0, cntlid @16 (0xFFFF = any on admin connect)
    uint16_t cntlid = qid == 0 ? 0xFFFF : 1;
    memcpy(data.data() + 16, &cntlid, 2);
    snprintf(reinterpret_cast<char*>(data.data() + 256), 256, "%s",
             subnqn.c_str());
    snprintf(reinterpret_cast<char*>(data.data() + 512), 256, "%s", kHostNqn);
    Cqe cqe = command(sqe, nullptr, data.data(), data.size());
    if (cqe.status_code() != kScSuccess) {
      throw std::runtime_error("nvmf: fabrics connect failed");
    }
  }
};

struct InflightIo {
  IoRequest req;
  uint64_t remaining_read = 0;
};

class NvmfChannel : public IoChannel {
 public:
  std::unique_ptr<QueuePair> qp;
  std::map<uint16_t, InflightIo> inflight;
  std::vector<std::pair<IoCompletion, int>> immediate;  // failed at submit
  std::string rxbuf;

  ~NvmfChannel() override = default;

  void set_nonblocking() {
    int flags = fcntl(qp->fd, F_GETFL, 0);
    fcntl(qp->fd, F_SETFL, flags | O_NONBLOCK);
  }
};

class NvmfBdev : public Bdev {
 public:
  NvmfBdev(const std::string& name, std::string traddr, uint16_t trsvcid,
           std::string subnqn, uint32_t nsid, bool digests,
           uint64_t block_size, uint64_t num_blocks)
      : Bdev(name, "NVMe-oF TCP disk", block_size, num_blocks),
        traddr_(std::move(traddr)),
        trsvcid_(trsvcid),
        subnqn_(std::move(subnqn)),
        nsid_(nsid),
        digests_(digests) {}

  std::shared_ptr<IoChannel> get_channel() override {
    auto channel = std::make_shared<NvmfChannel>();
    channel->qp = std::make_unique<QueuePair>();
    channel->qp->handshake(traddr_, trsvcid_, digests_);
    const uint16_t qid = next_qid_.fetch_add(1);
    channel->qp->fabrics_connect(subnqn_, qid, 127);
    channel->set_nonblocking();
    return channel;
  }

  void submit(IoChannel* ch, IoRequest req) override {
    auto* channel = static_cast<NvmfChannel*>(ch);
    QueuePair& qp = *channel->qp;
    if (req.op != IoOp::kFlush) {
      if (!check_bounds(req) ||
          (req.op == IoOp::kFill && req.fill != 0) ||
          req.length / block_size() > 0x10000) {
        channel->immediate.emplace_back(std::move(req.on_complete), kIoInvalid);
        return;
      }
    }
    account(req);
    Sqe sqe{};
    sqe.set_nsid(nsid_);
    const uint64_t slba = req.offset / block_size();
    const uint32_t nlb = req.length ? req.length / block_size() : 1;
    switch (req.op) {
      case IoOp::kRead: sqe.set_opc(kOpcRead); break;
      case IoOp::kWrite: sqe.set_opc(kOpcWrite); break;
      case IoOp::kFill: sqe.set_opc(kOpcWriteZeroes); break;
      case IoOp::kFlush: sqe.set_opc(kOpcFlush); break;
    }
    if (req.op == IoOp::kRead || req.op == IoOp::kWrite) {
      sqe.set_sgl_transport(req.length);
    }
    if (req.op != IoOp::kFlush) {
      sqe.set_cdw(10, static_cast<uint32_t>(slba));
      sqe.set_cdw(11, static_cast<uint32_t>(slba >> 32));
      sqe.set_cdw(12, nlb - 1);
    }
    const uint16_t cid = qp.next_cid++;
    sqe.set_cid(cid);
    InflightIo io;
    io.remaining_read = req.op == IoOp::kRead ? req.length : 0;
    io.req = std::move(req);
    channel->inflight.emplace(cid, std::move(io));
    if (!qp.send_capsule(sqe, nullptr, 0)) {
      auto it = channel->inflight.find(cid);
      channel->immediate.emplace_back(std::move(it->second.req.on_complete),
                                      kIoFailed);
      channel->inflight.erase(it);
    }
  }

  int poll(IoChannel* ch) override {
    auto* channel = static_cast<NvmfChannel*>(ch);
    int completed = 0;
    if (!channel->immediate.empty()) {
      // Swapped out first: callbacks may resubmit and append again.
      auto batch = std::move(channel->immediate);
      channel->immediate.clear();
      for (auto& [cb, status] : batch) {
        if (cb) cb(status);
        ++completed;
      }
    }
    // Drain the socket.
    char chunk[65536];
    while (true) {
      ssize_t r = recv(channel->qp->fd, chunk, sizeof(chunk), MSG_DONTWAIT);
      if (r > 0) {
        channel->rxbuf.append(chunk, r);
        continue;
      }
      if (r < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) break;
      if (r == 0 || (r < 0 && errno != EINTR)) {
        // connection failure: fail everything in flight
        for (auto& [cid, io] : channel->inflight) {
          if (io.req.on_complete) io.req.on_complete(kIoFailed);
          ++completed;
        }
        channel->inflight.clear();
        return completed;
      }
    }
    // Parse complete PDUs.
    size_t start = 0;
    const std::string& buf = channel->rxbuf;
    while (buf.size() - start >= sizeof(CommonHeader)) {
      CommonHeader ch_hdr;
      memcpy(&ch_hdr, buf.data() + start, sizeof(ch_hdr));
      if (ch_hdr.plen < ch_hdr.hlen || ch_hdr.plen > (64 << 20)) {
        // protocol violation: fail all
        for (auto& [cid, io] : channel->inflight) {
          if (io.req.on_complete) io.req.on_complete(kIoFailed);
          ++completed;
        }
        channel->inflight.clear();
        channel->rxbuf.clear();
        return completed;
      }
      if (buf.size() - start < ch_hdr.plen) break;
      completed += handle_pdu(channel,
                              reinterpret_cast<const uint8_t*>(buf.data()) +
                                  start,
                              ch_hdr);
      start += ch_hdr.plen;
    }
    channel->rxbuf.erase(0, start);
    return completed;
  }

 private:
  int handle_pdu(NvmfChannel* channel, const uint8_t* pdu,
                 const CommonHeader& ch) {
    QueuePair& qp = *channel->qp;
    if (qp.hdgst) {
      uint32_t hd;
      memcpy(&hd, pdu + ch.hlen, 4);
      if (hd != crc32c_sw(0, pdu, ch.hlen)) {
        return fail_cid_from_pdu(channel, pdu, ch);
      }
    }
    const uint32_t pdo = ch.pdo ? ch.pdo : ch.hlen + (qp.hdgst ? 4 : 0);
    uint32_t dlen = 0;
    if (ch.plen > pdo) dlen = ch.plen - pdo - (qp.ddgst ? 4 : 0);

    if (ch.type == kC2HData) {
      DataHeader dh;
      memcpy(&dh, pdu, sizeof(dh));
      auto it = channel->inflight.find(dh.cccid);
      if (it == channel->inflight.end()) return 0;
      InflightIo& io = it->second;
      if (qp.ddgst && dlen) {
        uint32_t dd;
        memcpy(&dd, pdu + pdo + dlen, 4);
        if (dd != crc32c_sw(0, pdu + pdo, dlen)) {
          if (io.req.on_complete) io.req.on_complete(kIoFailed);
          channel->inflight.erase(it);
          return 1;
        }
      }
      if (dh.datao + dlen <= io.req.length) {
        memcpy(static_cast<uint8_t*>(io.req.buffer) + dh.datao, pdu + pdo,
               dlen);
        io.remaining_read -= std::min<uint64_t>(io.remaining_read, dlen);
      }
      return 0;  // completion arrives as CapsuleResp
    }
    if (ch.type == kR2T) {
      DataHeader dh;
      memcpy(&dh, pdu, sizeof(dh));
      auto it = channel->inflight.find(dh.cccid);
      if (it == channel->inflight.end()) return 0;
      InflightIo& io = it->second;
      // Send the solicited range as H2CData chunks.
      uint32_t offset = dh.datao;
      uint32_t remaining = dh.datal;
      while (remaining > 0) {
        const uint32_t chunk = std::min(remaining, qp.maxh2cdata);
        DataHeader h2c{};
        const uint8_t h2c_pdo = 24 + (qp.hdgst ? 4 : 0);
        const bool last = chunk == remaining;
        h2c.ch = {kH2CData,
                  static_cast<uint8_t>((qp.hdgst ? kFlagHdgst : 0) |
                                       (qp.ddgst ? kFlagDdgst : 0) |
                                       (last ? kFlagLast : 0)),
                  24, h2c_pdo,
                  static_cast<uint32_t>(h2c_pdo + chunk + (qp.ddgst ? 4 : 0))};
        h2c.cccid = dh.cccid;
        h2c.ttag = dh.ttag;
        h2c.datao = offset;
        h2c.datal = chunk;
        const uint8_t* src = static_cast<const uint8_t*>(io.req.buffer) + offset;
        uint32_t dd = qp.ddgst ? crc32c_sw(0, src, chunk) : 0;
        if (!send_pdu(qp.fd, &h2c, 24, qp.hdgst, src, chunk, dd, qp.ddgst)) {
          if (io.req.on_complete) io.req.on_complete(kIoFailed);
          channel->inflight.erase(it);
          return 1;
        }
        offset += chunk;
        remaining -= chunk;
      }
      return 0;
    }
    if (ch.type == kCapsuleResp) {
      Cqe cqe;
      memcpy(cqe.bytes, pdu + 8, 16);
      auto it = channel->inflight.find(cqe.cid());
      if (it == channel->inflight.end()) return 0;
      InflightIo io = std::move(it->second);
      channel->inflight.erase(it);
      int status = kIoOk;
      if (cqe.status_code() != kScSuccess) status = kIoFailed;
      else if (io.remaining_read != 0) status = kIoFailed;  // short read
      if (io.req.on_complete) io.req.on_complete(status);
      return 1;
    }
    return 0;
  }

  int fail_cid_from_pdu(NvmfChannel* channel, const uint8_t*,
                        const CommonHeader&) {
    int n = 0;
    for (auto& [cid, io] : channel->inflight) {
      if (io.req.on_complete) io.req.on_complete(kIoFailed);
      ++n;
    }
    channel->inflight.clear();
    return n;
  }

  std::string traddr_;
  uint16_t trsvcid_;
  std::string subnqn_;
  uint32_t nsid_;
  bool digests_;
  std::atomic<uint16_t> next_qid_{1};
};

}  // namespace

BdevPtr create_nvmf_tcp_bdev(const std::string& name,
                             const std::string& traddr, uint16_t trsvcid,
                             const std::string& subnqn, uint32_t nsid,
                             bool enable_digests) {
  // Admin queue: connect, enable the controller, size the namespace.
  QueuePair admin;
  admin.handshake(traddr, trsvcid, enable_digests);
  admin.fabrics_connect(subnqn, /*qid=*/0, 31);
  {
    Sqe sqe{};
    sqe.set_opc(kOpcFabrics);
    sqe.set_fctype(kFctypePropertySet);
    sqe.bytes[40] = 1;  // 8-byte attribute
    sqe.set_cdw(11, kPropCc);
    sqe.set_cdw(12, 0x00460001);  // IOSQES=6, IOCQES=4, EN=1
    if (admin.command(sqe, nullptr).status_code() != kScSuccess) {
      throw std::runtime_error("nvmf: controller enable failed");
    }
  }
  {
    Sqe sqe{};
    sqe.set_opc(kOpcFabrics);
    sqe.set_fctype(kFctypePropertyGet);
    sqe.bytes[40] = 0;  // 4-byte attribute
    sqe.set_cdw(11, kPropCsts);
    Cqe cqe = admin.command(sqe, nullptr);
    if ((cqe.dw0() & 1) == 0) {
      throw std::runtime_error("nvmf: controller not ready");
    }
  }
  std::vector<uint8_t> ns_data;
  {
    Sqe sqe{};
    sqe.set_opc(kOpcIdentify);
    sqe.set_nsid(nsid);
    sqe.set_cdw(10, 0x00);  // CNS: namespace
    sqe.set_sgl_transport(4096);
    Cqe cqe = admin.command(sqe, &ns_data);
    if (cqe.status_code() != kScSuccess || ns_data.size() < 132) {
      throw std::runtime_error("nvmf: identify namespace failed");
    }
  }
  uint64_t nsze;
  memcpy(&nsze, ns_data.data(), 8);
  const uint8_t lbads = ns_data[130];
  if (nsze == 0 || lbads < 9 || lbads > 16) {
    throw std::runtime_error("nvmf: implausible namespace geometry");
  }
  // The admin connection closes here: our target keeps per-connection
  // state only, and the loopback pair does not enforce keep-alives.
  return std::make_shared<NvmfBdev>(name, traddr, trsvcid, subnqn, nsid,
                                    enable_digests, 1ull << lbads, nsze);
}

}  // namespace hipstore
