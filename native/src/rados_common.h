// Shared RADOS messenger (msgr v1) wire definitions (internal).
//
// From-scratch implementation of the subset of Ceph's legacy v1
// messenger protocol that the RBD data path needs (reference role:
// vendor/github.com/spdk/spdk/lib/bdev/rbd/bdev_rbd.c delegates this
// to librados; here both ends — the initiator in rados_client.cpp and
// the loopback fake cluster in rados_cluster.cpp — speak the protocol
// themselves, the same pattern as the NVMe/TCP pair).
//
// Wire layouts follow Ceph's msgr v1 (src/include/msgr.h,
// src/include/rados.h): 9-byte banner, entity_addr exchange,
// ceph_msg_connect / ceph_msg_connect_reply negotiation, then tagged
// frames; each CEPH_MSGR_TAG_MSG frame is ceph_msg_header (53 bytes,
// trailing header CRC32C) + front + middle + data + ceph_msg_footer
// (front/middle/data CRC32C). OSD ops use the ceph_osd_op layout
// (op, flags, extent union, payload_len) with the MODE|TYPE|n opcode
// composition scheme. The MOSDOp/MOSDOpReply *front* sections use a
// reduced-field encoding (documented at encode_osd_op_front below):
// the full MOSDOp v8 encoding drags in snapshot contexts, osdmap
// epochs and cap grants that have no meaning without a real monitor;
// the subset keeps the framing, CRCs and op structs bit-faithful and
// trims the front payload to the fields this cluster models.

#pragma once

#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

namespace hipstore {
namespace rados {

constexpr char kBanner[] = "ceph v027";   // CEPH_BANNER, 9 bytes on wire
constexpr size_t kBannerLen = 9;

// CEPH_MSGR_TAG_* (msgr.h)
enum Tag : uint8_t {
  kTagReady = 1,
  kTagResetSession = 4,
  kTagClose = 6,
  kTagMsg = 7,
  kTagAck = 8,
  kTagKeepalive2 = 14,
  kTagKeepalive2Ack = 15,
};

// CEPH_ENTITY_TYPE_* (msgr.h)
enum EntityType : uint32_t {
  kEntityMon = 0x01,
  kEntityOsd = 0x04,
  kEntityClient = 0x08,
};

// CEPH_MSG_* message types (ceph_fs.h)
enum MsgType : uint16_t {
  kMsgPing = 17,
  kMsgOsdOp = 42,
  kMsgOsdOpReply = 43,
};

// CEPH_OSD_OP_* composition scheme (rados.h): MODE | TYPE | number.
enum OsdOpMode : uint16_t {
  kOsdOpModeRd = 0x1000,
  kOsdOpModeWr = 0x2000,
  kOsdOpTypeData = 0x0200,
};

enum OsdOp : uint16_t {
  kOsdOpRead = kOsdOpModeRd | kOsdOpTypeData | 1,
  kOsdOpStat = kOsdOpModeRd | kOsdOpTypeData | 2,
  kOsdOpWrite = kOsdOpModeWr | kOsdOpTypeData | 1,
  kOsdOpWriteFull = kOsdOpModeWr | kOsdOpTypeData | 2,
  kOsdOpTruncate = kOsdOpModeWr | kOsdOpTypeData | 3,
  kOsdOpZero = kOsdOpModeWr | kOsdOpTypeData | 4,
  kOsdOpDelete = kOsdOpModeWr | kOsdOpTypeData | 7,
  kOsdOpCreate = kOsdOpModeWr | kOsdOpTypeData | 13,
};

#pragma pack(push, 1)

// entity_addr_t legacy encoding: type + nonce + sockaddr_storage
// (family and port big-endian, IPv4 address in network order).
struct EntityAddr {
  uint32_t type = 0;
  uint32_t nonce = 0;
  uint16_t family_be = 0;      // AF_INET big-endian
  uint16_t port_be = 0;
  uint8_t addr4[4] = {0, 0, 0, 0};
  uint8_t pad[120] = {0};      // rest of sockaddr_storage
};
static_assert(sizeof(EntityAddr) == 136, "entity_addr is 8+128 bytes");

// ceph_msg_connect (msgr.h)
struct MsgConnect {
  uint64_t features;
  uint32_t host_type;          // CEPH_ENTITY_TYPE_* of the sender
  uint32_t global_seq;
  uint32_t connect_seq;
  uint32_t protocol_version;
  uint32_t authorizer_protocol;  // CEPH_AUTH_NONE = 0
  uint32_t authorizer_len;
  uint8_t flags;
};
static_assert(sizeof(MsgConnect) == 33, "ceph_msg_connect is 33 bytes");

// ceph_msg_connect_reply (msgr.h)
struct MsgConnectReply {
  uint8_t tag;
  uint64_t features;
  uint32_t global_seq;
  uint32_t connect_seq;
  uint32_t protocol_version;
  uint32_t authorizer_len;
  uint8_t flags;
};
static_assert(sizeof(MsgConnectReply) == 26,
              "ceph_msg_connect_reply is 26 bytes");

// ceph_msg_header (ceph_fs.h): trailing crc is CRC32C(-1 seeded per
// Ceph convention? No — Ceph seeds msgr CRCs with 0) of bytes [0, 49).
struct MsgHeader {
  uint64_t seq;
  uint64_t tid;
  uint16_t type;       // MsgType
  uint16_t priority;
  uint16_t version;
  uint32_t front_len;
  uint32_t middle_len;
  uint32_t data_len;
  uint16_t data_off;
  uint8_t src_type;    // entity_name: type
  uint64_t src_num;    // entity_name: num
  uint16_t compat_version;
  uint16_t reserved;
  uint32_t crc;        // crc32c of the 49 bytes above
};
static_assert(sizeof(MsgHeader) == 53, "ceph_msg_header is 53 bytes");

// ceph_msg_footer (ceph_fs.h)
struct MsgFooter {
  uint32_t front_crc;
  uint32_t middle_crc;
  uint32_t data_crc;
  uint64_t sig;        // 0: CEPH_AUTH_NONE signs nothing
  uint8_t flags;       // bit0 FOOTER_COMPLETE
};
static_assert(sizeof(MsgFooter) == 21, "ceph_msg_footer is 21 bytes");

// ceph_osd_op (rados.h): opcode + flags + extent union + payload_len.
struct CephOsdOp {
  uint16_t op;
  uint32_t flags;
  uint64_t offset;         // extent.offset
  uint64_t length;         // extent.length
  uint64_t truncate_size;  // extent.truncate_size
  uint32_t truncate_seq;   // extent.truncate_seq
  uint32_t payload_len;    // bytes of this op's slice of the data section
};
static_assert(sizeof(CephOsdOp) == 38, "ceph_osd_op wire struct");

#pragma pack(pop)

// ---------------------------------------------------------------------------
// Ceph-style little-endian encode/decode helpers
// ---------------------------------------------------------------------------

inline void put_bytes(std::vector<uint8_t>* out, const void* p, size_t n) {
  const uint8_t* b = static_cast<const uint8_t*>(p);
  out->insert(out->end(), b, b + n);
}

template <typename T>
void put_le(std::vector<uint8_t>* out, T v) {
  put_bytes(out, &v, sizeof(v));
}

// Ceph string encoding: le32 length + bytes.
inline void put_string(std::vector<uint8_t>* out, const std::string& s) {
  put_le<uint32_t>(out, static_cast<uint32_t>(s.size()));
  put_bytes(out, s.data(), s.size());
}

class Decoder {
 public:
  Decoder(const uint8_t* p, size_t n) : p_(p), end_(p + n) {}
  explicit Decoder(const std::vector<uint8_t>& v)
      : Decoder(v.data(), v.size()) {}

  template <typename T>
  T get_le() {
    T v;
    take(&v, sizeof(v));
    return v;
  }

  std::string get_string() {
    const uint32_t n = get_le<uint32_t>();
    if (n > remaining()) throw std::runtime_error("rados: bad string len");
    std::string s(reinterpret_cast<const char*>(p_), n);
    p_ += n;
    return s;
  }

  void take(void* out, size_t n) {
    if (n > remaining()) throw std::runtime_error("rados: short decode");
    memcpy(out, p_, n);
    p_ += n;
  }

  size_t remaining() const { return static_cast<size_t>(end_ - p_); }

 private:
  const uint8_t* p_;
  const uint8_t* end_;
};

// ---------------------------------------------------------------------------
// MOSDOp / MOSDOpReply front sections (reduced encoding, see file
// header). Request front:
//   u8 struct_v(=1) | le64 pool | string oid | le16 num_ops |
//   num_ops x CephOsdOp
// The data section is the concatenation of each write-class op's
// payload (payload_len bytes each). Reply front:
//   u8 struct_v(=1) | string oid | le32 result(s32) | le16 num_ops |
//   num_ops x CephOsdOp (payload_len = that op's slice of reply data)
// ---------------------------------------------------------------------------

struct OsdOpRequest {
  uint64_t pool = 0;
  std::string oid;
  std::vector<CephOsdOp> ops;
};

struct OsdOpReply {
  std::string oid;
  int32_t result = 0;
  std::vector<CephOsdOp> ops;
};

inline std::vector<uint8_t> encode_osd_op_front(const OsdOpRequest& req) {
  std::vector<uint8_t> out;
  put_le<uint8_t>(&out, 1);  // struct_v
  put_le<uint64_t>(&out, req.pool);
  put_string(&out, req.oid);
  put_le<uint16_t>(&out, static_cast<uint16_t>(req.ops.size()));
  for (const CephOsdOp& op : req.ops) put_bytes(&out, &op, sizeof(op));
  return out;
}

inline OsdOpRequest decode_osd_op_front(const std::vector<uint8_t>& front) {
  Decoder d(front);
  OsdOpRequest req;
  if (d.get_le<uint8_t>() != 1) {
    throw std::runtime_error("rados: unsupported MOSDOp struct_v");
  }
  req.pool = d.get_le<uint64_t>();
  req.oid = d.get_string();
  const uint16_t n = d.get_le<uint16_t>();
  if (n > 16) throw std::runtime_error("rados: too many ops");
  req.ops.resize(n);
  for (CephOsdOp& op : req.ops) d.take(&op, sizeof(op));
  return req;
}

inline std::vector<uint8_t> encode_osd_op_reply_front(const OsdOpReply& r) {
  std::vector<uint8_t> out;
  put_le<uint8_t>(&out, 1);
  put_string(&out, r.oid);
  put_le<int32_t>(&out, r.result);
  put_le<uint16_t>(&out, static_cast<uint16_t>(r.ops.size()));
  for (const CephOsdOp& op : r.ops) put_bytes(&out, &op, sizeof(op));
  return out;
}

inline OsdOpReply decode_osd_op_reply_front(const std::vector<uint8_t>& f) {
  Decoder d(f);
  OsdOpReply r;
  if (d.get_le<uint8_t>() != 1) {
    throw std::runtime_error("rados: unsupported MOSDOpReply struct_v");
  }
  r.oid = d.get_string();
  r.result = d.get_le<int32_t>();
  const uint16_t n = d.get_le<uint16_t>();
  if (n > 16) throw std::runtime_error("rados: too many reply ops");
  r.ops.resize(n);
  for (CephOsdOp& op : r.ops) d.take(&op, sizeof(op));
  return r;
}

// ---------------------------------------------------------------------------
// Socket framing helpers (implemented in rados_client.cpp; shared with
// the cluster)
// ---------------------------------------------------------------------------

bool read_exact(int fd, void* buf, size_t n);
bool write_exact(int fd, const void* buf, size_t n);

// Complete a banner + addr + connect negotiation on a connected fd.
// `server` selects which side of the exchange to play. Returns false
// on any mismatch (caller closes the fd).
bool msgr_handshake(int fd, bool server, uint32_t my_entity_type);

// One CEPH_MSGR_TAG_MSG frame. Send computes header/front/data CRCs
// (data_crc may be precomputed by the GPU and passed in; pass
// ~0u to have the software CRC computed here).
bool msgr_send(int fd, uint64_t seq, uint64_t tid, uint16_t type,
               const std::vector<uint8_t>& front, const uint8_t* data,
               uint32_t data_len, uint32_t data_crc_precomputed = ~0u);

struct MsgrFrame {
  MsgHeader header;
  std::vector<uint8_t> front;
  std::vector<uint8_t> data;
  uint32_t footer_data_crc = 0;  // as received (GPU-verified later
                                 // when verify_data_crc was false)
};

// Receive the next TAG_MSG frame (skipping ACK/KEEPALIVE tags);
// verifies header and front CRCs. `verify_data_crc` additionally
// checks the data CRC here in software — the HBM-backed cluster skips
// that and verifies on the GPU after the payload lands in HBM.
// Returns false on EOF/short read/CRC mismatch.
bool msgr_recv(int fd, MsgrFrame* frame, bool verify_data_crc);

}  // namespace rados
}  // namespace hipstore
