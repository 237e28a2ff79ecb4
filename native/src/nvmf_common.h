// Shared NVMe/TCP PDU definitions and socket helpers (internal).
//
// Layouts follow NVMe-oF 1.1 / NVMe/TCP transport: 8-byte common
// header (type, flags, hlen, pdo, plen LE32) + type-specific header,
// optional header digest (CRC32C), optional padded data + data digest.

#pragma once

#include <cstdint>
#include <cstring>
#include <string>
#include <vector>

#include "hipstore/crc32c.h"

namespace hipstore {
namespace nvmf {

enum PduType : uint8_t {
  kIcReq = 0x00,
  kIcResp = 0x01,
  kH2CTermReq = 0x02,
  kC2HTermReq = 0x03,
  kCapsuleCmd = 0x04,
  kCapsuleResp = 0x05,
  kH2CData = 0x06,
  kC2HData = 0x07,
  kR2T = 0x09,
};

enum PduFlags : uint8_t {
  kFlagHdgst = 0x01,
  kFlagDdgst = 0x02,
  kFlagLast = 0x04,
  kFlagSuccess = 0x08,
};

#pragma pack(push, 1)
struct CommonHeader {
  uint8_t type;
  uint8_t flags;
  uint8_t hlen;
  uint8_t pdo;
  uint32_t plen;
};

struct IcReq {  // 128 bytes total
  CommonHeader ch;
  uint16_t pfv;
  uint8_t hpda;
  uint8_t dgst;  // bit0 hdgst, bit1 ddgst
  uint32_t maxr2t;
  uint8_t reserved[112];
};

struct IcResp {  // 128 bytes total
  CommonHeader ch;
  uint16_t pfv;
  uint8_t cpda;
  uint8_t dgst;
  uint32_t maxh2cdata;
  uint8_t reserved[112];
};

struct DataHeader {  // C2HData / H2CData / R2T type-specific header
  CommonHeader ch;
  uint16_t cccid;  // command capsule cid
  uint16_t ttag;
  uint32_t datao;  // (r2to for R2T)
  uint32_t datal;  // (r2tl for R2T)
  uint32_t reserved;
};
#pragma pack(pop)

static_assert(sizeof(CommonHeader) == 8, "CH must be 8 bytes");
static_assert(sizeof(IcReq) == 128, "ICReq must be 128 bytes");
static_assert(sizeof(IcResp) == 128, "ICResp must be 128 bytes");
static_assert(sizeof(DataHeader) == 24, "data PSH must be 24 bytes");

// NVMe SQE/CQE.
struct Sqe {
  uint8_t bytes[64];

  uint8_t opc() const { return bytes[0]; }
  void set_opc(uint8_t v) { bytes[0] = v; }
  uint16_t cid() const { uint16_t v; memcpy(&v, bytes + 2, 2); return v; }
  void set_cid(uint16_t v) { memcpy(bytes + 2, &v, 2); }
  uint32_t nsid() const { uint32_t v; memcpy(&v, bytes + 4, 4); return v; }
  void set_nsid(uint32_t v) { memcpy(bytes + 4, &v, 4); }
  uint8_t fctype() const { return bytes[4]; }  // fabrics commands only
  void set_fctype(uint8_t v) { bytes[4] = v; }
  uint32_t cdw(int n) const {  // n = 10..15
    uint32_t v;
    memcpy(&v, bytes + 4 * n, 4);
    return v;
  }
  void set_cdw(int n, uint32_t v) { memcpy(bytes + 4 * n, &v, 4); }
  // SGL1 (bytes 24..39): transport data block descriptor.
  void set_sgl_transport(uint32_t length) {
    memset(bytes + 24, 0, 16);
    memcpy(bytes + 32, &length, 4);
    bytes[39] = 0x5A;  // type 5 (transport data block), subtype 0xA
  }
  uint32_t sgl_length() const {
    uint32_t v;
    memcpy(&v, bytes + 32, 4);
    return v;
  }
};

struct Cqe {
  uint8_t bytes[16];

  uint32_t dw0() const { uint32_t v; memcpy(&v, bytes, 4); return v; }
  void set_dw0(uint32_t v) { memcpy(bytes, &v, 4); }
  uint64_t result64() const { uint64_t v; memcpy(&v, bytes, 8); return v; }
  void set_result64(uint64_t v) { memcpy(bytes, &v, 8); }
  uint16_t cid() const { uint16_t v; memcpy(&v, bytes + 12, 2); return v; }
  void set_cid(uint16_t v) { memcpy(bytes + 12, &v, 2); }
  uint16_t status() const { uint16_t v; memcpy(&v, bytes + 14, 2); return v; }
  void set_status(uint16_t sc) { uint16_t v = sc << 1; memcpy(bytes + 14, &v, 2); }
  uint16_t status_code() const { return (status() >> 1) & 0xFF; }
};

static_assert(sizeof(Sqe) == 64 && sizeof(Cqe) == 16, "sqe/cqe sizes");

// NVMe opcodes / fabrics types / properties used here.
constexpr uint8_t kOpcFabrics = 0x7F;
constexpr uint8_t kOpcIdentify = 0x06;
constexpr uint8_t kOpcKeepAlive = 0x18;
constexpr uint8_t kOpcWrite = 0x01;
constexpr uint8_t kOpcRead = 0x02;
constexpr uint8_t kOpcFlush = 0x00;
constexpr uint8_t kOpcWriteZeroes = 0x08;
constexpr uint8_t kFctypePropertySet = 0x00;
constexpr uint8_t kFctypeConnect = 0x01;
constexpr uint8_t kFctypePropertyGet = 0x04;
constexpr uint32_t kPropCap = 0x00;
constexpr uint32_t kPropCc = 0x14;
constexpr uint32_t kPropCsts = 0x1C;
constexpr uint16_t kScSuccess = 0x00;
constexpr uint16_t kScInvalidOpcode = 0x01;
constexpr uint16_t kScInvalidField = 0x02;
constexpr uint16_t kScInternalError = 0x06;
constexpr uint16_t kScLbaOutOfRange = 0x80;

constexpr size_t kConnectDataSize = 1024;
constexpr uint32_t kMaxDataPerPdu = 128 * 1024;

// Socket helpers (nvmf_common.cpp).
bool read_exact(int fd, void* buf, size_t n);
bool write_exact(int fd, const void* buf, size_t n);
// writev-style gathered send of header + optional digests + data.
bool send_pdu(int fd, const void* header, size_t hlen, bool hdgst,
              const void* data, size_t dlen, uint32_t ddgst_value,
              bool ddgst);

}  // namespace nvmf
}  // namespace hipstore
