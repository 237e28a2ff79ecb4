#include "hipstore/crc32c.h"

#include <vector>

#include "hipstore/engine.h"

namespace hipstore {

namespace {

// Castagnoli polynomial, reflected form.
constexpr uint32_t kPolyReflected = 0x82F63B78u;

struct Table {
  uint32_t t[256];
  Table() {
    for (uint32_t i = 0; i < 256; ++i) {
      uint32_t crc = i;
      for (int k = 0; k < 8; ++k) {
        crc = (crc & 1) ? (crc >> 1) ^ kPolyReflected : crc >> 1;
      }
      t[i] = crc;
    }
  }
};

const Table& table() {
  static Table t;
  return t;
}

}  // namespace

// Hardware path: x86 SSE4.2 carries a Castagnoli CRC instruction
// (the same accelerator SPDK's spdk_crc32c_update uses, reference
// lib/util/crc32c.c:56-80). ~0.1 cycle/byte vs ~4 for the byte table
// — at NVMe/TCP data-digest rates that is ~10 us/IO saved across the
// two ends. Table path remains the bit-exactness reference (KATs).
__attribute__((target("sse4.2")))
static uint32_t crc32c_hw(uint32_t crc, const uint8_t* p, size_t len) {
  while (len >= 8) {
    crc = static_cast<uint32_t>(
        __builtin_ia32_crc32di(crc, *reinterpret_cast<const uint64_t*>(p)));
    p += 8;
    len -= 8;
  }
  while (len > 0) {
    crc = __builtin_ia32_crc32qi(crc, *p);
    ++p;
    --len;
  }
  return crc;
}

static bool have_sse42() {
  static const bool ok = __builtin_cpu_supports("sse4.2");
  return ok;
}

uint32_t crc32c_sw(uint32_t init, const void* data, size_t len) {
  const uint8_t* p = static_cast<const uint8_t*>(data);
  uint32_t crc = ~init;
  if (have_sse42()) return ~crc32c_hw(crc, p, len);
  const Table& tab = table();
  for (size_t i = 0; i < len; ++i) {
    crc = (crc >> 8) ^ tab.t[(crc ^ p[i]) & 0xFF];
  }
  return ~crc;
}

// Table-only variant for tests: the bit-exactness reference the
// hardware and GPU paths are checked against.
uint32_t crc32c_table(uint32_t init, const void* data, size_t len) {
  const Table& tab = table();
  const uint8_t* p = static_cast<const uint8_t*>(data);
  uint32_t crc = ~init;
  for (size_t i = 0; i < len; ++i) {
    crc = (crc >> 8) ^ tab.t[(crc ^ p[i]) & 0xFF];
  }
  return ~crc;
}

namespace {

// GF(2) 32x32 matrix ops over the CRC register (zlib crc32_combine).
uint32_t gf2_times(const uint32_t* mat, uint32_t vec) {
  uint32_t sum = 0;
  int i = 0;
  while (vec) {
    if (vec & 1) sum ^= mat[i];
    vec >>= 1;
    ++i;
  }
  return sum;
}

void gf2_square(uint32_t* square, const uint32_t* mat) {
  for (int i = 0; i < 32; ++i) square[i] = gf2_times(mat, mat[i]);
}

}  // namespace

uint32_t crc32c_combine(uint32_t crc1, uint32_t crc2, size_t len2) {
  if (len2 == 0) return crc1;
  uint32_t even[32];  // operator for 2^(k) zero bytes
  uint32_t odd[32];
  // operator for one zero BIT: shift-right with poly feedback
  odd[0] = kPolyReflected;
  for (int i = 1; i < 32; ++i) odd[i] = 1u << (i - 1);
  gf2_square(even, odd);  // operator for 2 zero bits
  gf2_square(odd, even);  // operator for 4 zero bits
  // Square-and-multiply over len2 zero BYTES: the first loop square
  // yields the 8-bit (1 byte) operator (zlib crc32_combine ladder).
  do {
    gf2_square(even, odd);
    if (len2 & 1) crc1 = gf2_times(even, crc1);
    len2 >>= 1;
    if (len2 == 0) break;
    gf2_square(odd, even);
    if (len2 & 1) crc1 = gf2_times(odd, crc1);
    len2 >>= 1;
  } while (len2);
  return crc1 ^ crc2;
}

void crc32c_cpu_fallback(Bdev* bdev, uint64_t offset, uint32_t block_size,
                         uint32_t count, uint32_t* out) {
  std::vector<uint8_t> buf(block_size);
  for (uint32_t i = 0; i < count; ++i) {
    void* bounce = alloc_pinned(block_size);
    int status =
        bdev_read_sync(bdev, offset + static_cast<uint64_t>(i) * block_size,
                       bounce, block_size);
    if (status != kIoOk) {
      free_pinned(bounce);
      throw std::runtime_error("crc32c: read failed");
    }
    out[i] = crc32c_sw(0, bounce, block_size);
    free_pinned(bounce);
  }
}

}  // namespace hipstore
