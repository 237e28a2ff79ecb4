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

uint32_t crc32c_sw(uint32_t init, const void* data, size_t len) {
  const Table& tab = table();
  const uint8_t* p = static_cast<const uint8_t*>(data);
  uint32_t crc = ~init;
  for (size_t i = 0; i < len; ++i) {
    crc = (crc >> 8) ^ tab.t[(crc ^ p[i]) & 0xFF];
  }
  return ~crc;
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
