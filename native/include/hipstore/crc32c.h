// CRC32C (Castagnoli) — software reference + GPU block digests.
//
// The reference computes CRC32C in SPDK for NVMe-oF/TCP data digests
// (reference lib/util/crc32c.c, lib/nvme/nvme_tcp.c); here the software
// table variant is the bit-for-bit reference and the GPU path computes
// per-block digests of HBM-resident data (one lane per block, table in
// LDS), batched across the queue depth.

#pragma once

#include <cstddef>
#include <cstdint>

#include "hipstore/bdev.h"

namespace hipstore {

// Software CRC32C over a buffer. `init` is the running CRC (0 for a
// fresh digest); internally applies the standard ~ pre/post conditioning
// so crc32c_sw(0, ...) matches the usual CRC-32C value.
uint32_t crc32c_sw(uint32_t init, const void* data, size_t len);
// Byte-table-only variant (the bit-exactness reference; crc32c_sw
// auto-selects the SSE4.2 hardware instruction when available).
uint32_t crc32c_table(uint32_t init, const void* data, size_t len);

// CRC of a concatenation: crc32c(A||B) from crc32c(A), crc32c(B) and
// len(B) (zlib crc32_combine construction on the Castagnoli poly).
// Lets the GPU produce per-4KiB-block CRCs in parallel while the host
// folds them into one NVMe/TCP data digest.
uint32_t crc32c_combine(uint32_t crc1, uint32_t crc2, size_t len2);

// Per-block CRC32C of `count` consecutive `block_size`-byte blocks
// starting at byte `offset` of the bdev. HBM bdevs compute on-GPU;
// other bdevs fall back to reading + software CRC (CI path).
void crc32c_hbm_blocks(Bdev* bdev, uint64_t offset, uint32_t block_size,
                       uint32_t count, uint32_t* out);

}  // namespace hipstore
