// MI355X (gfx950) data-path engine: HBM-resident malloc bdev + batched
// LDS-staged block-copy kernels on per-queue HIP streams.
//
// Design (MI355X-first, not a port of SPDK's reactor model):
//   - The bdev backing store is one hipMalloc region in HBM3E.
//   - Each IoChannel owns a HIP stream, a pinned-host descriptor ring the
//     GPU reads directly (zero-copy over PCIe), and an event pool.
//   - submit() only queues; poll() coalesces every pending request into
//     ONE kernel launch (a batch), records an event, and retires
//     finished batches. This replaces SPDK's per-core poller loop
//     (reference lib/bdev/malloc/bdev_malloc.c submit/complete) with a
//     launch-batched GPU pipeline: at QD=32 a whole queue depth is one
//     launch (~4 us) + one 128 KiB PCIe burst.
//   - Kernels stage every 4 KiB block tile through LDS (16 B/lane
//     vectorized, one wave per tile, 4 waves per workgroup): the staging
//     hop costs little (LDS ~150 TB/s vs ~55 GB/s PCIe / 6.3 TB/s HBM)
//     and gives the CRC32C/verify paths a resident tile to chew on.

#include <hip/hip_runtime.h>
#include <pthread.h>
#include <sched.h>
#include <unistd.h>
#include <stdlib.h>

#include <algorithm>
#include <atomic>
#include <chrono>
#include <condition_variable>
#include <cstring>
#include <deque>
#include <functional>
#include <map>
#include <mutex>
#include <random>
#include <set>
#include <stdexcept>
#include <thread>
#include <utility>
#include <vector>

#include "hipstore/crc32c.h"
#include "hipstore/engine.h"

namespace hipstore {

// CPU fallback defined in crc32c.cpp.
void crc32c_cpu_fallback(Bdev* bdev, uint64_t offset, uint32_t block_size,
                         uint32_t count, uint32_t* out);

#define HIP_CHECK(expr)                                                   \
  do {                                                                    \
    hipError_t _e = (expr);                                               \
    if (_e != hipSuccess) {                                               \
      throw std::runtime_error(std::string("HIP error: ") +               \
                               hipGetErrorString(_e) + " at " #expr);     \
    }                                                                     \
  } while (0)

// ---------------------------------------------------------------------------
// Kernels
// ---------------------------------------------------------------------------

namespace {

constexpr uint32_t kTileBytes = 4096;   // one wave moves one tile
constexpr uint32_t kWavesPerWg = 4;     // 256-thread workgroups
constexpr uint32_t kRingSlots = 1 << 16;  // descriptor ring per channel
constexpr uint32_t kMaxBatchTiles = kRingSlots / 4;

struct BlockDesc {
  const uint8_t* src;  // null => fill
  uint8_t* dst;
  uint32_t bytes;      // multiple of 16, <= kTileBytes
  uint32_t fill;       // fill byte (replicated) when src == null
};
static_assert(sizeof(BlockDesc) == 24, "persistent engine reads 3 x u64");

// One wave per descriptor; 16 B/lane vector moves staged through LDS.
// Each lane re-reads exactly the bytes it wrote, so wave-internal
// lgkmcnt ordering (compiler-inserted) is the only sync needed — no
// __syncthreads() in the hot path.
__global__ __launch_bounds__(kWavesPerWg * 64) void k_copy_blocks(
    const BlockDesc* __restrict__ descs, uint32_t n) {
  __shared__ __attribute__((aligned(16))) uint8_t lds_raw[kWavesPerWg * kTileBytes];
  const uint32_t wave = threadIdx.x >> 6;
  const uint32_t lane = threadIdx.x & 63;
  const uint32_t idx = blockIdx.x * kWavesPerWg + wave;
  if (idx >= n) return;
  const BlockDesc d = descs[idx];
  float4* lds = reinterpret_cast<float4*>(lds_raw + wave * kTileBytes);
  const uint32_t n16 = d.bytes >> 4;  // 16-byte units
  if (d.src != nullptr) {
    const float4* __restrict__ src = reinterpret_cast<const float4*>(d.src);
#pragma unroll 4
    for (uint32_t i = lane; i < n16; i += 64) lds[i] = src[i];
  } else {
    const uint32_t b = d.fill & 0xFF;
    const uint32_t word = b | (b << 8) | (b << 16) | (b << 24);
    const float4 v = {__uint_as_float(word), __uint_as_float(word),
                      __uint_as_float(word), __uint_as_float(word)};
    for (uint32_t i = lane; i < n16; i += 64) lds[i] = v;
  }
  float4* __restrict__ dst = reinterpret_cast<float4*>(d.dst);
#pragma unroll 4
  for (uint32_t i = lane; i < n16; i += 64) dst[i] = lds[i];
}

// Contiguous device-to-device range copy (volume clone): grid-stride
// over 16-byte elements, staged through LDS like the block kernels.
// No descriptors — the common clone case is one contiguous extent and
// per-tile descriptor fetches over PCIe dominated a descriptor-based
// clone (measured ~4x).
__global__ __launch_bounds__(256) void k_copy_range(
    const float4* __restrict__ src, float4* __restrict__ dst, uint64_t n16) {
  __shared__ __attribute__((aligned(16))) float4 lds[256];
  const uint64_t stride = static_cast<uint64_t>(gridDim.x) * blockDim.x;
  for (uint64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n16;
       i += stride) {
    lds[threadIdx.x] = src[i];
    dst[i] = lds[threadIdx.x];
  }
}

// --- CRC32C (Castagnoli, reflected 0x82F63B78) -----------------------------
//
// One LANE per block: lanes of a wave digest 64 different blocks in
// parallel, the 256-entry table lives in LDS (one copy per workgroup,
// broadcast reads are conflict-free). Parallelism comes from batch
// width (count >= a few thousand blocks fills the chip), which matches
// the NVMe-oF/TCP digest use: one CRC per in-flight 4 KiB PDU.
__global__ __launch_bounds__(256) void k_crc32c_blocks(
    const uint8_t* __restrict__ base, uint32_t block_size, uint32_t count,
    uint32_t* __restrict__ out) {
  __shared__ uint32_t table[256];
  // Build the table once per workgroup (cheap: 8 iterations/entry).
  if (threadIdx.x < 256) {
    uint32_t crc = threadIdx.x;
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      crc = (crc & 1u) ? (crc >> 1) ^ 0x82F63B78u : crc >> 1;
    }
    table[threadIdx.x] = crc;
  }
  __syncthreads();
  const uint32_t idx = blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= count) return;
  const uint8_t* p = base + static_cast<uint64_t>(idx) * block_size;
  uint32_t crc = 0xFFFFFFFFu;
  // 4 bytes per iteration through a word load (block_size % 4 == 0 is
  // guaranteed by the 512-byte block granularity).
  const uint32_t* pw = reinterpret_cast<const uint32_t*>(p);
  for (uint32_t i = 0; i < block_size / 4; ++i) {
    uint32_t w = pw[i];
#pragma unroll
    for (int b = 0; b < 4; ++b) {
      crc = (crc >> 8) ^ table[(crc ^ (w & 0xFFu)) & 0xFFu];
      w >>= 8;
    }
  }
  out[idx] = ~crc;
}

// --- persistent service kernel ---------------------------------------------
//
// SPDK replaced kernel-driver queues with host polling reactors; the
// MI355X-native equivalent puts the poller ON the GPU: one persistent
// launch per queue whose waves claim submission-ring descriptors with a
// device-scope atomic (dequeue ~0.25-1.1 us, MI355X_MICROARCH price
// list), copy the 4 KiB tile through LDS, and publish a completion word
// to pinned host memory with a system-scope release. Submission latency
// becomes host-store -> PCIe poll -> copy (~5-10 us) instead of a
// kernel launch per batch.
//
// Liveness contract: waves poll with s_sleep and EXIT after
// `idle_timeout` ticks without new work (a crashed host can therefore
// never wedge the GPU); the host relaunches on demand with the claim
// counter reset to its completed prefix — descriptor replay is
// idempotent (same copy, same completion value), so the handoff is
// race-free without any further coordination.

__device__ __forceinline__ unsigned long long wave_bcast_u64(
    unsigned long long v) {
  int lo = __shfl(static_cast<int>(v & 0xFFFFFFFFu), 0);
  int hi = __shfl(static_cast<int>(v >> 32), 0);
  return (static_cast<unsigned long long>(static_cast<uint32_t>(hi)) << 32) |
         static_cast<uint32_t>(lo);
}

struct PersistentCtl {
  // Pinned host memory (GPU reads/writes over PCIe, uncached):
  const BlockDesc* sq;     // descriptor ring
  volatile unsigned long long* sq_tail;  // host-written monotonic count
  volatile unsigned long long* cq;  // per-slot completion: seq = idx+1
  volatile uint32_t* stop;     // host-set stop flag
  // Device memory. Only the LEADER wave touches host memory while
  // idle: with many queues, dozens of waves spin-reading a pinned
  // word would storm PCIe with small reads (measured: 4 channels x 16
  // spinning waves collapsed throughput ~30x). Workers poll the
  // device-resident mirror instead (relaxed agent loads = L2-served
  // sc1, no L1 invalidates — the monotonic counter IS the flag).
  unsigned long long* claim_counter;  // monotonic claim index
  unsigned long long* known_tail;     // leader-published tail mirror
  uint32_t* exit_flag;                // leader-published stop/idle exit
  uint32_t ring_mask;
  uint32_t idle_spins;         // leader spins before self-exit
};

__global__ __launch_bounds__(64) void k_persistent_copy(PersistentCtl ctl) {
  __shared__ __attribute__((aligned(16))) uint8_t lds_raw[kTileBytes];
  const uint32_t lane = threadIdx.x & 63;
  float4* lds = reinterpret_cast<float4*>(lds_raw);

  if (blockIdx.x == 0) {
    // Leader wave: the only PCIe poller. Mirrors sq_tail into device
    // memory and publishes stop/idle exits.
    if (lane != 0) return;
    unsigned long long known = __hip_atomic_load(
        ctl.known_tail, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    uint32_t spins = 0;
    while (true) {
      unsigned long long tail = __hip_atomic_load(
          const_cast<const unsigned long long*>(ctl.sq_tail),
          __ATOMIC_ACQUIRE, __HIP_MEMORY_SCOPE_SYSTEM);
      if (tail > known) {
        known = tail;
        __hip_atomic_store(ctl.known_tail, known, __ATOMIC_RELEASE,
                           __HIP_MEMORY_SCOPE_AGENT);
        spins = 0;
      }
      if (__hip_atomic_load(const_cast<const uint32_t*>(ctl.stop),
                            __ATOMIC_RELAXED,
                            __HIP_MEMORY_SCOPE_SYSTEM) != 0 ||
          ++spins > ctl.idle_spins) {
        __hip_atomic_store(ctl.exit_flag, 1u, __ATOMIC_RELEASE,
                           __HIP_MEMORY_SCOPE_AGENT);
        return;
      }
      __builtin_amdgcn_s_sleep(8);
    }
  }

  while (true) {
    // Bounded CAS claim: take a descriptor index only when the leader
    // has published work past it (claim-only-when-available, the
    // shared-service discipline). The original blind
    // fetch_add-then-wait form wedged the serving wave on current
    // pool firmware (round-2 diagnosis, tools/engine_diag*.sh) while
    // this form is proven on the same boxes by k_shared_service.
    const unsigned long long kt = __hip_atomic_load(
        ctl.known_tail, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    unsigned long long c = __hip_atomic_load(
        ctl.claim_counter, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    unsigned long long claim = ~0ull;
    while (c < kt) {
      unsigned long long witnessed = 0;
      if (lane == 0) {
        witnessed = atomicCAS(ctl.claim_counter, c, c + 1);
      }
      witnessed = wave_bcast_u64(witnessed);
      if (witnessed == c) {
        claim = c;
        break;
      }
      c = witnessed;
    }
    if (claim == ~0ull) {
      if (__hip_atomic_load(ctl.exit_flag, __ATOMIC_RELAXED,
                            __HIP_MEMORY_SCOPE_AGENT) != 0) {
        return;
      }
      __builtin_amdgcn_s_sleep(16);
      continue;
    }
    // Volatile loads: the descriptor address is computable BEFORE the
    // wait loop, so a plain load could be hoisted above it by the
    // compiler and read the slot before the host wrote it (observed as
    // null-dst faults / phantom zero-byte completions). Volatile pins
    // the reads after the tail match; host memory is uncached on the
    // GPU side, so the values are then current by construction. Three
    // independent u64 loads pipeline over PCIe (vs 4 dependent field
    // reads at ~1 us each).
    const volatile unsigned long long* vd =
        reinterpret_cast<const volatile unsigned long long*>(
            &ctl.sq[claim & ctl.ring_mask]);
    const unsigned long long w0 = vd[0];
    const unsigned long long w1 = vd[1];
    const unsigned long long w2 = vd[2];
    BlockDesc d;
    d.src = reinterpret_cast<const uint8_t*>(w0);
    d.dst = reinterpret_cast<uint8_t*>(w1);
    d.bytes = static_cast<uint32_t>(w2 & 0xFFFFFFFFu);
    d.fill = static_cast<uint32_t>(w2 >> 32);
    // Uniform u64 tile path, each lane staging and draining ITS OWN
    // LDS slots (slot i covered by lane i%64 in both loops — no
    // cross-lane LDS dependency, so wave-internal lgkmcnt ordering is
    // the only sync needed). Loads are agent-scope (sc1,
    // L2-bypassing): a RESIDENT kernel has no dispatch boundaries, so
    // a plain load can hit a stale line in this XCD's L2 written
    // around by another XCD's wave (caught as a read-after-write
    // corruption by TestVhostHbm). Batched kernels don't need this —
    // each dispatch's implicit acquire invalidates the caches.
    const uint32_t n8 = d.bytes >> 3;
    unsigned long long* lds64 = reinterpret_cast<unsigned long long*>(lds);
    if (d.src != nullptr) {
      const unsigned long long* __restrict__ src =
          reinterpret_cast<const unsigned long long*>(d.src);
#pragma unroll 4
      for (uint32_t i = lane; i < n8; i += 64) {
        lds64[i] = __hip_atomic_load(src + i, __ATOMIC_RELAXED,
                                     __HIP_MEMORY_SCOPE_AGENT);
      }
    } else {
      const uint32_t b = d.fill & 0xFF;
      const uint32_t word = b | (b << 8) | (b << 16) | (b << 24);
      const unsigned long long v =
          word | (static_cast<unsigned long long>(word) << 32);
      for (uint32_t i = lane; i < n8; i += 64) lds64[i] = v;
    }
    unsigned long long* __restrict__ dst =
        reinterpret_cast<unsigned long long*>(d.dst);
#pragma unroll 4
    for (uint32_t i = lane; i < n8; i += 64) dst[i] = lds64[i];
    // Publish the completion: data must be host-visible before the CQ
    // word. The system-scope RELEASE store carries that ordering by
    // itself (s_waitcnt on the wave's outstanding stores + L2
    // writeback before the store). An explicit __threadfence_system()
    // here is redundant — and on the current pool firmware it HANGS
    // the wave: round-2 diagnosis (tools/engine_diag3.sh) showed the
    // serving wave copying the data correctly, then never publishing
    // the CQ nor claiming again, with the fence the only instruction
    // between those two points. Do not reintroduce it.
    if (lane == 0) {
      __hip_atomic_store(&ctl.cq[claim & ctl.ring_mask], claim + 1,
                         __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_SYSTEM);
    }
  }
}

// Debug counters for the persistent engine (read via pybind
// persistent_stats(); helps separate relaunch thrash from scheduling
// issues on multi-queue collapses).
// Writes the persistent channel's three device control words
// ([claim, known_tail, exit_flag]) from kernel arguments — the
// launch-path alternative to a pageable-source hipMemcpyAsync on the
// service stream (engine_diag: candidate root cause of the fresh-box
// service stall).
__global__ void k_init_ctl(unsigned long long* dst, unsigned long long a,
                           unsigned long long b, unsigned long long c) {
  if (threadIdx.x == 0) {
    dst[0] = a;
    dst[1] = b;
    dst[2] = c;
  }
}

std::atomic<uint64_t> g_pers_launches{0};
std::atomic<uint64_t> g_pers_relaunches{0};
std::atomic<uint64_t> g_pers_stall_queries{0};

// --- shared service kernel (one per device) --------------------------------
//
// The per-queue persistent kernel needs one hardware queue per channel
// (GPU_MAX_HW_QUEUES caps that per process). The shared service is the
// scalable form: ONE resident kernel per device multiplexes up to 64
// submission rings. The leader workgroup polls every active ring's
// host tail IN PARALLEL (lane l owns slot l, so a full sweep is one
// PCIe round trip, not 64); worker waves sweep the slots and claim
// descriptors with a bounded CAS (claim-only-when-available — a blind
// fetch_add would strand a worker on one ring while others have work).
// Same liveness contract as the per-queue kernel: idle self-exit +
// idempotent-replay relaunch.

struct SharedSlot {
  // Written by the host (hipMemcpy) while inactive; read by waves via
  // relaxed agent loads. 8-byte fields so every access is one atom.
  const BlockDesc* sq;                    // pinned host ring (device view)
  volatile unsigned long long* sq_tail;   // pinned host tail (device view)
  volatile unsigned long long* cq;        // pinned host CQ (device view)
  unsigned long long claim;               // device-side claim counter
  unsigned long long known_tail;          // leader-mirrored tail
  unsigned long long ring_mask_active;    // low 32: mask, high 32: active
};
static_assert(sizeof(SharedSlot) == 48, "8-byte atom layout");

constexpr uint32_t kSharedSlots = 64;

struct SharedCtl {
  SharedSlot* slots;            // device memory, kSharedSlots entries
  volatile uint32_t* stop;      // pinned host
  uint32_t* exit_flag;          // device
  uint32_t idle_spins;
};

// Slot fields as 8-byte words for uniform agent-scope access (sc1
// loads bypass the CU's L1, which may hold stale lines from a previous
// occupant of a reused slot): [0]=sq [1]=sq_tail [2]=cq [3]=claim
// [4]=known_tail [5]=mask|active<<32.
__device__ __forceinline__ unsigned long long slot_ld(
    SharedSlot* slot, int word) {
  return __hip_atomic_load(
      reinterpret_cast<unsigned long long*>(slot) + word, __ATOMIC_RELAXED,
      __HIP_MEMORY_SCOPE_AGENT);
}

__device__ __forceinline__ void slot_st(SharedSlot* slot, int word,
                                        unsigned long long value) {
  __hip_atomic_store(reinterpret_cast<unsigned long long*>(slot) + word,
                     value, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_AGENT);
}

__global__ __launch_bounds__(64) void k_shared_service(SharedCtl ctl) {
  __shared__ __attribute__((aligned(16))) uint8_t lds_raw[kTileBytes];
  const uint32_t lane = threadIdx.x & 63;
  float4* lds = reinterpret_cast<float4*>(lds_raw);

  if (blockIdx.x == 0) {
    // Leader wave: lane l mirrors slot l's host tail — a full sweep of
    // all 64 slots is ONE parallel PCIe round trip.
    SharedSlot* slot = &ctl.slots[lane];
    unsigned long long known = 0;
    uint32_t spins = 0;
    while (true) {
      bool advanced = false;
      if (slot_ld(slot, 5) >> 32) {
        volatile unsigned long long* tail_ptr =
            reinterpret_cast<volatile unsigned long long*>(slot_ld(slot, 1));
        const unsigned long long tail = __hip_atomic_load(
            const_cast<const unsigned long long*>(tail_ptr),
            __ATOMIC_ACQUIRE, __HIP_MEMORY_SCOPE_SYSTEM);
        if (tail > known) {
          known = tail;
          slot_st(slot, 4, known);
          advanced = true;
        }
      } else {
        known = 0;  // slot detached: forget its generation
      }
      const bool stop_now =
          lane == 0 &&
          __hip_atomic_load(const_cast<const uint32_t*>(ctl.stop),
                            __ATOMIC_RELAXED,
                            __HIP_MEMORY_SCOPE_SYSTEM) != 0;
      if (__any(advanced)) spins = 0;
      if (__any(stop_now) || ++spins > ctl.idle_spins) {
        if (lane == 0) {
          __hip_atomic_store(ctl.exit_flag, 1u, __ATOMIC_RELEASE,
                             __HIP_MEMORY_SCOPE_AGENT);
        }
        return;
      }
      __builtin_amdgcn_s_sleep(8);
    }
  }

  // Worker wave: sweep the slots, claim and serve at most one
  // descriptor per slot per sweep (fairness across rings). A bounded
  // CAS claims work only when available — a blind fetch_add would park
  // this worker on one ring while others have work.
  const uint32_t start = blockIdx.x;  // stagger sweep origins
  while (true) {
    bool progress = false;
    for (uint32_t i = 0; i < kSharedSlots; ++i) {
      SharedSlot* slot = &ctl.slots[(start + i) % kSharedSlots];
      const unsigned long long ma = slot_ld(slot, 5);
      if (!(ma >> 32)) continue;
      const uint32_t ring_mask = static_cast<uint32_t>(ma);
      const unsigned long long kt = slot_ld(slot, 4);
      unsigned long long c = slot_ld(slot, 3);
      unsigned long long claim = ~0ull;
      while (c < kt) {
        unsigned long long witnessed = 0;
        if (lane == 0) {
          witnessed = atomicCAS(&slot->claim, c, c + 1);
        }
        witnessed = wave_bcast_u64(witnessed);
        if (witnessed == c) {
          claim = c;
          break;
        }
        c = witnessed;
      }
      if (claim == ~0ull) continue;
      // Serve it (volatile loads: pinned after the claim, never hoisted).
      const BlockDesc* sq = reinterpret_cast<const BlockDesc*>(slot_ld(slot, 0));
      const volatile unsigned long long* vd =
          reinterpret_cast<const volatile unsigned long long*>(
              &sq[claim & ring_mask]);
      const unsigned long long w0 = vd[0];
      const unsigned long long w1 = vd[1];
      const unsigned long long w2 = vd[2];
      const uint8_t* src = reinterpret_cast<const uint8_t*>(w0);
      uint8_t* dst = reinterpret_cast<uint8_t*>(w1);
      const uint32_t bytes = static_cast<uint32_t>(w2 & 0xFFFFFFFFu);
      // Uniform u64 same-slot tile path with agent-scope loads:
      // resident-kernel cross-XCD coherence, see the per-queue
      // kernel's copy loop.
      const uint32_t n8 = bytes >> 3;
      unsigned long long* lds64 =
          reinterpret_cast<unsigned long long*>(lds);
      if (src != nullptr) {
        const unsigned long long* __restrict__ s8 =
            reinterpret_cast<const unsigned long long*>(src);
#pragma unroll 4
        for (uint32_t k = lane; k < n8; k += 64) {
          lds64[k] = __hip_atomic_load(s8 + k, __ATOMIC_RELAXED,
                                       __HIP_MEMORY_SCOPE_AGENT);
        }
      } else {
        const uint32_t b = static_cast<uint32_t>(w2 >> 32) & 0xFF;
        const uint32_t word = b | (b << 8) | (b << 16) | (b << 24);
        const unsigned long long v =
            word | (static_cast<unsigned long long>(word) << 32);
        for (uint32_t k = lane; k < n8; k += 64) lds64[k] = v;
      }
      unsigned long long* __restrict__ d8 =
          reinterpret_cast<unsigned long long*>(dst);
#pragma unroll 4
      for (uint32_t k = lane; k < n8; k += 64) d8[k] = lds64[k];
      // Ordering carried by the system-scope release store; see the
      // per-queue kernel's publish comment (explicit
      // __threadfence_system hangs waves on current pool firmware).
      if (lane == 0) {
        volatile unsigned long long* cq =
            reinterpret_cast<volatile unsigned long long*>(slot_ld(slot, 2));
        __hip_atomic_store(&cq[claim & ring_mask], claim + 1,
                           __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_SYSTEM);
      }
      progress = true;
    }
    if (!progress) {
      if (__hip_atomic_load(ctl.exit_flag, __ATOMIC_RELAXED,
                            __HIP_MEMORY_SCOPE_AGENT) != 0) {
        return;
      }
      __builtin_amdgcn_s_sleep(16);
    }
  }
}

// --- liveness probe kernels -------------------------------------------------
//
// Minimal reproductions of the persistent-service idioms, used by
// persistent_probe() to pinpoint WHICH leg of the host<->GPU polling
// contract fails on a box (round-1 postmortem: the full service kernel
// sat resident serving nothing on fresh boxes, and the failing leg
// could not be identified from the outside):
//   leg 1: does a resident wave run at all? (heartbeat to device mem)
//   leg 2: GPU -> host visibility (heartbeat to pinned host mem)
//   leg 3: host -> GPU visibility (sees the host-written tail word)
//   leg 4: completion publish (system-release store to pinned CQ)
//   leg 5: cross-workgroup agent-scope visibility (leader->worker)

struct ProbeCtl {
  volatile unsigned long long* sq_tail;  // pinned (device view)
  volatile unsigned long long* hb_host;  // pinned (device view)
  volatile unsigned long long* cq;       // pinned (device view)
  unsigned long long* hb_dev;            // device
  unsigned long long* relay;             // device: leader->worker word
  unsigned long long* rmw_counter;       // device: worker fetch_add target
  uint32_t use_atomics;                  // 0 = volatile, 1 = scoped atomics
  uint32_t do_rmw;                       // worker does agent fetch_adds
  uint32_t do_fence;                     // __threadfence_system pre-publish
  uint32_t max_spins;
};

__global__ __launch_bounds__(64) void k_probe(ProbeCtl ctl) {
  __shared__ __attribute__((aligned(16))) uint8_t lds_raw[kTileBytes];
  const uint32_t lane = threadIdx.x & 63;
  (void)lds_raw;
  if (lane != 0) return;
  if (blockIdx.x != 0) {
    // Worker-style block: poll the leader's device-memory relay the
    // way service workers poll known_tail, then publish to cq[1].
    for (uint32_t i = 0; i < ctl.max_spins; ++i) {
      if (ctl.do_rmw) {
        (void)__hip_atomic_fetch_add(ctl.rmw_counter, 1ull,
                                     __ATOMIC_RELAXED,
                                     __HIP_MEMORY_SCOPE_AGENT);
      }
      unsigned long long v =
          ctl.use_atomics
              ? __hip_atomic_load(ctl.relay, __ATOMIC_RELAXED,
                                  __HIP_MEMORY_SCOPE_AGENT)
              : *reinterpret_cast<volatile unsigned long long*>(ctl.relay);
      if (v != 0) {
        __hip_atomic_store(&ctl.cq[1], v, __ATOMIC_RELEASE,
                           __HIP_MEMORY_SCOPE_SYSTEM);
        return;
      }
      __builtin_amdgcn_s_sleep(8);
    }
    return;
  }
  // Leader-style block: heartbeat + host-tail poll.
  for (uint32_t i = 0; i < ctl.max_spins; ++i) {
    if ((i & 1023) == 0) {
      const unsigned long long beat = i + 1;
      __hip_atomic_store(ctl.hb_dev, beat, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_AGENT);
      if (ctl.use_atomics) {
        __hip_atomic_store(ctl.hb_host, beat, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_SYSTEM);
      } else {
        *ctl.hb_host = beat;
      }
    }
    unsigned long long tail =
        ctl.use_atomics
            ? __hip_atomic_load(
                  const_cast<const unsigned long long*>(ctl.sq_tail),
                  __ATOMIC_ACQUIRE, __HIP_MEMORY_SCOPE_SYSTEM)
            : *ctl.sq_tail;
    if (tail != 0) {
      // Relay to the worker block (leader->worker handoff) and
      // publish the completion to the host.
      __hip_atomic_store(ctl.relay, tail, __ATOMIC_RELEASE,
                         __HIP_MEMORY_SCOPE_AGENT);
      if (ctl.do_fence) __threadfence_system();  // suspected wave-hang
      __hip_atomic_store(&ctl.cq[0], tail, __ATOMIC_RELEASE,
                         __HIP_MEMORY_SCOPE_SYSTEM);
      return;
    }
    __builtin_amdgcn_s_sleep(8);
  }
}

int g_device_count = -1;

// ROCm defaults to 4 hardware queues per process; more concurrent
// streams get gang-scheduled (preempted) onto them, which time-slices
// persistent service kernels — measured as a 30-50x multi-queue
// collapse with second-long p999s. Raise the cap before the HIP
// runtime initializes (no effect if the embedding process already
// initialized HIP; hipstored sets it in main() too). Users can
// override by exporting their own value.
struct HwQueueEnvInit {
  HwQueueEnvInit() {
    // HIPSTORE_NO_HWQ_INIT opts out (engine_diag.sh probes whether the
    // raised cap itself breaks service-kernel scheduling on a box).
    if (getenv("HIPSTORE_NO_HWQ_INIT") == nullptr) {
      setenv("GPU_MAX_HW_QUEUES", "24", /*overwrite=*/0);
    }
  }
};
HwQueueEnvInit g_hw_queue_env_init;

int device_count_cached() {
  static std::once_flag once;
  std::call_once(once, [] {
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) n = 0;
    g_device_count = n;
  });
  return g_device_count;
}

// Every host-side wait in this file is bounded (round-1 lesson: a
// resident-but-wedged service kernel turned two 30-minute driver
// budgets into GPU-busy hangs). Timeouts are env-tunable; on expiry
// the waiter FAILS the operation and, for teardown, LEAKS rings the
// GPU may still touch instead of freeing them (freeing pinned memory
// under a live kernel faults the device).
double env_seconds(const char* name, double fallback) {
  const char* env = getenv(name);
  if (env == nullptr) return fallback;
  const double v = atof(env);
  return v > 0 ? v : fallback;
}

double sync_io_timeout_s() {
  static const double s = env_seconds("HIPSTORE_SYNC_TIMEOUT_S", 30.0);
  return s;
}

double teardown_timeout_s() {
  static const double s = env_seconds("HIPSTORE_TEARDOWN_TIMEOUT_S", 10.0);
  return s;
}

// Bounded stream drain: true if the stream went idle (or errored —
// nothing left to wait for) within `seconds`.
bool stream_drain(hipStream_t stream, double seconds) {
  const auto deadline = std::chrono::steady_clock::now() +
                        std::chrono::duration<double>(seconds);
  while (true) {
    if (hipStreamQuery(stream) != hipErrorNotReady) return true;
    if (std::chrono::steady_clock::now() > deadline) return false;
    std::this_thread::sleep_for(std::chrono::microseconds(200));
  }
}

void dump_engine_stats(const char* why) {
  fprintf(stderr,
          "[hipstore] %s: persistent-engine stats: launches=%llu "
          "relaunches=%llu stall_queries=%llu\n",
          why,
          static_cast<unsigned long long>(g_pers_launches.load()),
          static_cast<unsigned long long>(g_pers_relaunches.load()),
          static_cast<unsigned long long>(g_pers_stall_queries.load()));
}

}  // namespace

uint64_t persistent_stat(int which) {
  switch (which) {
    case 0: return g_pers_launches.load();
    case 1: return g_pers_relaunches.load();
    case 2: return g_pers_stall_queries.load();
    default: return 0;
  }
}

bool gpu_available() { return device_count_cached() > 0; }
int gpu_device_count() { return device_count_cached(); }

std::string gpu_pci_address(int device) {
  char bus_id[32] = {0};
  if (hipDeviceGetPCIBusId(bus_id, sizeof(bus_id), device) != hipSuccess) {
    return "";
  }
  // hip returns "0000:c1:00.0" style already; normalize to lower-case.
  std::string s(bus_id);
  std::transform(s.begin(), s.end(), s.begin(), ::tolower);
  return s;
}

std::pair<uint64_t, uint64_t> hbm_info(int device) {
  if (!gpu_available() || device < 0 || device >= gpu_device_count()) {
    return {0, 0};
  }
  if (hipSetDevice(device) != hipSuccess) return {0, 0};
  size_t free_bytes = 0;
  size_t total_bytes = 0;
  if (hipMemGetInfo(&free_bytes, &total_bytes) != hipSuccess) return {0, 0};
  return {total_bytes, free_bytes};
}

void* alloc_pinned(size_t bytes) {
  if (!gpu_available()) return malloc(bytes);
  void* p = nullptr;
  HIP_CHECK(hipHostMalloc(&p, bytes, hipHostMallocMapped));
  return p;
}

void free_pinned(void* ptr) {
  if (ptr == nullptr) return;
  if (!gpu_available()) {
    free(ptr);
    return;
  }
  (void)hipHostFree(ptr);
}

// ---------------------------------------------------------------------------
// HBM bdev
// ---------------------------------------------------------------------------

namespace {

// Host pointer -> pointer the GPU may dereference.
template <typename T>
T* device_view(T* host_ptr) {
  void* dev = nullptr;
  HIP_CHECK(hipHostGetDevicePointer(&dev, const_cast<void*>(
      reinterpret_cast<const void*>(host_ptr)), 0));
  return reinterpret_cast<T*>(dev);
}

// Engine channels tag their kind so HbmBdev can dispatch without
// RTTI in the hot path (channels of different kinds coexist on one
// bdev when the auto-fallback kicks in).
class HbmChannelBase : public IoChannel {
 public:
  enum class Kind { kBatched, kPersistent, kShared };
  explicit HbmChannelBase(Kind kind) : kind(kind) {}
  const Kind kind;
};

struct PendingIo {
  IoRequest req;
  int status;
};

struct InflightBatch {
  hipEvent_t event = nullptr;
  uint32_t slots = 0;  // descriptor-ring slots to free on completion
  std::vector<std::pair<IoCompletion, int>> completions;
};

class HbmChannel : public HbmChannelBase {
 public:
  HbmChannel(int device, uint8_t* base, uint64_t size)
      : HbmChannelBase(Kind::kBatched), base_(base), size_(size) {
    HIP_CHECK(hipSetDevice(device));
    HIP_CHECK(hipStreamCreateWithFlags(&stream_, hipStreamNonBlocking));
    HIP_CHECK(hipHostMalloc(reinterpret_cast<void**>(&ring_),
                            kRingSlots * sizeof(BlockDesc), hipHostMallocMapped));
    ring_dev_ = device_view(ring_);
    device_ = device;
  }

  ~HbmChannel() override {
    // Drain so no kernel touches the ring after it is freed. Bounded:
    // if a launched batch never finishes, leak the ring (a kernel may
    // still read it) rather than hang or fault.
    if (!stream_drain(stream_, teardown_timeout_s())) {
      dump_engine_stats("batched-channel teardown timeout; leaking ring");
      return;
    }
    for (auto& batch : inflight_) (void)hipEventDestroy(batch.event);
    for (auto event : event_pool_) (void)hipEventDestroy(event);
    (void)hipHostFree(ring_);
    (void)hipStreamDestroy(stream_);
  }

  void enqueue(IoRequest req, int status) {
    pending_.push_back(PendingIo{std::move(req), status});
  }

  // Launch every pending request as one batch (if ring space allows).
  // Batches occupy CONTIGUOUS descriptor-ring slots [head, head+n) — the
  // kernel takes a flat pointer — so a batch never wraps: it is capped
  // at the space to the end of the ring, and the head snaps to 0 when
  // the ring is empty.
  void kick() {
    if (pending_.empty()) return;
    if (used_slots_ == 0) ring_head_ = 0;  // empty ring: reclaim tail space
    const uint32_t free_slots = kRingSlots - used_slots_;
    const uint32_t space_to_end = kRingSlots - ring_head_;
    const uint32_t cap =
        std::min({free_slots, space_to_end, kMaxBatchTiles});
    uint32_t n = 0;
    size_t take = 0;
    std::vector<std::pair<IoCompletion, int>> completions;
    for (const PendingIo& io : pending_) {
      uint32_t tiles = 0;
      if (io.status == kIoOk && io.req.op != IoOp::kFlush) {
        tiles = static_cast<uint32_t>(
            (io.req.length + kTileBytes - 1) / kTileBytes);
      }
      if (n + tiles > cap) break;
      n += tiles;
      ++take;
    }
    if (take == 0) return;  // no contiguous space yet; retire() will free it
    uint32_t slot = ring_head_;
    for (size_t k = 0; k < take; ++k) {
      PendingIo& io = pending_[k];
      if (io.status == kIoOk && io.req.op != IoOp::kFlush) {
        expand(io.req, &slot);
      }
      completions.emplace_back(std::move(io.req.on_complete), io.status);
    }
    pending_.erase(pending_.begin(), pending_.begin() + take);
    if (n > 0) {
      const uint32_t first = ring_head_;
      const uint32_t grid = (n + kWavesPerWg - 1) / kWavesPerWg;
      hipLaunchKernelGGL(k_copy_blocks, dim3(grid), dim3(kWavesPerWg * 64), 0,
                         stream_, ring_dev_ + first, n);
    }
    hipEvent_t event = get_event();
    HIP_CHECK(hipEventRecord(event, stream_));
    ring_head_ = (ring_head_ + n) % kRingSlots;
    used_slots_ += n;
    inflight_.push_back(InflightBatch{event, n, std::move(completions)});
  }

  int retire(bool wait) {
    int completed = 0;
    while (!inflight_.empty()) {
      InflightBatch& batch = inflight_.front();
      hipError_t st = wait ? hipEventSynchronize(batch.event)
                           : hipEventQuery(batch.event);
      if (st == hipErrorNotReady) break;
      if (st != hipSuccess) throw std::runtime_error("hip event failure");
      for (auto& [cb, status] : batch.completions) {
        if (cb) cb(status);
        ++completed;
      }
      used_slots_ -= batch.slots;
      event_pool_.push_back(batch.event);
      inflight_.pop_front();
    }
    return completed;
  }

  bool idle() const { return pending_.empty() && inflight_.empty(); }
  bool has_pending() const { return !pending_.empty(); }

 private:
  void expand(const IoRequest& req, uint32_t* slot) {
    // Split a request into <=4 KiB tiles, one ring descriptor each.
    // kick() capped the batch so [ring_head_, ring_head_+n) never wraps.
    uint64_t done = 0;
    while (done < req.length) {
      const uint32_t bytes = static_cast<uint32_t>(
          std::min<uint64_t>(kTileBytes, req.length - done));
      BlockDesc& d = ring_[*slot];
      if (req.op == IoOp::kRead) {
        d.src = base_ + req.offset + done;
        d.dst = buf_device(static_cast<uint8_t*>(req.buffer) + done);
      } else if (req.op == IoOp::kWrite) {
        d.src = buf_device(static_cast<uint8_t*>(req.buffer) + done);
        d.dst = base_ + req.offset + done;
      } else {  // kFill
        d.src = nullptr;
        d.dst = base_ + req.offset + done;
      }
      d.bytes = bytes;
      d.fill = req.fill;
      ++*slot;
      done += bytes;
    }
  }

  // The caller's buffer may be pinned (device-visible) or pageable.
  // Pinned path: translate; pageable would need a bounce buffer — the
  // daemon and bench always use alloc_pinned, so reject pageable here.
  uint8_t* buf_device(uint8_t* host) {
    void* dev = nullptr;
    hipError_t err = hipHostGetDevicePointer(&dev, host, 0);
    if (err != hipSuccess) {
      throw std::runtime_error(
          "hipstore: I/O buffer is not pinned host memory (use alloc_pinned)");
    }
    return static_cast<uint8_t*>(dev);
  }

  hipEvent_t get_event() {
    if (!event_pool_.empty()) {
      hipEvent_t e = event_pool_.back();
      event_pool_.pop_back();
      return e;
    }
    hipEvent_t e = nullptr;
    HIP_CHECK(hipEventCreateWithFlags(&e, hipEventDisableTiming));
    return e;
  }

  uint8_t* base_;
  [[maybe_unused]] uint64_t size_;  // geometry echo; bounds live in HbmBdev
  int device_ = 0;
  hipStream_t stream_ = nullptr;
  BlockDesc* ring_ = nullptr;      // pinned host
  BlockDesc* ring_dev_ = nullptr;  // GPU view of the ring
  uint32_t ring_head_ = 0;
  uint32_t used_slots_ = 0;
  std::vector<PendingIo> pending_;
  std::deque<InflightBatch> inflight_;
  std::vector<hipEvent_t> event_pool_;
};

// Host side of the persistent service kernel. One instance per channel.
// Live per-queue service kernels per device: beyond this, new
// persistent channels fall back to the shared service (each per-queue
// kernel needs a real hardware queue; GPU_MAX_HW_QUEUES=24 minus
// headroom for transient/utility streams).
inline std::atomic<int> g_per_queue_channels[64];
inline int per_queue_channel_cap() {
  static const int cap = [] {
    const char* env = getenv("HIPSTORE_PERQ_CAP");
    return env ? atoi(env) : 18;
  }();
  return cap;
}

// Which engine serves channels past the cap. Default "shared" since
// round 2: with the poll-dispatch fix in, the mixed-engine matrix
// (tools/wedge_experiments.sh) measures the shared-service fallback
// at p99 26-37 us vs 57-258 us for the batched fallback on the same
// shapes. HIPSTORE_FALLBACK=batched selects the old behavior.
inline bool fallback_is_shared() {
  static const bool shared = [] {
    const char* env = getenv("HIPSTORE_FALLBACK");
    return env == nullptr || strcmp(env, "batched") != 0;
  }();
  return shared;
}

class HbmPersistentChannel : public HbmChannelBase {
 public:
  static constexpr uint32_t kRing = 32768;      // descriptors (>= 2x max request tiles)
  static constexpr uint32_t kIdleSpins = 500000;  // ~1 s of s_sleep polling

  // Worker waves per queue (1-wave workgroups). 16 default; QD-deep
  // queues benefit from more (HIPSTORE_PERSISTENT_WORKERS overrides).
  static uint32_t workers() {
    static uint32_t w = [] {
      const char* env = getenv("HIPSTORE_PERSISTENT_WORKERS");
      int v = env ? atoi(env) : 16;
      return static_cast<uint32_t>(std::min(std::max(v, 1), 64));
    }();
    return w;
  }

  HbmPersistentChannel(int device, uint8_t* base)
      : HbmChannelBase(Kind::kPersistent), base_(base), device_(device) {
    // The creator (HbmBdev::get_channel) claimed the per-device slot
    // with fetch_add BEFORE constructing; release it in the dtor.
    HIP_CHECK(hipSetDevice(device));
    HIP_CHECK(hipStreamCreateWithFlags(&stream_, hipStreamNonBlocking));
    HIP_CHECK(hipHostMalloc(reinterpret_cast<void**>(&sq_),
                            kRing * sizeof(BlockDesc), hipHostMallocMapped));
    void* p_tail = nullptr;
    void* p_cq = nullptr;
    void* p_stop = nullptr;
    HIP_CHECK(hipHostMalloc(&p_tail, 8, hipHostMallocMapped));
    HIP_CHECK(hipHostMalloc(&p_cq, kRing * 8, hipHostMallocMapped));
    HIP_CHECK(hipHostMalloc(&p_stop, 4, hipHostMallocMapped));
    sq_tail_ = static_cast<volatile unsigned long long*>(p_tail);
    cq_ = static_cast<volatile unsigned long long*>(p_cq);
    stop_ = static_cast<volatile uint32_t*>(p_stop);
    *sq_tail_ = 0;
    *stop_ = 0;
    memset(const_cast<unsigned long long*>(cq_), 0, kRing * 8);
    // Device words: [0]=claim counter, [1]=known_tail, [2]=exit_flag.
    HIP_CHECK(hipMalloc(reinterpret_cast<void**>(&claim_ctr_), 24));
    HIP_CHECK(hipMemset(claim_ctr_, 0, 24));
    HIP_CHECK(hipStreamSynchronize(nullptr));  // memset before launch()
    desc_io_.resize(kRing, nullptr);
    launch();
  }

  ~HbmPersistentChannel() override {
    g_per_queue_channels[device_].fetch_sub(1, std::memory_order_relaxed);
    __atomic_store_n(const_cast<uint32_t*>(stop_), 1u, __ATOMIC_RELEASE);
    // A healthy service kernel sees `stop` within one poll tick; a
    // wedged one never exits, so the drain is bounded and a timeout
    // leaks the rings (the resident kernel may still load them).
    if (!stream_drain(stream_, teardown_timeout_s())) {
      dump_engine_stats(
          "persistent-channel teardown timeout; leaking rings");
      return;
    }
    (void)hipHostFree(sq_);
    (void)hipHostFree(const_cast<unsigned long long*>(sq_tail_));
    (void)hipHostFree(const_cast<unsigned long long*>(cq_));
    (void)hipHostFree(const_cast<uint32_t*>(stop_));
    (void)hipFree(claim_ctr_);
    (void)hipStreamDestroy(stream_);
    // Unfired completion callbacks (wedge teardown only): delete, never
    // call, so their captures are released without touching torn state.
    std::set<IoState*> leftovers(desc_io_.begin(), desc_io_.end());
    for (IoState* state : leftovers) delete state;
  }

  struct IoState {
    IoCompletion cb;
    uint32_t remaining;
    int status;
  };

  void enqueue(IoRequest req, int status) {
    if (status != kIoOk) {
      immediate_.emplace_back(std::move(req.on_complete), status);
      return;
    }
    if (req.op == IoOp::kFlush) {
      flushes_.push_back({tail_, std::move(req.on_complete)});
      return;
    }
    pending_.push_back(std::move(req));
    drain_pending();
  }

  int poll() {
    int completed = 0;
    // Swap out the immediate list before firing: a completion callback
    // may resubmit, and a failed resubmission appends to immediate_
    // again — mutating the vector mid-iteration.
    if (!immediate_.empty()) {
      auto batch = std::move(immediate_);
      immediate_.clear();
      for (auto& [cb, status] : batch) {
        if (cb) cb(status);
        ++completed;
      }
    }
    // Retire the contiguous completed prefix of the CQ.
    while (completed_ < tail_) {
      unsigned long long seq = __atomic_load_n(
          const_cast<const unsigned long long*>(&cq_[completed_ % kRing]),
          __ATOMIC_ACQUIRE);
      if (seq != completed_ + 1) break;
      IoState* state = desc_io_[completed_ % kRing];
      desc_io_[completed_ % kRing] = nullptr;
      ++completed_;
      if (state != nullptr && --state->remaining == 0) {
        if (state->cb) state->cb(state->status);
        delete state;
        ++completed;
      }
    }
    // Flush markers complete once their submission point is retired.
    while (!flushes_.empty() && flushes_.front().first <= completed_) {
      if (flushes_.front().second) flushes_.front().second(kIoOk);
      flushes_.pop_front();
      ++completed;
    }
    drain_pending();
    if (completed > 0) last_progress_ = std::chrono::steady_clock::now();
    // Liveness: if work is outstanding but nothing completed for a
    // while, the service kernel may have idle-exited in the submit
    // race window — relaunch from the completed prefix (descriptor
    // replay is idempotent). The 1 ms stall gate keeps hipStreamQuery
    // (a locked runtime call) out of the hot path: unthrottled, 8
    // polling threads serialized on it and collapsed throughput.
    if (completed_ < tail_) {
      const auto now = std::chrono::steady_clock::now();
      if (now - last_progress_ > std::chrono::milliseconds(1)) {
        g_pers_stall_queries.fetch_add(1, std::memory_order_relaxed);
        if (hipStreamQuery(stream_) == hipSuccess) {
          g_pers_relaunches.fetch_add(1, std::memory_order_relaxed);
          launch();
        }
        debug_dump_state();
        last_progress_ = now;  // gate the query itself to 1/ms
      }
    }
    return completed;
  }

  bool has_capacity(uint32_t tiles) const {
    return tail_ - completed_ + tiles <= kRing;
  }

  uint8_t* base() { return base_; }

 private:
  void launch() {
    g_pers_launches.fetch_add(1, std::memory_order_relaxed);
    (void)hipSetDevice(device_);
    // [claim, known_tail, exit_flag] reset to the completed prefix.
    // Default: a one-thread init kernel carries the values as kernel
    // arguments. The previous pageable-source hipMemcpyAsync on this
    // stream is kept behind HIPSTORE_LAUNCH_INIT=memcpy for A/B runs
    // (engine_diag found the service stream never advancing past it
    // on fresh boxes).
    static const bool use_memcpy = [] {
      const char* env = getenv("HIPSTORE_LAUNCH_INIT");
      return env != nullptr && strcmp(env, "memcpy") == 0;
    }();
    if (use_memcpy) {
      const unsigned long long init[3] = {completed_, completed_, 0};
      HIP_CHECK(hipMemcpyAsync(claim_ctr_, init, 24, hipMemcpyHostToDevice,
                               stream_));
    } else {
      hipLaunchKernelGGL(k_init_ctl, dim3(1), dim3(1), 0, stream_,
                         claim_ctr_, completed_, completed_, 0ull);
    }
    PersistentCtl ctl;
    ctl.sq = device_view(sq_);
    ctl.sq_tail = device_view(const_cast<unsigned long long*>(sq_tail_));
    ctl.cq = device_view(const_cast<unsigned long long*>(cq_));
    ctl.stop = device_view(const_cast<uint32_t*>(stop_));
    ctl.claim_counter = claim_ctr_;
    ctl.known_tail = claim_ctr_ + 1;
    ctl.exit_flag = reinterpret_cast<uint32_t*>(claim_ctr_ + 2);
    ctl.ring_mask = kRing - 1;
    ctl.idle_spins = kIdleSpins;
    hipLaunchKernelGGL(k_persistent_copy, dim3(workers() + 1), dim3(64), 0,
                       stream_, ctl);
    const hipError_t err = hipGetLastError();
    if (err != hipSuccess) {
      fprintf(stderr, "[hipstore] persistent service launch failed: %s\n",
              hipGetErrorString(err));
    }
    last_progress_ = std::chrono::steady_clock::now();
  }

  void drain_pending() {
    while (!pending_.empty()) {
      IoRequest& req = pending_.front();
      const uint32_t tiles = static_cast<uint32_t>(
          (req.length + kTileBytes - 1) / kTileBytes);
      if (!has_capacity(tiles)) return;
      auto* state = new IoState{std::move(req.on_complete), tiles, kIoOk};
      uint64_t done = 0;
      uint64_t t = tail_;
      while (done < req.length) {
        const uint32_t bytes = static_cast<uint32_t>(
            std::min<uint64_t>(kTileBytes, req.length - done));
        BlockDesc& d = sq_[t % kRing];
        if (req.op == IoOp::kRead) {
          d.src = base_ + req.offset + done;
          d.dst = buf_device(static_cast<uint8_t*>(req.buffer) + done);
        } else if (req.op == IoOp::kWrite) {
          d.src = buf_device(static_cast<uint8_t*>(req.buffer) + done);
          d.dst = base_ + req.offset + done;
        } else {
          d.src = nullptr;
          d.dst = base_ + req.offset + done;
        }
        d.bytes = bytes;
        d.fill = req.fill;
        desc_io_[t % kRing] = state;
        ++t;
        done += bytes;
      }
      tail_ = t;
      // Publish: descriptors before the tail, release order.
      __atomic_store_n(const_cast<unsigned long long*>(sq_tail_), tail_,
                       __ATOMIC_RELEASE);
      pending_.pop_front();
    }
  }

  uint8_t* buf_device(uint8_t* host) {
    void* dev = nullptr;
    hipError_t err = hipHostGetDevicePointer(&dev, host, 0);
    if (err != hipSuccess) {
      throw std::runtime_error(
          "hipstore: I/O buffer is not pinned host memory (use alloc_pinned)");
    }
    return static_cast<uint8_t*>(dev);
  }

  // HIPSTORE_DEBUG: once a second during a stall, read the device
  // control words back on a utility stream and print them — shows
  // whether the leader mirrored the tail and how far workers claimed.
  void debug_dump_state() {
    static const bool debug = getenv("HIPSTORE_DEBUG") != nullptr;
    if (!debug) return;
    const auto now = std::chrono::steady_clock::now();
    if (now - dbg_last_ < std::chrono::seconds(1)) return;
    dbg_last_ = now;
    if (dbg_stream_ == nullptr) {
      if (hipStreamCreateWithFlags(&dbg_stream_, hipStreamNonBlocking) !=
          hipSuccess) {
        return;
      }
    }
    unsigned long long words[3] = {0, 0, 0};
    (void)hipMemcpyAsync(words, claim_ctr_, 24, hipMemcpyDeviceToHost,
                         dbg_stream_);
    (void)hipStreamSynchronize(dbg_stream_);
    fprintf(stderr,
            "[hipstore-dbg] perq tail=%llu done=%llu sq_tail=%llu "
            "dev_claim=%llu dev_known=%llu dev_exit=%llu busy=%d\n",
            static_cast<unsigned long long>(tail_),
            static_cast<unsigned long long>(completed_), *sq_tail_,
            words[0], words[1], words[2],
            hipStreamQuery(stream_) == hipErrorNotReady);
  }

  std::chrono::steady_clock::time_point dbg_last_{};
  hipStream_t dbg_stream_ = nullptr;

  uint8_t* base_;
  int device_;
  hipStream_t stream_ = nullptr;
  BlockDesc* sq_ = nullptr;
  volatile unsigned long long* sq_tail_ = nullptr;
  volatile unsigned long long* cq_ = nullptr;
  volatile uint32_t* stop_ = nullptr;
  unsigned long long* claim_ctr_ = nullptr;
  std::chrono::steady_clock::time_point last_progress_;
  uint64_t tail_ = 0;
  uint64_t completed_ = 0;
  std::vector<IoState*> desc_io_;
  std::deque<IoRequest> pending_;
  std::deque<std::pair<uint64_t, IoCompletion>> flushes_;
  std::vector<std::pair<IoCompletion, int>> immediate_;
};

// Host side of the shared service kernel: one per device, multiplexing
// every shared-mode channel through the 64-slot table.
class SharedService {
 public:
  static constexpr uint32_t kIdleSpins = 500000;

  static std::shared_ptr<SharedService>& instance(int device) {
    static std::mutex mutex;
    static std::map<int, std::shared_ptr<SharedService>> services;
    std::lock_guard<std::mutex> lock(mutex);
    auto& service = services[device];
    if (!service) service = std::make_shared<SharedService>(device);
    return service;
  }

  explicit SharedService(int device) : device_(device) {
    dbg("svc:setdev");
    HIP_CHECK(hipSetDevice(device));
    HIP_CHECK(hipStreamCreateWithFlags(&stream_, hipStreamNonBlocking));
    // Slot config copies go on their OWN stream: an async copy queued
    // on the kernel's stream would wait behind the resident service
    // kernel and the slot would only activate after its idle-exit
    // (measured: first completions at 1.4-2.9 s).
    HIP_CHECK(hipStreamCreateWithFlags(&config_stream_, hipStreamNonBlocking));
    dbg("svc:malloc");
    HIP_CHECK(hipMalloc(reinterpret_cast<void**>(&slots_dev_),
                        kSharedSlots * sizeof(SharedSlot) + 8));
    dbg("svc:memset");
    HIP_CHECK(hipMemset(slots_dev_, 0, kSharedSlots * sizeof(SharedSlot) + 8));
    dbg("svc:nullsync");
    HIP_CHECK(hipStreamSynchronize(nullptr));  // memset before config copies
    dbg("svc:hostmalloc");
    exit_flag_dev_ = reinterpret_cast<uint32_t*>(slots_dev_ + kSharedSlots);
    void* p_stop = nullptr;
    HIP_CHECK(hipHostMalloc(&p_stop, 4, hipHostMallocMapped));
    stop_ = static_cast<volatile uint32_t*>(p_stop);
    *stop_ = 0;
    memset(mirror_, 0, sizeof(mirror_));
    memset(used_, 0, sizeof(used_));
    memset(prefix_, 0, sizeof(prefix_));
    // Warm both streams now so their hardware queues are allocated
    // HERE (bdev creation / first attach) and not while many channel
    // threads are concurrently creating queues and launching service
    // kernels — the mixed-engine wedge (profiles/README.md) is
    // consistent with a queue brought up in that storm never
    // dispatching its kernel.
    HIP_CHECK(hipMemsetAsync(exit_flag_dev_, 0, 4, stream_));
    HIP_CHECK(hipMemsetAsync(exit_flag_dev_, 0, 4, config_stream_));
    HIP_CHECK(hipStreamSynchronize(stream_));
    HIP_CHECK(hipStreamSynchronize(config_stream_));
    dbg("svc:ctor-done");
  }

  static void dbg(const char* what) {
    if (getenv("HIPSTORE_DEBUG") != nullptr) {
      fprintf(stderr, "[hipstore-dbg] %s\n", what);
    }
  }

  // The service lives for the process (kernels self-exit when idle, so
  // an abandoned service costs nothing); no teardown path needed.

  int attach(const BlockDesc* sq_dev, unsigned long long* tail_dev,
             unsigned long long* cq_dev, uint32_t ring_mask) {
    std::lock_guard<std::mutex> lock(mutex_);
    for (uint32_t i = 0; i < kSharedSlots; ++i) {
      if (used_[i]) continue;
      used_[i] = true;
      prefix_[i] = 0;
      SharedSlot& m = mirror_[i];
      m.sq = sq_dev;
      m.sq_tail = tail_dev;
      m.cq = cq_dev;
      m.claim = 0;
      m.known_tail = 0;
      m.ring_mask_active = ring_mask;  // active bit clear
      (void)hipSetDevice(device_);
      dbg("attach:copy");
      // Config words first, the active bit last (stream-ordered copies
      // so a running kernel never sees active before the pointers).
      HIP_CHECK(hipMemcpyAsync(&slots_dev_[i], &m, 40, hipMemcpyHostToDevice,
                               config_stream_));
      m.ring_mask_active = ring_mask | (1ull << 32);
      HIP_CHECK(hipMemcpyAsync(
          reinterpret_cast<unsigned long long*>(&slots_dev_[i]) + 5,
          &m.ring_mask_active, 8, hipMemcpyHostToDevice, config_stream_));
      dbg("attach:sync");
      HIP_CHECK(hipStreamSynchronize(config_stream_));
      dbg("attach:launch");
      ensure_running_locked(/*force_check=*/true);
      dbg("attach:done");
      return static_cast<int>(i);
    }
    throw std::runtime_error("shared service: no free ring slots (64 max)");
  }

  void detach(int slot) {
    std::lock_guard<std::mutex> lock(mutex_);
    // The owning channel drained first, so no worker can win a claim
    // on this slot; clearing the active bit stops sweeps touching it.
    (void)hipSetDevice(device_);
    mirror_[slot].ring_mask_active &= 0xFFFFFFFFull;
    (void)hipMemcpyAsync(
        reinterpret_cast<unsigned long long*>(&slots_dev_[slot]) + 5,
        &mirror_[slot].ring_mask_active, 8, hipMemcpyHostToDevice,
        config_stream_);
    (void)hipStreamSynchronize(config_stream_);
    used_[slot] = false;
  }

  void update_prefix(int slot, uint64_t completed) {
    // Racy-read tolerable: only consumed under mutex_ in relaunch.
    prefix_[slot] = completed;
  }

  // Called by channels when work is outstanding but completions stall:
  // relaunch the service kernel if it idle-exited (claims reset to each
  // channel's completed prefix; descriptor replay is idempotent).
  void ensure_running() {
    std::lock_guard<std::mutex> lock(mutex_);
    ensure_running_locked(false);
  }

 private:
  void ensure_running_locked(bool force_check) {
    const auto now = std::chrono::steady_clock::now();
    if (!force_check && now - last_check_ < std::chrono::milliseconds(1)) {
      return;
    }
    last_check_ = now;
    if (hipStreamQuery(stream_) != hipSuccess) return;  // still running
    (void)hipSetDevice(device_);
    for (uint32_t i = 0; i < kSharedSlots; ++i) {
      if (!used_[i]) continue;
      mirror_[i].claim = prefix_[i];
      mirror_[i].known_tail = prefix_[i];
    }
    HIP_CHECK(hipMemcpyAsync(slots_dev_, mirror_,
                             kSharedSlots * sizeof(SharedSlot),
                             hipMemcpyHostToDevice, stream_));
    HIP_CHECK(hipMemsetAsync(exit_flag_dev_, 0, 4, stream_));
    SharedCtl ctl;
    ctl.slots = slots_dev_;
    ctl.stop = device_view(const_cast<uint32_t*>(stop_));
    ctl.exit_flag = exit_flag_dev_;
    ctl.idle_spins = kIdleSpins;
    const char* env = getenv("HIPSTORE_SHARED_WORKERS");
    int workers = env ? atoi(env) : 48;
    workers = std::min(std::max(workers, 1), 255);
    hipLaunchKernelGGL(k_shared_service, dim3(workers + 1), dim3(64), 0,
                       stream_, ctl);
    if (getenv("HIPSTORE_DEBUG") != nullptr) {
      hipError_t launch_err = hipGetLastError();
      hipError_t query = hipStreamQuery(stream_);
      fprintf(stderr, "[hipstore-dbg] svc:launched err=%d query=%d\n",
              static_cast<int>(launch_err), static_cast<int>(query));
    }
    g_pers_launches.fetch_add(1, std::memory_order_relaxed);
  }

 public:
  // Debug: read a slot's device state back (claim, known_tail,
  // mask|active) — serialized on config_stream_ so it never touches
  // the service kernel's stream.
  void debug_read_slot(int slot, unsigned long long out[3]) {
    std::lock_guard<std::mutex> lock(mutex_);
    (void)hipSetDevice(device_);
    (void)hipMemcpyAsync(out,
                         reinterpret_cast<unsigned long long*>(
                             &slots_dev_[slot]) + 3,
                         24, hipMemcpyDeviceToHost, config_stream_);
    (void)hipStreamSynchronize(config_stream_);
  }

 private:

  int device_;
  hipStream_t stream_ = nullptr;
  hipStream_t config_stream_ = nullptr;
  SharedSlot* slots_dev_ = nullptr;
  uint32_t* exit_flag_dev_ = nullptr;
  volatile uint32_t* stop_ = nullptr;
  std::mutex mutex_;
  SharedSlot mirror_[kSharedSlots];
  bool used_[kSharedSlots];
  uint64_t prefix_[kSharedSlots];
  std::chrono::steady_clock::time_point last_check_{};
};

// Channel whose ring is serviced by the per-device shared kernel.
// Ring/completion bookkeeping mirrors HbmPersistentChannel; the
// difference is who runs the GPU side.
class HbmSharedChannel : public HbmChannelBase {
 public:
  static constexpr uint32_t kRing = 32768;

  HbmSharedChannel(int device, uint8_t* base)
      : HbmChannelBase(Kind::kShared), base_(base) {
    SharedService::dbg("chan:instance");
    service_ = SharedService::instance(device);
    SharedService::dbg("chan:hostmalloc");
    HIP_CHECK(hipSetDevice(device));
    HIP_CHECK(hipHostMalloc(reinterpret_cast<void**>(&sq_),
                            kRing * sizeof(BlockDesc), hipHostMallocMapped));
    void* p_tail = nullptr;
    void* p_cq = nullptr;
    HIP_CHECK(hipHostMalloc(&p_tail, 8, hipHostMallocMapped));
    HIP_CHECK(hipHostMalloc(&p_cq, kRing * 8, hipHostMallocMapped));
    sq_tail_ = static_cast<volatile unsigned long long*>(p_tail);
    cq_ = static_cast<volatile unsigned long long*>(p_cq);
    *sq_tail_ = 0;
    memset(const_cast<unsigned long long*>(cq_), 0, kRing * 8);
    desc_io_.resize(kRing, nullptr);
    SharedService::dbg("chan:attach");
    slot_ = service_->attach(
        device_view(sq_), device_view(const_cast<unsigned long long*>(sq_tail_)),
        device_view(const_cast<unsigned long long*>(cq_)), kRing - 1);
  }

  ~HbmSharedChannel() override {
    // Drain so no worker holds a claim below tail, then detach.
    const auto deadline = std::chrono::steady_clock::now() +
                          std::chrono::duration<double>(teardown_timeout_s());
    while (completed_ < tail_ &&
           std::chrono::steady_clock::now() < deadline) {
      poll();
    }
    service_->detach(slot_);
    std::set<IoState*> leftovers(desc_io_.begin(), desc_io_.end());
    for (IoState* state : leftovers) {
      if (state != nullptr) delete state;  // undrained multi-tile IOs
    }
    if (completed_ < tail_) {
      // The drain timed out: a wedged service worker may still hold a
      // claim into these rings — leak them rather than fault the GPU.
      dump_engine_stats("shared-channel teardown timeout; leaking rings");
      return;
    }
    (void)hipHostFree(sq_);
    (void)hipHostFree(const_cast<unsigned long long*>(sq_tail_));
    (void)hipHostFree(const_cast<unsigned long long*>(cq_));
  }

  struct IoState {
    IoCompletion cb;
    uint32_t remaining;
    int status;
  };

  void enqueue(IoRequest req, int status) {
    if (status != kIoOk) {
      immediate_.emplace_back(std::move(req.on_complete), status);
      return;
    }
    if (req.op == IoOp::kFlush) {
      flushes_.push_back({tail_, std::move(req.on_complete)});
      return;
    }
    pending_.push_back(std::move(req));
    drain_pending();
  }

  int poll() {
    static const bool debug_env = getenv("HIPSTORE_DEBUG") != nullptr;
    if (debug_env) {
      const auto now = std::chrono::steady_clock::now();
      if (now - dbg_last_ > std::chrono::seconds(2)) {
        dbg_last_ = now;
        unsigned long long st[3] = {0, 0, 0};
        service_->debug_read_slot(slot_, st);
        fprintf(stderr,
                "[hipstore-dbg] slot=%d tail=%llu done=%llu sqtail=%llu "
                "dev_claim=%llu dev_kt=%llu dev_ma=%llx\n",
                slot_, static_cast<unsigned long long>(tail_),
                static_cast<unsigned long long>(completed_), *sq_tail_,
                st[0], st[1], st[2]);
      }
    }
    int completed = 0;
    // Swap out the immediate list before firing: a completion callback
    // may resubmit, and a failed resubmission appends to immediate_
    // again — mutating the vector mid-iteration.
    if (!immediate_.empty()) {
      auto batch = std::move(immediate_);
      immediate_.clear();
      for (auto& [cb, status] : batch) {
        if (cb) cb(status);
        ++completed;
      }
    }
    while (completed_ < tail_) {
      unsigned long long seq = __atomic_load_n(
          const_cast<const unsigned long long*>(&cq_[completed_ % kRing]),
          __ATOMIC_ACQUIRE);
      if (seq != completed_ + 1) break;
      IoState* state = desc_io_[completed_ % kRing];
      desc_io_[completed_ % kRing] = nullptr;
      ++completed_;
      if (state != nullptr && --state->remaining == 0) {
        if (state->cb) state->cb(state->status);
        delete state;
        ++completed;
      }
    }
    if (completed > 0) {
      service_->update_prefix(slot_, completed_);
      last_progress_ = std::chrono::steady_clock::now();
    }
    while (!flushes_.empty() && flushes_.front().first <= completed_) {
      if (flushes_.front().second) flushes_.front().second(kIoOk);
      flushes_.pop_front();
      ++completed;
    }
    drain_pending();
    if (completed_ < tail_ &&
        std::chrono::steady_clock::now() - last_progress_ >
            std::chrono::milliseconds(1)) {
      service_->ensure_running();
      last_progress_ = std::chrono::steady_clock::now();
    }
    return completed;
  }

  bool has_capacity(uint32_t tiles) const {
    return tail_ - completed_ + tiles <= kRing;
  }

 private:
  void drain_pending() {
    while (!pending_.empty()) {
      IoRequest& req = pending_.front();
      const uint32_t tiles = static_cast<uint32_t>(
          (req.length + kTileBytes - 1) / kTileBytes);
      if (!has_capacity(tiles)) return;
      auto* state = new IoState{std::move(req.on_complete), tiles, kIoOk};
      uint64_t done = 0;
      uint64_t t = tail_;
      while (done < req.length) {
        const uint32_t bytes = static_cast<uint32_t>(
            std::min<uint64_t>(kTileBytes, req.length - done));
        BlockDesc& d = sq_[t % kRing];
        if (req.op == IoOp::kRead) {
          d.src = base_ + req.offset + done;
          d.dst = buf_device(static_cast<uint8_t*>(req.buffer) + done);
        } else if (req.op == IoOp::kWrite) {
          d.src = buf_device(static_cast<uint8_t*>(req.buffer) + done);
          d.dst = base_ + req.offset + done;
        } else {
          d.src = nullptr;
          d.dst = base_ + req.offset + done;
        }
        d.bytes = bytes;
        d.fill = req.fill;
        desc_io_[t % kRing] = state;
        ++t;
        done += bytes;
      }
      tail_ = t;
      __atomic_store_n(const_cast<unsigned long long*>(sq_tail_), tail_,
                       __ATOMIC_RELEASE);
      pending_.pop_front();
    }
  }

  uint8_t* buf_device(uint8_t* host) {
    void* dev = nullptr;
    hipError_t err = hipHostGetDevicePointer(&dev, host, 0);
    if (err != hipSuccess) {
      throw std::runtime_error(
          "hipstore: I/O buffer is not pinned host memory (use alloc_pinned)");
    }
    return static_cast<uint8_t*>(dev);
  }

  uint8_t* base_;
  std::shared_ptr<SharedService> service_;
  std::chrono::steady_clock::time_point dbg_last_{};
  int slot_ = -1;
  BlockDesc* sq_ = nullptr;
  volatile unsigned long long* sq_tail_ = nullptr;
  volatile unsigned long long* cq_ = nullptr;
  uint64_t tail_ = 0;
  uint64_t completed_ = 0;
  std::vector<IoState*> desc_io_;
  std::deque<IoRequest> pending_;
  std::deque<std::pair<uint64_t, IoCompletion>> flushes_;
  std::vector<std::pair<IoCompletion, int>> immediate_;
  std::chrono::steady_clock::time_point last_progress_{};
};

class HbmBdev : public Bdev {
 public:
  HbmBdev(const std::string& name, uint64_t block_size, uint64_t num_blocks,
          int device, bool persistent)
      : Bdev(name, "Malloc disk", block_size, num_blocks),
        device_(device),
        persistent_(persistent) {
    if (persistent_) {
      // HIPSTORE_SHARED=1 selects the shared per-device service kernel
      // (one HW queue total) instead of a kernel per channel.
      const char* env = getenv("HIPSTORE_SHARED");
      shared_ = env != nullptr && atoi(env) != 0;
    }
    HIP_CHECK(hipSetDevice(device));
    HIP_CHECK(hipMalloc(reinterpret_cast<void**>(&base_), size_bytes()));
    HIP_CHECK(hipMemset(base_, 0, size_bytes()));
    // hipMemset on the null stream is async with respect to the
    // engine's non-blocking streams: on a 64+ GiB bdev the zeroing
    // sweep was still running when the first writes landed and wiped
    // them (far offsets are memset last). Null-stream sync only — a
    // device-wide sync would block on live persistent service kernels.
    HIP_CHECK(hipStreamSynchronize(nullptr));
  }

  ~HbmBdev() override {
    (void)hipSetDevice(device_);
    (void)hipFree(base_);
  }

  int device() const { return device_; }
  void* device_base() override { return base_; }
  int gpu_device() const override { return device_; }

  // Engine channels capture base_ at creation, so the backing store
  // can only move while no channel is alive. resize() refuses
  // (kIoFailed = try again offline) when the count is non-zero.
  int resize(uint64_t new_num_blocks) override {
    std::lock_guard<std::mutex> lock(resize_mutex_);
    if (live_channels_->load(std::memory_order_acquire) != 0) {
      return kIoFailed;
    }
    const uint64_t old_bytes = size_bytes();
    const uint64_t new_bytes = new_num_blocks * block_size();
    if (new_bytes == old_bytes) return kIoOk;
    HIP_CHECK(hipSetDevice(device_));
    uint8_t* new_base = nullptr;
    if (hipMalloc(reinterpret_cast<void**>(&new_base), new_bytes) !=
        hipSuccess) {
      return kIoFailed;
    }
    hipStream_t stream = nullptr;
    HIP_CHECK(hipStreamCreateWithFlags(&stream, hipStreamNonBlocking));
    const uint64_t keep = std::min(old_bytes, new_bytes);
    if (new_bytes > old_bytes) {
      HIP_CHECK(hipMemsetAsync(new_base + keep, 0, new_bytes - keep, stream));
    }
    const uint64_t n16 = keep / 16;
    const uint32_t grid = static_cast<uint32_t>(
        std::min<uint64_t>(2048, (n16 + 255) / 256));
    hipLaunchKernelGGL(k_copy_range, dim3(grid), dim3(256), 0, stream,
                       reinterpret_cast<const float4*>(base_),
                       reinterpret_cast<float4*>(new_base), n16);
    hipError_t err = hipStreamSynchronize(stream);
    (void)hipStreamDestroy(stream);
    if (err != hipSuccess) {
      (void)hipFree(new_base);
      return kIoFailed;
    }
    (void)hipFree(base_);
    base_ = new_base;
    set_num_blocks(new_num_blocks);
    return kIoOk;
  }

  std::shared_ptr<IoChannel> get_channel() override {
    HIP_CHECK(hipSetDevice(device_));
    if (persistent_) {
      // Auto-fallback: per-queue service kernels give the best
      // latency but each needs a hardware queue; past the cap, new
      // channels fall back to the shared per-device service kernel
      // (p99 26-37 us in the mixed matrix vs 57-258 us for batched
      // fallbacks; round-1's "mixed-engine wedge" was a poll-dispatch
      // bug, fixed — see poll()). The slot is claimed atomically
      // BEFORE constructing — a check-then-create race let N
      // concurrent creators all pass the check and launch N > cap
      // service kernels, overrunning GPU_MAX_HW_QUEUES.
      if (shared_) {
        return track(std::make_shared<HbmSharedChannel>(device_, base_));
      }
      int prev = g_per_queue_channels[device_ & 63].fetch_add(
          1, std::memory_order_relaxed);
      if (prev < per_queue_channel_cap()) {
        return track(std::make_shared<HbmPersistentChannel>(device_, base_));
      }
      g_per_queue_channels[device_ & 63].fetch_sub(
          1, std::memory_order_relaxed);
      if (fallback_is_shared()) {
        return track(std::make_shared<HbmSharedChannel>(device_, base_));
      }
      return track(
          std::make_shared<HbmChannel>(device_, base_, size_bytes()));
    }
    return track(std::make_shared<HbmChannel>(device_, base_, size_bytes()));
  }

  // Wrap a channel so its lifetime is visible to resize(): the deleter
  // keeps the inner shared_ptr (and thus the channel) alive until the
  // caller drops the handle, then decrements the live count.
  template <typename T>
  std::shared_ptr<IoChannel> track(std::shared_ptr<T> channel) {
    auto counter = live_channels_;
    counter->fetch_add(1, std::memory_order_acq_rel);
    IoChannel* raw = channel.get();
    return std::shared_ptr<IoChannel>(
        raw, [channel, counter](IoChannel*) mutable {
          channel.reset();  // destroy the channel FIRST (drains rings)
          counter->fetch_sub(1, std::memory_order_acq_rel);
        });
  }

  void submit(IoChannel* ch, IoRequest req) override {
    int status = kIoOk;
    if (req.op != IoOp::kFlush) {
      // A request larger than half the descriptor ring could never form
      // a batch; reject it rather than deadlock the queue.
      if (!check_bounds(req) ||
          req.length > static_cast<uint64_t>(kMaxBatchTiles) * kTileBytes) {
        status = kIoInvalid;
      }
    }
    if (status == kIoOk) account(req);
    auto* channel = static_cast<HbmChannelBase*>(ch);
    switch (channel->kind) {
      case HbmChannelBase::Kind::kShared:
        static_cast<HbmSharedChannel*>(channel)->enqueue(std::move(req), status);
        break;
      case HbmChannelBase::Kind::kPersistent:
        static_cast<HbmPersistentChannel*>(channel)->enqueue(std::move(req),
                                                             status);
        break;
      case HbmChannelBase::Kind::kBatched:
        static_cast<HbmChannel*>(channel)->enqueue(std::move(req), status);
        break;
    }
  }

  int poll(IoChannel* ch) override {
    // Dispatch on the CHANNEL's kind, exactly like submit(): a
    // persistent bdev hands out batched/shared channels past the
    // per-queue cap, and polling those through bdev-level flags cast
    // them to the wrong class (round-1's "mixed-engine wedge": the
    // misdispatched poll read garbage — stalled batched fallbacks,
    // segfaulted shared ones — while the GPU side was healthy all
    // along). No HIP API calls on the persistent paths: poll() runs
    // in a tight loop on every submitter thread, and even
    // hipSetDevice takes the runtime's global lock — 4+ spinning
    // threads convoyed on it and starved each other.
    auto* channel = static_cast<HbmChannelBase*>(ch);
    switch (channel->kind) {
      case HbmChannelBase::Kind::kShared:
        return static_cast<HbmSharedChannel*>(channel)->poll();
      case HbmChannelBase::Kind::kPersistent:
        return static_cast<HbmPersistentChannel*>(channel)->poll();
      case HbmChannelBase::Kind::kBatched:
        break;
    }
    auto* batched = static_cast<HbmChannel*>(channel);
    int completed = batched->retire(/*wait=*/false);
    if (batched->has_pending()) {
      // Launches must come from the bdev's device; only pay the
      // runtime call when there is something to launch.
      (void)hipSetDevice(device_);
      batched->kick();
      completed += batched->retire(/*wait=*/false);
    }
    return completed;
  }

 private:
  uint8_t* base_ = nullptr;
  std::mutex resize_mutex_;
  std::shared_ptr<std::atomic<int>> live_channels_ =
      std::make_shared<std::atomic<int>>(0);
  int device_;
  bool persistent_;
  bool shared_ = false;
};

}  // namespace

BdevPtr create_hbm_bdev(const std::string& name, uint64_t block_size,
                        uint64_t num_blocks, int device, bool persistent) {
  if (!gpu_available()) {
    throw std::runtime_error("hipstore: no HIP device for HBM bdev");
  }
  if (device < 0 || device >= gpu_device_count()) {
    throw std::runtime_error("hipstore: bad device index");
  }
  if (persistent && getenv("HIPSTORE_EAGER_SHARED") != nullptr) {
    // Experimental (wedge bisection only, tools/wedge_experiments.sh):
    // bring the per-device shared service up NOW, in this
    // single-threaded moment, rather than lazily from whichever
    // channel thread first crosses the per-queue cap. Landed as the
    // default once and is the prime suspect for a whole-suite hang on
    // a fresh box (round-1 GPUTEST/BENCH rc=124), so it is opt-in
    // until proven on hardware.
    SharedService::instance(device);
  }
  return std::make_shared<HbmBdev>(name, block_size, num_blocks, device,
                                   persistent);
}

void crc32c_hbm_blocks(Bdev* bdev, uint64_t offset, uint32_t block_size,
                       uint32_t count, uint32_t* out) {
  if (block_size % 4 != 0) {
    throw std::runtime_error("crc32c: block_size must be a multiple of 4");
  }
  if (offset % bdev->block_size() != 0 ||
      offset + static_cast<uint64_t>(block_size) * count > bdev->size_bytes()) {
    throw std::runtime_error("crc32c: range out of bounds");
  }
  void* base = bdev->device_base();
  if (base == nullptr) {
    crc32c_cpu_fallback(bdev, offset, block_size, count, out);
    return;
  }
  HIP_CHECK(hipSetDevice(bdev->gpu_device()));
  // Thread-local stream + result buffer: hipHostMalloc and stream
  // creation cost ~1 ms each, which collapsed per-I/O digests (NVMe/TCP
  // targets call this once per large read). Own non-blocking stream +
  // stream sync, never a device-wide sync (it would block on live
  // persistent service kernels).
  struct DigestCtx {
    hipStream_t stream = nullptr;
    uint32_t* out = nullptr;
    uint32_t capacity = 0;

    ~DigestCtx() {
      if (stream != nullptr) (void)hipStreamDestroy(stream);
      free_pinned(out);
    }
  };
  thread_local DigestCtx ctx;
  if (ctx.stream == nullptr) {
    HIP_CHECK(hipStreamCreateWithFlags(&ctx.stream, hipStreamNonBlocking));
  }
  if (count > ctx.capacity) {
    free_pinned(ctx.out);
    ctx.capacity = std::max(count, std::max(1024u, ctx.capacity * 2));
    ctx.out = static_cast<uint32_t*>(alloc_pinned(ctx.capacity * 4));
  }
  const uint32_t grid = (count + 255) / 256;
  hipLaunchKernelGGL(k_crc32c_blocks, dim3(grid), dim3(256), 0, ctx.stream,
                     static_cast<const uint8_t*>(base) + offset, block_size,
                     count, device_view(ctx.out));
  HIP_CHECK(hipStreamSynchronize(ctx.stream));
  memcpy(out, ctx.out, count * 4);
}

int hbm_copy_sync(Bdev* src, uint64_t src_offset, Bdev* dst,
                  uint64_t dst_offset, uint64_t length) {
  uint8_t* src_base = static_cast<uint8_t*>(src->device_base());
  uint8_t* dst_base = static_cast<uint8_t*>(dst->device_base());
  if (src_base == nullptr || dst_base == nullptr) return kIoInvalid;
  if (length == 0 || length % 16 != 0 ||
      src_offset + length > src->size_bytes() ||
      dst_offset + length > dst->size_bytes()) {
    return kIoInvalid;
  }
  const int src_dev = src->gpu_device();
  const int dst_dev = dst->gpu_device();
  try {
    HIP_CHECK(hipSetDevice(dst_dev));
    hipStream_t stream = nullptr;
    HIP_CHECK(hipStreamCreateWithFlags(&stream, hipStreamNonBlocking));
    hipError_t err;
    if (src_dev != dst_dev) {
      // Cross-GPU: one xGMI peer copy (SDMA saturates the link).
      (void)hipSetDevice(src_dev);
      (void)hipDeviceEnablePeerAccess(dst_dev, 0);
      (void)hipSetDevice(dst_dev);
      (void)hipDeviceEnablePeerAccess(src_dev, 0);
      err = hipMemcpyPeerAsync(dst_base + dst_offset, dst_dev,
                               src_base + src_offset, src_dev, length,
                               stream);
    } else {
      // Same device: contiguous LDS-staged range copy, grid-stride
      // (2048 workgroups fill the chip; no per-tile descriptors).
      const uint64_t n16 = length / 16;
      const uint32_t grid = static_cast<uint32_t>(
          std::min<uint64_t>((n16 + 255) / 256, 2048));
      hipLaunchKernelGGL(k_copy_range, dim3(grid), dim3(256), 0, stream,
                         reinterpret_cast<const float4*>(src_base + src_offset),
                         reinterpret_cast<float4*>(dst_base + dst_offset),
                         n16);
      err = hipStreamSynchronize(stream);
    }
    if (err == hipSuccess) err = hipStreamSynchronize(stream);
    (void)hipStreamDestroy(stream);
    return err == hipSuccess ? kIoOk : kIoFailed;
  } catch (const std::exception&) {
    return kIoFailed;
  }
}

// ---------------------------------------------------------------------------
// Synchronous helpers
// ---------------------------------------------------------------------------

namespace {

// Large synchronous requests are chunked below the HBM queue's
// per-request cap (the descriptor ring holds 64 MiB of 4 KiB tiles).
constexpr uint64_t kSyncChunk = 32ull << 20;

int run_sync(Bdev* bdev, IoRequest req) {
  auto channel = bdev->get_channel();
  uint64_t done_bytes = 0;
  const uint64_t total = req.length;
  while (done_bytes < total || total == 0) {
    IoRequest chunk;
    chunk.op = req.op;
    chunk.fill = req.fill;
    chunk.offset = req.offset + done_bytes;
    chunk.length = std::min<uint64_t>(kSyncChunk, total - done_bytes);
    chunk.buffer = req.buffer == nullptr
                       ? nullptr
                       : static_cast<uint8_t*>(req.buffer) + done_bytes;
    // Heap state captured by value: on a timeout this frame returns
    // while the I/O is still outstanding, and a late completion (or
    // the channel teardown) must not touch dead stack slots.
    struct SyncState {
      std::atomic<int> result{kIoFailed};
      std::atomic<bool> done{false};
    };
    auto state = std::make_shared<SyncState>();
    chunk.on_complete = [state](int status) {
      state->result.store(status, std::memory_order_relaxed);
      state->done.store(true, std::memory_order_release);
    };
    const int op_code = static_cast<int>(chunk.op);
    bdev->submit(channel.get(), std::move(chunk));
    // Bounded wait: a wedged service kernel must fail the I/O in
    // seconds, not hang the caller at 99% GPU busy (round-1 failure
    // mode — both driver budgets burned on exactly this loop).
    const auto deadline = std::chrono::steady_clock::now() +
                          std::chrono::duration<double>(sync_io_timeout_s());
    while (!state->done.load(std::memory_order_acquire)) {
      bdev->poll(channel.get());
      if (std::chrono::steady_clock::now() > deadline) {
        fprintf(stderr,
                "[hipstore] sync I/O timeout after %.0fs "
                "(op=%d offset=%llu len=%llu)\n",
                sync_io_timeout_s(), op_code,
                static_cast<unsigned long long>(req.offset + done_bytes),
                static_cast<unsigned long long>(total));
        dump_engine_stats("sync I/O timeout");
        return kIoFailed;
      }
    }
    const int result = state->result.load(std::memory_order_relaxed);
    if (result != kIoOk || total == 0) return result;
    done_bytes += std::min<uint64_t>(kSyncChunk, total - done_bytes);
  }
  return kIoOk;
}

}  // namespace

int bdev_read_sync(Bdev* bdev, uint64_t offset, void* buf, uint64_t len) {
  IoRequest req;
  req.op = IoOp::kRead;
  req.offset = offset;
  req.length = len;
  req.buffer = buf;
  return run_sync(bdev, std::move(req));
}

int bdev_write_sync(Bdev* bdev, uint64_t offset, const void* buf, uint64_t len) {
  IoRequest req;
  req.op = IoOp::kWrite;
  req.offset = offset;
  req.length = len;
  req.buffer = const_cast<void*>(buf);
  return run_sync(bdev, std::move(req));
}

int bdev_fill_sync(Bdev* bdev, uint64_t offset, uint8_t value, uint64_t len) {
  IoRequest req;
  req.op = IoOp::kFill;
  req.offset = offset;
  req.length = len;
  req.fill = value;
  return run_sync(bdev, std::move(req));
}

// ---------------------------------------------------------------------------
// bdevperf harness
// ---------------------------------------------------------------------------

namespace {

// Per-queue latency accounting as a 1-us-resolution histogram: sorting
// per-I/O sample vectors between steps stalled the queues (~20% of a
// bench step at 10M IOPS went to std::sort while nothing was
// submitted). Exact to 1 us below the 65.5 ms clip.
struct LatHist {
  static constexpr uint32_t kBuckets = 65536;
  std::vector<uint32_t> buckets = std::vector<uint32_t>(kBuckets, 0);
  uint64_t count = 0;
  double sum = 0;
  uint32_t max = 0;

  void record(uint32_t us) {
    ++buckets[std::min(us, kBuckets - 1)];
    ++count;
    sum += us;
    if (us > max) max = us;
  }

  void reset() {
    std::fill(buckets.begin(), buckets.end(), 0);
    count = 0;
    sum = 0;
    max = 0;
  }

  void merge_into(LatHist& total) const {
    for (uint32_t i = 0; i < kBuckets; ++i) total.buckets[i] += buckets[i];
    total.count += count;
    total.sum += sum;
    total.max = std::max(total.max, max);
  }

  double percentile(double p) const {
    if (count == 0) return 0;
    const uint64_t target = static_cast<uint64_t>(p * (count - 1));
    uint64_t seen = 0;
    for (uint32_t i = 0; i < kBuckets; ++i) {
      seen += buckets[i];
      if (seen > target) return i;
    }
    return max;
  }

  void fill_result(PerfResult* r) const {
    if (count == 0) return;
    r->lat_avg_us = sum / count;
    r->lat_p50_us = percentile(0.50);
    r->lat_p90_us = percentile(0.90);
    r->lat_p99_us = percentile(0.99);
    r->lat_p999_us = percentile(0.999);
    r->lat_max_us = max;
  }
};

}  // namespace

namespace {

// Optional queue-thread affinity (HIPSTORE_AFFINITY_BASE=<core>): pin
// queue q to core base+q, SPDK-reactor style. Spinning submitter
// threads migrate under CFS otherwise, which adds multi-us jitter.
void maybe_pin_thread(int queue_index) {
  const char* env = getenv("HIPSTORE_AFFINITY_BASE");
  if (env == nullptr) return;
  const long ncpu = sysconf(_SC_NPROCESSORS_ONLN);
  const int core = (atoi(env) + queue_index) % static_cast<int>(ncpu);
  cpu_set_t set;
  CPU_ZERO(&set);
  CPU_SET(core, &set);
  (void)pthread_setaffinity_np(pthread_self(), sizeof(set), &set);
}

}  // namespace

PerfResult run_bdevperf(Bdev* bdev, const std::string& workload,
                        uint32_t io_size, uint32_t queue_depth,
                        int num_queues, double seconds, uint64_t max_ios) {
  using clock = std::chrono::steady_clock;
  const bool do_read = workload != "randwrite";
  const bool do_write = workload == "randwrite" || workload == "randrw";
  const uint64_t units = bdev->size_bytes() / io_size;
  if (units == 0) throw std::runtime_error("bdev smaller than io_size");

  struct QueueStats {
    uint64_t ios = 0;
    LatHist lat;
  };
  std::vector<QueueStats> stats(num_queues);
  std::vector<std::thread> threads;
  std::atomic<bool> failed{false};
  const auto t0 = clock::now();
  const auto deadline = t0 + std::chrono::duration<double>(seconds);
  const uint64_t per_queue_cap =
      max_ios ? (max_ios + num_queues - 1) / num_queues : 0;

  for (int q = 0; q < num_queues; ++q) {
    threads.emplace_back([&, q] {
      try {
        const bool dbg = getenv("HIPSTORE_DEBUG") != nullptr;
        maybe_pin_thread(q);
        auto channel = bdev->get_channel();
        if (dbg) fprintf(stderr, "[hipstore-dbg] q%d got-channel\n", q);
        uint8_t* buf = static_cast<uint8_t*>(
            alloc_pinned(static_cast<size_t>(io_size) * queue_depth));
        if (dbg) fprintf(stderr, "[hipstore-dbg] q%d pinned-buf\n", q);
        std::mt19937_64 rng(0x9E3779B97F4A7C15ULL ^ (q * 0x8DA6B343));
        QueueStats& st = stats[q];
        std::vector<clock::time_point> submit_ts(queue_depth);
        uint32_t inflight = 0;
        bool stopping = false;
        uint64_t submitted = 0;

        std::function<void(uint32_t)> submit_slot = [&](uint32_t slot) {
          IoRequest req;
          bool write = do_write && (!do_read || (rng() & 1));
          req.op = write ? IoOp::kWrite : IoOp::kRead;
          req.offset = (rng() % units) * io_size;
          req.length = io_size;
          req.buffer = buf + static_cast<size_t>(slot) * io_size;
          submit_ts[slot] = clock::now();
          req.on_complete = [&, slot](int status) {
            if (status != kIoOk) {
              // Abort the run on the first error (SPDK bdevperf
              // semantics): drain and stop rather than hammering a
              // dead/failing bdev for the rest of the duration.
              failed.store(true);
              stopping = true;
            }
            const auto now = clock::now();
            st.lat.record(static_cast<uint32_t>(std::min<int64_t>(
                std::chrono::duration_cast<std::chrono::microseconds>(
                    now - submit_ts[slot]).count(), UINT32_MAX)));
            ++st.ios;
            --inflight;
            if (!stopping) {
              submit_slot(slot);
              ++inflight;
              ++submitted;
            }
          };
          bdev->submit(channel.get(), std::move(req));
        };

        for (uint32_t slot = 0; slot < queue_depth; ++slot) {
          submit_slot(slot);
          ++inflight;
          ++submitted;
        }
        if (dbg) fprintf(stderr, "[hipstore-dbg] q%d submitted-initial\n", q);
        // Stall watchdog: if in-flight I/Os stop completing for the
        // sync-timeout window, abandon the queue (failed) instead of
        // spinning forever — checked only on empty polls, every 1024th.
        auto last_progress = clock::now();
        uint32_t empty_polls = 0;
        bool stalled = false;
        while (true) {
          if (bdev->poll(channel.get()) == 0) {
            __builtin_ia32_pause();  // spinning submitter hygiene
            if ((++empty_polls & 0x3FF) == 0 && inflight > 0 &&
                clock::now() - last_progress >
                    std::chrono::duration<double>(sync_io_timeout_s())) {
              stalled = true;
              break;
            }
          } else {
            last_progress = clock::now();
            empty_polls = 0;
          }
          if (!stopping &&
              (failed.load(std::memory_order_relaxed) ||
               clock::now() >= deadline ||
               (per_queue_cap && submitted >= per_queue_cap))) {
            stopping = true;
          }
          if (stopping && inflight == 0) break;
        }
        if (stalled) {
          fprintf(stderr,
                  "[hipstore] bdevperf q%d stalled (%u in flight); "
                  "abandoning queue\n", q, inflight);
          dump_engine_stats("bdevperf stall");
          failed.store(true);
          return;  // leak buf: outstanding descriptors reference it
        }
        free_pinned(buf);
      } catch (const std::exception&) {
        failed.store(true);
      }
    });
  }
  if (getenv("HIPSTORE_PERF_DEBUG") != nullptr) {
    // Diagnostics: report per-queue progress while workers run so a
    // wedged queue (vs a slow one) is identifiable from the outside.
    for (int tick = 0; tick < 600; ++tick) {
      std::this_thread::sleep_for(std::chrono::seconds(5));
      bool all_done = true;
      std::string line = "[bdevperf]";
      for (int q = 0; q < num_queues; ++q) {
        line += " q" + std::to_string(q) + "=" +
                std::to_string(stats[q].ios);
      }
      fprintf(stderr, "%s\n", line.c_str());
      for (auto& thread : threads) {
        if (thread.joinable()) all_done = false;
      }
      (void)all_done;
      if (clock::now() > deadline + std::chrono::seconds(15)) break;
    }
  }
  for (auto& thread : threads) thread.join();
  const double elapsed =
      std::chrono::duration<double>(clock::now() - t0).count();
  if (failed.load()) throw std::runtime_error("bdevperf: I/O failures");

  PerfResult result;
  result.seconds = elapsed;
  LatHist total;
  for (auto& st : stats) {
    result.io_count += st.ios;
    st.lat.merge_into(total);
  }
  result.iops = result.io_count / elapsed;
  result.throughput_mbps =
      result.io_count * static_cast<double>(io_size) / elapsed / 1e6;
  total.fill_result(&result);
  return result;
}

// ---------------------------------------------------------------------------
// PerfSession: persistent queues for stepped benchmarking
// ---------------------------------------------------------------------------

struct PerfSession::Impl {
  BdevPtr bdev;
  std::string workload;
  uint32_t io_size;
  uint32_t queue_depth;
  int num_queues;

  std::mutex mutex;
  std::condition_variable cv;
  uint64_t epoch = 0;           // incremented per step
  uint64_t per_queue_ios = 0;   // target for the current epoch
  bool stop = false;

  struct QueueStats {
    uint64_t done_epoch = 0;
    uint64_t ios = 0;
    LatHist lat;
    bool failed = false;
  };
  std::vector<QueueStats> stats;
  std::vector<std::thread> threads;

  void worker(int q) {
    using clock = std::chrono::steady_clock;
    try {
      maybe_pin_thread(q);
      auto channel = bdev->get_channel();
      uint8_t* buf = static_cast<uint8_t*>(
          alloc_pinned(static_cast<size_t>(io_size) * queue_depth));
      std::mt19937_64 rng(0x9E3779B97F4A7C15ULL ^ (q * 0x8DA6B343));
      const uint64_t units = bdev->size_bytes() / io_size;
      const bool do_read = workload != "randwrite";
      const bool do_write = workload == "randwrite" || workload == "randrw";
      std::vector<clock::time_point> submit_ts(queue_depth);
      QueueStats& st = stats[q];

      uint64_t my_epoch = 0;
      while (true) {
        {
          std::unique_lock<std::mutex> lock(mutex);
          cv.wait(lock, [&] { return stop || epoch > my_epoch; });
          if (stop) break;
          my_epoch = epoch;
        }
        const uint64_t target = per_queue_ios;
        uint64_t completed = 0, submitted = 0;
        uint32_t inflight = 0;
        st.lat.reset();

        std::function<void(uint32_t)> submit_slot = [&](uint32_t slot) {
          IoRequest req;
          bool write = do_write && (!do_read || (rng() & 1));
          req.op = write ? IoOp::kWrite : IoOp::kRead;
          req.offset = (rng() % units) * io_size;
          req.length = io_size;
          req.buffer = buf + static_cast<size_t>(slot) * io_size;
          submit_ts[slot] = clock::now();
          req.on_complete = [&, slot](int status) {
            if (status != kIoOk) st.failed = true;
            st.lat.record(static_cast<uint32_t>(std::min<int64_t>(
                std::chrono::duration_cast<std::chrono::microseconds>(
                    clock::now() - submit_ts[slot]).count(), UINT32_MAX)));
            ++completed;
            --inflight;
            if (submitted < target) {
              submit_slot(slot);
              ++inflight;
              ++submitted;
            }
          };
          bdev->submit(channel.get(), std::move(req));
        };

        const uint32_t initial =
            static_cast<uint32_t>(std::min<uint64_t>(queue_depth, target));
        for (uint32_t slot = 0; slot < initial; ++slot) {
          submit_slot(slot);
          ++inflight;
          ++submitted;
        }
        // Stall watchdog (checked on every 1024th empty poll): a
        // wedged engine fails the step in seconds instead of pinning
        // this thread — and step()'s cv.wait — forever.
        auto last_progress = clock::now();
        uint32_t empty_polls = 0;
        bool stalled = false;
        while (inflight > 0) {
          if (bdev->poll(channel.get()) == 0) {
            __builtin_ia32_pause();
            if ((++empty_polls & 0x3FF) == 0 &&
                clock::now() - last_progress >
                    std::chrono::duration<double>(sync_io_timeout_s())) {
              stalled = true;
              break;
            }
          } else {
            last_progress = clock::now();
            empty_polls = 0;
          }
        }
        if (stalled) {
          fprintf(stderr,
                  "[hipstore] perf-session q%d stalled (%u in flight); "
                  "failing the session\n", q, inflight);
          dump_engine_stats("perf-session stall");
          std::lock_guard<std::mutex> lock(mutex);
          st.failed = true;
          st.done_epoch = ~0ull;  // never blocks step()
          cv.notify_all();
          return;  // leak buf: outstanding descriptors reference it
        }
        st.ios = completed;
        {
          std::lock_guard<std::mutex> lock(mutex);
          st.done_epoch = my_epoch;
        }
        cv.notify_all();
      }
      free_pinned(buf);
    } catch (const std::exception&) {
      std::lock_guard<std::mutex> lock(mutex);
      stats[q].failed = true;
      stats[q].done_epoch = ~0ull;  // never blocks step()
      cv.notify_all();
    }
  }
};

PerfSession::PerfSession(BdevPtr bdev, std::string workload, uint32_t io_size,
                         uint32_t queue_depth, int num_queues)
    : impl_(new Impl) {
  impl_->bdev = std::move(bdev);
  impl_->workload = std::move(workload);
  impl_->io_size = io_size;
  impl_->queue_depth = queue_depth;
  impl_->num_queues = num_queues;
  impl_->stats.resize(num_queues);
  if (impl_->bdev->size_bytes() < io_size) {
    throw std::runtime_error("bdev smaller than io_size");
  }
  for (int q = 0; q < num_queues; ++q) {
    impl_->threads.emplace_back([this, q] { impl_->worker(q); });
  }
}

PerfSession::~PerfSession() {
  {
    std::lock_guard<std::mutex> lock(impl_->mutex);
    impl_->stop = true;
  }
  impl_->cv.notify_all();
  for (auto& t : impl_->threads) t.join();
}

PerfResult PerfSession::step(uint64_t total_ios) {
  using clock = std::chrono::steady_clock;
  auto& im = *impl_;
  const auto t0 = clock::now();
  uint64_t this_epoch;
  {
    std::lock_guard<std::mutex> lock(im.mutex);
    im.per_queue_ios = (total_ios + im.num_queues - 1) / im.num_queues;
    this_epoch = ++im.epoch;
  }
  im.cv.notify_all();
  {
    // Bounded: workers self-abort after the sync-timeout window when
    // their queue stalls, so this only expires if a worker thread is
    // itself stuck (e.g. inside a locked HIP runtime call).
    const auto step_deadline =
        std::chrono::duration<double>(2 * sync_io_timeout_s() + 30.0);
    std::unique_lock<std::mutex> lock(im.mutex);
    const bool done = im.cv.wait_for(lock, step_deadline, [&] {
      for (auto& st : im.stats) {
        if (st.done_epoch < this_epoch) return false;
      }
      return true;
    });
    if (!done) {
      dump_engine_stats("perf-session step timeout");
      throw std::runtime_error("perf session: step timed out (wedged queue)");
    }
  }
  const double elapsed = std::chrono::duration<double>(clock::now() - t0).count();

  PerfResult result;
  result.seconds = elapsed;
  LatHist total;
  bool failed = false;
  for (auto& st : im.stats) {
    result.io_count += st.ios;
    failed |= st.failed;
    st.lat.merge_into(total);
  }
  if (failed) throw std::runtime_error("perf session: I/O failures");
  result.iops = result.io_count / elapsed;
  result.throughput_mbps =
      result.io_count * static_cast<double>(im.io_size) / elapsed / 1e6;
  total.fill_result(&result);
  return result;
}

std::map<std::string, long long> persistent_probe(int device, int flags) {
  // flags: 1 = scoped atomics (else volatile), 2 = enqueue a
  // pageable-source hipMemcpyAsync on the KERNEL\'S stream right
  // before the launch (the service engine\'s launch() pattern),
  // 4 = workers hammer an agent-scope fetch_add, 8 = 17 workgroups
  // (the default service grid) instead of 2.
  std::map<std::string, long long> r;
  const bool use_atomics = flags & 1;
  const bool pre_memcpy = flags & 2;
  const bool do_rmw = flags & 4;
  const bool big_grid = flags & 8;
  const bool do_fence = flags & 16;
  r["flags"] = flags;
  HIP_CHECK(hipSetDevice(device));
  hipStream_t stream = nullptr;
  hipStream_t util = nullptr;
  HIP_CHECK(hipStreamCreateWithFlags(&stream, hipStreamNonBlocking));
  HIP_CHECK(hipStreamCreateWithFlags(&util, hipStreamNonBlocking));
  void* p_tail = nullptr;
  void* p_hb = nullptr;
  void* p_cq = nullptr;
  HIP_CHECK(hipHostMalloc(&p_tail, 8, hipHostMallocMapped));
  HIP_CHECK(hipHostMalloc(&p_hb, 8, hipHostMallocMapped));
  HIP_CHECK(hipHostMalloc(&p_cq, 16, hipHostMallocMapped));
  auto* tail = static_cast<volatile unsigned long long*>(p_tail);
  auto* hb = static_cast<volatile unsigned long long*>(p_hb);
  auto* cq = static_cast<volatile unsigned long long*>(p_cq);
  *tail = 0;
  *hb = 0;
  cq[0] = 0;
  cq[1] = 0;
  unsigned long long* dev_words = nullptr;  // [0]=hb_dev [1]=relay [2]=rmw
  HIP_CHECK(hipMalloc(reinterpret_cast<void**>(&dev_words), 24));
  HIP_CHECK(hipMemset(dev_words, 0, 24));
  HIP_CHECK(hipStreamSynchronize(nullptr));
  ProbeCtl ctl;
  ctl.sq_tail = device_view(const_cast<unsigned long long*>(tail));
  ctl.hb_host = device_view(const_cast<unsigned long long*>(hb));
  ctl.cq = device_view(const_cast<unsigned long long*>(cq));
  ctl.hb_dev = dev_words;
  ctl.relay = dev_words + 1;
  ctl.rmw_counter = dev_words + 2;
  ctl.use_atomics = use_atomics ? 1 : 0;
  ctl.do_rmw = do_rmw ? 1 : 0;
  ctl.do_fence = do_fence ? 1 : 0;
  ctl.max_spins = 4u << 20;  // ~4-8 s of polling
  if (pre_memcpy) {
    // The service launch() idiom under test: small H2D from a STACK
    // array, asynchronously, on the same stream the kernel follows on.
    const unsigned long long init[3] = {0, 0, 0};
    HIP_CHECK(hipMemcpyAsync(dev_words, init, 24, hipMemcpyHostToDevice,
                             stream));
  }
  hipLaunchKernelGGL(k_probe, dim3(big_grid ? 17 : 2), dim3(64), 0, stream,
                     ctl);
  r["launch_err"] = static_cast<long long>(hipGetLastError());
  using clock = std::chrono::steady_clock;
  const auto t0 = clock::now();
  std::this_thread::sleep_for(std::chrono::milliseconds(250));
  r["hb_host_early"] =
      static_cast<long long>(__atomic_load_n(hb, __ATOMIC_ACQUIRE));
  unsigned long long hb_dev_early = 0;
  (void)hipMemcpyAsync(&hb_dev_early, dev_words, 8, hipMemcpyDeviceToHost,
                       util);
  (void)hipStreamSynchronize(util);
  r["hb_dev_early"] = static_cast<long long>(hb_dev_early);
  __atomic_store_n(tail, 1ull, __ATOMIC_RELEASE);
  r["cq0_ms"] = -1;
  r["cq1_ms"] = -1;
  while (clock::now() - t0 < std::chrono::seconds(5)) {
    if (r["cq0_ms"] < 0 && __atomic_load_n(&cq[0], __ATOMIC_ACQUIRE) != 0) {
      r["cq0_ms"] = std::chrono::duration_cast<std::chrono::milliseconds>(
                        clock::now() - t0).count();
    }
    if (r["cq1_ms"] < 0 && __atomic_load_n(&cq[1], __ATOMIC_ACQUIRE) != 0) {
      r["cq1_ms"] = std::chrono::duration_cast<std::chrono::milliseconds>(
                        clock::now() - t0).count();
    }
    if (r["cq0_ms"] >= 0 && r["cq1_ms"] >= 0) break;
    std::this_thread::sleep_for(std::chrono::milliseconds(2));
  }
  r["hb_host_final"] =
      static_cast<long long>(__atomic_load_n(hb, __ATOMIC_ACQUIRE));
  unsigned long long dev_final[3] = {0, 0, 0};
  (void)hipMemcpyAsync(dev_final, dev_words, 24, hipMemcpyDeviceToHost,
                       util);
  (void)hipStreamSynchronize(util);
  r["hb_dev_final"] = static_cast<long long>(dev_final[0]);
  r["relay_final"] = static_cast<long long>(dev_final[1]);
  r["rmw_final"] = static_cast<long long>(dev_final[2]);
  const bool drained = stream_drain(stream, 10.0);
  r["stream_drained"] = drained;
  if (drained) {
    (void)hipHostFree(p_tail);
    (void)hipHostFree(p_hb);
    (void)hipHostFree(p_cq);
    (void)hipFree(dev_words);
    (void)hipStreamDestroy(stream);
  }  // else: leak — a resident wave may still touch these
  (void)hipStreamDestroy(util);
  return r;
}


std::map<std::string, long long> persistent_kernel_probe(int device,
                                                         int variant) {
  // Launches the REAL k_persistent_copy with a minimal hand-built
  // setup (same allocation sizes and ctl wiring as
  // HbmPersistentChannel::launch), serving exactly one descriptor:
  // variant 0 = pinned->pinned 64 B, 1 = pinned->HBM 4 KiB,
  // 2 = HBM->pinned 4 KiB. Separates "the kernel is broken" from
  // "the channel setup is broken".
  constexpr uint32_t kProbeRing = 32768;
  std::map<std::string, long long> r;
  r["variant"] = variant;
  HIP_CHECK(hipSetDevice(device));
  hipStream_t stream = nullptr;
  hipStream_t util = nullptr;
  HIP_CHECK(hipStreamCreateWithFlags(&stream, hipStreamNonBlocking));
  HIP_CHECK(hipStreamCreateWithFlags(&util, hipStreamNonBlocking));
  BlockDesc* sq = nullptr;
  HIP_CHECK(hipHostMalloc(reinterpret_cast<void**>(&sq),
                          kProbeRing * sizeof(BlockDesc),
                          hipHostMallocMapped));
  void* p_tail = nullptr;
  void* p_cq = nullptr;
  void* p_stop = nullptr;
  void* p_buf = nullptr;
  HIP_CHECK(hipHostMalloc(&p_tail, 8, hipHostMallocMapped));
  HIP_CHECK(hipHostMalloc(&p_cq, kProbeRing * 8, hipHostMallocMapped));
  HIP_CHECK(hipHostMalloc(&p_stop, 4, hipHostMallocMapped));
  HIP_CHECK(hipHostMalloc(&p_buf, 8192, hipHostMallocMapped));
  auto* tail = static_cast<volatile unsigned long long*>(p_tail);
  auto* cq = static_cast<volatile unsigned long long*>(p_cq);
  auto* stop = static_cast<volatile uint32_t*>(p_stop);
  auto* buf = static_cast<uint8_t*>(p_buf);
  *tail = 0;
  *stop = 0;
  memset(const_cast<unsigned long long*>(cq), 0, kProbeRing * 8);
  memset(buf, 0x5A, 4096);
  memset(buf + 4096, 0, 4096);
  uint8_t* hbm = nullptr;
  HIP_CHECK(hipMalloc(reinterpret_cast<void**>(&hbm), 8192));
  HIP_CHECK(hipMemset(hbm, 0xA7, 8192));
  unsigned long long* claim_ctr = nullptr;
  HIP_CHECK(hipMalloc(reinterpret_cast<void**>(&claim_ctr), 24));
  HIP_CHECK(hipMemset(claim_ctr, 0, 24));
  HIP_CHECK(hipStreamSynchronize(nullptr));
  PersistentCtl ctl;
  ctl.sq = device_view(sq);
  ctl.sq_tail = device_view(const_cast<unsigned long long*>(tail));
  ctl.cq = device_view(const_cast<unsigned long long*>(cq));
  ctl.stop = device_view(const_cast<uint32_t*>(stop));
  ctl.claim_counter = claim_ctr;
  ctl.known_tail = claim_ctr + 1;
  ctl.exit_flag = reinterpret_cast<uint32_t*>(claim_ctr + 2);
  ctl.ring_mask = kProbeRing - 1;
  ctl.idle_spins = 500000;
  hipLaunchKernelGGL(k_persistent_copy, dim3(17), dim3(64), 0, stream, ctl);
  r["launch_err"] = static_cast<long long>(hipGetLastError());
  // One descriptor.
  uint8_t* buf_dev = device_view(buf);
  BlockDesc& d = sq[0];
  switch (variant) {
    case 0:
      d.src = buf_dev;
      d.dst = buf_dev + 4096;
      d.bytes = 64;
      break;
    case 1:
      d.src = buf_dev;
      d.dst = hbm;
      d.bytes = 4096;
      break;
    default:
      d.src = hbm;
      d.dst = buf_dev + 4096;
      d.bytes = 4096;
      break;
  }
  d.fill = 0;
  __atomic_store_n(tail, 1ull, __ATOMIC_RELEASE);
  using clock = std::chrono::steady_clock;
  const auto t0 = clock::now();
  r["cq_ms"] = -1;
  while (clock::now() - t0 < std::chrono::seconds(4)) {
    if (__atomic_load_n(&cq[0], __ATOMIC_ACQUIRE) == 1) {
      r["cq_ms"] = std::chrono::duration_cast<std::chrono::milliseconds>(
                       clock::now() - t0).count();
      break;
    }
    std::this_thread::sleep_for(std::chrono::milliseconds(2));
  }
  unsigned long long words[3] = {0, 0, 0};
  (void)hipMemcpyAsync(words, claim_ctr, 24, hipMemcpyDeviceToHost, util);
  (void)hipStreamSynchronize(util);
  r["dev_claim"] = static_cast<long long>(words[0]);
  r["dev_known"] = static_cast<long long>(words[1]);
  r["dev_exit"] = static_cast<long long>(words[2]);
  r["data_ok"] = variant == 0   ? memcmp(buf, buf + 4096, 64) == 0
                 : variant == 2 ? buf[4096] == 0xA7 && buf[8191] == 0xA7
                                : -1;
  __atomic_store_n(stop, 1u, __ATOMIC_RELEASE);
  const bool drained = stream_drain(stream, 8.0);
  r["stream_drained"] = drained;
  if (drained) {
    (void)hipHostFree(sq);
    (void)hipHostFree(p_tail);
    (void)hipHostFree(p_cq);
    (void)hipHostFree(p_stop);
    (void)hipHostFree(p_buf);
    (void)hipFree(hbm);
    (void)hipFree(claim_ctr);
    (void)hipStreamDestroy(stream);
  }
  (void)hipStreamDestroy(util);
  return r;
}


}  // namespace hipstore
