#include "hipstore/composite.h"

#include <hip/hip_runtime.h>
#include <rccl/rccl.h>
#include <stdlib.h>

#include <deque>
#include <memory>
#include <mutex>
#include <set>
#include <stdexcept>

#include "hipstore/engine.h"

namespace hipstore {

namespace {

void validate_children(const std::vector<BdevPtr>& children, bool same_size) {
  if (children.empty()) throw std::runtime_error("composite: no children");
  for (const auto& child : children) {
    if (child->block_size() != children[0]->block_size()) {
      throw std::runtime_error("composite: mismatched block sizes");
    }
    if (same_size && child->num_blocks() != children[0]->num_blocks()) {
      throw std::runtime_error("composite: mismatched child sizes");
    }
  }
}

// Shared completion state for one parent request split across children.
struct SplitState {
  int remaining;
  int status = kIoOk;
  IoCompletion on_complete;

  void child_done(int child_status) {
    if (child_status != kIoOk && status == kIoOk) status = child_status;
    if (--remaining == 0 && on_complete) on_complete(status);
  }
};

class CompositeChannel : public IoChannel {
 public:
  std::vector<std::shared_ptr<IoChannel>> children;
  // Replication machinery (replicated bdev with HBM children only).
  hipStream_t rep_stream = nullptr;
  std::vector<hipEvent_t> event_pool;
  struct PendingRep {
    hipEvent_t event;
    std::shared_ptr<SplitState> state;
  };
  std::deque<PendingRep> pending_reps;

  ~CompositeChannel() override {
    if (rep_stream != nullptr) {
      (void)hipStreamSynchronize(rep_stream);
      for (auto& rep : pending_reps) (void)hipEventDestroy(rep.event);
      for (auto event : event_pool) (void)hipEventDestroy(event);
      (void)hipStreamDestroy(rep_stream);
    }
  }

  hipEvent_t get_event() {
    if (!event_pool.empty()) {
      hipEvent_t e = event_pool.back();
      event_pool.pop_back();
      return e;
    }
    hipEvent_t e = nullptr;
    if (hipEventCreateWithFlags(&e, hipEventDisableTiming) != hipSuccess) {
      throw std::runtime_error("composite: event creation failed");
    }
    return e;
  }

  int poll_reps() {
    int fired = 0;
    while (!pending_reps.empty()) {
      PendingRep& rep = pending_reps.front();
      hipError_t st = hipEventQuery(rep.event);
      if (st == hipErrorNotReady) break;
      rep.state->child_done(st == hipSuccess ? kIoOk : kIoFailed);
      event_pool.push_back(rep.event);
      pending_reps.pop_front();
      ++fired;
    }
    return fired;
  }
};

// ---------------------------------------------------------------------------
// Striped bdev
// ---------------------------------------------------------------------------

class StripedBdev : public Bdev {
 public:
  StripedBdev(const std::string& name, std::vector<BdevPtr> children,
              uint64_t stripe_size)
      : Bdev(name, "Striped Malloc disk", children[0]->block_size(),
             children[0]->num_blocks() * children.size()),
        children_(std::move(children)),
        stripe_size_(stripe_size) {}

  std::shared_ptr<IoChannel> get_channel() override {
    auto channel = std::make_shared<CompositeChannel>();
    for (const auto& child : children_) {
      channel->children.push_back(child->get_channel());
    }
    return channel;
  }

  void submit(IoChannel* ch, IoRequest req) override {
    auto* channel = static_cast<CompositeChannel*>(ch);
    if (req.op != IoOp::kFlush && !check_bounds(req)) {
      if (req.on_complete) req.on_complete(kIoInvalid);
      return;
    }
    account(req);
    // Split [offset, offset+length) at stripe boundaries; unit u maps
    // to child u % N at child offset (u / N) * stripe + intra.
    const uint64_t n = children_.size();
    auto state = std::make_shared<SplitState>();
    state->on_complete = std::move(req.on_complete);
    struct Piece {
      size_t child;
      IoRequest req;
    };
    std::vector<Piece> pieces;
    if (req.op == IoOp::kFlush) {
      for (size_t c = 0; c < n; ++c) {
        IoRequest sub;
        sub.op = IoOp::kFlush;
        pieces.push_back({c, std::move(sub)});
      }
    } else {
      uint64_t done = 0;
      while (done < req.length) {
        const uint64_t off = req.offset + done;
        const uint64_t unit = off / stripe_size_;
        const uint64_t intra = off % stripe_size_;
        const uint64_t span =
            std::min(req.length - done, stripe_size_ - intra);
        IoRequest sub;
        sub.op = req.op;
        sub.offset = (unit / n) * stripe_size_ + intra;
        sub.length = span;
        sub.fill = req.fill;
        sub.buffer = req.buffer == nullptr
                         ? nullptr
                         : static_cast<uint8_t*>(req.buffer) + done;
        pieces.push_back({static_cast<size_t>(unit % n), std::move(sub)});
        done += span;
      }
    }
    state->remaining = static_cast<int>(pieces.size());
    for (auto& piece : pieces) {
      piece.req.on_complete = [state](int status) {
        state->child_done(status);
      };
      children_[piece.child]->submit(
          channel->children[piece.child].get(), std::move(piece.req));
    }
  }

  int poll(IoChannel* ch) override {
    auto* channel = static_cast<CompositeChannel*>(ch);
    int completed = 0;
    for (size_t c = 0; c < children_.size(); ++c) {
      completed += children_[c]->poll(channel->children[c].get());
    }
    return completed;
  }

  const std::vector<BdevPtr>& children() const { return children_; }

 private:
  std::vector<BdevPtr> children_;
  uint64_t stripe_size_;
};

// ---------------------------------------------------------------------------
// Replicated bdev
// ---------------------------------------------------------------------------

// RCCL communicator clique over the replicas' devices + one stream per
// device, shared by the bdev's channels (serialized by a mutex: NCCL
// comms are not concurrency-safe). Broadcast fan-out beats a serial
// chain of peer copies from replica count >= 3: RCCL rings pipeline
// over several xGMI links instead of re-sending from the primary.
struct RcclClique {
  std::vector<ncclComm_t> comms;
  std::vector<hipStream_t> streams;
  std::vector<int> devices;
  std::mutex mutex;
  bool ok = false;

  explicit RcclClique(const std::vector<int>& devs) : devices(devs) {
    comms.resize(devs.size(), nullptr);
    if (ncclCommInitAll(comms.data(), static_cast<int>(devs.size()),
                        devs.data()) != ncclSuccess) {
      return;
    }
    streams.resize(devs.size(), nullptr);
    for (size_t i = 0; i < devs.size(); ++i) {
      if (hipSetDevice(devs[i]) != hipSuccess ||
          hipStreamCreateWithFlags(&streams[i], hipStreamNonBlocking) !=
              hipSuccess) {
        return;
      }
    }
    ok = true;
  }

  ~RcclClique() {
    for (auto s : streams) {
      if (s != nullptr) (void)hipStreamDestroy(s);
    }
    for (auto c : comms) {
      if (c != nullptr) (void)ncclCommDestroy(c);
    }
  }
};

class ReplicatedBdev : public Bdev {
 public:
  ReplicatedBdev(const std::string& name, std::vector<BdevPtr> children)
      : Bdev(name, "Replicated Malloc disk", children[0]->block_size(),
             children[0]->num_blocks()),
        children_(std::move(children)) {
    // xGMI fan-out path requires every child HBM-resident on a
    // distinct device with peer access enabled.
    peer_copy_ = true;
    std::set<int> devices;
    for (const auto& child : children_) {
      if (child->device_base() == nullptr) peer_copy_ = false;
      devices.insert(child->gpu_device());
    }
    if (devices.size() != children_.size()) peer_copy_ = false;
    if (peer_copy_) {
      const int primary = children_[0]->gpu_device();
      for (size_t i = 1; i < children_.size(); ++i) {
        (void)hipSetDevice(primary);
        hipError_t err =
            hipDeviceEnablePeerAccess(children_[i]->gpu_device(), 0);
        if (err != hipSuccess && err != hipErrorPeerAccessAlreadyEnabled) {
          peer_copy_ = false;
        }
        (void)hipSetDevice(children_[i]->gpu_device());
        err = hipDeviceEnablePeerAccess(primary, 0);
        if (err != hipSuccess && err != hipErrorPeerAccessAlreadyEnabled) {
          peer_copy_ = false;
        }
      }
    }
    // RCCL broadcast fan-out for >= 3 HBM replicas (HIPSTORE_RCCL=0
    // disables; peer copies remain the fallback and the 2-replica
    // path, where a single xGMI link is already optimal).
    const char* env = getenv("HIPSTORE_RCCL");
    if (peer_copy_ && children_.size() >= 3 && (!env || atoi(env) != 0)) {
      std::vector<int> devices;
      for (const auto& child : children_) {
        devices.push_back(child->gpu_device());
      }
      rccl_ = std::make_unique<RcclClique>(devices);
      if (!rccl_->ok) rccl_.reset();
    }
  }

  bool uses_rccl() const { return rccl_ != nullptr; }

  bool peer_copy() const { return peer_copy_; }

  std::shared_ptr<IoChannel> get_channel() override {
    auto channel = std::make_shared<CompositeChannel>();
    for (const auto& child : children_) {
      channel->children.push_back(child->get_channel());
    }
    if (peer_copy_) {
      (void)hipSetDevice(children_[0]->gpu_device());
      if (hipStreamCreateWithFlags(&channel->rep_stream,
                                   hipStreamNonBlocking) != hipSuccess) {
        throw std::runtime_error("replicated: stream creation failed");
      }
    }
    return channel;
  }

  void submit(IoChannel* ch, IoRequest req) override {
    auto* channel = static_cast<CompositeChannel*>(ch);
    if (req.op != IoOp::kFlush && !check_bounds(req)) {
      if (req.on_complete) req.on_complete(kIoInvalid);
      return;
    }
    account(req);
    const size_t n = children_.size();
    if (req.op == IoOp::kRead) {
      // Spread reads over replicas by stripe-ish hashing.
      const size_t child = (req.offset / (1 << 20)) % n;
      children_[child]->submit(channel->children[child].get(),
                               std::move(req));
      return;
    }
    if (req.op == IoOp::kFlush) {
      auto state = std::make_shared<SplitState>();
      state->on_complete = std::move(req.on_complete);
      state->remaining = static_cast<int>(n);
      for (size_t c = 0; c < n; ++c) {
        IoRequest sub;
        sub.op = IoOp::kFlush;
        sub.on_complete = [state](int status) { state->child_done(status); };
        children_[c]->submit(channel->children[c].get(), std::move(sub));
      }
      return;
    }
    // Write / fill.
    auto state = std::make_shared<SplitState>();
    state->on_complete = std::move(req.on_complete);
    if (peer_copy_ && req.op == IoOp::kWrite) {
      // Host -> primary HBM once, then primary -> replicas over xGMI:
      // RCCL ring broadcast for >= 3 replicas, direct peer copies
      // otherwise.
      state->remaining = static_cast<int>(n);  // primary + n-1 replicas
      IoRequest primary = req;
      const uint64_t offset = req.offset;
      const uint64_t length = req.length;
      primary.on_complete = [this, channel, state, offset, length](int status) {
        if (status != kIoOk) {
          // Primary failed: replicas would hold stale data; fail them.
          for (size_t i = 0; i < children_.size(); ++i) {
            state->child_done(status);
          }
          return;
        }
        state->child_done(kIoOk);
        if (rccl_ != nullptr && broadcast_rccl(channel, state, offset, length)) {
          return;
        }
        (void)hipSetDevice(children_[0]->gpu_device());
        uint8_t* src =
            static_cast<uint8_t*>(children_[0]->device_base()) + offset;
        for (size_t i = 1; i < children_.size(); ++i) {
          uint8_t* dst =
              static_cast<uint8_t*>(children_[i]->device_base()) + offset;
          hipError_t err = hipMemcpyPeerAsync(
              dst, children_[i]->gpu_device(), src,
              children_[0]->gpu_device(), length, channel->rep_stream);
          hipEvent_t event = channel->get_event();
          if (err != hipSuccess ||
              hipEventRecord(event, channel->rep_stream) != hipSuccess) {
            channel->event_pool.push_back(event);
            state->child_done(kIoFailed);
            continue;
          }
          channel->pending_reps.push_back({event, state});
        }
      };
      children_[0]->submit(channel->children[0].get(), std::move(primary));
      return;
    }
    // CPU fallback / fill: mirror the request to every child.
    state->remaining = static_cast<int>(n);
    for (size_t c = 0; c < n; ++c) {
      IoRequest sub = req;
      sub.on_complete = [state](int status) { state->child_done(status); };
      children_[c]->submit(channel->children[c].get(), std::move(sub));
    }
  }

  int poll(IoChannel* ch) override {
    auto* channel = static_cast<CompositeChannel*>(ch);
    int completed = 0;
    for (size_t c = 0; c < children_.size(); ++c) {
      completed += children_[c]->poll(channel->children[c].get());
    }
    completed += channel->poll_reps();
    return completed;
  }

  const std::vector<BdevPtr>& children() const { return children_; }

 private:
  // One ncclBroadcast across the clique (root = primary), completion
  // tracked per replica stream. Returns false to fall back to peer
  // copies (state untouched except on success).
  bool broadcast_rccl(CompositeChannel* channel,
                      const std::shared_ptr<SplitState>& state,
                      uint64_t offset, uint64_t length) {
    std::lock_guard<std::mutex> lock(rccl_->mutex);
    if (ncclGroupStart() != ncclSuccess) return false;
    bool ok = true;
    for (size_t i = 0; i < children_.size(); ++i) {
      void* buf = static_cast<uint8_t*>(children_[i]->device_base()) + offset;
      if (ncclBroadcast(buf, buf, length, ncclChar, /*root=*/0,
                        rccl_->comms[i], rccl_->streams[i]) != ncclSuccess) {
        ok = false;
      }
    }
    if (ncclGroupEnd() != ncclSuccess || !ok) return false;
    // Completion: one event per REPLICA stream (the primary already
    // holds the data; its stream entry just orders the collective).
    for (size_t i = 1; i < children_.size(); ++i) {
      (void)hipSetDevice(children_[i]->gpu_device());
      hipEvent_t event = channel->get_event();
      if (hipEventRecord(event, rccl_->streams[i]) != hipSuccess) {
        channel->event_pool.push_back(event);
        state->child_done(kIoFailed);
        continue;
      }
      channel->pending_reps.push_back({event, state});
    }
    return true;
  }

  std::vector<BdevPtr> children_;
  bool peer_copy_ = false;
  std::unique_ptr<RcclClique> rccl_;
};

}  // namespace

BdevPtr create_striped_bdev(const std::string& name,
                            std::vector<BdevPtr> children,
                            uint64_t stripe_size) {
  validate_children(children, /*same_size=*/true);
  if (stripe_size == 0 || stripe_size % children[0]->block_size() != 0) {
    throw std::runtime_error(
        "stripe size must be a positive multiple of the block size");
  }
  if (children[0]->size_bytes() % stripe_size != 0) {
    throw std::runtime_error("child size must be a multiple of stripe size");
  }
  return std::make_shared<StripedBdev>(name, std::move(children), stripe_size);
}

BdevPtr create_replicated_bdev(const std::string& name,
                               std::vector<BdevPtr> children) {
  validate_children(children, /*same_size=*/true);
  return std::make_shared<ReplicatedBdev>(name, std::move(children));
}

}  // namespace hipstore
