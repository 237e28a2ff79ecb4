// Kernel NBD export (reference lib/nbd/nbd.c rebuilt for hipstored).
//
// A socketpair connects the kernel's NBD block device to an in-process
// server thread; READ/WRITE payloads move through a pinned bounce
// buffer so the HBM bdev's GPU copy path applies. One pump thread per
// export (the reference polls from a reactor; here the kernel socket
// read is the natural wait point).

#include "hipstore/nbd.h"

#include <endian.h>
#include <fcntl.h>
#include <linux/nbd.h>
#include <sys/ioctl.h>
#include <sys/socket.h>
#include <sys/stat.h>
#include <unistd.h>

#include <cstring>
#include <map>
#include <memory>
#include <mutex>
#include <stdexcept>
#include <thread>

#include "hipstore/bdev.h"
#include "hipstore/engine.h"
#include "hipstore/rpc.h"

namespace hipstore {

namespace {

constexpr uint32_t kNbdRequestMagic = 0x25609513;
constexpr uint32_t kNbdReplyMagic = 0x67446698;

struct NbdExport {
  std::string bdev_name;
  std::string device;
  int dev_fd = -1;
  int kernel_sock = -1;
  int server_sock = -1;
  std::thread doit_thread;   // blocked in ioctl(NBD_DO_IT)
  std::thread pump_thread;   // serves the NBD protocol
};

struct NbdState {
  std::mutex mutex;
  std::map<std::string, std::unique_ptr<NbdExport>> exports;  // by device
};

NbdState& state() {
  static NbdState s;
  return s;
}

void pump(NbdExport* ex, BdevPtr bdev) {
  auto channel = bdev->get_channel();
  const size_t kMaxIo = 1 << 20;
  uint8_t* bounce = static_cast<uint8_t*>(alloc_pinned(kMaxIo));
  struct nbd_request req;
  struct nbd_reply reply;
  memset(&reply, 0, sizeof(reply));
  reply.magic = htobe32(kNbdReplyMagic);

  auto read_full = [&](void* buf, size_t n) -> bool {
    uint8_t* p = static_cast<uint8_t*>(buf);
    while (n > 0) {
      ssize_t r = read(ex->server_sock, p, n);
      if (r <= 0) return false;
      p += r;
      n -= r;
    }
    return true;
  };
  auto write_full = [&](const void* buf, size_t n) -> bool {
    const uint8_t* p = static_cast<const uint8_t*>(buf);
    while (n > 0) {
      ssize_t w = write(ex->server_sock, p, n);
      if (w <= 0) return false;
      p += w;
      n -= w;
    }
    return true;
  };

  while (read_full(&req, sizeof(req))) {
    if (be32toh(req.magic) != kNbdRequestMagic) break;
    const uint32_t type = be32toh(req.type) & 0xFFFF;
    const uint64_t from = be64toh(req.from);
    const uint32_t len = be32toh(req.len);
    memcpy(reply.handle, req.handle, sizeof(reply.handle));
    uint32_t error = 0;

    if (type == NBD_CMD_DISC) break;
    if (len > kMaxIo) {
      error = 22;  // EINVAL
      if (type == NBD_CMD_WRITE) {
        // Drain the payload we cannot take.
        size_t left = len;
        while (left > 0) {
          size_t n = std::min(left, kMaxIo);
          if (!read_full(bounce, n)) goto out;
          left -= n;
        }
      }
      reply.error = htobe32(error);
      if (!write_full(&reply, sizeof(reply))) break;
      continue;
    }

    switch (type) {
      case NBD_CMD_READ: {
        int status = bdev_read_sync(bdev.get(), from, bounce, len);
        reply.error = htobe32(status == kIoOk ? 0 : 5);
        if (!write_full(&reply, sizeof(reply))) goto out;
        if (status == kIoOk && !write_full(bounce, len)) goto out;
        break;
      }
      case NBD_CMD_WRITE: {
        if (!read_full(bounce, len)) goto out;
        int status = bdev_write_sync(bdev.get(), from, bounce, len);
        reply.error = htobe32(status == kIoOk ? 0 : 5);
        if (!write_full(&reply, sizeof(reply))) goto out;
        break;
      }
      case NBD_CMD_FLUSH: {
        reply.error = 0;
        if (!write_full(&reply, sizeof(reply))) goto out;
        break;
      }
      case NBD_CMD_TRIM: {
        int status = bdev_fill_sync(bdev.get(), from, 0, len);
        reply.error = htobe32(status == kIoOk ? 0 : 5);
        if (!write_full(&reply, sizeof(reply))) goto out;
        break;
      }
      default: {
        reply.error = htobe32(22);
        if (!write_full(&reply, sizeof(reply))) goto out;
        break;
      }
    }
  }
out:
  free_pinned(bounce);
}

}  // namespace

bool nbd_available() {
  struct stat st;
  return stat("/dev/nbd0", &st) == 0;
}

void nbd_start(const std::string& bdev_name, const std::string& device) {
  BdevPtr bdev = BdevManager::instance().find(bdev_name);
  if (!bdev) throw RpcError{kInvalidParams, "bdev " + bdev_name + " does not exist"};

  auto& s = state();
  std::lock_guard<std::mutex> lock(s.mutex);
  if (s.exports.count(device)) {
    throw RpcError{kInvalidParams, device + " already in use"};
  }

  auto ex = std::make_unique<NbdExport>();
  ex->bdev_name = bdev_name;
  ex->device = device;
  ex->dev_fd = open(device.c_str(), O_RDWR);
  if (ex->dev_fd < 0) {
    throw RpcError{kInternalError, "cannot open " + device +
                   " (nbd kernel module missing?)"};
  }
  int socks[2];
  if (socketpair(AF_UNIX, SOCK_STREAM, 0, socks) != 0) {
    close(ex->dev_fd);
    throw RpcError{kInternalError, "socketpair failed"};
  }
  ex->kernel_sock = socks[0];
  ex->server_sock = socks[1];

  if (ioctl(ex->dev_fd, NBD_SET_BLKSIZE, bdev->block_size()) != 0 ||
      ioctl(ex->dev_fd, NBD_SET_SIZE_BLOCKS,
            bdev->size_bytes() / bdev->block_size()) != 0 ||
      ioctl(ex->dev_fd, NBD_CLEAR_SOCK) != 0 ||
      ioctl(ex->dev_fd, NBD_SET_SOCK, ex->kernel_sock) != 0) {
    close(ex->dev_fd);
    close(socks[0]);
    close(socks[1]);
    throw RpcError{kInternalError, "NBD ioctl setup failed for " + device};
  }

  NbdExport* raw = ex.get();
  ex->doit_thread = std::thread([raw] {
    // Blocks until NBD_DISCONNECT / socket close.
    (void)ioctl(raw->dev_fd, NBD_DO_IT);
    (void)ioctl(raw->dev_fd, NBD_CLEAR_QUE);
    (void)ioctl(raw->dev_fd, NBD_CLEAR_SOCK);
  });
  ex->pump_thread = std::thread([raw, bdev] { pump(raw, bdev); });
  s.exports[device] = std::move(ex);
}

namespace {

void stop_export_locked(std::unique_ptr<NbdExport> ex) {
  (void)ioctl(ex->dev_fd, NBD_DISCONNECT);
  shutdown(ex->server_sock, SHUT_RDWR);
  close(ex->server_sock);
  if (ex->pump_thread.joinable()) ex->pump_thread.join();
  if (ex->doit_thread.joinable()) ex->doit_thread.join();
  close(ex->kernel_sock);
  close(ex->dev_fd);
}

}  // namespace

void nbd_stop(const std::string& device) {
  auto& s = state();
  std::unique_ptr<NbdExport> ex;
  {
    std::lock_guard<std::mutex> lock(s.mutex);
    auto it = s.exports.find(device);
    if (it == s.exports.end()) {
      throw RpcError{kInvalidParams, device + " is not exported"};
    }
    ex = std::move(it->second);
    s.exports.erase(it);
  }
  stop_export_locked(std::move(ex));
}

std::vector<std::pair<std::string, std::string>> nbd_list() {
  auto& s = state();
  std::lock_guard<std::mutex> lock(s.mutex);
  std::vector<std::pair<std::string, std::string>> out;
  for (const auto& [device, ex] : s.exports) {
    out.emplace_back(ex->bdev_name, device);
  }
  return out;
}

void nbd_stop_all() {
  auto& s = state();
  std::map<std::string, std::unique_ptr<NbdExport>> exports;
  {
    std::lock_guard<std::mutex> lock(s.mutex);
    exports.swap(s.exports);
  }
  for (auto& [_, ex] : exports) stop_export_locked(std::move(ex));
}

}  // namespace hipstore
