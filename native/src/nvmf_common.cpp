#include "nvmf_common.h"

#include <sys/socket.h>
#include <sys/uio.h>
#include <unistd.h>

#include <cerrno>

namespace hipstore {
namespace nvmf {

bool read_exact(int fd, void* buf, size_t n) {
  uint8_t* p = static_cast<uint8_t*>(buf);
  while (n > 0) {
    ssize_t r = recv(fd, p, n, 0);
    if (r < 0 && (errno == EINTR)) continue;
    if (r <= 0) return false;
    p += r;
    n -= r;
  }
  return true;
}

bool write_exact(int fd, const void* buf, size_t n) {
  const uint8_t* p = static_cast<const uint8_t*>(buf);
  while (n > 0) {
    ssize_t w = send(fd, p, n, MSG_NOSIGNAL);
    if (w < 0 && (errno == EINTR || errno == EAGAIN || errno == EWOULDBLOCK)) {
      continue;  // sockets are blocking except during poll-side recv
    }
    if (w <= 0) return false;
    p += w;
    n -= w;
  }
  return true;
}

bool send_pdu(int fd, const void* header, size_t hlen, bool hdgst,
              const void* data, size_t dlen, uint32_t ddgst_value,
              bool ddgst) {
  uint32_t hd = 0;
  struct iovec iov[4];
  int iovcnt = 0;
  iov[iovcnt++] = {const_cast<void*>(header), hlen};
  if (hdgst) {
    hd = crc32c_sw(0, header, hlen);
    iov[iovcnt++] = {&hd, 4};
  }
  if (data != nullptr && dlen > 0) {
    iov[iovcnt++] = {const_cast<void*>(data), dlen};
    if (ddgst) {
      iov[iovcnt++] = {&ddgst_value, 4};
    }
  }
  size_t total = 0;
  for (int i = 0; i < iovcnt; ++i) total += iov[i].iov_len;
  // writev loop handling partial sends
  struct msghdr msg = {};
  msg.msg_iov = iov;
  msg.msg_iovlen = iovcnt;
  size_t sent = 0;
  while (sent < total) {
    ssize_t w = sendmsg(fd, &msg, MSG_NOSIGNAL);
    if (w < 0 && errno == EINTR) continue;
    if (w <= 0) return false;
    sent += w;
    // advance iovecs
    size_t advance = w;
    while (advance > 0 && msg.msg_iovlen > 0) {
      if (msg.msg_iov[0].iov_len <= advance) {
        advance -= msg.msg_iov[0].iov_len;
        ++msg.msg_iov;
        --msg.msg_iovlen;
      } else {
        msg.msg_iov[0].iov_base =
            static_cast<uint8_t*>(msg.msg_iov[0].iov_base) + advance;
        msg.msg_iov[0].iov_len -= advance;
        advance = 0;
      }
    }
  }
  return true;
}

}  // namespace nvmf
}  // namespace hipstore
