#include "hipstore/rpc.h"

#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>

#include <cstring>
#include <mutex>

namespace hipstore {

RpcServer::RpcServer(std::string socket_path)
    : socket_path_(std::move(socket_path)) {}

RpcServer::~RpcServer() { stop(); }

void RpcServer::register_method(const std::string& name, RpcMethod fn) {
  std::lock_guard<std::mutex> lock(mutex_);
  methods_[name] = std::move(fn);
}

bool RpcServer::has_method(const std::string& name) const {
  std::lock_guard<std::mutex> lock(mutex_);
  return methods_.count(name) > 0;
}

void RpcServer::register_alias(const std::string& alias,
                               const std::string& existing) {
  std::lock_guard<std::mutex> lock(mutex_);
  auto it = methods_.find(existing);
  if (it == methods_.end()) return;
  methods_[alias] = it->second;
}

std::vector<std::string> RpcServer::method_names() const {
  std::lock_guard<std::mutex> lock(mutex_);
  std::vector<std::string> names;
  names.reserve(methods_.size());
  for (const auto& [name, fn] : methods_) names.push_back(name);
  return names;
}


void RpcServer::start() {
  listen_fd_ = socket(AF_UNIX, SOCK_STREAM | SOCK_CLOEXEC, 0);
  if (listen_fd_ < 0) throw std::runtime_error("rpc: socket() failed");
  sockaddr_un addr{};
  addr.sun_family = AF_UNIX;
  if (socket_path_.size() >= sizeof(addr.sun_path)) {
    throw std::runtime_error("rpc: socket path too long");
  }
  strncpy(addr.sun_path, socket_path_.c_str(), sizeof(addr.sun_path) - 1);
  unlink(socket_path_.c_str());
  if (bind(listen_fd_, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)) < 0) {
    throw std::runtime_error("rpc: bind(" + socket_path_ + ") failed");
  }
  if (listen(listen_fd_, 64) < 0) throw std::runtime_error("rpc: listen failed");
  running_.store(true);
  accept_thread_ = std::thread([this] { accept_loop(); });
}

void RpcServer::stop() {
  if (!running_.exchange(false)) return;
  ::shutdown(listen_fd_, SHUT_RDWR);
  ::close(listen_fd_);
  if (accept_thread_.joinable()) accept_thread_.join();
  {
    // Unblock connection threads parked in read() on live clients;
    // without this, shutdown waits for every client to hang up.
    std::lock_guard<std::mutex> lock(mutex_);
    for (auto& conn : connections_) {
      if (!conn.done->load()) ::shutdown(conn.fd, SHUT_RDWR);
    }
  }
  for (auto& conn : connections_) {
    if (conn.thread.joinable()) conn.thread.join();
  }
  connections_.clear();
  unlink(socket_path_.c_str());
}

void RpcServer::accept_loop() {
  while (running_.load()) {
    int fd = accept(listen_fd_, nullptr, nullptr);
    if (fd < 0) {
      if (!running_.load()) break;
      continue;
    }
    std::lock_guard<std::mutex> lock(mutex_);
    // Reap finished connection threads so a long-lived daemon does not
    // accumulate joinable handles (one per past client).
    for (auto it = connections_.begin(); it != connections_.end();) {
      if (it->done->load()) {
        it->thread.join();
        it = connections_.erase(it);
      } else {
        ++it;
      }
    }
    auto done = std::make_shared<std::atomic<bool>>(false);
    connections_.push_back(Connection{
        std::thread([this, fd, done] {
          serve_connection(fd);
          done->store(true);
        }),
        done, fd});
  }
}

namespace {

Json error_response(const Json& id, int code, const std::string& message) {
  JsonObject err;
  err["code"] = Json(static_cast<int64_t>(code));
  err["message"] = Json(message);
  JsonObject resp;
  resp["jsonrpc"] = Json("2.0");
  resp["id"] = id;
  resp["error"] = Json(std::move(err));
  return Json(std::move(resp));
}

}  // namespace

Json RpcServer::dispatch(const Json& request) {
  Json id = request.get("id") ? *request.get("id") : Json(nullptr);
  if (!request.is_object() || !request.has("method") ||
      !request.get("method")->is_string()) {
    return error_response(id, kInvalidRequest, "invalid request");
  }
  const std::string& method = request.get("method")->as_string();
  RpcMethod fn;
  {
    std::lock_guard<std::mutex> lock(mutex_);
    auto it = methods_.find(method);
    if (it != methods_.end()) fn = it->second;
  }
  if (!fn) {
    return error_response(id, kMethodNotFound, "Method not found");
  }
  Json params = request.get("params") ? *request.get("params") : Json(JsonObject{});
  try {
    Json result = fn(params);
    JsonObject resp;
    resp["jsonrpc"] = Json("2.0");
    resp["id"] = id;
    resp["result"] = std::move(result);
    return Json(std::move(resp));
  } catch (const RpcError& e) {
    return error_response(id, e.code, e.message);
  } catch (const std::exception& e) {
    return error_response(id, kInternalError, e.what());
  }
}

void RpcServer::serve_connection(int fd) {
  std::string buffer;
  char chunk[65536];
  while (running_.load()) {
    ssize_t n = read(fd, chunk, sizeof(chunk));
    if (n <= 0) break;
    buffer.append(chunk, n);
    // Drain every complete JSON value in the buffer.
    size_t start = 0;
    while (start < buffer.size()) {
      Json request;
      size_t consumed = 0;
      bool complete;
      try {
        complete = Json::parse_some(buffer.data() + start,
                                    buffer.data() + buffer.size(), &request,
                                    &consumed);
      } catch (const JsonError&) {
        Json resp = error_response(Json(nullptr), kParseError, "parse error");
        std::string out = resp.dump();
        (void)!write(fd, out.data(), out.size());
        close(fd);
        return;
      }
      if (!complete) break;
      start += consumed;
      Json resp = dispatch(request);
      std::string out = resp.dump();
      size_t off = 0;
      while (off < out.size()) {
        ssize_t w = write(fd, out.data() + off, out.size() - off);
        if (w <= 0) {
          close(fd);
          return;
        }
        off += w;
      }
    }
    buffer.erase(0, start);
  }
  close(fd);
}

}  // namespace hipstore
