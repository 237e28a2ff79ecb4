// JSON-RPC 2.0 server over a Unix socket, wire-compatible with the
// reference's SPDK RPC plane (reference lib/jsonrpc/jsonrpc_server.c +
// pkg/spdk/client.go framing: concatenated JSON objects, no length
// prefix) so the Go/Python clients work unchanged.

#pragma once

#include <atomic>
#include <functional>
#include <memory>
#include <map>
#include <mutex>
#include <string>
#include <thread>
#include <vector>

#include "hipstore/json.h"

namespace hipstore {

// JSON-RPC error codes (reference pkg/spdk/client.go:60-68). SPDK's
// "not found" frequently surfaces as kInvalidParams; callers tolerate it.
enum RpcErrorCode {
  kParseError = -32700,
  kInvalidRequest = -32600,
  kMethodNotFound = -32601,
  kInvalidParams = -32602,
  kInternalError = -32603,
};

struct RpcError {
  int code;
  std::string message;
};

using RpcMethod = std::function<Json(const Json& params)>;

class RpcServer {
 public:
  explicit RpcServer(std::string socket_path);
  ~RpcServer();

  void register_method(const std::string& name, RpcMethod fn);
  // Register `alias` to dispatch to already-registered `existing`.
  void register_alias(const std::string& alias, const std::string& existing);
  bool has_method(const std::string& name) const;
  std::vector<std::string> method_names() const;

  void start();  // binds + spawns accept loop
  void stop();
  const std::string& socket_path() const { return socket_path_; }

  // Dispatch a single already-parsed request; exposed for tests.
  Json dispatch(const Json& request);

 private:
  void accept_loop();
  void serve_connection(int fd);

  std::string socket_path_;
  int listen_fd_ = -1;
  std::atomic<bool> running_{false};
  std::thread accept_thread_;
  // Connection bookkeeping: the accept loop reaps finished threads,
  // and stop() shuts the live fds down so blocked reads unblock
  // (otherwise shutdown waits on connected-but-idle clients).
  struct Connection {
    std::thread thread;
    std::shared_ptr<std::atomic<bool>> done;
    int fd;
  };
  std::vector<Connection> connections_;
  std::map<std::string, RpcMethod> methods_;
  mutable std::mutex mutex_;
};

// Registers the SPDK-compatible method set
// against the global BdevManager. `use_hbm` selects HBM-resident malloc
// bdevs on `device`; false (or no GPU) falls back to host RAM.
void register_storage_methods(RpcServer* server, bool use_hbm, int device,
                              bool persistent = false);

}  // namespace hipstore
