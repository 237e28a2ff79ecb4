// hipstored — the MI355X data-path daemon.
//
// Replaces the reference's SPDK vhost app (reference app/vhost/vhost.c):
// serves the SPDK-compatible JSON-RPC socket, backs malloc bdevs with
// MI355X HBM3E (hipMalloc) and moves blocks with LDS-staged HIP kernels
// on per-queue streams. --cpu forces host-RAM bdevs (CI / config 1).

#include <getopt.h>
#include <signal.h>
#include <unistd.h>

#include <condition_variable>
#include <cstdio>
#include <cstring>
#include <mutex>

#include "hipstore/engine.h"
#include "hipstore/json.h"
#include "hipstore/nbd.h"
#include "hipstore/rpc.h"
#include "hipstore/vhost.h"

namespace {

std::mutex g_mutex;
std::condition_variable g_cv;
bool g_stop = false;

void handle_signal(int) {
  {
    std::lock_guard<std::mutex> lock(g_mutex);
    g_stop = true;
  }
  g_cv.notify_all();
}

void usage(const char* argv0) {
  fprintf(stderr,
          "usage: %s [-S socket] [-d device] [-C] \n"
          "  -S PATH   JSON-RPC unix socket (default /var/tmp/hipstored.sock)\n"
          "  -d N      HIP device index backing malloc bdevs (default 0)\n"
          "  -C        CPU mode: host-RAM bdevs even when a GPU is present\n"
          "  -P        persistent engine: on-GPU polling service kernels (default)\n"
          "  -B        batched engine: per-poll kernel launches\n"
          "  -V DIR    vhost-user socket directory (default: RPC socket dir)\n"
          "  -c FILE   apply a save_config snapshot at startup\n",
          argv0);
}

}  // namespace

int main(int argc, char** argv) {
  // Before any HIP call: the persistent engine needs one real hardware
  // queue per service kernel (ROCm default 4 -> gang-scheduling
  // preempts persistent kernels; see native/src/gpu.hip).
  setenv("GPU_MAX_HW_QUEUES", "24", /*overwrite=*/0);

  std::string socket_path = "/var/tmp/hipstored.sock";
  std::string vhost_dir;
  std::string config_file;
  int device = 0;
  bool cpu_only = false;
  bool persistent = true;  // the measured-better engine is the default

  int opt;
  while ((opt = getopt(argc, argv, "S:d:V:c:CPBh")) != -1) {
    switch (opt) {
      case 'S': socket_path = optarg; break;
      case 'd': device = atoi(optarg); break;
      case 'V': vhost_dir = optarg; break;
      case 'c': config_file = optarg; break;
      case 'C': cpu_only = true; break;
      case 'P': persistent = true; break;
      case 'B': persistent = false; break;
      case 'h': usage(argv[0]); return 0;
      default: usage(argv[0]); return 2;
    }
  }
  if (vhost_dir.empty()) {
    auto slash = socket_path.rfind('/');
    vhost_dir = slash == std::string::npos ? "." : socket_path.substr(0, slash);
  }
  hipstore::vhost_set_socket_dir(vhost_dir);

  const bool use_hbm = !cpu_only && hipstore::gpu_available();
  fprintf(stderr, "hipstored: socket=%s mode=%s device=%d gpus=%d\n",
          socket_path.c_str(), use_hbm ? "hbm" : "cpu", device,
          hipstore::gpu_device_count());

  hipstore::RpcServer server(socket_path);
  hipstore::register_storage_methods(&server, use_hbm, device, persistent);
  if (!config_file.empty()) {
    // Recreate the saved topology BEFORE serving clients, by
    // replaying the snapshot through the normal dispatch path.
    try {
      FILE* f = fopen(config_file.c_str(), "r");
      if (f == nullptr) throw std::runtime_error("cannot open " + config_file);
      std::string text;
      char buf[4096];
      size_t n;
      while ((n = fread(buf, 1, sizeof(buf), f)) > 0) text.append(buf, n);
      fclose(f);
      hipstore::JsonObject request;
      request["jsonrpc"] = hipstore::Json(std::string("2.0"));
      request["id"] = hipstore::Json(int64_t{0});
      request["method"] = hipstore::Json(std::string("load_config"));
      request["params"] = hipstore::Json::parse(text);
      hipstore::Json reply =
          server.dispatch(hipstore::Json(std::move(request)));
      if (reply.get("error") != nullptr) {
        throw std::runtime_error(
            reply.get("error")->get_string("message"));
      }
      fprintf(stderr, "hipstored: applied config %s\n", config_file.c_str());
    } catch (const std::exception& e) {
      fprintf(stderr, "hipstored: config load failed: %s\n", e.what());
      return 1;
    }
  }
  try {
    server.start();
  } catch (const std::exception& e) {
    fprintf(stderr, "hipstored: %s\n", e.what());
    return 1;
  }
  fprintf(stderr, "hipstored: ready\n");

  signal(SIGINT, handle_signal);
  signal(SIGTERM, handle_signal);
  {
    std::unique_lock<std::mutex> lock(g_mutex);
    g_cv.wait(lock, [] { return g_stop; });
  }
  fprintf(stderr, "hipstored: shutting down\n");
  hipstore::nbd_stop_all();
  hipstore::vhost_stop_all();
  server.stop();
  return 0;
}
