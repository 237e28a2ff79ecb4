// The SPDK-wire-compatible RPC method set.
//
// Exactly the methods the reference's Go client invokes
// (reference pkg/spdk/spdk.go), same parameter and result shapes, plus
// two native extras (get_rpc_methods, perf_run). "Not found" errors use
// kInvalidParams like SPDK does; all reference callers tolerate that
// (reference local.go:53-57, controller.go:76,204,239).

#include <algorithm>
#include <cinttypes>
#include <cstdio>
#include <cstring>
#include <mutex>

#include "hipstore/bdev.h"
#include "hipstore/composite.h"
#include "hipstore/engine.h"
#include "hipstore/json.h"
#include "hipstore/nbd.h"
#include "hipstore/nvmf.h"
#include "hipstore/rados.h"
#include "hipstore/ublk.h"
#include "hipstore/rpc.h"
#include "hipstore/vhost.h"

namespace hipstore {

namespace {

constexpr int kMaxScsiTargets = 8;  // reference vhost_scsi.c:952 default

struct ScsiLun {
  int id;
  std::string bdev_name;
};

struct ScsiTarget {
  bool used = false;
  std::string target_name;
  std::vector<ScsiLun> luns;
};

struct VhostController {
  std::string cpumask;
  ScsiTarget targets[kMaxScsiTargets];
  std::string blk_bdev;   // non-empty => virtio-blk personality
  bool blk_readonly = false;
  VhostDevPtr dev;  // live vhost-user server for this controller
};

struct VhostState {
  std::mutex mutex;
  std::map<std::string, VhostController> controllers;
};

VhostState& vhost_state() {
  static VhostState s;
  return s;
}

int g_malloc_seq = 0;
std::mutex g_name_mutex;

Json bdev_to_json(const BdevPtr& bdev) {
  JsonObject io_types;
  io_types["read"] = Json(true);
  io_types["write"] = Json(true);
  io_types["unmap"] = Json(true);
  io_types["write_zeroes"] = Json(true);
  io_types["flush"] = Json(true);
  io_types["reset"] = Json(false);
  io_types["nvme_admin"] = Json(false);
  io_types["nvme_io"] = Json(false);
  JsonObject o;
  o["name"] = Json(bdev->name());
  o["product_name"] = Json(bdev->product_name());
  o["uuid"] = Json(bdev->uuid());
  o["block_size"] = Json(static_cast<int64_t>(bdev->block_size()));
  o["num_blocks"] = Json(static_cast<int64_t>(bdev->num_blocks()));
  o["claimed"] = Json(bdev->claimed());
  o["supported_io_types"] = Json(std::move(io_types));
  if (bdev->gpu_device() >= 0) {
    JsonObject hbm;
    hbm["device"] = Json(static_cast<int64_t>(bdev->gpu_device()));
    hbm["pci_address"] = Json(gpu_pci_address(bdev->gpu_device()));
    JsonObject ds;
    ds["hbm"] = Json(std::move(hbm));
    o["driver_specific"] = Json(std::move(ds));
  } else if (!bdev->backing_path().empty()) {
    JsonObject aio;
    aio["filename"] = Json(bdev->backing_path());
    JsonObject ds;
    ds["aio"] = Json(std::move(aio));
    o["driver_specific"] = Json(std::move(ds));
  }
  return Json(std::move(o));
}

[[noreturn]] void not_found(const std::string& what) {
  throw RpcError{kInvalidParams, what + " does not exist"};
}

}  // namespace

void register_storage_methods(RpcServer* server, bool use_hbm, int device,
                              bool persistent) {
  auto& manager = BdevManager::instance();

  // Construction-parameter registry for save_config: composite, RBD
  // and clone bdevs cannot be introspected back into construct params
  // the way plain malloc bdevs can, so record the creating call.
  struct Creation {
    uint64_t seq;
    std::string method;
    Json params;
  };
  struct Creations {
    std::mutex mutex;
    uint64_t next_seq = 0;
    std::map<std::string, Creation> by_name;
  };
  auto creations = std::make_shared<Creations>();
  auto record_creation = [creations](const std::string& name,
                                     const std::string& method,
                                     const Json& params) {
    std::lock_guard<std::mutex> lock(creations->mutex);
    creations->by_name[name] = Creation{creations->next_seq++, method, params};
  };


  server->register_method("get_bdevs", [&manager](const Json& p) {
    const std::string name = p.get_string("name");
    JsonArray out;
    if (!name.empty()) {
      BdevPtr bdev = manager.find(name);
      if (!bdev) not_found("bdev " + name);
      out.push_back(bdev_to_json(bdev));
    } else {
      for (const auto& bdev : manager.list()) out.push_back(bdev_to_json(bdev));
    }
    return Json(std::move(out));
  });

  server->register_method("delete_bdev", [&manager, creations](const Json& p) {
    const std::string name = p.get_string("name");
    BdevPtr bdev = manager.find(name);
    if (!bdev) not_found("bdev " + name);
    if (bdev->claimed()) {
      throw RpcError{kInvalidParams, "bdev " + name + " is claimed"};
    }
    manager.remove(name);
    {
      std::lock_guard<std::mutex> lock(creations->mutex);
      creations->by_name.erase(name);
    }
    return Json(JsonObject{});
  });

  server->register_method(
      "construct_malloc_bdev",
      [&manager, use_hbm, device, persistent](const Json& p) {
        const int64_t num_blocks = p.get_int("num_blocks");
        const int64_t block_size = p.get_int("block_size");
        if (num_blocks <= 0 || block_size <= 0 || block_size % 512 != 0 ||
            num_blocks > (int64_t{1} << 62) / block_size) {
          // the product must not wrap uint64 (a wrapped size would
          // report a consistent-but-lying geometry)
          throw RpcError{kInvalidParams, "invalid num_blocks/block_size"};
        }
        std::string name = p.get_string("name");
        {
          std::lock_guard<std::mutex> lock(g_name_mutex);
          if (name.empty()) name = "Malloc" + std::to_string(g_malloc_seq++);
        }
        if (manager.find(name)) {
          throw RpcError{kInvalidParams, "bdev " + name + " already exists"};
        }
        BdevPtr bdev;
        if (use_hbm && gpu_available()) {
          bdev = create_hbm_bdev(name, block_size, num_blocks, device,
                                 p.get_bool("persistent", persistent));
        } else {
          bdev = create_malloc_bdev(name, block_size, num_blocks);
        }
        manager.add(bdev);
        return Json(name);
      });

  server->register_method(
      "construct_aio_bdev", [&manager, record_creation](const Json& p) {
        // SPDK aio bdev: file-backed, geometry from file size; the
        // only bdev whose data survives restarts.
        const std::string name = p.get_string("name");
        const std::string filename = p.get_string("filename");
        const int64_t block_size = p.get_int("block_size", 512);
        if (name.empty() || filename.empty() || block_size <= 0 ||
            block_size % 512 != 0) {
          throw RpcError{kInvalidParams,
                         "name/filename/block_size required"};
        }
        if (manager.find(name)) {
          throw RpcError{kInvalidParams, "bdev " + name + " already exists"};
        }
        BdevPtr bdev;
        try {
          bdev = create_file_bdev(name, filename, block_size);
        } catch (const std::exception& e) {
          throw RpcError{kInvalidParams, e.what()};
        }
        manager.add(bdev);
        record_creation(name, "construct_aio_bdev", p);
        return Json(name);
      });

  server->register_method(
      "construct_rbd_bdev",
      [&manager, use_hbm, device, record_creation](const Json& p) {
        // Ceph RBD path. With `config.mon_host` set, the bdev speaks
        // the RADOS messenger protocol over TCP to that cluster
        // endpoint (the in-repo loopback cluster started by
        // rados_cluster_start, rados_cluster.cpp — this environment
        // has no external network, so that is the reachable cluster).
        // Without mon_host the image falls back to a local emulated
        // store of `config.emu_size_mb` (default 1 GiB); the ceph-csi
        // parameter plumbing (user_id/pool/image/monitors/secret) is
        // exercised end-to-end either way.
        const std::string pool = p.get_string("pool_name");
        const std::string image = p.get_string("rbd_name");
        const int64_t block_size = p.get_int("block_size", 512);
        if (pool.empty() || image.empty()) {
          throw RpcError{kInvalidParams, "pool_name and rbd_name required"};
        }
        if (block_size <= 0 || block_size % 512 != 0) {
          throw RpcError{kInvalidParams, "invalid block_size"};
        }
        std::string name = p.get_string("name");
        if (name.empty()) name = pool + "/" + image;
        if (manager.find(name)) {
          throw RpcError{kInvalidParams, "bdev " + name + " already exists"};
        }
        int64_t emu_mb = 1024;
        int64_t object_mb = 4;  // RBD default object size (order 22)
        std::string mon_host;
        if (const Json* config = p.get("config")) {
          if (config->is_object()) {
            emu_mb = config->get_int("emu_size_mb", emu_mb);
            object_mb = config->get_int("object_mb", object_mb);
            mon_host = config->get_string("mon_host");
          }
        }
        if (object_mb <= 0 || object_mb > 64) {
          throw RpcError{kInvalidParams, "bad object_mb"};
        }
        if (!mon_host.empty()) {
          // SPDK's mon_host is comma-separated; this cluster is one
          // endpoint, so the first entry wins. object_mb must not
          // exceed the cluster's slot granularity
          // (rados_cluster_start object_mb).
          const size_t comma = mon_host.find(',');
          if (comma != std::string::npos) mon_host.resize(comma);
          BdevPtr bdev;
          try {
            bdev = create_rbd_bdev(
                name, mon_host, pool, image, block_size,
                static_cast<uint64_t>(emu_mb) << 20,
                static_cast<uint64_t>(object_mb) << 20);
          } catch (const std::exception& e) {
            throw RpcError{kInvalidParams, e.what()};
          }
          manager.add(bdev);
          record_creation(name, "construct_rbd_bdev", p);
          return Json(name);
        }
        const uint64_t num_blocks = emu_mb * 1024 * 1024 / block_size;
        BdevPtr inner;
        if (use_hbm && gpu_available()) {
          inner = create_hbm_bdev(name, block_size, num_blocks, device);
        } else {
          inner = create_malloc_bdev(name, block_size, num_blocks);
        }
        // Re-badge as an RBD disk: UnmapVolume deletes these
        // (product_name != "Malloc disk", reference controller.go:205).
        inner->set_product("Ceph Rbd Disk");
        manager.add(inner);
        record_creation(name, "construct_rbd_bdev", p);
        return Json(name);
      });

  server->register_method("start_nbd_disk", [&manager](const Json& p) {
    const std::string bdev_name = p.get_string("bdev_name");
    const std::string device = p.get_string("nbd_device");
    if (!manager.find(bdev_name)) not_found("bdev " + bdev_name);
    nbd_start(bdev_name, device);
    return Json(true);
  });

  server->register_method("get_nbd_disks", [](const Json&) {
    JsonArray out;
    for (const auto& [bdev_name, device] : nbd_list()) {
      JsonObject o;
      o["bdev_name"] = Json(bdev_name);
      o["nbd_device"] = Json(device);
      out.push_back(Json(std::move(o)));
    }
    return Json(std::move(out));
  });

  server->register_method("stop_nbd_disk", [](const Json& p) {
    nbd_stop(p.get_string("nbd_device"));
    return Json(true);
  });

  // --- ublk host attach (reference lib/nbd/nbd.c role; the pool's
  // kernels ship ublk_drv but no nbd module, so this is the path that
  // yields a real /dev/ublkbN on a GPU box) --------------------------------
  server->register_method("ublk_start_disk", [&manager](const Json& p) {
    const std::string bdev_name = p.get_string("bdev_name");
    if (!manager.find(bdev_name)) not_found("bdev " + bdev_name);
    UblkDisk disk;
    try {
      disk = ublk_start(bdev_name,
                        static_cast<int>(p.get_int("queue_depth", 32)));
    } catch (const std::exception& e) {
      throw RpcError{kInternalError, e.what()};
    }
    JsonObject o;
    o["dev_id"] = Json(static_cast<int64_t>(disk.dev_id));
    o["device"] = Json(disk.block_path);
    o["bdev_name"] = Json(disk.bdev_name);
    return Json(std::move(o));
  });

  server->register_method("ublk_stop_disk", [](const Json& p) {
    try {
      ublk_stop(static_cast<int>(p.get_int("dev_id")));
    } catch (const std::exception& e) {
      throw RpcError{kInvalidParams, e.what()};
    }
    return Json(JsonObject{});
  });

  server->register_method("ublk_get_disks", [](const Json&) {
    JsonArray out;
    for (const UblkDisk& disk : ublk_list()) {
      JsonObject o;
      o["dev_id"] = Json(static_cast<int64_t>(disk.dev_id));
      o["device"] = Json(disk.block_path);
      o["bdev_name"] = Json(disk.bdev_name);
      out.push_back(Json(std::move(o)));
    }
    return Json(std::move(out));
  });

  server->register_method("construct_vhost_scsi_controller", [](const Json& p) {
    const std::string name = p.get_string("ctrlr");
    if (name.empty()) throw RpcError{kInvalidParams, "ctrlr required"};
    auto& state = vhost_state();
    std::lock_guard<std::mutex> lock(state.mutex);
    if (state.controllers.count(name)) {
      throw RpcError{kInvalidParams, "controller " + name + " already exists"};
    }
    // Serve the vhost-user socket immediately (SPDK creates it at
    // construct time too); the resolver re-reads the target table per
    // command so add/remove_vhost_scsi_lun is visible without restart.
    VhostDevPtr dev;
    try {
      dev = vhost_start(name, [name](int target) -> BdevPtr {
        auto& st = vhost_state();
        std::lock_guard<std::mutex> l(st.mutex);
        auto it = st.controllers.find(name);
        if (it == st.controllers.end()) return nullptr;
        if (target < 0 || target >= kMaxScsiTargets) return nullptr;
        const ScsiTarget& t = it->second.targets[target];
        if (!t.used || t.luns.empty()) return nullptr;
        return BdevManager::instance().find(t.luns[0].bdev_name);
      });
    } catch (const std::exception& e) {
      throw RpcError{kInvalidParams, e.what()};
    }
    VhostController& ctrl = state.controllers[name];
    ctrl.cpumask = p.get_string("cpumask", "0x1");
    ctrl.dev = dev;
    return Json(JsonObject{});
  });

  server->register_method(
      "construct_vhost_blk_controller", [&manager](const Json& p) {
        // SPDK vhost-user-blk twin: one bdev, claimed for the
        // controller's lifetime; rings serve virtio-blk requests.
        const std::string name = p.get_string("ctrlr");
        const std::string dev_name = p.get_string("dev_name");
        if (name.empty() || dev_name.empty()) {
          throw RpcError{kInvalidParams, "ctrlr and dev_name required"};
        }
        auto& state = vhost_state();
        std::lock_guard<std::mutex> lock(state.mutex);
        if (state.controllers.count(name)) {
          throw RpcError{kInvalidParams,
                         "controller " + name + " already exists"};
        }
        BdevPtr bdev = manager.find(dev_name);
        if (!bdev) not_found("bdev " + dev_name);
        if (!bdev->claim()) {
          throw RpcError{kInvalidParams, "bdev " + dev_name + " is claimed"};
        }
        VhostDevPtr dev;
        try {
          dev = vhost_start_blk(
              name,
              [name](int) -> BdevPtr {
                auto& st = vhost_state();
                std::lock_guard<std::mutex> l(st.mutex);
                auto it = st.controllers.find(name);
                if (it == st.controllers.end()) return nullptr;
                return BdevManager::instance().find(it->second.blk_bdev);
              },
              p.get_bool("readonly", false));
        } catch (const std::exception& e) {
          bdev->release();
          throw RpcError{kInvalidParams, e.what()};
        }
        VhostController& ctrl = state.controllers[name];
        ctrl.cpumask = p.get_string("cpumask", "0x1");
        ctrl.blk_bdev = dev_name;
        ctrl.blk_readonly = p.get_bool("readonly", false);
        ctrl.dev = dev;
        return Json(JsonObject{});
      });

  server->register_method("add_vhost_scsi_lun", [&manager](const Json& p) {
    const std::string ctrlr = p.get_string("ctrlr");
    const int64_t target_num = p.get_int("scsi_target_num", -1);
    const std::string bdev_name = p.get_string("bdev_name");
    auto& state = vhost_state();
    std::lock_guard<std::mutex> lock(state.mutex);
    auto it = state.controllers.find(ctrlr);
    if (it == state.controllers.end()) not_found("controller " + ctrlr);
    if (!it->second.blk_bdev.empty()) {
      throw RpcError{kInvalidParams,
                     "controller " + ctrlr + " is a blk controller"};
    }
    if (target_num < 0 || target_num >= kMaxScsiTargets) {
      throw RpcError{kInvalidParams, "scsi_target_num out of range"};
    }
    BdevPtr bdev = manager.find(bdev_name);
    if (!bdev) not_found("bdev " + bdev_name);
    ScsiTarget& target = it->second.targets[target_num];
    if (target.used) {
      throw RpcError{kInvalidParams, "target already occupied"};
    }
    if (!bdev->claim()) {
      throw RpcError{kInvalidParams, "bdev " + bdev_name + " is claimed"};
    }
    target.used = true;
    target.target_name = "Target " + std::to_string(target_num);
    target.luns = {ScsiLun{0, bdev_name}};
    return Json(JsonObject{});
  });

  server->register_method("remove_vhost_scsi_target", [&manager](const Json& p) {
    const std::string ctrlr = p.get_string("ctrlr");
    const int64_t target_num = p.get_int("scsi_target_num", -1);
    auto& state = vhost_state();
    std::lock_guard<std::mutex> lock(state.mutex);
    auto it = state.controllers.find(ctrlr);
    if (it == state.controllers.end()) not_found("controller " + ctrlr);
    if (target_num < 0 || target_num >= kMaxScsiTargets ||
        !it->second.targets[target_num].used) {
      throw RpcError{kInvalidParams, "no such target"};
    }
    ScsiTarget& target = it->second.targets[target_num];
    for (const ScsiLun& lun : target.luns) {
      if (BdevPtr bdev = manager.find(lun.bdev_name)) bdev->release();
    }
    target = ScsiTarget{};
    return Json(JsonObject{});
  });

  server->register_method("remove_vhost_controller", [&manager](const Json& p) {
    const std::string ctrlr = p.get_string("ctrlr");
    auto& state = vhost_state();
    VhostDevPtr dev;
    {
      std::lock_guard<std::mutex> lock(state.mutex);
      auto it = state.controllers.find(ctrlr);
      if (it == state.controllers.end()) not_found("controller " + ctrlr);
      for (ScsiTarget& target : it->second.targets) {
        for (const ScsiLun& lun : target.luns) {
          if (BdevPtr bdev = manager.find(lun.bdev_name)) bdev->release();
        }
      }
      if (!it->second.blk_bdev.empty()) {
        if (BdevPtr bdev = manager.find(it->second.blk_bdev)) {
          bdev->release();
        }
      }
      dev = it->second.dev;
      state.controllers.erase(it);
    }
    // Outside the lock: ring workers may be blocked in the resolver
    // (which takes state.mutex) and stop() joins them.
    vhost_stop(dev);
    return Json(JsonObject{});
  });

  server->register_method("get_vhost_controllers", [](const Json&) {
    auto& state = vhost_state();
    std::lock_guard<std::mutex> lock(state.mutex);
    JsonArray out;
    for (const auto& [name, ctrl] : state.controllers) {
      JsonArray scsi;
      for (int t = 0; t < kMaxScsiTargets; ++t) {
        const ScsiTarget& target = ctrl.targets[t];
        if (!target.used) continue;
        JsonArray luns;
        for (const ScsiLun& lun : target.luns) {
          JsonObject lo;
          lo["id"] = Json(static_cast<int64_t>(lun.id));
          lo["bdev_name"] = Json(lun.bdev_name);
          luns.push_back(Json(std::move(lo)));
        }
        JsonObject to;
        to["target_name"] = Json(target.target_name);
        to["id"] = Json(static_cast<int64_t>(t));
        to["scsi_dev_num"] = Json(static_cast<int64_t>(t));
        to["luns"] = Json(std::move(luns));
        scsi.push_back(Json(std::move(to)));
      }
      JsonObject backend;
      if (!ctrl.blk_bdev.empty()) {
        JsonObject blk;
        blk["bdev"] = Json(ctrl.blk_bdev);
        blk["readonly"] = Json(ctrl.blk_readonly);
        backend["block"] = Json(std::move(blk));
      } else {
        backend["scsi"] = Json(std::move(scsi));
      }
      JsonObject o;
      o["ctrlr"] = Json(name);
      o["cpumask"] = Json(ctrl.cpumask);
      o["backend_specific"] = Json(std::move(backend));
      out.push_back(Json(std::move(o)));
    }
    return Json(std::move(out));
  });

  // --- native extras -------------------------------------------------------

  server->register_method("spdk_get_version", [](const Json&) {
    // SPDK wire parity: tools probe this to fingerprint the daemon.
    JsonObject version;
    version["version"] = Json(std::string("oim-amd 0.1.0 (hipstored)"));
    JsonObject fields;
    fields["major"] = Json(int64_t{0});
    fields["minor"] = Json(int64_t{1});
    fields["patch"] = Json(int64_t{0});
    fields["suffix"] = Json(std::string("-hipstored"));
    version["fields"] = Json(std::move(fields));
    return Json(std::move(version));
  });

  server->register_method("get_rpc_methods", [server](const Json&) {
    // Live registry, not a hand-kept list (SPDK semantics: the actual
    // dispatch table). Includes methods registered after this one.
    JsonArray out;
    for (const std::string& name : server->method_names()) {
      out.push_back(Json(name));
    }
    return Json(std::move(out));
  });

  // --- NVMe-oF/TCP (BASELINE config 3) -----------------------------------
  // nvmf_create_target exports existing bdevs as namespaces (nsid =
  // position in `bdevs` + 1); construct_nvme_tcp_bdev connects an
  // initiator bdev to any NVMe/TCP target (ours or a foreign one).
  struct TargetEntry {
    std::shared_ptr<NvmfTcpTarget> target;
    std::string listen_addr;
    bool digests;
    std::vector<std::string> ns_bdevs;  // nsid = index + 1
  };
  struct Targets {
    std::mutex mutex;
    std::map<std::string, TargetEntry> by_nqn;
  };
  auto targets = std::make_shared<Targets>();


  server->register_method(
      "nvmf_create_target", [&manager, targets](const Json& p) {
        const std::string subnqn =
            p.get_string("subnqn", "nqn.2026-01.com.amd:oim-amd");
        std::lock_guard<std::mutex> lock(targets->mutex);
        if (targets->by_nqn.count(subnqn)) {
          throw RpcError{kInvalidParams, "target " + subnqn + " exists"};
        }
        auto target = start_nvmf_tcp_target(
            p.get_string("listen_addr"),
            static_cast<uint16_t>(p.get_int("port", 0)), subnqn,
            p.get_bool("digests", true));
        if (const Json* bdevs = p.get("bdevs")) {
          for (const Json& name : bdevs->as_array()) {
            BdevPtr bdev = manager.find(name.as_string());
            if (!bdev) {
              target->stop();
              not_found("bdev " + name.as_string());
            }
            target->add_namespace(bdev);
          }
        }
        JsonObject o;
        o["port"] = Json(static_cast<int64_t>(target->port()));
        o["subnqn"] = Json(subnqn);
        TargetEntry entry;
        entry.listen_addr = p.get_string("listen_addr");
        entry.digests = p.get_bool("digests", true);
        if (const Json* bdevs = p.get("bdevs")) {
          for (const Json& name : bdevs->as_array()) {
            entry.ns_bdevs.push_back(name.as_string());
          }
        }
        entry.target = std::move(target);
        targets->by_nqn[subnqn] = std::move(entry);
        return Json(std::move(o));
      });

  server->register_method("nvmf_get_subsystems", [targets](const Json&) {
    std::lock_guard<std::mutex> lock(targets->mutex);
    JsonArray out;
    for (const auto& [subnqn, entry] : targets->by_nqn) {
      JsonArray namespaces;
      int64_t nsid = 1;
      for (const std::string& bdev : entry.ns_bdevs) {
        JsonObject ns;
        ns["nsid"] = Json(nsid++);
        ns["bdev_name"] = Json(bdev);
        namespaces.push_back(Json(std::move(ns)));
      }
      JsonArray addresses;
      JsonObject addr;
      addr["trtype"] = Json(std::string("TCP"));
      addr["traddr"] = Json(entry.listen_addr);
      addr["trsvcid"] = Json(std::to_string(entry.target->port()));
      addresses.push_back(Json(std::move(addr)));
      JsonObject o;
      o["nqn"] = Json(subnqn);
      o["subtype"] = Json(std::string("NVMe"));
      o["listen_addresses"] = Json(std::move(addresses));
      o["namespaces"] = Json(std::move(namespaces));
      out.push_back(Json(std::move(o)));
    }
    return Json(std::move(out));
  });

  server->register_method("nvmf_delete_target", [targets](const Json& p) {
    std::lock_guard<std::mutex> lock(targets->mutex);
    auto it = targets->by_nqn.find(p.get_string("subnqn"));
    if (it == targets->by_nqn.end()) not_found("nvmf target");
    it->second.target->stop();
    targets->by_nqn.erase(it);
    return Json(JsonObject{});
  });

  server->register_method(
      "construct_nvme_tcp_bdev", [&manager](const Json& p) {
        const std::string name = p.get_string("name");
        if (name.empty()) throw RpcError{kInvalidParams, "name required"};
        if (manager.find(name)) {
          throw RpcError{kInvalidParams, "bdev " + name + " already exists"};
        }
        try {
          BdevPtr bdev = create_nvmf_tcp_bdev(
              name, p.get_string("traddr", "127.0.0.1"),
              static_cast<uint16_t>(p.get_int("trsvcid")),
              p.get_string("subnqn", "nqn.2026-01.com.amd:oim-amd"),
              static_cast<uint32_t>(p.get_int("nsid", 1)),
              p.get_bool("digests", true));
          manager.add(bdev);
        } catch (const std::exception& e) {
          throw RpcError{kInternalError, e.what()};
        }
        return Json(name);
      });

  // --- loopback RADOS cluster (BASELINE config 4) ------------------------
  // An in-process fake Ceph cluster (msgr-v1 endpoint playing mon+osd,
  // rados_cluster.cpp) so construct_rbd_bdev's mon_host path has a
  // reachable peer in this no-network environment. Objects live in an
  // HBM arena on GPU boxes; data CRCs are GPU-computed there.
  struct RadosClusterEntry {
    std::shared_ptr<RadosCluster> cluster;
    Json params;  // replayable rados_cluster_start params
  };
  struct RadosClusters {
    std::mutex mutex;
    std::map<uint16_t, RadosClusterEntry> by_port;
  };
  auto rados_clusters = std::make_shared<RadosClusters>();

  server->register_method(
      "rados_cluster_start",
      [rados_clusters, use_hbm, device](const Json& p) {
        const int64_t arena_mb = p.get_int("arena_mb", 512);
        const int64_t object_mb = p.get_int("object_mb", 4);
        if (arena_mb <= 0 || object_mb <= 0 || object_mb > arena_mb) {
          throw RpcError{kInvalidParams, "bad arena/object size"};
        }
        std::shared_ptr<RadosCluster> cluster;
        try {
          cluster = start_rados_cluster(
              static_cast<uint16_t>(p.get_int("port", 0)),
              static_cast<uint64_t>(arena_mb),
              use_hbm && p.get_bool("use_hbm", true), device,
              static_cast<uint64_t>(object_mb) << 20);
        } catch (const std::exception& e) {
          throw RpcError{kInternalError, e.what()};
        }
        std::lock_guard<std::mutex> lock(rados_clusters->mutex);
        JsonObject replay;
        replay["port"] = Json(static_cast<int64_t>(cluster->port()));
        replay["arena_mb"] = Json(arena_mb);
        replay["object_mb"] = Json(object_mb);
        replay["use_hbm"] = Json(p.get_bool("use_hbm", true));
        rados_clusters->by_port[cluster->port()] =
            RadosClusterEntry{cluster, Json(std::move(replay))};
        JsonObject o;
        o["port"] = Json(static_cast<int64_t>(cluster->port()));
        o["mon_host"] =
            Json("127.0.0.1:" + std::to_string(cluster->port()));
        return Json(std::move(o));
      });

  server->register_method(
      "rados_cluster_stop", [rados_clusters](const Json& p) {
        std::lock_guard<std::mutex> lock(rados_clusters->mutex);
        auto it = rados_clusters->by_port.find(
            static_cast<uint16_t>(p.get_int("port")));
        if (it == rados_clusters->by_port.end()) {
          not_found("rados cluster");
        }
        it->second.cluster->stop();
        rados_clusters->by_port.erase(it);
        return Json(JsonObject{});
      });

  // --- composite bdevs (BASELINE config 5) -------------------------------
  // Striped / replicated malloc bdevs spanning the node's GPUs. The
  // `devices` param lists HIP device indices (one HBM child each); with
  // --cpu (or no GPU) `replicas` host-RAM children emulate the layout
  // for CI. These methods are native extensions — SPDK's raid bdev has
  // a different RPC shape and no GPU notion.
  auto make_children = [use_hbm, persistent](
                           const Json& p, const std::string& name,
                           int64_t num_blocks, int64_t block_size,
                           size_t count_hint) {
    std::vector<BdevPtr> children;
    std::vector<int> devices;
    if (const Json* d = p.get("devices")) {
      for (const Json& v : d->as_array()) {
        devices.push_back(static_cast<int>(v.as_int()));
      }
    }
    size_t count = devices.empty() ? count_hint : devices.size();
    if (count == 0) throw RpcError{kInvalidParams, "no devices/replicas"};
    for (size_t i = 0; i < count; ++i) {
      const std::string child_name = name + "." + std::to_string(i);
      if (use_hbm && gpu_available() && !devices.empty()) {
        children.push_back(create_hbm_bdev(
            child_name, block_size, num_blocks, devices[i],
            p.get_bool("persistent", persistent)));
      } else {
        children.push_back(
            create_malloc_bdev(child_name, block_size, num_blocks));
      }
    }
    return children;
  };

  server->register_method(
      "construct_striped_malloc_bdev",
      [&manager, make_children, record_creation](const Json& p) {
        const std::string name = p.get_string("name");
        const int64_t num_blocks = p.get_int("num_blocks");   // per child
        const int64_t block_size = p.get_int("block_size", 512);
        const int64_t stripe_kb = p.get_int("stripe_size_kb", 128);
        const int64_t replicas = p.get_int("count", 2);
        if (name.empty() || num_blocks <= 0 || block_size <= 0) {
          throw RpcError{kInvalidParams, "name/num_blocks/block_size required"};
        }
        if (manager.find(name)) {
          throw RpcError{kInvalidParams, "bdev " + name + " already exists"};
        }
        try {
          auto children = make_children(p, name, num_blocks, block_size,
                                        replicas);
          BdevPtr bdev = create_striped_bdev(name, std::move(children),
                                             stripe_kb * 1024);
          manager.add(bdev);
        } catch (const std::exception& e) {
          throw RpcError{kInvalidParams, e.what()};
        }
        record_creation(name, "construct_striped_malloc_bdev", p);
        return Json(name);
      });

  server->register_method(
      "construct_replicated_malloc_bdev",
      [&manager, make_children, record_creation](const Json& p) {
        const std::string name = p.get_string("name");
        const int64_t num_blocks = p.get_int("num_blocks");
        const int64_t block_size = p.get_int("block_size", 512);
        const int64_t replicas = p.get_int("count", 2);
        if (name.empty() || num_blocks <= 0 || block_size <= 0) {
          throw RpcError{kInvalidParams, "name/num_blocks/block_size required"};
        }
        if (manager.find(name)) {
          throw RpcError{kInvalidParams, "bdev " + name + " already exists"};
        }
        try {
          auto children = make_children(p, name, num_blocks, block_size,
                                        replicas);
          BdevPtr bdev = create_replicated_bdev(name, std::move(children));
          manager.add(bdev);
        } catch (const std::exception& e) {
          throw RpcError{kInvalidParams, e.what()};
        }
        record_creation(name, "construct_replicated_malloc_bdev", p);
        return Json(name);
      });

  server->register_method("get_bdevs_iostat", [&manager](const Json& p) {
    // SPDK-shaped per-bdev I/O counters.
    const std::string name = p.get_string("name");
    JsonArray bdevs;
    auto one = [](const BdevPtr& bdev) {
      auto& st = bdev->iostat();
      JsonObject o;
      o["name"] = Json(bdev->name());
      o["num_read_ops"] = Json(static_cast<int64_t>(st.num_read_ops.load()));
      o["num_write_ops"] = Json(static_cast<int64_t>(st.num_write_ops.load()));
      o["num_unmap_ops"] = Json(static_cast<int64_t>(st.num_unmap_ops.load()));
      o["bytes_read"] = Json(static_cast<int64_t>(st.bytes_read.load()));
      o["bytes_written"] = Json(static_cast<int64_t>(st.bytes_written.load()));
      return Json(std::move(o));
    };
    if (!name.empty()) {
      BdevPtr bdev = manager.find(name);
      if (!bdev) not_found("bdev " + name);
      bdevs.push_back(one(bdev));
    } else {
      for (const auto& bdev : manager.list()) bdevs.push_back(one(bdev));
    }
    JsonObject o;
    o["bdevs"] = Json(std::move(bdevs));
    return Json(std::move(o));
  });

  server->register_method("get_hbm_info", [device](const Json& p) {
    // Capacity source for CSI GetCapacity: HBM totals on GPU; in CPU
    // mode malloc bdevs consume host RAM, so report MemTotal /
    // MemAvailable instead (config-1 semantics).
    const int dev = static_cast<int>(p.get_int("device", device));
    auto [total, free_bytes] = hbm_info(dev);
    if (total == 0) {
      if (FILE* f = fopen("/proc/meminfo", "r")) {
        char key[64];
        unsigned long long kb = 0;
        while (fscanf(f, "%63s %llu kB\n", key, &kb) == 2) {
          if (strcmp(key, "MemTotal:") == 0) total = kb * 1024;
          if (strcmp(key, "MemAvailable:") == 0) free_bytes = kb * 1024;
        }
        fclose(f);
      }
    }
    JsonObject o;
    o["device"] = Json(static_cast<int64_t>(dev));
    o["total_bytes"] = Json(static_cast<int64_t>(total));
    o["free_bytes"] = Json(static_cast<int64_t>(free_bytes));
    o["pci_address"] = Json(gpu_available() ? gpu_pci_address(dev) : "");
    return Json(std::move(o));
  });

  server->register_method("resize_malloc_bdev", [&manager](const Json& p) {
    // Offline volume expansion (CSI ControllerExpandVolume): malloc
    // bdevs move their backing store, so live channels must be gone.
    BdevPtr bdev = manager.find(p.get_string("name"));
    if (!bdev) not_found("bdev " + p.get_string("name"));
    const int64_t size = p.get_int("size", -1);
    if (size <= 0 || size % static_cast<int64_t>(bdev->block_size()) != 0) {
      throw RpcError{kInvalidParams,
                     "size must be a positive multiple of the block size"};
    }
    const int status = bdev->resize(size / bdev->block_size());
    if (status == kIoInvalid) {
      throw RpcError{kInvalidParams,
                     "bdev " + bdev->name() + " cannot be resized"};
    }
    if (status != kIoOk) {
      throw RpcError{kInternalError,
                     "bdev " + bdev->name() +
                         " is busy (live channels); retry offline"};
    }
    JsonObject o;
    o["name"] = Json(bdev->name());
    o["num_blocks"] = Json(static_cast<int64_t>(bdev->num_blocks()));
    return Json(std::move(o));
  });

  server->register_method(
      "bdev_clone", [&manager, use_hbm, persistent](const Json& p) {
        // Volume clone: new malloc bdev + device-side range copy (HBM
        // rates; xGMI when cloning to another GPU via "device").
        BdevPtr src = manager.find(p.get_string("src"));
        if (!src) not_found("bdev " + p.get_string("src"));
        const std::string name = p.get_string("name");
        if (name.empty()) throw RpcError{kInvalidParams, "name required"};
        if (manager.find(name)) {
          throw RpcError{kInvalidParams, "bdev " + name + " already exists"};
        }
        BdevPtr clone;
        if (use_hbm && gpu_available() && src->device_base() != nullptr) {
          clone = create_hbm_bdev(
              name, src->block_size(), src->num_blocks(),
              static_cast<int>(p.get_int("device", src->gpu_device())),
              p.get_bool("persistent", persistent));
          int status = hbm_copy_sync(src.get(), 0, clone.get(), 0,
                                     src->size_bytes());
          if (status != kIoOk) {
            throw RpcError{kInternalError, "clone copy failed"};
          }
        } else {
          clone = create_malloc_bdev(name, src->block_size(),
                                     src->num_blocks());
          // Host-side copy for CPU bdevs (CI path).
          const uint64_t chunk = 8 << 20;
          std::vector<uint8_t> buf(chunk);
          for (uint64_t off = 0; off < src->size_bytes(); off += chunk) {
            const uint64_t n = std::min(chunk, src->size_bytes() - off);
            if (bdev_read_sync(src.get(), off, buf.data(), n) != kIoOk ||
                bdev_write_sync(clone.get(), off, buf.data(), n) != kIoOk) {
              throw RpcError{kInternalError, "clone copy failed"};
            }
          }
        }
        clone->set_product(src->product_name());
        manager.add(clone);
        return Json(name);
      });

  server->register_method("bdev_copy", [&manager](const Json& p) {
    // Device-side clone/rebuild: HBM->HBM at memory/xGMI rates.
    BdevPtr src = manager.find(p.get_string("src"));
    BdevPtr dst = manager.find(p.get_string("dst"));
    if (!src) not_found("bdev " + p.get_string("src"));
    if (!dst) not_found("bdev " + p.get_string("dst"));
    const uint64_t length = p.get_int("length",
                                      static_cast<int64_t>(src->size_bytes()));
    int status = hbm_copy_sync(src.get(), p.get_int("src_offset", 0),
                               dst.get(), p.get_int("dst_offset", 0), length);
    if (status != kIoOk) {
      throw RpcError{kInvalidParams, "bdev_copy failed (HBM bdevs only)"};
    }
    return Json(JsonObject{});
  });

  // Stepped benchmarking with persistent queues (bench.py contract):
  // perf_session_start -> id; perf_session_step runs a fixed I/O count
  // on the session's live queues; perf_session_stop tears down.
  struct Sessions {
    std::mutex mutex;
    std::map<int64_t, std::shared_ptr<PerfSession>> by_id;
    int64_t next_id = 1;
  };
  auto sessions = std::make_shared<Sessions>();

  server->register_method(
      "perf_session_start", [&manager, sessions](const Json& p) {
        BdevPtr bdev = manager.find(p.get_string("bdev_name"));
        if (!bdev) not_found("bdev " + p.get_string("bdev_name"));
        auto session = std::make_shared<PerfSession>(
            bdev, p.get_string("workload", "randread"),
            static_cast<uint32_t>(p.get_int("io_size", 4096)),
            static_cast<uint32_t>(p.get_int("queue_depth", 32)),
            static_cast<int>(p.get_int("num_queues", 8)));
        std::lock_guard<std::mutex> lock(sessions->mutex);
        int64_t id = sessions->next_id++;
        sessions->by_id[id] = std::move(session);
        JsonObject o;
        o["session_id"] = Json(id);
        return Json(std::move(o));
      });

  server->register_method("perf_session_step", [sessions](const Json& p) {
    std::shared_ptr<PerfSession> session;  // keeps the session alive
    {                                      // across a concurrent stop
      std::lock_guard<std::mutex> lock(sessions->mutex);
      auto it = sessions->by_id.find(p.get_int("session_id"));
      if (it == sessions->by_id.end()) not_found("perf session");
      session = it->second;
    }
    PerfResult r = session->step(
        static_cast<uint64_t>(p.get_int("total_ios", 1 << 17)));
    JsonObject o;
    o["seconds"] = Json(r.seconds);
    o["io_count"] = Json(static_cast<int64_t>(r.io_count));
    o["iops"] = Json(r.iops);
    o["throughput_mbps"] = Json(r.throughput_mbps);
    o["lat_avg_us"] = Json(r.lat_avg_us);
    o["lat_p50_us"] = Json(r.lat_p50_us);
    o["lat_p90_us"] = Json(r.lat_p90_us);
    o["lat_p99_us"] = Json(r.lat_p99_us);
    o["lat_p999_us"] = Json(r.lat_p999_us);
    o["lat_max_us"] = Json(r.lat_max_us);
    return Json(std::move(o));
  });

  server->register_method("perf_session_stop", [sessions](const Json& p) {
    std::lock_guard<std::mutex> lock(sessions->mutex);
    sessions->by_id.erase(p.get_int("session_id"));
    return Json(JsonObject{});
  });

  server->register_method("perf_run", [&manager](const Json& p) {
    // In-daemon bdevperf: the fio-shaped measurement loop runs next to
    // the data path, so the RPC socket is not in the hot path.
    BdevPtr bdev = manager.find(p.get_string("bdev_name"));
    if (!bdev) not_found("bdev " + p.get_string("bdev_name"));
    PerfResult r = run_bdevperf(
        bdev.get(), p.get_string("workload", "randread"),
        static_cast<uint32_t>(p.get_int("io_size", 4096)),
        static_cast<uint32_t>(p.get_int("queue_depth", 32)),
        static_cast<int>(p.get_int("num_queues", 1)),
        p.get("seconds") ? p.get("seconds")->as_double() : 2.0,
        static_cast<uint64_t>(p.get_int("max_ios", 0)));
    JsonObject o;
    o["seconds"] = Json(r.seconds);
    o["io_count"] = Json(static_cast<int64_t>(r.io_count));
    o["iops"] = Json(r.iops);
    o["throughput_mbps"] = Json(r.throughput_mbps);
    o["lat_avg_us"] = Json(r.lat_avg_us);
    o["lat_p50_us"] = Json(r.lat_p50_us);
    o["lat_p90_us"] = Json(r.lat_p90_us);
    o["lat_p99_us"] = Json(r.lat_p99_us);
    o["lat_p999_us"] = Json(r.lat_p999_us);
    o["lat_max_us"] = Json(r.lat_max_us);
    return Json(std::move(o));
  });

  // --- configuration snapshot (SPDK save_config/load_config shape) -------
  // Emits {"subsystems":[{"subsystem":..., "config":[{"method":...,
  // "params":...}]}]} so a daemon restart (or `hipstored -c FILE`)
  // recreates the control-plane topology. Data does NOT survive —
  // malloc bdevs are RAM/HBM by contract (reference spec.md:116-119) —
  // this is the SPDK-compatible control-state checkpoint. Composite
  // and RBD bdevs are not yet emitted (construction params are not
  // retained for them).
  server->register_method("save_config", [&manager, targets, creations,
                                          rados_clusters](const Json&) {
    // Loopback RADOS clusters replay FIRST (with their bound ports)
    // so rbd bdevs whose mon_host points at them reconnect.
    JsonArray rados_cfg;
    {
      std::lock_guard<std::mutex> lock(rados_clusters->mutex);
      for (const auto& [port, entry] : rados_clusters->by_port) {
        JsonObject e;
        e["method"] = Json(std::string("rados_cluster_start"));
        e["params"] = entry.params;
        rados_cfg.push_back(Json(std::move(e)));
      }
    }
    JsonArray bdev_cfg;
    std::vector<std::pair<uint64_t, Json>> recorded;
    {
      std::lock_guard<std::mutex> lock(creations->mutex);
      for (const auto& [name, creation] : creations->by_name) {
        JsonObject entry;
        entry["method"] = Json(creation.method);
        entry["params"] = creation.params;
        recorded.emplace_back(creation.seq, Json(std::move(entry)));
      }
    }
    for (const auto& bdev : manager.list()) {
      if (bdev->product_name() != "Malloc disk") continue;
      {
        std::lock_guard<std::mutex> lock(creations->mutex);
        if (creations->by_name.count(bdev->name())) continue;
      }
      JsonObject params;
      params["name"] = Json(bdev->name());
      params["num_blocks"] = Json(static_cast<int64_t>(bdev->num_blocks()));
      params["block_size"] = Json(static_cast<int64_t>(bdev->block_size()));
      JsonObject entry;
      entry["method"] = Json(std::string("construct_malloc_bdev"));
      entry["params"] = Json(std::move(params));
      bdev_cfg.push_back(Json(std::move(entry)));
    }
    // Recorded creations (composites, RBD) in creation order so
    // dependencies replay before dependents.
    std::sort(recorded.begin(), recorded.end(),
              [](const auto& a, const auto& b) { return a.first < b.first; });
    for (auto& [seq, entry] : recorded) {
      bdev_cfg.push_back(std::move(entry));
    }
    JsonArray vhost_cfg;
    {
      auto& state = vhost_state();
      std::lock_guard<std::mutex> lock(state.mutex);
      for (const auto& [name, ctrl] : state.controllers) {
        JsonObject params;
        params["ctrlr"] = Json(name);
        params["cpumask"] = Json(ctrl.cpumask);
        JsonObject entry;
        if (!ctrl.blk_bdev.empty()) {
          params["dev_name"] = Json(ctrl.blk_bdev);
          params["readonly"] = Json(ctrl.blk_readonly);
          entry["method"] =
              Json(std::string("construct_vhost_blk_controller"));
          entry["params"] = Json(std::move(params));
          vhost_cfg.push_back(Json(std::move(entry)));
          continue;
        }
        entry["method"] = Json(std::string("construct_vhost_scsi_controller"));
        entry["params"] = Json(std::move(params));
        vhost_cfg.push_back(Json(std::move(entry)));
        for (int t = 0; t < kMaxScsiTargets; ++t) {
          const ScsiTarget& target = ctrl.targets[t];
          if (!target.used || target.luns.empty()) continue;
          JsonObject lp;
          lp["ctrlr"] = Json(name);
          lp["scsi_target_num"] = Json(static_cast<int64_t>(t));
          lp["bdev_name"] = Json(target.luns[0].bdev_name);
          JsonObject le;
          le["method"] = Json(std::string("add_vhost_scsi_lun"));
          le["params"] = Json(std::move(lp));
          vhost_cfg.push_back(Json(std::move(le)));
        }
      }
    }
    JsonArray nvmf_cfg;
    {
      std::lock_guard<std::mutex> lock(targets->mutex);
      for (const auto& [subnqn, entry] : targets->by_nqn) {
        JsonObject params;
        params["subnqn"] = Json(subnqn);
        params["listen_addr"] = Json(entry.listen_addr);
        params["port"] = Json(static_cast<int64_t>(entry.target->port()));
        params["digests"] = Json(entry.digests);
        JsonArray ns;
        for (const std::string& name : entry.ns_bdevs) {
          ns.push_back(Json(name));
        }
        params["bdevs"] = Json(std::move(ns));
        JsonObject e;
        e["method"] = Json(std::string("nvmf_create_target"));
        e["params"] = Json(std::move(params));
        nvmf_cfg.push_back(Json(std::move(e)));
      }
    }
    auto subsystem = [](const char* name, JsonArray cfg) {
      JsonObject o;
      o["subsystem"] = Json(std::string(name));
      o["config"] = Json(std::move(cfg));
      return Json(std::move(o));
    };
    JsonArray nbd_cfg;
    for (const auto& [bdev_name, device] : nbd_list()) {
      JsonObject params;
      params["bdev_name"] = Json(bdev_name);
      params["nbd_device"] = Json(device);
      JsonObject e;
      e["method"] = Json(std::string("start_nbd_disk"));
      e["params"] = Json(std::move(params));
      nbd_cfg.push_back(Json(std::move(e)));
    }
    JsonArray ublk_cfg;
    for (const UblkDisk& disk : ublk_list()) {
      JsonObject params;
      params["bdev_name"] = Json(disk.bdev_name);
      JsonObject e;
      e["method"] = Json(std::string("ublk_start_disk"));
      e["params"] = Json(std::move(params));
      ublk_cfg.push_back(Json(std::move(e)));
    }
    JsonArray subsystems;
    subsystems.push_back(subsystem("rados", std::move(rados_cfg)));
    subsystems.push_back(subsystem("bdev", std::move(bdev_cfg)));
    subsystems.push_back(subsystem("vhost", std::move(vhost_cfg)));
    subsystems.push_back(subsystem("nvmf", std::move(nvmf_cfg)));
    subsystems.push_back(subsystem("nbd", std::move(nbd_cfg)));
    subsystems.push_back(subsystem("ublk", std::move(ublk_cfg)));
    JsonObject out;
    out["subsystems"] = Json(std::move(subsystems));
    return Json(std::move(out));
  });

  server->register_method("load_config", [server](const Json& p) {
    const Json* subsystems = p.get("subsystems");
    if (subsystems == nullptr) {
      throw RpcError{kInvalidParams, "subsystems required"};
    }
    int64_t applied = 0;
    for (const Json& subsystem : subsystems->as_array()) {
      const Json* config = subsystem.get("config");
      if (config == nullptr) continue;
      for (const Json& entry : config->as_array()) {
        JsonObject request;
        request["jsonrpc"] = Json(std::string("2.0"));
        request["id"] = Json(int64_t{0});
        request["method"] = Json(entry.get_string("method"));
        if (const Json* params = entry.get("params")) {
          request["params"] = *params;
        }
        Json reply = server->dispatch(Json(std::move(request)));
        if (reply.get("error") != nullptr) {
          throw RpcError{kInvalidParams,
                         "load_config failed at " +
                             entry.get_string("method") + ": " +
                             reply.get("error")->get_string("message")};
        }
        ++applied;
      }
    }
    return Json(applied);
  });

  // Modern SPDK method-name aliases (SPDK renamed its RPC surface in
  // v19.x; the reference client uses the old names, newer stock
  // tooling the new ones — serve both).
  for (const auto& [new_name, old_name] :
       std::initializer_list<std::pair<const char*, const char*>>{
           {"bdev_get_bdevs", "get_bdevs"},
           {"bdev_get_iostat", "get_bdevs_iostat"},
           {"bdev_malloc_create", "construct_malloc_bdev"},
           {"bdev_malloc_delete", "delete_bdev"},
           {"bdev_aio_create", "construct_aio_bdev"},
           {"bdev_aio_delete", "delete_bdev"},
           {"bdev_rbd_create", "construct_rbd_bdev"},
           {"bdev_rbd_delete", "delete_bdev"},
           {"nbd_start_disk", "start_nbd_disk"},
           {"nbd_get_disks", "get_nbd_disks"},
           {"nbd_stop_disk", "stop_nbd_disk"},
           {"vhost_create_scsi_controller",
            "construct_vhost_scsi_controller"},
           {"vhost_scsi_controller_add_target", "add_vhost_scsi_lun"},
           {"vhost_scsi_controller_remove_target",
            "remove_vhost_scsi_target"},
           {"vhost_create_blk_controller", "construct_vhost_blk_controller"},
           {"vhost_delete_controller", "remove_vhost_controller"},
           {"vhost_get_controllers", "get_vhost_controllers"},
           {"rpc_get_methods", "get_rpc_methods"},
       }) {
    server->register_alias(new_name, old_name);
  }
}

}  // namespace hipstore
