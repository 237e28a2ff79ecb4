// Python bindings over the hipstore C++/HIP core (oim_amd._hipstore).
//
// Used by tests, the fio-shaped benchmark harness (bench.py) and the
// CSI driver's local mode. The daemon (hipstored) links the same core,
// so everything measured through these bindings is the same code the
// control plane drives over JSON-RPC.

#include <pybind11/functional.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstring>
#include <stdexcept>

#include "hipstore/bdev.h"
#include "hipstore/composite.h"
#include "hipstore/crc32c.h"
#include "hipstore/engine.h"
#include "hipstore/nvmf.h"
#include "hipstore/rados.h"
#include "hipstore/vhost_master.h"

namespace py = pybind11;
using namespace hipstore;

namespace {

py::bytes bdev_read_py(Bdev& bdev, uint64_t offset, uint64_t length) {
  void* bounce = alloc_pinned(length);
  int status = bdev_read_sync(&bdev, offset, bounce, length);
  if (status != kIoOk) {
    free_pinned(bounce);
    throw std::runtime_error("bdev read failed: status " + std::to_string(status));
  }
  py::bytes out(static_cast<const char*>(bounce), length);
  free_pinned(bounce);
  return out;
}

void bdev_write_py(Bdev& bdev, uint64_t offset, py::buffer data) {
  py::buffer_info info = data.request();
  const uint64_t length = static_cast<uint64_t>(info.size) * info.itemsize;
  void* bounce = alloc_pinned(length);
  memcpy(bounce, info.ptr, length);
  int status = bdev_write_sync(&bdev, offset, bounce, length);
  free_pinned(bounce);
  if (status != kIoOk) {
    throw std::runtime_error("bdev write failed: status " + std::to_string(status));
  }
}

void bdev_fill_py(Bdev& bdev, uint64_t offset, int value, uint64_t length) {
  int status = bdev_fill_sync(&bdev, offset, static_cast<uint8_t>(value), length);
  if (status != kIoOk) {
    throw std::runtime_error("bdev fill failed: status " + std::to_string(status));
  }
}

py::dict perf_to_dict(const PerfResult& r) {
  py::dict d;
  d["seconds"] = r.seconds;
  d["io_count"] = r.io_count;
  d["iops"] = r.iops;
  d["throughput_mbps"] = r.throughput_mbps;
  d["lat_avg_us"] = r.lat_avg_us;
  d["lat_p50_us"] = r.lat_p50_us;
  d["lat_p90_us"] = r.lat_p90_us;
  d["lat_p99_us"] = r.lat_p99_us;
  d["lat_p999_us"] = r.lat_p999_us;
  d["lat_max_us"] = r.lat_max_us;
  return d;
}

}  // namespace

PYBIND11_MODULE(_hipstore, m) {
  m.doc() = "MI355X hipstore data-path core (HBM bdevs, LDS-staged block I/O)";

  m.def("gpu_available", &gpu_available);
  m.def("gpu_device_count", &gpu_device_count);
  m.def("gpu_pci_address", &gpu_pci_address, py::arg("device"));

  py::class_<Bdev, BdevPtr>(m, "Bdev")
      .def_property_readonly("name", &Bdev::name)
      .def_property_readonly("product_name", &Bdev::product_name)
      .def_property_readonly("uuid", &Bdev::uuid)
      .def_property_readonly("block_size", &Bdev::block_size)
      .def_property_readonly("num_blocks", &Bdev::num_blocks)
      .def_property_readonly("size_bytes", &Bdev::size_bytes)
      .def("resize",
           [](hipstore::Bdev& b, uint64_t num_blocks) {
             return b.resize(num_blocks);
           },
           py::arg("num_blocks"))
      .def("read", &bdev_read_py, py::arg("offset"), py::arg("length"),
           py::call_guard<py::gil_scoped_release>())
      .def("write", &bdev_write_py, py::arg("offset"), py::arg("data"))
      .def("fill", &bdev_fill_py, py::arg("offset"), py::arg("value"),
           py::arg("length"), py::call_guard<py::gil_scoped_release>());

  m.def("create_malloc_bdev", &create_malloc_bdev, py::arg("name"),
        py::arg("block_size"), py::arg("num_blocks"));
  m.def("create_hbm_bdev", &create_hbm_bdev, py::arg("name"),
        py::arg("block_size"), py::arg("num_blocks"), py::arg("device") = 0,
        py::arg("persistent") = false);
  m.def("create_file_bdev", &create_file_bdev, py::arg("name"),
        py::arg("path"), py::arg("block_size") = 512);
  m.def("create_striped_bdev", &create_striped_bdev, py::arg("name"),
        py::arg("children"), py::arg("stripe_size") = 131072);
  m.def("create_replicated_bdev", &create_replicated_bdev, py::arg("name"),
        py::arg("children"));

  py::class_<RadosCluster, std::shared_ptr<RadosCluster>>(m, "RadosCluster")
      .def("port", &RadosCluster::port)
      .def("object_count", &RadosCluster::object_count)
      .def("stop", &RadosCluster::stop,
           py::call_guard<py::gil_scoped_release>());
  m.def("start_rados_cluster", &start_rados_cluster, py::arg("port") = 0,
        py::arg("arena_mb") = 512, py::arg("use_hbm") = false,
        py::arg("device") = 0, py::arg("object_bytes") = 4ull << 20);
  m.def("create_rbd_bdev", &create_rbd_bdev, py::arg("name"),
        py::arg("mon_host"), py::arg("pool"), py::arg("image"),
        py::arg("block_size") = 512,
        py::arg("default_size_bytes") = 64ull << 20,
        py::arg("object_bytes") = 4ull << 20);

  py::class_<NvmfTcpTarget, std::shared_ptr<NvmfTcpTarget>>(m, "NvmfTcpTarget")
      .def_property_readonly("port", &NvmfTcpTarget::port)
      .def("add_namespace", &NvmfTcpTarget::add_namespace, py::arg("bdev"))
      .def("stop", &NvmfTcpTarget::stop,
           py::call_guard<py::gil_scoped_release>());
  m.def("start_nvmf_tcp_target", &start_nvmf_tcp_target,
        py::arg("listen_addr") = "", py::arg("port") = 0,
        py::arg("subnqn") = "nqn.2026-01.com.amd:oim-amd",
        py::arg("enable_digests") = true);
  m.def("create_nvmf_tcp_bdev", &create_nvmf_tcp_bdev, py::arg("name"),
        py::arg("traddr"), py::arg("trsvcid"), py::arg("subnqn"),
        py::arg("nsid") = 1, py::arg("enable_digests") = true,
        py::call_guard<py::gil_scoped_release>());

  py::class_<VhostMasterSession>(m, "VhostMasterSession")
      .def(py::init<const std::string&, const std::string&, int, int,
                    uint32_t, uint32_t, uint64_t>(),
           py::arg("socket_path"), py::arg("personality") = "scsi",
           py::arg("num_rings") = 4, py::arg("iodepth") = 32,
           py::arg("io_size") = 4096, py::arg("block_size") = 512,
           py::arg("capacity_bytes") = 0,
           py::call_guard<py::gil_scoped_release>())
      .def("run", [](VhostMasterSession& s, uint64_t total_ios,
                     const std::string& workload) {
        PerfResult r;
        {
          py::gil_scoped_release release;
          r = s.run(total_ios, workload);
        }
        return perf_to_dict(r);
      }, py::arg("total_ios"), py::arg("workload") = "randread");

  m.def(
      "vhost_master_bench",
      [](const std::string& socket_path, const std::string& personality,
         int num_rings, int iodepth, uint32_t io_size,
         const std::string& workload, uint64_t total_ios,
         uint32_t block_size, uint64_t capacity_bytes) {
        PerfResult r;
        {
          py::gil_scoped_release release;
          r = vhost_master_bench(socket_path, personality, num_rings,
                                 iodepth, io_size, workload, total_ios,
                                 block_size, capacity_bytes);
        }
        return perf_to_dict(r);
      },
      py::arg("socket_path"), py::arg("personality") = "scsi",
      py::arg("num_rings") = 4, py::arg("iodepth") = 32,
      py::arg("io_size") = 4096, py::arg("workload") = "randread",
      py::arg("total_ios") = 100000, py::arg("block_size") = 512,
      py::arg("capacity_bytes") = 0);

  py::class_<PerfSession>(m, "PerfSession")
      .def(py::init<BdevPtr, std::string, uint32_t, uint32_t, int>(),
           py::arg("bdev"), py::arg("workload") = "randread",
           py::arg("io_size") = 4096, py::arg("queue_depth") = 32,
           py::arg("num_queues") = 8)
      .def("step", [](PerfSession& s, uint64_t total_ios) {
        PerfResult r;
        {
          py::gil_scoped_release release;
          r = s.step(total_ios);
        }
        return perf_to_dict(r);
      }, py::arg("total_ios"));

  m.def(
      "run_bdevperf",
      [](BdevPtr bdev, const std::string& workload, uint32_t io_size,
         uint32_t queue_depth, int num_queues, double seconds,
         uint64_t max_ios) {
        PerfResult r;
        {
          py::gil_scoped_release release;
          r = run_bdevperf(bdev.get(), workload, io_size, queue_depth,
                           num_queues, seconds, max_ios);
        }
        return perf_to_dict(r);
      },
      py::arg("bdev"), py::arg("workload") = "randread",
      py::arg("io_size") = 4096, py::arg("queue_depth") = 32,
      py::arg("num_queues") = 1, py::arg("seconds") = 2.0,
      py::arg("max_ios") = 0);

  m.def("hbm_info", [](int device) {
    auto [total, free_bytes] = hbm_info(device);
    return py::make_tuple(total, free_bytes);
  }, py::arg("device") = 0);

  m.def("hbm_copy", [](BdevPtr src, uint64_t src_offset, BdevPtr dst,
                       uint64_t dst_offset, uint64_t length) {
    int status;
    {
      py::gil_scoped_release release;
      status = hbm_copy_sync(src.get(), src_offset, dst.get(), dst_offset,
                             length);
    }
    if (status != kIoOk) {
      throw std::runtime_error("hbm_copy failed: status " +
                               std::to_string(status));
    }
  }, py::arg("src"), py::arg("src_offset"), py::arg("dst"),
     py::arg("dst_offset"), py::arg("length"),
     "Device-side range copy between HBM bdevs (LDS-staged kernel "
     "same-device, xGMI peer copy cross-device)");

  m.def("persistent_probe", &persistent_probe, py::arg("device") = 0,
        py::arg("flags") = 1,
        py::call_guard<py::gil_scoped_release>());
  m.def("persistent_kernel_probe", &persistent_kernel_probe,
        py::arg("device") = 0, py::arg("variant") = 0,
        py::call_guard<py::gil_scoped_release>());

  m.def("persistent_stats", [] {
    py::dict d;
    d["launches"] = hipstore::persistent_stat(0);
    d["relaunches"] = hipstore::persistent_stat(1);
    d["stall_queries"] = hipstore::persistent_stat(2);
    return d;
  });

  m.def("crc32c_combine", &crc32c_combine, py::arg("crc1"), py::arg("crc2"),
        py::arg("len2"));

  m.def("crc32c_table", [](py::buffer data, uint32_t init) {
    py::buffer_info info = data.request();
    return crc32c_table(init, info.ptr,
                        static_cast<size_t>(info.size) * info.itemsize);
  }, py::arg("data"), py::arg("init") = 0);
  m.def("crc32c", [](py::buffer data, uint32_t init) {
    py::buffer_info info = data.request();
    return crc32c_sw(init, info.ptr, static_cast<size_t>(info.size) * info.itemsize);
  }, py::arg("data"), py::arg("init") = 0);

  m.def("crc32c_gpu_blocks", [](BdevPtr bdev, uint64_t offset, uint32_t block,
                                uint32_t count) {
    std::vector<uint32_t> out(count);
    {
      py::gil_scoped_release release;
      crc32c_hbm_blocks(bdev.get(), offset, block, count, out.data());
    }
    return out;
  }, py::arg("bdev"), py::arg("offset"), py::arg("block"), py::arg("count"),
     "CRC32C of `count` consecutive `block`-byte blocks of an HBM bdev, "
     "computed on-GPU");
}
