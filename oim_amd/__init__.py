"""oim-amd: an MI355X-native storage-accelerator control plane.

A from-scratch rebuild of the capabilities of intel/oim (reference:
/root/reference): an oim-registry / oim-controller / oim-csi-driver triad
speaking the oim.v0 gRPC API over mutual TLS, fronting a CDNA4 (gfx950)
data-path daemon ("hipstored") whose block devices live in MI355X HBM3E.

Layer map (mirrors reference SURVEY.md section 1):
  - oim_amd.spec       the oim.v0 protobuf API (runtime-built descriptors)
  - oim_amd.registry   registry service: key/value DB + transparent gRPC proxy
  - oim_amd.controller per-"card" (per-GPU) agent driving hipstored
  - oim_amd.csidriver  CSI driver (Identity/Controller/Node), local/remote
  - oim_amd.hipstore   JSON-RPC client for hipstored (reference pkg/spdk)
  - oim_amd.common     gRPC server/dial/TLS-CN helpers, PCI BDF, paths
  - oim_amd.log        context-attached structured logging (reference pkg/log)
  - oim_amd._hipstore  pybind11 module over the C++/HIP data-path core
"""

__version__ = "0.1.0"

# Must be in the environment BEFORE libamdhip64 initializes: the
# persistent engine needs one real hardware queue per service kernel,
# and ROCm's 4-queue default gang-schedules resident kernels (measured:
# 30-50x collapse, and a full wedge when per-queue and shared service
# kernels mix). The C++ static constructor in _hipstore is TOO LATE for
# in-process imports — shared-library dependencies (libamdhip64) run
# their initializers first — so set it here, before any native import.
import os as _os

_os.environ.setdefault("GPU_MAX_HW_QUEUES", "24")
