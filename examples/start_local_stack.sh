#!/bin/bash
# Demo: bring up the full oim-amd stack on one machine (the reference's
# `make start` demo cluster, test/start-stop.make, without VMs) and run
# a provisioning round trip + fio-shaped benchmark against it.
#
#   examples/start_local_stack.sh [workdir]
#
# Uses CPU-mode hipstored when no GPU is present, HBM mode otherwise.
set -euo pipefail

REPO="$(cd "$(dirname "$0")/.." && pwd)"
WORK="${1:-$(mktemp -d /tmp/oim-stack.XXXXXX)}"
mkdir -p "$WORK"
echo "workdir: $WORK"
cd "$REPO"

PIDS=()
cleanup() {
    for pid in "${PIDS[@]:-}"; do kill "$pid" 2>/dev/null || true; done
    wait 2>/dev/null || true
}
trap cleanup EXIT

# 1. data-path daemon (per GPU in production; one here)
./bin/hipstored -S "$WORK/hipstored.sock" 2>"$WORK/hipstored.log" &
PIDS+=($!)
for _ in $(seq 50); do [ -S "$WORK/hipstored.sock" ] && break; sleep 0.1; done

# 2. registry (durable file DB)
python -m oim_amd.cmd.oim_registry --endpoint "unix://$WORK/registry.sock" \
    --db-file "$WORK/registry.json" 2>"$WORK/registry.log" &
PIDS+=($!)
sleep 0.5

# 3. per-card controller with self-registration
python -m oim_amd.cmd.oim_controller --controllerid demo-gpu0 \
    --hipstored-socket "$WORK/hipstored.sock" \
    --endpoint "unix://$WORK/controller.sock" \
    --controller-address "unix://$WORK/controller.sock" \
    --registry "unix://$WORK/registry.sock" --registry-delay 5 \
    2>"$WORK/controller.log" &
PIDS+=($!)
sleep 1

# 4. CSI driver in remote mode
python -m oim_amd.cmd.oim_csi_driver --nodeid demo-node \
    --endpoint "unix://$WORK/csi.sock" \
    --oim-registry-address "unix://$WORK/registry.sock" \
    --controller-id demo-gpu0 2>"$WORK/csi.log" &
PIDS+=($!)
sleep 1

echo "--- registry contents (oimctl get) ---"
python -m oim_amd.cmd.oimctl --registry "unix://$WORK/registry.sock" get

echo "--- provisioning a volume through the proxy ---"
python - "$WORK" <<'EOF'
import sys
import grpc
from oim_amd import spec
from oim_amd.common.server import grpc_target

work = sys.argv[1]
with grpc.insecure_channel(grpc_target(f"unix://{work}/registry.sock")) as ch:
    stub = spec.ControllerStub(ch)
    md = ((spec.CONTROLLER_ID_KEY, "demo-gpu0"),)
    stub.ProvisionMallocBDev(
        spec.ProvisionMallocBDevRequest(bdev_name="demo-vol", size=64 << 20),
        metadata=md, timeout=30)
    reply = stub.MapVolume(
        spec.MapVolumeRequest(volume_id="demo-vol", malloc=spec.MallocParams()),
        metadata=md, timeout=30)
    print(f"mapped demo-vol at SCSI target {reply.scsi_disk.target} lun "
          f"{reply.scsi_disk.lun}")
EOF

echo "--- snapshot / clone / resize via oimctl ---"
oimctl_cmd() {
    python -m oim_amd.cmd.oimctl --registry "unix://$WORK/registry.sock" "$@"
}
oimctl_cmd clone  --controller demo-gpu0 demo-vol demo-vol-snap
oimctl_cmd resize --controller demo-gpu0 demo-vol 128MiB
oimctl_cmd check  --controller demo-gpu0 demo-vol-snap
oimctl_cmd provision --controller demo-gpu0 demo-vol-snap 0

echo "--- fio-shaped benchmark on the mapped volume ---"
python -m oim_amd.bench.fio_harness --socket "$WORK/hipstored.sock" \
    --bdev demo-vol --rw randread --bs 4096 --iodepth 32 --numjobs 4 \
    --runtime 3 --perfdash

echo "--- vhost-user (VM-attach) data-path benchmark (on a clone:"
echo "    demo-vol itself is claimed by its SCSI mapping) ---"
oimctl_cmd clone --controller demo-gpu0 demo-vol demo-vbench
python -m oim_amd.bench.vhost_harness --socket "$WORK/hipstored.sock" \
    --bdev demo-vbench --iodepth 16 --runtime 2 --ctrlr demo-vhbench \
    --master native
oimctl_cmd provision --controller demo-gpu0 demo-vbench 0

echo "demo complete; logs in $WORK"
