#!/bin/bash
# Round-2 perf batch: wedge-matrix re-check with the fixed engine,
# vhost data-path A/B (pipeline on/off), shared-engine A/B, and a
# rocprofv3 kernel-stats refresh for profiles/.
set -u
export HSA_ENABLE_IPC_MODE_LEGACY=0 PYTHONUNBUFFERED=1
OUT=gpurun_out/perf_batch.log
: > "$OUT"
log() { echo "$@" >> "$OUT"; }

log "=== ublk device-cgroup probe"
minor=$(awk '/ublk/{print $1}' /proc/misc)
mknod /dev/ublk-control c 10 "$minor" 2>> "$OUT" || true
python -c "
import os
try:
    fd = os.open('/dev/ublk-control', os.O_RDWR)
    print('open OK fd', fd)
    os.close(fd)
except OSError as e:
    print('open failed:', e)
" >> "$OUT" 2>&1
ls -la /dev/ublk-control >> "$OUT" 2>&1
cat /proc/self/status | grep -i cap >> "$OUT" 2>&1

log "=== wedge matrix (fixed engine)"
timeout -s KILL 700 bash tools/wedge_experiments.sh >> "$OUT" 2>&1 \
    || log "wedge rc=$?"

log "=== vhost data path: pipeline A/B (scsi + blk, via bench frontend)"
for pipe in 0 1; do
    for pers in scsi blk; do
        log "--- HIPSTORE_VHOST_PIPELINE=$pipe personality=$pers"
        timeout -s KILL 200 env HIPSTORE_VHOST_PIPELINE=$pipe \
            python bench.py --steps 6 --warmup 2 --frontend vhost \
            --personality $pers --queue-depth 32 --vhost-numjobs 4 \
            2> /dev/null | tail -1 >> "$OUT" || log "vhost $pipe/$pers rc=$?"
    done
done

log "=== shared-engine A/B (bench shape)"
timeout -s KILL 200 env HIPSTORE_SHARED_WORKERS=128 \
    python bench.py --steps 10 --warmup 3 --engine shared 2>/dev/null \
    | tail -1 >> "$OUT" || log "shared bench rc=$?"

log "=== rocprofv3 kernel stats (perf_run on persistent engine)"
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
timeout -s KILL 240 rocprofv3 --kernel-trace --stats \
    -d gpurun_out/prof_r2 -o r2stats -- \
    python -c "
import sys; sys.path.insert(0, '.')
import oim_amd._hipstore as hs
b = hs.create_hbm_bdev('prof', 4096, 1 << 21, device=0, persistent=True)
r = hs.run_bdevperf(b, 'randread', 4096, 32, 14, 6.0)
print('profiled %.0f IOPS p99=%sus' % (r['iops'], r['lat_p99_us']))
" >> "$OUT" 2>&1 || log "rocprof rc=$?"
ls gpurun_out/prof_r2 >> "$OUT" 2>&1 || true

cat "$OUT"
