#!/bin/bash
# One-shot GPU validation for an MI355X box (run via gpurun).  Writes all
# artifacts under gpurun_out/ so they merge back to the dev container.
# Usage: bash scripts/gpu_ci.sh [quick|full]
set -u
cd "$(dirname "$0")/.."
export TMPDIR=/tmp
mkdir -p gpurun_out
MODE="${1:-quick}"
PASS=0; FAIL=0
note() { echo "=== $1 ==="; }
step() { # step <name> <timeout_s> <cmd...>
    local name="$1" t="$2"; shift 2
    note "$name"
    if timeout "$t" "$@" > "gpurun_out/${name}.log" 2>&1; then
        echo "OK"; PASS=$((PASS+1))
    else
        echo "FAIL rc=$?"; FAIL=$((FAIL+1)); tail -6 "gpurun_out/${name}.log"
    fi
}

step pytest_gpu 600 python -m pytest tests -q -m gpu
step smoke 120 python -c 'from __graft_entry__ import smoke; smoke()'
# headline stability: 5 back-to-back repetitions -> median +- spread
note bench_repeat
for i in 1 2 3 4 5; do
    timeout 150 python bench.py --steps 20 --warmup 5 \
        > "gpurun_out/bench_rep$i.log" 2>&1 \
        && tail -1 "gpurun_out/bench_rep$i.log" || echo "rep$i FAIL rc=$?"
done
cp gpurun_out/bench_rep1.log gpurun_out/bench_1gpu.log 2>/dev/null || true

export PATH=/opt/conda/bin:$PATH
step pingpong 120 mpiexec -np 2 bench/bin/pingpong 24 100
step halo3d 90 mpiexec -np 2 bench/bin/halo3d 192 256 256 10
step gemm_check 90 bench/bin/gemm_pready --check
step gemm_overlap 120 mpiexec -np 2 bench/bin/gemm_pready 4096 4096 4096 5

if [ "$MODE" = full ]; then
    # classic-protocol A/B (fast-wait is the default since r02)
    note classic_pytest
    MPIX_FAST_WAIT=0 timeout 240 python -m pytest tests/test_gpu.py -q -m gpu \
        -k "loopback or partitioned" > gpurun_out/classic_pytest.log 2>&1 \
        && echo OK || { echo "FAIL rc=$?"; tail -4 gpurun_out/classic_pytest.log; }
    note classic_pingpong
    MPIX_FAST_WAIT=0 timeout 100 mpiexec -np 2 bench/bin/pingpong 18 100 \
        > gpurun_out/classic_pingpong.log 2>&1 \
        && { echo OK; head -6 gpurun_out/classic_pingpong.log; } \
        || echo "FAIL rc=$?"
    note stats_pingpong
    MPIX_STATS=1 timeout 100 mpiexec -np 2 bench/bin/pingpong 14 200 \
        > gpurun_out/stats_pingpong.log 2>&1 \
        && { echo OK; grep "per-leg" gpurun_out/stats_pingpong.log; } \
        || echo "FAIL rc=$?"
    note waitall_wave_kernel
    # single-wave waitall kernel under the kernel fallback: the r01 in-situ
    # hang was the stream->HSA-queue aliasing deadlock (see add_flag_node);
    # with the transport copy stream on its own queue this should pass
    MPIX_WAITALL_KERNEL=1 MPIX_DISABLE_MEMOPS=1 timeout 120 \
        python -m pytest tests/test_gpu.py -q -m gpu -k "waitall or loopback" \
        > gpurun_out/waitall_wave.log 2>&1 \
        && echo OK || { echo "FAIL rc=$?"; tail -4 gpurun_out/waitall_wave.log; }
    note kernelwait_pingpong
    MPIX_DISABLE_MEMOPS=1 timeout 100 mpiexec -np 2 bench/bin/pingpong 15 100 \
        > gpurun_out/kernelwait_pingpong.log 2>&1 \
        && { echo OK; head -5 gpurun_out/kernelwait_pingpong.log; } \
        || echo "FAIL rc=$?"
    note flush_probe
    ( timeout 60 tools/bin/flush_probe 1 && timeout 60 tools/bin/flush_probe 0 ) \
        > gpurun_out/flush_probe.log 2>&1 \
        && { echo OK; cat gpurun_out/flush_probe.log; } \
        || { echo "FAIL rc=$?"; cat gpurun_out/flush_probe.log; }
    note devpush_pingpong
    MPIX_DEV_PUSH_MAX=65536 timeout 100 mpiexec -np 2 bench/bin/pingpong 15 50 \
        > gpurun_out/devpush_pingpong.log 2>&1 \
        && { echo OK; head -5 gpurun_out/devpush_pingpong.log; } \
        || echo "FAIL rc=$?"
    step gemm_8k 90 bench/bin/gemm_pready 8192 8192 8192 5
    step gemm_v2_check 90 bench/bin/gemm_pready_v2 --check
    step gemm_v2_8k 90 bench/bin/gemm_pready_v2 8192 8192 8192 5
    tail -1 gpurun_out/gemm_v2_8k.log
    note rocprof
    ( cd /tmp && timeout 150 rocprofv3 --kernel-trace --stats \
        -d "$OLDPWD/gpurun_out/prof" -o ci \
        -- python "$OLDPWD/bench.py" --steps 5 --warmup 2 --msg-mib 64 \
           --pp-iters 100 > "$OLDPWD/gpurun_out/rocprof.log" 2>&1 ) \
        && echo OK || echo "FAIL rc=$?"
fi

echo "gpu_ci: $PASS ok, $FAIL failed ($MODE)"
exit "$FAIL"
