#!/bin/bash
# PMC counter capture (rocprofv3 --pmc + --kernel-trace ONLY — pool rule).
# CAVEAT discovered on-box: counter collection serializes kernel dispatch,
# so any workload where one kernel's exit depends on another kernel (or on
# proxy progress signalled between running kernels) DEADLOCKS under --pmc.
# Safe target: the MFMA GEMM with publish disabled (independent kernels).
set -u
cd "$(dirname "$0")/.."
export TMPDIR=/tmp
mkdir -p gpurun_out/pmc
PMC="SQ_WAVES SQ_WAVE_CYCLES SQ_INSTS_MFMA SQ_VALU_MFMA_BUSY_CYCLES SQ_INSTS_LDS SQ_LDS_BANK_CONFLICT TCC_REQ_sum TCC_EA0_RDREQ_sum"
echo "counters: $PMC"
cd /tmp
timeout 150 rocprofv3 --pmc $PMC --kernel-trace \
    -d "$OLDPWD/gpurun_out/pmc" -o gemm --output-format csv -- \
    "$OLDPWD/bench/bin/gemm_pready" 4096 4096 4096 3 \
    > "$OLDPWD/gpurun_out/pmc/run.log" 2>&1
rc=$?
echo "pmc_rc=$rc"
ls "$OLDPWD/gpurun_out/pmc" | tail -6
exit $rc
