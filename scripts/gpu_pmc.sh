#!/bin/bash
# PMC counter capture for the gfx950 flag kernels (k_set_flag /
# k_wait_and_set / k_pull_copy): runs the kernel-fallback pingpong under
# rocprofv3 --pmc.  NOTE: --pmc must NOT be combined with sys/runtime/hip
# trace domains (pool rule); only --kernel-trace here.
set -u
cd "$(dirname "$0")/.."
export TMPDIR=/tmp
mkdir -p gpurun_out/pmc
rocprofv3 --list-avail > gpurun_out/pmc/avail.txt 2>&1

# pick counters that exist on this build
want="SQ_WAVES SQ_WAVE_CYCLES SQ_INSTS_VALU TCC_REQ_sum TCC_HIT_sum TCC_MISS_sum TCC_EA0_RDREQ_sum TCC_EA0_WRREQ_sum"
PMC=""
for c in $want; do
    grep -q "\b$c\b" gpurun_out/pmc/avail.txt && PMC="$PMC $c"
done
echo "counters:$PMC"
[ -z "$PMC" ] && { echo "no counters matched"; exit 1; }

cd /tmp
MPIX_DISABLE_MEMOPS=1 timeout 150 rocprofv3 --pmc $PMC --kernel-trace \
    -d "$OLDPWD/gpurun_out/pmc" -o flagk --output-format csv -- \
    "$OLDPWD/bench/bin/pingpong" 15 50 \
    > "$OLDPWD/gpurun_out/pmc/run.log" 2>&1
rc=$?
echo "pmc_rc=$rc"
ls -la "$OLDPWD/gpurun_out/pmc" | tail -5
exit $rc
