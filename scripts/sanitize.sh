#!/bin/bash
# Sanitizer tiers (host-only, no GPU needed): builds instrumented copies of
# the library out-of-tree and runs the concurrency stress against them.
#   scripts/sanitize.sh tsan   — ThreadSanitizer (race detection)
#   scripts/sanitize.sh asan   — AddressSanitizer + LeakSanitizer
# The production build is untouched.
set -eu
cd "$(dirname "$0")/.."
MODE="${1:-tsan}"
case "$MODE" in
  tsan) FLAG=-fsanitize=thread ;;
  asan) FLAG=-fsanitize=address ;;
  *) echo "usage: $0 tsan|asan"; exit 2 ;;
esac
OUT="/tmp/mpix-$MODE"
mkdir -p "$OUT"
SRCS="state init proxy enqueue partitioned"
TSRCS="bootstrap native mpi"
for f in $SRCS; do
    hipcc -O1 -g $FLAG -std=c++17 -fPIC --offload-arch=gfx950 \
        -Iinclude -I/opt/conda/include -c src/$f.cpp -o "$OUT/$f.o" &
done
for f in $TSRCS; do
    hipcc -O1 -g $FLAG -std=c++17 -fPIC --offload-arch=gfx950 \
        -Iinclude -I/opt/conda/include -c src/transport/$f.cpp -o "$OUT/$f.o" &
done
wait
hipcc --offload-arch=gfx950 $FLAG "$OUT"/*.o -shared -L/opt/conda/lib -lmpi \
    -Wl,-rpath,/usr/lib/x86_64-linux-gnu -Wl,-rpath,/opt/conda/lib \
    -o "$OUT/libmpix.so"
# LLVM clang links the matching sanitizer runtime (GCC's libtsan lacks the
# __tsan_memcpy entry points the LLVM-instrumented library references)
/opt/rocm/lib/llvm/bin/amdclang++ -O1 -g $FLAG -Iinclude -I/opt/conda/include \
    -x c++ test/src/stress_tsan.c -x none -L"$OUT" -lmpix \
    /opt/conda/lib/libmpi.so -Wl,-rpath,"$OUT" \
    -Wl,-rpath,/usr/lib/x86_64-linux-gnu -Wl,-rpath,/opt/conda/lib \
    -o "$OUT/stress"
MPIX_FORCE_NO_GPU=1 MPIX_NFLAGS=128 "$OUT/stress"
echo "sanitize($MODE): OK"
