#!/usr/bin/env bash
# ASAN/UBSAN pass over the native core: rebuild _core with sanitizers, run
# the CPU protocol suite under libasan, then restore the normal build.
# (The rebuild's answer to the reference's compile-time safety: the C++
# coordinator/SDK/REST stack gets a memory-error pass in CI.)
set -euo pipefail
cd "$(dirname "$0")/.."

LIBASAN=$(gcc -print-file-name=libasan.so)
export CFLAGS="-fsanitize=address,undefined -fno-sanitize-recover=undefined -g -O1"
export LDSHARED="g++ -shared -fsanitize=address,undefined"

echo "== building _core with ASAN+UBSAN =="
rm -rf build/asan && mkdir -p build/asan
python setup.py build_ext --inplace --build-temp build/asan

echo "== running protocol suite under ASAN =="
ASAN_OPTIONS=detect_leaks=0:abort_on_error=1 \
UBSAN_OPTIONS=print_stacktrace=1 \
LD_PRELOAD="$LIBASAN" \
python -m pytest tests -q -m "not gpu" -x -p no:cacheprovider "$@"

echo "== restoring regular build =="
unset CFLAGS LDSHARED
python setup.py build_ext --inplace >/dev/null
echo "sanitize: OK"
