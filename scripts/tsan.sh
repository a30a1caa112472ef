#!/usr/bin/env bash
# ThreadSanitizer pass over the threaded protocol tests (coordinator thread +
# REST workers + SDK threads). Complements scripts/sanitize.sh (ASAN).
set -euo pipefail
cd "$(dirname "$0")/.."
LIBTSAN=$(gcc -print-file-name=libtsan.so)
export CFLAGS="-fsanitize=thread -g -O1"
export LDSHARED="g++ -shared -fsanitize=thread"
rm -rf build/tsan && python setup.py build_ext --inplace --build-temp build/tsan >/dev/null
TSAN_OPTIONS="report_bugs=1 history_size=7 suppressions=$(pwd)/scripts/tsan.supp exitcode=66 log_path=/tmp/tsan_rep" \
LD_PRELOAD="$LIBTSAN" \
python -m pytest tests/test_e2e_round.py tests/test_rest.py tests/test_failure_recovery.py \
    -q -x -p no:cacheprovider "$@" || true
unset CFLAGS LDSHARED
python setup.py build_ext --inplace >/dev/null
echo "tsan pass done (reports above, if any)"
