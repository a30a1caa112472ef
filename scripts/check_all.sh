#!/usr/bin/env bash
# One-command verification (no GPU needed): build all extensions, run the
# CPU protocol suite, then the ASAN/UBSAN and TSAN passes. GPU tiers
# (`pytest -m gpu`, bench.py) run on an MI355X box.
set -euo pipefail
cd "$(dirname "$0")/.."

echo "== build =="
python setup.py build_ext --inplace
python build_hip.py
python build_ffi.py

echo "== CPU suite =="
python -m pytest tests -q -m "not gpu"

echo "== sanitizers =="
bash scripts/sanitize.sh
bash scripts/tsan.sh

echo "check_all: OK"
