#!/bin/bash
# AddressSanitizer pass over the C++ data path (pileup/BAM/align): rebuild
# the _pileup extension instrumented and run the CPU data-path tests under
# libasan. The HIP extension is untouched (device code; use
# ROCM compute-sanitizer on a GPU box for that side).
#
#   scripts/sanitize.sh [pytest args...]
#
# Restores the normal build afterwards.
set -euo pipefail
cd "$(dirname "$0")/.."

SAN=${ROKO_SANITIZE:-address}
echo "== building _pileup with -fsanitize=$SAN =="
ROKO_SANITIZE=$SAN SKIP_HIP=1 python3 setup.py build_ext --inplace --force

ASAN_LIB=$(g++ -print-file-name=libasan.so)
STDCPP=$(g++ -print-file-name=libstdc++.so)
echo "== running data-path tests under $ASAN_LIB =="
# leak detection off: CPython itself 'leaks' interned objects at exit.
# libstdc++ must be preloaded too: CPython does not link it, so ASan's
# __cxa_throw interceptor finds no real symbol at init and CHECK-fails on
# the first C++ exception thrown from the (dlopen'ed) extension.
LD_PRELOAD="$ASAN_LIB $STDCPP" ASAN_OPTIONS=detect_leaks=0:abort_on_error=1 \
  python3 -m pytest tests/test_pileup.py tests/test_bamio.py \
  tests/test_labels.py tests/test_accuracy.py -q -m "not gpu" "${@}"

echo "== restoring normal build =="
SKIP_HIP=1 python3 setup.py build_ext --inplace --force >/dev/null
echo "sanitize pass OK"
