#!/bin/bash
# Native-code sanitizer tier: build fabricd/fabricctl with ASan+UBSan and run
# the fabric mesh test suites against the instrumented binaries.
# (The race-detection analog of the reference's `go test -race`; the Python
# side is covered by tests/test_concurrency.py.)
set -eu
cd "$(dirname "$0")/.."
OUT=${1:-/tmp/amddra-asan}
mkdir -p "$OUT"
CXXFLAGS="-O1 -g -std=c++17 -fsanitize=address,undefined -fno-omit-frame-pointer -pthread"
g++ $CXXFLAGS native/fabricd/fabricd.cpp -o "$OUT/fabricd" -ldl -lssl -lcrypto
g++ $CXXFLAGS native/fabricd/fabricctl.cpp -o "$OUT/fabricctl"
FABRICD_PATH="$OUT/fabricd" FABRICCTL_PATH="$OUT/fabricctl" \
  python -m pytest "tests/test_computedomain.py::TestFabricd" \
                   "tests/test_computedomain.py::TestFabricdRobustness" \
                   "tests/test_computedomain.py::TestEightDaemonMesh" \
                   tests/test_e2e.py -q "$@"
echo "sanitizer run clean"
