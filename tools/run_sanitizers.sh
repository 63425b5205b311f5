#!/bin/bash
# Build and run the standalone wire-codec test under ASAN+UBSAN
# (the reference repo has no sanitizer CI; the new build runs
# the C++ codec under sanitizers because it parses untrusted bytes).
set -euo pipefail
cd "$(dirname "$0")/.."
mkdir -p build/sanitize
g++ -std=c++17 -O1 -g -fsanitize=address,undefined -fno-omit-frame-pointer \
    tests/cpp/wire_test.cpp -o build/sanitize/wire_test_asan
./build/sanitize/wire_test_asan
echo "ASAN+UBSAN wire test passed"
g++ -std=c++17 -O1 -g -fsanitize=address,undefined -fno-omit-frame-pointer \
    tests/cpp/h2_test.cpp -o build/sanitize/h2_test_asan -lpthread
./build/sanitize/h2_test_asan
echo "ASAN+UBSAN h2/hpack test passed"
