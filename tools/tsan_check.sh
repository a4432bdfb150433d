#!/bin/bash
# Race-detection pass over the native control plane (reference `make
# unit-test-race` parity; here with ThreadSanitizer on the C++ core).
set -e
cd "$(dirname "$0")/.."
g++ -O1 -g -fsanitize=thread -std=c++17 -pthread tools/tsan_stress.cc -o /tmp/kvc_tsan_stress
TSAN_OPTIONS="halt_on_error=1" /tmp/kvc_tsan_stress
echo "TSAN: no races detected"
