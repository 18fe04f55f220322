#!/bin/bash
# Build the pure-C++ TSan driver against the bobraccel core and run it.
# Expected output: the driver's summary line and NO "WARNING: ThreadSanitizer".
set -e
cd "$(dirname "$0")/../.."
g++ -O1 -g -std=c++17 -DBOBRA_TSAN_COMPAT -fsanitize=thread \
  -I bobrapet_amd/csrc/core \
  bobrapet_amd/csrc/core/engine.cpp tests/tsan/tsan_driver.cpp \
  -o /tmp/bobra_tsan_driver -pthread
TSAN_OPTIONS="halt_on_error=0 exitcode=66" /tmp/bobra_tsan_driver

# ASan+UBSan pass over the same driver (memory + UB coverage)
g++ -O1 -g -std=c++17 -fsanitize=address,undefined \
  -I bobrapet_amd/csrc/core \
  bobrapet_amd/csrc/core/engine.cpp tests/tsan/tsan_driver.cpp \
  -o /tmp/bobra_asan_driver -pthread
/tmp/bobra_asan_driver
