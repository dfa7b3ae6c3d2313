#!/bin/bash
# TSAN build target for the torch-free C++ transport core.
set -e
cd "$(dirname "$0")/.."
mkdir -p build/tsan
g++ -std=c++17 -O1 -g -fsanitize=thread -pthread \
    tools/tsan_transport_test.cpp \
    torchmpi_amd/csrc/tcp.cpp torchmpi_amd/csrc/cpu_comm.cpp \
    torchmpi_amd/csrc/ps.cpp \
    -o build/tsan/transport_test
TSAN_OPTIONS="halt_on_error=0 exitcode=66" ./build/tsan/transport_test
echo "tsan run complete"
