#!/bin/bash
# Address+UB sanitizer pass over the CPU core (no HIP; plain g++).
set -e
cd "$(dirname "$0")/.."
g++ -std=c++17 -O1 -g -fsanitize=address,undefined -fno-omit-frame-pointer \
    csrc/core/mesh.cpp csrc/core/mesh_io.cpp csrc/core/osh_io.cpp \
    csrc/core/osh_omegah.cpp \
    csrc/core/engine_cpu.cpp csrc/core/partition.cpp \
    csrc/comm/comm_tcp.cpp tools/asan_check.cpp \
    -o /tmp/pt_asan_check -pthread -lz
/tmp/pt_asan_check
