#!/bin/bash
# ASan+UBSan over the host runtime + emulator protocol (2 forked ranks).
set -e
cd "$(dirname "$0")/.."
g++ -std=c++17 -O1 -g -fsanitize=address,undefined -fno-omit-frame-pointer \
  tests/cxx/asan_multirank.cpp accl_amd/csrc/core/util.cpp \
  accl_amd/csrc/core/accl.cpp accl_amd/csrc/emu/emudevice.cpp \
  -o /tmp/accl_asan -lpthread -w
ASAN_OPTIONS=detect_leaks=1 /tmp/accl_asan
