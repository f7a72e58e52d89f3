#!/bin/bash
# Drive the oracle's agg/sort/window paths under AddressSanitizer.
# (CPU-only; used during round-1 hardening — 25 randomized cases clean.)
set -e
cd "$(dirname "$0")/.."
gcc -fsanitize=address -O1 -fPIC -std=c11 -Wall -pthread -shared \
    -o /tmp/liboracle_asan.so oracle/oracle.c -pthread
LD_PRELOAD=$(gcc -print-file-name=libasan.so) ASAN_OPTIONS=detect_leaks=0 \
    python tools/asan_drive.py
