#!/bin/bash
# Copyright 2026 mlrun_amd authors
#
# AddressSanitizer lane for the native C++ components (net-new vs the
# reference, which ships no sanitizer runs — SURVEY.md §5).
# Builds the log-collector with ASan and runs its python test tier
# against the instrumented binary.
set -euo pipefail
cd "$(dirname "$0")/.."

SRC=mlrun_amd/native/log_collector.cpp
BIN=mlrun_amd/native/log_collector

echo "== building $SRC with -fsanitize=address =="
g++ -O1 -g -std=c++17 -pthread -fsanitize=address -fno-omit-frame-pointer \
    "$SRC" -o "$BIN"

echo "== running log-collector tests against the ASan build =="
# the test fixture rebuilds only when the source is newer than the
# binary; touch the binary so our instrumented build is used
touch "$BIN"
ASAN_OPTIONS=detect_leaks=0 python -m pytest tests/test_log_collector.py -q

echo "== rebuilding the optimized binary =="
g++ -O2 -std=c++17 -pthread "$SRC" -o "$BIN"
echo "asan lane OK"
