#!/usr/bin/env bash
# ASan+UBSan over the standalone C ABI library + compiled consumer
# (the reference has no sanitizer coverage; the C ABI is
# the surface foreign engines dlopen, so leaks/UB here corrupt *their*
# processes). The torch-linked extensions can't preload ASan cleanly, so
# this covers the pure-C++ library: reader/writer/MOR/merge-ops/filters
# /substrait decode + the metadata DAO layer.
set -euo pipefail
cd "$(dirname "$0")/.."
OUT=${TMPDIR:-/tmp}/lakesoul_asan
mkdir -p "$OUT"
SAN="-fsanitize=address,undefined -fno-sanitize-recover=undefined -g -O1"
g++ $SAN -std=c++17 -shared -fPIC \
    csrc/capi/lakesoul_c.cc csrc/capi/lakesoul_meta_c.cc \
    -o "$OUT/liblakesoul_amd_c_asan.so" \
    -l:libzstd.so.1 -l:libsqlite3.so.0 -l:libcrypto.so.3 -pthread
gcc $SAN -O1 -g csrc/capi/tests/capi_smoke.c -o "$OUT/capi_smoke_asan" \
    -ldl -lpthread
ASAN_OPTIONS=detect_leaks=1 UBSAN_OPTIONS=print_stacktrace=1 \
    "$OUT/capi_smoke_asan" "$OUT/liblakesoul_amd_c_asan.so" "$OUT"
echo "sanitize_capi: PASS"
