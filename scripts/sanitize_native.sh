#!/usr/bin/env bash
# ASan/UBSan build + run of the native hot-path cores (csrc/native_core.h)
# — the reference's `go test -race` CI lane analogue.
# CPU-only; runs in seconds. Invoked by tests/test_native_sanitize.py.
set -euo pipefail
cd "$(dirname "$0")/.."
out="${TMPDIR:-/tmp}/aigw_native_core_test"
g++ -std=c++17 -O1 -g -fno-omit-frame-pointer \
    -fsanitize=address,undefined -fno-sanitize-recover=all \
    -D_GLIBCXX_ASSERTIONS \
    csrc/native_core_test.cpp -o "$out"
"$out"
