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

# the native data-plane server (csrc/fastpath.cpp), GPU admission
# stubbed: concurrent clients + hot route swaps + retries + rate limits
# under ASan/UBSan
out2="${TMPDIR:-/tmp}/aigw_fastpath_sanitize_test"
g++ -std=c++17 -O1 -g -fno-omit-frame-pointer \
    -fsanitize=address,undefined -fno-sanitize-recover=all \
    -D_GLIBCXX_ASSERTIONS \
    csrc/fastpath_sanitize_test.cpp csrc/fastpath.cpp -o "$out2" -lpthread
"$out2"
