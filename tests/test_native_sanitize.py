"""ASan/UBSan lane for the native hot-path cores — the reference runs
`go test -race` over its Go hot path; the analogous
memory-safety check here compiles csrc/native_core.h's scanner and SSE
splitter into a standalone binary with -fsanitize=address,undefined and
runs extraction, rejection, and 45k-iteration fuzz loops."""

import shutil
import subprocess

import pytest


@pytest.mark.timeout(240)
def test_native_core_under_sanitizers():
    if shutil.which("g++") is None:
        pytest.skip("g++ not available")
    proc = subprocess.run(
        ["bash", "scripts/sanitize_native.sh"],
        capture_output=True,
        timeout=220,
        cwd=__file__.rsplit("/tests/", 1)[0],
    )
    assert proc.returncode == 0, (proc.stdout + proc.stderr).decode()[-3000:]
    assert b"passed" in proc.stdout
