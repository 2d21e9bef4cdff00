"""Sanitizer pass: the CPU oracle + packed-stream codec built
with ASan+UBSan and exercised end to end.  The harness (oracle/asan_check.c)
aborts on any out-of-bounds access, misaligned access, signed overflow or
leak — it already caught a real misaligned header store in vm_pack_blocks
when first introduced."""
import os
import shutil
import subprocess

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.skipif(shutil.which("gcc") is None, reason="no gcc")
def test_oracle_asan_ubsan_harness():
    subprocess.run(["make", "-C", os.path.join(REPO, "oracle"), "asan_check"],
                   check=True, capture_output=True)
    r = subprocess.run([os.path.join(REPO, "oracle", "asan_check")],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, f"sanitizer harness failed:\n{r.stdout}\n{r.stderr}"
    assert "asan_check OK" in r.stdout
