"""Sanitizer CI job: the native collator's index-building
core compiled under AddressSanitizer + UBSan and run over adversarial
shapes.  The harness compiles csrc/collate_core.h — the EXACT code the
extension executes (csrc/collate.cpp includes it) — torch-free, so the
ASan runtime links cleanly."""
import os
import shutil
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(300)
def test_collate_core_under_asan(tmp_path):
    gxx = shutil.which("g++")
    if gxx is None:
        pytest.skip("g++ not available")
    exe = str(tmp_path / "sanitize_collate")
    build = subprocess.run(
        [gxx, "-std=c++17", "-g", "-O1",
         "-fsanitize=address,undefined", "-fno-sanitize-recover=all",
         "-I", os.path.join(ROOT, "csrc"),
         os.path.join(ROOT, "csrc", "sanitize_main.cpp"), "-o", exe],
        capture_output=True, text=True, timeout=180,
    )
    assert build.returncode == 0, build.stderr[-2000:]
    run = subprocess.run([exe], capture_output=True, text=True, timeout=120)
    assert run.returncode == 0, (run.stdout[-1000:], run.stderr[-2000:])
    assert "all cases clean" in run.stdout
