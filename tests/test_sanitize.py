"""ASan/UBSan-clean planner (the reference relied on benign
races; the new build's logic layers must be sanitizer-clean)."""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_planner_sanitizer_clean(tmp_path):
    exe = str(tmp_path / "plan_sanitize")
    build = subprocess.run(
        ["g++", "-std=c++17", "-g", "-O1", "-fsanitize=address,undefined",
         "-fno-sanitize-recover=all",
         os.path.join(REPO, "tests", "cpp", "plan_sanitize.cpp"), "-o", exe],
        capture_output=True, text=True)
    assert build.returncode == 0, build.stderr
    run = subprocess.run([exe], capture_output=True, text=True, timeout=300)
    assert run.returncode == 0, run.stdout + run.stderr
    assert "plan_sanitize ok" in run.stdout
