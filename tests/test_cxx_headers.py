"""CPU-side compile check of the C++ surface headers (hipcc -fsyntax-only
builds the DIA test TU without a GPU), so header breaks are caught by the
CPU suite."""
import os
import subprocess

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_dia_surface_compiles():
    r = subprocess.run(
        ["hipcc", "-O0", "-std=c++17", "-fsyntax-only",
         "-I" + os.path.join(REPO, "include"),
         os.path.join(REPO, "tests", "cxx", "dia_test.cpp")],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
