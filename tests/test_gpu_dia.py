"""Run the compiled C++ DIA-surface test binary (tests/cxx/dia_test) on
the GPU. The binary is built in-tree by __graft_entry__.build() and mirrors
the reference's sort_node/word_count tests."""
import os
import subprocess

import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BIN = os.path.join(REPO, "tests", "cxx", "dia_test")


def test_dia_cxx_surface():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    if not os.path.exists(BIN):
        import __graft_entry__
        __graft_entry__.build()
    assert os.path.exists(BIN), "dia_test binary missing (build() failed?)"
    r = subprocess.run([BIN], cwd=REPO, capture_output=True, text=True,
                       timeout=600)
    print(r.stdout, r.stderr)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "all checks passed" in r.stdout
