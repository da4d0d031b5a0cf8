"""RCCL communicator bootstrap through the C ABI (t9_comm_id /
t9_comm_init) and the t9_alltoall loopback shortcut — the world-1 slice
of the product exchange that a single-GPU box can execute (real
multi-rank sends run in the driver's N>1 bench; their control flow is
covered by the gloo CPU tests)."""
import ctypes

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from tests import _gpu as G
    from thrill_amd import Native


@pytest.fixture(scope="module")
def nat():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    n = Native(device=0)
    yield n
    n.close()


def test_comm_bootstrap_world1():
    # ncclGetUniqueId + ncclCommInitRank(world=1) must succeed on
    # hardware; t9_destroy frees the owned communicator.
    nat = Native(device=0, rank=0, world=1)
    try:
        idb = nat.comm_id()
        assert len(idb) == nat._lib.t9_comm_id_size() > 0
        nat.comm_init(idb)
        # double init is an error (already connected)
        with pytest.raises(Exception):
            nat.comm_init(idb)
    finally:
        nat.close()


def test_rccl_selfsend_4gib_chunked():
    """regression for the silent RCCL p2p truncation at >= 2^32 bytes
    (profiles/r02_selfnccl_truncation.json): with t9_alltoall's 1 GiB
    chunking, a forced 4 GiB RCCL self-send must deliver every byte.
    Separate process: T9_A2A_SELF latches process-globally."""
    import json
    import os
    import subprocess
    import sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, os.path.join(repo, "scripts",
                                      "probe_self_nccl.py"),
         str(1 << 32)],
        capture_output=True, text=True, timeout=240)
    assert r.returncode == 0, r.stderr[-800:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["ok"] is True and d["bytes"] == 1 << 32


def test_alltoall_world1_shortcut(nat):
    # world==1: t9_alltoall is a device memcpy honoring displacements
    n = 100_000
    rng = np.random.default_rng(3)
    data = rng.integers(0, 2**63, n).astype(np.uint64)
    ds = G.dev(data)
    dr = G.empty(n, np.uint64)
    sc = np.array([n], dtype=np.uint64)
    sd = np.array([0], dtype=np.uint64)
    nat.alltoall(G.ptr(ds), ctypes.c_void_p(sc.ctypes.data),
                 ctypes.c_void_p(sd.ctypes.data), G.ptr(dr),
                 ctypes.c_void_p(sc.ctypes.data),
                 ctypes.c_void_p(sd.ctypes.data), 8, G.stream())
    assert np.array_equal(G.host(dr, np.uint64), data)
