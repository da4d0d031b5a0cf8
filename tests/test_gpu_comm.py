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
