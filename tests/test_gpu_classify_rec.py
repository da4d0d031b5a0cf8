"""GPU parity for full-record classification (acceptance total order):
t9_classify_rec vs the oracle's t9o_classify_rec with key_len = rec_size."""
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


def k64_of(recs):
    return np.array([int.from_bytes(r[:8].tobytes(), "big") for r in recs],
                    dtype=np.uint64)


@pytest.mark.parametrize("collisions", [False, True])
def test_classify_rec_parity(nat, oracle, collisions):
    n, p = 50_000, 8
    recs = oracle.gen_records(n, seed=21)
    if collisions:
        # force heavy u64-prefix collisions so the byte-fallback and the
        # gidx tiebreak paths are exercised
        recs[:, :8] = (np.arange(n, dtype=np.uint64) % 5
                       ).astype(">u8").view(np.uint8).reshape(n, 8)
    rng = np.random.default_rng(4)
    pos = np.sort(rng.choice(n, 64, replace=False)).astype(np.uint64)
    spl_all = recs[pos.astype(np.int64)]
    # select splitters with the reference rule over (record, idx) pairs
    order = sorted(range(len(pos)),
                   key=lambda t: (spl_all[t].tobytes(), int(pos[t])))
    step = len(pos) / p
    sel = [order[int(i * step)] for i in range(1, p)]
    spl_recs = np.stack([spl_all[t] for t in sel])
    spl_idx = np.array([pos[t] for t in sel], dtype=np.uint64)

    gidx0 = 777
    expect = oracle.classify_rec(recs, gidx0, 100, spl_recs, spl_idx, p)

    din = G.dev(recs.reshape(-1))
    dk = G.dev(k64_of(recs))
    dsr = G.dev(spl_recs.reshape(-1))
    dsk = G.dev(k64_of(spl_recs))
    dsi = G.dev(spl_idx)
    db = G.empty(n, np.uint32)
    dc = G.empty(p, np.uint64)
    nat.classify_rec(G.ptr(din), G.ptr(dk), n, gidx0, G.ptr(dsr),
                     G.ptr(dsk), G.ptr(dsi), p, 100, G.ptr(db), G.ptr(dc),
                     G.stream())
    got = G.host(db, np.uint32)
    assert np.array_equal(got, expect)
    assert np.array_equal(G.host(dc, np.uint64),
                          np.bincount(expect, minlength=p).astype(np.uint64))
