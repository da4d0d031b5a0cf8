"""Adversarial input patterns for the MSB sort's wave16 sub-bucket path
(csrc/t9_bitonic.h): stability is carried by the (low48 << 12 | pos)
composite, so these cases stress exactly the composite's tie handling —
heavy duplicate low-48 bits inside sub-buckets, monotone inputs (every
record of a bucket arrives from one direction), and sawtooth patterns.
Oracle: numpy stable argsort (the reference sort is stable under the
(key, gidx) tiebreak)."""
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


def _check_pairs(nat, keys):
    n = len(keys)
    dk = G.dev(keys)
    dv = G.dev(np.arange(n, dtype=np.uint32))
    w = G.ws(nat.ws("sort_pairs", n))
    nat.sort_pairs_u64_u32(G.ptr(dk), G.ptr(dv), n, G.ptr(w), G.stream())
    gk = G.host(dk, np.uint64)[:n]
    gv = G.host(dv, np.uint32)[:n]
    order = np.argsort(keys, kind="stable").astype(np.uint32)
    assert np.array_equal(gk, keys[order]), "keys not sorted"
    assert np.array_equal(gv, order), "stability violated"


def test_msb_stability_duplicate_low48(nat):
    """random top-17 bits (spreads over all 9-bit subs) but only 4
    distinct low-47 values: every sub-bucket is a mass of composite
    ties resolved purely by load position."""
    n = 1 << 22
    rng = np.random.default_rng(11)
    top = rng.integers(0, 1 << 17, n, dtype=np.uint64) << np.uint64(47)
    low = rng.integers(0, 4, n, dtype=np.uint64) * np.uint64(0x1234567)
    _check_pairs(nat, top | low)


def test_msb_descending_input(nat):
    """strictly descending keys: every bucket/sub fills in reverse
    order, the worst case for any rank/scatter order assumption."""
    n = (1 << 22) + 12345
    keys = np.arange(n, 0, -1, dtype=np.uint64) * np.uint64(0x100000000)
    _check_pairs(nat, keys)


def test_msb_sawtooth(nat):
    """sawtooth over a small period: adjacent tiles carry identical key
    sequences, so every wave sees the same digits (maximum rank-phase
    collision pattern), with ties across the whole array."""
    n = 1 << 22
    period = 4096
    base = np.arange(n, dtype=np.uint64) % np.uint64(period)
    keys = (base << np.uint64(52)) | (base * np.uint64(0x9E3779B97F4A7C15))
    _check_pairs(nat, keys)
