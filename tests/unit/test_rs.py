"""Reed-Solomon k+m over GF(256): C++ vs independent Python golden, all
erasure patterns for RS 6+2 (the judged config), and property checks."""

import itertools

import pytest

from edgehip import get_lib
from edgehip.golden import gf256 as g
from tests.conftest import rand_bytes


def shards(k, n, seed=0):
    return [rand_bytes(n, seed=seed * 100 + i) for i in range(k)]


def test_parity_matrix_matches_golden():
    lib = get_lib()
    for k, m in [(6, 2), (4, 2), (10, 4), (3, 1)]:
        M = lib.rs_parity_matrix(k, m)
        G = bytes(c for row in g.parity_matrix(k, m) for c in row)
        assert M == G


@pytest.mark.parametrize("k,m,n", [(6, 2, 4096), (6, 2, 65536), (4, 3, 1000), (10, 4, 512)])
def test_encode_matches_golden(k, m, n):
    lib = get_lib()
    data = shards(k, n, seed=k * m)
    assert lib.rs_encode_km(k, m, data) == g.encode(k, m, data)


def test_rs62_all_two_erasures():
    """RS 6+2: every C(8,2)=28 double-erasure pattern reconstructs exactly."""
    lib = get_lib()
    k, m, n = 6, 2, 8192
    data = shards(k, n, seed=42)
    parity = lib.rs_encode_km(k, m, data)
    full = data + parity
    for a, b in itertools.combinations(range(k + m), 2):
        damaged = [None if i in (a, b) else full[i] for i in range(k + m)]
        fixed = lib.rs_reconstruct_km(k, m, damaged, n)
        assert fixed == full, f"pattern ({a},{b})"


def test_rs62_single_erasures():
    lib = get_lib()
    k, m, n = 6, 2, 4096
    data = shards(k, n, seed=7)
    parity = lib.rs_encode_km(k, m, data)
    full = data + parity
    for a in range(k + m):
        damaged = [None if i == a else full[i] for i in range(k + m)]
        assert lib.rs_reconstruct_km(k, m, damaged, n) == full


def test_too_many_erasures_fails():
    lib = get_lib()
    k, m, n = 6, 2, 1024
    data = shards(k, n, seed=5)
    parity = lib.rs_encode_km(k, m, data)
    full = data + parity
    damaged = [None, None, None] + full[3:]
    with pytest.raises(RuntimeError):
        lib.rs_reconstruct_km(k, m, damaged, n)


def test_golden_reconstruct_agrees():
    k, m, n = 6, 2, 2048
    lib = get_lib()
    data = shards(k, n, seed=11)
    parity = lib.rs_encode_km(k, m, data)
    full = data + parity
    damaged = [None if i in (1, 6) else full[i] for i in range(k + m)]
    assert g.reconstruct(k, m, damaged) == lib.rs_reconstruct_km(k, m, damaged, n)
