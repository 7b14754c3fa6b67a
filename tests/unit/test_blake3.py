"""BLAKE3 correctness: official vectors + C++ vs independent Python golden."""

import pytest

from edgehip import get_lib
from edgehip.golden.blake3 import blake3 as py_blake3
from tests.conftest import rand_bytes

# Official BLAKE3 test vectors (input byte i % 251).
VECTORS = [
    (0, "af1349b9f5f9a1a6a0404dea36dcc9499bcb25c9adc112b7cc9a93cae41f3262"),
    (1, "2d3adedff11b61f14c886e35afa036736dcd87a74d27b5c1510225d0f592e213"),
    (63, "e9bc37a594daad83be9470df7f7b3798297c3d834ce80ba85d6e207627b7db7b"),
    (64, "4eed7141ea4a5cd4b788606bd23f46e212af9cacebacdc7d1f4c6dc7f2511b98"),
    (65, "de1e5fa0be70df6d2be8fffd0e99ceaa8eb6e8c93a63f2d8d1c30ecb6b263dee"),
    (127, "d81293fda863f008c09e92fc382a81f5a0b4a1251cba1634016a0f86a6bd640d"),
    (128, "f17e570564b26578c33bb7f44643f539624b05df1a76c81f30acd548c44b45ef"),
    (1023, "10108970eeda3eb932baac1428c7a2163b0e924c9a9e25b35bba72b28f70bd11"),
    (1024, "42214739f095a406f3fc83deb889744ac00df831c10daa55189b5d121c855af7"),
    (1025, "d00278ae47eb27b34faecf67b4fe263f82d5412916c1ffd97c8cb7fb814b8444"),
    (2048, "e776b6028c7cd22a4d0ba182a8bf62205d2ef576467e838ed6f2529b85fba24a"),
    (2049, "5f4d72f40d7a5f82b15ca2b2e44b1de3c2ef86c426c95c1af0b6879522563030"),
    (3072, "b98cb0ff3623be03326b373de6b9095218513e64f1ee2edd2525c7ad1e5cffd2"),
    (3073, "7124b49501012f81cc7f11ca069ec9226cecb8a2c850cfe644e327d22d3e1cd3"),
    (4096, "015094013f57a5277b59d8475c0501042c0b642e531b0a1c8f58d2163229e969"),
    (5120, "9cadc15fed8b5d854562b26a9536d9707cadeda9b143978f319ab34230535833"),
]


def seq_input(n):
    return bytes(i % 251 for i in range(n))


@pytest.mark.parametrize("n,want", VECTORS)
def test_official_vectors_cpp(n, want):
    assert get_lib().blake3(seq_input(n)).hex() == want


@pytest.mark.parametrize("n,want", VECTORS[:10])
def test_official_vectors_golden_py(n, want):
    assert py_blake3(seq_input(n)).hex() == want


@pytest.mark.parametrize("n", [0, 1, 31, 64, 1000, 1024, 4095, 65536, 1 << 20, (1 << 20) + 7])
def test_cpp_matches_golden(n):
    data = rand_bytes(n, seed=n)
    assert get_lib().blake3(data) == py_blake3(data)
