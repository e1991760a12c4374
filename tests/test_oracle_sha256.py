"""Pin the oracle's SHA256 against FIPS 180-4 KATs and hashlib (OpenSSL)."""
import ctypes
import hashlib
import os
import random


def oracle_sha(oracle, data: bytes) -> bytes:
    out = ctypes.create_string_buffer(32)
    oracle.m3x_oracle_sha256(data, ctypes.c_size_t(len(data)), out)
    return out.raw


def test_fips_kats(oracle):
    # FIPS 180-4 / NIST example vectors
    kats = {
        b"abc": "ba7816bf8f01cfea414140de5dae2223b00361a396177a9cb410ff61f20015ad",
        b"": "e3b0c44298fc1c149afbf4c8996fb92427ae41e4649b934ca495991b7852b855",
        b"abcdbcdecdefdefgefghfghighijhijkijkljklmklmnlmnomnopnopq": (
            "248d6a61d20638b8e5c026930c3e6039a33ce45964ff2167f6ecedd419db06c1"
        ),
    }
    for msg, want in kats.items():
        assert oracle_sha(oracle, msg).hex() == want


def test_million_a(oracle):
    assert (
        oracle_sha(oracle, b"a" * 1000000).hex()
        == "cdc76e5c9914fb9281a1c7e284d73e67f1809a48a497200e046d39ccc7112cd0"
    )


def test_vs_hashlib_random_lengths(oracle):
    rng = random.Random(0xC0FFEE)
    for ln in [1, 31, 32, 33, 55, 56, 57, 63, 64, 65, 119, 120, 121, 127, 128, 129, 1000, 4096]:
        data = bytes(rng.getrandbits(8) for _ in range(ln))
        assert oracle_sha(oracle, data) == hashlib.sha256(data).digest()


def test_hash64(oracle):
    left, right = os.urandom(32), os.urandom(32)
    out = ctypes.create_string_buffer(32)
    oracle.m3x_oracle_hash64(left, right, out)
    assert out.raw == hashlib.sha256(left + right).digest()
