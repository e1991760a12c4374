"""swap-or-not shuffle: pin the C oracle against an independent hashlib
restatement of shuffle_list.rs, plus the reference's own properties
(shuffle/unshuffle round trip, None conditions)."""
import ctypes
import hashlib

import numpy as np
import pytest


def ref_shuffle(inp, rounds, seed, forwards):
    """independent hashlib restatement (shuffle_list.rs:80-160)"""
    n = len(inp)
    if n == 0 or n > 2**24 or rounds == 0:
        return None
    inp = list(inp)
    r = 0 if forwards else rounds - 1
    while True:
        buf = bytearray(seed) + bytes([r])
        pivot = int.from_bytes(hashlib.sha256(bytes(buf)).digest()[:8], "little") % n
        mirror = (pivot + 1) >> 1
        source = hashlib.sha256(bytes(buf) + (pivot >> 8).to_bytes(4, "little")).digest()
        byte_v = source[(pivot & 0xFF) >> 3]
        for i in range(mirror):
            j = pivot - i
            if j & 0xFF == 0xFF:
                source = hashlib.sha256(bytes(buf) + (j >> 8).to_bytes(4, "little")).digest()
            if j & 0x07 == 0x07:
                byte_v = source[(j & 0xFF) >> 3]
            if (byte_v >> (j & 0x07)) & 1:
                inp[i], inp[j] = inp[j], inp[i]
        mirror = (pivot + n + 1) >> 1
        end = n - 1
        source = hashlib.sha256(bytes(buf) + (end >> 8).to_bytes(4, "little")).digest()
        byte_v = source[(end & 0xFF) >> 3]
        for li, i in enumerate(range(pivot + 1, mirror)):
            j = end - li
            if j & 0xFF == 0xFF:
                source = hashlib.sha256(bytes(buf) + (j >> 8).to_bytes(4, "little")).digest()
            if j & 0x07 == 0x07:
                byte_v = source[(j & 0xFF) >> 3]
            if (byte_v >> (j & 0x07)) & 1:
                inp[i], inp[j] = inp[j], inp[i]
        if forwards:
            r += 1
            if r == rounds:
                break
        else:
            if r == 0:
                break
            r -= 1
    return inp


def oracle_shuffle(oracle, indices, rounds, seed, forwards):
    arr = (ctypes.c_uint32 * len(indices))(*indices)
    rc = oracle.m3x_oracle_shuffle_list(
        arr, ctypes.c_uint64(len(indices)), ctypes.c_uint8(rounds), seed,
        1 if forwards else 0
    )
    return None if rc != 0 else list(arr)


def test_oracle_vs_ref(oracle):
    seed = hashlib.sha256(b"shuffle-seed").digest()
    for n in [1, 2, 10, 255, 256, 257, 1000]:
        for fwd in [True, False]:
            got = oracle_shuffle(oracle, list(range(n)), 90, seed, fwd)
            want = ref_shuffle(list(range(n)), 90, seed, fwd)
            assert got == want, (n, fwd)


def test_oracle_roundtrip(oracle):
    seed = hashlib.sha256(b"rt").digest()
    n = 2048
    fwd = oracle_shuffle(oracle, list(range(n)), 90, seed, True)
    back = oracle_shuffle(oracle, fwd, 90, seed, False)
    assert back == list(range(n))
    assert fwd != list(range(n))


def test_none_conditions(oracle):
    seed = b"\x2a" * 32
    assert oracle_shuffle(oracle, [], 90, seed, True) is None
    assert oracle_shuffle(oracle, [1, 2], 0, seed, True) is None


@pytest.mark.gpu
def test_gpu_vs_oracle(oracle):
    from lighthouse_amd import shuffle as sh

    seed = hashlib.sha256(b"gpu-shuffle").digest()
    for n in [1, 257, 4096, 1 << 20]:
        want = oracle_shuffle(oracle, list(range(n)), 90, seed, False)
        got = sh.shuffle_list(range(n), 90, seed, False)
        assert got.tolist() == want, n
    # round trip on GPU
    n = 1 << 16
    fwd = sh.shuffle_list(range(n), 90, seed, True)
    back = sh.shuffle_list(fwd, 90, seed, False)
    assert back.tolist() == list(range(n))
