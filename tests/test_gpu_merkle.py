"""GPU parity tests for the SSZ merkleize hot path: HIP kernels vs the CPU
oracle / independent hashlib restatement, on seeded inputs, including the
edge cases the reference's tree semantics exercise (empty, single, ragged,
full-capacity, deep sparse)."""
import ctypes
import random

import pytest

import ssz_ref

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ctx():
    from lighthouse_amd import _native

    return _native.default_ctx()


def test_merkleize_chunks_parity(ctx):
    from lighthouse_amd import tree_hash as th

    rng = random.Random(7)
    for depth, n in [
        (0, 0),
        (0, 1),
        (1, 1),
        (2, 3),
        (3, 8),
        (10, 1),
        (10, 700),
        (13, 8192),
        (20, 4097),
        (40, 5),
    ]:
        chunks = [bytes(rng.getrandbits(8) for _ in range(32)) for _ in range(n)]
        got = th.merkleize_chunks(b"".join(chunks), n, depth, -1, ctx=ctx)
        want = ssz_ref.merkleize(chunks, depth)
        assert got == want, (depth, n)


def test_merkleize_with_mix(ctx):
    from lighthouse_amd import tree_hash as th

    rng = random.Random(8)
    chunks = [bytes(rng.getrandbits(8) for _ in range(32)) for _ in range(100)]
    got = th.merkleize_chunks(b"".join(chunks), 100, 24, 100, ctx=ctx)
    want = ssz_ref.mix_in_length(ssz_ref.merkleize(chunks, 24), 100)
    assert got == want


def test_validator_registry_parity_small(ctx):
    from lighthouse_amd import tree_hash as th

    for n in [1, 2, 255, 256, 257, 1000]:
        ssz = b"".join(ssz_ref.synthetic_validator_ssz(i) for i in range(n))
        got = th.validator_registry_root(ssz, n, ctx=ctx)
        want = ssz_ref.validator_registry_root(ssz, n)
        assert got == want, n


def test_validator_registry_parity_64k_vs_oracle(ctx, oracle):
    from lighthouse_amd import state, tree_hash as th

    n = 65536
    ssz = state.validators_ssz(n)
    got = th.validator_registry_root(ssz, n, ctx=ctx)
    want = ctypes.create_string_buffer(32)
    oracle.m3x_oracle_validator_registry_root(ssz, ctypes.c_uint64(n), want)
    assert got == want.raw


def test_basic_list_roots(ctx):
    from lighthouse_amd import tree_hash as th

    rng = random.Random(9)
    data = bytes(rng.getrandbits(8) for _ in range(8 * 5000))
    got = th.basic_list_root(data, 5000, 8, 1 << 40, ctx=ctx)
    assert got == ssz_ref.basic_list_root(data, 5000, 8, 1 << 40)
    data = bytes(rng.getrandbits(8) for _ in range(3001))
    got = th.basic_list_root(data, 3001, 1, 1 << 40, ctx=ctx)
    assert got == ssz_ref.basic_list_root(data, 3001, 1, 1 << 40)
