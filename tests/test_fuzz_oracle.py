"""Property-based (hypothesis) fuzz parity: the C oracle vs the
independent hashlib/python restatement in ssz_ref.py, over randomized
shapes the hand-picked cases don't reach. CPU-only; bounded example
counts keep the suite fast."""
import ctypes

import ssz_ref
from hypothesis import given, settings, strategies as st

SET = settings(max_examples=25, deadline=None)


@SET
@given(st.binary(min_size=0, max_size=700))
def test_sha256_fuzz(oracle, data):
    out = ctypes.create_string_buffer(32)
    oracle.m3x_oracle_sha256(data, ctypes.c_size_t(len(data)), out)
    assert out.raw == ssz_ref.H(data)


@SET
@given(st.integers(0, 5), st.data())
def test_merkleize_fuzz(oracle, depth, data):
    cap = 1 << depth
    n = data.draw(st.integers(0, cap))
    chunks = [data.draw(st.binary(min_size=32, max_size=32)) for _ in range(n)]
    out = ctypes.create_string_buffer(32)
    oracle.m3x_oracle_merkleize(
        b"".join(chunks), ctypes.c_uint64(n), ctypes.c_uint32(depth), out
    )
    assert out.raw == ssz_ref.merkleize(chunks, depth)


@SET
@given(st.integers(0, 64), st.integers(6, 40))
def test_merkleize_sparse_fuzz(oracle, n, depth):
    # right-sparse deep trees (registry shape): n << 2^depth
    chunks = [bytes([i % 251] * 32) for i in range(n)]
    out = ctypes.create_string_buffer(32)
    oracle.m3x_oracle_merkleize(
        b"".join(chunks), ctypes.c_uint64(n), ctypes.c_uint32(depth), out
    )
    assert out.raw == ssz_ref.merkleize(chunks, depth)


@SET
@given(st.binary(min_size=121, max_size=121))
def test_validator_leaf_fuzz(oracle, ssz):
    out = ctypes.create_string_buffer(32)
    oracle.m3x_oracle_validator_leaf(ssz, out)
    assert out.raw == ssz_ref.validator_leaf(ssz)


@SET
@given(
    st.integers(1, 300),
    st.integers(1, 100),
    st.binary(min_size=32, max_size=32),
    st.booleans(),
)
def test_shuffle_fuzz(oracle, size, rounds, seed, forwards):
    from test_shuffle import oracle_shuffle, ref_shuffle

    got = oracle_shuffle(oracle, list(range(size)), rounds, seed, forwards)
    want = ref_shuffle(list(range(size)), rounds, seed, forwards)
    assert got == want


@SET
@given(st.integers(0, 40), st.integers(1, 33), st.data())
def test_basic_list_root_fuzz(oracle, n_elems, elem_size, data):
    limit_elems = n_elems + data.draw(st.integers(0, 1000))
    if limit_elems == 0:
        limit_elems = 1
    payload = data.draw(
        st.binary(min_size=n_elems * elem_size, max_size=n_elems * elem_size)
    )
    out = ctypes.create_string_buffer(32)
    oracle.m3x_oracle_basic_list_root(
        payload,
        ctypes.c_uint64(n_elems),
        ctypes.c_uint32(elem_size),
        ctypes.c_uint64(limit_elems),
        out,
    )
    assert out.raw == ssz_ref.basic_list_root(
        payload, n_elems, elem_size, limit_elems
    )
