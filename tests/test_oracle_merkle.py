"""Pin the oracle's SSZ merkleize against the independent hashlib-based
restatement in ssz_ref.py (semantics: merkle_proof/src/lib.rs:68-100,
deposit_data_tree.rs:26-38, validator.rs:25-35, eth_spec.rs:404)."""
import ctypes
import random

import ssz_ref


def o_merkleize(oracle, chunks: bytes, n: int, depth: int) -> bytes:
    out = ctypes.create_string_buffer(32)
    oracle.m3x_oracle_merkleize(chunks, ctypes.c_uint64(n), ctypes.c_uint32(depth), out)
    return out.raw


def test_zero_hashes(oracle):
    for d in [0, 1, 5, 20, 40]:
        out = ctypes.create_string_buffer(32)
        oracle.m3x_oracle_zero_hash(ctypes.c_uint32(d), out)
        assert out.raw == ssz_ref.ZEROS[d]


def test_merkleize_edge_cases(oracle):
    rng = random.Random(1)
    for depth in [0, 1, 2, 3, 4, 10]:
        cap = 1 << depth
        for n in sorted({0, 1, 2, 3, cap // 2, cap - 1, cap} & set(range(cap + 1))):
            chunks = [bytes(rng.getrandbits(8) for _ in range(32)) for _ in range(n)]
            want = ssz_ref.merkleize(chunks, depth)
            got = o_merkleize(oracle, b"".join(chunks), n, depth)
            assert got == want, (depth, n)


def test_merkleize_sparse_deep(oracle):
    # few leaves in a depth-40 tree (the registry shape)
    rng = random.Random(2)
    for n in [1, 2, 5, 100]:
        chunks = [bytes(rng.getrandbits(8) for _ in range(32)) for _ in range(n)]
        assert o_merkleize(oracle, b"".join(chunks), n, 40) == ssz_ref.merkleize(chunks, 40)


def test_mix_in_length(oracle):
    out = ctypes.create_string_buffer(32)
    root = bytes(range(32))
    oracle.m3x_oracle_mix_in_length(root, ctypes.c_uint64(123456789), out)
    assert out.raw == ssz_ref.mix_in_length(root, 123456789)


def test_validator_leaf(oracle):
    for i in [0, 1, 97, 12345]:
        ssz = ssz_ref.synthetic_validator_ssz(i)
        out = ctypes.create_string_buffer(32)
        oracle.m3x_oracle_validator_leaf(ssz, out)
        assert out.raw == ssz_ref.validator_leaf(ssz)


def test_validator_registry_root(oracle):
    for n in [0, 1, 2, 1000]:
        ssz = b"".join(ssz_ref.synthetic_validator_ssz(i) for i in range(n))
        out = ctypes.create_string_buffer(32)
        oracle.m3x_oracle_validator_registry_root(ssz, ctypes.c_uint64(n), out)
        assert out.raw == ssz_ref.validator_registry_root(ssz, n), n


def test_basic_list_and_vector_roots(oracle):
    rng = random.Random(3)
    out = ctypes.create_string_buffer(32)
    # balances-shaped: List[u64, 2^40], 1000 elems
    data = bytes(rng.getrandbits(8) for _ in range(8 * 1000))
    oracle.m3x_oracle_basic_list_root(
        data, ctypes.c_uint64(1000), ctypes.c_uint32(8), ctypes.c_uint64(1 << 40), out
    )
    assert out.raw == ssz_ref.basic_list_root(data, 1000, 8, 1 << 40)
    # participation-shaped: List[u8, 2^40]
    data = bytes(rng.getrandbits(8) for _ in range(1000))
    oracle.m3x_oracle_basic_list_root(
        data, ctypes.c_uint64(1000), ctypes.c_uint32(1), ctypes.c_uint64(1 << 40), out
    )
    assert out.raw == ssz_ref.basic_list_root(data, 1000, 1, 1 << 40)
    # slashings-shaped: Vector[u64, 8192]
    data = bytes(rng.getrandbits(8) for _ in range(8 * 8192))
    oracle.m3x_oracle_basic_vector_root(
        data, ctypes.c_uint64(8192), ctypes.c_uint32(8), out
    )
    assert out.raw == ssz_ref.basic_vector_root(data, 8192, 8)
    # block_roots-shaped: Vector[Hash256, 8192]
    roots = bytes(rng.getrandbits(8) for _ in range(32 * 8192))
    oracle.m3x_oracle_root_vector_root(roots, ctypes.c_uint64(8192), out)
    assert out.raw == ssz_ref.merkleize(ssz_ref.pack_bytes(roots), 13)
    # historical-roots-shaped list
    roots = bytes(rng.getrandbits(8) for _ in range(32 * 7))
    oracle.m3x_oracle_root_list_root(
        roots, ctypes.c_uint64(7), ctypes.c_uint64(1 << 24), out
    )
    assert out.raw == ssz_ref.mix_in_length(
        ssz_ref.merkleize(ssz_ref.pack_bytes(roots), 24), 7
    )
