"""Synthetic BeaconState-shaped workloads (BASELINE.json configs C3/C4).

Deterministic, seeded; shapes follow the reference's BeaconState fields
(consensus/types/src/beacon_state.rs, Deneb variant) at a given validator
count. Used by tests and bench.py (data generation only — no compute)."""
import hashlib
import numpy as np

EFFECTIVE_BALANCE = 32 * 10**9


def validators_ssz(n: int, seed: int = 0xC0FFEE) -> bytes:
    """Packed 121-byte SSZ records matching tests/ssz_ref.synthetic_validator_ssz
    (vectorized for large n)."""
    # pk (48B) and wc (32B) are per-index SHA256-derived; epochs patterned.
    out = np.zeros((n, 121), dtype=np.uint8)
    idx = np.arange(n, dtype=np.uint64)
    # derive 48B pk + 32B wc per validator with two sha256 calls per row is
    # slow in pure python for 1M; do blocks of vectorized hashing via
    # hashlib on concatenated counters (still python loop but cheap enough),
    # cache by (n, seed) on disk? Keep simple: single pass, ~4s at 1M.
    for i in range(n):
        ib = int(i).to_bytes(8, "little")
        h1 = hashlib.sha256(b"pk" + ib).digest()
        h2 = hashlib.sha256(b"pk2" + ib).digest()
        out[i, 0:48] = np.frombuffer((h1 + h2)[:48], dtype=np.uint8)
        out[i, 48:80] = np.frombuffer(hashlib.sha256(b"wc" + ib).digest(),
                                      dtype=np.uint8)
    out[:, 80:88] = np.frombuffer(
        np.full(n, EFFECTIVE_BALANCE, dtype="<u8").tobytes(), dtype=np.uint8
    ).reshape(n, 8)
    out[:, 88] = (idx % 97 == 0).astype(np.uint8)
    out[:, 89:97] = np.frombuffer((idx % 1024).astype("<u8").tobytes(),
                                  dtype=np.uint8).reshape(n, 8)
    out[:, 97:105] = np.frombuffer(((idx % 1024) + 1).astype("<u8").tobytes(),
                                   dtype=np.uint8).reshape(n, 8)
    far = np.full(n, 2**64 - 1, dtype=np.uint64)
    out[:, 105:113] = np.frombuffer(far.tobytes(), dtype=np.uint8).reshape(n, 8)
    out[:, 113:121] = np.frombuffer(far.tobytes(), dtype=np.uint8).reshape(n, 8)
    return out.tobytes()


def balances(n: int) -> bytes:
    return np.full(n, EFFECTIVE_BALANCE, dtype="<u8").tobytes()


def inactivity_scores(n: int) -> bytes:
    return np.zeros(n, dtype="<u8").tobytes()


def participation(n: int, fill: int = 7) -> bytes:
    return np.full(n, fill, dtype=np.uint8).tobytes()


def randao_mixes() -> bytes:
    rng = np.random.default_rng(1)
    return rng.integers(0, 256, size=65536 * 32, dtype=np.uint8).tobytes()


def roots_vector(count: int, seed: int) -> bytes:
    rng = np.random.default_rng(seed)
    return rng.integers(0, 256, size=count * 32, dtype=np.uint8).tobytes()
