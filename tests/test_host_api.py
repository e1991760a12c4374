"""Host-API mirrors added in round 2:

- aggregate_verify (blst.rs:263-274) composed on the batch pipeline
- gossip batch-false fallback (attestation_verification/batch.rs:109-127)
- the §8b threading contract: concurrent verify_signature_sets batches
  from multiple threads (per-thread contexts / streams)
"""
import ctypes
import hashlib

import pytest

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def material(oracle):
    """8 interop keypairs + per-key signatures over distinct messages."""
    n = 8
    sks = ctypes.create_string_buffer(32 * n)
    pks = ctypes.create_string_buffer(96 * n)
    oracle.m3x_oracle_bls_keypool(ctypes.c_uint64(n), sks, pks)
    msgs = b"".join(
        hashlib.sha256(b"host-api-%d" % i).digest() for i in range(n)
    )
    sigs = ctypes.create_string_buffer(96 * n)
    assert (
        oracle.m3x_oracle_bls_sign_batch(
            ctypes.c_uint64(n), sks.raw, msgs, sigs
        )
        == 0
    )
    return {
        "n": n,
        "pks": [pks.raw[96 * i : 96 * (i + 1)] for i in range(n)],
        "msgs": [msgs[32 * i : 32 * (i + 1)] for i in range(n)],
        "sigs": [sigs.raw[96 * i : 96 * (i + 1)] for i in range(n)],
    }


def _mk_sets(material, bls, corrupt_idx=None):
    sets = []
    for i in range(material["n"]):
        msg = material["msgs"][i]
        if i == corrupt_idx:
            msg = hashlib.sha256(b"tampered").digest()
        sets.append(
            bls.SignatureSet(
                signature=bls.Signature.from_compressed(material["sigs"][i]),
                signing_keys=[
                    bls.PublicKey.from_uncompressed(material["pks"][i])
                ],
                message=msg,
            )
        )
    return sets


def test_aggregate_verify(material):
    from lighthouse_amd import bls

    n = material["n"]
    agg = bls.aggregate_signatures(
        [bls.Signature.from_compressed(s) for s in material["sigs"]]
    )
    pks = [bls.PublicKey.from_uncompressed(p) for p in material["pks"]]
    assert bls.aggregate_verify(agg, material["msgs"], pks) is True
    # one tampered message -> False
    bad_msgs = list(material["msgs"])
    bad_msgs[2] = hashlib.sha256(b"wrong").digest()
    assert bls.aggregate_verify(agg, bad_msgs, pks) is False
    # pubkey/message permutation -> False
    assert (
        bls.aggregate_verify(agg, material["msgs"], pks[::-1]) is False
    )
    # empty / mismatched inputs -> False (blst.rs host rules)
    assert bls.aggregate_verify(agg, [], []) is False
    assert bls.aggregate_verify(agg, material["msgs"][:3], pks[:2]) is False
    # empty signature -> False
    assert (
        bls.aggregate_verify(bls.Signature.empty(), material["msgs"], pks)
        is False
    )
    assert n == len(pks)


def test_gossip_fallback_identifies_bad_set(material):
    from lighthouse_amd import bls

    good = _mk_sets(material, bls)
    assert bls.verify_signature_sets_with_fallback(good) == [True] * 8
    bad = _mk_sets(material, bls, corrupt_idx=3)
    got = bls.verify_signature_sets_with_fallback(bad)
    assert got == [True, True, True, False, True, True, True, True]


def test_concurrent_batches(material):
    """§8b: concurrent verify_signature_sets from 4 threads, each on its
    own per-thread context/stream, interleaving valid and invalid
    batches — all verdicts must stay correct (no cross-talk, no
    serialization deadlock)."""
    from concurrent.futures import ThreadPoolExecutor

    from lighthouse_amd import bls

    def worker(widx):
        results = []
        for it in range(3):
            good = _mk_sets(material, bls)
            results.append(bls.verify_signature_sets(good))
            bad = _mk_sets(material, bls, corrupt_idx=(widx + it) % 8)
            results.append(bls.verify_signature_sets(bad))
        return results

    with ThreadPoolExecutor(max_workers=4) as ex:
        outs = list(ex.map(worker, range(4)))
    for res in outs:
        assert res == [True, False] * 3
