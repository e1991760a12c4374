"""Host-side rule coverage that needs NO GPU: these paths return before
any device call (the reference keeps the same rules above its backend —
blst.rs:42-44, 80-89; generic_signature.rs:17-26;
generic_public_key.rs:86-94; generic_aggregate_signature.rs:198-210)."""
import pytest

from lighthouse_amd import bls


def test_empty_set_list_is_false():
    assert bls.verify_signature_sets([]) is False  # blst.rs:42-44


def test_empty_signature_is_false():
    s = bls.SignatureSet(
        bls.Signature.empty(), [bls.PublicKey(b"\x00" * 96)], b"\x00" * 32
    )
    assert bls.verify_signature_sets([s]) is False  # blst.rs:80-83


def test_empty_signing_keys_is_false():
    s = bls.SignatureSet(bls.Signature.infinity(), [], b"\x00" * 32)
    assert bls.verify_signature_sets([s]) is False  # blst.rs:86-89


def test_signature_wire_forms():
    assert bls.Signature.from_compressed(bytes(96)).is_empty()
    inf = bls.Signature.from_compressed(bls.INFINITY_SIGNATURE)
    assert inf.is_infinity and not inf.is_empty()
    with pytest.raises(bls.InvalidByteLength):
        bls.Signature.from_compressed(b"\x01" * 95)
    assert bls.Signature.empty().serialize() == bytes(96)
    assert bls.Signature.infinity().serialize() == bls.INFINITY_SIGNATURE


def test_infinity_pubkey_rejected_at_deserialize():
    # generic_public_key.rs:86-94: rejected BEFORE any device work
    with pytest.raises(bls.InvalidInfinityPublicKey):
        bls.PublicKey.deserialize(bls.INFINITY_PUBLIC_KEY)
    with pytest.raises(bls.InvalidByteLength):
        bls.PublicKey.deserialize(b"\x01" * 47)
    with pytest.raises(bls.InvalidByteLength):
        bls.PublicKey.from_uncompressed(b"\x01" * 95)


def test_aggregate_verify_host_rules():
    sig = bls.Signature.infinity()
    pk = bls.PublicKey(b"\x00" * 96)
    assert bls.aggregate_verify(sig, [], []) is False
    assert bls.aggregate_verify(sig, [b"\x00" * 32], [pk, pk]) is False
    assert (
        bls.aggregate_verify(bls.Signature.empty(), [b"\x00" * 32], [pk])
        is False
    )


def test_eth_fast_aggregate_verify_infinity_special_case():
    # generic_aggregate_signature.rs:198-210: infinity sig + NO pubkeys
    # is the one accepting combination, decided host-side
    assert (
        bls.eth_fast_aggregate_verify(
            bls.Signature.infinity(), b"\x00" * 32, []
        )
        is True
    )
    assert (
        bls.fast_aggregate_verify(bls.Signature.infinity(), b"\x00" * 32, [])
        is False
    )
    # empty signature with no pubkeys: not the infinity form -> False
    assert (
        bls.eth_fast_aggregate_verify(bls.Signature.empty(), b"\x00" * 32, [])
        is False
    )
