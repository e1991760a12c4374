"""Host mirror of the reference's `crypto/bls` surface over the m3x C-ABI
(hot path #1). Mirrors the define_mod! type surface and the host-side rules
the reference keeps above its backend (crypto/bls/src/lib.rs:86-141,
generic_signature.rs, generic_aggregate_signature.rs,
generic_public_key.rs:86-94, impls/blst.rs:37-119):

 - all-zeros serialization = "empty"; empty signature/keys => verify False
 - 0xc0... = point at infinity (valid G2 element; fails the equation)
 - infinity PUBKEY rejected at deserialize
 - r_i drawn host-side: uniform 64-bit, redrawn while zero (RAND_BITS=64)
 - verification failure is False, never an exception

GPU-only compute — no CPU fallback."""
import ctypes
import secrets

from . import _native

PUBLIC_KEY_BYTES_LEN = 48
PUBLIC_KEY_UNCOMPRESSED_BYTES_LEN = 96
SIGNATURE_BYTES_LEN = 96
INFINITY_PUBLIC_KEY = bytes([0xC0] + [0] * 47)
INFINITY_SIGNATURE = bytes([0xC0] + [0] * 95)
NONE_SIGNATURE = bytes(96)
RAND_BITS = 64


class Error(Exception):
    pass


class InvalidInfinityPublicKey(Error):
    pass


class InvalidByteLength(Error):
    pass


class BlstError(Error):
    pass


class PublicKey:
    """A validated BLS public key (uncompressed affine G1, 96B) — the
    decompressed pubkey-cache representation (validator_pubkey_cache.rs)."""

    __slots__ = ("_uncomp",)

    def __init__(self, uncomp: bytes):
        self._uncomp = bytes(uncomp)

    @classmethod
    def from_uncompressed(cls, b: bytes) -> "PublicKey":
        # deserialize_uncompressed: trusted bytes from serialize_uncompressed
        # (generic_public_key.rs:35-39)
        if len(b) != PUBLIC_KEY_UNCOMPRESSED_BYTES_LEN:
            raise InvalidByteLength(len(b))
        return cls(b)

    @classmethod
    def deserialize(cls, b: bytes, ctx=None) -> "PublicKey":
        """compressed-form deserialize = key_validate: subgroup + infinity
        checks (blst.rs:130-140; infinity: generic_public_key.rs:86-94)."""
        if len(b) != PUBLIC_KEY_BYTES_LEN:
            raise InvalidByteLength(len(b))
        if bytes(b) == INFINITY_PUBLIC_KEY:
            raise InvalidInfinityPublicKey()
        out, status = decompress_pubkeys(bytes(b), 1, ctx=ctx)
        if status[0] != 0:
            raise BlstError(f"key_validate failed ({status[0]})")
        return cls(out)

    def serialize_uncompressed(self) -> bytes:
        return self._uncomp


def decompress_pubkeys(comp: bytes, n: int, ctx=None):
    """Batch key_validate on GPU: returns (uncompressed n*96, status list)."""
    ctx = ctx or _native.default_ctx()
    out = ctypes.create_string_buffer(96 * n)
    status = (ctypes.c_int32 * n)()
    rc = ctx._lib.m3x_bls_pk_decompress(ctx.handle, comp, n, out, status)
    if rc != 0:
        raise Error(f"m3x_bls_pk_decompress rc={rc}")
    return out.raw, list(status)


class Signature:
    """A (possibly aggregate) signature: 96B compressed G2 wire form.
    point=None models the 'empty' (all-zero) value
    (generic_aggregate_signature.rs:87-105)."""

    __slots__ = ("_comp", "is_infinity")

    def __init__(self, comp, is_infinity=False):
        self._comp = comp  # None = empty
        self.is_infinity = is_infinity

    @classmethod
    def from_compressed(cls, b: bytes) -> "Signature":
        if len(b) != SIGNATURE_BYTES_LEN:
            raise InvalidByteLength(len(b))
        b = bytes(b)
        if b == NONE_SIGNATURE:
            return cls(None, False)
        return cls(b, b == INFINITY_SIGNATURE)

    @classmethod
    def empty(cls) -> "Signature":
        return cls(None, False)

    @classmethod
    def infinity(cls) -> "Signature":
        return cls(INFINITY_SIGNATURE, True)

    def is_empty(self) -> bool:
        return self._comp is None

    def serialize(self) -> bytes:
        return self._comp if self._comp is not None else NONE_SIGNATURE


def aggregate_signatures(sigs, ctx=None) -> "Signature":
    """Aggregate signatures on GPU (TAggregateSignature::add_assign chain,
    generic_aggregate_signature.rs:124-150): empty inputs are skipped (the
    reference's empty value contributes nothing); returns the compressed
    sum (possibly infinity)."""
    import ctypes as ct

    blobs = [s.serialize() for s in sigs if not s.is_empty()]
    if not blobs:
        return Signature.infinity()
    ctx = ctx or _native.default_ctx()
    out = ct.create_string_buffer(96)
    rc = ctx._lib.m3x_bls_sig_aggregate(
        ctx.handle, b"".join(blobs), len(blobs), out
    )
    if rc != 0:
        raise Error(f"m3x_bls_sig_aggregate rc={rc}")
    return Signature.from_compressed(out.raw)


AggregateSignature = Signature


class SignatureSet:
    """GenericSignatureSet (generic_signature_set.rs:61-121): an aggregate
    signature, >=1 signing keys, and one 32B signing-root message."""

    __slots__ = ("signature", "signing_keys", "message")

    def __init__(self, signature: Signature, signing_keys, message: bytes):
        assert len(message) == 32
        self.signature = signature
        self.signing_keys = list(signing_keys)
        self.message = bytes(message)


def _draw_rands(n):
    # blst.rs:53-68: 64-bit scalars, redrawn while zero
    out = []
    for _ in range(n):
        v = 0
        while v == 0:
            v = int.from_bytes(secrets.token_bytes(8), "little")
        out.append(v)
    return out


def verify(signature: "Signature", pubkey: PublicKey, msg: bytes,
           ctx=None) -> bool:
    """Single-set verify (TSignature::verify, blst.rs:196-200): the batch
    equation with r=1 degenerates to e(pk, H(m)) == e(g1, sigma)."""
    if signature.is_empty():
        return False
    s = SignatureSet(signature, [pubkey], msg)
    return verify_signature_sets([s], ctx=ctx, _rands=[1])


def fast_aggregate_verify(signature: "Signature", msg: bytes, pubkeys,
                          ctx=None) -> bool:
    """TAggregateSignature::fast_aggregate_verify (blst.rs:250-261;
    generic_aggregate_signature.rs:187-196): one message, k pubkeys."""
    pubkeys = list(pubkeys)
    if not pubkeys:
        return False
    if signature.is_empty():
        return False
    s = SignatureSet(signature, pubkeys, msg)
    return verify_signature_sets([s], ctx=ctx, _rands=[1])


def eth_fast_aggregate_verify(signature: "Signature", msg: bytes, pubkeys,
                              ctx=None) -> bool:
    """generic_aggregate_signature.rs:198-210: accepts the
    G2_POINT_AT_INFINITY signature when pubkeys is empty (sync-aggregate
    rule)."""
    pubkeys = list(pubkeys)
    if not pubkeys and signature.serialize() == INFINITY_SIGNATURE:
        return True
    return fast_aggregate_verify(signature, msg, pubkeys, ctx=ctx)


def aggregate_verify(signature: "Signature", msgs, pubkeys, ctx=None) -> bool:
    """TAggregateSignature::aggregate_verify (blst.rs:263-274): ONE
    aggregate signature over n distinct messages, one pubkey each —
    e(g1, sig) == prod_i e(pk_i, H(m_i)).

    Runs on the existing batch pipeline with no extra kernel: n sets
    {msg_i, pk_i} with sigma_1 = sig, sigma_i>1 = infinity and r_i = 1
    compute prod e(PK_i, H(m_i)) * e(-g1, sig) == 1 — exactly the
    aggregate_verify equation. r=1 is correct here: this is a single
    equation, not a batch of independent claims, and the reference's
    aggregate_verify uses no randomizers either. (The reference keeps
    this 'only for EF tests' — generic_aggregate_signature.rs:44-47.)"""
    msgs = [bytes(m) for m in msgs]
    pubkeys = list(pubkeys)
    if not msgs or len(msgs) != len(pubkeys):
        return False
    if signature.is_empty():
        return False
    sets = [
        SignatureSet(
            signature if i == 0 else Signature.infinity(), [pk], m
        )
        for i, (m, pk) in enumerate(zip(msgs, pubkeys))
    ]
    return verify_signature_sets(sets, ctx=ctx, _rands=[1] * len(sets))


def verify_signature_sets_with_fallback(sets, ctx=None):
    """Gossip-batch driver semantics (attestation_verification/
    batch.rs:109-127): one batch check; if the batch verdict is false,
    re-verify each set individually to identify the bad items (the
    reference maps each result back to accept/reject per attestation).
    Returns a per-set list of bools."""
    sets = list(sets)
    if verify_signature_sets(sets, ctx=ctx):
        return [True] * len(sets)
    # set.verify() path (generic_signature_set.rs:111-120): single-set
    # fast_aggregate_verify, no randomizer
    return [
        verify_signature_sets([s], ctx=ctx, _rands=[1]) for s in sets
    ]


def verify_signature_sets(sets, ctx=None, _rands=None) -> bool:
    """bls::verify_signature_sets (blst.rs:37-119). `_rands` exists for
    deterministic tests only."""
    sets = list(sets)
    if not sets:
        return False  # blst.rs:42-44
    msgs = bytearray()
    sigs = bytearray()
    pks = bytearray()
    offsets = [0]
    for s in sets:
        if s.signature.is_empty():
            return False  # blst.rs:80-83
        if not s.signing_keys:
            return False  # blst.rs:86-89
        msgs += s.message
        sigs += s.signature.serialize()
        for pk in s.signing_keys:
            pks += pk.serialize_uncompressed()
        offsets.append(offsets[-1] + len(s.signing_keys))
    n = len(sets)
    rands = _rands if _rands is not None else _draw_rands(n)
    ctx = ctx or _native.default_ctx()
    off_arr = (ctypes.c_uint32 * (n + 1))(*offsets)
    rand_arr = (ctypes.c_uint64 * n)(*rands)
    rc = ctx._lib.m3x_bls_verify_sets(
        ctx.handle, bytes(msgs), bytes(sigs), bytes(pks), off_arr, rand_arr, n
    )
    if rc < 0:
        raise Error(f"m3x_bls_verify_sets rc={rc}")
    return rc == 1
