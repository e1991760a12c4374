"""Batched signing-root computation — the message side of
signature-set construction: signing_root = hash_tree_root(SigningData {
object_root, domain }) (consensus/types/src/signing_data.rs:22-31), and the
AttestationData object roots that feed it
(signature_sets.rs:271-335 callers; AttestationData fields per
consensus/types/src/attestation_data.rs: slot, index, beacon_block_root,
source: Checkpoint, target: Checkpoint).

All hashing runs through one or two m3x_merkleize_batch launches."""
from . import tree_hash as th


def _c(b: bytes) -> bytes:
    return b + b"\x00" * (32 - len(b))


def _u64(v: int) -> bytes:
    return int(v).to_bytes(8, "little")


def signing_roots(object_roots, domains, ctx=None):
    """signing_root for each (object_root, domain) pair — one batched
    launch of 2-chunk trees (signing_data.rs:22-31)."""
    assert len(object_roots) == len(domains)
    groups = [r + d for r, d in zip(object_roots, domains)]
    return th.merkleize_batch(groups, ctx=ctx)


def attestation_data_roots(atts, ctx=None):
    """hash_tree_root for a batch of AttestationData tuples
    (slot, index, beacon_block_root, (src_epoch, src_root),
    (tgt_epoch, tgt_root)). Two batched launches: checkpoints, then the
    5-field containers."""
    ck_groups = []
    for (_, _, _, src, tgt) in atts:
        ck_groups.append(_c(_u64(src[0])) + src[1])
        ck_groups.append(_c(_u64(tgt[0])) + tgt[1])
    ck_roots = th.merkleize_batch(ck_groups, ctx=ctx) if ck_groups else []
    groups = []
    for i, (slot, index, bbr, _, _) in enumerate(atts):
        groups.append(
            _c(_u64(slot)) + _c(_u64(index)) + bbr
            + ck_roots[2 * i] + ck_roots[2 * i + 1]
        )
    return th.merkleize_batch(groups, ctx=ctx)


def attestation_signing_roots(atts, domain: bytes, ctx=None):
    """object roots + signing roots for a gossip attestation batch (the
    hot construction path of signature_sets.rs:271-335)."""
    roots = attestation_data_roots(atts, ctx=ctx)
    return signing_roots(roots, [domain] * len(roots), ctx=ctx)
