"""Independent (hashlib-based) restatement of the SSZ merkleize semantics,
used only to pin the C oracle in tests. Semantics follow
/root/reference/consensus/merkle_proof/src/lib.rs:9-14,68-100 (right-sparse
tree, zero ladder), deposit_data_tree.rs:26-38 (mix_in_length), and
tree_hash packing rules (32B chunks, little-endian)."""
import hashlib


def H(x: bytes) -> bytes:
    return hashlib.sha256(x).digest()


ZEROS = [b"\x00" * 32]
for _ in range(64):
    ZEROS.append(H(ZEROS[-1] + ZEROS[-1]))


def merkleize(chunks, depth):
    n = len(chunks)
    assert n <= (1 << depth)
    if n == 0:
        return ZEROS[depth]
    nodes = list(chunks)
    for level in range(depth):
        nxt = []
        for i in range(0, len(nodes), 2):
            left = nodes[i]
            right = nodes[i + 1] if i + 1 < len(nodes) else ZEROS[level]
            nxt.append(H(left + right))
        nodes = nxt
    return nodes[0]


def mix_in_length(root: bytes, length: int) -> bytes:
    return H(root + length.to_bytes(32, "little"))


def pack_bytes(data: bytes):
    chunks = []
    for i in range(0, len(data), 32):
        c = data[i : i + 32]
        chunks.append(c + b"\x00" * (32 - len(c)))
    return chunks


def ceil_log2(x: int) -> int:
    d = 0
    while (1 << d) < x:
        d += 1
    return d


def basic_list_root(data: bytes, n_elems: int, elem_size: int, limit: int):
    limit_chunks = max((limit * elem_size + 31) // 32, 1)
    root = merkleize(pack_bytes(data), ceil_log2(limit_chunks))
    return mix_in_length(root, n_elems)


def basic_vector_root(data: bytes, n_elems: int, elem_size: int):
    limit_chunks = max((n_elems * elem_size + 31) // 32, 1)
    return merkleize(pack_bytes(data), ceil_log2(limit_chunks))


def validator_leaf(ssz: bytes) -> bytes:
    """validator.rs:25-35: pubkey48 | wc32 | eff_bal8 | slashed1 | 4 epochs."""
    assert len(ssz) == 121
    chunks = [
        H(ssz[0:32] + ssz[32:48] + b"\x00" * 16),
        ssz[48:80],
        ssz[80:88] + b"\x00" * 24,
        ssz[88:89] + b"\x00" * 31,
        ssz[89:97] + b"\x00" * 24,
        ssz[97:105] + b"\x00" * 24,
        ssz[105:113] + b"\x00" * 24,
        ssz[113:121] + b"\x00" * 24,
    ]
    return merkleize(chunks, 3)


def validator_registry_root(ssz: bytes, n: int) -> bytes:
    leaves = [validator_leaf(ssz[121 * i : 121 * (i + 1)]) for i in range(n)]
    return mix_in_length(merkleize(leaves, 40), n)


def synthetic_validator_ssz(i: int) -> bytes:
    """Deterministic synthetic validator (shape of BASELINE config C3):
    pubkey bytes deterministic 48B (hashed as-is), wc = H(index), eff_bal
    32e9, epochs patterned."""
    pk = (H(b"pk" + i.to_bytes(8, "little")) + H(b"pk2" + i.to_bytes(8, "little")))[:48]
    wc = H(b"wc" + i.to_bytes(8, "little"))
    eff = (32 * 10**9).to_bytes(8, "little")
    slashed = b"\x01" if i % 97 == 0 else b"\x00"
    aee = (i % 1024).to_bytes(8, "little")
    ae = ((i % 1024) + 1).to_bytes(8, "little")
    ee = (2**64 - 1).to_bytes(8, "little")
    we = (2**64 - 1).to_bytes(8, "little")
    return pk + wc + eff + slashed + aee + ae + ee + we


# ---- full Deneb BeaconState root (mirror of lighthouse_amd.beacon_state,
# computed with hashlib; used to pin the GPU composition) ----

def _c(b):
    return b + b"\x00" * (32 - len(b))


def _u64le(v):
    return int(v).to_bytes(8, "little")


def _container(chunks):
    d = 0
    while (1 << d) < len(chunks):
        d += 1
    return merkleize(chunks, d)


def _list_of_roots(roots, n, limit):
    return mix_in_length(merkleize(roots, ceil_log2(max(limit, 1))), n)


def beacon_state_root_ref(st):
    f = []
    f.append(_c(_u64le(st["genesis_time"])))
    f.append(st["genesis_validators_root"])
    f.append(_c(_u64le(st["slot"])))
    pv, cv, ep = st["fork"]
    f.append(_container([_c(pv), _c(cv), _c(_u64le(ep))]))
    sl, pi, pr, sr, br = st["latest_block_header"]
    f.append(_container([_c(_u64le(sl)), _c(_u64le(pi)), pr, sr, br]))
    f.append(merkleize(pack_bytes(st["block_roots"]), 13))
    f.append(merkleize(pack_bytes(st["state_roots"]), 13))
    f.append(_list_of_roots(pack_bytes(st["historical_roots"]),
                            len(st["historical_roots"]) // 32, 1 << 24))

    def eth1_root(e):
        dr, dc, bh = e
        return _container([dr, _c(_u64le(dc)), bh])

    f.append(eth1_root(st["eth1_data"]))
    f.append(_list_of_roots([eth1_root(v) for v in st["eth1_data_votes"]],
                            len(st["eth1_data_votes"]), 2048))
    f.append(_c(_u64le(st["eth1_deposit_index"])))
    f.append(validator_registry_root(st["validators_ssz"], st["n_validators"]))
    f.append(basic_list_root(st["balances"], st["n_validators"], 8, 1 << 40))
    f.append(merkleize(pack_bytes(st["randao_mixes"]), 16))
    f.append(basic_vector_root(st["slashings"], 8192, 8))
    f.append(basic_list_root(st["previous_epoch_participation"],
                             st["n_validators"], 1, 1 << 40))
    f.append(basic_list_root(st["current_epoch_participation"],
                             st["n_validators"], 1, 1 << 40))
    f.append(_c(st["justification_bits"]))

    def ckpt(c):
        e, r = c
        return _container([_c(_u64le(e)), r])

    f.append(ckpt(st["previous_justified_checkpoint"]))
    f.append(ckpt(st["current_justified_checkpoint"]))
    f.append(ckpt(st["finalized_checkpoint"]))
    f.append(basic_list_root(st["inactivity_scores"], st["n_validators"], 8,
                             1 << 40))

    def sync_root(sc):
        pks, agg = sc
        leaves = [merkleize([_c(pk[:32]), _c(pk[32:48])], 1) for pk in pks]
        return _container([merkleize(leaves, 9),
                           merkleize([_c(agg[:32]), _c(agg[32:48])], 1)])

    f.append(sync_root(st["current_sync_committee"]))
    f.append(sync_root(st["next_sync_committee"]))
    h = st["latest_execution_payload_header"]
    ph = [
        h["parent_hash"], _c(h["fee_recipient"]), h["state_root"],
        h["receipts_root"], merkleize(pack_bytes(h["logs_bloom"]), 3),
        h["prev_randao"], _c(_u64le(h["block_number"])),
        _c(_u64le(h["gas_limit"])), _c(_u64le(h["gas_used"])),
        _c(_u64le(h["timestamp"])),
        basic_list_root(h["extra_data"], len(h["extra_data"]), 1, 32),
        h["base_fee_per_gas"], h["block_hash"], h["transactions_root"],
        h["withdrawals_root"], _c(_u64le(h["blob_gas_used"])),
        _c(_u64le(h["excess_blob_gas"])),
    ]
    f.append(_container(ph))
    f.append(_c(_u64le(st["next_withdrawal_index"])))
    f.append(_c(_u64le(st["next_withdrawal_validator_index"])))
    f.append(_list_of_roots([_container([a, b])
                             for a, b in st["historical_summaries"]],
                            len(st["historical_summaries"]), 1 << 24))
    assert len(f) == 28
    return _container(f)


def signing_root_ref(object_root: bytes, domain: bytes) -> bytes:
    """signing_data.rs:22-31: hash_tree_root(SigningData{object_root,domain})"""
    return merkleize([object_root, domain], 1)


def attestation_data_root_ref(slot, index, bbr, src, tgt) -> bytes:
    ck = lambda e, r: merkleize([_c(_u64le(e)), r], 1)
    return merkleize(
        [_c(_u64le(slot)), _c(_u64le(index)), bbr, ck(*src), ck(*tgt)], 3
    )
