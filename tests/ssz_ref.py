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
