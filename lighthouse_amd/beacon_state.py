"""Synthetic Deneb-shaped BeaconState + GPU hash_tree_root (hot path #2,
BASELINE config C3 proper).

Field list and SSZ order follow the reference's BeaconStateDeneb
(consensus/types/src/beacon_state.rs:224-560; sub-containers:
fork.rs, beacon_block_header.rs, eth1_data.rs, checkpoint.rs,
sync_committee.rs, execution_payload_header.rs Deneb variant,
historical_summary.rs). All hashing runs on the GPU through
lighthouse_amd.tree_hash; the layout constants are mainnet EthSpec typenums
(eth_spec.rs: SlotsPerHistoricalRoot=8192, HistoricalRootsLimit=2^24,
SlotsPerEth1VotingPeriod=2048, EpochsPerHistoricalVector=65536,
EpochsPerSlashingsVector=8192, SyncCommitteeSize=512,
ValidatorRegistryLimit=2^40)."""
import hashlib

import numpy as np

from . import state as synth, tree_hash as th

REGISTRY_LIMIT = 1 << 40
SLOTS_PER_HISTORICAL_ROOT = 8192
HISTORICAL_ROOTS_LIMIT = 1 << 24
SLOTS_PER_ETH1_VOTING = 2048
EPOCHS_PER_HISTORICAL_VECTOR = 65536
EPOCHS_PER_SLASHINGS = 8192
SYNC_COMMITTEE_SIZE = 512


def _h(tag: bytes) -> bytes:
    return hashlib.sha256(tag).digest()


def _u64(v: int) -> bytes:
    return int(v).to_bytes(8, "little")


def generate(n_validators: int, seed: int = 0xC0FFEE) -> dict:
    """Deterministic synthetic Deneb state (raw field bytes)."""
    rng = np.random.default_rng(seed)

    def rb(n):
        return rng.integers(0, 256, size=n, dtype=np.uint8).tobytes()

    n = n_validators
    st = {
        "genesis_time": 1606824023,
        "genesis_validators_root": _h(b"gvr"),
        "slot": 8192 * 300 + 17,
        "fork": (b"\x03\x00\x00\x00", b"\x04\x00\x00\x00", 269568),
        "latest_block_header": (st_slot := 8192 * 300 + 16, 42,
                                _h(b"parent"), b"\x00" * 32, _h(b"body")),
        "block_roots": rb(32 * SLOTS_PER_HISTORICAL_ROOT),
        "state_roots": rb(32 * SLOTS_PER_HISTORICAL_ROOT),
        "historical_roots": rb(32 * 7),  # 7 entries
        "eth1_data": (_h(b"dep_root"), n + 100, _h(b"blk")),
        "eth1_data_votes": [(_h(b"dr%d" % i), n + i, _h(b"bh%d" % i))
                            for i in range(33)],
        "eth1_deposit_index": n + 100,
        "validators_ssz": synth.validators_ssz(n),
        "n_validators": n,
        "balances": synth.balances(n),
        "randao_mixes": rb(32 * EPOCHS_PER_HISTORICAL_VECTOR),
        "slashings": np.zeros(EPOCHS_PER_SLASHINGS, dtype="<u8").tobytes(),
        "previous_epoch_participation": synth.participation(n, 7),
        "current_epoch_participation": synth.participation(n, 3),
        "justification_bits": b"\x0f",
        "previous_justified_checkpoint": (81919, _h(b"pjc")),
        "current_justified_checkpoint": (81920, _h(b"cjc")),
        "finalized_checkpoint": (81919, _h(b"fc")),
        "inactivity_scores": synth.inactivity_scores(n),
        "current_sync_committee": ([bytes(_h(b"sc%d" % i) + _h(b"sc2%d" % i))[:48]
                                    for i in range(SYNC_COMMITTEE_SIZE)],
                                   _h(b"agg") + _h(b"agg2")[:16]),
        "next_sync_committee": ([bytes(_h(b"ns%d" % i) + _h(b"ns2%d" % i))[:48]
                                 for i in range(SYNC_COMMITTEE_SIZE)],
                                _h(b"nagg") + _h(b"nagg2")[:16]),
        "latest_execution_payload_header": {
            "parent_hash": _h(b"eph"),
            "fee_recipient": _h(b"fee")[:20],
            "state_root": _h(b"esr"),
            "receipts_root": _h(b"err"),
            "logs_bloom": rb(256),
            "prev_randao": _h(b"rand"),
            "block_number": 20_000_000,
            "gas_limit": 30_000_000,
            "gas_used": 12_345_678,
            "timestamp": 1726000000,
            "extra_data": b"m3x",
            "base_fee_per_gas": (7 * 10**9).to_bytes(32, "little"),
            "block_hash": _h(b"ebh"),
            "transactions_root": _h(b"txr"),
            "withdrawals_root": _h(b"wdr"),
            "blob_gas_used": 131072,
            "excess_blob_gas": 0,
        },
        "next_withdrawal_index": 55_000_000,
        "next_withdrawal_validator_index": 12345,
        "historical_summaries": [(_h(b"hs%d" % i), _h(b"hss%d" % i))
                                 for i in range(293)],
    }
    return st


def _chunk(b: bytes) -> bytes:
    return b + b"\x00" * (32 - len(b))


def _container_root(chunks, ctx=None) -> bytes:
    n = len(chunks)
    d = 0
    while (1 << d) < n:
        d += 1
    return th.merkleize_chunks(b"".join(chunks), n, d, -1, ctx=ctx)


BIG_FIELDS = [
    # (name, pad-to-32 for chunk loads)
    "validators_ssz", "balances", "randao_mixes", "slashings",
    "previous_epoch_participation", "current_epoch_participation",
    "inactivity_scores", "block_roots", "state_roots",
]


def upload_fields(st: dict, ctx) -> dict:
    """Upload the large state fields to HBM once (bench: inputs resident
    when the timed region starts). Buffers are zero-padded to 32B."""
    dev = {}
    for name in BIG_FIELDS:
        data = st[name]
        pad = (-len(data)) % 32
        dev[name] = ctx.upload(data + b"\x00" * pad)
    return dev


def state_root(st: dict, ctx=None, dev=None, registry_root=None) -> bytes:
    """hash_tree_root of the synthetic Deneb state — every hash on GPU.
    With `dev` (from upload_fields) the big fields are read from HBM;
    with `registry_root` the validators field root is taken as given
    (multi-GPU sharded path computes it collectively). The many tiny
    containers go through ONE m3x_merkleize_batch launch; the full
    sync-committee pubkey vectors flatten into single merkleize calls
    (valid because the vectors are full: element roots at depth 1 of the
    combined chunk tree equal the per-element roots)."""
    from . import _native

    ctx = ctx or _native.default_ctx()
    C = _chunk
    u = _u64

    # ---- batch 1: every independent small container ----
    groups = []

    def grp(*chunks):
        groups.append(b"".join(chunks))
        return len(groups) - 1

    pv, cv, ep = st["fork"]
    gi_fork = grp(C(pv), C(cv), C(u(ep)))
    sl, pi, pr, sr, br = st["latest_block_header"]
    gi_hdr = grp(C(u(sl)), C(u(pi)), pr, sr, br)
    dr, dc, bh = st["eth1_data"]
    gi_eth1 = grp(dr, C(u(dc)), bh)
    gi_votes = [grp(v[0], C(u(v[1])), v[2]) for v in st["eth1_data_votes"]]
    gi_pjc = grp(C(u(st["previous_justified_checkpoint"][0])),
                 st["previous_justified_checkpoint"][1])
    gi_cjc = grp(C(u(st["current_justified_checkpoint"][0])),
                 st["current_justified_checkpoint"][1])
    gi_fc = grp(C(u(st["finalized_checkpoint"][0])),
                st["finalized_checkpoint"][1])
    h = st["latest_execution_payload_header"]
    # payload needs two sub-roots first (logs_bloom, extra_data) — both are
    # fixed 8-chunk / 1-chunk basic vectors: fold them in the same batch
    gi_bloom = grp(*[h["logs_bloom"][32 * i : 32 * (i + 1)] for i in range(8)])
    gi_agg1 = grp(C(st["current_sync_committee"][1][:32]),
                  C(st["current_sync_committee"][1][32:48]))
    gi_agg2 = grp(C(st["next_sync_committee"][1][:32]),
                  C(st["next_sync_committee"][1][32:48]))
    gi_hs = [grp(a, b) for a, b in st["historical_summaries"]]
    roots = th.merkleize_batch(groups, ctx=ctx)

    # ---- big/independent fields ----
    f6 = (th.root_vector_root_dev(dev["block_roots"], SLOTS_PER_HISTORICAL_ROOT, ctx)
          if dev else th.root_vector_root(st["block_roots"], SLOTS_PER_HISTORICAL_ROOT, ctx))
    f7 = (th.root_vector_root_dev(dev["state_roots"], SLOTS_PER_HISTORICAL_ROOT, ctx)
          if dev else th.root_vector_root(st["state_roots"], SLOTS_PER_HISTORICAL_ROOT, ctx))
    f8 = th.root_list_root(st["historical_roots"],
                           len(st["historical_roots"]) // 32,
                           HISTORICAL_ROOTS_LIMIT, ctx)
    votes_roots = b"".join(roots[i] for i in gi_votes)
    f10 = th.root_list_root(votes_roots, len(gi_votes), SLOTS_PER_ETH1_VOTING, ctx)
    n = st["n_validators"]
    if registry_root is not None:
        f12 = registry_root
    elif dev:
        f12 = th.validator_registry_root_dev(dev["validators_ssz"], n, ctx)
    else:
        f12 = th.validator_registry_root(st["validators_ssz"], n, ctx)
    if dev:
        f13 = th.basic_list_root_dev(dev["balances"], n, 8, REGISTRY_LIMIT, ctx)
        f14 = th.root_vector_root_dev(dev["randao_mixes"], EPOCHS_PER_HISTORICAL_VECTOR, ctx)
        f15 = th.basic_vector_root_dev(dev["slashings"], EPOCHS_PER_SLASHINGS, 8, ctx)
        f16 = th.basic_list_root_dev(dev["previous_epoch_participation"], n, 1, REGISTRY_LIMIT, ctx)
        f17 = th.basic_list_root_dev(dev["current_epoch_participation"], n, 1, REGISTRY_LIMIT, ctx)
        f22 = th.basic_list_root_dev(dev["inactivity_scores"], n, 8, REGISTRY_LIMIT, ctx)
    else:
        f13 = th.basic_list_root(st["balances"], n, 8, REGISTRY_LIMIT, ctx)
        f14 = th.root_vector_root(st["randao_mixes"], EPOCHS_PER_HISTORICAL_VECTOR, ctx)
        f15 = th.basic_vector_root(st["slashings"], EPOCHS_PER_SLASHINGS, 8, ctx)
        f16 = th.basic_list_root(st["previous_epoch_participation"], n, 1, REGISTRY_LIMIT, ctx)
        f17 = th.basic_list_root(st["current_epoch_participation"], n, 1, REGISTRY_LIMIT, ctx)
        f22 = th.basic_list_root(st["inactivity_scores"], n, 8, REGISTRY_LIMIT, ctx)

    # sync-committee pubkeys: FULL Vector[Bytes48, 512] -> flattened
    # 1024-chunk tree at depth 10 (single call)
    def pks_root(pks):
        flat = b"".join(C(pk[:32]) + C(pk[32:48]) for pk in pks)
        return th.merkleize_chunks(flat, 1024, 10, -1, ctx=ctx)

    pks1 = pks_root(st["current_sync_committee"][0])
    pks2 = pks_root(st["next_sync_committee"][0])
    hs_roots = b"".join(roots[i] for i in gi_hs)
    f28 = th.root_list_root(hs_roots, len(gi_hs), HISTORICAL_ROOTS_LIMIT, ctx)

    # ---- batch 2: containers that depend on batch-1/sub roots ----
    groups2 = [
        pks1 + roots[gi_agg1],  # current_sync_committee
        pks2 + roots[gi_agg2],  # next_sync_committee
        # payload header (17 field chunks)
        b"".join([
            h["parent_hash"], C(h["fee_recipient"]), h["state_root"],
            h["receipts_root"], roots[gi_bloom], h["prev_randao"],
            C(u(h["block_number"])), C(u(h["gas_limit"])),
            C(u(h["gas_used"])), C(u(h["timestamp"])),
            th.basic_list_root(h["extra_data"], len(h["extra_data"]), 1, 32, ctx),
            h["base_fee_per_gas"], h["block_hash"], h["transactions_root"],
            h["withdrawals_root"], C(u(h["blob_gas_used"])),
            C(u(h["excess_blob_gas"])),
        ]),
    ]
    r2 = th.merkleize_batch(groups2, ctx=ctx)

    f = [
        C(u(st["genesis_time"])), st["genesis_validators_root"],
        C(u(st["slot"])), roots[gi_fork], roots[gi_hdr], f6, f7, f8,
        roots[gi_eth1], f10, C(u(st["eth1_deposit_index"])), f12, f13, f14,
        f15, f16, f17, C(st["justification_bits"]), roots[gi_pjc],
        roots[gi_cjc], roots[gi_fc], f22, r2[0], r2[1], r2[2],
        C(u(st["next_withdrawal_index"])),
        C(u(st["next_withdrawal_validator_index"])), f28,
    ]
    assert len(f) == 28
    return _container_root(f, ctx)


def node_hash_count(n_validators: int) -> int:
    """two-to-one node hashes in a full rebuild (SURVEY §8a accounting)."""
    n = n_validators

    def list_hashes(n_chunks, limit_chunks):
        d = 0
        while (1 << d) < limit_chunks:
            d += 1
        # real tree + zero cap + mix
        real = max(n_chunks - 1, 0)
        dreal = 0
        while (1 << dreal) < max(n_chunks, 1):
            dreal += 1
        return real + (d - dreal) + 1

    total = 8 * n + list_hashes(n, REGISTRY_LIMIT)  # registry (leaf 8/val)
    total += list_hashes((n * 8 + 31) // 32, REGISTRY_LIMIT // 4)  # balances
    total += list_hashes((n + 31) // 32, REGISTRY_LIMIT // 32) * 2  # particip.
    total += list_hashes((n * 8 + 31) // 32, REGISTRY_LIMIT // 4)  # inactivity
    total += 65535 + 2 * 8191 + 2047  # randao + block/state roots + slashings
    total += 2 * (512 + 511 + 1 + 1 + 1)  # sync committees
    total += 33 * 2 + list_hashes(33, 2048)  # eth1 votes
    total += 293 + list_hashes(293, HISTORICAL_ROOTS_LIMIT)  # hist summaries
    total += 40  # small containers + top level, approx
    return total
