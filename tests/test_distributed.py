"""world_size-2 gloo CPU tests of the multi-GPU sharding logic,
driven with a stub compute backend (the reference's fake_crypto strategy —
crypto/bls/src/impls/fake_crypto.rs): the collectives and the shard/combine
math are exercised for real; the per-rank compute is a hashlib stub whose
answers are cross-checked against ssz_ref."""
import os
import sys
from pathlib import Path

import pytest
import torch.multiprocessing as mp

REPO = Path(__file__).resolve().parent.parent


def _worker(rank, world, port, results):
    import torch.distributed as dist

    sys.path.insert(0, str(REPO))
    sys.path.insert(0, str(REPO / "tests"))
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    import ssz_ref
    from lighthouse_amd import distributed as d

    # ---- partition: k-weighted, covers all, deterministic ----
    costs = [1] * 48 + [512] * 16
    parts = d.partition_sets(costs, world)
    assert sorted(i for p in parts for i in p) == list(range(64))
    loads = [sum(costs[i] for i in p) for p in parts]
    assert max(loads) - min(loads) <= 512

    # ---- verdict AND-reduce with a stub verifier ----
    # even ranks see only valid sets; an injected bad set on rank-assigned
    # index 5 must flip the global verdict to False
    def run_ok(idxs):
        return True

    def run_with_bad(idxs):
        return 5 not in idxs

    assert d.verify_sets_sharded(costs, run_ok) is True
    assert d.verify_sets_sharded(costs, run_with_bad) is False

    # ---- sharded registry root vs single-process ssz_ref ----
    n = 1024
    ssz = b"".join(ssz_ref.synthetic_validator_ssz(i) for i in range(n))

    def subtree_fn(start, count, depth):
        leaves = [
            ssz_ref.validator_leaf(ssz[121 * i : 121 * (i + 1)])
            for i in range(start, start + count)
        ]
        return ssz_ref.merkleize(leaves, depth)

    def hash2_fn(l, r):
        return ssz_ref.H(l + r)

    def finalize_fn(node, from_level, to_depth, mix_len):
        for lvl in range(from_level, to_depth):
            node = ssz_ref.H(node + ssz_ref.ZEROS[lvl])
        if mix_len >= 0:
            node = ssz_ref.mix_in_length(node, mix_len)
        return node

    root = d.registry_root_sharded(subtree_fn, hash2_fn, finalize_fn, n, n)
    if rank == 0:
        want = ssz_ref.validator_registry_root(ssz, n)
        assert root == want, "sharded root mismatch"
    dist.barrier()
    dist.destroy_process_group()
    results[rank] = "ok"


def test_gloo_world2():
    import random

    port = random.randint(29600, 29999)
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        procs = [
            ctx.Process(target=_worker, args=(r, 2, port, results))
            for r in range(2)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=180)
        for p in procs:
            assert p.exitcode == 0, f"worker failed (exit {p.exitcode})"
        assert results.get(0) == "ok" and results.get(1) == "ok"
