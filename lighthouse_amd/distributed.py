"""Multi-GPU sharding for the two hot paths.

Both paths shard embarrassingly across the ranks of one node:

 - BLS: signature sets are partitioned k-weighted round-robin; each rank
   runs its own complete batch check (including its own final
   exponentiation — the product of sub-batch checks with independent r_i is
   equivalent to the single batch check, the same math as the reference
   verifying chain segments separately, block_verification.rs:591-655);
   verdicts combine with ONE 4-byte all_reduce(MIN).
 - Merkleize: each rank takes a contiguous aligned leaf range of the
   registry, computes its depth-(20-log2 N) subtree root on its GPU;
   rank 0 gathers the N x 32B roots, merkleizes them and carries the zero
   cap to depth 40 + mix_in_length. Collectives are latency-bound: a
   gather + tiny all_reduce over xGMI, never a ring for 32 bytes.

The sharding/combine logic is backend-generic so the world_size-2 gloo CPU
tests can drive it with a stub compute backend (the reference's fake_crypto
test strategy, crypto/bls/src/impls/fake_crypto.rs)."""
import torch
import torch.distributed as dist


def partition_sets(costs, world_size):
    """k-weighted greedy round-robin: returns list of index lists per rank.
    Deterministic across ranks (same input -> same partition)."""
    order = sorted(range(len(costs)), key=lambda i: -costs[i])
    loads = [0] * world_size
    parts = [[] for _ in range(world_size)]
    for i in order:
        r = loads.index(min(loads))
        parts[r].append(i)
        loads[r] += costs[i]
    for p in parts:
        p.sort()
    return parts


def shard_range(n_leaves, rank, world_size):
    """Contiguous aligned leaf range for `rank`; n_leaves and world_size
    must make equal power-of-two shards for subtree alignment."""
    assert n_leaves % world_size == 0
    per = n_leaves // world_size
    assert per & (per - 1) == 0, "shard size must be a power of two"
    return rank * per, per


def verify_sets_sharded(sets_costs, run_rank_subset, group=None):
    """Partition sets by cost, run this rank's subset via
    run_rank_subset(indices)->bool, AND-combine verdicts with one 4-byte
    all_reduce(MIN). A rank with no sets contributes True (the global
    emptiness rule is enforced by the caller before sharding)."""
    rank = dist.get_rank(group)
    world = dist.get_world_size(group)
    parts = partition_sets(sets_costs, world)
    ok = bool(run_rank_subset(parts[rank])) if parts[rank] else True
    t = torch.tensor([1 if ok else 0], dtype=torch.int32)
    if dist.get_backend(group) == "nccl":
        t = t.cuda()
    dist.all_reduce(t, op=dist.ReduceOp.MIN, group=group)
    return bool(t.item() == 1)


def registry_root_sharded(subtree_fn, hash2_fn, finalize_fn, n_leaves,
                           n_validators, group=None):
    """subtree_fn(start, count, depth)->32B root for this rank's leaf
    range; hash2_fn(l, r)->32B two-to-one hash for the tiny top combine
    (world_size <= 8 -> at most 7 calls; GPU path serves it via
    m3x_merkleize_chunks); finalize_fn(node, from_level, to_depth, mix_len)
    carries the zero cap + mix_in_length on rank 0. Returns the full
    List[Validator, 2^40] root on rank 0, None elsewhere."""
    rank = dist.get_rank(group)
    world = dist.get_world_size(group)
    start, per = shard_range(n_leaves, rank, world)
    sub_depth = per.bit_length() - 1
    root = subtree_fn(start, per, sub_depth)
    t = torch.frombuffer(bytearray(root), dtype=torch.uint8).clone()
    if dist.get_backend(group) == "nccl":
        t = t.cuda()
    # all_gather: NCCL has no gather primitive (8x32B, latency-bound)
    gathered = [torch.zeros_like(t) for _ in range(world)]
    dist.all_gather(gathered, t, group=group)
    if rank != 0:
        return None
    nodes = [bytes(g.cpu().numpy().tobytes()) for g in gathered]
    level = sub_depth
    while len(nodes) > 1:
        nodes = [hash2_fn(nodes[i], nodes[i + 1]) for i in range(0, len(nodes), 2)]
        level += 1
    return finalize_fn(nodes[0], level, 40, n_validators)
