"""Host mirror of the reference's hash_tree_root seam (hot path #2).

Mirrors BeaconState::update_tree_hash_cache /
update_validators_tree_hash_cache
(/root/reference/consensus/types/src/beacon_state.rs:2031-2046) and the
tree_hash packing rules over the m3x C-ABI. GPU-only — no CPU fallback."""
import ctypes

from . import _native


def _ceil_log2(x: int) -> int:
    d = 0
    while (1 << d) < x:
        d += 1
    return d


def validator_registry_root(ssz: bytes, n: int, ctx=None) -> bytes:
    """hash_tree_root of List[Validator, 2^40] (eth_spec.rs:404) from packed
    121-byte SSZ records."""
    ctx = ctx or _native.default_ctx()
    out = ctypes.create_string_buffer(32)
    rc = ctx._lib.m3x_merkleize_validators(ctx.handle, ssz, n, out)
    if rc != 0:
        raise RuntimeError(f"m3x_merkleize_validators rc={rc}")
    return out.raw


def merkleize_chunks(chunks: bytes, n_chunks: int, depth: int,
                     mix_len: int = -1, ctx=None) -> bytes:
    ctx = ctx or _native.default_ctx()
    out = ctypes.create_string_buffer(32)
    rc = ctx._lib.m3x_merkleize_chunks(ctx.handle, chunks, n_chunks, depth,
                                       mix_len, out)
    if rc != 0:
        raise RuntimeError(f"m3x_merkleize_chunks rc={rc}")
    return out.raw


def basic_list_root(data: bytes, n_elems: int, elem_size: int,
                    limit_elems: int, ctx=None) -> bytes:
    """List of basic elements (u64 balances, u8 participation, ...):
    LE-packed 32B chunks, depth from the limit, mix_in_length(n)."""
    limit_chunks = max((limit_elems * elem_size + 31) // 32, 1)
    n_chunks = (len(data) + 31) // 32
    padded = data + b"\x00" * (n_chunks * 32 - len(data))
    return merkleize_chunks(padded, n_chunks, _ceil_log2(limit_chunks),
                            n_elems, ctx=ctx)


def basic_vector_root(data: bytes, n_elems: int, elem_size: int, ctx=None) -> bytes:
    limit_chunks = max((n_elems * elem_size + 31) // 32, 1)
    n_chunks = (len(data) + 31) // 32
    padded = data + b"\x00" * (n_chunks * 32 - len(data))
    return merkleize_chunks(padded, n_chunks, _ceil_log2(limit_chunks), -1,
                            ctx=ctx)


def merkleize_chunks_dev(dev_ptr, n_chunks: int, depth: int, mix_len: int = -1,
                         ctx=None) -> bytes:
    """device-resident chunk buffer (zero-padded tail chunk)."""
    import ctypes as ct

    ctx = ctx or _native.default_ctx()
    out = ct.create_string_buffer(32)
    rc = ctx._lib.m3x_merkleize_chunks_dev(ctx.handle, dev_ptr, n_chunks,
                                           depth, mix_len, out)
    if rc != 0:
        raise RuntimeError(f"m3x_merkleize_chunks_dev rc={rc}")
    return out.raw


def basic_list_root_dev(dev_ptr, n_elems: int, elem_size: int,
                        limit_elems: int, ctx=None) -> bytes:
    limit_chunks = max((limit_elems * elem_size + 31) // 32, 1)
    n_chunks = (n_elems * elem_size + 31) // 32
    return merkleize_chunks_dev(dev_ptr, n_chunks, _ceil_log2(limit_chunks),
                                n_elems, ctx=ctx)


def basic_vector_root_dev(dev_ptr, n_elems: int, elem_size: int, ctx=None) -> bytes:
    limit_chunks = max((n_elems * elem_size + 31) // 32, 1)
    n_chunks = (n_elems * elem_size + 31) // 32
    return merkleize_chunks_dev(dev_ptr, n_chunks, _ceil_log2(limit_chunks),
                                -1, ctx=ctx)


def root_vector_root_dev(dev_ptr, n: int, ctx=None) -> bytes:
    return merkleize_chunks_dev(dev_ptr, n, _ceil_log2(max(n, 1)), -1, ctx=ctx)


def validator_registry_root_dev(dev_ptr, n: int, ctx=None) -> bytes:
    import ctypes as ct

    ctx = ctx or _native.default_ctx()
    out = ct.create_string_buffer(32)
    rc = ctx._lib.m3x_merkleize_validators_dev(ctx.handle, dev_ptr, n, out)
    if rc != 0:
        raise RuntimeError(f"m3x_merkleize_validators_dev rc={rc}")
    return out.raw


def root_vector_root(roots: bytes, n: int, ctx=None) -> bytes:
    return merkleize_chunks(roots, n, _ceil_log2(max(n, 1)), -1, ctx=ctx)


def root_list_root(roots: bytes, n: int, limit: int, ctx=None) -> bytes:
    return merkleize_chunks(roots, n, _ceil_log2(max(limit, 1)), n, ctx=ctx)


def merkleize_batch(chunk_groups, ctx=None):
    """One GPU launch for many tiny containers: chunk_groups is a list of
    bytes (each a concatenation of 32B chunks, <=32 chunks per group);
    returns the list of hash_tree_roots (depth = ceil_log2(count))."""
    import ctypes as ct

    ctx = ctx or _native.default_ctx()
    offsets = [0]
    blob = bytearray()
    for g in chunk_groups:
        assert len(g) % 32 == 0 and len(g) <= 32 * 32
        blob += g
        offsets.append(offsets[-1] + len(g) // 32)
    n = len(chunk_groups)
    off = (ct.c_uint32 * (n + 1))(*offsets)
    out = ct.create_string_buffer(32 * n)
    rc = ctx._lib.m3x_merkleize_batch(ctx.handle, bytes(blob), off, n, out)
    if rc != 0:
        raise RuntimeError(f"m3x_merkleize_batch rc={rc}")
    return [out.raw[32 * i : 32 * (i + 1)] for i in range(n)]


class RegistryCache:
    """Incremental validator-registry merkleize (SURVEY §8f.3 — the
    milhouse cached-rehash seam, beacon_state.rs:1990-2021): the full tree
    lives in HBM; updates rehash only dirty leaves + root paths."""

    def __init__(self, ssz: bytes, n: int, ctx=None):
        import ctypes as ct

        self._ctx = ctx or _native.default_ctx()
        self._h = ct.c_void_p()
        rc = self._ctx._lib.m3x_registry_cache_create(
            self._ctx.handle, ssz, n, ct.byref(self._h)
        )
        if rc != 0:
            raise RuntimeError(f"m3x_registry_cache_create rc={rc}")

    def close(self):
        if self._h:
            self._ctx._lib.m3x_registry_cache_destroy(self._h)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    def root(self) -> bytes:
        import ctypes as ct

        out = ct.create_string_buffer(32)
        rc = self._ctx._lib.m3x_registry_cache_root(
            self._ctx.handle, self._h, out
        )
        if rc != 0:
            raise RuntimeError(f"m3x_registry_cache_root rc={rc}")
        return out.raw

    def update(self, indices, records: bytes) -> bytes:
        """apply updated/appended 121B records at `indices`; returns the
        new List[Validator, 2^40] root."""
        import ctypes as ct

        m = len(indices)
        assert len(records) == 121 * m
        idx = (ct.c_uint64 * max(m, 1))(*indices)
        out = ct.create_string_buffer(32)
        rc = self._ctx._lib.m3x_registry_cache_update(
            self._ctx.handle, self._h, idx, records, m, out
        )
        if rc != 0:
            raise RuntimeError(f"m3x_registry_cache_update rc={rc}")
        return out.raw
