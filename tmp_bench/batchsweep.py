"""Batch-size scaling curve for the BLS verify pipeline (evidence for the
occupancy/co-residency analysis in DESIGN.md). All-k=1 valid sets from
the 4096-key pool, repeated; one warm + one timed run per size."""
import ctypes, hashlib, sys, time
sys.path.insert(0, ".")
import numpy as np
from lighthouse_amd import _native

o = ctypes.CDLL("oracle/liboracle.so")
POOL = 4096
sks = ctypes.create_string_buffer(32 * POOL)
pks = ctypes.create_string_buffer(96 * POOL)
o.m3x_oracle_bls_keypool(ctypes.c_uint64(POOL), sks, pks)
msgs = b"".join(hashlib.sha256(b"sw%d" % i).digest() for i in range(POOL))
sigs = ctypes.create_string_buffer(96 * POOL)
assert o.m3x_oracle_bls_sign_batch(ctypes.c_uint64(POOL), sks.raw, msgs, sigs) == 0
ctx = _native.Ctx(0)
lib = ctx._lib
for logn in [11, 13, 14, 16, 17, 18, 19, 20]:
    n = 1 << logn
    reps = max(1, n // POOL)
    m = (msgs * reps)[: 32 * n]
    sg = (sigs.raw * reps)[: 96 * n]
    pk = (pks.raw * reps)[: 96 * n]
    dev = {
        "m": ctx.upload(m),
        "s": ctx.upload(sg),
        "p": ctx.upload(pk),
        "o": ctx.upload(np.arange(n + 1, dtype=np.uint32).tobytes()),
        "r": ctx.upload(np.asarray(
            [((i * 0x9E3779B97F4A7C15 + 5) | 1) & (2**64 - 1) for i in range(n)],
            dtype=np.uint64).tobytes()),
    }
    v = lib.m3x_bls_verify_sets_dev(ctx.handle, dev["m"], dev["s"], dev["p"], dev["o"], dev["r"], n)
    assert v == 1, (n, v)
    t0 = time.time()
    v = lib.m3x_bls_verify_sets_dev(ctx.handle, dev["m"], dev["s"], dev["p"], dev["o"], dev["r"], n)
    t = time.time() - t0
    assert v == 1
    print(f"n=2^{logn} ({n:>8}): {t*1e3:8.1f} ms = {n/t:9.0f} sets/s", flush=True)
    for pbuf in dev.values():
        ctx.free(pbuf)
