import ctypes, hashlib, sys, time
sys.path.insert(0, ".")
from lighthouse_amd import _native

o = ctypes.CDLL("oracle/liboracle.so")
n = 4096
sks = ctypes.create_string_buffer(32 * n)
pks = ctypes.create_string_buffer(96 * n)
o.m3x_oracle_bls_keypool(ctypes.c_uint64(n), sks, pks)
msgs = b"".join(hashlib.sha256(b"dbg%d" % i).digest() for i in range(n))
sigs = ctypes.create_string_buffer(96 * n)
assert o.m3x_oracle_bls_sign_batch(ctypes.c_uint64(n), sks.raw, msgs, sigs) == 0
offs = (ctypes.c_uint32 * (n + 1))(*range(n + 1))
rnds = (ctypes.c_uint64 * n)(*[(i * 0x9E37 + 1) | 1 for i in range(n)])
ctx = _native.Ctx(0)
print("ctx ok", flush=True)
t0 = time.time()
rc = ctx._lib.m3x_bls_verify_sets(ctx.handle, msgs, sigs.raw, pks.raw, offs, rnds, n)
print("verify(4096) rc=", rc, f"{time.time()-t0:.2f}s", flush=True)
