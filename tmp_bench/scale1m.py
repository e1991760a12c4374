"""1M-set all-k=1 single-GPU probe (BASELINE config-5 shape) on the
final build. Signing uses the OpenMP oracle (untimed)."""
import ctypes
import hashlib
import sys
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))
import torch  # noqa: E402  (before the m3x .so)
import numpy as np  # noqa: E402
from lighthouse_amd import _native  # noqa: E402

N = 1 << 20
POOL = 4096
ctx = _native.Ctx(0)
lib = ctx._lib
oracle = ctypes.CDLL(str(REPO / "oracle" / "liboracle.so"))

sks = ctypes.create_string_buffer(32 * POOL)
pks = ctypes.create_string_buffer(96 * POOL)
oracle.m3x_oracle_bls_keypool(ctypes.c_uint64(POOL), sks, pks)
t0 = time.time()
msgs = bytearray()
sign_sks = bytearray()
pkb = bytearray()
for i in range(N):
    msgs += hashlib.sha256(b"c5msg%d" % i).digest()
    j = i % POOL
    sign_sks += sks.raw[32 * j : 32 * (j + 1)]
    pkb += pks.raw[96 * j : 96 * (j + 1)]
sigs = ctypes.create_string_buffer(96 * N)
rc = oracle.m3x_oracle_bls_sign_batch(
    ctypes.c_uint64(N), bytes(sign_sks), bytes(msgs), sigs)
assert rc == 0
print(f"signed {N} in {time.time()-t0:.1f}s", flush=True)

d = {
    "msgs": ctx.upload(bytes(msgs)),
    "sigs": ctx.upload(sigs.raw),
    "pks": ctx.upload(bytes(pkb)),
    "offsets": ctx.upload(np.arange(N + 1, dtype=np.uint32).tobytes()),
    "rands": ctx.upload(
        np.asarray(
            [((i * 0x9E3779B97F4A7C15 + 0xC5) | 1) & 0xFFFFFFFFFFFFFFFF
             for i in range(N)],
            dtype=np.uint64,
        ).tobytes()
    ),
}
for rep in range(3):
    torch.cuda.synchronize()
    tb = time.time()
    v = lib.m3x_bls_verify_sets_dev(
        ctx.handle, d["msgs"], d["sigs"], d["pks"], d["offsets"], d["rands"], N)
    torch.cuda.synchronize()
    dt = time.time() - tb
    assert v == 1, v
    print(f"rep{rep}: {N/dt:,.0f} sets/s ({dt*1e3:.1f} ms/batch)", flush=True)
