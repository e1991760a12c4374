#!/bin/bash
# SHA CPU-baseline scaling probe (256-core box): threads x binding sweep
cd /root/repo
for T in 32 64 128 256; do
  for BIND in "" "spread"; do
    env OMP_NUM_THREADS=$T ${BIND:+OMP_PROC_BIND=$BIND OMP_PLACES=cores} python3 - <<'PY'
import ctypes, time, os
o = ctypes.CDLL("oracle/liboracle.so")
n = 1 << 20
ssz = bytes(121) * n
out = ctypes.create_string_buffer(32)
best = 9e9
for _ in range(3):
    t0 = time.time(); o.m3x_oracle_validator_registry_root(ssz, ctypes.c_uint64(n), out); best = min(best, time.time()-t0)
nodes = 8*n + (n-1) + 20 + 1
print(f"T={os.environ.get('OMP_NUM_THREADS')} bind={os.environ.get('OMP_PROC_BIND','-')}: {best*1e3:.0f} ms = {nodes/best/1e6:.0f}M node-hashes/s")
PY
  done
done
