"""Small-batch miller probe: C4-style block import (131 sets) and a
threshold sweep comparing per-lane k_bls_miller vs wave-per-set
k_bls_miller_small (M3X_SMALL_MILLER env toggles the dispatch)."""
import ctypes
import os
import sys
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))
import torch  # noqa: E402  (MUST load before the m3x .so: binding our
# ROCm-7.2-linked runtime first breaks torch's own HIP enumeration)
import bench  # noqa: E402
from bench import build_bls_workload, upload_bls, N_SETS, N_AGG  # noqa

import numpy as np  # noqa: E402
from lighthouse_amd import _native  # noqa: E402

ctx = _native.Ctx(0)
lib = ctx._lib
oracle = ctypes.CDLL(str(REPO / "oracle" / "liboracle.so"))
w = build_bls_workload(oracle)
print("workload ready", flush=True)


def subset_dev(idx):
    m = b"".join(w["msgs"][32 * i : 32 * (i + 1)] for i in idx)
    s = b"".join(w["sigs"][96 * i : 96 * (i + 1)] for i in idx)
    p = b""
    o = [0]
    for i in idx:
        a, b = w["offsets"][i], w["offsets"][i + 1]
        p += w["pks"][96 * a : 96 * b]
        o.append(o[-1] + (b - a))
    return {
        "msgs": ctx.upload(m),
        "sigs": ctx.upload(s),
        "pks": ctx.upload(p),
        "offsets": ctx.upload(np.asarray(o, dtype=np.uint32).tobytes()),
        "rands": ctx.upload(
            np.asarray([w["rands"][i] for i in idx], dtype=np.uint64).tobytes()
        ),
        "n": len(idx),
    }


def run(d, reps=3):
    import torch

    v = lib.m3x_bls_verify_sets_dev(
        ctx.handle, d["msgs"], d["sigs"], d["pks"], d["offsets"], d["rands"],
        d["n"])
    assert v == 1, f"verdict {v}"
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(reps):
        v = lib.m3x_bls_verify_sets_dev(
            ctx.handle, d["msgs"], d["sigs"], d["pks"], d["offsets"],
            d["rands"], d["n"])
    torch.cuda.synchronize()
    assert v == 1
    return (time.time() - t0) / reps * 1e3


# C4 shape: 128 aggregates k=512 + 3 singles
idx4 = list(range(N_SETS - N_AGG, N_SETS - N_AGG + 128)) + [0, 1, 2]
d4 = subset_dev(idx4)
for lbl, thr in (("per-lane", "0"), ("wave-per-set", "100000")):
    os.environ["M3X_SMALL_MILLER"] = thr
    print(f"C4 131 sets  {lbl:13s}: {run(d4):8.2f} ms", flush=True)

# per-kernel split of the wave-per-set C4 run (HIP-event timing; the
# timed runs above keep timing OFF to preserve stream overlap)
os.environ["M3X_SMALL_MILLER"] = "100000"
ctx.timing_enable(True)
run(d4, reps=3)
print("C4 kernel split:",
      {k: round(v[0] / max(v[1], 1), 2) for k, v in ctx.kernel_times().items()},
      flush=True)
ctx.timing_enable(False)

# threshold sweep on k=1 sets
for n in (512, 2048, 8192, 16384):
    idx = list(range(n))
    d = subset_dev(idx)
    for lbl, thr in (("per-lane", "0"), ("wave-per-set", "100000")):
        os.environ["M3X_SMALL_MILLER"] = thr
        print(f"n={n:6d} k=1  {lbl:13s}: {run(d):8.2f} ms", flush=True)
