"""Host mirror of the reference's swap-or-not shuffle seam
(consensus/swap_or_not_shuffle/src/shuffle_list.rs — committee shuffling,
SURVEY §8f.1). GPU-only."""
import ctypes

import numpy as np

from . import _native

SHUFFLE_ROUND_COUNT = 90  # ChainSpec mainnet, chain_spec.rs:632


def shuffle_list(indices, rounds: int, seed: bytes, forwards: bool,
                 ctx=None):
    """Returns the shuffled list (np.uint32 array), or None under the
    reference's None conditions (empty, > 2^24, rounds == 0)."""
    arr = np.ascontiguousarray(np.asarray(indices, dtype=np.uint32))
    n = arr.size
    if n == 0 or n > (1 << 24) or rounds == 0:
        return None
    assert len(seed) == 32
    ctx = ctx or _native.default_ctx()
    buf = arr.copy()
    rc = ctx._lib.m3x_shuffle_list(
        ctx.handle,
        buf.ctypes.data_as(ctypes.c_void_p),
        n,
        rounds,
        seed,
        1 if forwards else 0,
    )
    if rc != 0:
        raise RuntimeError(f"m3x_shuffle_list rc={rc}")
    return buf
