"""lighthouse_amd — MI355X-native implementations of a Lighthouse-class
consensus client's two compute hot paths:

  1. batched BLS12-381 signature-set verification (`lighthouse_amd.bls`),
     a drop-in for the crypto/bls backend trait semantics;
  2. SHA256 SSZ merkleization (`lighthouse_amd.tree_hash`), a drop-in for
     the BeaconState::update_tree_hash_cache seam.

Compute runs in hand-written HIP kernels for gfx950 behind the C-ABI in
include/m3x_consensus.h. There is no CPU fallback."""

__version__ = "0.1.0"

from . import _native  # noqa: F401
