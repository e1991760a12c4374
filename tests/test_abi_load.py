"""CPU-side check: the C-ABI library loads and exports every symbol
include/m3x_consensus.h declares (no compute without a GPU)."""
import ctypes
import re
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent

EXPECTED = [
    "m3x_abi_version",
    "m3x_ctx_create",
    "m3x_ctx_destroy",
    "m3x_dev_alloc",
    "m3x_dev_free",
    "m3x_h2d",
    "m3x_d2h",
    "m3x_merkleize_validators",
    "m3x_merkleize_validators_dev",
    "m3x_merkleize_chunks",
    "m3x_merkleize_chunks_dev",
    "m3x_merkleize_batch",
    "m3x_finalize_root",
    "m3x_timing_enable",
    "m3x_kernel_ms",
    "m3x_validator_subtree_root_dev",
    "m3x_bls_pk_decompress",
    "m3x_bls_verify_sets",
    "m3x_bls_verify_sets_dev",
]


def test_header_declares_expected():
    hdr = (REPO / "include" / "m3x_consensus.h").read_text()
    declared = set(re.findall(r"\b(m3x_\w+)\s*\(", hdr))
    for sym in EXPECTED:
        assert sym in declared, sym


def test_library_exports_all():
    import lighthouse_amd._native as native

    lib = native.load()
    for sym in EXPECTED:
        assert hasattr(lib, sym), f"missing export {sym}"
    assert lib.m3x_abi_version() >= 1
