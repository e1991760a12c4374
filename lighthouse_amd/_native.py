"""ctypes binding to libm3x_consensus.so (the C-ABI in include/m3x_consensus.h).

Product path only — no CPU fallback: if a GPU is present and the library is
missing or fails to load, every call raises. The oracle is never imported
here (see DESIGN.md §Parity strategy)."""
import ctypes
import os
from pathlib import Path

_PKG = Path(__file__).resolve().parent
_LIB_PATH = _PKG / "libm3x_consensus.so"

_lib = None
_load_error = None


def _bind(lib):
    lib.m3x_abi_version.restype = ctypes.c_int32
    lib.m3x_ctx_create.argtypes = [ctypes.POINTER(ctypes.c_void_p), ctypes.c_int32]
    lib.m3x_ctx_create.restype = ctypes.c_int32
    lib.m3x_ctx_destroy.argtypes = [ctypes.c_void_p]
    lib.m3x_dev_alloc.argtypes = [
        ctypes.c_void_p,
        ctypes.c_uint64,
        ctypes.POINTER(ctypes.c_void_p),
    ]
    lib.m3x_dev_alloc.restype = ctypes.c_int32
    lib.m3x_dev_free.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
    lib.m3x_dev_free.restype = ctypes.c_int32
    lib.m3x_h2d.argtypes = [
        ctypes.c_void_p,
        ctypes.c_void_p,
        ctypes.c_char_p,
        ctypes.c_uint64,
    ]
    lib.m3x_h2d.restype = ctypes.c_int32
    lib.m3x_d2h.argtypes = [
        ctypes.c_void_p,
        ctypes.c_void_p,
        ctypes.c_void_p,
        ctypes.c_uint64,
    ]
    lib.m3x_d2h.restype = ctypes.c_int32
    for name, argtypes in [
        (
            "m3x_merkleize_validators",
            [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_uint64, ctypes.c_char_p],
        ),
        (
            "m3x_merkleize_validators_dev",
            [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_uint64, ctypes.c_char_p],
        ),
        (
            "m3x_merkleize_chunks",
            [
                ctypes.c_void_p,
                ctypes.c_char_p,
                ctypes.c_uint64,
                ctypes.c_uint32,
                ctypes.c_int64,
                ctypes.c_char_p,
            ],
        ),
        (
            "m3x_merkleize_chunks_dev",
            [
                ctypes.c_void_p,
                ctypes.c_void_p,
                ctypes.c_uint64,
                ctypes.c_uint32,
                ctypes.c_int64,
                ctypes.c_char_p,
            ],
        ),
        (
            "m3x_validator_subtree_root_dev",
            [
                ctypes.c_void_p,
                ctypes.c_void_p,
                ctypes.c_uint64,
                ctypes.c_uint32,
                ctypes.c_char_p,
            ],
        ),
        ("m3x_timing_enable", [ctypes.c_void_p, ctypes.c_int32]),
        (
            "m3x_bls_sig_aggregate",
            [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_uint64,
             ctypes.c_char_p],
        ),
        (
            "m3x_registry_cache_create",
            [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_uint64,
             ctypes.POINTER(ctypes.c_void_p)],
        ),
        ("m3x_registry_cache_destroy", [ctypes.c_void_p]),
        (
            "m3x_registry_cache_root",
            [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_char_p],
        ),
        (
            "m3x_registry_cache_update",
            [ctypes.c_void_p, ctypes.c_void_p,
             ctypes.POINTER(ctypes.c_uint64), ctypes.c_char_p,
             ctypes.c_uint64, ctypes.c_char_p],
        ),
        (
            "m3x_shuffle_list",
            [
                ctypes.c_void_p,
                ctypes.c_void_p,
                ctypes.c_uint64,
                ctypes.c_uint32,
                ctypes.c_char_p,
                ctypes.c_int32,
            ],
        ),
        (
            "m3x_shuffle_list_dev",
            [
                ctypes.c_void_p,
                ctypes.c_void_p,
                ctypes.c_uint64,
                ctypes.c_uint32,
                ctypes.c_char_p,
                ctypes.c_int32,
            ],
        ),
        (
            "m3x_merkleize_batch",
            [
                ctypes.c_void_p,
                ctypes.c_char_p,
                ctypes.POINTER(ctypes.c_uint32),
                ctypes.c_uint64,
                ctypes.c_char_p,
            ],
        ),
        (
            "m3x_kernel_ms",
            [
                ctypes.c_void_p,
                ctypes.c_int32,
                ctypes.POINTER(ctypes.c_double),
                ctypes.POINTER(ctypes.c_uint64),
            ],
        ),
        (
            "m3x_finalize_root",
            [
                ctypes.c_void_p,
                ctypes.c_char_p,
                ctypes.c_uint32,
                ctypes.c_uint32,
                ctypes.c_int64,
                ctypes.c_char_p,
            ],
        ),
        (
            "m3x_bls_expand_test",
            [
                ctypes.c_void_p,
                ctypes.c_char_p,
                ctypes.c_uint32,
                ctypes.c_char_p,
                ctypes.c_uint32,
                ctypes.c_uint32,
                ctypes.c_char_p,
            ],
        ),
        (
            "m3x_bls_h2c_test",
            [
                ctypes.c_void_p,
                ctypes.c_char_p,
                ctypes.c_uint32,
                ctypes.c_char_p,
                ctypes.c_uint32,
                ctypes.c_char_p,
                ctypes.c_char_p,
            ],
        ),
        (
            "m3x_bls_pk_decompress",
            [
                ctypes.c_void_p,
                ctypes.c_char_p,
                ctypes.c_uint64,
                ctypes.c_char_p,
                ctypes.POINTER(ctypes.c_int32),
            ],
        ),
        (
            "m3x_bls_verify_sets",
            [
                ctypes.c_void_p,
                ctypes.c_char_p,
                ctypes.c_char_p,
                ctypes.c_char_p,
                ctypes.POINTER(ctypes.c_uint32),
                ctypes.POINTER(ctypes.c_uint64),
                ctypes.c_uint64,
            ],
        ),
        (
            "m3x_bls_verify_sets_dev",
            [
                ctypes.c_void_p,
                ctypes.c_void_p,
                ctypes.c_void_p,
                ctypes.c_void_p,
                ctypes.c_void_p,
                ctypes.c_void_p,
                ctypes.c_uint64,
            ],
        ),
    ]:
        try:
            fn = getattr(lib, name)
        except AttributeError:
            continue
        fn.argtypes = argtypes
        fn.restype = (
            None if name == "m3x_registry_cache_destroy" else ctypes.c_int32
        )
    return lib


def load():
    """Load (and cache) the native library; raises RuntimeError if absent."""
    global _lib, _load_error
    if _lib is not None:
        return _lib
    if not _LIB_PATH.exists():
        raise RuntimeError(
            f"m3x_consensus native library missing: {_LIB_PATH}. "
            "Build it with `python -c 'import __graft_entry__; "
            "__graft_entry__.build()'` — there is no CPU fallback."
        )
    try:
        _lib = _bind(ctypes.CDLL(str(_LIB_PATH)))
    except OSError as e:
        _load_error = e
        raise RuntimeError(f"failed to load {_LIB_PATH}: {e}") from e
    return _lib


class Ctx:
    """One GPU context (device + stream + scratch)."""

    def __init__(self, device: int = 0):
        self._lib = load()
        self._h = ctypes.c_void_p()
        rc = self._lib.m3x_ctx_create(ctypes.byref(self._h), device)
        if rc != 0:
            raise RuntimeError(
                f"m3x_ctx_create failed (rc={rc}) — is an MI355X visible?"
            )

    def close(self):
        if self._h:
            self._lib.m3x_ctx_destroy(self._h)
            self._h = ctypes.c_void_p()

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    @property
    def handle(self):
        return self._h

    # ---- device buffers ----
    def alloc(self, nbytes: int) -> ctypes.c_void_p:
        p = ctypes.c_void_p()
        rc = self._lib.m3x_dev_alloc(self._h, nbytes, ctypes.byref(p))
        if rc != 0:
            raise RuntimeError(f"m3x_dev_alloc({nbytes}) rc={rc}")
        return p

    def free(self, p):
        self._lib.m3x_dev_free(self._h, p)

    def h2d(self, dev, host_bytes: bytes):
        rc = self._lib.m3x_h2d(self._h, dev, host_bytes, len(host_bytes))
        if rc != 0:
            raise RuntimeError(f"m3x_h2d rc={rc}")

    def upload(self, host_bytes: bytes):
        p = self.alloc(max(len(host_bytes), 4))
        self.h2d(p, host_bytes)
        return p

    # ---- kernel timing (bench/roofline) ----
    def timing_enable(self, on: bool = True):
        self._lib.m3x_timing_enable(self._h, 1 if on else 0)

    KERNEL_NAMES = [
        "leaves",
        "reduce",
        "finalize",
        "bls_prepare",
        "bls_h2c",
        "bls_miller",
        "bls_reduce",
        "bls_finish",
        "bls_agg",
    ]

    def kernel_times(self):
        out = {}
        for i, name in enumerate(self.KERNEL_NAMES):
            ms = ctypes.c_double()
            n = ctypes.c_uint64()
            if self._lib.m3x_kernel_ms(self._h, i, ctypes.byref(ms), ctypes.byref(n)) == 0:
                out[name] = (ms.value, n.value)
        return out

    def finalize_root(self, node: bytes, from_level: int, to_depth: int,
                      mix_len: int = -1) -> bytes:
        out = ctypes.create_string_buffer(32)
        rc = self._lib.m3x_finalize_root(self._h, node, from_level, to_depth,
                                         mix_len, out)
        if rc != 0:
            raise RuntimeError(f"m3x_finalize_root rc={rc}")
        return out.raw


_tls = None


def default_ctx() -> Ctx:
    """Per-THREAD context (SURVEY §8b threading contract: the backend is
    called concurrently from N beacon-processor workers, so it must be
    re-entrant and support concurrent batches). Each thread gets its own
    context — own HIP streams + scratch — so concurrent
    verify_signature_sets calls from different threads run on different
    streams without serializing on one context's lock."""
    global _tls
    if _tls is None:
        import threading

        _tls = threading.local()
    ctx = getattr(_tls, "ctx", None)
    if ctx is None:
        ctx = Ctx(int(os.environ.get("M3X_DEVICE", "0")))
        _tls.ctx = ctx
    return ctx
