"""ctypes loader for libuidalgo.so (the product C-ABI, include/uidalgo.h).

Fails loudly if the extension is missing — the product path has no CPU
fallback by design (the Go caller keeps its own fallback).
"""
import ctypes as C
import os

_DIR = os.path.dirname(os.path.abspath(__file__))
# UA_LIB_PATH overrides the library for perf-bisection builds (tools/ubench.py)
_LIB_PATH = os.environ.get("UA_LIB_PATH", os.path.join(_DIR, "libuidalgo.so"))

UA_OK = 0

_u64 = C.c_uint64
_u64p = C.POINTER(C.c_uint64)
_u32p = C.POINTER(C.c_uint32)
_u8p = C.POINTER(C.c_uint8)
_i64p = C.POINTER(C.c_int64)
_voidpp = C.POINTER(C.c_void_p)


class UaDPair(C.Structure):
    _fields_ = [
        ("u", C.c_void_p),
        ("n", _u64),
        ("v", C.c_void_p),
        ("m", _u64),
        ("out", C.c_void_p),
    ]


class UaDPack(C.Structure):
    _fields_ = [
        ("block_size", C.c_uint32),
        ("n_blocks", _u64),
        ("bases", C.c_void_p),
        ("num_uids", C.c_void_p),
        ("delta_offs", C.c_void_p),
        ("deltas", C.c_void_p),
        ("total_uids", _u64),
    ]


class UaDFilter(C.Structure):
    _fields_ = [
        ("u", C.c_void_p),
        ("n", _u64),
        ("mask", C.c_void_p),
        ("out", C.c_void_p),
    ]


class UaDSeg(C.Structure):
    _fields_ = [
        ("data", C.c_void_p),
        ("n", _u64),
        ("tmp", C.c_void_p),
    ]


class UaPTask(C.Structure):
    _fields_ = [
        ("v", C.c_void_p),
        ("m", _u64),
        ("out", C.c_void_p),
        ("after_uid", _u64),
    ]


class UaBlock(C.Structure):
    _fields_ = [
        ("base", _u64),
        ("num_uids", C.c_uint32),
        ("deltas_len", C.c_uint32),
        ("deltas", _u8p),
    ]


class UaPack(C.Structure):
    _fields_ = [
        ("block_size", C.c_uint32),
        ("n_blocks", _u64),
        ("blocks", C.POINTER(UaBlock)),
    ]


_lib = None


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(_LIB_PATH):
            raise RuntimeError(
                f"libuidalgo.so not found at {_LIB_PATH}: the HIP extension is "
                "mandatory — run __graft_entry__.build() (hipcc "
                "--offload-arch=gfx950).")
        L = C.CDLL(_LIB_PATH)
        L.ua_strerror.restype = C.c_char_p
        L.ua_strerror.argtypes = [C.c_int]
        L.ua_version.restype = C.c_int
        L.ua_ctx_create.argtypes = [_voidpp, C.c_int]
        L.ua_ctx_destroy.argtypes = [C.c_void_p]
        L.ua_dev_alloc.argtypes = [C.c_void_p, _u64, _voidpp]
        L.ua_dev_free.argtypes = [C.c_void_p, C.c_void_p]
        L.ua_h2d.argtypes = [C.c_void_p, C.c_void_p, C.c_void_p, _u64]
        L.ua_d2h.argtypes = [C.c_void_p, C.c_void_p, C.c_void_p, _u64]
        L.ua_sync.argtypes = [C.c_void_p]
        L.ua_stats_reset.argtypes = [C.c_void_p]
        L.ua_stats_get.argtypes = [C.c_void_p, _u64p, C.POINTER(C.c_double), _u64p]
        L.ua_intersect_batch_dev.argtypes = [C.c_void_p, C.POINTER(UaDPair), C.c_int, _u64p]
        L.ua_batch_create.argtypes = [C.c_void_p, C.POINTER(UaDPair), C.c_int, _voidpp]
        L.ua_batch_run.argtypes = [C.c_void_p, C.c_void_p, C.c_int, _u64p]
        L.ua_batch_run_n.argtypes = [C.c_void_p, C.c_void_p, C.c_int, C.c_int, _u64p]
        L.ua_batch_destroy.argtypes = [C.c_void_p, C.c_void_p]
        L.ua_batch_destroy.restype = None
        L.ua_merge_batch_dev.argtypes = [C.c_void_p, C.POINTER(UaDPair), C.c_int, _u64p]
        L.ua_difference_batch_dev.argtypes = [C.c_void_p, C.POINTER(UaDPair), C.c_int, _u64p]
        L.ua_merge_all_batch_dev.argtypes = [C.c_void_p, C.POINTER(UaDPair), C.c_int, _u64p]
        L.ua_sort_segments_dev.argtypes = [C.c_void_p, C.POINTER(UaDSeg), C.c_int]
        L.ua_index_of_batch_dev.argtypes = [C.c_void_p, C.c_void_p, _u64, C.c_void_p, _u64,
                                            C.c_void_p]
        L.ua_apply_filter_batch_dev.argtypes = [C.c_void_p, C.POINTER(UaDFilter),
                                                C.c_int, _u64p]
        L.ua_apply_filter.argtypes = [C.c_void_p, _u64p, _u64, _u8p, _u64p]
        L.ua_intersect_k_dev.argtypes = [C.c_void_p, _voidpp, _u64p, C.c_int, C.c_void_p, _u64p]
        L.ua_merge_k_dev.argtypes = [C.c_void_p, _voidpp, _u64p, C.c_int, C.c_void_p, _u64p]
        L.ua_intersect_packed_dev.argtypes = [C.c_void_p, C.POINTER(UaDPack), _u64, C.c_void_p,
                                              _u64, C.c_void_p, _u64p]
        L.ua_intersect_packed_batch_dev.argtypes = [
            C.c_void_p, C.c_void_p, C.c_void_p, C.c_void_p, C.c_void_p,
            _u64p, C.c_int, C.POINTER(UaPTask), _u64p]
        L.ua_pbatch_create.argtypes = [
            C.c_void_p, C.c_void_p, C.c_void_p, C.c_void_p, C.c_void_p,
            _u64p, C.c_int, C.POINTER(UaPTask), _voidpp]
        L.ua_pbatch_run.argtypes = [C.c_void_p, C.c_void_p, _u64p]
        L.ua_pbatch_destroy.argtypes = [C.c_void_p, C.c_void_p]
        L.ua_pbatch_destroy.restype = None
        L.ua_decode_dev.argtypes = [C.c_void_p, C.POINTER(UaDPack), _u64, C.c_void_p, _u64p]
        L.ua_encode.argtypes = [_u64p, _u64, C.c_uint32, _voidpp]
        L.ua_encode_dev.argtypes = [C.c_void_p, C.c_void_p, _u64, C.c_uint32,
                                    C.c_void_p, C.c_void_p, C.c_void_p, C.c_void_p,
                                    _u64p, _u64p]
        L.ua_owned_pack_view.restype = C.POINTER(UaPack)
        L.ua_owned_pack_view.argtypes = [C.c_void_p]
        L.ua_owned_pack_free.argtypes = [C.c_void_p]
        L.ua_pack_exact_len.restype = _u64
        L.ua_pack_exact_len.argtypes = [C.POINTER(UaPack)]
        L.ua_pack_approx_len.restype = _u64
        L.ua_pack_approx_len.argtypes = [C.POINTER(UaPack)]
        L.ua_pack_flat_sizes.argtypes = [C.POINTER(UaPack), _u64p, _u64p, _u64p]
        L.ua_pack_flatten.argtypes = [C.POINTER(UaPack), _u64p, _u32p, _u64p, _u8p]
        L.ua_intersect.argtypes = [C.c_void_p, _u64p, _u64, _u64p, _u64, _u64p, _u64p]
        L.ua_intersect_k.argtypes = [C.c_void_p, C.POINTER(_u64p), _u64p, C.c_int, _u64p, _u64p]
        L.ua_merge_k.argtypes = [C.c_void_p, C.POINTER(_u64p), _u64p, C.c_int, _u64p, _u64p]
        L.ua_difference.argtypes = [C.c_void_p, _u64p, _u64, _u64p, _u64, _u64p, _u64p]
        L.ua_index_of.restype = C.c_int64
        L.ua_index_of.argtypes = [_u64p, _u64, _u64]
        L.ua_intersect_packed.argtypes = [C.c_void_p, C.POINTER(UaPack), _u64, _u64p, _u64,
                                          _u64p, _u64p]
        _lib = L
    return _lib


def check(rc):
    if rc != UA_OK:
        msg = lib().ua_strerror(rc).decode()
        raise RuntimeError(f"uidalgo error {rc}: {msg}")
