"""ctypes bindings for liboracle.so — TEST INFRASTRUCTURE ONLY.

Bit-exact CPU restatement of dgraph's algo/uidlist.go + codec/codec.go
(see oracle.c for file:line citations).  Used by tests/, smoke() and
bench.py's cpu_baseline leg; never by the product path.
"""
import ctypes as C
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_DIR, "liboracle.so")

SEEK_START = 0
SEEK_CURRENT = 1

_u64p = C.POINTER(C.c_uint64)
_szp = C.POINTER(C.c_size_t)


class _Block(C.Structure):
    _fields_ = [
        ("base", C.c_uint64),
        ("num_uids", C.c_uint32),
        ("deltas_len", C.c_uint32),
        ("deltas", C.POINTER(C.c_uint8)),
    ]


class _Pack(C.Structure):
    _fields_ = [
        ("block_size", C.c_uint32),
        ("n_blocks", C.c_size_t),
        ("blocks", C.POINTER(_Block)),
    ]


class _Dec(C.Structure):
    _fields_ = [
        ("pack", C.POINTER(_Pack)),
        ("block_idx", C.c_int),
        ("uids", _u64p),
        ("n_uids", C.c_size_t),
        ("buf", _u64p),
        ("buf_cap", C.c_size_t),
    ]


def build():
    subprocess.run(["make", "-s", "-C", _DIR], check=True)


_lib = None


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(_LIB_PATH):
            build()
        L = C.CDLL(_LIB_PATH)
        L.orc_encode.restype = C.POINTER(_Pack)
        L.orc_encode.argtypes = [_u64p, C.c_size_t, C.c_int]
        L.orc_pack_free.argtypes = [C.POINTER(_Pack)]
        L.orc_pack_approx_len.restype = C.c_size_t
        L.orc_pack_approx_len.argtypes = [C.POINTER(_Pack)]
        L.orc_pack_exact_len.restype = C.c_size_t
        L.orc_pack_exact_len.argtypes = [C.POINTER(_Pack)]
        L.orc_decode.restype = C.c_size_t
        L.orc_decode.argtypes = [C.POINTER(_Pack), C.c_uint64, _u64p]
        L.orc_dec_init.argtypes = [C.POINTER(_Dec), C.POINTER(_Pack)]
        L.orc_dec_free.argtypes = [C.POINTER(_Dec)]
        L.orc_dec_seek.argtypes = [C.POINTER(_Dec), C.c_uint64, C.c_int]
        L.orc_dec_seek_to_block.argtypes = [C.POINTER(_Dec), C.c_uint64, C.c_int]
        L.orc_dec_linear_seek.argtypes = [C.POINTER(_Dec), C.c_uint64]
        L.orc_dec_next.argtypes = [C.POINTER(_Dec)]
        L.orc_dec_valid.restype = C.c_int
        L.orc_dec_valid.argtypes = [C.POINTER(_Dec)]
        L.orc_intersect_with.restype = C.c_size_t
        L.orc_intersect_with.argtypes = [_u64p, C.c_size_t, _u64p, C.c_size_t, _u64p]
        L.orc_intersect_sorted.restype = C.c_size_t
        L.orc_intersect_sorted.argtypes = [C.POINTER(_u64p), _szp, C.c_size_t, _u64p]
        L.orc_merge_sorted.restype = C.c_size_t
        L.orc_merge_sorted.argtypes = [C.POINTER(_u64p), _szp, C.c_size_t, _u64p]
        L.orc_difference.restype = C.c_size_t
        L.orc_difference.argtypes = [_u64p, C.c_size_t, _u64p, C.c_size_t, _u64p]
        L.orc_index_of.restype = C.c_int64
        L.orc_index_of.argtypes = [_u64p, C.c_size_t, C.c_uint64]
        L.orc_apply_filter.restype = C.c_size_t
        L.orc_apply_filter.argtypes = [_u64p, C.c_size_t, C.POINTER(C.c_uint8)]
        L.orc_intersect_compressed_with.restype = C.c_size_t
        L.orc_intersect_compressed_with.argtypes = [
            C.POINTER(_Pack), C.c_uint64, _u64p, C.c_size_t, _u64p]
        L.orc_intersect_batch_cpu.argtypes = [
            C.c_int, C.POINTER(_u64p), _szp, C.POINTER(_u64p), _szp,
            C.POINTER(_u64p), _szp, C.c_int, C.POINTER(C.c_int)]
        L.orc_omp_max_threads.restype = C.c_int
        _lib = L
    return _lib


def _arr(x):
    a = np.ascontiguousarray(np.asarray(x, dtype=np.uint64))
    return a


_EMPTY = np.empty(1, dtype=np.uint64)  # non-NULL pointer for empty slices
                                       # (NULL is reserved for Go's nil *pb.List)


def _ptr(a):
    if a.size == 0:
        return _EMPTY.ctypes.data_as(_u64p)
    return a.ctypes.data_as(_u64p)


def intersect_with(u, v):
    u, v = _arr(u), _arr(v)
    out = np.empty(min(u.size, v.size), dtype=np.uint64)
    n = lib().orc_intersect_with(_ptr(u), u.size, _ptr(v), v.size, _ptr(out))
    return out[:n].copy()


def difference(u, v):
    u, v = _arr(u), _arr(v)
    out = np.empty(u.size, dtype=np.uint64)
    n = lib().orc_difference(_ptr(u), u.size, _ptr(v), v.size, _ptr(out))
    return out[:n].copy()


def _lists_args(lists):
    arrs = [_arr(x) for x in lists]
    k = len(arrs)
    ptrs = (_u64p * max(k, 1))(*[_ptr(a) for a in arrs])
    lens = (C.c_size_t * max(k, 1))(*[a.size for a in arrs])
    return arrs, ptrs, lens, k


def intersect_sorted(lists):
    arrs, ptrs, lens, k = _lists_args(lists)
    cap = min((a.size for a in arrs), default=0)
    if k == 1:
        cap = arrs[0].size
    out = np.empty(max(cap, 1), dtype=np.uint64)
    n = lib().orc_intersect_sorted(ptrs, lens, k, _ptr(out))
    return out[:n].copy()


def merge_sorted(lists):
    arrs, ptrs, lens, k = _lists_args(lists)
    cap = sum(a.size for a in arrs)
    out = np.empty(max(cap, 1), dtype=np.uint64)
    n = lib().orc_merge_sorted(ptrs, lens, k, _ptr(out))
    return out[:n].copy()


def index_of(u, uid):
    u = _arr(u)
    return int(lib().orc_index_of(_ptr(u), u.size, C.c_uint64(uid)))


def apply_filter(u, mask):
    """uidlist.go:21 ApplyFilter (mask = precomputed f(uid, i))."""
    u = _arr(u).copy()  # in-place in C; keep the caller's array intact
    m = np.ascontiguousarray(np.asarray(mask, dtype=bool).view(np.uint8))
    assert m.size == u.size
    n = lib().orc_apply_filter(_ptr(u), u.size,
                               m.ctypes.data_as(C.POINTER(C.c_uint8)))
    return u[:n].copy()


class Pack:
    """Owns an orc_pack* (mirror of pb.UidPack)."""

    def __init__(self, uids, block_size):
        self._uids = _arr(uids)
        self.p = lib().orc_encode(_ptr(self._uids), self._uids.size, block_size)

    def __del__(self):
        try:
            if self.p:
                lib().orc_pack_free(self.p)
                self.p = None
        except Exception:
            pass

    @property
    def n_blocks(self):
        return int(self.p.contents.n_blocks)

    def exact_len(self):
        return int(lib().orc_pack_exact_len(self.p))

    def approx_len(self):
        return int(lib().orc_pack_approx_len(self.p))

    def decode(self, seek=0):
        # exact_len, not approx_len: BlockSize=0 packs give ApproxLen 0
        out = np.empty(max(self.exact_len(), 1), dtype=np.uint64)
        n = lib().orc_decode(self.p, C.c_uint64(seek), _ptr(out))
        return out[:n].copy()

    def flatten(self):
        """Return (bases, num_uids, delta_offsets, deltas_blob) numpy arrays —
        the flat device layout the product engine consumes."""
        nb = self.n_blocks
        bases = np.empty(nb, dtype=np.uint64)
        nums = np.empty(nb, dtype=np.uint32)
        offs = np.empty(nb + 1, dtype=np.uint64)
        total = 0
        blocks = self.p.contents.blocks
        for i in range(nb):
            b = blocks[i]
            bases[i] = b.base
            nums[i] = b.num_uids
            offs[i] = total
            total += b.deltas_len
        offs[nb] = total
        blob = np.empty(total, dtype=np.uint8)
        pos = 0
        for i in range(nb):
            b = blocks[i]
            if b.deltas_len:
                blob[pos:pos + b.deltas_len] = np.ctypeslib.as_array(
                    b.deltas, shape=(b.deltas_len,))
            pos += b.deltas_len
        return bases, nums, offs, blob


class Dec:
    """Wraps orc_dec (codec.Decoder)."""

    def __init__(self, pack: Pack):
        self.pack = pack
        self.d = _Dec()
        lib().orc_dec_init(C.byref(self.d), pack.p)

    def __del__(self):
        try:
            lib().orc_dec_free(C.byref(self.d))
        except Exception:
            pass

    def uids(self):
        n = self.d.n_uids
        if n == 0:
            return np.empty(0, dtype=np.uint64)
        return np.ctypeslib.as_array(self.d.uids, shape=(n,)).copy()

    def seek(self, uid, whence=SEEK_START):
        lib().orc_dec_seek(C.byref(self.d), C.c_uint64(uid), whence)
        return self.uids()

    def seek_to_block(self, uid, whence=SEEK_CURRENT):
        lib().orc_dec_seek_to_block(C.byref(self.d), C.c_uint64(uid), whence)
        return self.uids()

    def linear_seek(self, uid):
        lib().orc_dec_linear_seek(C.byref(self.d), C.c_uint64(uid))
        return self.uids()

    def next(self):
        lib().orc_dec_next(C.byref(self.d))
        return self.uids()

    def valid(self):
        return bool(lib().orc_dec_valid(C.byref(self.d)))


def intersect_compressed_with(pack: Pack, after, v):
    v = _arr(v)
    cap = max(pack.exact_len(), v.size, 1)
    out = np.empty(cap, dtype=np.uint64)
    n = lib().orc_intersect_compressed_with(pack.p, C.c_uint64(after), _ptr(v), v.size, _ptr(out))
    return out[:n].copy()


def intersect_batch_cpu(us, vs, n_threads=0, return_threads_used=False):
    """OpenMP batched IntersectWith across pairs — the CPU baseline leg."""
    us = [_arr(u) for u in us]
    vs = [_arr(v) for v in vs]
    k = len(us)
    outs = [np.empty(min(u.size, v.size) or 1, dtype=np.uint64) for u, v in zip(us, vs)]
    up = (_u64p * k)(*[_ptr(a) for a in us])
    vp = (_u64p * k)(*[_ptr(a) for a in vs])
    op = (_u64p * k)(*[_ptr(a) for a in outs])
    un = (C.c_size_t * k)(*[a.size for a in us])
    vn = (C.c_size_t * k)(*[a.size for a in vs])
    on = (C.c_size_t * k)()
    tu = C.c_int(0)
    lib().orc_intersect_batch_cpu(k, up, un, vp, vn, op, on, n_threads, C.byref(tu))
    res = [outs[i][:on[i]].copy() for i in range(k)]
    if return_threads_used:
        return res, tu.value
    return res


def omp_max_threads():
    return int(lib().orc_omp_max_threads())
