"""Host-side mirror of dgraph's `algo` package over the uidalgo C-ABI.

Function-for-function surface of /root/reference/algo/uidlist.go (same names'
semantics, same argument meaning, status-code errors), plus the batched
device-resident engine the reference's per-key fan-out maps onto
(worker/task.go:834-971 -> one grid).

UID lists are sorted uint64.  torch has no uint64 CUDA dtype, so device
buffers are int64 tensors REINTERPRETED as u64 by the kernels; numpy uint64
arrays cross the host boundary bit-for-bit via .view(int64).

No CPU fallback: every op goes through libuidalgo.so and raises without it.
"""
import ctypes as C
import sys

import numpy as np

from dgraph_amd._lib import (UaDPack, UaDPair, check, lib, _u64p)

_u64 = C.c_uint64


def _np_u64(x):
    a = np.ascontiguousarray(np.asarray(x, dtype=np.uint64))
    return a


_KEEP = np.empty(1, dtype=np.uint64)


def _hptr(a):
    if a.size == 0:
        return _KEEP.ctypes.data_as(_u64p)
    return a.ctypes.data_as(_u64p)


class Engine:
    """One GPU + one HIP stream (wraps ua_ctx)."""

    def __init__(self, device=0):
        self._ctx = C.c_void_p()
        rc = lib().ua_ctx_create(C.byref(self._ctx), device)
        if rc != 0:
            raise RuntimeError(
                f"uidalgo: cannot create GPU context on device {device}: "
                f"{lib().ua_strerror(rc).decode()} — the HIP engine is "
                "mandatory, there is no CPU fallback.")
        self.device = device

    def close(self):
        if self._ctx:
            lib().ua_ctx_destroy(self._ctx)
            self._ctx = C.c_void_p()

    def __del__(self):
        try:
            if sys.is_finalizing():
                return  # HIP runtime may already be torn down at interpreter exit
            self.close()
        except Exception:
            pass

    # ---- stats (dominant-kernel HIP-event time) ----
    def stats_reset(self):
        check(lib().ua_stats_reset(self._ctx))

    def stats(self):
        n = _u64()
        ms = C.c_double()
        b = _u64()
        check(lib().ua_stats_get(self._ctx, C.byref(n), C.byref(ms), C.byref(b)))
        return {"launches": n.value, "kernel_ms": ms.value, "bytes_algorithmic": b.value}

    # ---- batched device-resident ops (torch CUDA int64 tensors) ----
    def _run_pairs(self, fn, us, vs, outs):
        np_ = len(us)
        pairs = (UaDPair * np_)()
        for i in range(np_):
            pairs[i].u = us[i].data_ptr()
            pairs[i].n = us[i].numel()
            pairs[i].v = vs[i].data_ptr()
            pairs[i].m = vs[i].numel()
            pairs[i].out = outs[i].data_ptr()
        lens = (_u64 * np_)()
        check(fn(self._ctx, pairs, np_, lens))
        return [int(lens[i]) for i in range(np_)]

    def intersect_pairs(self, us, vs, outs=None):
        """Batched algo.IntersectWith.  us/vs: lists of CUDA int64 tensors
        (u64 bits).  Returns (outs, lens); outs[i][:lens[i]] is the result."""
        import torch
        if outs is None:
            outs = [torch.empty(min(u.numel(), v.numel()), dtype=torch.int64,
                                device=u.device) for u, v in zip(us, vs)]
        lens = self._run_pairs(lib().ua_intersect_batch_dev, us, vs, outs)
        return outs, lens

    def merge_pairs(self, us, vs, outs=None):
        import torch
        if outs is None:
            outs = [torch.empty(u.numel() + v.numel(), dtype=torch.int64,
                                device=u.device) for u, v in zip(us, vs)]
        lens = self._run_pairs(lib().ua_merge_batch_dev, us, vs, outs)
        return outs, lens

    def difference_pairs(self, us, vs, outs=None):
        import torch
        if outs is None:
            outs = [torch.empty(max(u.numel(), 1), dtype=torch.int64, device=u.device)
                    for u in us]
        lens = self._run_pairs(lib().ua_difference_batch_dev, us, vs, outs)
        return outs, lens

    def make_batch(self, us, vs, outs):
        """Prepared batch: descriptor upload + merge-path partition done once
        (ua_batch_create); run ops repeatedly with Batch.run().  The input
        tensors' contents must stay unchanged while the batch lives."""
        np_ = len(us)
        pairs = (UaDPair * np_)()
        for i in range(np_):
            pairs[i].u = us[i].data_ptr()
            pairs[i].n = us[i].numel()
            pairs[i].v = vs[i].data_ptr()
            pairs[i].m = vs[i].numel()
            pairs[i].out = outs[i].data_ptr()
        h = C.c_void_p()
        check(lib().ua_batch_create(self._ctx, pairs, np_, C.byref(h)))
        b = Batch(self, h, np_, (us, vs, outs))
        b._caps = [(us[i].numel(), vs[i].numel(), outs[i].numel())
                   for i in range(np_)]
        return b

    def intersect_sorted(self, lists):
        """algo.IntersectSorted (uidlist.go:297): k-way fold, smallest first."""
        import torch
        k = len(lists)
        if k == 0:
            return torch.empty(0, dtype=torch.int64)
        ptrs = (C.c_void_p * k)(*[int(t.data_ptr()) for t in lists])
        lens = (_u64 * k)(*[t.numel() for t in lists])
        cap = min(t.numel() for t in lists)
        out = torch.empty(max(cap, 1), dtype=torch.int64, device=lists[0].device)
        out_n = _u64()
        check(lib().ua_intersect_k_dev(self._ctx, C.cast(ptrs, C.POINTER(C.c_void_p)),
                                       lens, k, C.c_void_p(out.data_ptr()),
                                       C.byref(out_n)))
        return out[:out_n.value]

    def merge_sorted(self, lists):
        """algo.MergeSorted (uidlist.go:448): dedup'd k-way union."""
        import torch
        k = len(lists)
        if k == 0:
            return torch.empty(0, dtype=torch.int64)
        ptrs = (C.c_void_p * k)(*[int(t.data_ptr()) for t in lists])
        lens = (_u64 * k)(*[t.numel() for t in lists])
        cap = sum(t.numel() for t in lists)
        out = torch.empty(max(cap, 1), dtype=torch.int64, device=lists[0].device)
        out_n = _u64()
        check(lib().ua_merge_k_dev(self._ctx, C.cast(ptrs, C.POINTER(C.c_void_p)),
                                   lens, k, C.c_void_p(out.data_ptr()), C.byref(out_n)))
        return out[:out_n.value]

    def merge_all_pairs(self, us, vs, outs=None):
        """Batched duplicate-KEEPING merge of sorted runs (no dedup)."""
        import torch
        if outs is None:
            outs = [torch.empty(max(u.numel() + v.numel(), 1), dtype=torch.int64,
                                device=u.device) for u, v in zip(us, vs)]
        lens = self._run_pairs(lib().ua_merge_all_batch_dev, us, vs, outs)
        return outs, lens

    def sort_segments(self, tensors):
        """Batched segmented sort in place (u64 ascending, duplicates kept):
        the sort-path primitive (worker/sort.go UidMatrix shapes, SURVEY
        §8f row 3)."""
        import torch
        from dgraph_amd._lib import UaDSeg
        k = len(tensors)
        if k == 0:
            return tensors
        tmps = [torch.empty(max(t.numel(), 1), dtype=torch.int64, device=t.device)
                for t in tensors]
        segs = (UaDSeg * k)()
        for i, t in enumerate(tensors):
            segs[i].data = t.data_ptr()
            segs[i].n = t.numel()
            segs[i].tmp = tmps[i].data_ptr()
        check(lib().ua_sort_segments_dev(self._ctx, segs, k))
        return tensors

    def apply_filter_batch(self, us, masks, outs=None):
        """Batched algo.ApplyFilter (uidlist.go:21): order-preserving mask
        compaction on device.  us: CUDA int64 tensors; masks: CUDA uint8/bool
        tensors (nonzero = keep).  outs may be the us themselves (in-place,
        the reference's shape).  Returns (outs, lens)."""
        import torch
        from dgraph_amd._lib import UaDFilter
        k = len(us)
        if outs is None:
            outs = us  # in-place like the reference
        tasks = (UaDFilter * max(k, 1))()
        for i in range(k):
            m = masks[i]
            if m.dtype == torch.bool:
                m = m.view(torch.uint8)
            assert m.dtype == torch.uint8 and m.numel() == us[i].numel()
            tasks[i].u = us[i].data_ptr()
            tasks[i].n = us[i].numel()
            tasks[i].mask = m.data_ptr()
            tasks[i].out = outs[i].data_ptr()
        lens = (_u64 * max(k, 1))()
        check(lib().ua_apply_filter_batch_dev(self._ctx, tasks, k, lens))
        return outs, [int(lens[i]) for i in range(k)]

    def index_of_batch(self, u, queries):
        """Batched algo.IndexOf: u, queries CUDA int64; returns int64 tensor
        of positions (-1 = absent)."""
        import torch
        out = torch.empty(queries.numel(), dtype=torch.int64, device=u.device)
        check(lib().ua_index_of_batch_dev(
            self._ctx, C.c_void_p(u.data_ptr()), u.numel(),
            C.c_void_p(queries.data_ptr()), queries.numel(),
            C.c_void_p(out.data_ptr())))
        return out

    # ---- packed (codec) path ----
    def upload_pack(self, bases, num_uids, delta_offs, deltas_blob, block_size):
        """Upload flat pack arrays (from encode_flat) -> DPack on this GPU."""
        import torch
        dev = f"cuda:{self.device}"
        t_bases = torch.from_numpy(_np_u64(bases).view(np.int64)).to(dev)
        t_nums = torch.from_numpy(np.ascontiguousarray(num_uids, dtype=np.uint32)
                                  .view(np.int32)).to(dev)
        t_offs = torch.from_numpy(_np_u64(delta_offs).view(np.int64)).to(dev)
        blob = np.ascontiguousarray(deltas_blob, dtype=np.uint8)
        t_blob = torch.from_numpy(blob.view(np.int8)).to(dev) if blob.size else \
            torch.zeros(1, dtype=torch.int8, device=dev)
        total = int(np.asarray(num_uids, dtype=np.uint64).sum())
        return DPack(t_bases, t_nums, t_offs, t_blob, int(block_size), total)

    def upload_pack_batch(self, packs):
        """Concatenate many flat packs (tuples from encode_flat) into one
        device arena + pack boundaries — the collected form of a
        handleUidPostings fan-out (worker/task.go:834-971)."""
        import torch
        dev = f"cuda:{self.device}"
        pbb = np.zeros(len(packs) + 1, dtype=np.uint64)
        blob_base = 0
        all_bases, all_nums, all_offs, all_blobs, totals = [], [], [], [], []
        for i, (bases, nums, offs, blob, total) in enumerate(packs):
            pbb[i + 1] = pbb[i] + bases.size
            all_bases.append(bases)
            all_nums.append(nums)
            all_offs.append(np.asarray(offs[:-1], dtype=np.uint64) + np.uint64(blob_base))
            all_blobs.append(blob)
            totals.append(total)
            blob_base += blob.size
        cat = lambda xs, dt: np.ascontiguousarray(
            np.concatenate(xs) if xs else np.empty(0, dtype=dt), dtype=dt)
        bases = cat(all_bases, np.uint64)
        nums = cat(all_nums, np.uint32)
        offs = np.concatenate([cat(all_offs, np.uint64),
                               np.array([blob_base], dtype=np.uint64)])
        blob = cat(all_blobs, np.uint8)
        dp = self.upload_pack(bases, nums, offs, blob, 0)
        dp.pbb = pbb
        dp.pack_totals = totals
        return dp

    def intersect_packed_batch(self, dpb, vs, outs=None, afters=None):
        """Batched algo.IntersectCompressedWith: every pack of the fan-out in
        ONE grid.  vs: per-pack CUDA int64 tensors (may repeat one shared
        tensor — the q.UidList shape).  Returns (outs, lens)."""
        import torch
        from dgraph_amd._lib import UaPTask
        n = len(dpb.pbb) - 1
        assert len(vs) == n
        if afters is None:
            afters = [0] * n
        if outs is None:
            outs = [torch.empty(max(min(dpb.pack_totals[i], vs[i].numel()), 1),
                                dtype=torch.int64, device=vs[i].device)
                    for i in range(n)]
        tasks = (UaPTask * n)()
        for i in range(n):
            tasks[i].v = vs[i].data_ptr()
            tasks[i].m = vs[i].numel()
            tasks[i].out = outs[i].data_ptr()
            tasks[i].after_uid = afters[i]
        pbb = (C.c_uint64 * (n + 1))(*[int(x) for x in dpb.pbb])
        lens = (C.c_uint64 * n)()
        check(lib().ua_intersect_packed_batch_dev(
            self._ctx, C.c_void_p(dpb.bases.data_ptr()),
            C.c_void_p(dpb.num_uids.data_ptr()), C.c_void_p(dpb.delta_offs.data_ptr()),
            C.c_void_p(dpb.deltas.data_ptr()), pbb, n, tasks, lens))
        return outs, [int(lens[i]) for i in range(n)]

    def make_pack_batch(self, dpb, vs, outs=None, afters=None):
        """Prepared pack fan-out (ua_pbatch): a standing query plan — tasks
        and boundaries upload once, each run() is launches only.  vs may
        repeat one shared tensor (the q.UidList shape)."""
        import torch
        from dgraph_amd._lib import UaPTask
        n = len(dpb.pbb) - 1
        assert len(vs) == n
        if afters is None:
            afters = [0] * n
        if outs is None:
            outs = [torch.empty(max(min(dpb.pack_totals[i], vs[i].numel()), 1),
                                dtype=torch.int64, device=vs[i].device)
                    for i in range(n)]
        tasks = (UaPTask * n)()
        for i in range(n):
            tasks[i].v = vs[i].data_ptr()
            tasks[i].m = vs[i].numel()
            tasks[i].out = outs[i].data_ptr()
            tasks[i].after_uid = afters[i]
        pbb = (C.c_uint64 * (n + 1))(*[int(x) for x in dpb.pbb])
        h = C.c_void_p()
        check(lib().ua_pbatch_create(
            self._ctx, C.c_void_p(dpb.bases.data_ptr()),
            C.c_void_p(dpb.num_uids.data_ptr()), C.c_void_p(dpb.delta_offs.data_ptr()),
            C.c_void_p(dpb.deltas.data_ptr()), pbb, n, tasks, C.byref(h)))
        return PackBatch(self, h, n, (dpb, vs, outs))

    def intersect_packed(self, dpack, after, v, out=None):
        """algo.IntersectCompressedWith (uidlist.go:33): fused decode+intersect."""
        import torch
        if out is None:
            out = torch.empty(max(min(dpack.total_uids, v.numel()), 1),
                              dtype=torch.int64, device=v.device)
        out_n = _u64()
        pk = dpack.struct()
        check(lib().ua_intersect_packed_dev(
            self._ctx, C.byref(pk), _u64(after), C.c_void_p(v.data_ptr()),
            v.numel(), C.c_void_p(out.data_ptr()), C.byref(out_n)))
        return out[:out_n.value]

    def encode_dev(self, uids, block_size=256):
        """codec.Encode on the GPU (codec.go:393): sorted device int64 (u64
        bits) -> DPack resident in HBM.  Byte-identical to the reference
        group-varint block format."""
        import torch
        n = uids.numel()
        dev = uids.device
        mb = max(n, 1)
        bases = torch.empty(mb, dtype=torch.int64, device=dev)
        nums = torch.empty(mb, dtype=torch.int32, device=dev)
        offs = torch.empty(mb + 1, dtype=torch.int64, device=dev)
        blob = torch.empty(6 * mb + 24, dtype=torch.int8, device=dev)
        nb = _u64()
        db = _u64()
        check(lib().ua_encode_dev(
            self._ctx, C.c_void_p(uids.data_ptr()), n, block_size,
            C.c_void_p(bases.data_ptr()), C.c_void_p(nums.data_ptr()),
            C.c_void_p(offs.data_ptr()), C.c_void_p(blob.data_ptr()),
            C.byref(nb), C.byref(db)))
        k = nb.value
        return DPack(bases[:max(k, 1)], nums[:max(k, 1)], offs[:k + 1],
                     blob[:max(db.value, 1)], int(block_size), n) if k else \
            DPack(bases[:0], nums[:0], offs[:1], blob[:1], int(block_size), 0)

    # ---- packed compositions (algo/packed.go) ----
    def merge_sorted_packed(self, lists, block_size=256):
        """algo.MergeSortedPacked (packed.go:222): dedup k-way union, packed
        on-device."""
        return self.encode_dev(self.merge_sorted(lists), block_size)

    def intersect_sorted_packed(self, dpacks):
        """algo.IntersectSortedPacked (packed.go:100): decode each pack,
        fold-intersect smallest-first, re-pack (block size of the first
        list, like the reference)."""
        bs = dpacks[0].block_size if dpacks else 10
        lists = [self.decode_pack(dp) for dp in dpacks]
        return self.encode_dev(self.intersect_sorted(lists), bs)

    def apply_filter_packed(self, dpack, mask_fn):
        """algo.ApplyFilterPacked (packed.go:16): decode, boolean-mask
        compaction, re-pack."""
        dec = self.decode_pack(dpack)
        return self.encode_dev(dec[mask_fn(dec)], dpack.block_size)

    def decode_pack(self, dpack, seek=0, out=None):
        """codec.Decode(pack, seek) on the GPU (codec.go:444)."""
        import torch
        if out is None:
            out = torch.empty(max(dpack.total_uids, 1), dtype=torch.int64,
                              device=dpack.bases.device)
        out_n = _u64()
        pk = dpack.struct()
        check(lib().ua_decode_dev(self._ctx, C.byref(pk), _u64(seek),
                                  C.c_void_p(out.data_ptr()), C.byref(out_n)))
        return out[:out_n.value]


OP_INTERSECT = 0
OP_MERGE = 1
OP_DIFFERENCE = 2


class Batch:
    """Prepared pair batch (ua_batch): run() executes one op over all pairs
    as one grid, returning per-pair output lengths."""

    def __init__(self, engine, handle, n_pairs, keepalive):
        self._eng = engine
        self._h = handle
        self.n_pairs = n_pairs
        self._keep = keepalive  # input/output tensors must outlive the batch

    def _check_caps(self, op):
        """Per-pair output capacity contract (uidalgo.h: intersect needs
        min(n,m), difference n, union n+m).  The C ABI sees raw pointers;
        this layer knows the tensor sizes, so an undersized output becomes a
        clean error instead of OOB device writes."""
        caps = getattr(self, "_caps", None)
        if caps is None:
            return
        for i, (n, m, o) in enumerate(caps):
            need = min(n, m) if op == OP_INTERSECT else (
                n if op == OP_DIFFERENCE else n + m)
            if o < need:
                raise ValueError(
                    f"pair {i}: out capacity {o} < required {need} for op {op}")

    def run(self, op=OP_INTERSECT):
        self._check_caps(op)
        lens = (C.c_uint64 * self.n_pairs)()
        check(lib().ua_batch_run(self._eng._ctx, self._h, op, lens))
        return [int(lens[i]) for i in range(self.n_pairs)]

    def run_n(self, op, n_runs):
        """n_runs passes enqueued back-to-back, ONE sync at the end (the
        repeated-query serving shape — no host round-trip between runs)."""
        self._check_caps(op)
        lens = (C.c_uint64 * self.n_pairs)()
        check(lib().ua_batch_run_n(self._eng._ctx, self._h, op, n_runs, lens))
        return [int(lens[i]) for i in range(self.n_pairs)]

    def close(self):
        if self._h:
            lib().ua_batch_destroy(self._eng._ctx, self._h)
            self._h = C.c_void_p()

    def __del__(self):
        try:
            if sys.is_finalizing():
                return
            self.close()
        except Exception:
            pass


class PackBatch:
    """Prepared pack fan-out (ua_pbatch)."""

    def __init__(self, engine, handle, n_packs, keepalive):
        self._eng = engine
        self._h = handle
        self.n_packs = n_packs
        self.outs = keepalive[2]
        self._keep = keepalive

    def run(self):
        lens = (C.c_uint64 * self.n_packs)()
        check(lib().ua_pbatch_run(self._eng._ctx, self._h, lens))
        return [int(lens[i]) for i in range(self.n_packs)]

    def close(self):
        if self._h:
            lib().ua_pbatch_destroy(self._eng._ctx, self._h)
            self._h = C.c_void_p()

    def __del__(self):
        try:
            if sys.is_finalizing():
                return
            self.close()
        except Exception:
            pass


class DPack:
    """Flattened pb.UidPack resident in HBM (engine-native layout)."""

    def __init__(self, bases, num_uids, delta_offs, deltas, block_size, total_uids):
        self.bases = bases
        self.num_uids = num_uids
        self.delta_offs = delta_offs
        self.deltas = deltas
        self.block_size = block_size
        self.total_uids = total_uids

    def struct(self):
        pk = UaDPack()
        pk.block_size = self.block_size
        pk.n_blocks = self.bases.numel()
        pk.bases = self.bases.data_ptr()
        pk.num_uids = self.num_uids.data_ptr()
        pk.delta_offs = self.delta_offs.data_ptr()
        pk.deltas = self.deltas.data_ptr()
        pk.total_uids = self.total_uids
        return pk


# ---- host-side codec (product encoder; codec.Encode restated in C++) ----

def encode_flat(uids, block_size=256):
    """codec.Encode(uids, blockSize) -> flat arrays
    (bases u64[nb], num_uids u32[nb], delta_offs u64[nb+1], deltas u8[...])."""
    uids = _np_u64(uids)
    h = C.c_void_p()
    check(lib().ua_encode(_hptr(uids), uids.size, block_size, C.byref(h)))
    try:
        view = lib().ua_owned_pack_view(h)
        nb = _u64()
        db = _u64()
        tu = _u64()
        check(lib().ua_pack_flat_sizes(view, C.byref(nb), C.byref(db), C.byref(tu)))
        bases = np.empty(nb.value, dtype=np.uint64)
        nums = np.empty(nb.value, dtype=np.uint32)
        offs = np.empty(nb.value + 1, dtype=np.uint64)
        blob = np.empty(max(db.value, 1), dtype=np.uint8)
        check(lib().ua_pack_flatten(
            view, _hptr(bases), nums.ctypes.data_as(C.POINTER(C.c_uint32)),
            _hptr(offs), blob.ctypes.data_as(C.POINTER(C.c_uint8))))
        return bases, nums, offs, blob[:db.value], tu.value
    finally:
        lib().ua_owned_pack_free(h)


# ---- host-pointer convenience mirror (the cgo drop-in surface) ----
# These run on the GPU via upload/compute/download; they exist so the cgo shim
# maps 1:1 onto algo's signatures.

def intersect_with(engine, u, v):
    u, v = _np_u64(u), _np_u64(v)
    out = np.empty(max(min(u.size, v.size), 1), dtype=np.uint64)
    n = _u64()
    check(lib().ua_intersect(engine._ctx, _hptr(u), u.size, _hptr(v), v.size,
                             _hptr(out), C.byref(n)))
    return out[:n.value]


def difference(engine, u, v):
    u, v = _np_u64(u), _np_u64(v)
    out = np.empty(max(u.size, 1), dtype=np.uint64)
    n = _u64()
    check(lib().ua_difference(engine._ctx, _hptr(u), u.size, _hptr(v), v.size,
                              _hptr(out), C.byref(n)))
    return out[:n.value]


def _host_lists(lists):
    arrs = [_np_u64(x) for x in lists]
    k = len(arrs)
    ptrs = (_u64p * max(k, 1))(*[_hptr(a) for a in arrs])
    lens = (_u64 * max(k, 1))(*[a.size for a in arrs])
    return arrs, ptrs, lens, k


def intersect_sorted(engine, lists):
    arrs, ptrs, lens, k = _host_lists(lists)
    cap = min((a.size for a in arrs), default=0)
    out = np.empty(max(cap, 1), dtype=np.uint64)
    n = _u64()
    check(lib().ua_intersect_k(engine._ctx, ptrs, lens, k, _hptr(out), C.byref(n)))
    return out[:n.value]


def merge_sorted(engine, lists):
    arrs, ptrs, lens, k = _host_lists(lists)
    cap = sum(a.size for a in arrs)
    out = np.empty(max(cap, 1), dtype=np.uint64)
    n = _u64()
    check(lib().ua_merge_k(engine._ctx, ptrs, lens, k, _hptr(out), C.byref(n)))
    return out[:n.value]


def index_of(u, uid):
    """algo.IndexOf — host binary search, like the reference (uidlist.go:546)."""
    u = _np_u64(u)
    return int(lib().ua_index_of(_hptr(u), u.size, _u64(uid)))


def apply_filter(u, mask, engine=None):
    """algo.ApplyFilter (uidlist.go:21): boolean-mask compaction.

    The reference takes a Go closure; across a C ABI the filter arrives as a
    precomputed mask (the callers evaluate per-uid predicates upstream).
    With an Engine, host arrays go through the ua_apply_filter C-ABI export
    (in-place device compaction — the cgo drop-in path); CUDA tensors go
    through the batched device export."""
    import torch
    if isinstance(u, torch.Tensor):
        if engine is not None and u.is_cuda:
            m = mask if isinstance(mask, torch.Tensor) else \
                torch.as_tensor(np.asarray(mask, dtype=np.uint8), device=u.device)
            out = torch.empty_like(u)
            outs, lens = engine.apply_filter_batch([u], [m], [out])
            return out[:lens[0]]
        return u[mask]
    u = np.ascontiguousarray(_np_u64(u))
    m = np.ascontiguousarray(np.asarray(mask, dtype=bool).view(np.uint8))
    if engine is not None:
        from dgraph_amd._lib import _u8p
        n = _u64()
        check(lib().ua_apply_filter(engine._ctx, _hptr(u), u.size,
                                    m.ctypes.data_as(_u8p), C.byref(n)))
        return u[:n.value]
    return u[m.view(bool)]
