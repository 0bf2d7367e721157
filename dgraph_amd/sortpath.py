"""GPU orchestration of the worker/sort.go sort path.

The product-side restatement of the reference's two sort strategies over the
uidalgo engine — every heavy step runs on the GPU through the C-ABI:

 - sort_without_index (worker/sort.go:139-189): per-row sort-by-value becomes
   ONE batched segmented sort (`ua_sort_segments_dev`) over packed
   (key32 << 32 | row_position) u64 keys; pagination (paginate :740,
   x.PageRange x/x.go:815) is O(log) host logic on the sorted rows.
 - sort_with_index (worker/sort.go:189-375): each index bucket intersects
   with every still-unfilled UidMatrix row as ONE batched intersect grid
   (`ua_intersect_batch_dev` — the engine form of pl.Uids(Intersect),
   sort.go:641); removeDuplicates/offset/count bookkeeping (intersectBucket
   :606-719) is host logic over the downloaded lengths; the nullNodes tail
   (:320-357) runs as a batched segmented sort of present-uids plus a
   batched Difference.
 - dest_uids (sort.go:566): MergeSorted over the result rows (the engine's
   dedup union IS destUids' sorted dedup'd set).

Boundary contract (what stays upstream, SURVEY.md §2 OOS: schema, tokenizers,
Badger, types): sort keys arrive as uint32 ranks, `keys_of(uid)` -> key or
None; index buckets arrive as token-ordered sorted uid tensors.  Tie order
among equal keys follows the (key, uid) refinement — the reference leaves it
unspecified (types.Sort is unstable sort.Sort, types/sort.go:135).

Parity: oracle/sortref.py restates the same reference lines in pure Python;
tests/test_sortpath.py compares the two on seeded inputs.
"""
import numpy as np


def page_range(count, offset, n):
    """x.PageRange (x/x.go:815)."""
    if n == 0:
        return 0, 0
    if count == 0 and offset == 0:
        return 0, n
    if count < 0:
        if -count > n:
            count = -n
        return (((n + count) % n) + n) % n, n
    start = min(max(offset, 0), n)
    if count == 0:
        return start, n
    return start, min(start + count, n)


def _to_dev(a, device):
    import torch
    a = np.ascontiguousarray(a, dtype=np.uint64)
    if a.size == 0:
        return torch.empty(0, dtype=torch.int64, device=device)
    return torch.from_numpy(a.view(np.int64)).to(device)


def sort_without_index(eng, uid_matrix, keys_of, offset, count, desc=False,
                       multi=False, device="cuda:0"):
    """sortWithoutIndex (sort.go:139-189) on the GPU.

    uid_matrix: list of numpy uint64 rows (sorted, duplicate-free);
    keys_of(uid) -> uint32 key or None.  Returns (rows, multi_sort_offsets)
    as numpy arrays / ints."""
    # sortByValue (:775): valued uids sort by key; nulls append in input
    # order.  Pack (key << 32 | position) so ONE u64 segmented sort orders
    # every row at once; desc flips the key bits (keys are u32).
    packed_rows, null_tails, valued_uids = [], [], []
    for ul in uid_matrix:
        keys = np.array([keys_of(int(u)) for u in ul], dtype=object)
        has = np.array([k is not None for k in keys], dtype=bool)
        vk = keys[has].astype(np.uint64)
        if desc:
            vk = np.uint64(0xFFFFFFFF) - vk
        pos = np.arange(len(ul), dtype=np.uint64)[has]
        packed_rows.append((vk << np.uint64(32)) | pos)
        null_tails.append(np.asarray(ul, dtype=np.uint64)[~has])
        valued_uids.append(np.asarray(ul, dtype=np.uint64))
    dev_rows = [_to_dev(p, device) for p in packed_rows]
    eng.sort_segments(dev_rows)  # ua_sort_segments_dev: one batched grid
    rows, ms_offsets = [], []
    for i, dr in enumerate(dev_rows):
        packed = dr.cpu().numpy().view(np.uint64)
        pos = (packed & np.uint64(0xFFFFFFFF)).astype(np.int64)
        uids = np.concatenate([valued_uids[i][pos], null_tails[i]])
        keyvals = list(packed >> np.uint64(32)) + [None] * len(null_tails[i])
        # paginate (:740): PageRange + multi equal-value extension
        start, end = page_range(count, offset, len(uids))
        if multi:
            while 0 < start < len(keyvals) and keyvals[start] == keyvals[start - 1]:
                start -= 1
            while end < len(uids) and keyvals[end - 1] == keyvals[end]:
                end += 1
            ms_offsets.append(offset - start if start < offset else 0)
        rows.append(uids[start:end])
    return rows, ms_offsets


def sort_with_index(eng, uid_matrix, buckets, offset, count, multi=False,
                    device="cuda:0"):
    """sortWithIndex (sort.go:189-375) on the GPU over pre-supplied index
    buckets (token-ordered sorted uid tensors/arrays).  count must be
    positive (the reference's count==0 path trips x.AssertTruef :716).
    Returns rows as numpy uint64 arrays."""
    assert count > 0
    n = len(uid_matrix)
    rows_dev = [_to_dev(ul, device) for ul in uid_matrix]
    buckets_dev = [_to_dev(b, device) for b in buckets]
    ulist = [[] for _ in range(n)]
    skipped = [[] for _ in range(n)]
    uset = [set() for _ in range(n)]
    offs = [offset] * n
    ms_off = [0] * n

    for bd in buckets_dev:  # BUCKETS loop (:278)
        active = [i for i in range(n)
                  if len(ulist[i]) - ms_off[i] < count]
        if not active:
            break
        # pl.Uids(Intersect: ul) for every active row x this bucket — one
        # batched grid (intersectBucket :641)
        outs, lens = eng.intersect_pairs([bd] * len(active),
                                         [rows_dev[i] for i in active])
        for j, i in enumerate(active):
            result = outs[j][:lens[j]].cpu().numpy().view(np.uint64).tolist()
            # removeDuplicates (:648, :726)
            result = [u for u in result if u not in uset[i]]
            uset[i].update(result)
            nn = len(result)
            if offs[i] >= nn:  # skip whole intersection (:652)
                offs[i] -= nn
                skipped[i].extend(result)
                continue
            if offs[i] > 0:  # apply offset (:672)
                if not multi:
                    skipped[i].extend(result[:offs[i]])
                    result = result[offs[i]:]
                else:
                    ms_off[i] = offs[i]
                offs[i] = 0
                nn = len(result)
            if not multi:  # count slack (:694)
                slack = count - len(ulist[i])
                if slack < nn:
                    nn = slack
            ulist[i].extend(result[:nn])
        if all(len(ulist[i]) - ms_off[i] >= count for i in range(n)):
            break  # errDone (:709-718)

    # nullNodes tail (:320-357): present = ulist ∪ skipped (disjoint by
    # uset); GPU-sort the present sets, then batched Difference preserves
    # ul's (sorted) order exactly like the reference's in-order scan.
    present_dev = []
    for i in range(n):
        present_dev.append(_to_dev(
            np.array(ulist[i] + skipped[i], dtype=np.uint64), device))
    eng.sort_segments(present_dev)
    null_outs, null_lens = eng.difference_pairs(rows_dev, present_dev)
    rows = []
    for i in range(n):
        null_nodes = null_outs[i][:null_lens[i]].cpu().numpy().view(np.uint64)
        if offs[i] < len(null_nodes):
            if offs[i] >= 0:
                null_nodes = null_nodes[offs[i]:]
        else:
            null_nodes = null_nodes[:0]
        remaining = count - len(ulist[i])
        # bug-compatible with sort.go:349's uint64 cast of a negative
        # remainder (multi case): appends ALL nullNodes
        can_append = len(null_nodes) if remaining < 0 else \
            min(remaining, len(null_nodes))
        rows.append(np.array(ulist[i] + list(null_nodes[:can_append]),
                             dtype=np.uint64))
    return rows


def dest_uids(eng, rows, device="cuda:0"):
    """destUids (sort.go:566): sorted dedup'd union of the result rows ==
    the engine's MergeSorted over per-row GPU sorts (rows arrive in bucket
    order, not sorted)."""
    devs = [_to_dev(np.asarray(r, dtype=np.uint64), device) for r in rows]
    eng.sort_segments(devs)
    return eng.merge_sorted(devs).cpu().numpy().view(np.uint64)
