"""Oracle restatement of the worker/sort.go sort path.

TEST INFRASTRUCTURE ONLY — like oracle.c, this module is the parity checker
for dgraph_amd.sortpath; nothing in the product path imports it.

Restates, file-for-file (all /root/reference/worker/sort.go unless noted):
 - sortWithoutIndex       sort.go:139-189  (per-row sort-by-value + paginate)
 - sortByValue            sort.go:775-821  (nulls appended after valued uids)
 - paginate               sort.go:740-772  (multi-sort equal-value extension)
 - x.PageRange            x/x.go:815-843
 - sortWithIndex          sort.go:189-375  (bucket loop + nullNodes tail)
 - intersectBucket        sort.go:606-719  (offset/skipped/count bookkeeping)
 - removeDuplicates       sort.go:726-738

Engine-level boundary (what stays upstream, SURVEY.md §2 OOS): schema/
tokenizer/Badger/types.  Sort keys arrive as uint32 ranks (what a sortable
index tokenizer produces: order-preserving byte strings ranked per bucket),
uids as uint64; `keys_of` maps uid -> key or None (null value, sort.go:790).
Tie order among equal keys is UNSPECIFIED in the reference (types.Sort is
sort.Sort, types/sort.go:135 — unstable); both restatements refine it to
(key, uid-ascending), which is inside the reference's contract.
"""


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
    start = offset
    if start < 0:
        start = 0
    if start > n:
        return n, n
    if count == 0:
        return start, n
    end = start + count
    if end > n:
        end = n
    return start, end


def _sort_by_value(uids, keys_of, desc):
    """sortByValue (sort.go:775): valued uids sorted by key, nulls appended
    in input order.  Returns (sorted_uids, sorted_keys_with_None_tail)."""
    valued = [(keys_of(u), u) for u in uids if keys_of(u) is not None]
    nulls = [u for u in uids if keys_of(u) is None]
    valued.sort(key=lambda kv: ((-kv[0] if desc else kv[0]), kv[1]))
    out_uids = [u for _, u in valued] + nulls
    out_keys = [k for k, _ in valued] + [None] * len(nulls)
    return out_uids, out_keys


def _paginate(count, offset, uids, keys, multi):
    """paginate (sort.go:740): PageRange + multi-sort equal-value extension."""
    start, end = page_range(count, offset, len(uids))
    if multi:
        while 0 < start < len(keys) and keys[start] == keys[start - 1]:
            start -= 1
        while end < len(uids) and keys[end - 1] == keys[end]:
            end += 1
    return start, end


def sort_without_index(uid_matrix, keys_of, offset, count, desc=False,
                       multi=False):
    """sortWithoutIndex (sort.go:139-189).  Returns (rows, multi_sort_offsets)
    — multi_sort_offsets is [] unless multi (len(ts.Order) > 1, :168-176)."""
    rows = []
    ms_offsets = []
    for ul in uid_matrix:
        uids, keys = _sort_by_value(list(ul), keys_of, desc)
        start, end = _paginate(count, offset, uids, keys, multi)
        if multi:
            ms_offsets.append(offset - start if start < offset else 0)
        rows.append(uids[start:end])
    return rows, ms_offsets


def sort_with_index(uid_matrix, buckets, offset, count, multi=False):
    """sortWithIndex (sort.go:189-375) over pre-supplied index buckets
    (token-ordered sorted uid lists — what the Badger iterator + pl.Uids
    deliver upstream of the boundary).  Single-language (no cross-bucket
    duplicates beyond what uset removes).  Returns rows (UidMatrix out)."""
    n = len(uid_matrix)
    # out[i] = intersectedList (sort.go:597-604)
    ulist = [[] for _ in range(n)]
    skipped = [[] for _ in range(n)]
    uset = [set() for _ in range(n)]
    offs = [offset] * n
    ms_off = [0] * n

    # The reference's sortWithIndex is only reached with a positive count
    # (its count==0 path trips x.AssertTruef at sort.go:716).
    assert count > 0
    for bucket in buckets:  # BUCKETS loop, sort.go:278
        bset = set(bucket)
        for i, ul in enumerate(uid_matrix):  # intersectBucket, sort.go:606
            if len(ulist[i]) - ms_off[i] >= count:
                continue
            result = [u for u in ul if u in bset]  # pl.Uids(Intersect), :641
            # removeDuplicates (:648, :726)
            result = [u for u in result if u not in uset[i]]
            uset[i].update(result)
            nn = len(result)
            if offs[i] >= nn:  # skip whole intersection (:652-658)
                offs[i] -= nn
                skipped[i].extend(result)
                continue
            # within the page (:663+); bucket-internal sortByValue is a
            # no-op at this boundary (single lang, all values == token)
            if offs[i] > 0:  # apply offset (:672-689)
                if not multi:
                    skipped[i].extend(result[:offs[i]])
                    result = result[offs[i]:]
                else:
                    ms_off[i] = offs[i]
                offs[i] = 0
                nn = len(result)
            if not multi:  # count slack (:694-699)
                slack = count - len(ulist[i])
                if slack < nn:
                    nn = slack
            ulist[i].extend(result[:nn])
        # errContinue/errDone (:709-718)
        if all(len(ulist[i]) - ms_off[i] >= count for i in range(n)):
            break

    # nullNodes tail (sort.go:320-357)
    rows = []
    for i, ul in enumerate(uid_matrix):
        present = set(ulist[i]) | set(skipped[i])
        null_nodes = [u for u in ul if u not in present]
        if offs[i] < len(null_nodes):
            if offs[i] >= 0:
                null_nodes = null_nodes[offs[i]:]
        else:
            null_nodes = []
        remaining = count - len(ulist[i])
        # sort.go:349 casts remainingCount to uint64 before x.Min: a negative
        # remainder (possible in the multi case, where ulist may exceed
        # count) wraps and appends ALL nullNodes — restated bug-compatibly.
        if remaining < 0:
            can_append = len(null_nodes)
        else:
            can_append = min(remaining, len(null_nodes))
        rows.append(list(ulist[i]) + null_nodes[:can_append])
    return rows


def dest_uids(rows):
    """destUids (sort.go:566-580): dedup'd sorted union of the result rows
    (the SrcUids of multiSort's follow-up value fetches)."""
    return sorted(set(u for row in rows for u in row))
