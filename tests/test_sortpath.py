"""Sort-path parity: the GPU orchestration
(dgraph_amd.sortpath) vs the oracle restatement (oracle/sortref.py) of
worker/sort.go:139-375, on seeded UidMatrix + synthetic index buckets.

CPU tests pin the oracle restatement's own invariants (PageRange table from
x/x.go:815, bucket bookkeeping properties); GPU tests are the parity proper.
"""
import numpy as np
import pytest

from oracle import sortref

SEED = 0xD6A77


# ---------------- CPU: the oracle restatement itself ----------------

# x.PageRange behavior transcribed from x/x.go:815-843
PAGE_RANGE_TABLE = [
    # (count, offset, n) -> (start, end)
    ((0, 0, 0), (0, 0)),
    ((0, 0, 10), (0, 10)),
    ((5, 0, 10), (0, 5)),
    ((5, 8, 10), (8, 10)),
    ((5, 20, 10), (10, 10)),
    ((0, 3, 10), (3, 10)),
    ((-3, 0, 10), (7, 10)),     # negative count: from the back
    ((-15, 0, 10), (0, 10)),    # clamped to -n
    ((5, -2, 10), (0, 5)),      # negative offset clamps to 0
]


@pytest.mark.parametrize("args,want", PAGE_RANGE_TABLE)
def test_page_range(args, want):
    assert sortref.page_range(*args) == want
    from dgraph_amd import sortpath
    assert sortpath.page_range(*args) == want


def _mk_matrix(rng, n_rows, max_len, limit):
    rows = []
    for _ in range(n_rows):
        ln = int(rng.integers(0, max_len))
        rows.append(np.sort(rng.choice(limit, size=ln, replace=False))
                    .astype(np.uint64) if ln else np.empty(0, dtype=np.uint64))
    return rows


def _mk_keys(rng, rows, null_frac=0.2, nkeys=50):
    keys = {}
    for row in rows:
        for u in row:
            u = int(u)
            if u not in keys:
                keys[u] = None if rng.random() < null_frac \
                    else int(rng.integers(0, nkeys))
    return keys


def _mk_buckets(keys, nkeys):
    """Index buckets: bucket k = sorted uids whose key == k, in key (token)
    order — what the Badger index iterator supplies (sort.go:278)."""
    buckets = [[] for _ in range(nkeys)]
    for u, k in keys.items():
        if k is not None:
            buckets[k].append(u)
    return [np.array(sorted(b), dtype=np.uint64) for b in buckets if b]


def test_oracle_sort_without_index_basic():
    rows, ms = sortref.sort_without_index(
        [[10, 20, 30, 40]], {10: 3, 20: 1, 30: 2, 40: None}.get,
        offset=0, count=0)
    assert rows[0] == [20, 30, 10, 40]  # by key, null last
    assert ms == []
    rows, _ = sortref.sort_without_index(
        [[10, 20, 30, 40]], {10: 3, 20: 1, 30: 2, 40: None}.get,
        offset=1, count=2)
    assert rows[0] == [30, 10]
    rows, _ = sortref.sort_without_index(
        [[10, 20, 30]], {10: 3, 20: 1, 30: 2}.get, offset=0, count=0,
        desc=True)
    assert rows[0] == [10, 30, 20]


def test_oracle_sort_without_index_multi_extension():
    # equal keys at the page edge are kept in multi mode (sort.go:745-768)
    keys = {1: 5, 2: 5, 3: 5, 4: 7, 5: 8}.get
    rows, ms = sortref.sort_without_index([[1, 2, 3, 4, 5]], keys,
                                          offset=2, count=2, multi=True)
    # start backs up over the equal-5 run: start 2 -> 0; end 4 stays (7 != 8)
    assert rows[0] == [1, 2, 3, 4]
    assert ms == [2]  # remaining offset applied after multi sort


def test_oracle_sort_with_index_equals_without_on_full_pages():
    """With enough count to cover everything and offset 0, the index path's
    rows (bucket order = key order, nulls appended) must equal the
    no-index path's sorted rows."""
    rng = np.random.default_rng(SEED)
    rows_in = _mk_matrix(rng, 8, 60, 500)
    keys = _mk_keys(rng, rows_in)
    buckets = _mk_buckets(keys, 50)
    big = 10_000
    want, _ = sortref.sort_without_index(rows_in, keys.get, 0, big)
    got = sortref.sort_with_index(rows_in, buckets, 0, big)
    for w, g in zip(want, got):
        assert list(g) == list(w)


def test_oracle_sort_with_index_offset_count_window():
    """The index path with (offset, count) returns exactly the no-index
    path's page for single-order queries (valued uids; nulls fill the
    tail only when the valued page runs short)."""
    rng = np.random.default_rng(SEED + 1)
    rows_in = _mk_matrix(rng, 10, 80, 400)
    keys = _mk_keys(rng, rows_in, null_frac=0.3)
    buckets = _mk_buckets(keys, 50)
    for offset, count in [(0, 5), (3, 4), (10, 7), (0, 1), (100, 3)]:
        want, _ = sortref.sort_without_index(rows_in, keys.get, offset, count)
        got = sortref.sort_with_index(rows_in, buckets, offset, count)
        for w, g in zip(want, got):
            assert list(g) == list(w), (offset, count)


def test_oracle_dest_uids():
    assert sortref.dest_uids([[3, 1], [2, 3], []]) == [1, 2, 3]


# ---------------- GPU: product orchestration vs oracle ----------------

@pytest.mark.gpu
class TestSortPathGPU:
    @pytest.fixture(scope="class")
    def eng(self):
        from dgraph_amd import algo
        e = algo.Engine(0)
        yield e
        e.close()

    @pytest.mark.parametrize("desc", [False, True])
    @pytest.mark.parametrize("offset,count,multi", [
        (0, 0, False), (0, 5, False), (7, 9, False), (3, 0, False),
        (2, 2, True), (0, 4, True)])
    def test_sort_without_index_parity(self, eng, offset, count, multi, desc):
        from dgraph_amd import sortpath
        rng = np.random.default_rng(SEED + offset * 31 + count)
        rows_in = _mk_matrix(rng, 12, 3000, 100_000)
        keys = _mk_keys(rng, rows_in, nkeys=200)
        want, wms = sortref.sort_without_index(
            rows_in, keys.get, offset, count, desc=desc, multi=multi)
        got, gms = sortpath.sort_without_index(
            eng, rows_in, keys.get, offset, count, desc=desc, multi=multi)
        assert gms == wms
        for w, g in zip(want, got):
            assert g.tolist() == [int(x) for x in w]

    @pytest.mark.parametrize("offset,count,multi", [
        (0, 5, False), (4, 6, False), (11, 2, False), (0, 1, False),
        (2, 3, True)])
    def test_sort_with_index_parity(self, eng, offset, count, multi):
        from dgraph_amd import sortpath
        rng = np.random.default_rng(SEED + offset * 17 + count)
        rows_in = _mk_matrix(rng, 10, 2000, 50_000)
        keys = _mk_keys(rng, rows_in, null_frac=0.25, nkeys=64)
        buckets = _mk_buckets(keys, 64)
        want = sortref.sort_with_index(rows_in, buckets, offset, count,
                                       multi=multi)
        got = sortpath.sort_with_index(eng, rows_in, buckets, offset, count,
                                       multi=multi)
        for w, g in zip(want, got):
            assert g.tolist() == [int(x) for x in w]

    def test_dest_uids_parity(self, eng):
        from dgraph_amd import sortpath
        rng = np.random.default_rng(SEED)
        rows = [rng.permutation(rng.choice(10_000, size=200, replace=False))
                .astype(np.uint64) for _ in range(6)]
        want = sortref.dest_uids([r.tolist() for r in rows])
        got = sortpath.dest_uids(eng, rows)
        assert got.tolist() == want
