"""GPU parity: the HIP engine vs the CPU oracle (bit-exact restatement of the
Go reference) on the same seeded inputs.  The bar is BIT-EXACT equality on
duplicate-free sorted inputs — the parity domain the reference itself pins
(uidlist_test.go:394, :536-542).

Everything here goes through the C-ABI of libuidalgo.so (the product path);
the oracle appears only as the checker.
"""
import numpy as np
import pytest
import torch

from dgraph_amd import algo, synth
from oracle import bind as orc

pytestmark = pytest.mark.gpu

SEED = synth.SEED


@pytest.fixture(scope="module")
def eng():
    e = algo.Engine(0)
    yield e
    e.close()


def to_dev(a):
    a = np.ascontiguousarray(a, dtype=np.uint64)
    if a.size == 0:
        return torch.empty(0, dtype=torch.int64, device="cuda:0")
    return torch.from_numpy(a.view(np.int64)).to("cuda:0")


def to_np(t):
    return t.cpu().numpy().view(np.uint64)


# ---------- golden duplicate-free cases through the host-convenience API ----------

GOLDEN_INTERSECT = [
    ([1, 2, 3], [], []),
    ([1, 2, 3], [1, 2, 3, 4, 5], [1, 2, 3]),
    ([1, 2, 3], [2], [2]),
    ([1, 2, 3], [0, 5], []),
    ([1, 2, 3], [3, 5], [3]),
]


@pytest.mark.parametrize("u,v,want", GOLDEN_INTERSECT)
def test_golden_intersect_host_api(eng, u, v, want):
    got = algo.intersect_with(eng, u, v)
    assert got.tolist() == want


GOLDEN_DIFF = [
    ([1, 2, 3], [1], [2, 3]),
    ([1, 2, 3], [2], [1, 3]),
    ([1, 2, 3], [3], [1, 2]),
    ([1, 2, 3], [], [1, 2, 3]),
    ([], [1, 2], []),
    ([1, 2, 3], [2, 3, 4, 5], [1]),
    ([10, 12, 13], [2, 3, 4, 13], [10, 12]),
]


@pytest.mark.parametrize("u,v,want", GOLDEN_DIFF)
def test_golden_difference_host_api(eng, u, v, want):
    got = algo.difference(eng, u, v)
    assert got.tolist() == want


GOLDEN_MERGE = [
    ([[55]], [55]),
    ([[1, 3, 6, 8, 10], [2, 4, 5, 7, 15]], [1, 2, 3, 4, 5, 6, 7, 8, 10, 15]),
    ([[1, 3, 6, 8, 10], []], [1, 3, 6, 8, 10]),
    ([[], [1, 3, 6, 8, 10]], [1, 3, 6, 8, 10]),
    ([[], []], []),
    ([[5, 6, 7], [3, 4], [1, 2], []], [1, 2, 3, 4, 5, 6, 7]),
    ([], []),
    # duplicate-input dedup tables (uidlist_test.go:65,144,151): MergeSorted
    # collapses ALL duplicates, including within one list
    ([[11, 13, 16, 18, 20], [12, 14, 15, 15, 16, 16, 17, 25], [1, 2]],
     [1, 2, 11, 12, 13, 14, 15, 16, 17, 18, 20, 25]),
    ([[1, 1, 1]], [1]),
    ([[1, 2, 3, 3, 6], [4, 8, 9]], [1, 2, 3, 4, 6, 8, 9]),
]


@pytest.mark.parametrize("lists,want", GOLDEN_MERGE)
def test_golden_merge_host_api(eng, lists, want):
    got = algo.merge_sorted(eng, lists)
    assert got.tolist() == want


GOLDEN_INTERSECT_K = [
    ([[1, 2, 3], [2, 3, 4, 5]], [2, 3]),
    ([[1, 2, 3]], [1, 2, 3]),
    ([], []),
    ([[100, 101]], [100, 101]),
    ([[1, 2, 3], [2, 3, 4, 5], [4, 5, 6]], []),
    ([[10, 12, 13], [2, 3, 4, 13], [4, 5, 6]], []),
]


@pytest.mark.parametrize("lists,want", GOLDEN_INTERSECT_K)
def test_golden_intersect_k_host_api(eng, lists, want):
    got = algo.intersect_sorted(eng, lists)
    assert got.tolist() == want


# ---------- randomized parity vs oracle, device-resident batched path ----------

SIZES = [
    (0, 0), (1, 1), (1, 1000), (10, 1_000_000),  # ratio 1e5 (the 766ns/op shape)
    (1000, 1000), (4096, 100_000), (100_000, 100_000),
    (1_000_000, 1_000_000),
]


@pytest.mark.parametrize("n,m", SIZES)
def test_intersect_pair_vs_oracle(eng, n, m):
    rng = np.random.default_rng(SEED + n * 7 + m)
    limit = max((n + m) * 3, 100)
    u = synth.gen_sorted_unique(rng, n, limit)
    v = synth.gen_sorted_unique(rng, m, limit)
    outs, lens = eng.intersect_pairs([to_dev(u)], [to_dev(v)])
    got = to_np(outs[0][:lens[0]])
    want = orc.intersect_with(u, v)
    assert got.tolist() == want.tolist()


@pytest.mark.parametrize("n,m", SIZES)
def test_difference_pair_vs_oracle(eng, n, m):
    rng = np.random.default_rng(SEED + n * 13 + m)
    limit = max((n + m) * 3, 100)
    u = synth.gen_sorted_unique(rng, n, limit)
    v = synth.gen_sorted_unique(rng, m, limit)
    outs, lens = eng.difference_pairs([to_dev(u)], [to_dev(v)])
    got = to_np(outs[0][:lens[0]])
    assert got.tolist() == orc.difference(u, v).tolist()


@pytest.mark.parametrize("n,m", SIZES)
def test_merge_pair_vs_oracle(eng, n, m):
    rng = np.random.default_rng(SEED + n * 17 + m)
    limit = max((n + m) * 3, 100)
    u = synth.gen_sorted_unique(rng, n, limit)
    v = synth.gen_sorted_unique(rng, m, limit)
    outs, lens = eng.merge_pairs([to_dev(u)], [to_dev(v)])
    got = to_np(outs[0][:lens[0]])
    assert got.tolist() == orc.merge_sorted([u, v]).tolist()


@pytest.mark.parametrize("n,m", [(100, 100), (5000, 3000), (200_000, 100_000)])
def test_merge_with_duplicates_vs_oracle(eng, n, m):
    """MergeSorted's dedup is reference-pinned ON duplicate inputs
    (uidlist_test.go:65,144,151) — sorted-with-dups lists, k-way + pairwise."""
    rng = np.random.default_rng(SEED + n)
    u = np.sort(rng.integers(0, (n + m) // 2, size=n, dtype=np.uint64))
    v = np.sort(rng.integers(0, (n + m) // 2, size=m, dtype=np.uint64))
    w = np.sort(rng.integers(0, (n + m) // 4, size=m // 2, dtype=np.uint64))
    outs, lens = eng.merge_pairs([to_dev(u)], [to_dev(v)])
    assert to_np(outs[0][:lens[0]]).tolist() == orc.merge_sorted([u, v]).tolist()
    got_k = to_np(eng.merge_sorted([to_dev(u), to_dev(v), to_dev(w)]))
    assert got_k.tolist() == orc.merge_sorted([u, v, w]).tolist()
    got_1 = to_np(eng.merge_sorted([to_dev(u)]))
    assert got_1.tolist() == orc.merge_sorted([u]).tolist()


def test_cfg2_planted_overlap(eng):
    """cfg 2: 1M x 1M, 1% planted overlap — exact output known by construction."""
    rng = np.random.default_rng(SEED)
    u, v, common = synth.gen_pair(rng, 1_000_000, 1_000_000, 10_000, 100_000_000)
    outs, lens = eng.intersect_pairs([to_dev(u)], [to_dev(v)])
    got = to_np(outs[0][:lens[0]])
    assert lens[0] == common.size
    assert np.array_equal(got, common)
    # and vs oracle
    assert np.array_equal(got, orc.intersect_with(u, v))


def test_batched_zipf_vs_oracle(eng):
    """cfg 3 (scaled): 96 Zipf-sized pairs in ONE grid, each bit-exact."""
    rng = np.random.default_rng(SEED + 3)
    sizes = synth.zipf_sizes(rng, 96, lo=100, hi=200_000)
    us, vs, wants = [], [], []
    for i, sz in enumerate(sizes):
        n = int(sz)
        m = int(max(100, n // (10 ** int(rng.integers(0, 3)))))
        u = synth.gen_sorted_unique(rng, n, 3 * (n + m))
        v = synth.gen_sorted_unique(rng, m, 3 * (n + m))
        us.append(u)
        vs.append(v)
    d_us = [to_dev(x) for x in us]
    d_vs = [to_dev(x) for x in vs]
    outs, lens = eng.intersect_pairs(d_us, d_vs)
    wants = orc.intersect_batch_cpu(us, vs)
    for i in range(len(us)):
        assert to_np(outs[i][:lens[i]]).tolist() == wants[i].tolist(), f"pair {i}"
    # merge + difference over the same batch
    m_outs, m_lens = eng.merge_pairs(d_us, d_vs)
    d_outs, d_lens = eng.difference_pairs(d_us, d_vs)
    for i in range(len(us)):
        assert to_np(m_outs[i][:m_lens[i]]).tolist() == \
            orc.merge_sorted([us[i], vs[i]]).tolist(), f"merge pair {i}"
        assert to_np(d_outs[i][:d_lens[i]]).tolist() == \
            orc.difference(us[i], vs[i]).tolist(), f"diff pair {i}"


def test_prepared_batch_vs_oracle(eng):
    """ua_batch: partition cached at create; repeated runs of all three ops
    over one prepared batch are bit-exact and stable across runs."""
    import torch
    rng = np.random.default_rng(SEED + 77)
    us, vs = [], []
    for _ in range(32):
        n = int(rng.integers(0, 50_000))
        m = int(rng.integers(0, 50_000))
        us.append(synth.gen_sorted_unique(rng, n, 3 * (n + m) + 10))
        vs.append(synth.gen_sorted_unique(rng, m, 3 * (n + m) + 10))
    d_us = [to_dev(x) for x in us]
    d_vs = [to_dev(x) for x in vs]
    # out capacity n+m fits every op
    outs = [torch.empty(max(u.numel() + v.numel(), 1), dtype=torch.int64,
                        device="cuda:0") for u, v in zip(d_us, d_vs)]
    batch = eng.make_batch(d_us, d_vs, outs)
    for op, ref in [(algo.OP_INTERSECT, orc.intersect_with),
                    (algo.OP_DIFFERENCE, orc.difference),
                    (algo.OP_MERGE, lambda u, v: orc.merge_sorted([u, v])),
                    (algo.OP_INTERSECT, orc.intersect_with)]:  # re-run intersect
        lens = batch.run(op)
        for i in range(len(us)):
            got = to_np(outs[i][:lens[i]])
            assert got.tolist() == ref(us[i], vs[i]).tolist(), f"op={op} pair={i}"
    batch.close()


@pytest.mark.parametrize("k", [1, 2, 3, 8, 150])
def test_kway_vs_oracle(eng, k):
    rng = np.random.default_rng(SEED + k)
    lists = [synth.gen_sorted_unique(rng, int(rng.integers(0, 20_000)), 60_000)
             for _ in range(k)]
    d_lists = [to_dev(x) for x in lists]
    got_m = to_np(eng.merge_sorted(d_lists))
    assert got_m.tolist() == orc.merge_sorted(lists).tolist()
    got_i = to_np(eng.intersect_sorted(d_lists))
    assert got_i.tolist() == orc.intersect_sorted(lists).tolist()


def test_merge_all_pairs(eng):
    """Duplicate-keeping merge of sorted runs (the segmented-sort building
    block): equals np.sort of the concatenation."""
    rng = np.random.default_rng(SEED + 5)
    us, vs = [], []
    for _ in range(16):
        n, m = int(rng.integers(0, 30_000)), int(rng.integers(0, 30_000))
        us.append(np.sort(rng.integers(0, 40_000, size=n, dtype=np.uint64)))
        vs.append(np.sort(rng.integers(0, 40_000, size=m, dtype=np.uint64)))
    outs, lens = eng.merge_all_pairs([to_dev(u) for u in us], [to_dev(v) for v in vs])
    for i in range(16):
        assert lens[i] == us[i].size + vs[i].size
        got = to_np(outs[i][:lens[i]])
        want = np.sort(np.concatenate([us[i], vs[i]]))
        assert np.array_equal(got, want), f"pair {i}"


@pytest.mark.parametrize("sizes", [
    [0, 1, 5, 2047, 2048, 2049],
    [100_000, 3, 1_000_000],
    [65_536] * 8,
])
def test_sort_segments(eng, sizes):
    """Segmented sort (bitonic chunks + merge-all tree) vs np.sort — with
    duplicates, u64-extreme values, in place."""
    rng = np.random.default_rng(SEED + sum(sizes))
    arrs = []
    for n in sizes:
        a = rng.integers(0, max(n, 10) * 2, size=n, dtype=np.uint64)
        if n > 10:  # sprinkle extremes and duplicates
            a[:: max(n // 7, 1)] = np.uint64(2**64 - 1)
            a[1:: max(n // 5, 1)] = a[0]
        arrs.append(a)
    tensors = [to_dev(a) for a in arrs]
    eng.sort_segments(tensors)
    for a, t in zip(arrs, tensors):
        assert np.array_equal(to_np(t), np.sort(a))


def test_trigram_term_combinators(eng):
    """worker/trigram.go:56-96 + needsIntersect allof/anyof (task.go:303):
    the index-token combinators are exactly IntersectSorted (allof) and
    MergeSorted (anyof) over token posting lists — §8f row 4, no new
    kernels.  8 token lists of skewed sizes, bit-exact vs oracle."""
    rng = np.random.default_rng(SEED + 888)
    token_lists = [synth.gen_sorted_unique(rng, int(sz), 500_000)
                   for sz in [120_000, 90_000, 60_000, 30_000, 8_000, 2_000, 500, 50]]
    d_lists = [to_dev(t) for t in token_lists]
    allof = to_np(eng.intersect_sorted(d_lists))       # "allof" fold
    anyof = to_np(eng.merge_sorted(d_lists))           # "anyof" union
    assert allof.tolist() == orc.intersect_sorted(token_lists).tolist()
    assert anyof.tolist() == orc.merge_sorted(token_lists).tolist()
    # uids ∩ filtered-list shape (trigram.go:87: result ∩ original uid list)
    uids = synth.gen_sorted_unique(rng, 40_000, 500_000)
    outs, lens = eng.intersect_pairs([to_dev(uids)], [to_dev(anyof)])
    assert to_np(outs[0][:lens[0]]).tolist() == \
        orc.intersect_with(uids, anyof).tolist()


def test_index_of_batch_vs_oracle(eng):
    rng = np.random.default_rng(SEED)
    u = synth.gen_sorted_unique(rng, 100_000, 1_000_000)
    q = np.concatenate([u[::37], rng.integers(0, 1_000_000, 1000, dtype=np.uint64)])
    got = eng.index_of_batch(to_dev(u), to_dev(q)).cpu().numpy()
    want = [orc.index_of(u, int(x)) for x in q]
    assert got.tolist() == want


def test_randomized_small_batch_fuzz(eng):
    """512 random small/edge-shaped pairs in ONE batch per op, bit-exact vs
    the oracle — dense coverage of tile-boundary and shape edge cases."""
    rng = np.random.default_rng(SEED + 1234)
    us, vs = [], []
    for i in range(512):
        shape = i % 8
        if shape == 0:
            n, m = 0, int(rng.integers(0, 50))
        elif shape == 1:
            n, m = int(rng.integers(0, 50)), 0
        elif shape == 2:
            n, m = 1, int(rng.integers(1, 3000))
        elif shape == 3:  # straddles one tile boundary
            n, m = 2048, int(rng.integers(1, 64))
        elif shape == 4:  # exactly multiple tiles
            n, m = 2048, 2048
        else:
            n, m = int(rng.integers(1, 4000)), int(rng.integers(1, 4000))
        lim = max(3 * (n + m), 16)
        us.append(synth.gen_sorted_unique(rng, n, lim))
        vs.append(synth.gen_sorted_unique(rng, m, lim))
    d_us = [to_dev(x) for x in us]
    d_vs = [to_dev(x) for x in vs]
    i_o, i_l = eng.intersect_pairs(d_us, d_vs)
    m_o, m_l = eng.merge_pairs(d_us, d_vs)
    d_o, d_l = eng.difference_pairs(d_us, d_vs)
    for i in range(512):
        assert to_np(i_o[i][:i_l[i]]).tolist() == \
            orc.intersect_with(us[i], vs[i]).tolist(), f"intersect {i}"
        assert to_np(m_o[i][:m_l[i]]).tolist() == \
            orc.merge_sorted([us[i], vs[i]]).tolist(), f"merge {i}"
        assert to_np(d_o[i][:d_l[i]]).tolist() == \
            orc.difference(us[i], vs[i]).tolist(), f"diff {i}"


def test_extreme_values(eng):
    """Boundary values incl. 2^63 crossing and UINT64_MAX (u64 compare, not i64)."""
    u = np.array([0, 1, 2**32, 2**63 - 1, 2**63, 2**64 - 2, 2**64 - 1], dtype=np.uint64)
    v = np.array([1, 2**63, 2**64 - 1], dtype=np.uint64)
    outs, lens = eng.intersect_pairs([to_dev(u)], [to_dev(v)])
    assert to_np(outs[0][:lens[0]]).tolist() == orc.intersect_with(u, v).tolist()
    m_outs, m_lens = eng.merge_pairs([to_dev(u)], [to_dev(v)])
    assert to_np(m_outs[0][:m_lens[0]]).tolist() == orc.merge_sorted([u, v]).tolist()
    d_outs, d_lens = eng.difference_pairs([to_dev(u)], [to_dev(v)])
    assert to_np(d_outs[0][:d_lens[0]]).tolist() == orc.difference(u, v).tolist()


# ---------- packed (codec) path ----------

@pytest.mark.parametrize("size,bs,m", [
    (0, 256, 100), (1, 256, 100), (300, 10, 300), (5000, 0, 500),
    (200_000, 256, 50_000), (2_000_000, 256, 1_000_000),
])
def test_packed_intersect_vs_oracle(eng, size, bs, m):
    rng = np.random.default_rng(SEED + size + bs)
    pack_uids = synth.getuids_geometric(rng, max(size, 1))[:size]
    pack_uids = np.unique(pack_uids)  # duplicate-free parity domain
    # v: half sampled from the pack (matches), half random
    take = rng.choice(pack_uids.size, size=min(m // 2, pack_uids.size),
                      replace=False) if pack_uids.size else []
    hi = int(pack_uids[-1]) + 1000 if pack_uids.size else 1000
    v = np.unique(np.concatenate([
        pack_uids[np.sort(take)] if pack_uids.size else np.empty(0, np.uint64),
        rng.integers(0, hi, size=m // 2, dtype=np.uint64)]))

    bases, nums, offs, blob, total = algo.encode_flat(pack_uids, bs)
    if total == 0:
        return
    dp = eng.upload_pack(bases, nums, offs, blob, bs)
    got = to_np(eng.intersect_packed(dp, 0, to_dev(v)))

    opack = orc.Pack(pack_uids, bs)
    want = orc.intersect_compressed_with(opack, 0, v)
    assert got.tolist() == want.tolist()

    # decode parity (codec.Decode)
    got_dec = to_np(eng.decode_pack(dp, 0))
    assert got_dec.tolist() == opack.decode(0).tolist()

    # after/seek parity
    if pack_uids.size > 10:
        for after in [int(pack_uids[pack_uids.size // 3]),
                      int(pack_uids[pack_uids.size // 3]) + 1,
                      int(pack_uids[-1]) + 1]:
            got_a = to_np(eng.intersect_packed(dp, after, to_dev(v)))
            want_a = orc.intersect_compressed_with(opack, after, v)
            assert got_a.tolist() == want_a.tolist(), f"after={after}"
            got_d = to_np(eng.decode_pack(dp, after))
            assert got_d.tolist() == opack.decode(after).tolist()


def test_packed_32msb_splits(eng):
    """Blocks forced by 32-MSB changes (codec.go:117) incl. huge bases."""
    rng = np.random.default_rng(SEED)
    big = [0xf000000000000000, 0xf00f000000000000, 0x00f00f0000000000,
           0x000f0f0000000000, 0x0f0f0f0f00000000]
    vals = [np.uint64(rng.integers(0, 2**32)) for _ in range(50)]
    vals += [np.uint64(rng.integers(0, 2**32)) + np.uint64(big[rng.integers(0, 5)])
             for _ in range(50)]
    uids = np.unique(np.array(vals, dtype=np.uint64))
    v = uids[::3].copy()
    bases, nums, offs, blob, total = algo.encode_flat(uids, 256)
    dp = eng.upload_pack(bases, nums, offs, blob, 256)
    got = to_np(eng.intersect_packed(dp, 0, to_dev(v)))
    opack = orc.Pack(uids, 256)
    assert got.tolist() == orc.intersect_compressed_with(opack, 0, v).tolist()
    assert to_np(eng.decode_pack(dp, 0)).tolist() == uids.tolist()


@pytest.mark.parametrize("size,bs", [(0, 256), (1, 256), (5, 0), (300, 10),
                                     (5000, 1), (200_000, 256), (2_000_000, 256)])
def test_encode_dev_matches_oracle_bytes(eng, size, bs):
    """GPU codec.Encode vs the oracle encoder: byte-identical flat pack
    (blocks split on 32-MSB + block_size, group-varint deltas)."""
    rng = np.random.default_rng(SEED + size + bs)
    uids = synth.getuids_geometric(rng, max(size, 1))[:size]
    if size > 10:  # force 32-MSB splits (codec.go:117)
        uids = uids.copy()
        uids[size // 2:] += np.uint64(1) << np.uint64(33)
        uids[3 * size // 4:] += np.uint64(1) << np.uint64(41)
    dp = eng.encode_dev(to_dev(uids), bs)

    opack = orc.Pack(uids, bs)
    obases, onums, ooffs, oblob = opack.flatten()
    assert dp.bases.numel() == obases.size
    assert to_np(dp.bases).tolist() == obases.tolist()
    assert dp.num_uids.cpu().numpy().view(np.uint32).tolist() == onums.tolist()
    assert to_np(dp.delta_offs).tolist() == ooffs.tolist()
    got_blob = dp.deltas.cpu().numpy().view(np.uint8)[:oblob.size]
    assert got_blob.tolist() == oblob.tolist()
    if size:
        # and the GPU decoder inverts the GPU encoder
        assert to_np(eng.decode_pack(dp)).tolist() == uids.tolist()


def test_packed_compositions_vs_oracle(eng):
    """MergeSortedPacked / IntersectSortedPacked / ApplyFilterPacked
    (algo/packed.go:222,100,16) as engine compositions."""
    rng = np.random.default_rng(SEED)
    lists = [synth.gen_sorted_unique(rng, int(rng.integers(100, 20_000)), 60_000)
             for _ in range(5)]
    d_lists = [to_dev(x) for x in lists]

    mp = eng.merge_sorted_packed(d_lists, 256)
    want_merge = orc.merge_sorted(lists)
    assert to_np(eng.decode_pack(mp)).tolist() == want_merge.tolist()
    # byte parity of the resulting pack vs oracle-encoding the oracle merge
    ob, on, oo, obl = orc.Pack(want_merge, 256).flatten()
    assert to_np(mp.bases).tolist() == ob.tolist()
    assert mp.deltas.cpu().numpy().view(np.uint8)[:obl.size].tolist() == obl.tolist()

    dpacks = [eng.encode_dev(t, 10) for t in d_lists]
    ip = eng.intersect_sorted_packed(dpacks)
    want_i = orc.intersect_sorted(lists)
    assert to_np(eng.decode_pack(ip)).tolist() == want_i.tolist()

    fp = eng.apply_filter_packed(dpacks[0], lambda t: (t % 2) == 1)
    want_f = lists[0][lists[0] % 2 == 1]
    assert to_np(eng.decode_pack(fp)).tolist() == want_f.tolist()


def test_packed_fanout_batch_vs_oracle(eng):
    """The handleUidPostings fan-out shape (worker/task.go:834-971): many
    keys' packs intersected in ONE grid — half against one shared q.UidList,
    half against per-key lists, with per-pack after cursors."""
    rng = np.random.default_rng(SEED + 99)
    n_packs = 64
    packs_np, flat, vs_np, afters = [], [], [], []
    shared = synth.gen_sorted_unique(rng, 50_000, 2_000_000)
    d_shared = to_dev(shared)
    d_vs = []
    bss = []
    for i in range(n_packs):
        size = int(rng.integers(1, 60_000))
        uids = np.unique(synth.getuids_geometric(rng, size))
        packs_np.append(uids)
        bs = int(rng.choice([10, 64, 256]))
        bss.append(bs)
        flat.append(algo.encode_flat(uids, bs))
        if i % 2 == 0:
            vs_np.append(shared)
            d_vs.append(d_shared)
        else:
            mlen = int(rng.integers(1, 30_000))
            v = synth.gen_sorted_unique(
                rng, mlen, max(int(uids[-1]) + 1000 if uids.size else 1000, 3 * mlen))
            vs_np.append(v)
            d_vs.append(to_dev(v))
        afters.append(0 if i % 4 else int(uids[uids.size // 2]) if uids.size else 0)

    dpb = eng.upload_pack_batch(flat)
    outs, lens = eng.intersect_packed_batch(dpb, d_vs, afters=afters)
    for i in range(n_packs):
        got = to_np(outs[i][:lens[i]])
        opack = orc.Pack(packs_np[i], bss[i])
        want = orc.intersect_compressed_with(opack, afters[i], vs_np[i])
        assert got.tolist() == want.tolist(), f"pack {i}"


def test_prepared_pack_batch_vs_oneshot(eng):
    """ua_pbatch (standing query plan): repeated runs equal the one-shot
    batched call; v contents may change between runs."""
    import torch
    rng = np.random.default_rng(SEED + 321)
    packs_np, flat = [], []
    for _ in range(32):
        uids = np.unique(synth.getuids_geometric(rng, int(rng.integers(1, 40_000))))
        packs_np.append(uids)
        flat.append(algo.encode_flat(uids, 256))
    shared = synth.gen_sorted_unique(rng, 30_000, 1_500_000)
    d_shared = to_dev(shared)
    dpb = eng.upload_pack_batch(flat)

    pb = eng.make_pack_batch(dpb, [d_shared] * 32)
    ref_outs, ref_lens = eng.intersect_packed_batch(dpb, [d_shared] * 32)
    for _ in range(3):
        lens = pb.run()
        for i in range(32):
            assert lens[i] == ref_lens[i]
            assert torch.equal(pb.outs[i][:lens[i]], ref_outs[i][:ref_lens[i]])
    # mutate the shared v in place: the plan has no data-dependent cache
    shared2 = synth.gen_sorted_unique(rng, 30_000, 1_500_000)
    d_shared.copy_(torch.from_numpy(shared2.view(np.int64)).cuda())
    lens2 = pb.run()
    for i in [0, 15, 31]:
        want = orc.intersect_compressed_with(orc.Pack(packs_np[i], 256), 0, shared2)
        assert to_np(pb.outs[i][:lens2[i]]).tolist() == want.tolist()
    pb.close()


def test_host_packed_api(eng):
    """ua_intersect_packed host-pointer convenience (the cgo surface)."""
    import ctypes as Ct
    from dgraph_amd import _lib
    rng = np.random.default_rng(SEED)
    uids = synth.getuids_geometric(rng, 10_000)
    uids = np.unique(uids)
    v = uids[::7].copy()
    # build host ua_pack via ua_encode
    h = Ct.c_void_p()
    u_arr = np.ascontiguousarray(uids)
    _lib.check(_lib.lib().ua_encode(
        u_arr.ctypes.data_as(Ct.POINTER(Ct.c_uint64)), u_arr.size, 256, Ct.byref(h)))
    try:
        view = _lib.lib().ua_owned_pack_view(h)
        out = np.empty(uids.size, dtype=np.uint64)
        out_n = Ct.c_uint64()
        _lib.check(_lib.lib().ua_intersect_packed(
            eng._ctx, view, 0, v.ctypes.data_as(Ct.POINTER(Ct.c_uint64)), v.size,
            out.ctypes.data_as(Ct.POINTER(Ct.c_uint64)), Ct.byref(out_n)))
        opack = orc.Pack(uids, 256)
        assert out[:out_n.value].tolist() == \
            orc.intersect_compressed_with(opack, 0, v).tolist()
    finally:
        _lib.lib().ua_owned_pack_free(h)


# ---------- ApplyFilter across the C-ABI (uidlist.go:21; worker/task.go:1403) ----------

@pytest.mark.parametrize("n", [0, 1, 7, 2048, 2049, 100_000])
def test_apply_filter_host_abi(eng, n):
    """ua_apply_filter (host-pointer, in-place like the reference) vs oracle."""
    rng = np.random.default_rng(SEED + n)
    u = np.sort(rng.choice(np.uint64(1) << np.uint64(48), size=n, replace=False)) \
        if n else np.empty(0, dtype=np.uint64)
    u = u.astype(np.uint64)
    mask = rng.integers(0, 2, size=n, dtype=np.uint8).astype(bool)
    got = algo.apply_filter(u.copy(), mask, engine=eng)
    want = orc.apply_filter(u, mask)
    assert got.tolist() == want.tolist()


@pytest.mark.parametrize("inplace", [True, False])
def test_apply_filter_batch_dev(eng, inplace):
    """Batched device mask compaction, mixed sizes incl. tile boundaries;
    in-place (out == u, the reference's shape) and separate-out forms."""
    rng = np.random.default_rng(SEED)
    sizes = [0, 1, 5, 2048, 2049, 4096, 40_000, 1_000_000]
    us_np, masks_np = [], []
    for i, n in enumerate(sizes):
        u = np.sort(rng.choice(np.uint64(1) << np.uint64(40), size=n,
                               replace=False)).astype(np.uint64) \
            if n else np.empty(0, dtype=np.uint64)
        if i % 3 == 0:
            m = rng.integers(0, 2, size=n, dtype=np.uint8)
        elif i % 3 == 1:
            m = np.ones(n, dtype=np.uint8)   # keep all
        else:
            m = np.zeros(n, dtype=np.uint8)  # drop all
        us_np.append(u)
        masks_np.append(m)
    us = [to_dev(u) for u in us_np]
    masks = [torch.from_numpy(m.view(np.int8)).to("cuda:0").view(torch.uint8)
             for m in masks_np]
    if inplace:
        outs, lens = eng.apply_filter_batch(us, masks)
    else:
        outs = [torch.empty_like(u) for u in us]
        outs, lens = eng.apply_filter_batch(us, masks, outs)
    for i, n in enumerate(sizes):
        want = orc.apply_filter(us_np[i], masks_np[i].astype(bool))
        assert lens[i] == want.size
        assert to_np(outs[i][:lens[i]]).tolist() == want.tolist()


# ---------- cfg 3 extreme Zipf shapes (BASELINE cfg 3 clamp [1k, 10M]) ----------

def test_zipf_extreme_pair_shapes(eng):
    """Tile-count skew inside one grid: 10M-element lists (≈9766 tiles)
    batched next to 1k ones (1 tile), all three ops vs the oracle."""
    rng = np.random.default_rng(SEED + 33)
    def gen(n):
        deltas = rng.integers(1, 17, size=n, dtype=np.uint64)
        return np.cumsum(deltas).astype(np.uint64)
    TEN_M = 10_000_000
    shapes = [(TEN_M, TEN_M), (TEN_M, 1000), (1000, TEN_M), (1000, 1000),
              (TEN_M, 100_000)]
    us_np = [gen(n) for n, _ in shapes]
    vs_np = [gen(m) for _, m in shapes]
    us = [to_dev(u) for u in us_np]
    vs = [to_dev(v) for v in vs_np]
    for op_name in ["intersect", "merge", "difference"]:
        if op_name == "intersect":
            outs, lens = eng.intersect_pairs(us, vs)
            want = [orc.intersect_with(u, v) for u, v in zip(us_np, vs_np)]
        elif op_name == "merge":
            outs, lens = eng.merge_pairs(us, vs)
            want = [orc.merge_sorted([u, v]) for u, v in zip(us_np, vs_np)]
        else:
            outs, lens = eng.difference_pairs(us, vs)
            want = [orc.difference(u, v) for u, v in zip(us_np, vs_np)]
        for i in range(len(shapes)):
            assert lens[i] == want[i].size, (op_name, i)
            got = to_np(outs[i][:lens[i]])
            assert np.array_equal(got, want[i]), (op_name, i)


def test_batch_run_n_matches_run(eng):
    """ua_batch_run_n (pipelined passes, one sync) returns the same lens and
    outputs as per-call runs, for all three ops."""
    rng = np.random.default_rng(SEED + 77)
    us_np = [np.sort(rng.choice(1_000_000, size=n, replace=False)).astype(np.uint64)
             for n in [5000, 70_000, 1]]
    vs_np = [np.sort(rng.choice(1_000_000, size=n, replace=False)).astype(np.uint64)
             for n in [60_000, 3000, 400_000]]
    us = [to_dev(u) for u in us_np]
    vs = [to_dev(v) for v in vs_np]
    outs = [torch.empty(u.numel() + v.numel(), dtype=torch.int64, device="cuda:0")
            for u, v in zip(us, vs)]
    b = eng.make_batch(us, vs, outs)
    from dgraph_amd.algo import OP_INTERSECT, OP_MERGE, OP_DIFFERENCE
    for op in [OP_INTERSECT, OP_MERGE, OP_DIFFERENCE]:
        lens1 = b.run(op)
        snap1 = [to_np(outs[i][:lens1[i]]).copy() for i in range(3)]
        lens5 = b.run_n(op, 5)
        assert lens5 == lens1
        for i in range(3):
            assert np.array_equal(to_np(outs[i][:lens5[i]]), snap1[i])
    b.close()


def test_batch_out_capacity_guard(eng):
    """Undersized outputs raise a clean ValueError BEFORE any launch
    (uidalgo.h capacity contract: intersect min(n,m), diff n, union n+m).
    Found by the r02 soak: a union run into min(n,m)-sized outs is OOB
    device writes the raw-pointer C ABI cannot catch."""
    rng = np.random.default_rng(SEED + 99)
    u = to_dev(np.sort(rng.choice(100_000, size=5000, replace=False)).astype(np.uint64))
    v = to_dev(np.sort(rng.choice(100_000, size=4000, replace=False)).astype(np.uint64))
    small = torch.empty(4000, dtype=torch.int64, device="cuda:0")  # min(n,m)
    b = eng.make_batch([u], [v], [small])
    from dgraph_amd.algo import OP_INTERSECT, OP_MERGE, OP_DIFFERENCE
    b.run(OP_INTERSECT)  # fits
    with pytest.raises(ValueError):
        b.run(OP_MERGE)  # needs n+m
    with pytest.raises(ValueError):
        b.run_n(OP_DIFFERENCE, 2)  # needs n = 5000
    b.close()


def test_extreme_ratio_pairs(eng):
    """Tile-skew extreme: a 10M-element list against tiny and
    mid-sized partners in one batch — a 10M list is ~5.9k merge tiles while
    its 1k partner contributes none of the path, so per-pair tile counts in
    the grid differ by three orders of magnitude.  Parity vs the oracle for
    all three ops, through the prepared batch (split cache exercised by the
    second intersect run)."""
    import torch
    rng = np.random.default_rng(SEED + 424)
    big = synth.gen_sorted_unique(rng, 10_000_000, 40_000_000)
    shapes = [
        (synth.gen_sorted_unique(rng, 1_000, 40_000_000), big),
        (big, synth.gen_sorted_unique(rng, 1_000, 40_000_000)),
        (synth.gen_sorted_unique(rng, 1, 40_000_000), big),
        (big, synth.gen_sorted_unique(rng, 300_000, 40_000_000)),
    ]
    us = [to_dev(u) for u, _ in shapes]
    vs = [to_dev(v) for _, v in shapes]
    outs = [torch.empty(u.numel() + v.numel(), dtype=torch.int64, device="cuda:0")
            for u, v in zip(us, vs)]
    batch = eng.make_batch(us, vs, outs)
    for op, ref in [(algo.OP_INTERSECT, orc.intersect_with),
                    (algo.OP_MERGE, lambda u, v: orc.merge_sorted([u, v])),
                    (algo.OP_DIFFERENCE, orc.difference),
                    (algo.OP_INTERSECT, orc.intersect_with)]:  # cached re-run
        lens = batch.run(op)
        for i, (u, v) in enumerate(shapes):
            want = ref(u, v)
            assert lens[i] == want.size, (op, i)
            got = to_np(outs[i][:lens[i]])
            assert np.array_equal(got, want), (op, i)
    batch.close()
