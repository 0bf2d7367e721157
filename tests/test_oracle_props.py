"""Property tests of the oracle on seeded random inputs.

Independent cross-check: on duplicate-free sorted inputs the reference's ops
are plain set algebra, so numpy's set routines give a second, independent
restatement (the golden tables in test_oracle_golden.py give the first).
Also mirrors the reference's randomized roundtrip tests
(codec_test.go:37-58 TestUidPack, :190-217 TestDecoder,
uidlist_test.go:607-681 compressed-intersect vs constructed ground truth).
"""
import numpy as np
import pytest

from oracle import bind as orc

SEED = 0xD6A77


def gen_sorted_unique(rng, n, limit):
    if n == 0:
        return np.empty(0, dtype=np.uint64)
    # over-draw, unique, trim
    draw = rng.integers(0, limit, size=int(n * 1.3) + 16, dtype=np.uint64)
    un = np.unique(draw)
    while un.size < n:
        extra = rng.integers(0, limit, size=n, dtype=np.uint64)
        un = np.unique(np.concatenate([un, extra]))
    idx = np.sort(rng.choice(un.size, size=n, replace=False))
    return un[idx]


def getuids(rng, size):
    """Mirrors codec_test.go:26-35 getUids: start <100, deltas Uniform[0,33)."""
    deltas = rng.integers(0, 33, size=size, dtype=np.uint64)
    deltas[0] = rng.integers(0, 100)
    return np.cumsum(deltas).astype(np.uint64)


@pytest.mark.parametrize("n,m,limit", [
    (0, 0, 10), (1, 1, 10), (100, 100, 300), (1000, 10, 10_000),
    (10, 1000, 10_000), (1000, 1000, 3000), (5000, 4096, 20_000),
    (10_000, 100, 1_000_000),  # ratio 100 -> Jump path
    (100_000, 100, 1_000_000),  # ratio 1000 -> Bin path
])
def test_intersect_difference_vs_numpy(n, m, limit):
    rng = np.random.default_rng(SEED + n * 31 + m)
    u = gen_sorted_unique(rng, n, limit)
    v = gen_sorted_unique(rng, m, limit)
    assert orc.intersect_with(u, v).tolist() == np.intersect1d(u, v).tolist()
    assert orc.difference(u, v).tolist() == np.setdiff1d(u, v).tolist()


@pytest.mark.parametrize("k,n", [(1, 100), (2, 1000), (3, 500), (7, 99), (150, 40)])
def test_merge_intersect_k_vs_numpy(k, n):
    rng = np.random.default_rng(SEED + k)
    lists = [gen_sorted_unique(rng, rng.integers(0, n + 1), n * 3) for _ in range(k)]
    got = orc.merge_sorted(lists)
    want = np.unique(np.concatenate(lists)) if lists else np.empty(0, dtype=np.uint64)
    assert got.tolist() == want.tolist()

    got_i = orc.intersect_sorted(lists)
    want_i = lists[0]
    for l in lists[1:]:
        want_i = np.intersect1d(want_i, l)
    assert got_i.tolist() == want_i.tolist()


def test_index_of():
    rng = np.random.default_rng(SEED)
    u = gen_sorted_unique(rng, 1000, 100_000)
    for uid in list(u[::97]) + [0, 2**63, u[0] + 1]:
        want = int(np.searchsorted(u, uid))
        if want < u.size and u[want] == uid:
            assert orc.index_of(u, int(uid)) == want
        else:
            assert orc.index_of(u, int(uid)) == -1


@pytest.mark.parametrize("size", [1, 5, 100, 255, 256, 257, 1000, 100_000])
@pytest.mark.parametrize("block_size", [0, 10, 128, 256])
def test_codec_roundtrip(size, block_size):
    rng = np.random.default_rng(SEED + size + block_size)
    uids = getuids(rng, size)
    pack = orc.Pack(uids, block_size)
    assert pack.exact_len() == np.unique(uids).size or pack.exact_len() == uids.size
    got = pack.decode(0)
    assert got.tolist() == uids.tolist()


def test_codec_roundtrip_large():
    rng = np.random.default_rng(SEED)
    uids = getuids(rng, 2_000_000)
    pack = orc.Pack(uids, 256)
    assert pack.exact_len() == uids.size
    assert pack.decode(0).tolist() == uids.tolist()


def test_decoder_seek_decode_suffix():
    # TestDecoder codec_test.go:190-217: Decode(pack, seek) returns the suffix.
    N = 10001
    uids = np.arange(3, N, 3, dtype=np.uint64)
    pack = orc.Pack(uids, 10)
    dec = orc.Dec(pack)
    for i in range(3, N, 33):  # step 33 to keep runtime sane; same shape
        got = dec.seek(i, orc.SEEK_START)
        assert got[0] == i
        got = dec.seek(i - 1, orc.SEEK_START)
        assert got[0] == i
        got = dec.seek(i - 2, orc.SEEK_START)
        assert got[0] == i
        start = i // 3 - 1
        assert pack.decode(i).tolist() == uids[start:].tolist()


def test_seek_to_block_semantics():
    """codec.Decoder.SeekToBlock (codec.go:219): positions at the block that
    could contain uid and returns the WHOLE block untruncated (unlike Seek);
    stateful prevBlockIdx fast path included."""
    uids = np.arange(0, 10001, 10, dtype=np.uint64)
    pack = orc.Pack(uids, 10)
    bases, nums, offs, blob = pack.flatten()
    dec = orc.Dec(pack)
    dec.seek(0, orc.SEEK_START)
    for x in range(5, 10000, 97):
        got = dec.seek_to_block(x, orc.SEEK_CURRENT)
        # whence=SeekCurrent: last block with base <= x; if x exceeds that
        # block's last uid, SeekToBlock falls through to Next() (codec.go:270)
        bidx = max(int(np.searchsorted(bases, x, side="right")) - 1, 0)
        last_uid = int(bases[bidx]) + (int(nums[bidx]) - 1) * 10
        if x > last_uid:
            bidx += 1
        assert got.size > 0 and got[0] == bases[bidx], (x, got[:3])
        # whole block, no truncation to >= x
        assert got.size == nums[bidx]
    # exhausted decoder (past the end) returns empty where Go would panic
    dec2 = orc.Dec(pack)
    dec2.seek(10**9, orc.SEEK_START)
    got = dec2.seek_to_block(50, orc.SEEK_CURRENT)
    assert got.size == 0


def test_decoder_valid_and_next():
    uids = np.arange(0, 100, 3, dtype=np.uint64)
    pack = orc.Pack(uids, 10)
    dec = orc.Dec(pack)
    dec.seek(0, orc.SEEK_START)
    seen = []
    while True:
        u = dec.uids()
        if u.size == 0:
            break
        seen.extend(u.tolist())
        if not dec.valid():
            break
        dec.next()
    assert seen == uids.tolist()


def fill_nums(rng, n1, n2):
    """Mirrors uidlist_test.go:583-605 fillNums: returns (common, block, other)."""
    common = rng.integers(0, 2**64, size=n1, dtype=np.uint64)
    block = np.concatenate([common, rng.integers(0, 2**64, size=n2, dtype=np.uint64)])
    other = np.concatenate([common, rng.integers(0, 2**64, size=n2, dtype=np.uint64)])
    return np.sort(common), np.sort(block), np.sort(other)


@pytest.mark.parametrize("n1", [0, 1, 3, 11, 100])
@pytest.mark.parametrize("n2", [0, 1, 3, 11, 100])
def test_intersect_compressed_vs_ground_truth(n1, n2):
    # uidlist_test.go:607-681 (both LinJump and Bin paths via the dispatcher,
    # plus BlockSize=0 which forces 1-uid blocks).
    rng = np.random.default_rng(SEED + n1 * 7 + n2)
    common, block, other = fill_nums(rng, n1, n2)
    for bs in (10, 0):
        pack = orc.Pack(block, bs)
        got = orc.intersect_compressed_with(pack, 0, other)
        assert got.tolist() == common.tolist()


def test_intersect_compressed_after_uid():
    rng = np.random.default_rng(SEED)
    common, block, other = fill_nums(rng, 50, 500)
    pack = orc.Pack(block, 10)
    after = int(common[25])
    got = orc.intersect_compressed_with(pack, after, other)
    # after positions the decoder at the first uid >= after (Seek SeekStart)
    want = np.intersect1d(block[block >= after], other)
    assert got.tolist() == want.tolist()


def test_batch_cpu_matches_single():
    rng = np.random.default_rng(SEED)
    us = [gen_sorted_unique(rng, int(rng.integers(0, 5000)), 20_000) for _ in range(32)]
    vs = [gen_sorted_unique(rng, int(rng.integers(0, 5000)), 20_000) for _ in range(32)]
    outs = orc.intersect_batch_cpu(us, vs)
    for u, v, o in zip(us, vs, outs):
        assert o.tolist() == orc.intersect_with(u, v).tolist()
