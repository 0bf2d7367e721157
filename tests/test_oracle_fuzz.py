"""Seeded randomized fuzzing of the oracle against numpy set algebra and
codec roundtrips — property-level insurance on top of the transcribed golden
tables.  Plain seeded loops (hypothesis's big-int strategies proved
pathologically slow here); a few thousand cases run in seconds."""
import numpy as np

from oracle import bind as orc

SEED = 0xFA22


def rand_list(rng, max_len=200, full_range=False):
    n = int(rng.integers(0, max_len + 1))
    hi = 2**64 if full_range else int(rng.integers(1, 4 * max_len + 2))
    return np.sort(rng.integers(0, hi, size=n, dtype=np.uint64))


def test_fuzz_intersect_difference():
    rng = np.random.default_rng(SEED)
    for case in range(600):
        u = np.unique(rand_list(rng, full_range=(case % 5 == 0)))
        v = np.unique(rand_list(rng, full_range=(case % 5 == 0)))
        assert orc.intersect_with(u, v).tolist() == np.intersect1d(u, v).tolist()
        assert orc.difference(u, v).tolist() == np.setdiff1d(u, v).tolist()


def test_fuzz_merge_sorted_with_dups():
    rng = np.random.default_rng(SEED + 1)
    for case in range(300):
        k = int(rng.integers(0, 7))
        arrs = [rand_list(rng) for _ in range(k)]  # dups allowed
        got = orc.merge_sorted(arrs)
        want = (np.unique(np.concatenate(arrs)) if arrs
                else np.empty(0, dtype=np.uint64))
        assert got.tolist() == want.tolist()


def test_fuzz_codec_roundtrip():
    rng = np.random.default_rng(SEED + 2)
    for case in range(300):
        bs = int(rng.choice([0, 1, 3, 10, 256]))
        uids = rand_list(rng, full_range=(case % 4 == 0))  # dups allowed
        pack = orc.Pack(uids, bs)
        assert pack.decode(0).tolist() == uids.tolist()
        assert pack.exact_len() == uids.size


def test_fuzz_intersect_compressed():
    rng = np.random.default_rng(SEED + 3)
    for case in range(200):
        bs = int(rng.choice([0, 7, 256]))
        pack_uids = np.unique(rand_list(rng))
        v = np.unique(rand_list(rng))
        pack = orc.Pack(pack_uids, bs)
        got = orc.intersect_compressed_with(pack, 0, v)
        assert got.tolist() == np.intersect1d(pack_uids, v).tolist()
