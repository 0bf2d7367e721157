"""Seeded synthetic UID-list generators for bench + parity tests
(SURVEY.md §8d).  All duplicate-free sorted uint64."""
import numpy as np

SEED = 0xD6A77


def gen_sorted_unique(rng, n, limit):
    """n distinct sorted values uniform over [0, limit).  limit is widened to
    2n when the range is too small to hold n distinct values."""
    if n == 0:
        return np.empty(0, dtype=np.uint64)
    if limit < 2 * n:
        limit = 2 * n
    draw = rng.integers(0, limit, size=int(n * 1.3) + 16, dtype=np.uint64)
    un = np.unique(draw)
    while un.size < n:
        un = np.unique(np.concatenate(
            [un, rng.integers(0, limit, size=n, dtype=np.uint64)]))
    idx = np.sort(rng.choice(un.size, size=n, replace=False))
    return un[idx]


def gen_pair(rng, n, m, overlap, limit):
    """Sorted duplicate-free (u, v) with exactly `overlap` planted common
    values (cfg 2: n=m=1M, overlap=10k, limit=1e8)."""
    pool = gen_sorted_unique(rng, n + m - overlap, limit)
    perm = rng.permutation(pool.size)
    common = pool[perm[:overlap]]
    u_only = pool[perm[overlap:n]]
    v_only = pool[perm[n:]]
    u = np.sort(np.concatenate([common, u_only]))
    v = np.sort(np.concatenate([common, v_only]))
    return u, v, np.sort(common)


def offset_pair(u, v, common, pair_idx):
    """Shift a pair into a distinct 32-MSB range: distinct buffers + distinct
    values per pair without regenerating (values < 2^32 required)."""
    off = np.uint64(pair_idx) << np.uint64(32)
    return u + off, v + off, common + off


def zipf_sizes(rng, k, s=1.07, lo=1000, hi=10_000_000):
    """cfg 3: k list sizes ~ Zipf(s) clamped to [lo, hi]."""
    raw = rng.zipf(s, size=k).astype(np.float64)
    sizes = np.clip(raw * lo, lo, hi).astype(np.int64)
    return sizes


def getuids_geometric(rng, size, mean_delta=33):
    """cfg 4 input (mirrors codec_test.go:26-35 getUids): start < 100,
    deltas uniform [0, 33)."""
    deltas = rng.integers(0, mean_delta, size=size, dtype=np.uint64)
    deltas[0] = rng.integers(0, 100)
    return np.cumsum(deltas).astype(np.uint64)
