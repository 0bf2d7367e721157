#!/usr/bin/env python3
"""Fast engine canary for fresh GPU boxes: one small pass over every lookback
code path (one-shot batch, prepared batch, union, difference, merge tree,
apply_filter, sort segments) with correctness asserts.  Run under `timeout`
FIRST on a box so a regression hangs this tiny probe, not the full suite."""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from dgraph_amd import algo, synth  # noqa: E402


def dev(a):
    a = np.ascontiguousarray(a, dtype=np.uint64)
    if a.size == 0:
        return torch.empty(0, dtype=torch.int64, device="cuda:0")
    return torch.from_numpy(a.view(np.int64)).to("cuda:0")


def main():
    eng = algo.Engine(0)
    rng = np.random.default_rng(synth.SEED)
    # one-shot small batch, all ops
    u = np.arange(0, 20000, 2, dtype=np.uint64)
    v = np.arange(0, 30000, 3, dtype=np.uint64)
    us, vs = [dev(u), dev(v), dev(u[:0])], [dev(v), dev(u), dev(v)]
    outs, lens = eng.intersect_pairs(us, vs)
    want = np.intersect1d(u, v)
    assert lens[0] == want.size and lens[1] == want.size and lens[2] == 0
    assert np.array_equal(outs[0][:lens[0]].cpu().numpy().view(np.uint64), want)
    outs, lens = eng.merge_pairs(us, vs)
    assert lens[0] == np.union1d(u, v).size
    outs, lens = eng.difference_pairs(us, vs)
    assert lens[0] == np.setdiff1d(u, v).size
    # prepared batch incl. a 1Mx1M pair (multi-tile lookback chains)
    u0, v0, common0 = synth.gen_pair(rng, 1_000_000, 1_000_000, 10_000,
                                     100_000_000)
    du, dv = dev(u0), dev(v0)
    douts = [torch.empty(1_000_000, dtype=torch.int64, device="cuda:0"),
             torch.empty(30000, dtype=torch.int64, device="cuda:0")]
    b = eng.make_batch([du, dev(u)], [dv, dev(v)], douts)
    for _ in range(3):
        lens = b.run(algo.OP_INTERSECT)
        assert lens[0] == 10_000 and lens[1] == want.size
        assert np.array_equal(douts[0][:10_000].cpu().numpy().view(np.uint64),
                              common0)
    mlens = None
    douts2 = [torch.empty(2_000_000, dtype=torch.int64, device="cuda:0"),
              torch.empty(60000, dtype=torch.int64, device="cuda:0")]
    b2 = eng.make_batch([du, dev(u)], [dv, dev(v)], douts2)
    for _ in range(2):
        mlens = b2.run(algo.OP_MERGE)
    assert mlens[0] == np.union1d(u0, v0).size
    b.close()
    b2.close()
    # merge-k device tree
    lists = [dev(np.sort(rng.choice(100_000, size=5000, replace=False)))
             for _ in range(7)]
    got = eng.merge_sorted(lists)
    wantm = np.unique(np.concatenate(
        [x.cpu().numpy().view(np.uint64) for x in lists]))
    assert np.array_equal(got.cpu().numpy().view(np.uint64), wantm)
    # apply_filter batched + host
    m = (u % 4 == 0)
    got = algo.apply_filter(u.copy(), m, engine=eng)
    assert np.array_equal(got, u[m])
    # segmented sort
    t = dev(rng.permutation(np.arange(10000, dtype=np.uint64)))
    eng.sort_segments([t])
    assert np.array_equal(t.cpu().numpy().view(np.uint64),
                          np.arange(10000, dtype=np.uint64))
    eng.close()
    print("canary ok")


if __name__ == "__main__":
    main()
