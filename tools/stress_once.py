#!/usr/bin/env python3
"""One-off large-scale stress checks (run on a GPU box):
1. 64-way MergeSorted over duplicate-heavy 2M lists, bit-exact vs oracle.
2. 1024-pair (32 GB) prepared-batch intersect — robustness at scale
   (bounded allocation, planted-overlap check per pair sample)."""
import sys
import time

import numpy as np
import torch

sys.path.insert(0, "/root/repo")
from dgraph_amd import algo, synth  # noqa: E402
from oracle import bind as orc  # noqa: E402

eng = algo.Engine(0)
rng = np.random.default_rng(synth.SEED + 7)

# 1. dup-heavy k-way merge at scale
lists = [np.sort(rng.integers(0, 40_000_000, size=2_000_000, dtype=np.uint64))
         for _ in range(64)]
d_lists = [torch.from_numpy(l.view(np.int64)).cuda() for l in lists]
t0 = time.perf_counter()
got = eng.merge_sorted(d_lists)
torch.cuda.synchronize()
el = time.perf_counter() - t0
want = orc.merge_sorted(lists)
assert np.array_equal(got.cpu().numpy().view(np.uint64), want), "merge mismatch"
print(f"64-way dup-heavy merge of 128M elems: {el*1e3:.1f} ms, out={want.size}, bit-exact OK")
del d_lists, got
torch.cuda.empty_cache()

# 2. 1024-pair batch (32 GB)
P = 1024
u0, v0, common0 = synth.gen_pair(rng, 1_000_000, 1_000_000, 10_000, 100_000_000)
us, vs = [], []
for p in range(P):
    up, vp, _ = synth.offset_pair(u0, v0, common0, p)
    us.append(torch.from_numpy(up.view(np.int64)).cuda())
    vs.append(torch.from_numpy(vp.view(np.int64)).cuda())
outs = [torch.empty(1_000_000, dtype=torch.int64, device="cuda") for _ in range(P)]
batch = eng.make_batch(us, vs, outs)
lens = batch.run(algo.OP_INTERSECT)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(20):
    lens = batch.run(algo.OP_INTERSECT)
torch.cuda.synchronize()
el = (time.perf_counter() - t0) / 20
for p in [0, 511, 1023]:
    want_p = common0 + (np.uint64(p) << np.uint64(32))
    assert lens[p] == 10_000
    assert np.array_equal(outs[p][:lens[p]].cpu().numpy().view(np.uint64), want_p)
print(f"1024-pair (32 GB) batch: {el*1e3:.2f} ms/step = {P/el:.0f} pairs/s, spot checks OK")
