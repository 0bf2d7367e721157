#!/usr/bin/env python3
"""Repro & localize UA_AISECT parity divergence: the failing test's exact
pairs, per-pair diff positions (mod tile/window) against numpy."""
import os, sys
import numpy as np
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from dgraph_amd import algo, synth
from tests.test_parity_gpu import SEED  # same seed as the suite

def main():
    rng = np.random.default_rng(SEED + 77)
    us, vs = [], []
    for _ in range(32):
        n = int(rng.integers(0, 50_000)); m = int(rng.integers(0, 50_000))
        us.append(synth.gen_sorted_unique(rng, n, 3 * (n + m) + 10))
        vs.append(synth.gen_sorted_unique(rng, m, 3 * (n + m) + 10))
    d_us = [torch.from_numpy(u.view(np.int64)).cuda() for u in us]
    d_vs = [torch.from_numpy(v.view(np.int64)).cuda() for v in vs]
    outs = [torch.empty(max(u.numel() + v.numel(), 1), dtype=torch.int64, device="cuda")
            for u, v in zip(d_us, d_vs)]
    eng = algo.Engine(0)
    batch = eng.make_batch(d_us, d_vs, outs)
    for op, name, ref_fn in [(algo.OP_INTERSECT, "inter", np.intersect1d),
                             (algo.OP_DIFFERENCE, "diff", np.setdiff1d)]:
        lens = batch.run(op)
        bad = 0
        for i in range(32):
            got = outs[i][:lens[i]].cpu().numpy().view(np.uint64)
            ref = ref_fn(us[i], vs[i])
            if not np.array_equal(got, ref):
                bad += 1
                miss = np.setdiff1d(ref, got); extra = np.setdiff1d(got, ref)
                # positions of missing in A
                mpos = np.searchsorted(us[i], miss[:4])
                print(f"{name} pair {i}: n={len(us[i])} m={len(vs[i])} "
                      f"got={len(got)} ref={len(ref)} miss={len(miss)} extra={len(extra)}")
                for mv, mp in zip(miss[:4], mpos):
                    bp = int(np.searchsorted(vs[i], mv))
                    print(f"   missing val={mv} Aidx={mp} (tile {mp//512}, "
                          f"in-tile {mp%512}, win {mp%512//64}) Bidx={bp}")
                if len(extra):
                    print(f"   extra vals {extra[:4]}")
        print(f"{name}: {bad} bad pairs")
    batch.close()

if __name__ == "__main__":
    main()
