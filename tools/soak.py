#!/usr/bin/env python3
"""Serving-shape soak + leak sanity: a long run_n burst with stable
results, then repeated batch and engine create/destroy cycles with HBM
free-memory checks (catches workspace leaks across the C-ABI)."""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from dgraph_amd import algo, synth  # noqa: E402


def free_mb():
    free, _ = torch.cuda.mem_get_info()
    return free // (1 << 20)


def main():
    steps = int(sys.argv[1]) if len(sys.argv) > 1 else 2000
    rng = np.random.default_rng(synth.SEED)
    u0, v0, common0 = synth.gen_pair(rng, 1_000_000, 1_000_000, 10_000, 100_000_000)
    us, vs, outs = [], [], []
    for p in range(48):
        up, vp, _ = synth.offset_pair(u0, v0, common0, p)
        us.append(torch.from_numpy(up.view(np.int64)).cuda())
        vs.append(torch.from_numpy(vp.view(np.int64)).cuda())
        # n+m capacity: phase 2 runs OP_MERGE too (union outputs up to n+m;
        # the 1M-cap first version of this script was itself the OOB bug the
        # first soak run "found")
        outs.append(torch.empty(2_000_000, dtype=torch.int64, device="cuda"))

    eng = algo.Engine(0)
    batch = eng.make_batch(us, vs, outs)
    print("batch created", flush=True)
    ref = batch.run(algo.OP_INTERSECT)
    print("single run ok", flush=True)

    lens = batch.run_n(algo.OP_INTERSECT, steps)
    assert lens == ref, "run_n soak diverged from single run"
    print(f"soak ok: {steps} pipelined runs, lens stable ({sum(ref)} total)", flush=True)

    base = free_mb()
    for i in range(100):
        b2 = eng.make_batch(us[:8], vs[:8], outs[:8])
        b2.run(algo.OP_INTERSECT)
        b2.run(algo.OP_MERGE)
        b2.close()
    drift = base - free_mb()
    print(f"batch create/destroy x100: free-memory drift {drift} MB")
    assert drift < 256, f"leak suspected: {drift} MB"

    base = free_mb()
    for i in range(20):
        e2 = algo.Engine(0)
        b2 = e2.make_batch(us[:4], vs[:4], outs[:4])
        b2.run(algo.OP_DIFFERENCE)
        b2.close()
        e2.close()
    drift = base - free_mb()
    print(f"engine create/destroy x20: free-memory drift {drift} MB")
    assert drift < 256, f"leak suspected: {drift} MB"
    print("leak sanity ok")


if __name__ == "__main__":
    main()
