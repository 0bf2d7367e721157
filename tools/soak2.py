#!/usr/bin/env python3
"""Bisect the create/run/close crash: phases A (create/close only),
B (+intersect), C (+merge), D (interleaved like soak.py phase 2)."""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from dgraph_amd import algo, synth  # noqa: E402


def main():
    phase = sys.argv[1] if len(sys.argv) > 1 else "D"
    iters = int(sys.argv[2]) if len(sys.argv) > 2 else 100
    rng = np.random.default_rng(synth.SEED)
    u0, v0, common0 = synth.gen_pair(rng, 1_000_000, 1_000_000, 10_000, 100_000_000)
    us, vs, outs = [], [], []
    for p in range(8):
        up, vp, _ = synth.offset_pair(u0, v0, common0, p)
        us.append(torch.from_numpy(up.view(np.int64)).cuda())
        vs.append(torch.from_numpy(vp.view(np.int64)).cuda())
        outs.append(torch.empty(2_000_000, dtype=torch.int64, device="cuda"))
    eng = algo.Engine(0)
    for i in range(iters):
        b = eng.make_batch(us, vs, outs)
        if phase in ("B", "D"):
            b.run(algo.OP_INTERSECT)
        if phase in ("C", "D"):
            b.run(algo.OP_MERGE)
        b.close()
        if (i + 1) % 20 == 0:
            print(f"phase {phase}: {i+1}/{iters} ok", flush=True)
    torch.cuda.synchronize()
    print(f"phase {phase} done", flush=True)


if __name__ == "__main__":
    main()
