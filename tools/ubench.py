#!/usr/bin/env python3
"""Microbench for kernel bisection: runs the batched intersect and reports
step time + dominant-kernel HIP-event time.  No correctness asserts (ablation
builds return garbage by design).  Select the .so with UA_LIB_PATH."""
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from dgraph_amd import algo, synth  # noqa: E402


def main():
    pairs = int(sys.argv[1]) if len(sys.argv) > 1 else 48
    steps = int(sys.argv[2]) if len(sys.argv) > 2 else 50
    tag = os.environ.get("UA_LIB_PATH", "default")
    eng = algo.Engine(0)
    rng = np.random.default_rng(synth.SEED)
    u0, v0, common0 = synth.gen_pair(rng, 1_000_000, 1_000_000, 10_000, 100_000_000)
    us, vs = [], []
    for p in range(pairs):
        up, vp, _ = synth.offset_pair(u0, v0, common0, p)
        us.append(torch.from_numpy(up.view(np.int64)).cuda())
        vs.append(torch.from_numpy(vp.view(np.int64)).cuda())
    outs = [torch.empty(1_000_000, dtype=torch.int64, device="cuda") for _ in range(pairs)]
    # prepared batch: mirrors bench.py (partition + split cache warm after
    # the warmup runs)
    batch = eng.make_batch(us, vs, outs)
    for _ in range(5):
        batch.run(algo.OP_INTERSECT)
    torch.cuda.synchronize()
    eng.stats_reset()
    t0 = time.perf_counter()
    batch.run_n(algo.OP_INTERSECT, steps)
    torch.cuda.synchronize()
    el = time.perf_counter() - t0
    st = eng.stats()
    kms = st["kernel_ms"] / max(st["launches"], 1)
    bytes_per_launch = pairs * 2 * 8_000_000
    print(f"[{tag}] pairs={pairs} steps={steps} ms/step={el/steps*1e3:.3f} "
          f"k_tiles_ms={kms:.3f} k_tiles_GB/s={bytes_per_launch/kms*1e3/1e9:.0f}")


if __name__ == "__main__":
    main()
