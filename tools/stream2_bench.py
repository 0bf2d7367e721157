#!/usr/bin/env python3
"""Two-stream convoy-break experiment: split the cfg2 batch across TWO
engine contexts (each ua_ctx has its own HIP stream) and run both halves
concurrently from two host threads (ctypes releases the GIL; run_n syncs
once per thread).

Rationale: phase ablations show fill-only 0.474 ms and walk-only 0.476 but
fill+walk 0.652 — equal-share HBM service synchronizes fill completions
across the single grid's workgroups, so whole CUs alternate all-fill /
all-walk and HBM idles ~27% of the time.  Blocks from two INDEPENDENT
kernels co-scheduled on the same CUs are not phase-locked to each other;
if the mixture de-convoys, the aggregate beats the single grid.

Usage: python tools/stream2_bench.py [pairs_total] [steps]
"""
import os
import sys
import threading
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from dgraph_amd import algo, synth  # noqa: E402


def make_half(eng, rng_pairs, base_idx):
    us, vs, outs = [], [], []
    for p, (up, vp) in enumerate(rng_pairs):
        us.append(torch.from_numpy(up.view(np.int64)).cuda())
        vs.append(torch.from_numpy(vp.view(np.int64)).cuda())
        outs.append(torch.empty(1_000_000, dtype=torch.int64, device="cuda"))
    return eng.make_batch(us, vs, outs), (us, vs, outs)


def main():
    total = int(sys.argv[1]) if len(sys.argv) > 1 else 192
    steps = int(sys.argv[2]) if len(sys.argv) > 2 else 60
    half = total // 2
    rng = np.random.default_rng(synth.SEED)
    u0, v0, common0 = synth.gen_pair(rng, 1_000_000, 1_000_000, 10_000, 100_000_000)
    pairs = [synth.offset_pair(u0, v0, common0, p)[:2] for p in range(total)]

    eng_a, eng_b = algo.Engine(0), algo.Engine(0)
    batch_a, keep_a = make_half(eng_a, pairs[:half], 0)
    batch_b, keep_b = make_half(eng_b, pairs[half:], half)
    # single-engine reference: the same pairs in ONE grid
    eng_c = algo.Engine(0)
    batch_c, keep_c = make_half(eng_c, pairs, 0)

    for _ in range(3):
        batch_c.run_n(algo.OP_INTERSECT, 2)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    batch_c.run_n(algo.OP_INTERSECT, steps)
    el_one = time.perf_counter() - t0

    def worker(batch):
        batch.run_n(algo.OP_INTERSECT, steps)

    for _ in range(3):  # warmup both streams
        ta = threading.Thread(target=worker, args=(batch_a,))
        tb = threading.Thread(target=worker, args=(batch_b,))
        ta.start(); tb.start(); ta.join(); tb.join()
    t0 = time.perf_counter()
    ta = threading.Thread(target=worker, args=(batch_a,))
    tb = threading.Thread(target=worker, args=(batch_b,))
    ta.start(); tb.start(); ta.join(); tb.join()
    el_two = time.perf_counter() - t0

    print(f"single-grid {total}p: {el_one/steps*1e3:.3f} ms/step "
          f"({total*steps/el_one:.0f} pairs/s)")
    print(f"two-stream 2x{half}p: {el_two/steps*1e3:.3f} ms/step "
          f"({total*steps/el_two:.0f} pairs/s)  "
          f"ratio {el_one/el_two:.3f}x")


if __name__ == "__main__":
    main()
