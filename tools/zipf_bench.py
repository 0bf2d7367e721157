#!/usr/bin/env python3
"""cfg 3 shape: one grid over thousands of Zipf-skewed
predicate list-pairs — intersect, pairwise MergeSorted, Difference.
Validates the batched launcher under size skew and reports throughput.
Prints one JSON line; parity for this shape is gated by
tests/test_parity_gpu.py::test_batched_zipf_vs_oracle.
"""
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from dgraph_amd import algo, synth  # noqa: E402


def gen_sorted(rng, n):
    deltas = rng.integers(1, 17, size=n, dtype=np.uint64)
    return (np.uint64(rng.integers(0, 1000)) + np.cumsum(deltas)).astype(np.uint64)


def main():
    pairs = int(sys.argv[1]) if len(sys.argv) > 1 else 4096
    steps = int(sys.argv[2]) if len(sys.argv) > 2 else 20
    hi = int(sys.argv[3]) if len(sys.argv) > 3 else 10_000_000
    mem_budget = float(os.environ.get("UA_ZIPF_GB", "200")) * 1e9
    rng = np.random.default_rng(synth.SEED + 3)
    sizes_u = synth.zipf_sizes(rng, pairs, lo=1000, hi=hi)
    sizes_v = synth.zipf_sizes(rng, pairs, lo=1000, hi=hi)
    # BASELINE cfg 3 at full spec (4096 pairs, Zipf[1k,10M]) sums to ~350 GB
    # of inputs — more than one GPU's 288 GB HBM (it is cfg 5's multi-GPU
    # working set).  On one GPU: keep the SAME size distribution and take
    # the largest prefix of pairs whose in+out working set fits the budget.
    requested = pairs
    per_pair = (sizes_u + sizes_v) * 8 * 3  # u+v + out(n+m) bytes
    cum = np.cumsum(per_pair)
    pairs = int(np.searchsorted(cum, mem_budget)) or 1
    sizes_u, sizes_v = sizes_u[:pairs], sizes_v[:pairs]
    total = int(sizes_u.sum() + sizes_v.sum())
    eng = algo.Engine(0)
    us, vs = [], []
    for p in range(pairs):
        us.append(torch.from_numpy(gen_sorted(rng, int(sizes_u[p])).view(np.int64)).cuda())
        vs.append(torch.from_numpy(gen_sorted(rng, int(sizes_v[p])).view(np.int64)).cuda())
    m_outs = [torch.empty(u.numel() + v.numel(), dtype=torch.int64, device="cuda")
              for u, v in zip(us, vs)]

    res = {"workload": "cfg3_zipf_batch", "pairs": pairs,
           "pairs_requested": requested, "size_hi_clamp": hi,
           "total_elems": total, "total_MB": round(total * 8 / 1e6, 1),
           "size_min": int(min(sizes_u.min(), sizes_v.min())),
           "size_max": int(max(sizes_u.max(), sizes_v.max()))}
    # prepared batch with n+m capacity fits all three ops
    batch = eng.make_batch(us, vs, m_outs)
    for name, op in [("intersect", algo.OP_INTERSECT), ("merge", algo.OP_MERGE),
                     ("difference", algo.OP_DIFFERENCE)]:
        batch.run(op)
        torch.cuda.synchronize()
        eng.stats_reset()
        t0 = time.perf_counter()
        for _ in range(steps):
            batch.run(op)
        torch.cuda.synchronize()
        el = time.perf_counter() - t0
        st = eng.stats()
        res[f"{name}_ms_per_batch"] = round(el / steps * 1e3, 3)
        res[f"{name}_Gelems_per_s"] = round(total * steps / el / 1e9, 1)
        res[f"{name}_kernel_ms"] = round(st["kernel_ms"] / max(st["launches"], 1), 3)
    print(json.dumps(res))


if __name__ == "__main__":
    main()
