#!/usr/bin/env python3
"""Segmented-sort throughput (the §8f row-3 primitive): many mixed segments
and one large segment, u64 keys with duplicates.  Prints one JSON line."""
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from dgraph_amd import algo, synth  # noqa: E402


def run(eng, arrs, steps, tag, res):
    tensors = [torch.from_numpy(a.view(np.int64)).cuda() for a in arrs]
    total = sum(a.size for a in arrs)
    eng.sort_segments(tensors)  # warm (also leaves them sorted; re-sorting sorted
    # data is the cheap case, so re-upload fresh tensors per timing run)
    runs = []
    for _ in range(steps):
        fresh = [torch.from_numpy(a.view(np.int64)).cuda() for a in arrs]
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        eng.sort_segments(fresh)
        torch.cuda.synchronize()
        runs.append(time.perf_counter() - t0)
    el = min(runs)
    # correctness
    got = fresh[0].cpu().numpy().view(np.uint64)
    assert np.array_equal(got, np.sort(arrs[0]))
    res[f"{tag}_ms"] = round(el * 1e3, 3)
    res[f"{tag}_Melems_per_s"] = round(total / el / 1e6, 1)
    res[f"{tag}_total_elems"] = total


def main():
    steps = int(sys.argv[1]) if len(sys.argv) > 1 else 5
    eng = algo.Engine(0)
    rng = np.random.default_rng(synth.SEED)
    res = {"workload": "segmented_sort"}
    mixed = [rng.integers(0, 2**63, size=int(s), dtype=np.uint64)
             for s in synth.zipf_sizes(rng, 4096, lo=100, hi=100_000)]
    run(eng, mixed, steps, "zipf4096", res)
    big = [rng.integers(0, 2**63, size=10_000_000, dtype=np.uint64)]
    run(eng, big, steps, "single_10M", res)
    print(json.dumps(res))


if __name__ == "__main__":
    main()
