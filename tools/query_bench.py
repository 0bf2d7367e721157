#!/usr/bin/env python3
"""End-to-end query-shape benchmark: the reference's read-query combine
 as engine calls —

  1. handleUidPostings fan-out (worker/task.go:834-971): N keys' packs each
     intersected with the shared q.UidList  -> UidMatrix   [one grid]
  2. DestUIDs = MergeSorted(matrix rows)    (query.go:2290, "or" combine)
  3. "and" filter = IntersectSorted(rows),  "not" = Difference(DestUIDs, and)
     (query.go:2357-2371)
  4. updateUidMatrix (query.go:1425-1436): every row ∩ DestUIDs  [one grid]

Spot-checked against the oracle on a few rows; per-phase times + whole-shape
rate printed as one JSON line.
"""
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from dgraph_amd import algo, synth  # noqa: E402
from oracle import bind as orc  # noqa: E402  (checker only)


def main():
    n_keys = int(sys.argv[1]) if len(sys.argv) > 1 else 256
    steps = int(sys.argv[2]) if len(sys.argv) > 2 else 20
    eng = algo.Engine(0)
    rng = np.random.default_rng(synth.SEED + 42)

    packs_np, flat = [], []
    for _ in range(n_keys):
        uids = np.unique(synth.getuids_geometric(rng, 200_000))
        packs_np.append(uids)
        flat.append(algo.encode_flat(uids, 256))
    q_list = synth.gen_sorted_unique(rng, 200_000, int(max(p[-1] for p in packs_np)))
    dpb = eng.upload_pack_batch(flat)
    d_q = torch.from_numpy(q_list.view(np.int64)).cuda()

    plan = eng.make_pack_batch(dpb, [d_q] * n_keys)  # standing query plan

    def one_step():
        t = {}
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        lens = plan.run()
        matrix = [plan.outs[i][:lens[i]] for i in range(n_keys)]
        torch.cuda.synchronize()
        t["fanout_ms"] = (time.perf_counter() - t0) * 1e3

        t0 = time.perf_counter()
        dest = eng.merge_sorted(matrix)
        torch.cuda.synchronize()
        t["merge_or_ms"] = (time.perf_counter() - t0) * 1e3

        t0 = time.perf_counter()
        and_f = eng.intersect_sorted(matrix[:8])
        d_outs, d_lens = eng.difference_pairs([dest], [and_f])
        not_f = d_outs[0][:d_lens[0]]
        torch.cuda.synchronize()
        t["filters_ms"] = (time.perf_counter() - t0) * 1e3

        t0 = time.perf_counter()
        u_outs, u_lens = eng.intersect_pairs(matrix, [dest] * n_keys)
        torch.cuda.synchronize()
        t["update_matrix_ms"] = (time.perf_counter() - t0) * 1e3
        return t, matrix, dest, and_f, not_f, u_outs, u_lens

    # warm + oracle spot check on 3 rows
    t, matrix, dest, and_f, not_f, u_outs, u_lens = one_step()
    for i in [0, n_keys // 2, n_keys - 1]:
        want = orc.intersect_compressed_with(orc.Pack(packs_np[i], 256), 0, q_list)
        got = matrix[i].cpu().numpy().view(np.uint64)
        assert np.array_equal(got, want), f"fanout row {i}"
        want_u = orc.intersect_with(got, dest.cpu().numpy().view(np.uint64))
        assert np.array_equal(u_outs[i][:u_lens[i]].cpu().numpy().view(np.uint64),
                              want_u), f"updateUidMatrix row {i}"

    phases = {k: 0.0 for k in t}
    t0 = time.perf_counter()
    for _ in range(steps):
        t, *_ = one_step()
        for k, v in t.items():
            phases[k] += v
    total = time.perf_counter() - t0

    res = {
        "workload": "query_shape_fanout_combine",
        "n_keys": n_keys,
        "pack_uids_per_key": 200_000,
        "q_list_len": int(q_list.size),
        "dest_uids": int(dest.numel()),
        "shape_ms": round(total / steps * 1e3, 3),
        "shapes_per_s": round(steps / total, 1),
    }
    for k, v in phases.items():
        res[k] = round(v / steps, 3)
    print(json.dumps(res))


if __name__ == "__main__":
    main()
