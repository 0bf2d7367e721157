#!/usr/bin/env python3
"""cfg 5 shape (BASELINE configs[4]): shard-parallel list-pairs + MergeSorted
reduce over torch.distributed (RCCL/xGMI on GPUs).

Run under torchrun (one rank per GPU):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 tools/cfg5_bench.py [n_pairs] [hi]

Each rank: greedy bin-packed share of the Zipf pair set (shard.partition_pairs
— every rank computes the same partition from the same seed), batched
intersect on its GPU, per-rank MergeSorted of its outputs (the engine's
device-chained union tree), then shard.merge_reduce over nccl(=RCCL)
point-to-point — the reference's final MergeSorted combine (query.go:2290).
Rank 0 prints one JSON line with the whole-job shape and per-phase times.

At world_size 1 this still exercises the full nccl init + partition +
reduce wiring (merge_reduce degenerates to identity) — the VERDICT r01 ask
that the RCCL leg not first execute on the driver's 8-GPU run.
"""
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402

from dgraph_amd import algo, shard, synth  # noqa: E402


def gen_sorted(rng, n):
    deltas = rng.integers(1, 17, size=n, dtype=np.uint64)
    return (np.uint64(rng.integers(0, 1000)) + np.cumsum(deltas)).astype(np.uint64)


def main():
    n_pairs = int(sys.argv[1]) if len(sys.argv) > 1 else 2048
    hi = int(sys.argv[2]) if len(sys.argv) > 2 else 1_000_000
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_gpu = torch.cuda.is_available()
    backend = "nccl" if use_gpu else "gloo"
    if world > 1 or os.environ.get("UA_FORCE_DIST"):
        dist.init_process_group(backend)
    else:
        dist.init_process_group(
            backend, init_method="tcp://127.0.0.1:29701", rank=0, world_size=1)
    if use_gpu:
        torch.cuda.set_device(local_rank)
    dev = f"cuda:{local_rank}"

    # identical partition on every rank (same seed -> same sizes)
    rng = np.random.default_rng(synth.SEED + 5)
    sizes_u = synth.zipf_sizes(rng, n_pairs, lo=1000, hi=hi)
    sizes_v = synth.zipf_sizes(rng, n_pairs, lo=1000, hi=hi)
    buckets = shard.partition_pairs(sizes_u + sizes_v, world)
    mine = buckets[rank]

    eng = algo.Engine(local_rank)
    # per-pair value streams seeded by PAIR INDEX so ranks are reproducible
    us, vs = [], []
    for idx in mine:
        prng = np.random.default_rng(synth.SEED + 100 + idx)
        us.append(torch.from_numpy(
            gen_sorted(prng, int(sizes_u[idx])).view(np.int64)).to(dev))
        vs.append(torch.from_numpy(
            gen_sorted(prng, int(sizes_v[idx])).view(np.int64)).to(dev))
    outs = [torch.empty(min(u.numel(), v.numel()), dtype=torch.int64, device=dev)
            for u, v in zip(us, vs)]

    torch.cuda.synchronize()
    dist.barrier()
    t0 = time.perf_counter()
    outs, lens = eng.intersect_pairs(us, vs, outs)
    torch.cuda.synchronize()
    t1 = time.perf_counter()
    # per-rank MergeSorted over its intersect outputs (device union tree)
    local_merged = eng.merge_sorted([o[:n] for o, n in zip(outs, lens)])
    torch.cuda.synchronize()
    t2 = time.perf_counter()

    def gpu_merge(a, b):
        mouts, mlens = eng.merge_pairs([a], [b])
        return mouts[0][:mlens[0]]

    reduced = shard.merge_reduce(local_merged, gpu_merge)
    torch.cuda.synchronize()
    dist.barrier()
    t3 = time.perf_counter()

    times = torch.tensor([t1 - t0, t2 - t1, t3 - t2], dtype=torch.float64,
                         device=dev if use_gpu else "cpu")
    dist.all_reduce(times, op=dist.ReduceOp.MAX)
    total_elems = int(sizes_u.sum() + sizes_v.sum())
    if rank == 0:
        print(json.dumps({
            "workload": "cfg5_shard_parallel", "n_pairs": n_pairs,
            "world": world, "backend": backend, "size_hi": hi,
            "total_elems": total_elems,
            "intersect_s": round(float(times[0]), 4),
            "local_merge_s": round(float(times[1]), 4),
            "reduce_s": round(float(times[2]), 4),
            "reduced_len": int(reduced.numel()),
            "elems_per_s": round(total_elems / float(times.sum()), 1),
        }), flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
