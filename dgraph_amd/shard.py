"""Multi-GPU sharding over RCCL/xGMI.

List-pairs partition embarrassingly across the 8 GPUs of one node — the
reference itself fans out independent per-key goroutines
(worker/task.go:816, x/x.go:1005 DivideAndRule) and predicate-sharded groups
(worker/groups.go).  The only data exchange is the final MergeSorted reduce
(query.go:2290): recursive halving over torch.distributed point-to-point
send/recv — "nccl" (= RCCL) over xGMI on GPUs, "gloo" on CPU for tests.
Sorted-merge is not a sum-reduce, so no collective op applies; each round a
sender ships its run and the receiver merges two sorted runs.
"""
import numpy as np
import torch
import torch.distributed as dist


def partition_pairs(sizes, world_size):
    """Greedy bin-pack pair indices by total bytes (n+m) across ranks.
    Returns list of index-lists, one per rank."""
    order = np.argsort(sizes)[::-1]
    loads = [0] * world_size
    buckets = [[] for _ in range(world_size)]
    for idx in order:
        r = int(np.argmin(loads))
        buckets[r].append(int(idx))
        loads[r] += int(sizes[idx])
    return buckets


def merge_reduce(local, merge_fn, device=None, group=None):
    """Recursive-halving merge of per-rank sorted lists; result lands on
    rank 0 (other ranks return None).  `local` is a 1-D int64 tensor (u64
    bits); `merge_fn(a, b) -> tensor` is the dedup'd sorted merge (the GPU
    engine's merge_pairs on CUDA; a reference merge in CPU tests).
    log2(world) rounds of p2p — 3 rounds at 8 GPUs."""
    rank = dist.get_rank(group)
    world = dist.get_world_size(group)
    cur = local
    step = 1
    while step < world:
        if rank % (2 * step) == step:
            dst = rank - step
            n = torch.tensor([cur.numel()], dtype=torch.int64, device=cur.device)
            dist.send(n, dst=dst, group=group)
            if cur.numel() > 0:
                dist.send(cur, dst=dst, group=group)
            return None  # this rank is done
        elif rank % (2 * step) == 0 and rank + step < world:
            src = rank + step
            n = torch.tensor([0], dtype=torch.int64, device=cur.device)
            dist.recv(n, src=src, group=group)
            if int(n.item()) > 0:
                other = torch.empty(int(n.item()), dtype=torch.int64,
                                    device=cur.device)
                dist.recv(other, src=src, group=group)
                cur = merge_fn(cur, other)
        step *= 2
    return cur if rank == 0 else None
