"""Engine robustness on GPU: error paths, concurrent use of one context,
multiple contexts on one device, stats accounting."""
import threading

import numpy as np
import pytest
import torch

from dgraph_amd import algo, synth
from oracle import bind as orc

pytestmark = pytest.mark.gpu

SEED = synth.SEED


def to_dev(a):
    a = np.ascontiguousarray(a, dtype=np.uint64)
    if a.size == 0:
        return torch.empty(0, dtype=torch.int64, device="cuda:0")
    return torch.from_numpy(a.view(np.int64)).to("cuda:0")


def to_np(t):
    return t.cpu().numpy().view(np.uint64)


def test_encode_dev_rejects_oversize_block():
    eng = algo.Engine(0)
    try:
        with pytest.raises(RuntimeError):
            eng.encode_dev(to_dev(np.arange(10, dtype=np.uint64)), 512)
    finally:
        eng.close()


def test_bad_device_fails_loudly():
    with pytest.raises(RuntimeError):
        algo.Engine(127)


def test_two_engines_one_device():
    e1 = algo.Engine(0)
    e2 = algo.Engine(0)
    try:
        rng = np.random.default_rng(SEED)
        u = synth.gen_sorted_unique(rng, 100_000, 400_000)
        v = synth.gen_sorted_unique(rng, 100_000, 400_000)
        want = orc.intersect_with(u, v)
        o1, l1 = e1.intersect_pairs([to_dev(u)], [to_dev(v)])
        o2, l2 = e2.merge_pairs([to_dev(u)], [to_dev(v)])
        assert to_np(o1[0][:l1[0]]).tolist() == want.tolist()
        assert to_np(o2[0][:l2[0]]).tolist() == orc.merge_sorted([u, v]).tolist()
    finally:
        e1.close()
        e2.close()


def test_concurrent_threads_one_engine():
    """8 threads hammering one ctx with compound ops: whole-op locking must
    keep workspace reuse safe (results all bit-exact)."""
    eng = algo.Engine(0)
    try:
        rng = np.random.default_rng(SEED)
        datasets = []
        for _ in range(8):
            u = synth.gen_sorted_unique(rng, int(rng.integers(1000, 60_000)), 300_000)
            v = synth.gen_sorted_unique(rng, int(rng.integers(1000, 60_000)), 300_000)
            w = synth.gen_sorted_unique(rng, int(rng.integers(1000, 60_000)), 300_000)
            datasets.append((u, v, w))
        errs = []

        def worker(i):
            try:
                u, v, w = datasets[i]
                for _ in range(5):
                    got_i = algo.intersect_sorted(eng, [u, v, w])
                    assert got_i.tolist() == orc.intersect_sorted([u, v, w]).tolist()
                    got_m = algo.merge_sorted(eng, [u, v, w])
                    assert got_m.tolist() == orc.merge_sorted([u, v, w]).tolist()
            except Exception as e:  # noqa: BLE001
                errs.append((i, repr(e)))

        ts = [threading.Thread(target=worker, args=(i,)) for i in range(8)]
        for t in ts:
            t.start()
        for t in ts:
            t.join(timeout=120)
        assert not errs, errs
    finally:
        eng.close()


def test_stats_accounting():
    eng = algo.Engine(0)
    try:
        rng = np.random.default_rng(SEED)
        u = synth.gen_sorted_unique(rng, 500_000, 2_000_000)
        v = synth.gen_sorted_unique(rng, 500_000, 2_000_000)
        du, dv = to_dev(u), to_dev(v)
        eng.stats_reset()
        outs, lens = eng.intersect_pairs([du], [dv])
        st = eng.stats()
        assert st["launches"] == 1
        assert st["kernel_ms"] > 0
        want_bytes = 8 * (u.size + v.size + lens[0])
        assert st["bytes_algorithmic"] == want_bytes
        eng.stats_reset()
        assert eng.stats()["launches"] == 0
    finally:
        eng.close()
