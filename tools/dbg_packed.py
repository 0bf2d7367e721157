import numpy as np
import torch
import sys
sys.path.insert(0, "/root/repo")
from dgraph_amd import algo, synth
from oracle import bind as orc

eng = algo.Engine(0)

def run(size, bs, m, seed):
    rng = np.random.default_rng(seed)
    pack_uids = np.unique(synth.getuids_geometric(rng, max(size, 1))[:size])
    take = rng.choice(pack_uids.size, size=min(m // 2, pack_uids.size), replace=False)
    hi = int(pack_uids[-1]) + 1000
    v = np.unique(np.concatenate([pack_uids[np.sort(take)],
                                  rng.integers(0, hi, size=m // 2, dtype=np.uint64)]))
    bases, nums, offs, blob, total = algo.encode_flat(pack_uids, bs)
    dp = eng.upload_pack(bases, nums, offs, blob, bs)
    dv = torch.from_numpy(v.view(np.int64)).cuda()
    opack = orc.Pack(pack_uids, bs)
    for after in [0, int(pack_uids[pack_uids.size // 3]), int(pack_uids[pack_uids.size // 3]) + 1]:
        got = eng.intersect_packed(dp, after, dv).cpu().numpy().view(np.uint64)
        want = orc.intersect_compressed_with(opack, after, v)
        ok = got.tolist() == want.tolist()
        print(f"size={size} bs={bs} m={m} after={after}: got={got.size} want={want.size} ok={ok}")
        if not ok and got.size and want.size:
            # first mismatch
            n = min(got.size, want.size)
            d = np.nonzero(got[:n] != want[:n])[0]
            print("  first diff idx:", d[:5], "got:", got[d[:5]] if d.size else None,
                  "want:", want[d[:5]] if d.size else None)
        gd = eng.decode_pack(dp, after).cpu().numpy().view(np.uint64)
        wd = opack.decode(after)
        print(f"   decode after={after}: got={gd.size} want={wd.size} ok={gd.tolist()==wd.tolist()}")

run(300, 10, 300, 0xD6A77 + 300 + 10)
run(5000, 0, 500, 0xD6A77 + 5000 + 0)
run(200000, 256, 50000, 0xD6A77 + 200000 + 256)
