#!/usr/bin/env python3
"""cfg 4 measurement: fused group-varint decode +
IntersectCompressedWith on a 21M-uid synthetic pack (geometric deltas,
mirrors codec_test.go:26-35) vs a 1M list, plus pure GPU decode rate.

The reference's own published codec numbers (codec/benchmark/benchmark.go:
50-52): pack 30 M uids/s, unpack 116 M uids/s, single-core Go.
Parity of this path is gated by tests/test_parity_gpu.py; this tool reports
throughput only.  Prints one JSON line.
"""
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from dgraph_amd import algo, synth  # noqa: E402


def main():
    steps = int(sys.argv[1]) if len(sys.argv) > 1 else 50
    eng = algo.Engine(0)
    rng = np.random.default_rng(synth.SEED)
    pack_uids = np.unique(synth.getuids_geometric(rng, 21_000_000))
    v = np.unique(np.concatenate([
        pack_uids[np.sort(rng.choice(pack_uids.size, 500_000, replace=False))],
        rng.integers(0, int(pack_uids[-1]) + 1000, size=500_000, dtype=np.uint64)]))

    t0 = time.perf_counter()
    bases, nums, offs, blob, total = algo.encode_flat(pack_uids, 256)
    enc_s = time.perf_counter() - t0
    dp = eng.upload_pack(bases, nums, offs, blob, 256)
    dv = torch.from_numpy(v.view(np.int64)).cuda()

    # warmup + sanity: decode inverts encode
    out_i = eng.intersect_packed(dp, 0, dv)
    dec = eng.decode_pack(dp, 0)
    assert dec.numel() == pack_uids.size
    assert np.array_equal(dec.cpu().numpy().view(np.uint64), pack_uids)

    out_buf = torch.empty(min(total, v.size), dtype=torch.int64, device="cuda")
    torch.cuda.synchronize()
    eng.stats_reset()
    t0 = time.perf_counter()
    for _ in range(steps):
        eng.intersect_packed(dp, 0, dv, out=out_buf)
    torch.cuda.synchronize()
    el_i = time.perf_counter() - t0
    st_i = eng.stats()

    out_d = torch.empty(total, dtype=torch.int64, device="cuda")
    torch.cuda.synchronize()
    eng.stats_reset()
    t0 = time.perf_counter()
    for _ in range(steps):
        eng.decode_pack(dp, 0, out=out_d)
    torch.cuda.synchronize()
    el_d = time.perf_counter() - t0
    st_d = eng.stats()

    # GPU encode (ua_encode_dev)
    d_uids = torch.from_numpy(pack_uids.view(np.int64)).cuda()
    dp2 = eng.encode_dev(d_uids, 256)  # warm
    assert dp2.bases.numel() == bases.size
    torch.cuda.synchronize()
    eng.stats_reset()
    t0 = time.perf_counter()
    for _ in range(steps):
        eng.encode_dev(d_uids, 256)
    torch.cuda.synchronize()
    el_e = time.perf_counter() - t0
    st_e = eng.stats()

    res = {
        "workload": "cfg4_packed_21M",
        "gpu_encode_Muids_per_s": round(pack_uids.size * steps / el_e / 1e6, 1),
        "gpu_encode_kernel_ms": round(st_e["kernel_ms"] / max(st_e["launches"], 1), 4),
        "pack_uids": int(pack_uids.size),
        "pack_bytes": int(blob.size + 20 * bases.size),
        "bytes_per_uid": round((blob.size + 20 * bases.size) / pack_uids.size, 3),
        "v_len": int(v.size),
        "intersect_out": int(out_i.numel()),
        "fused_decode_intersect_Muids_per_s": round(pack_uids.size * steps / el_i / 1e6, 1),
        "fused_kernel_ms": round(st_i["kernel_ms"] / max(st_i["launches"], 1), 4),
        "decode_Muids_per_s": round(pack_uids.size * steps / el_d / 1e6, 1),
        "decode_kernel_ms": round(st_d["kernel_ms"] / max(st_d["launches"], 1), 4),
        "decode_kernel_Muids_per_s": round(
            pack_uids.size / (st_d["kernel_ms"] / max(st_d["launches"], 1)) / 1e3, 1),
        "host_encode_Muids_per_s": round(pack_uids.size / enc_s / 1e6, 1),
        "reference_go_unpack_Muids_per_s": 116,
        "reference_go_pack_Muids_per_s": 30,
    }
    print(json.dumps(res))


if __name__ == "__main__":
    main()
