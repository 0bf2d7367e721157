"""CPU-side checks of the product C-ABI library (no GPU compute):
- libuidalgo.so loads and exports every symbol include/uidalgo.h declares,
- the host-side codec (ua_encode/flatten) and ua_index_of work,
- the product encoder agrees byte-for-byte with the oracle encoder
  (both restate the go-groupvarint + codec.go block format),
- GPU context creation fails LOUDLY (UA_ERR_NO_GPU) when no device exists.
"""
import ctypes as C
import os
import re

import numpy as np
import pytest

from dgraph_amd import _lib
from dgraph_amd import algo
from oracle import bind as orc

HDR = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                   "include", "uidalgo.h")


def header_exports():
    txt = open(HDR).read()
    # function declarations: "int ua_xxx(...)" / "void ua_xxx(...)" etc.
    names = re.findall(r"^\s*(?:const char|int64_t|uint64_t|int|void)\s+\*?(ua_\w+)\s*\(",
                       txt, re.M)
    return sorted(set(names))


def test_lib_loads_and_exports_all():
    L = _lib.lib()
    names = header_exports()
    assert len(names) >= 25
    dll = C.CDLL(os.path.join(os.path.dirname(_lib.__file__), "libuidalgo.so"))
    for n in names:
        assert hasattr(dll, n), f"missing C-ABI export {n}"
    assert L.ua_version() >= 1


def test_no_gpu_fails_loudly():
    import torch
    if torch.cuda.is_available():
        pytest.skip("GPU present")
    ctx = C.c_void_p()
    rc = _lib.lib().ua_ctx_create(C.byref(ctx), 0)
    assert rc != 0  # UA_ERR_NO_GPU — no silent CPU fallback


def test_host_index_of():
    u = np.array([1, 5, 9, 11], dtype=np.uint64)
    assert algo.index_of(u, 9) == 2
    assert algo.index_of(u, 4) == -1
    assert algo.index_of(np.empty(0, dtype=np.uint64), 4) == -1


@pytest.mark.parametrize("size,bs", [(0, 256), (1, 256), (5, 0), (1000, 10),
                                     (100_000, 256), (4096, 1)])
def test_product_encoder_matches_oracle_bytes(size, bs):
    rng = np.random.default_rng(0xD6A77 + size)
    deltas = rng.integers(0, 33, size=max(size, 1), dtype=np.uint64)
    uids = np.cumsum(deltas).astype(np.uint64)[:size]
    # sprinkle 32-MSB jumps to force splits (codec.go:117)
    if size > 10:
        uids[size // 2:] += np.uint64(1) << np.uint64(33)
        uids[3 * size // 4:] += np.uint64(1) << np.uint64(40)

    bases, nums, offs, blob, total = algo.encode_flat(uids, bs)
    assert total == uids.size

    opack = orc.Pack(uids, bs)
    obases, onums, ooffs, oblob = opack.flatten()
    assert bases.tolist() == obases.tolist()
    assert nums.tolist() == onums.tolist()
    assert offs.tolist() == ooffs.tolist()
    assert blob.tolist() == oblob.tolist()
    # and the oracle decoder inverts the product encoder
    assert opack.decode(0).tolist() == uids.tolist()


def test_apply_filter_mask():
    u = np.array([1, 2, 3, 4, 5], dtype=np.uint64)
    got = algo.apply_filter(u, (u % 2) == 1)
    assert got.tolist() == [1, 3, 5]


def test_oracle_apply_filter():
    """orc_apply_filter (uidlist.go:21) vs numpy + the golden table shape."""
    rng = np.random.default_rng(0xD6A77)
    for n in [0, 1, 5, 1000, 4097]:
        u = np.sort(rng.integers(0, 1 << 48, size=n, dtype=np.uint64))
        mask = rng.integers(0, 2, size=n, dtype=np.uint8).astype(bool)
        got = orc.apply_filter(u, mask)
        assert got.tolist() == u[mask].tolist()
    u = np.array([1, 2, 3, 4, 5], dtype=np.uint64)
    assert orc.apply_filter(u, (u % 2) == 1).tolist() == [1, 3, 5]
