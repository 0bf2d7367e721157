"""Oracle vs the reference's own known-answer tables (tests/golden/golden.json,
transcribed from algo/uidlist_test.go + codec_test.go) — this is what pins the
oracle to the reference."""
import json
import os

import numpy as np
import pytest

from oracle import bind as orc

GOLDEN = json.load(open(os.path.join(os.path.dirname(__file__), "golden", "golden.json")))


def u64(x):
    return np.array(x, dtype=np.uint64)


@pytest.mark.parametrize("case", GOLDEN["merge_sorted"], ids=lambda c: c["name"])
def test_merge_sorted_golden(case):
    got = orc.merge_sorted([u64(l) for l in case["input"]])
    assert got.tolist() == case["out"]


@pytest.mark.parametrize("case", GOLDEN["intersect_sorted"], ids=lambda c: c["name"])
def test_intersect_sorted_golden(case):
    got = orc.intersect_sorted([u64(l) for l in case["input"]])
    assert got.tolist() == case["out"]


@pytest.mark.parametrize("case", GOLDEN["difference"], ids=lambda c: c["name"])
def test_difference_golden(case):
    got = orc.difference(u64(case["u"]), u64(case["v"]))
    assert got.tolist() == case["out"]


@pytest.mark.parametrize("case", GOLDEN["intersect_with"], ids=lambda c: c["name"])
def test_intersect_with_golden(case):
    got = orc.intersect_with(u64(case["u"]), u64(case["v"]))
    assert got.tolist() == case["out"]


@pytest.mark.parametrize("case", GOLDEN["seek"]["cases"],
                         ids=lambda c: f"in={c['in']}-{c['whence']}")
def test_seek_golden(case):
    spec = GOLDEN["seek"]["uids_spec"]
    uids = np.arange(spec["start"], spec["stop"], spec["step"], dtype=np.uint64)
    pack = orc.Pack(uids, GOLDEN["seek"]["block_size"])
    dec = orc.Dec(pack)
    whence = orc.SEEK_START if case["whence"] == "start" else orc.SEEK_CURRENT
    got = dec.seek(case["in"], whence)
    if case.get("empty"):
        assert got.size == 0
    else:
        assert got.size > 0 and int(got[0]) == case["out"]


def test_linear_seek_sweep_golden():
    # codec_test.go:155-159 (tail of TestSeek) + TestLinearSeek :162-188
    uids = np.arange(0, 10001, 10, dtype=np.uint64)
    pack = orc.Pack(uids, 10)
    dec = orc.Dec(pack)
    for i in range(100, 10000, 100):
        got = dec.linear_seek(i)
        assert i in got.tolist()

    dec2 = orc.Dec(pack)
    N = 10001
    for i in range(0, 2 * N, 10):
        got = dec2.linear_seek(i)
        if i < N:
            assert i in got.tolist()
        else:
            assert i not in got.tolist()
    # decoder is stateful: blockIdx parked at the last block; earlier values
    # are no longer reachable (codec_test.go:182-188)
    for i in range(0, 9990, 10):
        got = dec2.linear_seek(i)
        assert i not in got.tolist()


def test_encoding_32msb_golden():
    # codec_test.go:306-334: high-32-MSB values force block splits; roundtrip.
    bigints = [np.uint64(x) for x in GOLDEN["encoding_32msb"]["bigints"]]
    rng = np.random.default_rng(0xD6A77)
    for n in [0, 1, 2, 3, 5, 13, 18, 100, 99, 98]:
        vals = []
        for i in range(min(50, n)):
            vals.append(np.uint64(rng.integers(0, 2**32)))
        for i in range(50, n):
            vals.append(np.uint64(rng.integers(0, 2**32)) + bigints[rng.integers(0, 5)])
        ints = np.sort(np.array(vals, dtype=np.uint64))
        pack = orc.Pack(ints, 256)
        got = pack.decode(0)
        assert got.tolist() == ints.tolist()
        # every block must share one 32-MSB prefix (codec.go:117)
        bases, nums, offs, blob = pack.flatten()
        dec_all = pack.decode(0)
        pos = 0
        for b in range(len(bases)):
            blk = dec_all[pos:pos + nums[b]]
            assert np.all((blk >> np.uint64(32)) == (blk[0] >> np.uint64(32)))
            pos += nums[b]


def test_apply_filter_golden():
    # ApplyFilter (uidlist.go:21) is host-side compaction; the engine's
    # equivalent is a boolean-mask compaction. Semantics check vs numpy.
    inp = u64(GOLDEN["apply_filter"]["input"])
    keep = (inp % 2) == 1
    assert inp[keep].tolist() == GOLDEN["apply_filter"]["out"]
