"""General key shapes (VERDICT round-1 item 8): mixed/long user keys up to
48 B through the prefix-normkey + full-key side table; outputs bit-exact
vs the oracle (db/dbformat.h:1057-1096 arbitrary-length bytewise
contract).  Jobs beyond the envelope are refused."""
import os
import random

import pytest

import oracle
import toplingdb_amd as dcw

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def gpu():
    dcw.init(0)
    yield
    dcw.shutdown()


def ikey_sort(kvs):
    return sorted(kvs, key=lambda e: (e[0], -((e[1] << 8) | e[2])))


def _write(tmp_path, name, kvs):
    es = [(oracle.make_ikey(k, s, t), v) for k, s, t, v in ikey_sort(kvs)]
    p = str(tmp_path / name)
    with open(p, "wb") as f:
        f.write(oracle.build_sst(es))
    return p


def run_both(tmp_path, runs, **kw):
    og = tmp_path / "g"
    oo = tmp_path / "o"
    og.mkdir()
    oo.mkdir()
    rg = dcw.execute(dcw.make_job(runs, str(og), **kw))
    ro = oracle.execute(oracle.make_job(runs, str(oo), **kw))
    assert rg["out_entries"] == ro["out_entries"]
    assert len(rg["files"]) == len(ro["files"])
    for fg, fo in zip(rg["files"], ro["files"]):
        assert open(fg["path"], "rb").read() == open(fo["path"], "rb").read(), \
            fg["path"]
        assert fg["smallest"] == fo["smallest"]
        assert fg["largest"] == fo["largest"]
    return rg, ro


def test_mixed_length_keys(tmp_path):
    rnd = random.Random(4)
    runs = []
    seq = 1
    for r in range(3):
        kvs = []
        for _ in range(15000):
            ln = rnd.choice([6, 10, 16, 24, 33, 48])
            uk = bytes(rnd.getrandbits(8) for _ in range(ln))
            kvs.append((uk, seq, 1, b"v%d" % seq))
            seq += 1
        runs.append([_write(tmp_path, "m%d.sst" % r, kvs)])
    run_both(tmp_path, runs, bottommost_level=1, compression=1)


def test_prefix_relationship_ordering(tmp_path):
    # "ab" < "ab\\x00" < "ab\\x00x" < "abc": the exact case the zero-padded
    # prefix normkey alone cannot order (needs the length tie-break)
    kvs = []
    seq = 1
    for uk in [b"ab", b"ab\x00", b"ab\x00x", b"abc", b"abcd" * 8,
               b"ab" + b"\x00" * 20, b"ab" + b"\x00" * 20 + b"x"]:
        for _ in range(3):
            kvs.append((uk, seq, 1, b"v%d" % seq))
            seq += 1
    p = _write(tmp_path, "p.sst", kvs)
    rg, _ = run_both(tmp_path, [[p]], bottommost_level=1)
    assert rg["out_entries"] == 7  # newest version of each key


def test_long_uniform_keys_48(tmp_path):
    rnd = random.Random(9)
    runs = []
    for r in range(2):
        kvs = [(b"%048d" % rnd.randrange(10**12), 100 + r * 9000 + i, 1,
                b"val%d" % i) for i in range(9000)]
        kvs = list({k: (k, s, t, v) for k, s, t, v in kvs}.values())
        runs.append([_write(tmp_path, "l%d.sst" % r, kvs)])
    run_both(tmp_path, runs, bottommost_level=1, compression=1,
             target_file_size=2 << 20)


def test_mixed_keys_with_tombstones_fsm(tmp_path):
    # deletes/single-deletes still flow through the FSM in general mode
    rnd = random.Random(21)
    kvs = []
    seq = 1
    keys = [bytes(rnd.getrandbits(8) for _ in range(rnd.choice([5, 20, 40])))
            for _ in range(800)]
    for uk in keys:
        for _ in range(rnd.randrange(1, 4)):
            t = 0 if rnd.random() < 0.3 else 1
            kvs.append((uk, seq, t, b"" if t == 0 else b"v%d" % seq))
            seq += 1
    p = _write(tmp_path, "t.sst", kvs)
    for bottom in (1, 0):
        og = tmp_path / ("bg%d" % bottom)
        oo = tmp_path / ("bo%d" % bottom)
        og.mkdir()
        oo.mkdir()
        rg = dcw.execute(dcw.make_job([[p]], str(og), bottommost_level=bottom))
        ro = oracle.execute(oracle.make_job([[p]], str(oo),
                                            bottommost_level=bottom))
        assert rg["out_entries"] == ro["out_entries"]
        for fg, fo in zip(rg["files"], ro["files"]):
            assert open(fg["path"], "rb").read() == open(fo["path"], "rb").read()


def test_keys_over_48_refused(tmp_path):
    kvs = [(b"x" * 60, 5, 1, b"v")]
    p = _write(tmp_path, "big.sst", kvs)
    out = tmp_path / "out"
    out.mkdir()
    with pytest.raises(RuntimeError):
        dcw.execute(dcw.make_job([[p]], str(out)))


def test_general_plus_grandparents_refused(tmp_path):
    kvs = [(bytes([i]) * (8 + i % 20), 10 + i, 1, b"v") for i in range(64)]
    p = _write(tmp_path, "gg.sst", kvs)
    out = tmp_path / "out"
    out.mkdir()
    jd = dcw.make_job([[p]], str(out),
                      grandparents=[(b"a" * 16, b"z" * 16, 1 << 20)])
    with pytest.raises(RuntimeError, match="envelope"):
        dcw.execute(jd)
