"""GPU range-deletion parity (SURVEY §8f-5 envelope subset): output SSTs
bit-identical to the oracle on jobs whose inputs carry range tombstones;
out-of-envelope jobs refused by both sides."""
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


def K(i):
    return b"k%015d" % i


def _write(tmp_path, name, kvs, tombstones=()):
    p = str(tmp_path / name)
    with open(p, "wb") as f:
        f.write(oracle.build_sst(kvs, tombstones=tombstones))
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
        assert open(fg["path"], "rb").read() == open(fo["path"], "rb").read()
    return rg, ro


def test_rangedel_covered_keys_dropped(tmp_path):
    kvs = [(oracle.make_ikey(K(i), 100 + i, 1), b"v%d" % i)
           for i in range(20000)]
    p = _write(tmp_path, "a.sst", kvs,
               [(K(1000), K(3000), 900_000), (K(10_000), K(10_500), 900_001)])
    rg, _ = run_both(tmp_path, [[p]], bottommost_level=1)
    assert rg["out_entries"] == 20000 - 2000 - 500


def test_rangedel_across_runs_fuzz(tmp_path):
    rnd = random.Random(17)
    runs = []
    seq = 1
    all_ts = []
    for r in range(3):
        kvs = []
        for i in range(8000):
            kvs.append((oracle.make_ikey(K(rnd.randrange(30000)),
                                         seq, 1), b"v%d" % seq))
            seq += 1
        kvs = sorted({k: v for k, v in kvs}.items(),
                     key=lambda e: (e[0][:-8],
                                    -int.from_bytes(e[0][-8:], "little")))
        ts = []
        for _ in range(rnd.randrange(4)):
            a = rnd.randrange(30000)
            b = a + rnd.randrange(1, 4000)
            ts.append((K(a), K(b), seq))
            seq += 1
        all_ts += ts
        runs.append([_write(tmp_path, "r%d.sst" % r, kvs, ts)])
    run_both(tmp_path, runs, bottommost_level=1)


def test_rangedel_newer_keys_survive(tmp_path):
    kvs = [(oracle.make_ikey(K(i), 500_000 + i, 1), b"n") for i in range(5000)]
    p = _write(tmp_path, "n.sst", kvs, [(K(0), K(9000), 1000)])
    rg, _ = run_both(tmp_path, [[p]], bottommost_level=1)
    assert rg["out_entries"] == 5000


def test_rangedel_envelope_refused_both(tmp_path):
    kvs = [(oracle.make_ikey(K(i), 100 + i, 1), b"v") for i in range(100)]
    p = _write(tmp_path, "e.sst", kvs, [(K(0), K(50), 500)])
    for kw in (dict(bottommost_level=0),
               dict(bottommost_level=1, snapshots=[400])):
        og = tmp_path / ("ge%d" % kw.get("bottommost_level"))
        og.mkdir(exist_ok=True)
        with pytest.raises(RuntimeError, match="envelope"):
            dcw.execute(dcw.make_job([[p]], str(og), **kw))
        with pytest.raises(RuntimeError, match="envelope"):
            oracle.execute(oracle.make_job([[p]], str(og), **kw))


def test_rangedel_with_dzt_output(tmp_path):
    # tombstones + DcwZipTable output: the FSM drop applies before the
    # DZT build; bit-exact vs oracle
    kvs = [(oracle.make_ikey(K(i), 100 + i, 1), b"val-%d" % i)
           for i in range(15000)]
    p = _write(tmp_path, "z.sst", kvs, [(K(2000), K(6000), 777_000)])
    rg, _ = run_both(tmp_path, [[p]], bottommost_level=1,
                     output_table_factory=1)
    assert rg["out_entries"] == 11000
