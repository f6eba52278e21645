"""GPU bloom-filter build parity: output SSTs with filter
blocks bit-identical to the oracle (XXPH3 + FastLocalBloom on device)."""
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


def gen_runs(tmp_path, n_runs, entries, seed0=1, compression=1):
    runs = []
    for r in range(n_runs):
        p = str(tmp_path / ("f%d.sst" % r))
        dcw.gen_sst(p, seed=seed0 + r, num_entries=entries,
                    seq_base=1 + r * entries, compression=compression)
        runs.append([p])
    return runs


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
        da = open(fg["path"], "rb").read()
        db = open(fo["path"], "rb").read()
        assert da == db, fg["path"]
    return rg, ro


def test_filter_bit_exact(tmp_path):
    runs = gen_runs(tmp_path, 4, 40000)
    rg, _ = run_both(tmp_path, runs, compression=1, bottommost_level=1,
                     bloom_millibits_per_key=10000)
    assert b"fullfilter" in open(rg["files"][0]["path"], "rb").read()


def test_filter_multi_file_cuts(tmp_path):
    runs = gen_runs(tmp_path, 4, 60000)
    rg, _ = run_both(tmp_path, runs, compression=1, bottommost_level=1,
                     bloom_millibits_per_key=10000, target_file_size=4 << 20)
    assert len(rg["files"]) > 1


def test_filter_other_bits_per_key(tmp_path):
    runs = gen_runs(tmp_path, 2, 20000)
    for mb in (6000, 15500, 23000):
        og = tmp_path / ("g%d" % mb)
        oo = tmp_path / ("o%d" % mb)
        og.mkdir()
        oo.mkdir()
        rg = dcw.execute(dcw.make_job(runs, str(og), compression=1,
                                      bloom_millibits_per_key=mb))
        ro = oracle.execute(oracle.make_job(runs, str(oo), compression=1,
                                            bloom_millibits_per_key=mb))
        for fg, fo in zip(rg["files"], ro["files"]):
            assert open(fg["path"], "rb").read() == open(fo["path"], "rb").read()


def test_filter_general_keys(tmp_path):
    rnd = random.Random(8)
    kvs = []
    seq = 1
    for _ in range(20000):
        uk = bytes(rnd.getrandbits(8) for _ in range(rnd.choice([8, 16, 40])))
        kvs.append((uk, seq, 1, b"v%d" % seq))
        seq += 1
    kvs.sort(key=lambda e: (e[0], -e[1]))
    es = [(oracle.make_ikey(k, s, t), v) for k, s, t, v in kvs]
    p = str(tmp_path / "gk.sst")
    with open(p, "wb") as f:
        f.write(oracle.build_sst(es))
    run_both(tmp_path, [[p]], bottommost_level=1,
             bloom_millibits_per_key=10000)


def test_filter_with_flush(tmp_path):
    rnd = random.Random(9)
    entries = []
    seq = 1
    for _ in range(15000):
        uk = b"k" + bytes(rnd.getrandbits(8) for _ in range(15))
        entries.append((uk, seq, 1, b"v%d" % seq))
        seq += 1
    entries.sort(key=lambda e: (e[0], -e[1]))
    es = [(oracle.make_ikey(k, s, t), v) for k, s, t, v in entries]
    og = tmp_path / "g"
    oo = tmp_path / "o"
    og.mkdir()
    oo.mkdir()
    rg = dcw.execute(dcw.make_job([], str(og), flush_entries=es,
                                  output_level=0, compression=1,
                                  bloom_millibits_per_key=10000))
    ro = oracle.execute(oracle.make_job([], str(oo), flush_entries=es,
                                        output_level=0, compression=1,
                                        bloom_millibits_per_key=10000))
    for fg, fo in zip(rg["files"], ro["files"]):
        assert open(fg["path"], "rb").read() == open(fo["path"], "rb").read()
