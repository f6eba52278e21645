"""Flush offload (BuildTable db/builder.cc:56): a sorted
raw-KV memtable stream through the worker produces L0 SSTs bit-identical
to the oracle's flush of the same stream."""
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


def memtable_stream(rnd, nkeys, key_len=16, sd_frac=0.0):
    entries = []
    seq = 1
    for _ in range(nkeys):
        uk = b"k" + bytes(rnd.getrandbits(8) for _ in range(key_len - 1))
        nver = rnd.randrange(1, 3)
        for _ in range(nver):
            t = 7 if (sd_frac and rnd.random() < sd_frac) else \
                (0 if rnd.random() < 0.15 else 1)
            entries.append((uk, seq, t, b"" if t != 1 else b"v%d" % seq))
            seq += 1
    entries.sort(key=lambda e: (e[0], -e[1]))
    return [(oracle.make_ikey(k, s, t), v) for k, s, t, v in entries], seq


def run_both(tmp_path, flush_entries, **kw):
    og = tmp_path / "g"
    oo = tmp_path / "o"
    og.mkdir()
    oo.mkdir()
    rg = dcw.execute(dcw.make_job([], str(og), flush_entries=flush_entries,
                                  **kw))
    ro = oracle.execute(oracle.make_job([], str(oo),
                                        flush_entries=flush_entries, **kw))
    assert rg["out_entries"] == ro["out_entries"]
    assert len(rg["files"]) == len(ro["files"])
    for fg, fo in zip(rg["files"], ro["files"]):
        assert open(fg["path"], "rb").read() == open(fo["path"], "rb").read()
    return rg, ro


def test_flush_basic(tmp_path):
    es, _ = memtable_stream(random.Random(1), 30000)
    rg, _ = run_both(tmp_path, es, compression=1, output_level=0)
    assert rg["in_bytes"] == sum(8 + len(k) + len(v) for k, v in es)


def test_flush_dedup_within_memtable(tmp_path):
    # several versions per key: flush keeps only the newest visible ones
    es, seq = memtable_stream(random.Random(2), 8000)
    rg, ro = run_both(tmp_path, es, compression=1, output_level=0)
    assert rg["out_entries"] < len(es)


def test_flush_with_snapshots(tmp_path):
    es, seq = memtable_stream(random.Random(3), 8000)
    snaps = [seq // 3, 2 * seq // 3]
    run_both(tmp_path, es, compression=1, output_level=0, snapshots=snaps,
             earliest_write_conflict_snapshot=snaps[0])


def test_flush_mixed_key_lengths(tmp_path):
    rnd = random.Random(4)
    entries = []
    seq = 1
    for _ in range(5000):
        uk = bytes(rnd.getrandbits(8) for _ in range(rnd.choice([8, 16, 30])))
        entries.append((uk, seq, 1, b"v%d" % seq))
        seq += 1
    entries.sort(key=lambda e: (e[0], -e[1]))
    es = [(oracle.make_ikey(k, s, t), v) for k, s, t, v in entries]
    run_both(tmp_path, es, compression=1, output_level=0)


def test_flush_rejects_runs_plus_blob(tmp_path):
    es, _ = memtable_stream(random.Random(5), 100)
    p = str(tmp_path / "in.sst")
    dcw.gen_sst(p, seed=1, num_entries=100)
    out = tmp_path / "out"
    out.mkdir()
    jd = dcw.make_job([[p]], str(out), flush_entries=es)
    with pytest.raises(RuntimeError, match="no SST runs"):
        dcw.execute(jd)
