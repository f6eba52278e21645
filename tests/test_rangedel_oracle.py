"""Range-deletion support (envelope subset): the
"rocksdb.range_del" meta block round-trips through the table
builder/reader, covered point keys are dropped at bottommost compaction
(range_del_aggregator.cc:407-413 + compaction_iterator.cc:1056-1063), and
out-of-envelope jobs are refused (DB-side local fallback)."""
import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import oracle


def K(i):
    return b"k%015d" % i


def test_tombstone_block_roundtrip(tmp_path):
    kvs = [(oracle.make_ikey(K(i), 100 + i, 1), b"v%d" % i) for i in range(100)]
    ts = [(K(10), K(20), 500), (K(50), K(90), 700)]
    data = oracle.build_sst(kvs, tombstones=ts)
    assert oracle.read_tombstones(data) == ts
    # point entries unaffected
    assert len(oracle.read_sst(data)) == 100
    # absent block -> empty list
    assert oracle.read_tombstones(oracle.build_sst(kvs)) == []


def _run(tmp_path, name, kvs, tombstones=()):
    p = str(tmp_path / name)
    with open(p, "wb") as f:
        f.write(oracle.build_sst(kvs, tombstones=tombstones))
    return p


def _compact(tmp_path, runs, **kw):
    out = tmp_path / ("out%d" % len(list(tmp_path.iterdir())))
    out.mkdir()
    r = oracle.execute(oracle.make_job(runs, str(out), **kw))
    stream = []
    for f in r["files"]:
        stream += oracle.read_sst(open(f["path"], "rb").read())
    return r, stream


def test_covered_keys_dropped_bottommost(tmp_path):
    kvs = [(oracle.make_ikey(K(i), 100 + i, 1), b"v%d" % i) for i in range(100)]
    # tombstone at seq 500 covers [k10, k20): keys k10..k19 have seqs
    # 110..119 < 500 -> dropped; k20 survives (end exclusive)
    p1 = _run(tmp_path, "a.sst", kvs, [(K(10), K(20), 500)])
    r, stream = _compact(tmp_path, [[p1]], bottommost_level=1)
    uks = [ik[:-8] for ik, _ in stream]
    assert K(9) in uks and K(20) in uks
    for i in range(10, 20):
        assert K(i) not in uks
    assert len(stream) == 90
    # outputs carry no tombstones (all obsolete at bottommost, no snapshots)
    for f in r["files"]:
        assert oracle.read_tombstones(open(f["path"], "rb").read()) == []


def test_tombstone_seq_ordering(tmp_path):
    # keys NEWER than the tombstone survive
    kvs = [(oracle.make_ikey(K(i), 1000 + i, 1), b"new%d" % i)
           for i in range(30)]
    p1 = _run(tmp_path, "n.sst", kvs, [(K(0), K(100), 500)])
    _, stream = _compact(tmp_path, [[p1]], bottommost_level=1)
    assert len(stream) == 30  # all newer than seq 500


def test_tombstone_across_runs(tmp_path):
    # tombstone in run A covers keys in run B (the aggregator is job-wide)
    old = [(oracle.make_ikey(K(i), 100 + i, 1), b"old%d" % i)
           for i in range(50)]
    pb = _run(tmp_path, "old.sst", old)
    pa = _run(tmp_path, "del.sst",
              [(oracle.make_ikey(K(200), 900, 1), b"x")],
              [(K(5), K(45), 800)])
    _, stream = _compact(tmp_path, [[pa], [pb]], bottommost_level=1)
    uks = [ik[:-8] for ik, _ in stream]
    assert len(stream) == 11  # 50 - 40 dropped + the k200 put
    assert K(4) in uks and K(45) in uks and K(5) not in uks


def test_overlapping_tombstones_max_seq(tmp_path):
    # overlapping tombstones: coverage by the max seq per fragment
    kvs = [(oracle.make_ikey(K(i), 600, 1), b"v") for i in range(40)]
    p1 = _run(tmp_path, "ov.sst", kvs,
              [(K(0), K(30), 500), (K(20), K(40), 700)])
    _, stream = _compact(tmp_path, [[p1]], bottommost_level=1)
    uks = [ik[:-8] for ik, _ in stream]
    # seq600 keys survive the seq500 tombstone on [0,20), die on [20,40)
    for i in range(0, 20):
        assert K(i) in uks
    for i in range(20, 40):
        assert K(i) not in uks


def test_tombstone_covers_point_delete(tmp_path):
    # a point Delete covered by a newer range tombstone is dropped too;
    # bottommost deletes vanish anyway — make the delete KEEP-worthy by
    # covering a key so the delete would otherwise surface... at
    # bottommost all deletes drop; just check no crash + counts
    kvs = [(oracle.make_ikey(K(5), 300, 0), b""),
           (oracle.make_ikey(K(6), 301, 1), b"v6")]
    p1 = _run(tmp_path, "pd.sst", kvs, [(K(0), K(10), 400)])
    _, stream = _compact(tmp_path, [[p1]], bottommost_level=1)
    assert stream == []  # delete dropped (bottommost), k6 covered


def test_envelope_refusals(tmp_path):
    kvs = [(oracle.make_ikey(K(i), 100 + i, 1), b"v") for i in range(10)]
    p1 = _run(tmp_path, "e.sst", kvs, [(K(0), K(5), 500)])
    out = tmp_path / "eo"
    out.mkdir()
    # non-bottommost
    with pytest.raises(RuntimeError, match="envelope"):
        oracle.execute(oracle.make_job([[p1]], str(out), bottommost_level=0))
    # snapshots
    with pytest.raises(RuntimeError, match="envelope"):
        oracle.execute(oracle.make_job([[p1]], str(out), bottommost_level=1,
                                       snapshots=[400]))
