"""DcwZipTable ("DZT1", BASELINE.json configs[3]) oracle tests: the dict
codec round-trip, the build path through the compaction oracle, the
searchable property (point lookups without whole-file decode), and file
cutting.  The GPU builder is bit-compared against these outputs in
tests/test_dzt_gpu.py."""
import os
import random
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import oracle


def test_dict_codec_roundtrip():
    rnd = random.Random(7)
    base = bytes(rnd.getrandbits(8) for _ in range(512)) * 8  # 4KB pattern
    dictionary = base[:2048]
    for payload in (
        b"",
        b"abc",
        base[100:3000],                      # mostly dict-matchable
        bytes(rnd.getrandbits(8) for _ in range(5000)),  # incompressible
        (b"hello world " * 400)[:4096],      # self-compressible
    ):
        enc = oracle.snappy_compress_dict(dictionary, payload)
        dec = oracle.snappy_uncompress_dict(dictionary, enc)
        assert dec == payload
    # dict matches shrink the stream vs no-dict when content IS dict-like
    with_dict = oracle.snappy_compress_dict(dictionary, base[64:2048])
    without = oracle.snappy_compress_dict(b"", base[64:2048])
    assert len(with_dict) < len(without)


def test_dict_codec_adversarial_match_mix():
    # alternating 4-byte dict matches + 1-byte literals: the densest
    # copy/literal interleaving the encoder can emit.  (With the DZT
    # constants — dict <= 48 KiB, blocks <= 16 KiB — the max virtual
    # offset is 65536, so the 5-byte copy form is reachable only at one
    # corner; the device/oracle output slots are nevertheless sized for
    # the 1.4x worst case defensively.)
    rnd = random.Random(99)
    dictionary = bytes(rnd.getrandbits(8) for _ in range(48 * 1024))
    payload = bytearray()
    i = 0
    while len(payload) < 12000:
        payload += dictionary[i:i + 4] + bytes([rnd.getrandbits(8)])
        i = (i + 97 * 4) % (len(dictionary) - 4)
    payload = bytes(payload)
    enc = oracle.snappy_compress_dict(dictionary, payload)
    assert oracle.snappy_uncompress_dict(dictionary, enc) == payload


def test_dict_codec_large_offsets():
    # dictionary matches reach past 64 KiB -> 4-byte-offset copy form
    rnd = random.Random(11)
    dictionary = bytes(rnd.getrandbits(8) for _ in range(48 * 1024))
    payload = dictionary[100:108] + bytes(rnd.getrandbits(8) for _ in range(9000)) + \
        dictionary[40000:40100]
    enc = oracle.snappy_compress_dict(dictionary, payload)
    assert oracle.snappy_uncompress_dict(dictionary, enc) == payload


def _gen_runs(tmp_path, n_runs=2, entries=8000, seed0=50):
    runs = []
    for r in range(n_runs):
        p = str(tmp_path / ("dz_in%d.sst" % r))
        es = []
        rnd = random.Random(seed0 + r)
        for i in range(entries):
            uk = b"k%015d" % rnd.randrange(entries * 4)
            es.append((uk, 1 + r * entries + i, 1, b"value-%d-%d" % (r, i)))
        es = sorted(set(e[0] for e in es))
        kvs = [(oracle.make_ikey(k, 100 + j + r * entries, 1),
                b"v%d.%d:" % (r, j) + k * 3) for j, k in enumerate(es)]
        with open(p, "wb") as f:
            f.write(oracle.build_sst(kvs))
        runs.append([p])
    return runs


def _job_streams(tmp_path, runs, **kw):
    """execute with BBT and DZT factories; return both KV streams."""
    obbt = tmp_path / "bbt"
    odzt = tmp_path / "dzt"
    obbt.mkdir()
    odzt.mkdir()
    rb = oracle.execute(oracle.make_job(runs, str(obbt), **kw))
    rz = oracle.execute(oracle.make_job(runs, str(odzt),
                                        output_table_factory=1, **kw))
    bbt_stream = []
    for f in rb["files"]:
        bbt_stream += oracle.read_sst(open(f["path"], "rb").read())
    dzt_stream = []
    for f in rz["files"]:
        dzt_stream += oracle.dzt_read(open(f["path"], "rb").read())
    return bbt_stream, dzt_stream, rb, rz


def test_dzt_build_stream_equals_bbt(tmp_path):
    runs = _gen_runs(tmp_path)
    bbt, dzt, rb, rz = _job_streams(tmp_path, runs, compression=1,
                                    bottommost_level=1)
    assert rb["out_entries"] == rz["out_entries"]
    assert bbt == dzt  # same merged KV stream through either table format


def test_dzt_searchable_point_lookups(tmp_path):
    runs = _gen_runs(tmp_path, entries=5000)
    _, dzt, _, rz = _job_streams(tmp_path, runs, compression=1,
                                 bottommost_level=1)
    assert len(rz["files"]) == 1
    data = open(rz["files"][0]["path"], "rb").read()
    rnd = random.Random(3)
    for ik, v in rnd.sample(dzt, 50):
        uk = ik[:-8]
        got = oracle.dzt_get(data, uk)
        assert got is not None, uk
        assert got[0] == v
        assert got[1] == int.from_bytes(ik[-8:], "little")
    # absent keys
    for _ in range(20):
        uk = b"z%015d" % rnd.randrange(10**9)
        assert oracle.dzt_get(data, uk) is None


def test_dzt_newest_version_wins(tmp_path):
    # equal user keys under a snapshot: both versions survive; get()
    # returns the newest (file order)
    kvs = []
    for i in range(200):
        uk = b"k%015d" % i
        kvs.append((oracle.make_ikey(uk, 900 + i, 1), b"new-%d" % i))
        kvs.append((oracle.make_ikey(uk, 100 + i, 1), b"old-%d" % i))
    p = str(tmp_path / "two_ver.sst")
    with open(p, "wb") as f:
        f.write(oracle.build_sst(kvs))
    out = tmp_path / "out"
    out.mkdir()
    r = oracle.execute(oracle.make_job([[p]], str(out), snapshots=[500],
                                       output_table_factory=1))
    data = open(r["files"][0]["path"], "rb").read()
    stream = oracle.dzt_read(data)
    assert len(stream) == 400  # both versions kept (snapshot at 500)
    got = oracle.dzt_get(data, b"k%015d" % 7)
    assert got[0] == b"new-7"


def test_dzt_file_cuts(tmp_path):
    runs = _gen_runs(tmp_path, n_runs=2, entries=12000)
    bbt, dzt, _, rz = _job_streams(tmp_path, runs, compression=1,
                                   bottommost_level=1,
                                   target_file_size=256 * 1024)
    assert len(rz["files"]) > 1
    assert bbt == dzt
    # per-file metadata is consistent
    for f in rz["files"]:
        data = open(f["path"], "rb").read()
        assert f["file_size"] == len(data)
        stream = oracle.dzt_read(data)
        assert f["num_entries"] == len(stream)
        assert stream[0][0] == f["smallest"]
        assert stream[-1][0] == f["largest"]


def test_dzt_uncompressed_mode(tmp_path):
    runs = _gen_runs(tmp_path, entries=2000)
    bbt, dzt, _, _ = _job_streams(tmp_path, runs, compression=0,
                                  bottommost_level=1)
    assert bbt == dzt
