"""Bloom filter build path: XXPH3 pinned against the
reference's own util/xxph3.h (golden vectors), and the FastLocalBloom
filter block (full_filter_block.cc + util/bloom_impl.h:144-223) verified
through the oracle compaction path: every key may-match, absent keys
mostly don't."""
import json
import os
import struct
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import oracle

GOLDEN = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden",
                      "xxph3_vectors.json")


def test_xxph3_reference_vectors():
    doc = json.load(open(GOLDEN))
    x = 0x12345678
    buf = bytearray()
    for _ in range(256):
        x ^= (x << 13) & 0xffffffff
        x ^= x >> 17
        x ^= (x << 5) & 0xffffffff
        buf.append(x & 0xff)
    for v in doc["vectors"]:
        data = bytes(buf[:v["len"]])
        assert oracle.xxph3_64(data) == int(v["hash"], 16), v["len"]


def _bloom_may_match(filter_block: bytes, ukey: bytes) -> bool:
    # HashMayMatch (bloom_impl.h:225-243) restated for the test
    data, meta = filter_block[:-5], filter_block[-5:]
    assert meta[0] == 0xFF and meta[1] == 0
    probes = meta[2]
    h64 = oracle.xxph3_64(ukey)
    h1, h2 = h64 & 0xffffffff, h64 >> 32
    length = len(data)
    line = ((h1 * (length >> 6)) >> 32) << 6
    h = h2
    for _ in range(probes):
        bitpos = h >> (32 - 9)
        if not (data[line + (bitpos >> 3)] >> (bitpos & 7)) & 1:
            return False
        h = (h * 0x9e3779b9) & 0xffffffff
    return True


def _filter_block_of(sst: bytes) -> bytes:
    # footer -> metaindex -> "fullfilter.rocksdb.BuiltinBloomFilter"
    f = sst[-53:]
    assert f[41:45] == struct.pack("<I", 5)

    def get_varint(b, i):
        r, sh = 0, 0
        while True:
            x = b[i]
            i += 1
            r |= (x & 0x7F) << sh
            if not x & 0x80:
                return r, i
            sh += 7

    mi_off, i = get_varint(f, 1)
    mi_sz, i = get_varint(f, i)
    blk = sst[mi_off:mi_off + mi_sz]
    nres = struct.unpack("<I", blk[-4:])[0] & 0x7FFFFFFF
    end = len(blk) - 4 - 4 * nres
    p, key = 0, b""
    while p < end:
        sh, p = get_varint(blk, p)
        ns, p = get_varint(blk, p)
        vl, p = get_varint(blk, p)
        key = key[:sh] + blk[p:p + ns]
        p += ns
        if key == b"fullfilter.rocksdb.BuiltinBloomFilter":
            off, q = get_varint(blk, p)
            sz, q = get_varint(blk, q)
            return sst[off:off + sz]
        p += vl
    raise AssertionError("no filter block in metaindex")


def test_filter_through_compaction(tmp_path):
    kvs = [(oracle.make_ikey(b"k%015d" % i, 100 + i, 1), b"v%d" % i)
           for i in range(20000)]
    p = str(tmp_path / "in.sst")
    with open(p, "wb") as fh:
        fh.write(oracle.build_sst(kvs))
    out = tmp_path / "out"
    out.mkdir()
    r = oracle.execute(oracle.make_job([[p]], str(out), bottommost_level=1,
                                       compression=1,
                                       bloom_millibits_per_key=10000))
    sst = open(r["files"][0]["path"], "rb").read()
    fb = _filter_block_of(sst)
    # ~10 bits/key: len = round64(20000*10/8) + 5
    assert len(fb) == ((20000 * 10000 // 8000 + 63) & ~63) + 5
    for i in range(0, 20000, 97):
        assert _bloom_may_match(fb, b"k%015d" % i)
    fp = sum(_bloom_may_match(fb, b"z%015d" % i) for i in range(2000))
    assert fp < 2000 * 0.05  # ~1% expected at 10 bits/key


def test_filter_dedups_consecutive_versions(tmp_path):
    # equal user keys under a snapshot count once in num_filter_entries
    kvs = []
    for i in range(1000):
        uk = b"k%015d" % i
        kvs.append((oracle.make_ikey(uk, 900 + i, 1), b"new"))
        kvs.append((oracle.make_ikey(uk, 100 + i, 1), b"old"))
    p = str(tmp_path / "two.sst")
    with open(p, "wb") as fh:
        fh.write(oracle.build_sst(kvs))
    out = tmp_path / "out"
    out.mkdir()
    r = oracle.execute(oracle.make_job([[p]], str(out), snapshots=[500],
                                       bloom_millibits_per_key=10000))
    # old versions survive only under the snapshot (seq <= 500): 1000 new
    # + 401 old
    assert r["out_entries"] == 1401
    sst = open(r["files"][0]["path"], "rb").read()
    fb = _filter_block_of(sst)
    # 1000 distinct keys -> filter sized for 1000 entries, not 2000
    assert len(fb) == ((1000 * 10000 // 8000 + 63) & ~63) + 5
