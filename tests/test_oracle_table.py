"""Oracle SST builder/reader: round-trips, block cutting, footer layout
(table/format.cc:191-259), compression acceptance
(GoodCompressionRatio, block_based_table_builder.cc)."""
import os
import random
import struct

import oracle as o

MAGIC = 0x88E241B785F4CFF7


def entries(n, klen=12, vlen=20, seq0=1000, vtype=1):
    out = []
    for i in range(n):
        uk = b"k%0*d" % (klen - 1, i)
        out.append((o.make_ikey(uk, seq0 + i, vtype), bytes([i & 0xFF]) * vlen))
    return out


def test_roundtrip_small():
    es = entries(5)
    data = o.build_sst(es)
    assert o.read_sst(data) == es


def test_roundtrip_multiblock():
    es = entries(5000, klen=16, vlen=100)
    data = o.build_sst(es)
    assert o.read_sst(data) == es


def test_roundtrip_varied_key_lengths():
    random.seed(3)
    uks = sorted({bytes(random.randrange(97, 123) for _ in range(random.randrange(1, 40)))
                  for _ in range(2000)})
    es = [(o.make_ikey(uk, 10 + i, 1), b"v" * random.randrange(0, 200))
          for i, uk in enumerate(uks)]
    data = o.build_sst(es)
    assert o.read_sst(data) == es


def test_footer_layout():
    data = o.build_sst(entries(10))
    f = data[-53:]
    assert f[0] == 4  # kXXH3
    assert struct.unpack("<I", f[41:45])[0] == 5  # format_version
    assert struct.unpack("<Q", f[45:53])[0] == MAGIC


def test_block_cut_at_4096():
    # 4 KiB block_size: raw entries ~124 B -> ~33 per block; every data block's
    # unompressed payload estimate stays near 4096
    es = entries(1000, klen=16, vlen=100)
    data = o.build_sst(es)
    assert o.read_sst(data) == es
    # count data blocks via num-restarts trick is internal; just check size sane
    assert len(data) > 1000 * 110


def test_snappy_sst_roundtrip():
    opts = o.default_table_opts(compression=1)
    es = entries(3000, klen=16, vlen=100)
    data = o.build_sst(es, opts)
    nocomp = o.build_sst(es)
    assert len(data) < len(nocomp)  # values are constant-byte runs -> compress
    assert o.read_sst(data) == es


def test_incompressible_blocks_stored_raw():
    random.seed(11)
    es = []
    for i in range(500):
        uk = b"k%011d" % i
        es.append((o.make_ikey(uk, 1 + i, 1),
                   bytes(random.randrange(256) for _ in range(100))))
    opts = o.default_table_opts(compression=1)
    data = o.build_sst(es, opts)
    assert o.read_sst(data) == es


def test_checksum_detects_corruption():
    data = bytearray(o.build_sst(entries(100)))
    data[10] ^= 0xFF
    try:
        o.read_sst(bytes(data))
        # corruption may be in a data block -> iterate raises; in index -> open
        # raises. Either way we must not get identical entries back silently.
        raised = False
    except ValueError:
        raised = True
    assert raised


def test_crc32c_checksum_type():
    opts = o.default_table_opts(checksum_type=1)
    es = entries(200)
    data = o.build_sst(es, opts)
    assert data[-53] == 1
    assert o.read_sst(data) == es


def test_deterministic():
    es = entries(500)
    assert o.build_sst(es) == o.build_sst(es)
