"""Oracle primitive KATs: crc32c (reference util/crc32c_test.cc:67-94 vectors),
XXH3 (golden vectors generated from the reference's vendored util/xxhash.h),
snappy codec, internal-key ordering (dbformat.h:1057-1096)."""
import json
import os
import random

import oracle as o

HERE = os.path.dirname(os.path.abspath(__file__))


def test_crc32c_rfc3720_kats():
    # transcribed from util/crc32c_test.cc TEST(CRC, StandardResults)
    assert o.crc32c(b"\x00" * 32) == 0x8A9136AA
    assert o.crc32c(b"\xff" * 32) == 0x62A8AB43
    assert o.crc32c(bytes(range(32))) == 0x46DD794E
    assert o.crc32c(bytes(reversed(range(32)))) == 0x113FDB5C
    iscsi = bytes([
        0x01, 0xC0, 0x00, 0x00, 0x00, 0x00, 0x00, 0x00, 0x00, 0x00, 0x00, 0x00,
        0x00, 0x00, 0x00, 0x00, 0x14, 0x00, 0x00, 0x00, 0x00, 0x00, 0x04, 0x00,
        0x00, 0x00, 0x00, 0x14, 0x00, 0x00, 0x00, 0x18, 0x28, 0x00, 0x00, 0x00,
        0x00, 0x00, 0x00, 0x00, 0x02, 0x00, 0x00, 0x00, 0x00, 0x00, 0x00, 0x00,
    ])
    assert o.crc32c(iscsi) == 0xD9963A56


def test_crc32c_mask():
    # Mask(crc) = ((crc >> 15) | (crc << 17)) + 0xa282ead8 (util/crc32c.h:44-46)
    data = b"hello world"
    crc = o.crc32c(data)
    masked = o.crc32c_masked(data)
    assert masked == (((crc >> 15) | (crc << 17)) + 0xA282EAD8) & 0xFFFFFFFF


def test_xxh3_golden_vectors():
    with open(os.path.join(HERE, "golden", "xxh3_vectors.json")) as f:
        g = json.load(f)
    buf = bytes(((i * 2654435761) >> 24) & 0xFF for i in range(1 << 22))
    for v in g["vectors"]:
        assert o.xxh3_64(buf[: v["len"]]) == int(v["xxh3_64"], 16), v


def test_block_checksum_xxh3_last_byte():
    # kXXH3: Lower32(XXH3(data)) ^ last*0x6b9083d9 (format.cc:433-439,497-503)
    data = b"block contents here" * 100
    want = (o.xxh3_64(data) & 0xFFFFFFFF) ^ ((1 * 0x6B9083D9) & 0xFFFFFFFF)
    assert o.block_checksum(4, data, 1) == want
    # kCRC32c path covers the last byte via Extend
    assert o.block_checksum(1, data, 0) != o.block_checksum(1, data, 1)


def test_snappy_roundtrip():
    random.seed(7)
    cases = [b"", b"a", b"abcabcabcabcabcabc" * 100,
             bytes(random.randrange(256) for _ in range(4096)),
             bytes(random.randrange(4) for _ in range(65536)),
             b"\x00" * 100000]
    for data in cases:
        c = o.snappy_compress(data)
        assert o.snappy_uncompress(c) == data


def test_snappy_decodes_reference_format():
    # hand-built snappy stream: varint len, literal, 1-byte-offset copy
    # "abcdabcdabcd": literal "abcd" + copy(offset=4, len=8)
    raw = b"abcdabcdabcd"
    stream = bytes([len(raw)]) + bytes([(4 - 1) << 2]) + b"abcd" + bytes(
        [1 | ((8 - 4) << 2) | ((4 >> 8) << 5), 4 & 0xFF])
    assert o.snappy_uncompress(stream) == raw


def test_ikey_ordering():
    # user key asc, tie -> seq desc, then type desc (tag as u64 desc)
    k = o.make_ikey
    assert o.ikey_compare(k(b"a", 5, 1), k(b"b", 5, 1)) < 0
    assert o.ikey_compare(k(b"a", 9, 1), k(b"a", 5, 1)) < 0  # higher seq first
    assert o.ikey_compare(k(b"a", 5, 1), k(b"a", 5, 0)) < 0  # higher type first
    assert o.ikey_compare(k(b"a", 5, 1), k(b"a", 5, 1)) == 0
    assert o.ikey_compare(k(b"a", 5, 1), k(b"ab", 9, 1)) < 0  # prefix first
    # 8-byte-word compare path: keys longer than 8 bytes
    assert o.ikey_compare(k(b"aaaaaaaaz", 1, 1), k(b"aaaaaaab", 1, 1)) < 0
